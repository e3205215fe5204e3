"""Config + CLI tests (model: reference main.rs colocated tests :1810+)."""
import pytest

from smg_amd.cli import to_router_config
from smg_amd.config import ConfigError, PolicyConfig, RouterConfig, RoutingMode


def test_default_config_validates():
    RouterConfig().validate()


def test_cli_defaults():
    cfg = to_router_config(["launch"])
    assert cfg.port == 30000
    assert cfg.policy.name == "cache_aware"
    assert cfg.policy.cache_threshold == 0.3
    assert cfg.policy.max_tree_size == 67_108_864


def test_cli_policy_flags():
    cfg = to_router_config(
        [
            "launch", "--policy", "least_load", "--worker-urls", "http://a:1", "http://b:2",
            "--least-load-kv-pressure-weight", "0.5", "--cache-threshold", "0.7",
        ]
    )
    assert cfg.policy.name == "least_load"
    assert cfg.policy.least_load_kv_pressure_weight == 0.5
    assert cfg.worker_urls == ["http://a:1", "http://b:2"]
    cfg.validate()


def test_cli_prefill_pairs():
    cfg = to_router_config(
        ["launch", "--prefill", "http://p1:800", "9000", "--prefill", "http://p2:800",
         "--decode", "http://d1:801", "--policy", "round_robin"]
    )
    assert cfg.mode == RoutingMode.PREFILL_DECODE
    assert cfg.prefill_urls == [("http://p1:800", 9000), ("http://p2:800", None)]
    assert cfg.decode_urls == ["http://d1:801"]
    cfg.validate()


def test_cli_auth_and_tenant_keys():
    cfg = to_router_config(["launch", "--api-key", "k1", "--tenant-api-key", "tk=acme"])
    assert cfg.auth.api_key == "k1"
    assert cfg.auth.tenant_api_keys == {"tk": "acme"}


def test_cli_model_alias():
    cfg = to_router_config(["launch", "--model-alias", "gpt-4=mock-model"])
    assert cfg.model_aliases == {"gpt-4": "mock-model"}


def test_validation_rejects_bad_urls():
    cfg = RouterConfig(worker_urls=["not-a-url"])
    with pytest.raises(ConfigError):
        cfg.validate()


def test_validation_rejects_bad_policy():
    with pytest.raises(ValueError):
        PolicyConfig(name="nope").validate()


def test_validation_pd_requires_workers():
    cfg = RouterConfig(mode=RoutingMode.PREFILL_DECODE)
    with pytest.raises(ConfigError):
        cfg.validate()


def test_validation_block_size_power_of_two():
    with pytest.raises(ValueError):
        PolicyConfig(block_size=13).validate()
