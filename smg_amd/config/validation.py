"""Config validation (reference: model_gateway/src/config/validation.rs, re-validated
at startup server.rs:992).  Raises ConfigError with a plain-language message."""
from __future__ import annotations

from .types import ConnectionMode, RouterConfig, RoutingMode


class ConfigError(ValueError):
    pass


def validate_config(cfg: RouterConfig) -> None:
    if not (0 < cfg.port < 65536):
        raise ConfigError(f"port {cfg.port} out of range")
    if cfg.health_check_port is not None and cfg.health_check_port == cfg.port:
        raise ConfigError("health_check_port must differ from port")
    if cfg.prometheus_port is not None and cfg.prometheus_port == cfg.port:
        raise ConfigError("prometheus_port must differ from port")

    cfg.policy.validate()
    for sub in (cfg.prefill_policy, cfg.decode_policy, cfg.encode_policy):
        if sub is not None:
            sub.validate()

    if cfg.mode == RoutingMode.PREFILL_DECODE and cfg.connection_mode != ConnectionMode.RCCL:
        # rccl PD derives its prefill/decode fleets from the torchrun ranks
        # (odd = prefill), not from URLs
        if not cfg.prefill_urls and not cfg.discovery.enabled:
            raise ConfigError("prefill_decode mode requires --prefill URLs or service discovery")
        if not cfg.decode_urls and not cfg.discovery.enabled:
            raise ConfigError("prefill_decode mode requires --decode URLs or service discovery")
    if cfg.mode == RoutingMode.ENCODE_PREFILL_DECODE and not cfg.encode_urls and not cfg.discovery.enabled:
        raise ConfigError("encode_prefill_decode mode requires --encode URLs or service discovery")

    for url in cfg.worker_urls + cfg.decode_urls:
        _validate_url(url)
    for url, _port in cfg.prefill_urls + cfg.encode_urls:
        _validate_url(url)

    if cfg.connection_mode == ConnectionMode.RCCL and cfg.rccl.world_size < 1:
        raise ConfigError("rccl connection mode requires world_size >= 1")

    if cfg.rate_limit.max_concurrent_requests == 0:
        raise ConfigError("max_concurrent_requests must be -1 (unlimited) or > 0")
    if cfg.retry.max_retries < 0:
        raise ConfigError("retry.max_retries must be >= 0")
    if not (0.0 <= cfg.retry.jitter_factor <= 1.0):
        raise ConfigError("retry.jitter_factor must be in [0,1]")
    if cfg.circuit_breaker.failure_threshold <= 0 or cfg.circuit_breaker.success_threshold <= 0:
        raise ConfigError("circuit breaker thresholds must be positive")
    if (cfg.tls_cert_path is None) != (cfg.tls_key_path is None):
        raise ConfigError("tls_cert_path and tls_key_path must be set together")
    if cfg.storage.backend not in ("memory", "none", "postgres", "redis", "oracle"):
        raise ConfigError(f"unknown storage backend {cfg.storage.backend!r}")


def _validate_url(url: str) -> None:
    if "://" not in url:
        raise ConfigError(f"worker url {url!r} must include a scheme (http:// or grpc://)")
    scheme = url.split("://", 1)[0]
    if scheme not in ("http", "https", "grpc", "grpcs", "rccl", "sim"):
        raise ConfigError(f"worker url {url!r} has unsupported scheme {scheme!r}")
