"""PolicyRegistry: per-model policy instances + the sticky routing-key override
layer (reference: model_gateway/src/policies/registry.rs, 835 LoC;
routing_key_override layer registered in main.rs:190)."""
from __future__ import annotations

from typing import Dict, Optional

from ..config import PolicyConfig
from .base import LoadBalancingPolicy
from .cache_aware import CacheAwarePolicy
from .classic import (
    BucketPolicy,
    ConsistentHashingPolicy,
    LeastLoadPolicy,
    ManualPolicy,
    MinimumTokensPolicy,
    PassthroughPolicy,
    PowerOfTwoPolicy,
    PrefixHashPolicy,
    RandomPolicy,
    RoundRobinPolicy,
)


def create_policy(cfg: PolicyConfig, indexer=None, seed: Optional[int] = None) -> LoadBalancingPolicy:
    name = cfg.name
    if name == "random":
        return RandomPolicy(seed)
    if name == "round_robin":
        return RoundRobinPolicy()
    if name == "passthrough":
        return PassthroughPolicy()
    if name == "power_of_two":
        return PowerOfTwoPolicy(seed)
    if name == "least_load":
        return LeastLoadPolicy(cfg)
    if name == "prefix_hash":
        return PrefixHashPolicy(cfg)
    if name == "consistent_hashing":
        return ConsistentHashingPolicy()
    if name == "bucket":
        return BucketPolicy(cfg)
    if name == "manual":
        return ManualPolicy(cfg, seed)
    if name == "cache_aware":
        return CacheAwarePolicy(cfg, indexer=indexer)
    raise ValueError(f"unknown policy {name!r}")


class PolicyRegistry:
    """One policy instance per (model, role); stateful policies are not shared
    across models.  Roles: regular / prefill / decode / encode."""

    def __init__(
        self,
        default_cfg: PolicyConfig,
        indexer=None,
        prefill_cfg: Optional[PolicyConfig] = None,
        decode_cfg: Optional[PolicyConfig] = None,
        encode_cfg: Optional[PolicyConfig] = None,
        seed: Optional[int] = None,
    ):
        self.default_cfg = default_cfg
        self.role_cfgs: Dict[str, PolicyConfig] = {
            "regular": default_cfg,
            "prefill": prefill_cfg or default_cfg,
            "decode": decode_cfg or default_cfg,
            "encode": encode_cfg or default_cfg,
        }
        self.indexer = indexer
        self._seed = seed
        self._policies: Dict[tuple, LoadBalancingPolicy] = {}
        self._model_overrides: Dict[str, PolicyConfig] = {}
        self.dp_policy = MinimumTokensPolicy()

    def set_model_policy(self, model_id: str, cfg: PolicyConfig) -> None:
        self._model_overrides[model_id] = cfg
        for key in [k for k in self._policies if k[0] == model_id]:
            del self._policies[key]

    def get(self, model_id: Optional[str] = None, role: str = "regular") -> LoadBalancingPolicy:
        model_id = model_id or "default"
        key = (model_id, role)
        policy = self._policies.get(key)
        if policy is None:
            cfg = self._model_overrides.get(model_id, self.role_cfgs.get(role, self.default_cfg))
            policy = create_policy(cfg, indexer=self.indexer, seed=self._seed)
            self._policies[key] = policy
        return policy

    def on_worker_removed(self, worker) -> None:
        for p in self._policies.values():
            p.on_worker_removed(worker)

    def all_policies(self):
        return list(self._policies.values())
