from .base import DPRankLoadPolicy, LoadBalancingPolicy, SelectWorkerInfo
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
from .registry import PolicyRegistry, create_policy

__all__ = [
    "BucketPolicy",
    "CacheAwarePolicy",
    "ConsistentHashingPolicy",
    "DPRankLoadPolicy",
    "LeastLoadPolicy",
    "LoadBalancingPolicy",
    "ManualPolicy",
    "MinimumTokensPolicy",
    "PassthroughPolicy",
    "PolicyRegistry",
    "PowerOfTwoPolicy",
    "PrefixHashPolicy",
    "RandomPolicy",
    "RoundRobinPolicy",
    "SelectWorkerInfo",
    "create_policy",
]
