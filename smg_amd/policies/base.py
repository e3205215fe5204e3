"""Load-balancing policy interface (reference: model_gateway/src/policies/mod.rs:47
`trait LoadBalancingPolicy::select_worker`, `on_request_complete` mod.rs:62,
`update_loads` mod.rs:77; `trait DPRankLoadPolicy::select_dp_rank` mod.rs:101).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

from ..workers.worker import Worker


@dataclass
class SelectWorkerInfo:
    """Per-request routing context handed to a policy."""

    request_id: str = ""
    model_id: Optional[str] = None
    text: Optional[str] = None  # HTTP path: routing text (extracted prompt)
    tokens: Optional[Sequence[int]] = None  # gRPC/tokenized path
    routing_key: Optional[str] = None  # sticky X-SMG-Routing-Key
    tenant_id: Optional[str] = None
    est_tokens: int = 0
    headers: Dict[str, str] = field(default_factory=dict)


class LoadBalancingPolicy:
    """Selects one worker index from a candidate list.  Stateless policies are
    shared across models; stateful ones are instantiated per model by the
    PolicyRegistry (reference registry.rs)."""

    name = "base"

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        raise NotImplementedError

    def select_worker_pair(
        self, prefill: Sequence[Worker], decode: Sequence[Worker], info: SelectWorkerInfo
    ) -> Optional[tuple]:
        """PD disaggregation: pick (prefill_idx, decode_idx); default independent."""
        p = self.select_worker(prefill, info)
        d = self.select_worker(decode, info)
        if p is None or d is None:
            return None
        return p, d

    def on_request_complete(self, worker: Worker, info: SelectWorkerInfo, success: bool) -> None:
        """Called when a routed request finishes (reference mod.rs:62)."""

    def update_loads(self, workers: Sequence[Worker]) -> None:
        """Called by the load monitor after a GetLoads poll (reference mod.rs:77)."""

    def on_worker_removed(self, worker: Worker) -> None:
        pass

    def needs_tokens(self) -> bool:
        """True if the policy benefits from tokenized input (cache_aware/prefix_hash)."""
        return False

    def reset(self) -> None:
        pass


class DPRankLoadPolicy:
    """Selects a data-parallel rank inside one engine (reference mod.rs:101)."""

    def select_dp_rank(self, worker: Worker) -> Optional[int]:
        raise NotImplementedError


def filter_available(workers: Sequence[Worker]) -> List[int]:
    return [i for i, w in enumerate(workers) if w.is_available()]
