"""Worker abstraction (reference: model_gateway/src/worker/worker.rs:233 `trait Worker`).

A worker is one engine endpoint: URL (http/grpc/rccl scheme), model id, worker type
(regular / prefill / decode / encode), live load counters, a circuit breaker, and
optional DP-rank topology.  Load accounting is plain ints mutated from the event
loop; the RCCL data plane reads them through the registry snapshot.
"""
from __future__ import annotations

import enum
import itertools
import time
from typing import Any, Dict, List, Optional

from ..config import CircuitBreakerConfig
from .circuit_breaker import CircuitBreaker

_worker_ids = itertools.count(1)


class WorkerType(str, enum.Enum):
    REGULAR = "regular"
    PREFILL = "prefill"
    DECODE = "decode"
    ENCODE = "encode"


class HealthState(str, enum.Enum):
    UNKNOWN = "unknown"
    HEALTHY = "healthy"
    UNHEALTHY = "unhealthy"
    DRAINING = "draining"


class Worker:
    """One engine endpoint and its gateway-side state."""

    __slots__ = (
        "worker_id",
        "url",
        "model_id",
        "model_aliases",
        "worker_type",
        "labels",
        "priority",
        "cost",
        "bootstrap_host",
        "bootstrap_port",
        "dp_size",
        "api_key",
        "circuit_breaker",
        "health",
        "_health_failures",
        "_health_successes",
        "active_requests",
        "queued_tokens",
        "inflight_tokens",
        "token_usage",
        "gen_throughput",
        "dp_loads",
        "processed_requests",
        "added_at",
        "sampling_defaults",
        "rccl_rank",
        "extra",
    )

    def __init__(
        self,
        url: str,
        model_id: str = "default",
        worker_type: WorkerType = WorkerType.REGULAR,
        labels: Optional[Dict[str, str]] = None,
        circuit_breaker_config: Optional[CircuitBreakerConfig] = None,
        bootstrap_host: Optional[str] = None,
        bootstrap_port: Optional[int] = None,
        dp_size: int = 0,
        priority: int = 0,
        cost: float = 1.0,
        api_key: Optional[str] = None,
        model_aliases: Optional[List[str]] = None,
        rccl_rank: Optional[int] = None,
    ):
        self.worker_id = next(_worker_ids)
        self.url = url.rstrip("/")
        self.model_id = model_id
        self.model_aliases = list(model_aliases or [])
        self.worker_type = worker_type
        self.labels = dict(labels or {})
        self.priority = priority
        self.cost = cost
        self.bootstrap_host = bootstrap_host
        self.bootstrap_port = bootstrap_port
        self.dp_size = dp_size
        self.api_key = api_key
        self.circuit_breaker = CircuitBreaker(circuit_breaker_config)
        self.health = HealthState.UNKNOWN
        self._health_failures = 0
        self._health_successes = 0
        # live load
        self.active_requests = 0
        self.queued_tokens = 0
        self.inflight_tokens = 0
        self.token_usage: Optional[float] = None  # engine KV utilization 0..1
        self.gen_throughput: Optional[float] = None  # tokens/s reported by engine
        self.dp_loads: List[int] = [0] * dp_size if dp_size else []
        self.processed_requests = 0
        self.added_at = time.time()
        self.sampling_defaults: Dict[str, Any] = {}
        self.rccl_rank = rccl_rank  # on-node xGMI data-plane rank, None = off-node
        self.extra: Dict[str, Any] = {}

    # ---- load accounting -------------------------------------------------
    def incr_load(self, tokens: int = 0) -> None:
        self.active_requests += 1
        self.inflight_tokens += tokens

    def decr_load(self, tokens: int = 0) -> None:
        self.active_requests = max(0, self.active_requests - 1)
        self.inflight_tokens = max(0, self.inflight_tokens - tokens)
        self.processed_requests += 1

    @property
    def load(self) -> int:
        return self.active_requests

    def is_available(self) -> bool:
        return self.health != HealthState.UNHEALTHY and self.circuit_breaker.can_execute()

    def record_outcome(self, success: bool) -> None:
        self.circuit_breaker.record_outcome(success)

    # ---- health gating (reference HealthCheckConfig thresholds) ----------
    def observe_health(self, ok: bool, failure_threshold: int, success_threshold: int) -> bool:
        """Returns True when the health state changed."""
        if ok:
            self._health_failures = 0
            self._health_successes += 1
            if self.health != HealthState.HEALTHY and self._health_successes >= success_threshold:
                self.health = HealthState.HEALTHY
                return True
        else:
            self._health_successes = 0
            self._health_failures += 1
            if self.health != HealthState.UNHEALTHY and self._health_failures >= failure_threshold:
                self.health = HealthState.UNHEALTHY
                return True
        return False

    def serves_model(self, model_id: Optional[str]) -> bool:
        return model_id is None or model_id == self.model_id or model_id in self.model_aliases

    def to_dict(self) -> Dict[str, Any]:
        return {
            "id": str(self.worker_id),
            "url": self.url,
            "model_id": self.model_id,
            "worker_type": self.worker_type.value,
            "health": self.health.value,
            "circuit_breaker": self.circuit_breaker.state.value,
            "labels": self.labels,
            "load": self.active_requests,
            "inflight_tokens": self.inflight_tokens,
            "token_usage": self.token_usage,
            "gen_throughput": self.gen_throughput,
            "dp_size": self.dp_size,
            "priority": self.priority,
            "cost": self.cost,
            "bootstrap_host": self.bootstrap_host,
            "bootstrap_port": self.bootstrap_port,
            "rccl_rank": self.rccl_rank,
            "processed_requests": self.processed_requests,
        }

    def __repr__(self) -> str:  # pragma: no cover
        return f"<Worker {self.worker_id} {self.worker_type.value} {self.url} model={self.model_id} load={self.active_requests}>"
