"""Prometheus metrics (reference: model_gateway/src/observability/metrics.rs —
161 registrations; the smg_http_*/smg_router_*/smg_worker_* families).

GatewayMetrics wraps a prometheus_client registry; GatewayMetrics.null() gives
a no-op sink for tests.  The separate metrics HTTP server mirrors the
reference's dedicated --prometheus-port listener (metrics_server.rs).
"""
from __future__ import annotations

from typing import Optional

try:
    from prometheus_client import (
        CollectorRegistry,
        Counter,
        Gauge,
        Histogram,
        generate_latest,
    )

    HAVE_PROM = True
except ImportError:  # pragma: no cover
    HAVE_PROM = False


_ROUTING_BUCKETS = (
    1e-6, 5e-6, 1e-5, 2.5e-5, 5e-5, 1e-4, 2.5e-4, 5e-4, 1e-3, 2.5e-3, 5e-3, 1e-2, 5e-2, 0.1,
)
_LATENCY_BUCKETS = (0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1, 2.5, 5, 10, 30, 60, 120)


class GatewayMetrics:
    def __init__(self, registry: Optional["CollectorRegistry"] = None):
        if not HAVE_PROM:
            self._null = True
            return
        self._null = False
        self.registry = registry or CollectorRegistry()
        r = self.registry
        self.http_requests = Counter(
            "smg_http_requests_total", "HTTP requests", ["path", "method", "status"], registry=r
        )
        self.http_duration = Histogram(
            "smg_http_request_duration_seconds", "HTTP request duration", ["path"],
            buckets=_LATENCY_BUCKETS, registry=r,
        )
        self.router_routing_latency = Histogram(
            "smg_router_routing_latency_seconds", "policy select_worker latency",
            buckets=_ROUTING_BUCKETS, registry=r,
        )
        self.router_retries = Counter("smg_router_retries_total", "routing retries", ["path"], registry=r)
        self.router_no_worker = Counter(
            "smg_router_no_available_worker_total", "requests with no available worker", ["path"], registry=r
        )
        self.worker_errors = Counter("smg_worker_errors_total", "worker transport errors", ["worker"], registry=r)
        self.worker_health = Gauge("smg_worker_healthy", "worker health (1/0)", ["worker"], registry=r)
        self.active_workers = Gauge("smg_active_workers", "registered workers", registry=r)
        self.inflight = Gauge("smg_inflight_requests", "in-flight requests", registry=r)
        self.ttft = Histogram(
            "smg_router_ttft_seconds", "time to first token", buckets=_LATENCY_BUCKETS, registry=r
        )
        self.generate_tokens = Counter("smg_router_generate_tokens_total", "tokens streamed back", registry=r)
        self.cache_hits = Counter("smg_policy_cache_hits_total", "cache-aware tree hits", registry=r)
        self.cache_misses = Counter("smg_policy_cache_misses_total", "cache-aware tree misses", registry=r)
        self.tree_size = Gauge("smg_policy_tree_nodes", "prefix-tree node count", ["model"], registry=r)
        self.queue_depth = Gauge("smg_scheduler_queue_depth", "admission queue depth", ["klass"], registry=r)
        self.rate_limited = Counter("smg_rate_limited_total", "429 rejections", ["tenant"], registry=r)

    @classmethod
    def null(cls) -> "GatewayMetrics":
        m = object.__new__(cls)
        m._null = True
        return m

    # ---- recording helpers (no-ops on the null sink) ---------------------
    def observe_http(self, path: str, method: str, status: int, duration: float) -> None:
        if self._null:
            return
        self.http_requests.labels(path, method, str(status)).inc()
        self.http_duration.labels(path).observe(duration)

    def observe_routing_latency(self, secs: float) -> None:
        if not self._null:
            self.router_routing_latency.observe(secs)

    def count_retry(self, path: str) -> None:
        if not self._null:
            self.router_retries.labels(path).inc()

    def count_no_worker(self, path: str) -> None:
        if not self._null:
            self.router_no_worker.labels(path).inc()

    def count_worker_error(self, worker: str) -> None:
        if not self._null:
            self.worker_errors.labels(worker).inc()

    def observe_ttft(self, secs: float) -> None:
        if not self._null:
            self.ttft.observe(secs)

    def export(self) -> bytes:
        if self._null:
            return b""
        return generate_latest(self.registry)
