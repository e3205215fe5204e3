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
        # ---- extended families (reference observability/metrics.rs groups) --
        self.tpot = Histogram(
            "smg_router_tpot_seconds", "time per output token", buckets=_ROUTING_BUCKETS, registry=r
        )
        self.request_input_tokens = Histogram(
            "smg_router_input_tokens", "prompt tokens per request",
            buckets=(16, 64, 256, 1024, 4096, 16384, 65536), registry=r,
        )
        self.request_output_tokens = Histogram(
            "smg_router_output_tokens", "output tokens per request",
            buckets=(1, 8, 32, 128, 512, 2048, 8192), registry=r,
        )
        # PD disaggregation (metrics.rs:207-227)
        self.pd_prefill_duration = Histogram(
            "smg_pd_prefill_duration_seconds", "prefill-leg duration", buckets=_LATENCY_BUCKETS, registry=r
        )
        self.pd_bootstrap_failures = Counter(
            "smg_pd_bootstrap_failures_total", "PD bootstrap injection failures", registry=r
        )
        self.pd_dual_dispatch = Counter(
            "smg_pd_dual_dispatch_total", "prefill+decode dual dispatches", registry=r
        )
        # circuit breaker (metrics.rs:272-288)
        self.cb_state = Gauge(
            "smg_worker_circuit_breaker_state", "0=closed 1=half-open 2=open", ["worker"], registry=r
        )
        self.cb_transitions = Counter(
            "smg_worker_circuit_breaker_transitions_total", "state transitions", ["worker", "to"], registry=r
        )
        # scheduler (middleware/scheduler)
        self.scheduler_admitted = Counter(
            "smg_scheduler_admitted_total", "requests admitted", ["klass"], registry=r
        )
        self.scheduler_preempted = Counter(
            "smg_scheduler_preempted_total", "requests preempted", ["klass"], registry=r
        )
        self.scheduler_timeout = Counter(
            "smg_scheduler_queue_timeout_total", "admission queue timeouts", ["klass"], registry=r
        )
        # mesh (mesh/src/metrics.rs)
        self.mesh_gossip_rounds = Counter("smg_mesh_gossip_rounds_total", "gossip rounds", registry=r)
        self.mesh_ops_applied = Counter("smg_mesh_ops_applied_total", "CRDT ops applied", ["ns"], registry=r)
        self.mesh_peers = Gauge("smg_mesh_alive_peers", "alive mesh peers", registry=r)
        # tokenizer caches (tokenizer/src/cache)
        self.tokenizer_l0_hits = Counter("smg_tokenizer_l0_hits_total", "L0 exact-match hits", registry=r)
        self.tokenizer_l1_hits = Counter("smg_tokenizer_l1_hits_total", "L1 prefix hits", registry=r)
        # MCP / plugins
        self.mcp_tool_calls = Counter("smg_mcp_tool_calls_total", "MCP tool invocations", ["server"], registry=r)
        self.plugin_short_circuits = Counter(
            "smg_plugin_short_circuits_total", "plugin OnRequest short-circuits", registry=r
        )
        # KV-event index (worker/kv_event_monitor.rs)
        self.kv_events_applied = Counter(
            "smg_kv_events_applied_total", "KV cache events applied", ["kind"], registry=r
        )
        # runtime self-observability (observability/runtime_metrics.rs canary)
        self.event_loop_lag = Histogram(
            "smg_event_loop_lag_seconds", "asyncio scheduling lag (canary)",
            buckets=(0.0005, 0.001, 0.005, 0.01, 0.05, 0.1, 0.5), registry=r,
        )
        # inflight age buckets (inflight_tracker.rs:22)
        self.inflight_age = Gauge(
            "smg_inflight_request_age_bucket", "in-flight requests older than the bucket", ["ge_seconds"],
            registry=r,
        )

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
