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
        # ---- full reference family coverage (observability/metrics.rs) -----
        # router core
        self.router_requests = Counter(
            "smg_router_requests_total", "router requests", ["endpoint", "router_type"], registry=r)
        self.router_request_errors = Counter(
            "smg_router_request_errors_total", "router request errors", ["endpoint", "error_type"], registry=r)
        self.router_request_duration = Histogram(
            "smg_router_request_duration_seconds", "full request duration", ["endpoint"],
            buckets=_LATENCY_BUCKETS, registry=r)
        self.router_generation_duration = Histogram(
            "smg_router_generation_duration_seconds", "generation (stream) duration",
            buckets=_LATENCY_BUCKETS, registry=r)
        self.router_stage_duration = Histogram(
            "smg_router_stage_duration_seconds", "gRPC pipeline stage duration", ["stage"],
            buckets=_ROUTING_BUCKETS + (0.5, 1.0), registry=r)
        self.router_tokens = Counter(
            "smg_router_tokens_total", "tokens by direction", ["token_type"], registry=r)
        self.router_upstream_responses = Counter(
            "smg_router_upstream_responses_total", "upstream status codes", ["status_code"], registry=r)
        self.http_responses = Counter(
            "smg_http_responses_total", "HTTP responses", ["status_code"], registry=r)
        self.http_connections_active = Gauge(
            "smg_http_connections_active", "open client connections", registry=r)
        self.http_rate_limit = Counter(
            "smg_http_rate_limit_total", "admission outcomes", ["outcome"], registry=r)
        # worker layer (metrics.rs smg_worker_*)
        self.worker_pool_size = Gauge("smg_worker_pool_size", "workers per model", ["model"], registry=r)
        self.worker_requests_active = Gauge(
            "smg_worker_requests_active", "active requests per worker", ["worker"], registry=r)
        self.worker_connections_active = Gauge(
            "smg_worker_connections_active", "open connections per worker", ["worker"], registry=r)
        self.worker_health_checks = Counter(
            "smg_worker_health_checks_total", "health checks", ["worker", "outcome"], registry=r)
        self.worker_selection = Counter(
            "smg_worker_selection_total", "policy selections", ["policy", "outcome"], registry=r)
        self.worker_retries_exhausted = Counter(
            "smg_worker_retries_exhausted_total", "requests failing all retries", registry=r)
        self.worker_retry_backoff = Histogram(
            "smg_worker_retry_backoff_seconds", "retry backoff slept",
            buckets=(0.01, 0.05, 0.1, 0.5, 1, 5), registry=r)
        self.worker_routing_keys_active = Gauge(
            "smg_worker_routing_keys_active", "sticky routing keys", registry=r)
        self.cb_outcomes = Counter(
            "smg_worker_cb_outcomes_total", "circuit-breaker recorded outcomes", ["worker", "outcome"],
            registry=r)
        self.cb_consecutive_failures = Gauge(
            "smg_worker_cb_consecutive_failures", "current failure streak", ["worker"], registry=r)
        self.cb_consecutive_successes = Gauge(
            "smg_worker_cb_consecutive_successes", "current success streak", ["worker"], registry=r)
        # engine GetLoads re-export (metrics.rs smg_engine_*, --engine-metrics)
        self.engine_running_requests = Gauge(
            "smg_engine_running_requests", "engine running requests", ["worker"], registry=r)
        self.engine_waiting_requests = Gauge(
            "smg_engine_waiting_requests", "engine waiting requests", ["worker"], registry=r)
        self.engine_token_usage = Gauge(
            "smg_engine_token_usage", "engine KV utilization 0..1", ["worker"], registry=r)
        self.engine_cache_hit_rate = Gauge(
            "smg_engine_cache_hit_rate", "engine prefix-cache hit rate", ["worker"], registry=r)
        self.engine_gen_throughput = Gauge(
            "smg_engine_gen_throughput", "engine tokens/s", ["worker"], registry=r)
        self.engine_pd_prefill_queue = Gauge(
            "smg_engine_pd_prefill_queue_reqs", "PD prefill queue depth", ["worker"], registry=r)
        self.engine_pd_decode_queue = Gauge(
            "smg_engine_pd_decode_queue_reqs", "PD decode queue depth", ["worker"], registry=r)
        self.engine_pd_kv_transfer_speed = Gauge(
            "smg_engine_pd_kv_transfer_speed_gb_s", "PD KV transfer speed", ["worker"], registry=r)
        self.engine_pd_kv_transfer_latency = Gauge(
            "smg_engine_pd_kv_transfer_latency_ms", "PD KV transfer latency", ["worker"], registry=r)
        # PD transfer (metrics.rs:207-227 remainder)
        self.pd_kv_transfer_duration = Histogram(
            "smg_pd_kv_transfer_duration_seconds", "KV handoff duration",
            buckets=_LATENCY_BUCKETS, registry=r)
        self.pd_kv_transfer_failures = Counter(
            "smg_pd_kv_transfer_failures_total", "KV handoff failures", registry=r)
        self.pd_kv_connector_mode = Counter(
            "smg_pd_kv_connector_mode_total", "KV connector chosen", ["connection_mode"], registry=r)
        self.pd_ttft = Histogram(
            "smg_pd_ttft_seconds", "PD time to first token", buckets=_LATENCY_BUCKETS, registry=r)
        # policy branch counters (metrics.rs per-policy)
        self.policy_branch = Counter(
            "smg_policy_branch_total", "policy decision branches", ["policy", "branch"], registry=r)
        self.manual_policy_cache_entries = Gauge(
            "smg_manual_policy_cache_entries", "manual policy sticky entries", registry=r)
        # multimodal transport (metrics.rs smg_mm_*)
        self.mm_tensors = Counter(
            "smg_mm_tensors_total", "multimodal tensors moved", ["transport"], registry=r)
        self.mm_tensor_bytes = Counter(
            "smg_mm_tensor_bytes_total", "multimodal tensor bytes", ["transport"], registry=r)
        self.mm_shm_write_failures = Counter(
            "smg_mm_shm_write_failures_total", "SHM write failures", registry=r)
        # storage (metrics.rs smg_db_*)
        self.db_operations = Counter(
            "smg_db_operations_total", "storage operations", ["storage_type", "operation"], registry=r)
        self.db_operation_duration = Histogram(
            "smg_db_operation_duration_seconds", "storage op duration", ["storage_type"],
            buckets=_LATENCY_BUCKETS, registry=r)
        self.db_items_stored = Gauge(
            "smg_db_items_stored", "stored items", ["storage_type"], registry=r)
        self.db_connections_active = Gauge(
            "smg_db_connections_active", "open storage connections", ["storage_type"], registry=r)
        # discovery (metrics.rs smg_discovery_*)
        self.discovery_registrations = Counter(
            "smg_discovery_registrations_total", "discovered worker adds", registry=r)
        self.discovery_deregistrations = Counter(
            "smg_discovery_deregistrations_total", "discovered worker removals", registry=r)
        self.discovery_workers = Gauge(
            "smg_discovery_workers_discovered", "currently discovered workers", registry=r)
        self.discovery_sync_duration = Histogram(
            "smg_discovery_sync_duration_seconds", "discovery sync pass duration",
            buckets=_LATENCY_BUCKETS, registry=r)
        # MCP depth (metrics.rs smg_mcp_*)
        self.mcp_servers_active = Gauge("smg_mcp_servers_active", "connected MCP servers", registry=r)
        self.mcp_tool_duration = Histogram(
            "smg_mcp_tool_duration_seconds", "MCP tool call duration", ["tool_name"],
            buckets=_LATENCY_BUCKETS, registry=r)
        self.mcp_tool_iterations = Counter(
            "smg_mcp_tool_iterations_total", "tool-loop iterations", registry=r)
        # KV events
        self.kv_event_subscription_failures = Counter(
            "smg_kv_event_subscription_failures_total", "KV event stream failures", ["worker"],
            registry=r)
        # ---- MI355X-native subsystems (no reference equivalent) ------------
        # RCCL/xGMI serving plane (comm/plane.py + routers/rccl_router.py)
        self.plane_ticks = Counter("smg_plane_ticks_total", "lockstep plane ticks", registry=r)
        self.plane_requests_shipped = Counter(
            "smg_plane_requests_shipped_total", "requests shipped over xGMI", ["rank"], registry=r)
        self.plane_events_received = Counter(
            "smg_plane_events_received_total", "token events received over xGMI", registry=r)
        self.plane_tick_duration = Histogram(
            "smg_plane_tick_duration_seconds", "gateway tick duration",
            buckets=_ROUTING_BUCKETS + (0.5, 1.0), registry=r)
        self.plane_phase_seconds = Counter(
            "smg_plane_phase_seconds_total", "cumulative tick phase time", ["phase"], registry=r)
        # GPU radix tree (csrc/gpu_tree.hip)
        self.gpu_tree_nodes_live = Gauge("smg_gpu_tree_nodes_live", "tenanted device nodes", ["model"], registry=r)
        self.gpu_tree_nodes_allocated = Gauge(
            "smg_gpu_tree_nodes_allocated", "bump-allocated device nodes", ["model"], registry=r)
        self.gpu_tree_nodes_reclaimed = Counter(
            "smg_gpu_tree_nodes_reclaimed_total", "nodes recycled to the free list", ["model"], registry=r)
        self.gpu_tree_batch_size = Histogram(
            "smg_gpu_tree_batch_size", "requests per match/insert kernel launch",
            buckets=(1, 4, 16, 64, 256, 1024, 4096), registry=r)
        # mesh hardening (partition.rs equivalents)
        self.mesh_partitioned = Gauge("smg_mesh_partitioned", "1 while partitioned", registry=r)
        self.mesh_partition_heals = Counter("smg_mesh_partition_heals_total", "partitions healed", registry=r)
        self.mesh_repair_pages = Counter("smg_mesh_repair_pages_total", "repair pages served", registry=r)
        self.mesh_repairs_completed = Counter(
            "smg_mesh_repairs_completed_total", "peer repairs completed", registry=r)
        # local GPU engine (engine/torch_engine.py)
        self.engine_prefix_cache_hits = Counter(
            "smg_engine_prefix_cache_hits_total", "engine prefix-KV restores", registry=r)
        self.engine_decode_steps = Counter(
            "smg_engine_decode_steps_total", "fused decode iterations", registry=r)
        self.engine_prefill_tokens = Counter(
            "smg_engine_prefill_tokens_total", "prompt tokens prefilled", registry=r)
        # tokenizer families beyond the caches
        self.tokenize_duration = Histogram(
            "smg_tokenize_duration_seconds", "encode latency", buckets=_ROUTING_BUCKETS, registry=r)
        self.tokenizers_registered = Gauge("smg_tokenizers_registered", "registered tokenizers", registry=r)
        # auth/audit
        self.auth_failures = Counter("smg_auth_failures_total", "auth rejections", ["reason"], registry=r)
        self.admin_denied = Counter("smg_admin_denied_total", "non-admin control-plane attempts", registry=r)

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
