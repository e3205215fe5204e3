"""AppContext — the DI container (reference: model_gateway/src/app_context.rs:52-85).

Holds every shared subsystem: config, worker registry, policy registry, router
manager, monitor, metrics, rate limiter, tokenizer registry, parsers, storages.
Constructed once at startup; handlers reach it through the aiohttp app.
"""
from __future__ import annotations

from typing import Optional

from ..config import RouterConfig
from ..observability.metrics import GatewayMetrics
from ..policies import PolicyRegistry
from ..workers.monitor import WorkerMonitor
from ..workers.registry import WorkerRegistry
from ..workers.worker import Worker, WorkerType


class AppContext:
    def __init__(self, config: RouterConfig, metrics: Optional[GatewayMetrics] = None):
        from ..config import ConnectionMode
        from ..kvindex.event_index import PositionalIndexer

        self.config = config
        self.metrics = metrics or GatewayMetrics()
        self.worker_registry = WorkerRegistry()
        indexer = None
        if config.connection_mode == ConnectionMode.GRPC:
            indexer = PositionalIndexer(block_size=config.policy.block_size)
            if config.policy.gpu_tree:
                try:
                    from ..kvindex import gpu_available
                    from ..kvindex.gpu_event_index import GpuPositionalIndexer

                    if gpu_available():
                        indexer = GpuPositionalIndexer(block_size=config.policy.block_size)
                except ImportError:
                    pass
        self.policy_registry = PolicyRegistry(
            config.policy,
            indexer=indexer,
            prefill_cfg=config.prefill_policy,
            decode_cfg=config.decode_policy,
            encode_cfg=config.encode_policy,
        )
        self.worker_monitor = WorkerMonitor(
            self.worker_registry,
            config.health_check,
            load_interval_secs=config.load_monitor_interval,
            policy_registry=self.policy_registry,
        )
        self.router_manager = None  # wired by startup()
        # distributed tracing (reference otel_trace.rs): spans + optional
        # wire-level OTLP/HTTP export when --otlp-traces-endpoint is set
        from ..observability.tracing import OtlpHttpExporter, Tracer

        exporter = (
            OtlpHttpExporter(config.trace.otlp_endpoint)
            if (config.trace.enabled and config.trace.otlp_endpoint) else None
        )
        self.tracer = Tracer(enabled=config.trace.enabled, otlp_exporter=exporter)
        self.tokenizer_registry = None
        self.rate_limiter = None
        self.scheduler = None
        self.mesh = None
        self.storage = None
        self.mcp = None
        self.plugins = None
        self.kv_event_monitor = None
        self.inflight = 0
        self._background: list = []

    # ---- worker bootstrap (reference Job::InitializeWorkersFromConfig) ----
    def init_workers_from_config(self) -> None:
        cfg = self.config
        for url in cfg.worker_urls:
            self.worker_registry.register(
                Worker(
                    url,
                    model_id=cfg.model_path or "default",
                    worker_type=WorkerType.REGULAR,
                    circuit_breaker_config=cfg.circuit_breaker,
                )
            )
        for url, bport in cfg.prefill_urls:
            self.worker_registry.register(
                Worker(
                    url,
                    model_id=cfg.model_path or "default",
                    worker_type=WorkerType.PREFILL,
                    bootstrap_port=bport,
                    circuit_breaker_config=cfg.circuit_breaker,
                )
            )
        for url in cfg.decode_urls:
            self.worker_registry.register(
                Worker(
                    url,
                    model_id=cfg.model_path or "default",
                    worker_type=WorkerType.DECODE,
                    circuit_breaker_config=cfg.circuit_breaker,
                )
            )
        for url, port in cfg.encode_urls:
            self.worker_registry.register(
                Worker(
                    url,
                    model_id=cfg.model_path or "default",
                    worker_type=WorkerType.ENCODE,
                    bootstrap_port=port,
                    circuit_breaker_config=cfg.circuit_breaker,
                )
            )
        self.worker_registry.subscribe(lambda kind, w: self.policy_registry.on_worker_removed(w) if kind == "remove" else None)

    async def start_background(self) -> None:
        await self.worker_monitor.start()
        if self.policy_registry.indexer is not None:
            from ..workers.kv_event_monitor import KvEventMonitor

            self.kv_event_monitor = KvEventMonitor(self.worker_registry, self.policy_registry.indexer)
            await self.kv_event_monitor.start()

    async def shutdown(self) -> None:
        await self.worker_monitor.stop()
        if self.scheduler is not None and hasattr(self.scheduler, "stop"):
            await self.scheduler.stop()
        if self.kv_event_monitor is not None:
            await self.kv_event_monitor.stop()
        if self.mesh is not None:
            await self.mesh.stop()
            runner = getattr(self, "_mesh_runner", None)
            if runner is not None:
                await runner.cleanup()
        for t in self._background:
            t.cancel()
        if self.router_manager is not None:
            await self.router_manager.shutdown()
