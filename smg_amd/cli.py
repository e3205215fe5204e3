"""`smg launch` CLI (reference: model_gateway/src/main.rs — Cli :103,
CliArgs :161, ~130 flags in 26 help groups, to_router_config :1296; the
--prefill/--encode URL [bootstrap_port] positional pairs parse :1692-1711).

Flag names match the reference so launch scripts port unchanged.
"""
from __future__ import annotations

import argparse
import asyncio
import sys
from typing import List, Optional, Tuple

from .config import (
    ConnectionMode,
    PolicyConfig,
    RouterConfig,
    RoutingMode,
)


def _extract_url_port_pairs(argv: List[str], flag: str) -> Tuple[List[Tuple[str, Optional[int]]], List[str]]:
    """--prefill URL [bootstrap_port] may repeat (reference main.rs:1692-1711)."""
    out: List[Tuple[str, Optional[int]]] = []
    rest: List[str] = []
    i = 0
    while i < len(argv):
        if argv[i] == flag and i + 1 < len(argv):
            url = argv[i + 1]
            port: Optional[int] = None
            skip = 2
            if i + 2 < len(argv) and not argv[i + 2].startswith("--"):
                try:
                    port = int(argv[i + 2])
                    skip = 3
                except ValueError:
                    pass
            out.append((url, port))
            i += skip
        else:
            rest.append(argv[i])
            i += 1
    return out, rest


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="smg",
        description="smg — MI355X-native model-routing gateway (SMG-compatible CLI)",
    )
    p.add_argument("command", nargs="?", default="launch", choices=["launch", "start", "serve"])
    g = p.add_argument_group("Network")
    g.add_argument("--host", default="0.0.0.0")
    g.add_argument("--port", type=int, default=30000)
    g.add_argument("--health-check-port", type=int, default=None)

    g = p.add_argument_group("Worker Configuration")
    g.add_argument("--worker-urls", nargs="*", default=[])
    g.add_argument("--decode", action="append", default=[], dest="decode_urls")
    g.add_argument("--worker-startup-timeout-secs", type=int, default=600)
    g.add_argument("--worker-startup-delay", type=float, default=0.0)
    g.add_argument("--worker-startup-check-interval", type=int, default=30)
    g.add_argument("--load-monitor-interval", type=int, default=5)
    g.add_argument("--engine-metrics", action="store_true")

    g = p.add_argument_group("Routing Policy")
    g.add_argument("--policy", default="cache_aware", choices=[
        "random", "round_robin", "passthrough", "cache_aware", "power_of_two",
        "least_load", "prefix_hash", "consistent_hashing", "manual", "bucket"])
    g.add_argument("--cache-threshold", type=float, default=0.3)
    g.add_argument("--balance-abs-threshold", type=int, default=64)
    g.add_argument("--balance-rel-threshold", type=float, default=1.5)
    g.add_argument("--balance-token-usage-threshold", type=float, default=1.0)
    g.add_argument("--overload-token-usage-threshold", type=float, default=1.0)
    g.add_argument("--eviction-interval", type=int, default=120)
    g.add_argument("--max-tree-size", type=int, default=67_108_864)
    g.add_argument("--block-size", type=int, default=16)
    g.add_argument("--max-idle-secs", type=int, default=14_400)
    g.add_argument("--assignment-mode", default="random", choices=["random", "min_load", "min_group"])
    g.add_argument("--prefix-token-count", type=int, default=256)
    g.add_argument("--prefix-hash-load-factor", type=float, default=1.25)
    g.add_argument("--least-load-kv-pressure-weight", type=float, default=0.15)
    g.add_argument("--least-load-default-throughput", type=float, default=2000.0)
    g.add_argument("--least-load-mean-prefill-tokens", type=int, default=1024)
    g.add_argument("--dp-aware", action="store_true")
    g.add_argument("--routing-key-override", default=None)
    g.add_argument("--enable-igw", action="store_true")
    g.add_argument("--dp-minimum-tokens-scheduler", action="store_true")
    g.add_argument("--no-gpu-tree", action="store_true", help="MI355X: disable the GPU-resident prefix tree")

    g = p.add_argument_group("PD Disaggregation")
    g.add_argument("--pd-disaggregation", action="store_true")
    g.add_argument("--epd-disaggregation", action="store_true")
    g.add_argument("--prefill-policy", default=None)
    g.add_argument("--decode-policy", default=None)
    g.add_argument("--encode-policy", default=None)

    g = p.add_argument_group("Connection")
    g.add_argument("--connection-mode", default="http", choices=["http", "grpc", "rccl"])
    g.add_argument("--multimodal-tensor-transport", default="inline", choices=["inline", "shm", "rdma", "xgmi"])
    g.add_argument("--multimodal-shm-min-bytes", type=int, default=65_536)

    g = p.add_argument_group("Model / Tokenizer")
    g.add_argument("--model-path", default=None)
    g.add_argument("--tokenizer-path", default=None)
    g.add_argument("--chat-template", default=None)
    g.add_argument("--disable-tokenizer-autoload", action="store_true")
    g.add_argument("--tokenizer-cache-enable-l0", action="store_true", default=True)
    g.add_argument("--tokenizer-cache-l0-max-entries", type=int, default=8192)
    g.add_argument("--tokenizer-cache-enable-l1", action="store_true")
    g.add_argument("--tokenizer-cache-l1-max-memory", type=int, default=64 << 20)
    g.add_argument("--reasoning-parser", default=None)
    g.add_argument("--tool-call-parser", default=None)
    g.add_argument("--model-alias", action="append", default=[], help="alias=model_id, repeatable")

    g = p.add_argument_group("Rate Limiting / Admission")
    g.add_argument("--max-concurrent-requests", type=int, default=-1)
    g.add_argument("--queue-size", type=int, default=100)
    g.add_argument("--queue-timeout-secs", type=int, default=60)
    g.add_argument("--rate-limit-tokens-per-second", type=int, default=None)
    g.add_argument("--priority-scheduler-enabled", action="store_true")
    g.add_argument("--priority-scheduler-default-max-class", default="default")
    g.add_argument("--priority-scheduler-config", default=None)
    g.add_argument("--priority-scheduler-tenant-metric-top-n", type=int, default=0)
    g.add_argument("--tenant-rate-limit-enabled", action="store_true")
    g.add_argument("--tenant-rate-limit-config", default=None)

    g = p.add_argument_group("Retries")
    g.add_argument("--retry-max-retries", type=int, default=5)
    g.add_argument("--retry-initial-backoff-ms", type=int, default=50)
    g.add_argument("--retry-max-backoff-ms", type=int, default=30_000)
    g.add_argument("--retry-backoff-multiplier", type=float, default=1.5)
    g.add_argument("--retry-jitter-factor", type=float, default=0.2)
    g.add_argument("--disable-retries", action="store_true")

    g = p.add_argument_group("Circuit Breaker")
    g.add_argument("--cb-failure-threshold", type=int, default=10)
    g.add_argument("--cb-success-threshold", type=int, default=3)
    g.add_argument("--cb-timeout-duration-secs", type=int, default=60)
    g.add_argument("--cb-window-duration-secs", type=int, default=120)
    g.add_argument("--disable-circuit-breaker", action="store_true")

    g = p.add_argument_group("Health Checks")
    g.add_argument("--health-failure-threshold", type=int, default=3)
    g.add_argument("--health-success-threshold", type=int, default=2)
    g.add_argument("--health-check-timeout-secs", type=int, default=5)
    g.add_argument("--health-check-interval-secs", type=int, default=60)
    g.add_argument("--health-check-endpoint", default="/health")
    g.add_argument("--disable-health-check", action="store_true")
    g.add_argument("--remove-unhealthy-workers", action="store_true")
    g.add_argument("--drain-settle-secs", type=int, default=0)

    g = p.add_argument_group("Authentication")
    g.add_argument("--api-key", default=None)
    g.add_argument("--tenant-api-key", action="append", default=[], dest="tenant_api_keys",
                   help="key=tenant, repeatable")
    g.add_argument("--control-plane-api-keys", nargs="*", default=[])
    g.add_argument("--jwt-issuer", default=None)
    g.add_argument("--jwt-audience", default=None)
    g.add_argument("--jwt-jwks-uri", default=None,
                   help="http(s) JWKS URI or a local JWKS json file path")
    g.add_argument("--jwt-role-claim", default="roles")
    g.add_argument("--jwt-leeway-secs", type=float, default=60.0)
    g.add_argument("--jwt-enable-jti-check", action="store_true",
                   help="replay protection via a JTI LRU cache")
    g.add_argument("--jwt-allow-missing-exp", action="store_true",
                   help="accept tokens without an exp claim (default: reject, as the reference does)")
    g.add_argument("--admin-role", default="admin",
                   help="JWT role allowed to hit control-plane mutation endpoints")
    g.add_argument("--plugin-dir", default=None,
                   help="only plugin modules inside this directory may be loaded via POST /wasm")
    g.add_argument("--disable-audit-logging", action="store_true")
    g.add_argument("--trust-tenant-header", action="store_true")
    g.add_argument("--tenant-header-name", default="x-smg-tenant")

    g = p.add_argument_group("Storage")
    g.add_argument("--backend", default="memory", choices=["memory", "none", "sqlite", "postgres", "redis", "oracle"])
    g.add_argument("--history-backend", default=None)
    g.add_argument("--postgres-db-url", default=None)
    g.add_argument("--redis-url", default=None)

    g = p.add_argument_group("Observability")
    g.add_argument("--log-dir", default=None)
    g.add_argument("--log-level", default="info")
    g.add_argument("--log-json", action="store_true")
    g.add_argument("--prometheus-port", type=int, default=29000)
    g.add_argument("--prometheus-host", default="0.0.0.0")
    g.add_argument("--request-id-headers", nargs="*",
                   default=["x-request-id", "x-correlation-id", "x-trace-id", "request-id"])
    g.add_argument("--enable-trace", action="store_true")
    g.add_argument("--otlp-traces-endpoint", default=None)

    g = p.add_argument_group("HA Mesh")
    g.add_argument("--enable-mesh", action="store_true")
    g.add_argument("--mesh-server-name", default=None)
    g.add_argument("--mesh-host", default="0.0.0.0")
    g.add_argument("--mesh-advertise-host", default=None)
    g.add_argument("--mesh-port", type=int, default=32300)
    g.add_argument("--mesh-peer-urls", nargs="*", default=[])
    g.add_argument("--mesh-mtls-cert", default=None, help="mesh mTLS: this node's certificate (PEM)")
    g.add_argument("--mesh-mtls-key", default=None)
    g.add_argument("--mesh-mtls-ca", default=None, help="mesh CA every replica chains to")

    g = p.add_argument_group("Service Discovery (Kubernetes)")
    g.add_argument("--service-discovery", action="store_true")
    g.add_argument("--selector", nargs="*", default=[])
    g.add_argument("--service-discovery-port", type=int, default=8000)
    g.add_argument("--service-discovery-namespace", default=None)

    g = p.add_argument_group("MI355X Data Plane (RCCL over xGMI)")
    g.add_argument("--rccl-world-size", type=int, default=0, help=">0 enables the RCCL tick plane")
    g.add_argument("--rccl-tick-us", type=int, default=200)

    g = p.add_argument_group("Serve (worker spawn; reference bindings/python serve.py)")
    g.add_argument("--kv-fp8", action="store_true",
                   help="local GPU engines store their KV cache as fp8 e4m3 (compute stays bf16)")
    g.add_argument("--local-workers", type=int, default=0,
                   help="serve mode: spawn N in-process workers (GPU TorchEngine when available, mock simulator otherwise)")
    g.add_argument("--local-worker-model", default="local-model")

    g = p.add_argument_group("Misc")
    g.add_argument("--request-timeout-secs", type=int, default=1800)
    g.add_argument("--max-payload-size", type=int, default=512 << 20)
    g.add_argument("--cors-allowed-origins", nargs="*", default=[])
    g.add_argument("--mcp-config-path", default=None)
    g.add_argument("--tls-cert-path", default=None)
    g.add_argument("--tls-key-path", default=None)
    g.add_argument("--version", "-V", action="version", version="smg-amd 0.1.0")
    return p


def _policy_cfg(args, name: Optional[str] = None) -> PolicyConfig:
    return PolicyConfig(
        name=name or args.policy,
        cache_threshold=args.cache_threshold,
        balance_abs_threshold=args.balance_abs_threshold,
        balance_rel_threshold=args.balance_rel_threshold,
        balance_token_usage_threshold=args.balance_token_usage_threshold,
        overload_token_usage_threshold=args.overload_token_usage_threshold,
        eviction_interval_secs=args.eviction_interval,
        max_tree_size=args.max_tree_size,
        block_size=args.block_size,
        max_idle_secs=args.max_idle_secs,
        assignment_mode=args.assignment_mode,
        prefix_token_count=args.prefix_token_count,
        prefix_hash_load_factor=args.prefix_hash_load_factor,
        least_load_kv_pressure_weight=args.least_load_kv_pressure_weight,
        least_load_default_throughput=args.least_load_default_throughput,
        least_load_mean_prefill_tokens=args.least_load_mean_prefill_tokens,
        gpu_tree=not args.no_gpu_tree,
    )


def to_router_config(argv: Optional[List[str]] = None) -> RouterConfig:
    """CLI args -> RouterConfig (reference main.rs:1296 to_router_config)."""
    argv = list(sys.argv[1:] if argv is None else argv)
    prefill_urls, argv = _extract_url_port_pairs(argv, "--prefill")
    encode_urls, argv = _extract_url_port_pairs(argv, "--encode")
    args = build_parser().parse_args(argv)

    mode = RoutingMode.REGULAR
    if args.epd_disaggregation or encode_urls:
        mode = RoutingMode.ENCODE_PREFILL_DECODE
    elif args.pd_disaggregation or prefill_urls:
        mode = RoutingMode.PREFILL_DECODE

    cfg = RouterConfig(
        host=args.host,
        port=args.port,
        health_check_port=args.health_check_port,
        worker_urls=args.worker_urls,
        prefill_urls=prefill_urls,
        decode_urls=args.decode_urls,
        encode_urls=encode_urls,
        mode=mode,
        connection_mode=ConnectionMode(args.connection_mode),
        enable_igw=args.enable_igw,
        policy=_policy_cfg(args),
        prefill_policy=_policy_cfg(args, args.prefill_policy) if args.prefill_policy else None,
        decode_policy=_policy_cfg(args, args.decode_policy) if args.decode_policy else None,
        encode_policy=_policy_cfg(args, args.encode_policy) if args.encode_policy else None,
        dp_aware=args.dp_aware,
        dp_minimum_tokens_scheduler=args.dp_minimum_tokens_scheduler,
        routing_key_override=args.routing_key_override,
        model_path=args.model_path,
        tokenizer_path=args.tokenizer_path,
        chat_template=args.chat_template,
        disable_tokenizer_autoload=args.disable_tokenizer_autoload,
        reasoning_parser=args.reasoning_parser,
        tool_call_parser=args.tool_call_parser,
        request_timeout_secs=args.request_timeout_secs,
        max_payload_size=args.max_payload_size,
        cors_allowed_origins=args.cors_allowed_origins,
        request_id_headers=args.request_id_headers,
        trust_tenant_header=args.trust_tenant_header,
        tenant_header_name=args.tenant_header_name,
        log_dir=args.log_dir,
        log_level=args.log_level,
        log_json=args.log_json,
        prometheus_port=args.prometheus_port,
        prometheus_host=args.prometheus_host,
        mcp_config_path=args.mcp_config_path,
        tls_cert_path=args.tls_cert_path,
        tls_key_path=args.tls_key_path,
        worker_startup_timeout_secs=args.worker_startup_timeout_secs,
        worker_startup_delay=args.worker_startup_delay,
        load_monitor_interval=args.load_monitor_interval,
        engine_metrics=args.engine_metrics,
        multimodal_tensor_transport=args.multimodal_tensor_transport,
        multimodal_shm_min_bytes=args.multimodal_shm_min_bytes,
    )
    cfg.model_aliases = dict(a.split("=", 1) for a in args.model_alias if "=" in a)
    cfg.retry.max_retries = args.retry_max_retries
    cfg.retry.initial_backoff_ms = args.retry_initial_backoff_ms
    cfg.retry.max_backoff_ms = args.retry_max_backoff_ms
    cfg.retry.backoff_multiplier = args.retry_backoff_multiplier
    cfg.retry.jitter_factor = args.retry_jitter_factor
    cfg.retry.disable = args.disable_retries
    cfg.circuit_breaker.failure_threshold = args.cb_failure_threshold
    cfg.circuit_breaker.success_threshold = args.cb_success_threshold
    cfg.circuit_breaker.timeout_duration_secs = args.cb_timeout_duration_secs
    cfg.circuit_breaker.window_duration_secs = args.cb_window_duration_secs
    cfg.circuit_breaker.disable = args.disable_circuit_breaker
    cfg.health_check.failure_threshold = args.health_failure_threshold
    cfg.health_check.success_threshold = args.health_success_threshold
    cfg.health_check.timeout_secs = args.health_check_timeout_secs
    cfg.health_check.check_interval_secs = args.health_check_interval_secs
    cfg.health_check.endpoint = args.health_check_endpoint
    cfg.health_check.disable = args.disable_health_check
    cfg.health_check.remove_unhealthy_workers = args.remove_unhealthy_workers
    cfg.health_check.drain_settle_secs = args.drain_settle_secs
    cfg.rate_limit.max_concurrent_requests = args.max_concurrent_requests
    cfg.rate_limit.queue_size = args.queue_size
    cfg.rate_limit.queue_timeout_secs = args.queue_timeout_secs
    cfg.rate_limit.tokens_per_second = args.rate_limit_tokens_per_second
    cfg.priority_scheduler.enabled = args.priority_scheduler_enabled
    cfg.priority_scheduler.default_max_class = args.priority_scheduler_default_max_class
    cfg.priority_scheduler.config_path = args.priority_scheduler_config
    cfg.priority_scheduler.tenant_metric_top_n = args.priority_scheduler_tenant_metric_top_n
    cfg.tenant_rate_limit.enabled = args.tenant_rate_limit_enabled
    cfg.tenant_rate_limit.config_path = args.tenant_rate_limit_config
    cfg.auth.api_key = args.api_key
    cfg.auth.tenant_api_keys = dict(a.split("=", 1) for a in args.tenant_api_keys if "=" in a)
    cfg.auth.control_plane_api_keys = args.control_plane_api_keys
    cfg.auth.jwt_issuer = args.jwt_issuer
    cfg.auth.jwt_audience = args.jwt_audience
    cfg.auth.jwt_jwks_uri = args.jwt_jwks_uri
    cfg.auth.jwt_role_claim = args.jwt_role_claim
    cfg.auth.jwt_leeway_secs = args.jwt_leeway_secs
    cfg.auth.jwt_enable_jti_check = args.jwt_enable_jti_check
    cfg.auth.jwt_require_exp = not args.jwt_allow_missing_exp
    cfg.auth.admin_role = args.admin_role
    cfg.plugin_dir = args.plugin_dir
    cfg.auth.disable_audit_logging = args.disable_audit_logging
    cfg.storage.backend = args.backend
    cfg.storage.history_backend = args.history_backend
    cfg.storage.postgres_db_url = args.postgres_db_url
    cfg.storage.redis_url = args.redis_url
    cfg.mesh.enabled = args.enable_mesh
    cfg.mesh.server_name = args.mesh_server_name
    cfg.mesh.host = args.mesh_host
    cfg.mesh.advertise_host = args.mesh_advertise_host
    cfg.mesh.port = args.mesh_port
    cfg.mesh.peer_urls = args.mesh_peer_urls
    cfg.mesh.mtls_cert = args.mesh_mtls_cert
    cfg.mesh.mtls_key = args.mesh_mtls_key
    cfg.mesh.mtls_ca = args.mesh_mtls_ca
    cfg.discovery.enabled = args.service_discovery
    cfg.discovery.selector = dict(s.split("=", 1) for s in args.selector if "=" in s)
    cfg.discovery.port = args.service_discovery_port
    cfg.discovery.namespace = args.service_discovery_namespace
    cfg.trace.enabled = args.enable_trace
    cfg.trace.otlp_endpoint = args.otlp_traces_endpoint
    cfg.tokenizer_cache.enable_l0 = args.tokenizer_cache_enable_l0
    cfg.tokenizer_cache.l0_max_entries = args.tokenizer_cache_l0_max_entries
    cfg.tokenizer_cache.enable_l1 = args.tokenizer_cache_enable_l1
    cfg.tokenizer_cache.l1_max_memory = args.tokenizer_cache_l1_max_memory
    if args.rccl_world_size > 0:
        cfg.rccl.enabled = True
        cfg.rccl.world_size = args.rccl_world_size
        cfg.rccl.tick_interval_us = args.rccl_tick_us
    return cfg


def rccl_worker_main(cfg) -> None:
    """Worker-rank entry for `smg launch --connection-mode rccl` under
    torchrun: ranks >= 1 run the lockstep plane <-> engine loop instead of
    the HTTP server (routers/rccl_router.py run_worker_loop)."""
    import os

    import torch
    import torch.distributed as dist

    from .comm.plane import PlaneConfig, WorkerPlane
    from .engine.torch_engine import TorchEngine, TorchEngineConfig
    from .routers.rccl_router import run_worker_loop

    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    use_gpu = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if use_gpu else "cpu"
    if use_gpu:
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        dist.init_process_group(backend="nccl" if use_gpu else "gloo")
    eng = TorchEngine(
        TorchEngineConfig() if use_gpu else TorchEngineConfig.tiny(),
        device=device, graphs=use_gpu,
    )
    plane = WorkerPlane(
        PlaneConfig(
            max_reqs_per_tick=cfg.rccl.max_batch_requests,
            max_prompt=cfg.rccl.max_tokens_per_msg,
            device=device if use_gpu else "cpu",
        )
    )
    from .config import RoutingMode
    from .routers.rccl_router import epd_rank_roles, pd_rank_roles

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if cfg.mode == RoutingMode.PREFILL_DECODE:
        role = pd_rank_roles(world).get(rank, "regular")
    elif cfg.mode == RoutingMode.ENCODE_PREFILL_DECODE:
        role = epd_rank_roles(world).get(rank, "regular")
    else:
        role = "regular"
    if role == "encode":
        # EPD encode rank: the vision tower instead of a text engine
        from .multimodal.encoder import EncodeWorker, ToyVisionEncoder

        eng = EncodeWorker(ToyVisionEncoder(
            eng.cfg.d_model, image_size=64, patch=16,
            device=device, dtype=eng.dtype))
    run_worker_loop(eng, plane, role=role)
    dist.destroy_process_group()


def main(argv: Optional[List[str]] = None) -> None:
    raw_argv = list(sys.argv[1:] if argv is None else argv)
    cfg = to_router_config(argv)
    from .observability.logging import setup_logging

    setup_logging(cfg.log_level, cfg.log_json, cfg.log_dir)
    import os as _os

    from .config import ConnectionMode

    if cfg.connection_mode == ConnectionMode.RCCL and int(_os.environ.get("RANK", "0")) > 0:
        rccl_worker_main(cfg)
        return
    # `smg serve`: spawn local workers next to the router (reference
    # bindings/python serve.py spawns engine workers + router together)
    n_local = 0
    model = "local-model"
    if "--local-workers" in raw_argv:
        i = raw_argv.index("--local-workers")
        n_local = int(raw_argv[i + 1])
    if "--local-worker-model" in raw_argv:
        model = raw_argv[raw_argv.index("--local-worker-model") + 1]

    from .server.app import startup

    async def _run():
        ctx = await startup(cfg)
        engines = []
        if n_local:
            from .workers.worker import Worker

            try:
                import torch

                use_gpu = torch.cuda.is_available()
            except ImportError:
                use_gpu = False
            for i in range(n_local):
                if use_gpu:
                    from .engine.torch_engine import TorchEngine, TorchEngineConfig
                    from .grpc.servicer import EngineAdapter

                    ecfg = TorchEngineConfig.bench_1b()
                    ecfg.kv_fp8 = "--kv-fp8" in raw_argv
                    eng = TorchEngine(ecfg, device=f"cuda:{i % torch.cuda.device_count()}")
                    adapter = EngineAdapter(eng)
                    await adapter.start()
                    engines.append(adapter)
                    w = Worker(f"sim://gpu-{i}", model_id=model)
                    w.extra["engine"] = _TorchHttpShim(adapter, model)
                else:
                    from .mock.engine import MockWorkerEngine

                    eng = MockWorkerEngine()
                    eng.config.model_id = model
                    await eng.start()
                    engines.append(eng)
                    w = Worker(f"sim://local-{i}", model_id=model)
                    w.extra["engine"] = eng
                ctx.worker_registry.register(w)
        try:
            await asyncio.Event().wait()
        finally:
            for e in engines:
                await e.stop()
            await ctx.shutdown()

    try:
        asyncio.run(_run())
    except KeyboardInterrupt:
        pass


class _TorchHttpShim:
    """Minimal handle() shim presenting the GPU engine over the sim:// transport."""

    def __init__(self, adapter, model):
        self.adapter = adapter
        self.model = model

    async def handle(self, path, body, headers):
        import json as _json

        from .grpc import api as _api

        if path in ("/health", "/health_generate"):
            return 200, {}, b'{"status":"ok"}'
        if path == "/get_loads":
            return 200, {}, _json.dumps({"loads": self.adapter.loads()}).encode()
        if path in ("/v1/chat/completions", "/v1/completions", "/generate"):
            body = body or {}
            text = ""
            msgs = body.get("messages")
            if msgs:
                text = "\n".join(m.get("content", "") for m in msgs if isinstance(m.get("content"), str))
            else:
                text = body.get("prompt") or body.get("text") or ""
            ids = [hash(text[i: i + 4]) & 0x7FFF for i in range(0, len(text), 4)] or [1]
            req = _api.GenerateRequest("r", input_ids=ids,
                                       sampling=_api.SamplingParams(max_new_tokens=int(body.get("max_tokens") or 16)))
            toks = []
            async for chunk in self.adapter.generate(req):
                toks.extend(chunk.token_ids)
                if chunk.finished:
                    break
            content = "".join(f" tok{t}" for t in toks)
            return 200, {}, _json.dumps({
                "id": "cmpl-local", "object": "chat.completion", "model": self.model,
                "choices": [{"index": 0, "message": {"role": "assistant", "content": content},
                             "text": content, "finish_reason": "stop"}],
                "usage": {"prompt_tokens": len(ids), "completion_tokens": len(toks),
                          "total_tokens": len(ids) + len(toks)},
            }).encode()
        return 404, {}, b'{"error":"not found"}'


if __name__ == "__main__":
    main()
