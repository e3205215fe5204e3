"""PD / EPD disaggregation tests, HTTP and gRPC modes (reference:
http/pd_router.rs, grpc request_execution.rs PD :253 EPD :302,
stages/encode.rs; e2e_test/router/test_pd_*.py)."""
import json

import pytest

from smg_amd.config import ConnectionMode, PolicyConfig, RouterConfig, RoutingMode
from smg_amd.mock.engine import MockWorkerEngine, SimConfig
from smg_amd.routers.base import RouteRequest
from smg_amd.routers.factory import RouterManager
from smg_amd.server.app_context import AppContext
from smg_amd.workers.worker import Worker, WorkerType

from tests.test_multimodal import png_data_url


def _req(path, body):
    return RouteRequest(path=path, body=body, raw_body=json.dumps(body).encode(), request_id="pd1")


def test_http_pd_bootstrap_injection(runner):
    async def run():
        cfg = RouterConfig(
            policy=PolicyConfig(name="round_robin", gpu_tree=False),
            mode=RoutingMode.PREFILL_DECODE,
        )
        cfg.health_check.disable = True
        ctx = AppContext(cfg)
        seen_bodies = []

        class SpyEngine(MockWorkerEngine):
            async def handle(self, path, body, headers):
                seen_bodies.append(dict(body or {}))
                return await super().handle(path, body, headers)

        engines = []
        for i, wt in enumerate([WorkerType.PREFILL, WorkerType.DECODE]):
            eng = SpyEngine(SimConfig(speedup=50.0, model_id="m"))
            await eng.start()
            w = Worker(f"sim://pd-{i}", model_id="m", worker_type=wt, bootstrap_port=9000 + i)
            w.extra["engine"] = eng
            ctx.worker_registry.register(w)
            engines.append(eng)
        ctx.router_manager = RouterManager(ctx, cfg)
        resp = await ctx.router_manager.route(
            _req("/v1/chat/completions", {"model": "m", "max_tokens": 2,
                                          "messages": [{"role": "user", "content": "hi"}]}))
        assert resp.status == 200, resp.body
        assert len(seen_bodies) == 2  # both legs dispatched
        for b in seen_bodies:
            assert "bootstrap_host" in b and "bootstrap_room" in b
            assert b["bootstrap_port"] == 9000  # the prefill worker's port
        for e in engines:
            await e.stop()
        await ctx.router_manager.shutdown()

    runner(run())


async def grpc_pd_setup(mode, with_encode=False):
    from smg_amd.grpc.servicer import serve_grpc_worker
    from smg_amd.routers.grpc.router import GrpcRouter

    cfg = RouterConfig(
        policy=PolicyConfig(name="round_robin", gpu_tree=False),
        connection_mode=ConnectionMode.GRPC,
        mode=mode,
    )
    cfg.health_check.disable = True
    ctx = AppContext(cfg)
    servers = []
    roles = [WorkerType.PREFILL, WorkerType.DECODE] + ([WorkerType.ENCODE] if with_encode else [])
    for wt in roles:
        server, adapter, port = await serve_grpc_worker(port=0, sim_config=SimConfig(speedup=50.0, model_id="m"))
        servers.append((server, adapter))
        ctx.worker_registry.register(
            Worker(f"grpc://127.0.0.1:{port}", model_id="m", worker_type=wt, bootstrap_port=7001)
        )
    return ctx, GrpcRouter(ctx, cfg), servers


def test_grpc_pd_dual_dispatch(runner):
    async def run():
        ctx, router, servers = await grpc_pd_setup(RoutingMode.PREFILL_DECODE)
        try:
            resp = await router.route(
                _req("/v1/chat/completions", {"model": "m", "max_tokens": 3,
                                              "messages": [{"role": "user", "content": "hello pd"}]}))
            assert resp.status == 200, resp.body
            data = json.loads(resp.body)
            assert data["choices"][0]["message"]["content"]
            import asyncio

            await asyncio.sleep(0.2)  # let the prefill leg drain
            prefill = ctx.worker_registry.by_type(WorkerType.PREFILL)[0]
            decode = ctx.worker_registry.by_type(WorkerType.DECODE)[0]
            assert prefill.active_requests == 0 and decode.active_requests == 0
            # both engines saw work
            total = sum(s[1].engine.sim.total_generated for s in servers)
            assert total >= 3
        finally:
            await router.shutdown()
            for server, adapter in servers:
                await adapter.stop()
                server.stop(grace=None)

    runner(run())


def test_grpc_epd_encode_leg(runner):
    async def run():
        ctx, router, servers = await grpc_pd_setup(RoutingMode.ENCODE_PREFILL_DECODE, with_encode=True)
        try:
            body = {
                "model": "m", "max_tokens": 2,
                "messages": [{
                    "role": "user",
                    "content": [
                        {"type": "text", "text": "describe"},
                        {"type": "image_url", "image_url": {"url": png_data_url(32, 32)}},
                    ],
                }],
            }
            resp = await router.route(_req("/v1/chat/completions", body))
            assert resp.status == 200, resp.body
        finally:
            await router.shutdown()
            for server, adapter in servers:
                await adapter.stop()
                server.stop(grace=None)

    runner(run())
