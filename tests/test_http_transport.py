"""Real-socket HTTP transport e2e: gateway -> aiohttp client -> mock worker
HTTP server (the production off-node path; reference mock_worker http.rs +
monitor GetLoads polling)."""
import asyncio
import json

import pytest

from smg_amd.config import HealthCheckConfig, PolicyConfig, RouterConfig
from smg_amd.mock.engine import SimConfig
from smg_amd.mock.server import serve_mock_worker
from smg_amd.routers.base import RouteRequest
from smg_amd.routers.http_router import HttpRouter
from smg_amd.server.app_context import AppContext
from smg_amd.workers.monitor import WorkerMonitor
from smg_amd.workers.worker import Worker


def _req(path, body):
    return RouteRequest(path=path, body=body, raw_body=json.dumps(body).encode(), request_id="h1")


def test_http_proxy_unary_and_stream(runner):
    async def run():
        engine, runner_srv = await serve_mock_worker(port=0, config=SimConfig(speedup=50.0))
        port = runner_srv.addresses[0][1]
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        ctx = AppContext(cfg)
        ctx.worker_registry.register(Worker(f"http://127.0.0.1:{port}", model_id="mock-model"))
        router = HttpRouter(ctx.worker_registry, ctx.policy_registry, cfg)
        try:
            # unary over a real socket
            resp = await router.route(_req("/v1/chat/completions",
                                           {"model": "mock-model", "max_tokens": 3,
                                            "messages": [{"role": "user", "content": "hi"}]}))
            assert resp.status == 200, resp.body
            data = json.loads(resp.body)
            assert data["usage"]["completion_tokens"] == 3
            # SSE stream over a real socket
            resp = await router.route(_req("/v1/chat/completions",
                                           {"model": "mock-model", "max_tokens": 4, "stream": True,
                                            "messages": [{"role": "user", "content": "hi"}]}))
            assert resp.is_stream
            frames = b""
            async for chunk in resp.stream:
                frames += chunk
            assert b"data: [DONE]" in frames
            assert frames.count(b"chat.completion.chunk") >= 4
            w = ctx.worker_registry.all()[0]
            assert w.active_requests == 0  # guards released
        finally:
            await router.shutdown()
            await engine.stop()
            await runner_srv.cleanup()

    runner(run())


def test_monitor_polls_health_and_loads_over_http(runner):
    async def run():
        engine, runner_srv = await serve_mock_worker(port=0, config=SimConfig(speedup=50.0))
        port = runner_srv.addresses[0][1]
        ctx = AppContext(RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False)))
        w = ctx.worker_registry.register(Worker(f"http://127.0.0.1:{port}", model_id="mock-model"))
        hc = HealthCheckConfig(check_interval_secs=1, success_threshold=1, failure_threshold=1)
        monitor = WorkerMonitor(ctx.worker_registry, hc, load_interval_secs=0.2,
                                policy_registry=ctx.policy_registry)
        await monitor.start()
        try:
            # submit work so loads are non-trivial
            engine.sim.submit(list(range(64)), 50)
            for _ in range(30):
                if w.health.value == "healthy" and w.token_usage is not None:
                    break
                await asyncio.sleep(0.1)
            assert w.health.value == "healthy"
            assert w.token_usage is not None and w.token_usage > 0
        finally:
            await monitor.stop()
            await engine.stop()
            await runner_srv.cleanup()

    runner(run())


def test_monitor_marks_dead_worker_unhealthy(runner):
    async def run():
        ctx = AppContext(RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False)))
        w = ctx.worker_registry.register(Worker("http://127.0.0.1:1", model_id="m"))  # nothing there
        hc = HealthCheckConfig(check_interval_secs=0.2, failure_threshold=2, timeout_secs=1)
        monitor = WorkerMonitor(ctx.worker_registry, hc, load_interval_secs=60,
                                policy_registry=ctx.policy_registry)
        await monitor.start()
        try:
            for _ in range(40):
                if w.health.value == "unhealthy":
                    break
                await asyncio.sleep(0.1)
            assert w.health.value == "unhealthy"
            assert not w.is_available()
        finally:
            await monitor.stop()

    runner(run())
