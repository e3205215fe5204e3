"""Metrics breadth + OTLP wire export + tracing middleware (reference:
observability/metrics.rs 161 registrations, otel_trace.rs OTLP batch
exporter, W3C propagation http/router.rs:466)."""
import asyncio
import json

import pytest
from aiohttp import web
from aiohttp.test_utils import TestServer

from smg_amd.observability.metrics import GatewayMetrics
from smg_amd.observability.tracing import OtlpHttpExporter, Span, Tracer


def test_metric_family_breadth():
    """>=100 families, covering the reference's smg_http_*/
    smg_router_*/smg_worker_*/smg_engine_*/smg_pd_*/smg_db_*/smg_discovery_*
    groups plus the MI355X-native plane/tree families."""
    m = GatewayMetrics()
    fams = {c.name for c in m.registry.collect()}
    assert len(fams) >= 100, len(fams)
    for expected in [
        "smg_http_requests", "smg_router_requests", "smg_router_stage_duration_seconds",
        "smg_worker_selection", "smg_worker_cb_outcomes", "smg_engine_token_usage",
        "smg_pd_kv_transfer_duration_seconds", "smg_db_operations",
        "smg_discovery_registrations", "smg_mm_tensors", "smg_mcp_tool_duration_seconds",
        "smg_plane_ticks", "smg_gpu_tree_nodes_live", "smg_mesh_partitioned",
    ]:
        assert any(f.startswith(expected) for f in fams), expected


def test_otlp_exporter_wire_format(runner):
    async def run():
        received = []

        async def collect(request):
            received.append(await request.json())
            return web.json_response({})

        app = web.Application()
        app.router.add_post("/v1/traces", collect)
        server = TestServer(app)
        await server.start_server()
        try:
            exp = OtlpHttpExporter(f"http://127.0.0.1:{server.port}", max_batch=2)
            tracer = Tracer(enabled=True, otlp_exporter=exp)
            with tracer.span("req-a", method="POST"):
                pass
            with tracer.span("req-b"):
                pass
            await exp.flush()
            assert exp.exported == 2
            payload = received[0]
            spans = payload["resourceSpans"][0]["scopeSpans"][0]["spans"]
            assert [s["name"] for s in spans] == ["req-a", "req-b"]
            s0 = spans[0]
            assert len(s0["traceId"]) == 32 and len(s0["spanId"]) == 16
            assert int(s0["endTimeUnixNano"]) >= int(s0["startTimeUnixNano"])
            attrs = {a["key"]: a["value"]["stringValue"] for a in s0["attributes"]}
            assert attrs["method"] == "POST"
            res = payload["resourceSpans"][0]["resource"]["attributes"]
            assert {"key": "service.name", "value": {"stringValue": "smg-amd"}} in res
        finally:
            await server.close()

    runner(run())


def test_otlp_export_error_counted(runner):
    async def run():
        exp = OtlpHttpExporter("http://127.0.0.1:1")  # nothing listening
        exp((lambda: Span("x", "ab" * 16, "cd" * 8))())
        await exp.flush()
        assert exp.export_errors == 1 and exp.exported == 0

    runner(run())


def test_tracing_middleware_e2e(runner):
    from aiohttp.test_utils import TestClient

    from smg_amd.config import PolicyConfig, RouterConfig
    from smg_amd.server.app import build_app
    from smg_amd.server.app_context import AppContext
    from smg_amd.routers.factory import RouterManager

    async def run():
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        cfg.health_check.disable = True
        cfg.trace.enabled = True
        ctx = AppContext(cfg)
        ctx.router_manager = RouterManager(ctx, cfg)
        client = TestClient(TestServer(build_app(ctx)))
        await client.start_server()
        try:
            tp = "00-" + "ab" * 16 + "-" + "cd" * 8 + "-01"
            r = await client.get("/health", headers={"traceparent": tp})
            assert r.status == 200
            r = await client.get("/v1/models", headers={"traceparent": tp})
            assert r.status == 200
            spans = [s for s in ctx.tracer.finished if s.trace_id == "ab" * 16]
            assert spans, "incoming traceparent must continue the trace"
            assert spans[-1].attributes.get("status") == 200
        finally:
            await client.close()

    runner(run())


def test_scrape_gauge_refresh(runner):
    from aiohttp.test_utils import TestClient

    from smg_amd.config import PolicyConfig, RouterConfig
    from smg_amd.server.app import build_app
    from smg_amd.server.app_context import AppContext
    from smg_amd.routers.factory import RouterManager
    from smg_amd.workers.worker import Worker

    async def run():
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        cfg.health_check.disable = True
        ctx = AppContext(cfg)
        ctx.worker_registry.register(Worker("http://w1:1", model_id="m"))
        ctx.worker_registry.register(Worker("http://w2:1", model_id="m"))
        ctx.router_manager = RouterManager(ctx, cfg)
        client = TestClient(TestServer(build_app(ctx)))
        await client.start_server()
        try:
            r = await client.get("/metrics")
            text = await r.text()
            assert 'smg_worker_pool_size{model="m"} 2.0' in text
        finally:
            await client.close()

    runner(run())
