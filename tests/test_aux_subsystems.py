"""Realtime WS proxy, discovery, tracing, plugin hooks
(reference: realtime/ws.rs, service_discovery.rs, otel_trace.rs, wasm)."""
import asyncio
import json
import os

import pytest
from aiohttp import WSMsgType, web
from aiohttp.test_utils import TestServer

from smg_amd.discovery import FileDiscovery
from smg_amd.discovery.source import reconcile
from smg_amd.observability.tracing import InFlightTracker, Tracer, inject_trace_context
from smg_amd.plugins import PluginManager, ShortCircuit
from smg_amd.workers.registry import WorkerRegistry

from tests.test_gateway_e2e import make_ctx, start_client, stop_all


class TestDiscovery:
    def test_file_reconcile(self, runner, tmp_path):
        async def run():
            path = tmp_path / "workers.json"
            path.write_text(json.dumps([{"url": "http://d1:8000", "model_id": "m"}]))
            reg = WorkerRegistry()
            src = FileDiscovery(str(path))
            await reconcile(reg, src)
            assert reg.get_by_url("http://d1:8000") is not None
            # file changes: worker removed
            path.write_text(json.dumps([{"url": "http://d2:8000", "model_id": "m"}]))
            await reconcile(reg, src)
            assert reg.get_by_url("http://d1:8000") is None
            assert reg.get_by_url("http://d2:8000") is not None

        runner(run())

    def test_static_workers_not_removed(self, runner, tmp_path):
        async def run():
            from smg_amd.workers.worker import Worker

            path = tmp_path / "w.json"
            path.write_text("[]")
            reg = WorkerRegistry()
            reg.register(Worker("http://static:1"))
            await reconcile(reg, FileDiscovery(str(path)))
            assert reg.get_by_url("http://static:1") is not None

        runner(run())

    def test_k8s_outside_cluster_raises(self, monkeypatch):
        # the raw-HTTP watch needs no client package; outside a cluster the
        # api_base resolution fails loudly
        from smg_amd.discovery.source import KubernetesDiscovery

        monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
        src = KubernetesDiscovery({"app": "x"}, 8000)
        with pytest.raises(RuntimeError, match="KUBERNETES_SERVICE_HOST"):
            src.api_base()


class TestTracing:
    def test_span_nesting_and_traceparent(self):
        tr = Tracer(enabled=True)
        with tr.span("outer") as outer:
            headers = {}
            inject_trace_context(headers)
            assert headers["traceparent"].startswith(f"00-{outer.trace_id}-")
            with tr.span("inner") as inner:
                assert inner.trace_id == outer.trace_id
                assert inner.parent_id == outer.span_id
        assert len(tr.finished) == 2

    def test_incoming_traceparent_continues_trace(self):
        tr = Tracer(enabled=True)
        with tr.span("handler", traceparent="00-" + "ab" * 16 + "-" + "cd" * 8 + "-01") as s:
            assert s.trace_id == "ab" * 16
            assert s.parent_id == "cd" * 8

    def test_inflight_tracker(self):
        t = InFlightTracker()
        t.start("r1")
        t.start("r2")
        assert len(t) == 2
        hist = t.age_histogram()
        assert hist["<1s"] == 2
        t.finish("r1")
        assert len(t) == 1


class TestPlugins:
    def write_plugin(self, tmp_path, body):
        p = tmp_path / "plug.py"
        p.write_text(body)
        return str(p)

    def test_on_request_mutation(self, tmp_path):
        pm = PluginManager()
        pm.add_module(self.write_plugin(tmp_path, "def on_request(ctx):\n    return {'tenant_id': 'from-plugin'}\n"))
        ctx = pm.run_phase("on_request", {"path": "/x"})
        assert ctx["tenant_id"] == "from-plugin"

    def test_short_circuit(self, tmp_path):
        pm = PluginManager()
        pm.add_module(self.write_plugin(
            tmp_path,
            "from smg_amd.plugins import ShortCircuit\n"
            "def on_request(ctx):\n    raise ShortCircuit(403, b'{\"blocked\":true}')\n",
        ))
        with pytest.raises(ShortCircuit) as e:
            pm.run_phase("on_request", {})
        assert e.value.status == 403

    def test_rest_add_remove(self, runner, tmp_path):
        async def run():
            plug = self.write_plugin(tmp_path, "def on_response(ctx):\n    return {'extra_headers': {'x-plugged': '1'}}\n")
            ctx, engines = make_ctx()
            ctx.config.plugin_dir = str(tmp_path)
            client = await start_client(ctx, engines)
            try:
                # paths outside --plugin-dir are refused (never exec
                # request-supplied native code)
                resp = await client.post("/wasm", json={"path": "/etc/passwd", "name": "evil"})
                assert resp.status == 403
                resp = await client.post("/wasm", json={"path": plug, "name": "test"})
                assert resp.status == 201
                mod_id = (await resp.json())["module_uuid"]
                resp = await client.get("/wasm")
                assert len((await resp.json())["modules"]) == 1
                # plugin adds a response header on inference routes
                resp = await client.post("/v1/completions", json={"model": "mock-model", "prompt": "x", "max_tokens": 1})
                assert resp.headers.get("x-plugged") == "1"
                resp = await client.delete(f"/wasm/{mod_id}")
                assert resp.status == 200
            finally:
                await stop_all(client, engines)

        runner(run())


class TestRealtime:
    def test_session_mint_and_ws_relay(self, runner):
        async def run():
            # mock worker WS: echoes frames with a prefix
            async def ws_handler(request):
                ws = web.WebSocketResponse()
                await ws.prepare(request)
                async for msg in ws:
                    if msg.type == WSMsgType.TEXT:
                        await ws.send_str("echo:" + msg.data)
                        if msg.data == "bye":
                            await ws.close()
                return ws

            worker_app = web.Application()
            worker_app.router.add_get("/v1/realtime", ws_handler)
            worker_srv = TestServer(worker_app)
            await worker_srv.start_server()

            ctx, engines = make_ctx(n_workers=0)
            from smg_amd.workers.worker import Worker

            ctx.worker_registry.register(
                Worker(f"http://127.0.0.1:{worker_srv.port}", model_id="mock-model")
            )
            client = await start_client(ctx, engines)
            try:
                resp = await client.post("/v1/realtime/sessions", json={"model": "mock-model"})
                sess = await resp.json()
                assert sess["object"] == "realtime.session"
                assert sess["client_secret"]["value"].startswith("ek_")

                ws = await client.ws_connect("/v1/realtime?model=mock-model")
                # the relay validates client events (realtime_events.rs), so
                # frames must be well-formed realtime JSON
                await ws.send_str(json.dumps({"type": "input_audio_buffer.append", "audio": "QQ=="}))
                msg = await ws.receive(timeout=5)
                assert msg.data.startswith("echo:") and "input_audio_buffer.append" in msg.data
                await ws.send_str(json.dumps({"type": "response.create"}))
                msg = await ws.receive(timeout=5)
                assert "response.create" in msg.data
                await ws.close()
            finally:
                await stop_all(client, engines)
                await worker_srv.close()

        runner(run())


class TestWebRtcSignaling:
    """WebRTC call signaling broker (reference webrtc_bridge.rs:147-270):
    SDP offer validation, upstream brokering, call session management."""

    OFFER = (
        "v=0\r\no=- 1 1 IN IP4 0.0.0.0\r\ns=-\r\nt=0 0\r\n"
        "m=audio 9 UDP/TLS/RTP/SAVPF 111\r\n"
        "a=ice-ufrag:abcd\r\na=ice-pwd:efgh1234567890\r\n"
        "a=fingerprint:sha-256 AA:BB\r\n"
    )

    def test_invalid_sdp_rejected(self, runner):
        async def run():
            ctx, engines = make_ctx(n_workers=0)
            client = await start_client(ctx, engines)
            try:
                resp = await client.post("/v1/realtime/calls", data=b"not sdp at all")
                assert resp.status == 400
            finally:
                await stop_all(client, engines)

        runner(run())

    def test_no_upstream_501_with_guidance(self, runner):
        async def run():
            ctx, engines = make_ctx(n_workers=0)
            client = await start_client(ctx, engines)
            try:
                resp = await client.post("/v1/realtime/calls", data=self.OFFER.encode())
                assert resp.status == 501
                assert "webrtc=true" in await resp.text()
            finally:
                await stop_all(client, engines)

        runner(run())

    def test_broker_to_upstream_and_manage_call(self, runner):
        from aiohttp import web as aioweb
        from aiohttp.test_utils import TestServer as AioTestServer

        from smg_amd.workers.worker import Worker

        async def run():
            hangups = []

            async def upstream_calls(request):
                sdp = await request.text()
                assert "a=ice-ufrag:" in sdp
                return aioweb.Response(
                    status=201, text=sdp.replace("a=ice-ufrag:abcd", "a=ice-ufrag:srvr"),
                    headers={"Location": "/v1/realtime/calls/up_1",
                             "Content-Type": "application/sdp"})

            async def upstream_hangup(request):
                hangups.append(request.match_info["cid"])
                return aioweb.json_response({})

            up = aioweb.Application()
            up.router.add_post("/v1/realtime/calls", upstream_calls)
            up.router.add_delete("/v1/realtime/calls/{cid}", upstream_hangup)
            upstream = AioTestServer(up)
            await upstream.start_server()

            ctx, engines = make_ctx(n_workers=0)
            ctx.worker_registry.register(Worker(
                f"http://127.0.0.1:{upstream.port}", model_id="rt",
                labels={"webrtc": "true"}))
            client = await start_client(ctx, engines)
            try:
                resp = await client.post("/v1/realtime/calls", data=self.OFFER.encode())
                assert resp.status == 201, await resp.text()
                answer = await resp.text()
                assert "a=ice-ufrag:srvr" in answer  # upstream's SDP answer relayed
                loc = resp.headers["Location"]
                assert loc.startswith("/v1/realtime/calls/rtc_")
                # call is managed
                resp = await client.get(loc)
                assert resp.status == 200
                assert (await resp.json())["state"] == "active"
                resp = await client.delete(loc)
                assert resp.status == 200
                assert (await resp.json())["state"] == "ended"
                resp = await client.get(loc)
                assert resp.status == 404
            finally:
                await stop_all(client, engines)
                await upstream.close()

        runner(run())
