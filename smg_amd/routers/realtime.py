"""Realtime API: WebSocket proxy + session registry (reference:
model_gateway/src/routers/common/realtime/ — ws.rs/proxy.rs WebSocket proxy,
registry.rs with TTL reaper; the WebRTC relay (webrtc_bridge.rs, str0m) maps
to a data-channel relay over the same registry and is out of scope for a
CPU-only test path).

GET /v1/realtime upgrades the client connection and relays frames to the
selected worker's /v1/realtime WebSocket; POST /v1/realtime/sessions mints an
ephemeral session token.
"""
from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import Dict, Optional

import aiohttp
from aiohttp import WSMsgType, web

from ..policies import SelectWorkerInfo
from ..protocols.openai import error_body


class RealtimeRegistry:
    """Session registry with TTL reaping (reference registry.rs; reaper wired
    in server.rs:762)."""

    def __init__(self, session_ttl: float = 3600.0, pending_ttl: float = 30.0):
        self.sessions: Dict[str, dict] = {}
        self.session_ttl = session_ttl
        self.pending_ttl = pending_ttl

    def create_session(self, model: Optional[str], config: Optional[dict] = None) -> dict:
        sid = f"sess_{uuid.uuid4().hex}"
        sess = {
            "id": sid,
            "object": "realtime.session",
            "model": model,
            "client_secret": {
                "value": f"ek_{uuid.uuid4().hex}",
                "expires_at": int(time.time() + self.pending_ttl),
            },
            "created_at": int(time.time()),
            "_expires": time.monotonic() + self.session_ttl,
            "config": config or {},
            "state": "pending",
        }
        self.sessions[sid] = sess
        return sess

    def reap(self) -> int:
        now = time.monotonic()
        stale = [k for k, s in self.sessions.items() if s["_expires"] < now]
        for k in stale:
            del self.sessions[k]
        return len(stale)


async def v1_realtime_session(request: web.Request):
    from ..server.app import CTX_KEY

    ctx = request.app[CTX_KEY]
    if not hasattr(ctx, "realtime_registry"):
        ctx.realtime_registry = RealtimeRegistry()
    body = {}
    try:
        raw = await request.read()
        if raw:
            body = json.loads(raw)
    except json.JSONDecodeError:
        pass
    sess = ctx.realtime_registry.create_session(body.get("model"), body)
    out = {k: v for k, v in sess.items() if not k.startswith("_")}
    return web.json_response(out)


async def v1_realtime_ws(request: web.Request):
    """Bidirectional WS relay: client <-> gateway <-> worker."""
    from ..server.app import CTX_KEY

    ctx = request.app[CTX_KEY]
    model = request.query.get("model")
    workers = ctx.worker_registry.for_model(ctx.worker_registry.resolve_model(model))
    if not workers:
        return web.Response(status=503, body=error_body("no realtime worker available", 503),
                            content_type="application/json")
    policy = ctx.policy_registry.get(model)
    idx = policy.select_worker(workers, SelectWorkerInfo(model_id=model, routing_key=request.headers.get("x-smg-routing-key")))
    if idx is None:
        return web.Response(status=503, body=error_body("selection failed", 503), content_type="application/json")
    worker = workers[idx]

    ws_client = web.WebSocketResponse(heartbeat=30)
    await ws_client.prepare(request)
    worker.incr_load()
    try:
        url = worker.url.replace("http://", "ws://").replace("https://", "wss://") + "/v1/realtime"
        if model:
            url += f"?model={model}"
        async with aiohttp.ClientSession() as session:
            async with session.ws_connect(url) as ws_worker:
                async def pump(src, dst):
                    async for msg in src:
                        if msg.type == WSMsgType.TEXT:
                            await dst.send_str(msg.data)
                        elif msg.type == WSMsgType.BINARY:
                            await dst.send_bytes(msg.data)
                        elif msg.type in (WSMsgType.CLOSE, WSMsgType.ERROR):
                            break
                    await dst.close()

                await asyncio.gather(pump(ws_client, ws_worker), pump(ws_worker, ws_client))
        worker.record_outcome(True)
    except Exception:
        worker.record_outcome(False)
        if not ws_client.closed:
            await ws_client.close()
    finally:
        worker.decr_load()
    return ws_client


async def v1_realtime_calls(request: web.Request):
    """WebRTC call setup (reference realtime/webrtc_bridge.rs — str0m-based
    media relay).  This build ships the WebSocket relay only; WebRTC needs a
    media stack (ICE/DTLS/SRTP) that is out of scope here, so the endpoint
    exists and says so instead of 404ing."""
    from ..protocols.openai import error_body

    return web.Response(
        status=501,
        body=error_body(
            "WebRTC realtime calls are not supported by this build; connect via"
            " the WebSocket endpoint GET /v1/realtime instead",
            501,
        ),
        content_type="application/json",
    )


def add_realtime_routes(app: web.Application) -> None:
    app.router.add_get("/v1/realtime", v1_realtime_ws)
    app.router.add_post("/v1/realtime/sessions", v1_realtime_session)
    app.router.add_post("/v1/realtime/client_secrets", v1_realtime_session)
    app.router.add_post("/v1/realtime/transcription_sessions", v1_realtime_session)
    app.router.add_post("/v1/realtime/calls", v1_realtime_calls)
