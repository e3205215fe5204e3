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
        self.calls: Dict[str, "CallSession"] = {}
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
                from ..protocols.realtime_events import RealtimeEventError, parse_event

                async def pump(src, dst, validate_client=False):
                    async for msg in src:
                        if msg.type == WSMsgType.TEXT:
                            if validate_client:
                                # typed event validation (reference
                                # realtime_events.rs ClientEvent): malformed
                                # events answer an `error` event, not a relay
                                try:
                                    ev = json.loads(msg.data)
                                    parse_event(ev, "client")
                                except (json.JSONDecodeError, RealtimeEventError) as exc:
                                    await src.send_str(json.dumps({
                                        "type": "error",
                                        "error": {"type": "invalid_request_error",
                                                  "message": str(exc)},
                                    }))
                                    continue
                            await dst.send_str(msg.data)
                        elif msg.type == WSMsgType.BINARY:
                            await dst.send_bytes(msg.data)
                        elif msg.type in (WSMsgType.CLOSE, WSMsgType.ERROR):
                            break
                    await dst.close()

                await asyncio.gather(pump(ws_client, ws_worker, validate_client=True),
                                     pump(ws_worker, ws_client))
        worker.record_outcome(True)
    except Exception:
        worker.record_outcome(False)
        if not ws_client.closed:
            await ws_client.close()
    finally:
        worker.decr_load()
    return ws_client


def _validate_sdp_offer(sdp: str) -> Optional[str]:
    """Minimal SDP sanity (webrtc_bridge.rs parses with str0m; we validate
    the signaling invariants): session header, at least one media section,
    ICE credentials for trickle."""
    if not sdp.startswith("v=0"):
        return "SDP must start with v=0"
    if "\nm=" not in sdp and "\rm=" not in sdp:
        return "SDP offer has no media section"
    if "a=ice-ufrag:" not in sdp:
        return "SDP offer has no ICE credentials"
    return None


class CallSession:
    __slots__ = ("call_id", "model", "state", "created", "upstream_url", "upstream_call_id")

    def __init__(self, call_id, model, upstream_url, upstream_call_id=None):
        self.call_id = call_id
        self.model = model
        self.state = "active"
        self.created = time.time()
        self.upstream_url = upstream_url
        self.upstream_call_id = upstream_call_id


async def v1_realtime_calls(request: web.Request):
    """WebRTC call signaling (reference webrtc_bridge.rs:147-270 — SDP
    offer/answer brokering; server.rs:855 route).  The gateway validates the
    client's SDP offer and brokers it to a WebRTC-capable upstream (a worker
    labeled webrtc=true, which answers /v1/realtime/calls itself); the SDP
    answer and call id relay back and the call is tracked for GET/DELETE
    management.  Media flows peer-to-peer between client and upstream — the
    in-process str0m-style media RELAY of the reference needs a media stack
    this image does not carry, so relay mode stays unimplemented (501 when no
    upstream can take the call)."""
    from ..server.app import CTX_KEY

    ctx = request.app[CTX_KEY]
    if not hasattr(ctx, "realtime_registry"):
        ctx.realtime_registry = RealtimeRegistry()
    sdp = (await request.read()).decode("utf-8", "replace")
    err = _validate_sdp_offer(sdp)
    if err:
        return web.Response(status=400, body=error_body(f"invalid SDP offer: {err}"),
                            content_type="application/json")
    model = request.query.get("model") or "default"
    workers = [w for w in ctx.worker_registry.all() if w.labels.get("webrtc") == "true"]
    if not workers:
        return web.Response(
            status=501,
            body=error_body(
                "no WebRTC-capable upstream registered (label webrtc=true); this build "
                "brokers SDP signaling but carries no media stack for relay mode — "
                "connect via WebSocket /v1/realtime instead", 501),
            content_type="application/json")
    worker = workers[0]
    import aiohttp as _aiohttp

    session = getattr(ctx, "client_session", None)
    close_session = False
    if session is None:
        session = _aiohttp.ClientSession()
        close_session = True
    try:
        async with session.post(
            f"{worker.url}/v1/realtime/calls", data=sdp.encode(),
            params={"model": model}, headers={"Content-Type": "application/sdp"},
            timeout=_aiohttp.ClientTimeout(total=15),
        ) as resp:
            answer = await resp.text()
            if resp.status // 100 != 2:
                return web.Response(status=502, body=error_body(
                    f"upstream signaling failed: HTTP {resp.status}", 502),
                    content_type="application/json")
            upstream_call = resp.headers.get("Location", "").rsplit("/", 1)[-1] or None
    finally:
        if close_session:
            await session.close()
    call_id = "rtc_" + uuid.uuid4().hex[:20]
    reg = ctx.realtime_registry
    reg.calls[call_id] = CallSession(call_id, model, worker.url, upstream_call)
    return web.Response(
        status=201, body=answer.encode(),
        headers={"Location": f"/v1/realtime/calls/{call_id}",
                 "Content-Type": "application/sdp"})


async def v1_realtime_call_get(request: web.Request):
    from ..server.app import CTX_KEY

    ctx = request.app[CTX_KEY]
    if not hasattr(ctx, "realtime_registry"):
        ctx.realtime_registry = RealtimeRegistry()
    reg = ctx.realtime_registry
    call = reg.calls.get(request.match_info["call_id"])
    if call is None:
        return web.Response(status=404, body=error_body("call not found", 404),
                            content_type="application/json")
    return web.json_response({
        "id": call.call_id, "model": call.model, "state": call.state,
        "created": call.created, "upstream": call.upstream_url,
    })


async def v1_realtime_call_hangup(request: web.Request):
    from ..server.app import CTX_KEY

    ctx = request.app[CTX_KEY]
    if not hasattr(ctx, "realtime_registry"):
        ctx.realtime_registry = RealtimeRegistry()
    reg = ctx.realtime_registry
    call = reg.calls.pop(request.match_info["call_id"], None)
    if call is None:
        return web.Response(status=404, body=error_body("call not found", 404),
                            content_type="application/json")
    call.state = "ended"
    if call.upstream_call_id:
        try:
            import aiohttp as _aiohttp

            session = getattr(ctx, "client_session", None)
            if session is not None:
                await session.delete(
                    f"{call.upstream_url}/v1/realtime/calls/{call.upstream_call_id}",
                    timeout=_aiohttp.ClientTimeout(total=5))
        except Exception:
            pass
    return web.json_response({"id": call.call_id, "state": "ended"})


async def _v1_realtime_calls_legacy(request: web.Request):
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
    app.router.add_get("/v1/realtime/calls/{call_id}", v1_realtime_call_get)
    app.router.add_delete("/v1/realtime/calls/{call_id}", v1_realtime_call_hangup)
