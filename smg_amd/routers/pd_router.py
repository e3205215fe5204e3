"""Prefill/decode disaggregation router over HTTP (reference:
model_gateway/src/routers/http/pd_router.rs — bootstrap injection :217-266,
random room id pd_types.rs:15, concurrent dual dispatch, stream from decode).

Selects a (prefill, decode) pair with the per-leg policies, injects
bootstrap_host/bootstrap_port/bootstrap_room into the JSON body, dispatches
prefill and decode concurrently, and streams the decode response back.  The
KV handoff itself is engine-side (Mooncake/NIXL in the reference; xGMI
peer-copy for the RCCL data plane).
"""
from __future__ import annotations

import asyncio
import json
import logging
import random
import uuid
from urllib.parse import urlparse

from ..policies import SelectWorkerInfo
from ..protocols.openai import ProtocolError, error_body, parse_request
from ..workers.worker import WorkerType
from .base import RouteRequest, RouteResponse
from .http_router import HttpRouter

log = logging.getLogger("smg.router.pd")


class PDRouter(HttpRouter):
    router_id = "http-pd"

    async def route(self, req: RouteRequest) -> RouteResponse:
        try:
            view = parse_request(req.path, req.body if req.body is not None else {})
        except ProtocolError as e:
            return RouteResponse(status=e.code, body=error_body(str(e), e.code, e.err_type))

        model_id = self.registry.resolve_model(req.model_override or view.model)
        info = SelectWorkerInfo(
            request_id=req.request_id or uuid.uuid4().hex,
            model_id=model_id,
            text=view.routing_text() or None,
            routing_key=req.routing_key,
            tenant_id=req.tenant_id,
            est_tokens=view.est_prompt_tokens(),
        )

        prefill_pool = self.registry.for_model(model_id, worker_type=WorkerType.PREFILL)
        decode_pool = self.registry.for_model(model_id, worker_type=WorkerType.DECODE)
        if not prefill_pool or not decode_pool:
            return RouteResponse(
                status=503, body=error_body("no prefill/decode workers available", 503, "service_unavailable")
            )
        p_idx = self.policies.get(model_id, "prefill").select_worker(prefill_pool, info)
        d_idx = self.policies.get(model_id, "decode").select_worker(decode_pool, info)
        if p_idx is None or d_idx is None:
            return RouteResponse(status=503, body=error_body("selection failed", 503, "service_unavailable"))
        prefill, decode = prefill_pool[p_idx], decode_pool[d_idx]

        # bootstrap metadata injection (reference pd_router.rs:217-266)
        host = prefill.bootstrap_host or urlparse(prefill.url).hostname or "127.0.0.1"
        room = random.getrandbits(63)
        body = dict(req.body or {})
        body["bootstrap_host"] = host
        body["bootstrap_port"] = prefill.bootstrap_port
        body["bootstrap_room"] = room
        raw = json.dumps(body).encode()
        pd_req = RouteRequest(
            path=req.path,
            method=req.method,
            body=body,
            raw_body=raw,
            headers=req.headers,
            request_id=info.request_id,
            tenant_id=req.tenant_id,
        )

        prefill.incr_load(info.est_tokens)
        decode.incr_load(info.est_tokens)
        try:
            prefill_task = asyncio.ensure_future(self._leg(prefill, pd_req))
            decode_resp = await self._leg(decode, pd_req)
            prefill_resp = await prefill_task
            ok = decode_resp.status < 500
            prefill.record_outcome(prefill_resp.status < 500)
            if decode_resp.is_stream:
                decode_resp.stream = self._guarded_stream(
                    decode_resp.stream, lambda s: (decode.decr_load(info.est_tokens), decode.record_outcome(s)), ok
                )
                prefill.decr_load(info.est_tokens)
            else:
                decode.record_outcome(ok)
                decode.decr_load(info.est_tokens)
                prefill.decr_load(info.est_tokens)
            return decode_resp
        except Exception as exc:
            prefill.decr_load(info.est_tokens)
            decode.decr_load(info.est_tokens)
            prefill.record_outcome(False)
            decode.record_outcome(False)
            return RouteResponse(status=502, body=error_body(f"pd dispatch error: {exc}", 502, "bad_gateway"))

    async def _leg(self, worker, req: RouteRequest) -> RouteResponse:
        engine = worker.extra.get("engine")
        if engine is not None:
            return await self._dispatch_sim(engine, worker, req)
        return await self._dispatch_http(worker, req)
