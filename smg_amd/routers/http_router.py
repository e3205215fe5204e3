"""HTTP pass-through router (reference: model_gateway/src/routers/http/router.rs —
route_typed_request :316, route_typed_request_once :419, select_worker_for_model :195).

Flow per request: parse view -> alias resolve -> retry loop { policy select ->
load guard -> proxy to worker (aiohttp or in-process sim transport) -> record
circuit-breaker outcome } -> stream SSE bytes back verbatim.

Transports:
  * http(s)://  — aiohttp client with connection pooling;
  * sim://name  — direct in-process call into a mock/GPU engine object
    registered on the worker (worker.extra["engine"]), used by tests and the
    routing benchmark (no socket hop, mirrors the reference's in-process
    test_app harness).
"""
from __future__ import annotations

import asyncio
import json
import logging
import time
import uuid
from typing import AsyncIterator, Dict, List, Optional

import aiohttp

from ..config import RouterConfig
from ..observability.metrics import GatewayMetrics
from ..policies import PolicyRegistry, SelectWorkerInfo
from ..protocols.openai import ProtocolError, error_body, parse_request
from ..workers.registry import WorkerRegistry
from ..workers.worker import Worker
from .base import RouteRequest, RouteResponse, Router
from .retry import RetryExecutor, is_retryable_status

log = logging.getLogger("smg.router.http")

HOP_HEADERS = {
    "host",
    "content-length",
    "connection",
    "keep-alive",
    "transfer-encoding",
    "te",
    "upgrade",
    "proxy-authorization",
    "proxy-authenticate",
}


class HttpRouter(Router):
    router_id = "http-regular"

    def __init__(
        self,
        registry: WorkerRegistry,
        policies: PolicyRegistry,
        config: Optional[RouterConfig] = None,
        session: Optional[aiohttp.ClientSession] = None,
        metrics: Optional[GatewayMetrics] = None,
    ):
        self.registry = registry
        self.policies = policies
        self.config = config or RouterConfig()
        self._session = session
        self.metrics = metrics or GatewayMetrics.null()
        self.retry_executor = RetryExecutor(self.config.retry)
        self._model_retry_cache: dict = {}

    async def session(self) -> aiohttp.ClientSession:
        if self._session is None or self._session.closed:
            self._session = aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=self.config.request_timeout_secs, sock_connect=10),
                connector=aiohttp.TCPConnector(limit=0, ttl_dns_cache=300),
            )
        return self._session

    async def shutdown(self) -> None:
        if self._session is not None and not self._session.closed:
            await self._session.close()

    # ---- selection -------------------------------------------------------
    def select_worker(self, req: RouteRequest, model_id: Optional[str], info: SelectWorkerInfo) -> Optional[Worker]:
        workers = self.registry.for_model(model_id)
        if not workers:
            return None
        policy = self.policies.get(model_id)
        t0 = time.perf_counter_ns()
        idx = policy.select_worker(workers, info)
        self.metrics.observe_routing_latency((time.perf_counter_ns() - t0) / 1e9)
        if idx is None:
            return None
        return workers[idx]

    # ---- entry -----------------------------------------------------------
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
        tried: List[str] = []

        async def attempt(attempt_no: int) -> RouteResponse:
            worker = self.select_worker(req, model_id, info)
            if worker is None:
                self.metrics.count_no_worker(req.path)
                return RouteResponse(
                    status=503,
                    body=error_body(
                        f"no available worker for model {model_id or 'any'}", 503, "service_unavailable"
                    ),
                )
            tried.append(worker.url)
            return await self._dispatch(worker, req, info)

        def should_retry(resp: RouteResponse) -> bool:
            return not resp.is_stream and is_retryable_status(resp.status)

        def on_retry(attempt_no: int, resp) -> None:
            self.metrics.count_retry(req.path)

        resp = await self._retry_for(model_id).execute(attempt, should_retry, on_retry)
        return resp

    def _retry_for(self, model_id):
        """Per-model retry override (reference registry.rs model_retry_configs
        consulted by route_typed_request)."""
        if not model_id:
            return self.retry_executor
        cfg = self.registry.get_model_retry_config(model_id)
        if cfg is None:
            return self.retry_executor
        ex = self._model_retry_cache.get(model_id)
        if ex is None or ex.config is not cfg:
            ex = RetryExecutor(cfg)
            self._model_retry_cache[model_id] = ex
        return ex

    # ---- dispatch --------------------------------------------------------
    async def _dispatch(self, worker: Worker, req: RouteRequest, info: SelectWorkerInfo) -> RouteResponse:
        worker.incr_load(info.est_tokens)
        released = False

        def release(success: bool) -> None:
            nonlocal released
            if released:
                return
            released = True
            worker.decr_load(info.est_tokens)
            worker.record_outcome(success)
            policy = self.policies.get(info.model_id)
            policy.on_request_complete(worker, info, success)

        try:
            engine = worker.extra.get("engine")
            provider = worker.labels.get("provider")
            if engine is not None:
                resp = await self._dispatch_sim(engine, worker, req)
            elif provider in ("anthropic", "gemini") and req.path == "/v1/chat/completions" and not (req.body or {}).get("stream"):
                # vendor translation (reference openai/provider registry)
                from .providers import dispatch_to_provider

                body = await dispatch_to_provider(await self.session(), worker, req.body or {})
                resp = RouteResponse(status=200, body=json.dumps(body).encode())
            else:
                resp = await self._dispatch_http(worker, req)
        except asyncio.CancelledError:
            release(False)
            raise
        except Exception as exc:  # transport failure
            release(False)
            log.warning("dispatch to %s failed: %s", worker.url, exc)
            self.metrics.count_worker_error(worker.url)
            return RouteResponse(status=502, body=error_body(f"upstream error: {exc}", 502, "bad_gateway"))

        success = 200 <= resp.status < 500  # 4xx = client error, not worker fault
        if resp.is_stream:
            resp.stream = self._guarded_stream(resp.stream, release, success)
        else:
            release(success)
        if resp.status >= 500:
            self.metrics.count_worker_error(worker.url)
        return resp

    async def _guarded_stream(self, inner: AsyncIterator[bytes], release, success: bool) -> AsyncIterator[bytes]:
        ok = success
        try:
            async for chunk in inner:
                yield chunk
        except Exception:
            ok = False
            raise
        finally:
            release(ok)

    async def _dispatch_sim(self, engine, worker: Worker, req: RouteRequest) -> RouteResponse:
        status, headers, payload = await engine.handle(req.path, req.body, req.headers)
        if hasattr(payload, "__aiter__"):
            return RouteResponse(status=status, headers=headers, stream=payload)
        return RouteResponse(status=status, headers=headers, body=payload)

    async def _dispatch_http(self, worker: Worker, req: RouteRequest) -> RouteResponse:
        session = await self.session()
        headers = {k: v for k, v in req.headers.items() if k.lower() not in HOP_HEADERS}
        headers["content-type"] = "application/json"
        if req.request_id:
            headers["x-request-id"] = req.request_id
        # W3C trace propagation to the worker (reference inject_trace_context_http)
        from ..observability.tracing import inject_trace_context

        inject_trace_context(headers)
        if worker.api_key:
            headers["authorization"] = f"Bearer {worker.api_key}"
        url = worker.url + req.path
        resp = await session.request(req.method, url, data=req.raw_body or None, headers=headers)
        out_headers = {
            k: v
            for k, v in resp.headers.items()
            if k.lower() in ("content-type", "x-request-id", "cache-control")
        }
        ctype = resp.headers.get("content-type", "")
        if "text/event-stream" in ctype or resp.headers.get("transfer-encoding") == "chunked" and "json" not in ctype:
            return RouteResponse(status=resp.status, headers=out_headers, stream=self._body_stream(resp))
        body = await resp.read()
        resp.release()
        return RouteResponse(status=resp.status, headers=out_headers, body=body)

    async def _body_stream(self, resp: aiohttp.ClientResponse) -> AsyncIterator[bytes]:
        try:
            async for chunk in resp.content.iter_any():
                yield chunk
        finally:
            resp.release()

    # ---- aux -------------------------------------------------------------
    async def get_loads(self) -> Dict[str, object]:
        return {"workers": [w.to_dict() for w in self.registry.all()]}

    async def flush_cache(self) -> RouteResponse:
        for policy in self.policies.all_policies():
            policy.reset()
        return RouteResponse(status=200, body=b'{"status":"cache flushed"}')
