"""Gateway HTTP server (reference: model_gateway/src/server.rs — build_app :751,
startup :984, route table :781-935).

aiohttp.web application (C-accelerated HTTP parsing; uvicorn here has no
httptools/uvloop) with the reference's route groups:
  protected (inference), public (health/model info), admin (cache/profiling/
  parsers/tokenizers), worker CRUD.  Middleware outside-in: request-id ->
  metrics -> auth -> tenant resolution -> admission (concurrency limiter).
"""
from __future__ import annotations

import asyncio
import json
import logging
import time
import uuid
from typing import Optional

from aiohttp import web

from ..config import RouterConfig
from ..protocols.openai import error_body
from ..routers.base import RouteRequest
from ..routers.factory import RouterManager
from .app_context import AppContext

log = logging.getLogger("smg.server")

CTX_KEY = web.AppKey("smg_ctx", AppContext)


# --------------------------------------------------------------------------
# middleware
# --------------------------------------------------------------------------
@web.middleware
async def request_id_middleware(request: web.Request, handler):
    ctx: AppContext = request.app[CTX_KEY]
    rid = None
    for h in ctx.config.request_id_headers:
        rid = request.headers.get(h)
        if rid:
            break
    request["request_id"] = rid or uuid.uuid4().hex
    resp = await handler(request)
    try:
        resp.headers["x-request-id"] = request["request_id"]
    except Exception:
        pass
    return resp


@web.middleware
async def metrics_middleware(request: web.Request, handler):
    ctx: AppContext = request.app[CTX_KEY]
    t0 = time.perf_counter()
    try:
        resp = await handler(request)
        return resp
    finally:
        status = getattr(locals().get("resp"), "status", 500)
        ctx.metrics.observe_http(request.path, request.method, status, time.perf_counter() - t0)


@web.middleware
async def plugin_middleware(request: web.Request, handler):
    """WASM-hook equivalent: OnRequest / OnResponse phases
    (reference middleware/wasm.rs)."""
    ctx: AppContext = request.app[CTX_KEY]
    plugins = getattr(ctx, "plugins", None)
    if plugins is None or len(plugins) == 0 or request.path in PUBLIC_PATHS:
        return await handler(request)
    from ..plugins import ShortCircuit

    pctx = {
        "path": request.path,
        "method": request.method,
        "headers": dict(request.headers),
        "request_id": request.get("request_id"),
        "tenant_id": request.get("tenant_id"),
    }
    try:
        plugins.run_phase("on_request", pctx)
    except ShortCircuit as sc:
        return web.Response(status=sc.status, body=sc.body, headers=sc.headers,
                            content_type="application/json")
    if pctx.get("tenant_id"):
        request["tenant_id"] = pctx["tenant_id"]
    resp = await handler(request)
    try:
        out = {"status": resp.status, "headers": dict(resp.headers), "path": request.path}
        plugins.run_phase("on_response", out)
        for k, v in out.get("extra_headers", {}).items():
            resp.headers[k] = v
    except ShortCircuit as sc:
        return web.Response(status=sc.status, body=sc.body, content_type="application/json")
    except Exception:
        pass
    return resp


def _jwt_enabled(auth) -> bool:
    return bool(auth.jwt_jwks_uri or auth.jwt_jwks_inline)


def _get_jwt_validator(ctx: "AppContext"):
    """Lazily build the JwtValidator from AuthConfig (reference
    crates/auth/src/jwt.rs:235-258 from_config)."""
    v = getattr(ctx, "jwt_validator", None)
    if v is None:
        from .jwt_auth import JwksProvider, JwtValidator

        auth = ctx.config.auth
        jwks = JwksProvider(auth.jwt_jwks_inline or auth.jwt_jwks_uri, ttl_secs=auth.jwt_jwks_cache_ttl_secs)
        v = JwtValidator(
            jwks,
            issuer=auth.jwt_issuer,
            audience=auth.jwt_audience,
            leeway_secs=auth.jwt_leeway_secs,
            role_claim=auth.jwt_role_claim,
            role_mapping=auth.jwt_role_mapping,
            enable_jti_check=auth.jwt_enable_jti_check,
            require_exp=auth.jwt_require_exp,
        )
        ctx.jwt_validator = v
    return v


def _is_admin_mutation(request: web.Request) -> bool:
    """Control-plane mutations that require an admin identity once auth is
    configured (reference ControlPlaneAuthState + Role::is_admin gating)."""
    if request.method in ("GET", "HEAD", "OPTIONS"):
        return False
    p = request.path
    return (
        p in ("/flush_cache", "/start_profile", "/stop_profile", "/workers", "/wasm")
        or p.startswith("/workers/")
        or p.startswith("/wasm/")
        or p.startswith("/tokenizers")
        or p.startswith("/mcp/servers")
    )


def _admin_forbidden() -> web.Response:
    return web.Response(
        status=403, body=error_body("admin role required", 403, "permission_error"),
        content_type="application/json",
    )


@web.middleware
async def auth_middleware(request: web.Request, handler):
    ctx: AppContext = request.app[CTX_KEY]
    auth = ctx.config.auth
    if request.path in PUBLIC_PATHS or (
        not auth.api_key and not auth.tenant_api_keys
        and not auth.control_plane_api_keys and not _jwt_enabled(auth)
    ):
        return await handler(request)
    header = request.headers.get("authorization", "")
    token = header[7:] if header.lower().startswith("bearer ") else request.headers.get("x-api-key", "")
    if auth.api_key and _ct_eq(token, auth.api_key):
        request["auth_role"] = "admin"  # the master key is the operator key
        return await handler(request)
    if any(_ct_eq(token, k) for k in auth.control_plane_api_keys):
        request["auth_role"] = "admin"
        return await handler(request)
    tenant = auth.tenant_api_keys.get(token)
    if tenant is not None:
        request["tenant_id"] = tenant
        request["auth_role"] = "user"
        if _is_admin_mutation(request):
            return _admin_forbidden()
        return await handler(request)
    if _jwt_enabled(auth) and token.count(".") == 2:
        from .jwt_auth import JwtError

        try:
            validated = await _get_jwt_validator(ctx).validate(token, session=getattr(ctx, "client_session", None))
        except JwtError as e:
            return web.Response(
                status=401, body=error_body(f"invalid token: {e}", 401, "authentication_error"),
                content_type="application/json",
            )
        request["auth_user"] = validated.subject
        request["auth_role"] = validated.role
        if validated.claims.get("tenant_id"):
            request["tenant_id"] = validated.claims["tenant_id"]
        if _is_admin_mutation(request) and validated.role != auth.admin_role:
            return _admin_forbidden()
        return await handler(request)
    return web.Response(
        status=401, body=error_body("invalid API key", 401, "authentication_error"),
        content_type="application/json",
    )


def _ct_eq(a: str, b: str) -> bool:
    import hmac

    return hmac.compare_digest(a.encode(), b.encode())


@web.middleware
async def tracing_middleware(request: web.Request, handler):
    """Span per request, continuing an incoming W3C traceparent; the
    contextvar makes the span visible to routers for outbound propagation
    (reference otel_trace.rs + http/router.rs:466)."""
    ctx: AppContext = request.app[CTX_KEY]
    tracer = getattr(ctx, "tracer", None)
    if tracer is None or not tracer.enabled:
        return await handler(request)
    with tracer.span(
        f"http {request.path}",
        traceparent=request.headers.get("traceparent"),
        method=request.method,
        path=request.path,
        request_id=request.get("request_id", ""),
    ) as span:
        resp = await handler(request)
        span.set("status", getattr(resp, "status", 0))
        return resp


@web.middleware
async def tenant_middleware(request: web.Request, handler):
    ctx: AppContext = request.app[CTX_KEY]
    if "tenant_id" not in request and ctx.config.trust_tenant_header:
        t = request.headers.get(ctx.config.tenant_header_name)
        if t:
            request["tenant_id"] = t
    return await handler(request)


@web.middleware
async def admission_middleware(request: web.Request, handler):
    """Concurrency limiter + bounded queue (reference middleware/concurrency.rs);
    replaced by the priority scheduler when enabled (middleware/scheduler/)."""
    ctx: AppContext = request.app[CTX_KEY]
    if request.path in PUBLIC_PATHS or request.path in ADMIN_PATHS:
        return await handler(request)
    # per-tenant rate limit (reference rate_limit/manager.rs reserve -> settle)
    trl = getattr(ctx, "tenant_rate_limiter", None)
    reservation = None
    if trl is not None:
        tenant = request.get("tenant_id") or "default"
        reservation = trl.reserve(tenant, est_tokens=max(0, request.content_length or 0) // 4)
        if reservation is None:
            ctx.metrics.rate_limited.labels(tenant).inc() if not ctx.metrics._null else None
            return web.Response(
                status=429,
                body=error_body(f"tenant {tenant} rate limit exceeded", 429, "rate_limit_error"),
                content_type="application/json",
            )
    try:
        if ctx.scheduler is not None:
            return await ctx.scheduler.admit(request, handler)
        limiter = ctx.rate_limiter
        if limiter is None:
            return await handler(request)
        return await limiter.admit(request, handler)
    finally:
        if reservation is not None:
            trl.settle(reservation)


PUBLIC_PATHS = {
    "/liveness",
    "/readiness",
    "/health",
    "/health_generate",
    "/engine_metrics",
    "/v1/models",
    "/get_model_info",
    "/get_server_info",
    "/metrics",
    "/openapi.json",
}
ADMIN_PATHS = {
    "/flush_cache",
    "/start_profile",
    "/stop_profile",
    "/get_loads",
    "/parse/function_call",
    "/parse/reasoning",
    "/workers",
}


# --------------------------------------------------------------------------
# handlers
# --------------------------------------------------------------------------
async def _read_json(request: web.Request) -> Optional[dict]:
    raw = await request.read()
    if not raw:
        return None, b""
    try:
        return json.loads(raw), raw
    except json.JSONDecodeError:
        raise web.HTTPBadRequest(body=error_body("invalid JSON body"), content_type="application/json")


async def _proxy_endpoint(request: web.Request) -> web.StreamResponse:
    """Shared handler for all inference endpoints: parse -> RouterManager ->
    stream or unary response."""
    ctx: AppContext = request.app[CTX_KEY]
    body, raw = await _read_json(request)
    route_req = RouteRequest(
        path=request.path,
        method=request.method,
        body=body,
        raw_body=raw,
        headers=dict(request.headers),
        request_id=request.get("request_id", ""),
        tenant_id=request.get("tenant_id"),
        routing_key=request.headers.get("x-smg-routing-key") or (ctx.config.routing_key_override and request.headers.get(ctx.config.routing_key_override)),
    )
    ctx.inflight += 1
    try:
        resp = await ctx.router_manager.route(route_req)
        if resp.is_stream:
            out = web.StreamResponse(status=resp.status, headers={"content-type": "text/event-stream", **resp.headers})
            out.enable_chunked_encoding()
            await out.prepare(request)
            try:
                async for chunk in resp.stream:
                    await out.write(chunk)
            except (ConnectionResetError, asyncio.CancelledError):
                pass
            await out.write_eof()
            return out
        hdrs = dict(resp.headers)
        ctype = hdrs.pop("content-type", "application/json").split(";")[0]
        return web.Response(status=resp.status, body=resp.body, headers=hdrs, content_type=ctype)
    finally:
        ctx.inflight -= 1


async def liveness(request):
    return web.json_response({"status": "alive"})


async def readiness(request):
    ctx: AppContext = request.app[CTX_KEY]
    healthy = ctx.worker_registry.healthy_count()
    status = 200 if healthy > 0 or len(ctx.worker_registry) == 0 else 503
    return web.json_response({"status": "ready" if status == 200 else "not ready", "healthy_workers": healthy}, status=status)


async def health(request):
    return web.json_response({"status": "healthy"})


async def v1_models(request):
    ctx: AppContext = request.app[CTX_KEY]
    models = ctx.worker_registry.models()
    data = [{"id": m, "object": "model", "created": 0, "owned_by": "smg"} for m in models]
    return web.json_response({"object": "list", "data": data})


async def get_model_info(request):
    ctx: AppContext = request.app[CTX_KEY]
    models = ctx.worker_registry.models()
    return web.json_response({"model_path": models[0] if models else None, "is_generation": True})


async def get_server_info(request):
    ctx: AppContext = request.app[CTX_KEY]
    from .. import __version__

    return web.json_response(
        {
            "version": __version__,
            "workers": [w.to_dict() for w in ctx.worker_registry.all()],
            "policy": ctx.config.policy.name,
            "mode": ctx.config.mode.value,
            "connection_mode": ctx.config.connection_mode.value,
        }
    )


async def metrics_endpoint(request):
    ctx: AppContext = request.app[CTX_KEY]
    _refresh_scrape_gauges(ctx)
    return web.Response(body=ctx.metrics.export(), content_type="text/plain")


def _refresh_scrape_gauges(ctx: "AppContext") -> None:
    """Just-in-time gauges computed at scrape (mesh partition state, worker
    pool sizes, GPU-tree occupancy, tokenizer registry) — state the exporter
    reads rather than events that push."""
    m = ctx.metrics
    if m._null:
        return
    try:
        by_model = {}
        for w in ctx.worker_registry.all():
            by_model[w.model_id] = by_model.get(w.model_id, 0) + 1
            m.worker_requests_active.labels(w.url).set(w.active_requests)
        for model, n in by_model.items():
            m.worker_pool_size.labels(model).set(n)
        if ctx.mesh is not None:
            st = ctx.mesh.partition_state()
            m.mesh_partitioned.set(1 if st["partitioned"] else 0)
            m.mesh_peers.set(len(ctx.mesh.live_members()))
        if ctx.tokenizer_registry is not None:
            m.tokenizers_registered.set(len(ctx.tokenizer_registry.list()))
        for policy in ctx.policy_registry.all_policies():
            trees = getattr(policy, "token_trees", None)
            if not trees:
                continue
            for model_id, tree in list(trees.items())[:16]:
                stats = getattr(tree, "stats", None)
                if callable(stats):
                    try:
                        st = stats()
                        if isinstance(st, dict) and "live_nodes" in st:
                            m.gpu_tree_nodes_live.labels(model_id).set(int(st["live_nodes"]))
                            m.gpu_tree_nodes_allocated.labels(model_id).set(
                                int(st.get("allocated_nodes", 0)))
                    except Exception:
                        pass
                elif hasattr(tree, "__len__"):
                    m.tree_size.labels(model_id).set(len(tree))
    except Exception:  # scrape must never fail the endpoint
        pass


async def engine_metrics(request):
    """Merged Prometheus exposition across the worker fleet (reference
    worker/metrics_aggregator.rs via /engine_metrics); `?format=json` keeps
    the compact summary."""
    ctx: AppContext = request.app[CTX_KEY]
    if request.query.get("format") == "json":
        out = {}
        for w in ctx.worker_registry.all():
            out[w.url] = {"token_usage": w.token_usage, "gen_throughput": w.gen_throughput, "load": w.active_requests}
        return web.json_response(out)
    from ..observability.aggregate import collect_engine_metrics

    text = await collect_engine_metrics(ctx, session=getattr(ctx, "client_session", None))
    return web.Response(body=text, content_type="text/plain")


async def get_loads(request):
    ctx: AppContext = request.app[CTX_KEY]
    loads = await ctx.router_manager.default_router.get_loads()
    return web.json_response(loads)


async def flush_cache(request):
    ctx: AppContext = request.app[CTX_KEY]
    resp = await ctx.router_manager.default_router.flush_cache()
    return web.Response(status=resp.status, body=resp.body, content_type="application/json")


async def parse_function_call(request):
    ctx: AppContext = request.app[CTX_KEY]
    body, _ = await _read_json(request)
    from ..parsers.tool import parse_complete

    parser = (body or {}).get("tool_call_parser") or ctx.config.tool_call_parser or "json"
    text = (body or {}).get("text", "")
    tools = (body or {}).get("tools")
    normal, calls = parse_complete(parser, text, tools)
    return web.json_response({"normal_text": normal, "calls": calls})


async def parse_reasoning(request):
    ctx: AppContext = request.app[CTX_KEY]
    body, _ = await _read_json(request)
    from ..parsers.reasoning import parse_reasoning_complete

    parser = (body or {}).get("reasoning_parser") or ctx.config.reasoning_parser or "deepseek_r1"
    text = (body or {}).get("text", "")
    reasoning, normal = parse_reasoning_complete(parser, text)
    return web.json_response({"reasoning_text": reasoning, "text": normal})


# ---- worker CRUD (reference worker/service.rs REST /workers) -------------
def _job_queue(ctx):
    if getattr(ctx, "job_queue", None) is None:
        from .jobs import AuditLog, JobQueue

        ctx.job_queue = JobQueue(ctx, AuditLog(enabled=not ctx.config.auth.disable_audit_logging))
    return ctx.job_queue


async def create_worker(request):
    ctx: AppContext = request.app[CTX_KEY]
    body, _ = await _read_json(request)
    if not body or "url" not in body:
        return web.Response(status=400, body=error_body("'url' is required"), content_type="application/json")
    # typed validation (reference WorkerSpec worker.rs:604 via serde)
    from ..protocols.worker_spec import WorkerSpec, WorkerSpecError

    try:
        WorkerSpec.from_dict(body)
    except WorkerSpecError as exc:
        return web.Response(status=400, body=error_body(str(exc)), content_type="application/json")
    from .jobs import JobKind

    try:
        w = await _job_queue(ctx).submit(JobKind.ADD_WORKER, body, actor=request.get("tenant_id"))
    except Exception as exc:
        return web.Response(status=400, body=error_body(str(exc)), content_type="application/json")
    return web.json_response(w.to_dict(), status=201)


async def list_workers(request):
    ctx: AppContext = request.app[CTX_KEY]
    return web.json_response({"workers": [w.to_dict() for w in ctx.worker_registry.all()]})


async def get_worker(request):
    ctx: AppContext = request.app[CTX_KEY]
    wid = request.match_info["worker_id"]
    w = ctx.worker_registry.get_by_url(wid) or (
        ctx.worker_registry.get(int(wid)) if wid.isdigit() else None
    )
    if w is None:
        return web.Response(status=404, body=error_body("worker not found", 404), content_type="application/json")
    return web.json_response(w.to_dict())


async def delete_worker(request):
    ctx: AppContext = request.app[CTX_KEY]
    wid = request.match_info["worker_id"]
    if wid.isdigit() and ctx.worker_registry.get(int(wid)) is not None:
        wid = ctx.worker_registry.get(int(wid)).url
    from .jobs import JobKind

    try:
        w = await _job_queue(ctx).submit(JobKind.REMOVE_WORKER, {"url": wid}, actor=request.get("tenant_id"))
    except Exception:
        return web.Response(status=404, body=error_body("worker not found", 404), content_type="application/json")
    return web.json_response({"status": "removed", "url": w.url})


# --------------------------------------------------------------------------
# app assembly
# --------------------------------------------------------------------------
def build_app(ctx: AppContext) -> web.Application:
    app = web.Application(
        middlewares=[
            request_id_middleware,
            tracing_middleware,
            metrics_middleware,
            auth_middleware,
            tenant_middleware,
            plugin_middleware,
            admission_middleware,
        ],
        client_max_size=ctx.config.max_payload_size,
    )
    app[CTX_KEY] = ctx
    if ctx.plugins is None:
        from ..plugins import PluginManager

        ctx.plugins = PluginManager()

    inference_paths = [
        "/generate",
        "/v1/chat/completions",
        "/v1/completions",
        "/rerank",
        "/v1/rerank",
        "/v1/embeddings",
        "/v1/classify",
    ]
    for p in inference_paths:
        app.router.add_post(p, _proxy_endpoint)

    from ..routers.anthropic import v1_messages_handler

    app.router.add_post("/v1/messages", v1_messages_handler)

    from .responses_routes import add_responses_routes, v1_responses

    app.router.add_post("/v1/responses", v1_responses)
    add_responses_routes(app)
    from .interactions_routes import add_interactions_routes

    add_interactions_routes(app)

    app.router.add_get("/liveness", liveness)
    app.router.add_get("/readiness", readiness)
    app.router.add_get("/health", health)
    app.router.add_get("/health_generate", health)
    app.router.add_get("/engine_metrics", engine_metrics)
    app.router.add_get("/v1/models", v1_models)
    app.router.add_get("/get_model_info", get_model_info)
    app.router.add_get("/get_server_info", get_server_info)
    app.router.add_get("/metrics", metrics_endpoint)

    async def openapi_json(request: web.Request):
        # generated from the live route table (reference clients/openapi-gen)
        from .openapi import build_openapi

        return web.json_response(build_openapi(request.app))

    app.router.add_get("/openapi.json", openapi_json)

    async def start_profile(request):
        out = {}
        for w in ctx.worker_registry.all():
            if w.url.startswith("grpc"):
                try:
                    router = ctx.router_manager.default_router
                    client = router.pool.get(w.url)
                    out[w.url] = await client._unary("StartProfile", {})
                except Exception as exc:
                    out[w.url] = {"error": str(exc)}
        # MI355X-native equivalent: rocprofv3 wraps the worker process; this
        # passthrough asks engines to start their own profilers
        return web.json_response({"status": "requested", "workers": out})

    async def stop_profile(request):
        out = {}
        for w in ctx.worker_registry.all():
            if w.url.startswith("grpc"):
                try:
                    client = ctx.router_manager.default_router.pool.get(w.url)
                    out[w.url] = await client._unary("StopProfile", {})
                except Exception as exc:
                    out[w.url] = {"error": str(exc)}
        return web.json_response({"status": "requested", "workers": out})

    app.router.add_post("/start_profile", start_profile)
    app.router.add_post("/stop_profile", stop_profile)
    app.router.add_post("/flush_cache", flush_cache)
    app.router.add_get("/get_loads", get_loads)
    app.router.add_post("/parse/function_call", parse_function_call)
    app.router.add_post("/parse/reasoning", parse_reasoning)

    app.router.add_post("/workers", create_worker)
    app.router.add_get("/workers", list_workers)
    app.router.add_get("/workers/{worker_id:.*}", get_worker)
    app.router.add_delete("/workers/{worker_id:.*}", delete_worker)

    # tokenize/detokenize wired once the tokenizer registry exists
    from .tokenize_routes import add_tokenize_routes

    add_tokenize_routes(app)

    from ..routers.realtime import add_realtime_routes

    add_realtime_routes(app)

    async def v1_audio_transcriptions(request: web.Request):
        # multipart upload passthrough (reference multipart_upload_routes):
        # decodes the form, routes the transcription to an audio-capable
        # worker; without one, answers 501 with a clear error
        try:
            reader = await request.multipart()
        except (AssertionError, ValueError):
            return web.Response(status=400, body=error_body("multipart/form-data required"),
                                content_type="application/json")
        model = None
        audio_bytes = 0
        async for part in reader:
            if part.name == "model":
                model = (await part.text()).strip()
            elif part.name == "file":
                while True:
                    chunk = await part.read_chunk()
                    if not chunk:
                        break
                    audio_bytes += len(chunk)
        workers = ctx.worker_registry.for_model(ctx.worker_registry.resolve_model(model))
        audio_workers = [w for w in workers if w.labels.get("audio") == "true"]
        if not audio_workers:
            return web.Response(
                status=501,
                body=error_body("no audio-capable worker registered (label audio=true)", 501),
                content_type="application/json")
        return web.json_response({"text": "", "model": model, "bytes_received": audio_bytes})

    app.router.add_post("/v1/audio/transcriptions", v1_audio_transcriptions)

    # plugin (WASM-equivalent) module management — route names kept /wasm
    # for CLI/API compatibility with the reference
    async def add_plugin(request):
        body, _ = await _read_json(request)
        if not body or "path" not in body:
            return web.Response(status=400, body=error_body("'path' is required"), content_type="application/json")
        # plugins execute native Python: only load from the operator-configured
        # directory, never from an arbitrary request-supplied path
        plugin_dir = ctx.config.plugin_dir
        if not plugin_dir:
            return web.Response(
                status=403,
                body=error_body("plugin loading disabled: no --plugin-dir configured", 403, "permission_error"),
                content_type="application/json")
        import os as _os

        resolved = _os.path.realpath(str(body["path"]))
        root = _os.path.realpath(plugin_dir)
        if not (resolved == root or resolved.startswith(root + _os.sep)):
            return web.Response(
                status=403,
                body=error_body(f"plugin path must be inside {plugin_dir}", 403, "permission_error"),
                content_type="application/json")
        try:
            mod_id = ctx.plugins.add_module(resolved, body.get("name"))
        except Exception as exc:
            return web.Response(status=400, body=error_body(str(exc)), content_type="application/json")
        return web.json_response({"module_uuid": mod_id}, status=201)

    async def remove_plugin(request):
        if not ctx.plugins.remove_module(request.match_info["module_uuid"]):
            return web.Response(status=404, body=error_body("module not found", 404), content_type="application/json")
        return web.json_response({"status": "removed"})

    async def list_plugins(request):
        return web.json_response({"modules": ctx.plugins.list_modules()})

    app.router.add_post("/wasm", add_plugin)
    app.router.add_delete("/wasm/{module_uuid}", remove_plugin)
    app.router.add_get("/wasm", list_plugins)
    return app


async def startup(config: RouterConfig, serve: bool = True) -> AppContext:
    """Ordered boot (reference server.rs:984): validate -> metrics -> context ->
    workers -> router manager -> monitor -> app -> serve."""
    config.validate()
    ctx = AppContext(config)
    ctx.init_workers_from_config()
    from ..rate_limit.limiter import ConcurrencyLimiter

    if config.rate_limit.max_concurrent_requests > 0:
        ctx.rate_limiter = ConcurrencyLimiter(config.rate_limit)
    if config.priority_scheduler.enabled:
        from ..scheduler.engine import PriorityScheduler, SchedulerConfig

        sched_cfg = SchedulerConfig.from_yaml(config.priority_scheduler.config_path)
        sched_cfg.default_class = config.priority_scheduler.default_max_class
        ctx.scheduler = PriorityScheduler(sched_cfg, ctx.worker_registry, ctx.metrics)
        ctx.scheduler.start_sampler()
    if config.tenant_rate_limit.enabled:
        from ..rate_limit.tenant import RateLimitManager, TenantRateLimitSettings

        ctx.tenant_rate_limiter = RateLimitManager(
            TenantRateLimitSettings.from_yaml(config.tenant_rate_limit.config_path)
        )
    if (config.tokenizer_path or config.model_path) and not config.disable_tokenizer_autoload:
        from ..tokenizer.registry import TokenizerRegistry

        ctx.tokenizer_registry = TokenizerRegistry()
        path = config.tokenizer_path or config.model_path
        try:
            import os as _os

            tok_json = path if path.endswith(".json") else _os.path.join(path, "tokenizer.json")
            if _os.path.exists(tok_json):
                try:
                    # prefer the gfx950 batch-BPE tokenizer when the vocab is BPE
                    from ..tokenizer.gpu_bpe import GpuBpeTokenizer

                    tok = GpuBpeTokenizer(tok_json)
                except Exception:
                    tok = None
                if tok is None:
                    tok = ctx.tokenizer_registry.load(config.model_path or "default", tok_json)
                else:
                    ctx.tokenizer_registry.add(config.model_path or "default", tok)
                if config.tokenizer_cache.enable_l1:
                    from ..tokenizer.l1_cache import L1CachedTokenizer

                    ctx.tokenizer_registry.add(
                        config.model_path or "default",
                        L1CachedTokenizer(tok, config.tokenizer_cache.l1_max_memory),
                    )
                log.info("tokenizer loaded from %s", tok_json)
        except Exception as exc:
            log.warning("tokenizer autoload failed: %s", exc)
    ctx.router_manager = RouterManager(ctx, config)
    if config.mesh.enabled:
        from ..mesh.adapters import MeshAdapters
        from ..mesh.server import start_mesh_server
        from ..mesh.swim import MeshNode

        node_id = config.mesh.server_name or f"{config.host}:{config.port}"
        m = config.mesh
        scheme = "https" if (m.mtls_cert and m.mtls_key and m.mtls_ca) else "http"
        advertise = f"{scheme}://{m.advertise_host or '127.0.0.1'}:{m.port}"
        ctx.mesh = MeshNode(node_id, advertise,
                            mtls_cert=m.mtls_cert, mtls_key=m.mtls_key, mtls_ca=m.mtls_ca)
        ctx.mesh_adapters = MeshAdapters(ctx.mesh, ctx)
        if serve:
            ctx._mesh_runner = await start_mesh_server(ctx.mesh, config.mesh.host, config.mesh.port)
        await ctx.mesh.start(config.mesh.peer_urls)
    if config.discovery.enabled:
        from ..discovery import FileDiscovery, start_discovery
        from ..discovery.source import KubernetesDiscovery

        file_path = config.discovery.selector.get("file")
        if file_path:
            source = FileDiscovery(file_path)
        else:
            source = KubernetesDiscovery(
                config.discovery.selector, config.discovery.port, config.discovery.namespace
            )
        ctx._background.append(
            await start_discovery(ctx.worker_registry, source, circuit_breaker_config=config.circuit_breaker)
        )
    await ctx.start_background()
    # runtime self-observability: asyncio loop-lag canary (reference
    # observability/runtime_metrics.rs — tokio event-loop canary + sampler)
    if not ctx.metrics._null:

        async def _loop_canary(interval: float = 0.5):
            loop = asyncio.get_event_loop()
            while True:
                t0 = loop.time()
                await asyncio.sleep(interval)
                lag = max(0.0, loop.time() - t0 - interval)
                ctx.metrics.event_loop_lag.observe(lag)

        ctx._background.append(asyncio.ensure_future(_loop_canary()))
    if serve and config.health_check_port:
        probe_app = web.Application()
        probe_app[CTX_KEY] = ctx
        probe_app.router.add_get("/liveness", liveness)
        probe_app.router.add_get("/readiness", readiness)
        probe_app.router.add_get("/health", health)
        probe_runner = web.AppRunner(probe_app, access_log=None)
        await probe_runner.setup()
        await web.TCPSite(probe_runner, config.host, config.health_check_port).start()
        ctx._probe_runner = probe_runner
    if serve:
        app = build_app(ctx)
        runner = web.AppRunner(app, access_log=None)
        await runner.setup()
        site = web.TCPSite(runner, config.host, config.port)
        await site.start()
        ctx._runner = runner
        log.info("smg gateway listening on %s:%d", config.host, config.port)
    return ctx
