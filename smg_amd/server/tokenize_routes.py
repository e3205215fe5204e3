"""/v1/tokenize, /v1/detokenize and /v1/tokenizers admin routes (reference:
server.rs:821-822 and the tokenizer management group :909-918)."""
from __future__ import annotations


from aiohttp import web

from ..protocols.openai import error_body


def add_tokenize_routes(app: web.Application) -> None:
    app.router.add_post("/v1/tokenize", v1_tokenize)
    app.router.add_post("/v1/detokenize", v1_detokenize)
    app.router.add_post("/v1/tokenizers", v1_tokenizers_add)
    app.router.add_get("/v1/tokenizers", v1_tokenizers_list)
    app.router.add_get("/v1/tokenizers/{tid}", v1_tokenizers_get)
    app.router.add_delete("/v1/tokenizers/{tid}", v1_tokenizers_remove)


def _registry(request):
    from .app import CTX_KEY

    ctx = request.app[CTX_KEY]
    if ctx.tokenizer_registry is None:
        from ..tokenizer.registry import TokenizerRegistry

        ctx.tokenizer_registry = TokenizerRegistry()
    return ctx.tokenizer_registry


async def v1_tokenize(request):
    reg = _registry(request)
    body = await request.json()
    tok = reg.get(body.get("model") or body.get("tokenizer"))
    if tok is None:
        return web.Response(status=404, body=error_body("no tokenizer loaded", 404), content_type="application/json")
    prompt = body.get("prompt") or body.get("text") or ""
    if isinstance(prompt, list):
        ids = [tok.encode(p) for p in prompt]
        count = sum(len(x) for x in ids)
    else:
        ids = tok.encode(prompt)
        count = len(ids)
    return web.json_response({"tokens": ids, "count": count, "max_model_len": tok.model_max_length})


async def v1_detokenize(request):
    reg = _registry(request)
    body = await request.json()
    tok = reg.get(body.get("model") or body.get("tokenizer"))
    if tok is None:
        return web.Response(status=404, body=error_body("no tokenizer loaded", 404), content_type="application/json")
    tokens = body.get("tokens") or []
    return web.json_response({"text": tok.decode(tokens)})


async def v1_tokenizers_add(request):
    reg = _registry(request)
    body = await request.json()
    name = body.get("name") or body.get("tokenizer_id")
    path = body.get("path") or body.get("tokenizer_path")
    if not name or not path:
        return web.Response(status=400, body=error_body("'name' and 'path' required"), content_type="application/json")
    try:
        reg.load(name, path, chat_template=body.get("chat_template"))
    except Exception as exc:
        return web.Response(status=400, body=error_body(f"load failed: {exc}"), content_type="application/json")
    return web.json_response({"status": "loaded", "id": name}, status=201)


async def v1_tokenizers_list(request):
    reg = _registry(request)
    return web.json_response({"tokenizers": reg.list()})


async def v1_tokenizers_get(request):
    reg = _registry(request)
    tid = request.match_info["tid"]
    info = reg.info(tid)
    if info is None:
        return web.Response(status=404, body=error_body("not found", 404), content_type="application/json")
    return web.json_response(info)


async def v1_tokenizers_remove(request):
    reg = _registry(request)
    tid = request.match_info["tid"]
    if not reg.remove(tid):
        return web.Response(status=404, body=error_body("not found", 404), content_type="application/json")
    return web.json_response({"status": "removed", "id": tid})
