"""OpenAI Responses API + Conversations API (reference: routers/{responses,
conversations}/ + common/persistence_utils.rs; format translation
common/openai_bridge/transformer.rs).

POST /v1/responses translates to a chat completion, routes through the
RouterManager, translates back to a Response object and persists it;
conversation CRUD backs onto the data-connector storage.
"""
from __future__ import annotations

import json
import time
import uuid
from typing import Any, Dict, List

from aiohttp import web

from ..protocols.openai import error_body
from ..routers.base import RouteRequest


def _ctx(request):
    from .app import CTX_KEY

    return request.app[CTX_KEY]


def _storages(ctx):
    if ctx.storage is None:
        from ..storage import make_storage

        backend = ctx.config.storage.backend
        try:
            ctx.storage = make_storage(backend)
        except Exception:
            ctx.storage = make_storage("memory")
    return ctx.storage


# ---- format translation (openai_bridge equivalents) -----------------------
def responses_to_chat(body: Dict[str, Any]) -> Dict[str, Any]:
    messages: List[Dict[str, Any]] = []
    instructions = body.get("instructions")
    if instructions:
        messages.append({"role": "system", "content": instructions})
    inp = body.get("input")
    if isinstance(inp, str):
        messages.append({"role": "user", "content": inp})
    elif isinstance(inp, list):
        for item in inp:
            if not isinstance(item, dict):
                continue
            if item.get("type") in (None, "message"):
                content = item.get("content")
                if isinstance(content, list):
                    text = "".join(
                        p.get("text", "") for p in content if isinstance(p, dict) and p.get("type") in ("input_text", "output_text", "text")
                    )
                else:
                    text = content or ""
                messages.append({"role": item.get("role", "user"), "content": text})
    chat = {
        "model": body.get("model"),
        "messages": messages,
        "stream": bool(body.get("stream", False)),
    }
    if body.get("max_output_tokens"):
        chat["max_tokens"] = body["max_output_tokens"]
    for k in ("temperature", "top_p", "tools"):
        if k in body and body[k] is not None:
            chat[k] = body[k]
    return chat


def chat_to_response(body: Dict[str, Any], chat_resp: Dict[str, Any]) -> Dict[str, Any]:
    rid = f"resp_{uuid.uuid4().hex}"
    msg = (chat_resp.get("choices") or [{}])[0].get("message", {})
    output = []
    if msg.get("content"):
        output.append(
            {
                "type": "message",
                "id": f"msg_{uuid.uuid4().hex[:16]}",
                "role": "assistant",
                "status": "completed",
                "content": [{"type": "output_text", "text": msg["content"], "annotations": []}],
            }
        )
    for tc in msg.get("tool_calls") or []:
        output.append(
            {
                "type": "function_call",
                "id": tc.get("id"),
                "call_id": tc.get("id"),
                "name": tc.get("function", {}).get("name"),
                "arguments": tc.get("function", {}).get("arguments"),
                "status": "completed",
            }
        )
    usage = chat_resp.get("usage", {})
    return {
        "id": rid,
        "object": "response",
        "created_at": int(time.time()),
        "status": "completed",
        "model": chat_resp.get("model") or body.get("model"),
        "instructions": body.get("instructions"),
        "output": output,
        "output_text": msg.get("content") or "",
        "usage": {
            "input_tokens": usage.get("prompt_tokens", 0),
            "output_tokens": usage.get("completion_tokens", 0),
            "total_tokens": usage.get("total_tokens", 0),
        },
        "metadata": body.get("metadata") or {},
    }


# ---- handlers --------------------------------------------------------------
async def v1_responses(request: web.Request):
    ctx = _ctx(request)
    try:
        body = json.loads(await request.read() or b"{}")
    except json.JSONDecodeError:
        return web.Response(status=400, body=error_body("invalid JSON"), content_type="application/json")
    if body.get("stream"):
        return web.Response(
            status=400,
            body=error_body("streaming responses not supported on this endpoint yet", 400),
            content_type="application/json",
        )
    chat_body = responses_to_chat(body)
    # conversation context injection
    conv_id = body.get("conversation") if isinstance(body.get("conversation"), str) else None
    resp_store, conv_store = _storages(ctx)
    if conv_id:
        prior = await conv_store.list_items(conv_id, limit=100)
        history = [
            {"role": it.get("role", "user"), "content": it.get("content", "")}
            for it in prior
            if it.get("type") in (None, "message")
        ]
        chat_body["messages"] = history + chat_body["messages"]
    route_req = RouteRequest(
        path="/v1/chat/completions",
        body=chat_body,
        raw_body=json.dumps(chat_body).encode(),
        headers=dict(request.headers),
        request_id=request.get("request_id", ""),
        tenant_id=request.get("tenant_id"),
    )
    resp = await ctx.router_manager.route(route_req)
    if resp.status != 200:
        return web.Response(status=resp.status, body=resp.body, content_type="application/json")
    chat_resp = json.loads(resp.body)
    response_obj = chat_to_response(body, chat_resp)
    if body.get("store", True):
        response_obj["_input_items"] = [
            {"type": "message", "role": m["role"], "content": m["content"]} for m in chat_body["messages"]
        ]
        await resp_store.store_response(response_obj)
    if conv_id:
        items = [{"type": "message", "role": m["role"], "content": m["content"]} for m in chat_body["messages"][-1:]]
        items.append({"type": "message", "role": "assistant", "content": response_obj["output_text"]})
        await conv_store.add_items(conv_id, items)
    return web.json_response(response_obj)


async def v1_responses_get(request):
    resp_store, _ = _storages(_ctx(request))
    obj = await resp_store.get_response(request.match_info["response_id"])
    if obj is None:
        return web.Response(status=404, body=error_body("response not found", 404), content_type="application/json")
    return web.json_response(obj)


async def v1_responses_delete(request):
    resp_store, _ = _storages(_ctx(request))
    ok = await resp_store.delete_response(request.match_info["response_id"])
    if not ok:
        return web.Response(status=404, body=error_body("response not found", 404), content_type="application/json")
    return web.json_response({"id": request.match_info["response_id"], "deleted": True})


async def v1_responses_cancel(request):
    resp_store, _ = _storages(_ctx(request))
    obj = await resp_store.get_response(request.match_info["response_id"])
    if obj is None:
        return web.Response(status=404, body=error_body("response not found", 404), content_type="application/json")
    return web.json_response(obj)  # synchronous responses are already terminal


async def v1_responses_input_items(request):
    resp_store, _ = _storages(_ctx(request))
    items = await resp_store.list_input_items(request.match_info["response_id"])
    return web.json_response({"object": "list", "data": items})


async def v1_conversations_create(request):
    _, conv_store = _storages(_ctx(request))
    body = json.loads(await request.read() or b"{}")
    conv = await conv_store.create_conversation(body.get("metadata"))
    if body.get("items"):
        await conv_store.add_items(conv["id"], body["items"])
    return web.json_response(conv)


async def v1_conversations_get(request):
    _, conv_store = _storages(_ctx(request))
    conv = await conv_store.get_conversation(request.match_info["conversation_id"])
    if conv is None:
        return web.Response(status=404, body=error_body("conversation not found", 404), content_type="application/json")
    return web.json_response(conv)


async def v1_conversations_update(request):
    _, conv_store = _storages(_ctx(request))
    body = json.loads(await request.read() or b"{}")
    conv = await conv_store.update_conversation(request.match_info["conversation_id"], body.get("metadata") or {})
    if conv is None:
        return web.Response(status=404, body=error_body("conversation not found", 404), content_type="application/json")
    return web.json_response(conv)


async def v1_conversations_delete(request):
    _, conv_store = _storages(_ctx(request))
    ok = await conv_store.delete_conversation(request.match_info["conversation_id"])
    if not ok:
        return web.Response(status=404, body=error_body("conversation not found", 404), content_type="application/json")
    return web.json_response({"id": request.match_info["conversation_id"], "deleted": True})


async def v1_conversations_items(request):
    _, conv_store = _storages(_ctx(request))
    cid = request.match_info["conversation_id"]
    if request.method == "POST":
        body = json.loads(await request.read() or b"{}")
        try:
            items = await conv_store.add_items(cid, body.get("items") or [])
        except Exception as exc:
            return web.Response(status=404, body=error_body(str(exc), 404), content_type="application/json")
        return web.json_response({"object": "list", "data": items})
    items = await conv_store.list_items(cid, limit=int(request.query.get("limit", 100)), after=request.query.get("after"))
    return web.json_response({"object": "list", "data": items})


async def v1_conversations_item(request):
    _, conv_store = _storages(_ctx(request))
    cid, iid = request.match_info["conversation_id"], request.match_info["item_id"]
    if request.method == "DELETE":
        ok = await conv_store.delete_item(cid, iid)
        if not ok:
            return web.Response(status=404, body=error_body("item not found", 404), content_type="application/json")
        return web.json_response({"id": iid, "deleted": True})
    item = await conv_store.get_item(cid, iid)
    if item is None:
        return web.Response(status=404, body=error_body("item not found", 404), content_type="application/json")
    return web.json_response(item)


def add_responses_routes(app: web.Application) -> None:
    app.router.add_get("/v1/responses/{response_id}", v1_responses_get)
    app.router.add_delete("/v1/responses/{response_id}", v1_responses_delete)
    app.router.add_post("/v1/responses/{response_id}/cancel", v1_responses_cancel)
    app.router.add_get("/v1/responses/{response_id}/input_items", v1_responses_input_items)
    app.router.add_post("/v1/conversations", v1_conversations_create)
    app.router.add_get("/v1/conversations/{conversation_id}", v1_conversations_get)
    app.router.add_post("/v1/conversations/{conversation_id}", v1_conversations_update)
    app.router.add_delete("/v1/conversations/{conversation_id}", v1_conversations_delete)
    app.router.add_post("/v1/conversations/{conversation_id}/items", v1_conversations_items)
    app.router.add_get("/v1/conversations/{conversation_id}/items", v1_conversations_items)
    app.router.add_get("/v1/conversations/{conversation_id}/items/{item_id}", v1_conversations_item)
    app.router.add_delete("/v1/conversations/{conversation_id}/items/{item_id}", v1_conversations_item)
