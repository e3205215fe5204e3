"""Gemini Interactions API (reference: routers/gemini/ — route_interactions
router.rs:75, driver.rs step pipeline; protocol crates/protocols/src/
interactions.rs — InteractionsRequest/Interaction; route server.rs:818).

POST /v1/interactions translates to a chat completion, routes through the
RouterManager, and returns an Interaction object.  `previous_interaction_id`
chains prior turns into the context (the reference persists interactions the
same way; storage here is the in-process interaction store on the ctx).
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


def _interaction_store(ctx) -> Dict[str, Dict]:
    store = getattr(ctx, "interactions", None)
    if store is None:
        store = {}
        ctx.interactions = store
    return store


def _input_to_messages(inp: Any) -> List[Dict[str, Any]]:
    if isinstance(inp, str):
        return [{"role": "user", "content": inp}]
    messages = []
    if isinstance(inp, list):
        for item in inp:
            if isinstance(item, str):
                messages.append({"role": "user", "content": item})
            elif isinstance(item, dict):
                role = item.get("role", "user")
                if role == "model":  # Gemini role name
                    role = "assistant"
                text = item.get("text")
                if text is None and isinstance(item.get("parts"), list):
                    text = "".join(p.get("text", "") for p in item["parts"] if isinstance(p, dict))
                messages.append({"role": role, "content": text or ""})
    return messages


def interactions_to_chat(body: Dict[str, Any], history: List[Dict[str, Any]]) -> Dict[str, Any]:
    messages: List[Dict[str, Any]] = []
    if body.get("system_instruction"):
        messages.append({"role": "system", "content": body["system_instruction"]})
    messages.extend(history)
    messages.extend(_input_to_messages(body.get("input")))
    chat: Dict[str, Any] = {
        "model": body.get("model") or body.get("agent"),
        "messages": messages,
        "stream": bool(body.get("stream", False)),
    }
    gen = body.get("generation_config") or {}
    if gen.get("max_output_tokens"):
        chat["max_tokens"] = gen["max_output_tokens"]
    for src, dst in (("temperature", "temperature"), ("top_p", "top_p")):
        if gen.get(src) is not None:
            chat[dst] = gen[src]
    if body.get("tools"):
        tools = []
        for t in body["tools"]:
            decls = t.get("function_declarations") if isinstance(t, dict) else None
            if decls:
                for d in decls:
                    tools.append({"type": "function", "function": d})
            elif isinstance(t, dict) and t.get("name"):
                tools.append({"type": "function", "function": t})
        if tools:
            chat["tools"] = tools
    if body.get("response_format"):
        chat["response_format"] = {
            "type": "json_schema",
            "json_schema": {"name": "response", "schema": body["response_format"]},
        }
    return chat


def chat_to_interaction(body: Dict[str, Any], chat_resp: Dict[str, Any]) -> Dict[str, Any]:
    msg = (chat_resp.get("choices") or [{}])[0].get("message", {})
    outputs = []
    if msg.get("content"):
        outputs.append({"type": "text", "text": msg["content"], "role": "model"})
    for tc in msg.get("tool_calls") or []:
        fn = tc.get("function", {})
        try:
            args = json.loads(fn.get("arguments") or "{}")
        except json.JSONDecodeError:
            args = {"_raw": fn.get("arguments")}
        outputs.append({"type": "function_call", "id": tc.get("id"), "name": fn.get("name"), "args": args})
    usage = chat_resp.get("usage", {})
    now = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
    return {
        "object": "interaction",
        "id": f"interaction_{uuid.uuid4().hex}",
        "model": chat_resp.get("model") or body.get("model"),
        "agent": body.get("agent"),
        "status": "completed",
        "created": now,
        "updated": now,
        "role": "model",
        "outputs": outputs,
        "usage": {
            "input_tokens": usage.get("prompt_tokens", 0),
            "output_tokens": usage.get("completion_tokens", 0),
            "total_tokens": usage.get("total_tokens", 0),
        },
        "previous_interaction_id": body.get("previous_interaction_id"),
    }


async def v1_interactions(request: web.Request):
    ctx = _ctx(request)
    try:
        body = json.loads(await request.read() or b"{}")
    except json.JSONDecodeError:
        return web.Response(status=400, body=error_body("invalid JSON"), content_type="application/json")
    if not body.get("model") and not body.get("agent"):
        return web.Response(
            status=400, body=error_body("one of 'model' or 'agent' is required"), content_type="application/json"
        )
    store = _interaction_store(ctx)
    # chain prior turns (reference driver context assembly)
    history: List[Dict[str, Any]] = []
    prev_id = body.get("previous_interaction_id")
    chain: List[Dict] = []
    seen = set()
    while prev_id and prev_id in store and prev_id not in seen:
        seen.add(prev_id)
        entry = store[prev_id]
        chain.append(entry)
        prev_id = entry["interaction"].get("previous_interaction_id")
    for entry in reversed(chain):
        history.extend(entry["input_messages"])
        text = "".join(
            o.get("text", "") for o in entry["interaction"].get("outputs") or [] if o.get("type") == "text"
        )
        if text:
            history.append({"role": "assistant", "content": text})
    chat_body = interactions_to_chat(body, history)
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
    interaction = chat_to_interaction(body, json.loads(resp.body))
    if body.get("store", True):
        store[interaction["id"]] = {
            "interaction": interaction,
            "input_messages": _input_to_messages(body.get("input")),
        }
    return web.json_response(interaction)


def add_interactions_routes(app: web.Application) -> None:
    app.router.add_post("/v1/interactions", v1_interactions)
