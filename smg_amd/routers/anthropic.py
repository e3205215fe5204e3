"""Anthropic Messages API over self-hosted workers (reference:
model_gateway/src/routers/anthropic/ — Messages <-> Chat translation, SSE
event-stream translation anthropic/sse.rs (799 LoC)).

POST /v1/messages accepts the Anthropic wire format, translates to an OpenAI
chat completion for the routed worker, and translates the result (unary or
SSE) back into Anthropic events: message_start -> content_block_start ->
content_block_delta* -> content_block_stop -> message_delta -> message_stop.
"""
from __future__ import annotations

import json
import uuid
from typing import Any, AsyncIterator, Dict, List

from aiohttp import web

from ..protocols.openai import error_body
from .base import RouteRequest


def messages_to_chat(body: Dict[str, Any]) -> Dict[str, Any]:
    messages: List[Dict[str, Any]] = []
    system = body.get("system")
    if isinstance(system, str) and system:
        messages.append({"role": "system", "content": system})
    elif isinstance(system, list):
        text = "".join(b.get("text", "") for b in system if isinstance(b, dict) and b.get("type") == "text")
        if text:
            messages.append({"role": "system", "content": text})
    for m in body.get("messages") or []:
        content = m.get("content")
        role = m.get("role", "user")
        if isinstance(content, list):
            parts = []
            tool_calls = []
            tool_results = []
            for b in content:
                if not isinstance(b, dict):
                    continue
                btype = b.get("type")
                if btype == "text":
                    parts.append({"type": "text", "text": b.get("text", "")})
                elif btype == "image":
                    src = b.get("source", {})
                    if src.get("type") == "base64":
                        parts.append({
                            "type": "image_url",
                            "image_url": {"url": f"data:{src.get('media_type','image/png')};base64,{src.get('data','')}"},
                        })
                    elif src.get("type") == "url":
                        parts.append({"type": "image_url", "image_url": {"url": src.get("url", "")}})
                elif btype == "tool_use":
                    tool_calls.append({
                        "id": b.get("id"),
                        "type": "function",
                        "function": {"name": b.get("name"), "arguments": json.dumps(b.get("input") or {})},
                    })
                elif btype == "tool_result":
                    rc = b.get("content")
                    if isinstance(rc, list):
                        rc = "".join(x.get("text", "") for x in rc if isinstance(x, dict))
                    tool_results.append({"role": "tool", "tool_call_id": b.get("tool_use_id"),
                                         "content": rc if isinstance(rc, str) else json.dumps(rc)})
            if tool_results:
                messages.extend(tool_results)
                continue
            msg = {"role": role}
            if len(parts) == 1 and parts[0]["type"] == "text":
                msg["content"] = parts[0]["text"]
            elif parts:
                msg["content"] = parts
            else:
                msg["content"] = ""
            if tool_calls:
                msg["tool_calls"] = tool_calls
                if not parts:
                    msg["content"] = None
            messages.append(msg)
        else:
            messages.append({"role": role, "content": content or ""})
    chat: Dict[str, Any] = {
        "model": body.get("model"),
        "messages": messages,
        "max_tokens": body.get("max_tokens", 256),
        "stream": bool(body.get("stream", False)),
    }
    for k in ("temperature", "top_p", "top_k"):
        if body.get(k) is not None:
            chat[k] = body[k]
    if body.get("stop_sequences"):
        chat["stop"] = body["stop_sequences"]
    if body.get("tools"):
        chat["tools"] = [
            {
                "type": "function",
                "function": {
                    "name": t.get("name"),
                    "description": t.get("description"),
                    "parameters": t.get("input_schema", {}),
                },
            }
            for t in body["tools"]
        ]
    return chat


STOP_REASON = {"stop": "end_turn", "length": "max_tokens", "tool_calls": "tool_use"}


def chat_to_message(body: Dict[str, Any], chat_resp: Dict[str, Any]) -> Dict[str, Any]:
    choice = (chat_resp.get("choices") or [{}])[0]
    msg = choice.get("message", {})
    content: List[Dict[str, Any]] = []
    if msg.get("reasoning_content"):
        content.append({"type": "thinking", "thinking": msg["reasoning_content"]})
    if msg.get("content"):
        content.append({"type": "text", "text": msg["content"]})
    for tc in msg.get("tool_calls") or []:
        fn = tc.get("function", {})
        try:
            args = json.loads(fn.get("arguments") or "{}")
        except json.JSONDecodeError:
            args = {}
        content.append({"type": "tool_use", "id": tc.get("id"), "name": fn.get("name"), "input": args})
    usage = chat_resp.get("usage", {})
    return {
        "id": f"msg_{uuid.uuid4().hex[:24]}",
        "type": "message",
        "role": "assistant",
        "model": chat_resp.get("model") or body.get("model"),
        "content": content,
        "stop_reason": STOP_REASON.get(choice.get("finish_reason"), "end_turn"),
        "stop_sequence": None,
        "usage": {
            "input_tokens": usage.get("prompt_tokens", 0),
            "output_tokens": usage.get("completion_tokens", 0),
        },
    }


async def translate_sse(chat_stream: AsyncIterator[bytes], model: str) -> AsyncIterator[bytes]:
    """OpenAI chunk SSE -> Anthropic event SSE."""
    mid = f"msg_{uuid.uuid4().hex[:24]}"

    def ev(name: str, payload: dict) -> bytes:
        return f"event: {name}\ndata: {json.dumps(payload)}\n\n".encode()

    yield ev(
        "message_start",
        {
            "type": "message_start",
            "message": {
                "id": mid, "type": "message", "role": "assistant", "model": model,
                "content": [], "stop_reason": None, "usage": {"input_tokens": 0, "output_tokens": 0},
            },
        },
    )
    block_open = False
    out_tokens = 0
    finish = "end_turn"
    buf = b""
    async for raw in chat_stream:
        buf += raw
        while b"\n\n" in buf:
            frame, buf = buf.split(b"\n\n", 1)
            line = frame.decode().strip()
            if not line.startswith("data: "):
                continue
            data = line[6:]
            if data == "[DONE]":
                continue
            try:
                chunk = json.loads(data)
            except json.JSONDecodeError:
                continue
            for choice in chunk.get("choices") or []:
                delta = choice.get("delta") or {}
                text = delta.get("content")
                if text:
                    if not block_open:
                        yield ev("content_block_start", {"type": "content_block_start", "index": 0,
                                                         "content_block": {"type": "text", "text": ""}})
                        block_open = True
                    out_tokens += 1
                    yield ev("content_block_delta", {"type": "content_block_delta", "index": 0,
                                                     "delta": {"type": "text_delta", "text": text}})
                if choice.get("finish_reason"):
                    finish = STOP_REASON.get(choice["finish_reason"], "end_turn")
    if block_open:
        yield ev("content_block_stop", {"type": "content_block_stop", "index": 0})
    yield ev("message_delta", {"type": "message_delta", "delta": {"stop_reason": finish, "stop_sequence": None},
                               "usage": {"output_tokens": out_tokens}})
    yield ev("message_stop", {"type": "message_stop"})


async def v1_messages_handler(request: web.Request):
    from ..server.app import CTX_KEY

    ctx = request.app[CTX_KEY]
    try:
        body = json.loads(await request.read() or b"{}")
    except json.JSONDecodeError:
        return web.Response(status=400, body=error_body("invalid JSON"), content_type="application/json")
    if not body.get("messages"):
        return web.Response(
            status=400, body=error_body("'messages' is required"), content_type="application/json"
        )
    chat_body = messages_to_chat(body)
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
    if resp.is_stream:
        out = web.StreamResponse(status=200, headers={"content-type": "text/event-stream"})
        out.enable_chunked_encoding()
        await out.prepare(request)
        async for chunk in translate_sse(resp.stream, body.get("model") or "unknown"):
            await out.write(chunk)
        await out.write_eof()
        return out
    chat_resp = json.loads(resp.body)
    return web.json_response(chat_to_message(body, chat_resp))
