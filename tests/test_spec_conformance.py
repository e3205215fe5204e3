"""API schema conformance (reference: model_gateway/tests/spec/ — response
shapes must match the OpenAI / Anthropic wire contracts)."""
import json

import pytest

from tests.test_gateway_e2e import CHAT_BODY, make_ctx, start_client, stop_all


def test_chat_completion_schema(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post("/v1/chat/completions", json=CHAT_BODY)
            d = await resp.json()
            for field in ("id", "object", "created", "model", "choices", "usage"):
                assert field in d, field
            assert d["object"] == "chat.completion"
            ch = d["choices"][0]
            assert set(ch) >= {"index", "message", "finish_reason"}
            assert set(ch["message"]) >= {"role", "content"}
            assert set(d["usage"]) >= {"prompt_tokens", "completion_tokens", "total_tokens"}
        finally:
            await stop_all(client, engines)

    runner(run())


def test_completion_schema(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post("/v1/completions",
                                     json={"model": "mock-model", "prompt": "x", "max_tokens": 2})
            d = await resp.json()
            assert d["object"] == "text_completion"
            assert set(d["choices"][0]) >= {"index", "text", "finish_reason"}
        finally:
            await stop_all(client, engines)

    runner(run())


def test_error_envelope_schema(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post("/v1/chat/completions", json={"model": "m", "messages": []})
            d = await resp.json()
            assert set(d["error"]) >= {"message", "type", "code"}
        finally:
            await stop_all(client, engines)

    runner(run())


def test_models_schema(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            d = await (await client.get("/v1/models")).json()
            assert d["object"] == "list"
            assert set(d["data"][0]) >= {"id", "object"}
        finally:
            await stop_all(client, engines)

    runner(run())


def test_anthropic_message_schema(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post("/v1/messages",
                                     json={"model": "mock-model", "max_tokens": 2,
                                           "messages": [{"role": "user", "content": "hi"}]})
            d = await resp.json()
            for field in ("id", "type", "role", "model", "content", "stop_reason", "usage"):
                assert field in d, field
            assert d["content"][0]["type"] == "text"
            assert set(d["usage"]) >= {"input_tokens", "output_tokens"}
        finally:
            await stop_all(client, engines)

    runner(run())


def test_anthropic_tool_blocks_roundtrip():
    from smg_amd.routers.anthropic import messages_to_chat

    body = {
        "model": "m", "max_tokens": 5,
        "messages": [
            {"role": "user", "content": "weather?"},
            {"role": "assistant", "content": [
                {"type": "text", "text": "checking"},
                {"type": "tool_use", "id": "tu_1", "name": "get_weather", "input": {"city": "Oslo"}},
            ]},
            {"role": "user", "content": [
                {"type": "tool_result", "tool_use_id": "tu_1", "content": "rainy"},
            ]},
        ],
    }
    chat = messages_to_chat(body)
    roles = [m["role"] for m in chat["messages"]]
    assert roles == ["user", "assistant", "tool"]
    tc = chat["messages"][1]["tool_calls"][0]
    assert tc["function"]["name"] == "get_weather"
    assert json.loads(tc["function"]["arguments"]) == {"city": "Oslo"}
    assert chat["messages"][2]["tool_call_id"] == "tu_1"


def test_anthropic_image_block():
    from smg_amd.routers.anthropic import messages_to_chat

    body = {
        "model": "m", "max_tokens": 5,
        "messages": [{"role": "user", "content": [
            {"type": "text", "text": "what is this"},
            {"type": "image", "source": {"type": "base64", "media_type": "image/png", "data": "QUJD"}},
        ]}],
    }
    chat = messages_to_chat(body)
    parts = chat["messages"][0]["content"]
    assert parts[1]["type"] == "image_url"
    assert parts[1]["image_url"]["url"].startswith("data:image/png;base64,")


def test_openapi_schema(runner):
    """GET /openapi.json covers the live route table with valid structure
    (reference clients/openapi-gen)."""
    async def run():
        from tests.test_gateway_e2e import make_ctx, start_client, stop_all

        ctx, engines = make_ctx(n_workers=1)
        client = await start_client(ctx, engines)
        try:
            r = await client.get("/openapi.json")
            assert r.status == 200
            doc = await r.json()
            assert doc["openapi"].startswith("3.")
            for path in ("/v1/chat/completions", "/v1/completions", "/v1/embeddings",
                         "/v1/responses", "/v1/messages", "/workers", "/health"):
                assert path in doc["paths"], path
            post = doc["paths"]["/v1/chat/completions"]["post"]
            ref = post["requestBody"]["content"]["application/json"]["schema"]["$ref"]
            assert ref.endswith("ChatCompletionRequest")
            # every $ref resolves
            import json as _json

            blob = _json.dumps(doc)
            import re as _re

            for m in set(_re.findall(r'#/components/schemas/(\w+)', blob)):
                assert m in doc["components"]["schemas"], m
        finally:
            await stop_all(client, engines)

    runner(run())
