"""Responses API, Conversations API and Anthropic Messages translation tests
(reference: routers/{responses,conversations,anthropic} behavior + spec/)."""
import json

import pytest

from tests.test_gateway_e2e import CHAT_BODY, make_ctx, start_client, stop_all


def test_responses_roundtrip(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post(
                "/v1/responses",
                json={"model": "mock-model", "input": "hello there", "max_output_tokens": 4,
                      "instructions": "be brief"},
            )
            assert resp.status == 200, await resp.text()
            data = await resp.json()
            assert data["object"] == "response"
            assert data["status"] == "completed"
            assert data["output"][0]["content"][0]["type"] == "output_text"
            rid = data["id"]
            # fetch
            resp = await client.get(f"/v1/responses/{rid}")
            assert resp.status == 200
            # input items
            resp = await client.get(f"/v1/responses/{rid}/input_items")
            items = (await resp.json())["data"]
            assert any(i["role"] == "system" for i in items)
            # delete
            resp = await client.delete(f"/v1/responses/{rid}")
            assert resp.status == 200
            resp = await client.get(f"/v1/responses/{rid}")
            assert resp.status == 404
        finally:
            await stop_all(client, engines)

    runner(run())


def test_conversations_crud_and_context(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post("/v1/conversations", json={"metadata": {"topic": "t"}})
            conv = await resp.json()
            cid = conv["id"]
            resp = await client.post(
                f"/v1/conversations/{cid}/items",
                json={"items": [{"type": "message", "role": "user", "content": "earlier context"}]},
            )
            assert resp.status == 200
            # run a response against the conversation: history + new turn recorded
            resp = await client.post(
                "/v1/responses",
                json={"model": "mock-model", "input": "next turn", "max_output_tokens": 2,
                      "conversation": cid},
            )
            assert resp.status == 200
            resp = await client.get(f"/v1/conversations/{cid}/items")
            items = (await resp.json())["data"]
            roles = [i.get("role") for i in items]
            assert "assistant" in roles and len(items) >= 3
            # item get/delete
            iid = items[0]["id"]
            resp = await client.get(f"/v1/conversations/{cid}/items/{iid}")
            assert resp.status == 200
            resp = await client.delete(f"/v1/conversations/{cid}/items/{iid}")
            assert resp.status == 200
            resp = await client.delete(f"/v1/conversations/{cid}")
            assert resp.status == 200
            resp = await client.get(f"/v1/conversations/{cid}")
            assert resp.status == 404
        finally:
            await stop_all(client, engines)

    runner(run())


def test_anthropic_messages_unary(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post(
                "/v1/messages",
                json={"model": "mock-model", "max_tokens": 4, "system": "sys prompt",
                      "messages": [{"role": "user", "content": "hi"}]},
            )
            assert resp.status == 200, await resp.text()
            data = await resp.json()
            assert data["type"] == "message"
            assert data["role"] == "assistant"
            assert data["content"][0]["type"] == "text"
            assert data["stop_reason"] in ("end_turn", "max_tokens")
            assert data["usage"]["output_tokens"] == 4
        finally:
            await stop_all(client, engines)

    runner(run())


def test_anthropic_messages_stream(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post(
                "/v1/messages",
                json={"model": "mock-model", "max_tokens": 3, "stream": True,
                      "messages": [{"role": "user", "content": [{"type": "text", "text": "hi"}]}]},
            )
            assert resp.status == 200
            events = []
            async for line in resp.content:
                line = line.decode().strip()
                if line.startswith("event: "):
                    events.append(line[7:])
            assert events[0] == "message_start"
            assert "content_block_delta" in events
            assert events[-1] == "message_stop"
        finally:
            await stop_all(client, engines)

    runner(run())


def test_anthropic_requires_messages(runner):
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            resp = await client.post("/v1/messages", json={"model": "mock-model", "max_tokens": 4})
            assert resp.status == 400
        finally:
            await stop_all(client, engines)

    runner(run())


def test_interactions_api(runner):
    """Gemini Interactions API (reference routers/gemini/, interactions.rs)."""
    async def run():
        ctx, engines = make_ctx()
        client = await start_client(ctx, engines)
        try:
            r = await client.post("/v1/interactions", json={"input": "hello"})
            assert r.status == 400  # model or agent required
            r = await client.post(
                "/v1/interactions",
                json={"model": "mock-model", "input": "what is the answer to everything?",
                      "generation_config": {"max_output_tokens": 8}},
            )
            assert r.status == 200
            first = await r.json()
            assert first["object"] == "interaction" and first["status"] == "completed"
            assert first["outputs"] and first["outputs"][0]["type"] == "text"
            assert first["usage"]["output_tokens"] > 0
            # chained turn via previous_interaction_id
            r = await client.post(
                "/v1/interactions",
                json={"model": "mock-model", "input": "and why?",
                      "previous_interaction_id": first["id"],
                      "generation_config": {"max_output_tokens": 8}},
            )
            assert r.status == 200
            second = await r.json()
            assert second["previous_interaction_id"] == first["id"]
            # system_instruction + list input forms
            r = await client.post(
                "/v1/interactions",
                json={"model": "mock-model", "system_instruction": "be brief",
                      "input": [{"role": "user", "parts": [{"text": "hi"}]}],
                      "generation_config": {"max_output_tokens": 4}},
            )
            assert r.status == 200
        finally:
            await stop_all(client, engines)

    runner(run())
