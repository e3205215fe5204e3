"""Cloud provider translation tests: mock anthropic/gemini upstreams behind
the HTTP router (reference: routers/openai provider registry behavior)."""
import json

import pytest
from aiohttp import web
from aiohttp.test_utils import TestServer

from smg_amd.routers.providers import AnthropicProvider, GeminiProvider, dispatch_to_provider
from smg_amd.workers.worker import Worker


class TestTranslation:
    def test_anthropic_request(self):
        p = AnthropicProvider()
        chat = {"model": "claude-x", "max_tokens": 10, "stop": ["END"],
                "messages": [{"role": "system", "content": "be terse"},
                             {"role": "user", "content": "hi"}]}
        out = p.translate_request(chat)
        assert out["system"] == "be terse"
        assert out["messages"] == [{"role": "user", "content": "hi"}]
        assert out["stop_sequences"] == ["END"]

    def test_anthropic_response(self):
        p = AnthropicProvider()
        vendor = {"model": "claude-x", "stop_reason": "end_turn",
                  "content": [{"type": "text", "text": "hello"}],
                  "usage": {"input_tokens": 3, "output_tokens": 2}}
        out = p.translate_response(vendor, {"model": "claude-x"})
        assert out["choices"][0]["message"]["content"] == "hello"
        assert out["usage"]["total_tokens"] == 5

    def test_gemini_request(self):
        p = GeminiProvider()
        chat = {"model": "gemini-pro", "max_tokens": 8,
                "messages": [{"role": "system", "content": "s"},
                             {"role": "user", "content": "u"},
                             {"role": "assistant", "content": "a"}]}
        out = p.translate_request(chat)
        assert out["systemInstruction"]["parts"][0]["text"] == "s"
        assert [c["role"] for c in out["contents"]] == ["user", "model"]
        assert out["generationConfig"]["maxOutputTokens"] == 8

    def test_gemini_response(self):
        p = GeminiProvider()
        vendor = {"candidates": [{"content": {"parts": [{"text": "out"}]}, "finishReason": "STOP"}],
                  "usageMetadata": {"promptTokenCount": 1, "candidatesTokenCount": 1, "totalTokenCount": 2}}
        out = p.translate_response(vendor, {"model": "gemini-pro"})
        assert out["choices"][0]["message"]["content"] == "out"


def test_dispatch_to_mock_anthropic_upstream(runner):
    async def run():
        import aiohttp

        seen = {}

        async def handler(request):
            seen["body"] = await request.json()
            seen["key"] = request.headers.get("x-api-key")
            return web.json_response({
                "model": "claude-x", "stop_reason": "end_turn",
                "content": [{"type": "text", "text": "upstream says hi"}],
                "usage": {"input_tokens": 2, "output_tokens": 3},
            })

        app = web.Application()
        app.router.add_post("/v1/messages", handler)
        server = TestServer(app)
        await server.start_server()
        worker = Worker(f"http://127.0.0.1:{server.port}", model_id="claude-x",
                        labels={"provider": "anthropic"}, api_key="sk-ant-test")
        async with aiohttp.ClientSession() as session:
            out = await dispatch_to_provider(session, worker,
                                             {"model": "claude-x", "max_tokens": 5,
                                              "messages": [{"role": "user", "content": "hi"}]})
        await server.close()
        assert out["choices"][0]["message"]["content"] == "upstream says hi"
        assert seen["key"] == "sk-ant-test"
        assert seen["body"]["max_tokens"] == 5

    runner(run())


def test_dispatch_to_mock_gemini_upstream(runner):
    async def run():
        import aiohttp

        async def handler(request):
            assert "generateContent" in request.path
            return web.json_response({
                "candidates": [{"content": {"parts": [{"text": "g"}]}, "finishReason": "STOP"}],
                "usageMetadata": {"totalTokenCount": 2},
            })

        app = web.Application()
        app.router.add_post("/v1beta/models/{tail:.*}", handler)
        server = TestServer(app)
        await server.start_server()
        worker = Worker(f"http://127.0.0.1:{server.port}", model_id="gemini-pro",
                        labels={"provider": "gemini"}, api_key="g-key")
        async with aiohttp.ClientSession() as session:
            out = await dispatch_to_provider(session, worker,
                                             {"model": "gemini-pro",
                                              "messages": [{"role": "user", "content": "q"}]})
        await server.close()
        assert out["choices"][0]["message"]["content"] == "g"

    runner(run())
