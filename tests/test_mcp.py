"""MCP client tests against the mock stdio server (reference mcp_test.rs)."""
import json
import os
import sys

import pytest

from smg_amd.mcp.client import (
    ApprovalEngine,
    McpError,
    McpOrchestrator,
    McpServerConfig,
    run_tool_loop,
)

SERVER_CMD = [sys.executable, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "scripts", "mock_mcp_server.py")]


def make_cfg(name="mock", **kw):
    return McpServerConfig(name=name, transport="stdio", command=SERVER_CMD, **kw)


def test_register_and_list(runner):
    async def run():
        orch = McpOrchestrator()
        n = await orch.register_server(make_cfg())
        assert n == 2
        assert "mock.echo" in orch.inventory
        tools = orch.tools_for_tenant()
        names = {t["function"]["name"] for t in tools}
        assert names == {"mock.echo", "mock.add"}
        await orch.shutdown()

    runner(run())


def test_call_tool(runner):
    async def run():
        orch = McpOrchestrator()
        await orch.register_server(make_cfg())
        res = await orch.call_tool("mock.add", {"a": 2, "b": 3})
        assert res["content"][0]["text"] == "5"
        await orch.shutdown()

    runner(run())


def test_approval_deny(runner):
    async def run():
        orch = McpOrchestrator(ApprovalEngine(mode="deny", allow=["mock.echo"]))
        await orch.register_server(make_cfg())
        res = await orch.call_tool("mock.echo", {"text": "hi"})
        assert res["content"][0]["text"] == "hi"
        with pytest.raises(McpError):
            await orch.call_tool("mock.add", {"a": 1, "b": 1})
        assert orch.approval.audit[-1]["approved"] is False
        await orch.shutdown()

    runner(run())


def test_tenant_binding(runner):
    async def run():
        orch = McpOrchestrator()
        await orch.register_server(make_cfg(allowed_tenants=["acme"]))
        assert orch.tools_for_tenant("acme")
        assert orch.tools_for_tenant("other") == []
        with pytest.raises(McpError):
            await orch.call_tool("mock.echo", {"text": "x"}, tenant="other")
        await orch.shutdown()

    runner(run())


def test_tool_loop(runner):
    async def run():
        orch = McpOrchestrator()
        await orch.register_server(make_cfg())
        rounds = [0]

        async def chat_fn(messages, tools):
            rounds[0] += 1
            if rounds[0] == 1:
                return {
                    "role": "assistant",
                    "content": None,
                    "tool_calls": [
                        {"id": "c1", "type": "function",
                         "function": {"name": "mock.add", "arguments": json.dumps({"a": 4, "b": 5})}}
                    ],
                }
            return {"role": "assistant", "content": f"the answer is in {messages[-1]['content']}"}

        messages = await run_tool_loop(orch, chat_fn, [{"role": "user", "content": "add 4+5"}])
        assert rounds[0] == 2
        tool_msgs = [m for m in messages if m.get("role") == "tool"]
        assert tool_msgs and "9" in tool_msgs[0]["content"]
        assert messages[-1]["role"] == "assistant"
        await orch.shutdown()

    runner(run())


def test_remove_server(runner):
    async def run():
        orch = McpOrchestrator()
        await orch.register_server(make_cfg())
        assert await orch.remove_server("mock")
        assert orch.inventory == {}

    runner(run())


class TestSseTransport:
    """MCP HTTP+SSE transport (reference core/transports SSE): endpoint
    event handshake + id-matched responses off the event stream."""

    @staticmethod
    async def make_sse_server():
        import asyncio

        from aiohttp import web
        from aiohttp.test_utils import TestServer

        queues = {}

        async def sse(request):
            resp = web.StreamResponse(headers={"Content-Type": "text/event-stream"})
            await resp.prepare(request)
            q = asyncio.Queue()
            queues["q"] = q
            await resp.write(b"event: endpoint\ndata: /messages\n\n")
            try:
                while True:
                    msg = await q.get()
                    await resp.write(b"event: message\ndata: " + json.dumps(msg).encode() + b"\n\n")
            except (ConnectionResetError, asyncio.CancelledError):
                pass
            return resp

        async def messages(request):
            msg = await request.json()
            method = msg.get("method")
            if method == "initialize":
                result = {"serverInfo": {"name": "sse-mock"}}
            elif method == "tools/list":
                result = {"tools": [{"name": "ping", "description": "pong",
                                     "inputSchema": {"type": "object"}}]}
            elif method == "tools/call":
                result = {"content": [{"type": "text",
                                       "text": msg["params"]["arguments"].get("x", "") + "!"}]}
            else:
                result = {}
            await queues["q"].put({"jsonrpc": "2.0", "id": msg["id"], "result": result})
            return web.json_response({}, status=202)

        app = web.Application()
        app.router.add_get("/sse", sse)
        app.router.add_post("/messages", messages)
        server = TestServer(app)
        await server.start_server()
        return server

    def test_sse_register_and_call(self, runner):
        async def run():
            server = await self.make_sse_server()
            orch = McpOrchestrator()
            try:
                cfg = McpServerConfig(name="ssemock", transport="sse",
                                      url=f"http://127.0.0.1:{server.port}/sse")
                n = await orch.register_server(cfg)
                assert n == 1 and "ssemock.ping" in orch.inventory
                res = await orch.call_tool("ssemock.ping", {"x": "hello"})
                assert res["content"][0]["text"] == "hello!"
            finally:
                await orch.shutdown()
                await server.close()

        runner(run())


def test_reconnect_after_server_death(runner):
    """ResilientTransport: kill the stdio server mid-session; the next
    request reconnects with backoff, replays initialize, refreshes the
    inventory, and succeeds."""

    async def run():
        orch = McpOrchestrator()
        await orch.register_server(make_cfg())
        res = await orch.call_tool("mock.add", {"a": 1, "b": 1})
        assert res
        # kill the inner subprocess
        transport = orch.servers["mock"]
        transport._inner._proc.kill()
        await transport._inner._proc.wait()
        res = await orch.call_tool("mock.add", {"a": 2, "b": 5})
        assert res["content"][0]["text"] == "7"
        assert transport.reconnects >= 1
        assert "mock.add" in orch.inventory  # inventory refreshed on reconnect
        await orch.shutdown()

    runner(run())
