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
