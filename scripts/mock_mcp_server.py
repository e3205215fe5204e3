#!/usr/bin/env python3
"""Minimal stdio MCP server for tests (reference: model_gateway/tests/common/
mock_mcp_server.rs).  Newline-delimited JSON-RPC 2.0; tools: echo, add."""
import json
import sys

TOOLS = [
    {
        "name": "echo",
        "description": "echo the input back",
        "inputSchema": {"type": "object", "properties": {"text": {"type": "string"}}},
    },
    {
        "name": "add",
        "description": "add two numbers",
        "inputSchema": {
            "type": "object",
            "properties": {"a": {"type": "number"}, "b": {"type": "number"}},
            "required": ["a", "b"],
        },
    },
]


def handle(msg):
    method = msg.get("method")
    if method == "initialize":
        return {"protocolVersion": "2024-11-05", "capabilities": {"tools": {}}, "serverInfo": {"name": "mock"}}
    if method == "tools/list":
        return {"tools": TOOLS}
    if method == "tools/call":
        params = msg.get("params", {})
        name = params.get("name")
        args = params.get("arguments", {})
        if name == "echo":
            return {"content": [{"type": "text", "text": args.get("text", "")}]}
        if name == "add":
            return {"content": [{"type": "text", "text": str(args.get("a", 0) + args.get("b", 0))}]}
        raise ValueError(f"unknown tool {name}")
    raise ValueError(f"unknown method {method}")


def main():
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        msg = json.loads(line)
        try:
            result = handle(msg)
            out = {"jsonrpc": "2.0", "id": msg.get("id"), "result": result}
        except Exception as e:
            out = {"jsonrpc": "2.0", "id": msg.get("id"), "error": {"code": -32000, "message": str(e)}}
        sys.stdout.write(json.dumps(out) + "\n")
        sys.stdout.flush()


if __name__ == "__main__":
    main()
