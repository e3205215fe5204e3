"""MCP (Model Context Protocol) client (reference: crates/mcp — orchestrator +
sessions + connection pool (core/), transports stdio/SSE/streamable-HTTP,
tool inventory with qualified names (inventory/), approval engine (approval/),
per-tenant bindings (tenant.rs)).

JSON-RPC 2.0 over newline-delimited stdio or streamable HTTP.  Tools are
qualified as "<server>.<tool>" in the inventory; the approval engine gates
calls by policy (always_allow / always_deny / interactive callback).
"""
from __future__ import annotations

import asyncio
import itertools
import json
import logging
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

log = logging.getLogger("smg.mcp")


class McpError(RuntimeError):
    pass


@dataclass
class McpServerConfig:
    name: str
    transport: str = "stdio"  # stdio | http
    command: Optional[List[str]] = None  # stdio
    url: Optional[str] = None  # http
    env: Dict[str, str] = field(default_factory=dict)
    allowed_tenants: Optional[List[str]] = None  # None = all


# ---------------------------------------------------------------------------
# transports
# ---------------------------------------------------------------------------
class StdioTransport:
    def __init__(self, command: List[str], env: Optional[Dict[str, str]] = None):
        self.command = command
        self.env = env
        self._proc: Optional[asyncio.subprocess.Process] = None
        self._ids = itertools.count(1)
        self._pending: Dict[int, asyncio.Future] = {}
        self._reader_task: Optional[asyncio.Task] = None

    async def start(self) -> None:
        import os

        env = dict(os.environ)
        env.update(self.env or {})
        self._proc = await asyncio.create_subprocess_exec(
            *self.command,
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.DEVNULL,
            env=env,
        )
        self._reader_task = asyncio.ensure_future(self._read_loop())

    async def _read_loop(self) -> None:
        try:
            while True:
                line = await self._proc.stdout.readline()
                if not line:
                    break
                try:
                    msg = json.loads(line)
                except json.JSONDecodeError:
                    continue
                mid = msg.get("id")
                fut = self._pending.pop(mid, None)
                if fut is not None and not fut.done():
                    fut.set_result(msg)
        except asyncio.CancelledError:
            pass
        for fut in self._pending.values():
            if not fut.done():
                fut.set_exception(McpError("mcp server exited"))

    async def request(self, method: str, params: Optional[dict] = None, timeout: float = 30.0) -> dict:
        if self._proc is None or self._proc.returncode is not None:
            raise McpError("stdio transport not running")
        mid = next(self._ids)
        msg = {"jsonrpc": "2.0", "id": mid, "method": method, "params": params or {}}
        fut = asyncio.get_event_loop().create_future()
        self._pending[mid] = fut
        self._proc.stdin.write(json.dumps(msg).encode() + b"\n")
        await self._proc.stdin.drain()
        resp = await asyncio.wait_for(fut, timeout)
        if "error" in resp:
            raise McpError(str(resp["error"]))
        return resp.get("result", {})

    async def close(self) -> None:
        if self._reader_task:
            self._reader_task.cancel()
        if self._proc and self._proc.returncode is None:
            self._proc.terminate()
            try:
                await asyncio.wait_for(self._proc.wait(), 3)
            except asyncio.TimeoutError:
                self._proc.kill()


class HttpTransport:
    """Streamable-HTTP MCP transport: POST JSON-RPC to the server URL, over
    a pooled connector (reference core/ connection pool)."""

    def __init__(self, url: str, pool_size: int = 8):
        self.url = url
        self.pool_size = pool_size
        self._session = None
        self._ids = itertools.count(1)

    async def start(self) -> None:
        import aiohttp

        self._session = aiohttp.ClientSession(
            connector=aiohttp.TCPConnector(limit=self.pool_size),
            timeout=aiohttp.ClientTimeout(total=30),
        )

    async def request(self, method: str, params: Optional[dict] = None, timeout: float = 30.0) -> dict:
        msg = {"jsonrpc": "2.0", "id": next(self._ids), "method": method, "params": params or {}}
        async with self._session.post(self.url, json=msg) as resp:
            if resp.status != 200:
                raise McpError(f"mcp http {resp.status}")
            data = await resp.json()
        if "error" in data:
            raise McpError(str(data["error"]))
        return data.get("result", {})

    async def close(self) -> None:
        if self._session:
            await self._session.close()


class SseTransport:
    """MCP HTTP+SSE transport (reference core/transports SSE): a long-lived
    GET to the SSE endpoint carries server->client messages; the server's
    initial `endpoint` event names the URL to POST client->server JSON-RPC.
    Responses are matched to requests by id off the event stream."""

    def __init__(self, url: str):
        self.url = url
        self._session = None
        self._post_url: Optional[str] = None
        self._ids = itertools.count(1)
        self._pending: Dict[object, asyncio.Future] = {}
        self._reader_task: Optional[asyncio.Task] = None
        self._endpoint_ready: Optional[asyncio.Event] = None

    async def start(self) -> None:
        import aiohttp

        self._session = aiohttp.ClientSession(timeout=aiohttp.ClientTimeout(total=None, sock_read=None))
        self._endpoint_ready = asyncio.Event()
        self._reader_task = asyncio.ensure_future(self._read_stream())
        await asyncio.wait_for(self._endpoint_ready.wait(), 10.0)

    async def _read_stream(self) -> None:
        from urllib.parse import urljoin

        try:
            async with self._session.get(self.url, headers={"Accept": "text/event-stream"}) as resp:
                if resp.status != 200:
                    raise McpError(f"mcp sse {resp.status}")
                event, data_lines = None, []
                async for raw in resp.content:
                    line = raw.decode("utf-8", "replace").rstrip("\n\r")
                    if line.startswith("event:"):
                        event = line[6:].strip()
                    elif line.startswith("data:"):
                        data_lines.append(line[5:].strip())
                    elif line == "":
                        data = "\n".join(data_lines)
                        data_lines = []
                        if event == "endpoint":
                            self._post_url = urljoin(self.url, data)
                            self._endpoint_ready.set()
                        elif data:
                            try:
                                msg = json.loads(data)
                            except json.JSONDecodeError:
                                continue
                            fut = self._pending.pop(msg.get("id"), None)
                            if fut is not None and not fut.done():
                                fut.set_result(msg)
                        event = None
        except asyncio.CancelledError:
            pass
        except Exception as exc:
            log.debug("mcp sse stream ended: %s", exc)
        for fut in self._pending.values():
            if not fut.done():
                fut.set_exception(McpError("mcp sse stream closed"))
        self._pending.clear()

    async def request(self, method: str, params: Optional[dict] = None, timeout: float = 30.0) -> dict:
        if self._post_url is None or (self._reader_task and self._reader_task.done()):
            raise McpError("sse transport not running")
        mid = next(self._ids)
        msg = {"jsonrpc": "2.0", "id": mid, "method": method, "params": params or {}}
        fut = asyncio.get_event_loop().create_future()
        self._pending[mid] = fut
        async with self._session.post(self._post_url, json=msg) as resp:
            if resp.status not in (200, 202):
                self._pending.pop(mid, None)
                raise McpError(f"mcp sse post {resp.status}")
        resp_msg = await asyncio.wait_for(fut, timeout)
        if "error" in resp_msg:
            raise McpError(str(resp_msg["error"]))
        return resp_msg.get("result", {})

    async def close(self) -> None:
        if self._reader_task:
            self._reader_task.cancel()
            try:
                await self._reader_task
            except (asyncio.CancelledError, Exception):
                pass
        if self._session:
            await self._session.close()


class ResilientTransport:
    """Reconnect-with-backoff wrapper (reference core/ session reconnect;
    backoff ladder mirrors kv_event_monitor.rs 100ms -> 30s).  A failed
    request tears the inner transport down, reconnects, replays the MCP
    `initialize` handshake, and retries the request once per attempt."""

    BACKOFF_S = (0.1, 0.5, 2.0, 10.0, 30.0)

    def __init__(self, make_transport, max_attempts: int = 4, on_reconnect=None):
        self._make = make_transport
        self._inner = None
        self.max_attempts = max_attempts
        self.reconnects = 0
        self._on_reconnect = on_reconnect  # async hook(transport) after handshake

    async def start(self) -> None:
        self._inner = self._make()
        await self._inner.start()

    async def _handshake(self) -> None:
        await self._inner.request(
            "initialize",
            {"protocolVersion": "2024-11-05", "capabilities": {},
             "clientInfo": {"name": "smg", "version": "0.1"}},
        )
        if self._on_reconnect is not None:
            await self._on_reconnect(self._inner)

    async def request(self, method: str, params: Optional[dict] = None, timeout: float = 30.0) -> dict:
        last: Optional[Exception] = None
        for attempt in range(self.max_attempts):
            if self._inner is None:
                await self.start()
            try:
                return await self._inner.request(method, params, timeout)
            except (McpError, asyncio.TimeoutError, ConnectionError, OSError) as exc:
                last = exc
                try:
                    await self._inner.close()
                except Exception:
                    pass
                self._inner = None
                if attempt + 1 >= self.max_attempts:
                    break
                await asyncio.sleep(self.BACKOFF_S[min(attempt, len(self.BACKOFF_S) - 1)])
                try:
                    await self.start()
                    self.reconnects += 1
                    await self._handshake()
                except Exception as exc2:
                    last = exc2
                    self._inner = None
        raise McpError(f"mcp request failed after {self.max_attempts} attempts: {last}")

    async def close(self) -> None:
        if self._inner is not None:
            await self._inner.close()
            self._inner = None


# ---------------------------------------------------------------------------
# approval engine (reference approval/: policy + interactive + audit)
# ---------------------------------------------------------------------------
class ApprovalEngine:
    def __init__(
        self,
        mode: str = "auto",  # auto | deny | interactive
        allow: Optional[List[str]] = None,
        deny: Optional[List[str]] = None,
        interactive_cb: Optional[Callable[[str, dict], bool]] = None,
    ):
        self.mode = mode
        self.allow = set(allow or [])
        self.deny = set(deny or [])
        self.interactive_cb = interactive_cb
        self.audit: List[dict] = []

    def check(self, qualified_tool: str, args: dict) -> bool:
        decision: bool
        if qualified_tool in self.deny:
            decision = False
        elif qualified_tool in self.allow:
            decision = True
        elif self.mode == "deny":
            decision = False
        elif self.mode == "interactive" and self.interactive_cb is not None:
            decision = bool(self.interactive_cb(qualified_tool, args))
        else:
            decision = True
        self.audit.append({"tool": qualified_tool, "approved": decision})
        return decision


# ---------------------------------------------------------------------------
# orchestrator (reference core/ + inventory/)
# ---------------------------------------------------------------------------
class McpOrchestrator:
    def __init__(self, approval: Optional[ApprovalEngine] = None):
        self.servers: Dict[str, object] = {}  # name -> transport
        self.configs: Dict[str, McpServerConfig] = {}
        self.inventory: Dict[str, dict] = {}  # qualified name -> tool schema
        self.approval = approval or ApprovalEngine()

    @classmethod
    def from_yaml(cls, path: Optional[str]) -> "McpOrchestrator":
        orch = cls()
        if not path:
            return orch
        import yaml

        with open(path) as f:
            data = yaml.safe_load(f) or {}
        for s in data.get("servers", []):
            orch.configs[s["name"]] = McpServerConfig(
                name=s["name"],
                transport=s.get("transport", "stdio"),
                command=s.get("command"),
                url=s.get("url"),
                env=s.get("env") or {},
                allowed_tenants=s.get("allowed_tenants"),
            )
        ap = data.get("approval") or {}
        orch.approval = ApprovalEngine(
            mode=ap.get("mode", "auto"), allow=ap.get("allow"), deny=ap.get("deny")
        )
        return orch

    async def register_server(self, cfg: McpServerConfig) -> int:
        """Connect + list tools; returns number of tools discovered.  Every
        transport is wrapped in ResilientTransport: a dead session reconnects
        with backoff, replays initialize, and refreshes this server's
        inventory (reference core/ session + reconnect)."""

        def make():
            if cfg.transport == "stdio":
                return StdioTransport(cfg.command or [], cfg.env)
            if cfg.transport == "http":
                return HttpTransport(cfg.url or "")
            if cfg.transport == "sse":
                return SseTransport(cfg.url or "")
            raise McpError(f"unknown transport {cfg.transport}")

        async def refresh_inventory(inner) -> None:
            result = await inner.request("tools/list")
            for q in [q for q in self.inventory if q.startswith(cfg.name + ".")]:
                del self.inventory[q]
            for tool in result.get("tools", []):
                self.inventory[f"{cfg.name}.{tool['name']}"] = tool

        t = ResilientTransport(make, on_reconnect=refresh_inventory)
        await t.start()
        await t._handshake()  # initialize + initial tools/list
        self.servers[cfg.name] = t
        self.configs[cfg.name] = cfg
        return sum(1 for q in self.inventory if q.startswith(cfg.name + "."))

    async def start_all(self) -> None:
        for cfg in list(self.configs.values()):
            if cfg.name not in self.servers:
                try:
                    await self.register_server(cfg)
                except Exception as exc:
                    log.warning("mcp server %s failed to start: %s", cfg.name, exc)

    def tools_for_tenant(self, tenant: Optional[str] = None) -> List[dict]:
        out = []
        for qname, tool in self.inventory.items():
            server = qname.split(".", 1)[0]
            cfg = self.configs.get(server)
            if cfg and cfg.allowed_tenants is not None and tenant not in cfg.allowed_tenants:
                continue
            out.append(
                {
                    "type": "function",
                    "function": {
                        "name": qname,
                        "description": tool.get("description", ""),
                        "parameters": tool.get("inputSchema", {}),
                    },
                }
            )
        return out

    async def call_tool(self, qualified: str, args: dict, tenant: Optional[str] = None) -> dict:
        if qualified not in self.inventory:
            raise McpError(f"unknown tool {qualified}")
        server, tool = qualified.split(".", 1)
        cfg = self.configs.get(server)
        if cfg and cfg.allowed_tenants is not None and tenant not in cfg.allowed_tenants:
            raise McpError(f"tenant {tenant} not allowed on {server}")
        if not self.approval.check(qualified, args):
            raise McpError(f"tool call {qualified} denied by approval policy")
        t = self.servers[server]
        return await t.request("tools/call", {"name": tool, "arguments": args})

    async def remove_server(self, name: str) -> bool:
        t = self.servers.pop(name, None)
        self.configs.pop(name, None)
        for q in [q for q in self.inventory if q.startswith(name + ".")]:
            del self.inventory[q]
        if t is not None:
            await t.close()
            return True
        return False

    async def shutdown(self) -> None:
        for name in list(self.servers):
            await self.remove_server(name)


# ---------------------------------------------------------------------------
# Responses-API tool loop (reference openai/mcp/tool_loop.rs: model <-> tool
# execution cycles with approval)
# ---------------------------------------------------------------------------
async def run_tool_loop(
    orchestrator: McpOrchestrator,
    chat_fn,  # async (messages, tools) -> assistant message dict
    messages: List[dict],
    tenant: Optional[str] = None,
    max_rounds: int = 8,
) -> List[dict]:
    """Drive model <-> tool cycles until the model stops calling tools.
    Returns the grown message list (assistant + tool results appended)."""
    tools = orchestrator.tools_for_tenant(tenant)
    for _ in range(max_rounds):
        msg = await chat_fn(messages, tools)
        messages.append(msg)
        calls = msg.get("tool_calls") or []
        if not calls:
            return messages
        for call in calls:
            fn = call.get("function", {})
            try:
                args = json.loads(fn.get("arguments") or "{}")
            except json.JSONDecodeError:
                args = {}
            try:
                result = await orchestrator.call_tool(fn.get("name", ""), args, tenant)
                content = json.dumps(result.get("content", result))
            except McpError as exc:
                content = json.dumps({"error": str(exc)})
            messages.append({"role": "tool", "tool_call_id": call.get("id"), "content": content})
    return messages
