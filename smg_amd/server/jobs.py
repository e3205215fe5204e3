"""Control-plane job queue (reference: model_gateway/src/workflow/job_queue.rs:35-69
— Job enum Add/Update/RemoveWorker, InitializeWorkersFromConfig,
InitializeMcpServers, RegisterMcpServer, Add/RemoveTokenizer,
Add/RemoveWasmModule; serialized so control-plane mutations never race).

Typed async jobs drained by one worker task; submit() returns a future for
callers that need the result.  An audit trail records every mutation
(reference smg-auth audit.rs).
"""
from __future__ import annotations

import asyncio
import enum
import logging
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

log = logging.getLogger("smg.jobs")


class JobKind(str, enum.Enum):
    ADD_WORKER = "add_worker"
    UPDATE_WORKER = "update_worker"
    REMOVE_WORKER = "remove_worker"
    INIT_WORKERS_FROM_CONFIG = "init_workers_from_config"
    INIT_MCP_SERVERS = "init_mcp_servers"
    REGISTER_MCP_SERVER = "register_mcp_server"
    ADD_TOKENIZER = "add_tokenizer"
    REMOVE_TOKENIZER = "remove_tokenizer"
    ADD_PLUGIN = "add_plugin"
    REMOVE_PLUGIN = "remove_plugin"


@dataclass
class Job:
    kind: JobKind
    payload: Dict[str, Any] = field(default_factory=dict)
    actor: Optional[str] = None  # who asked (api key id / tenant)
    future: Optional[asyncio.Future] = None


class AuditLog:
    """Control-plane audit trail (reference audit.rs): ring of mutation records."""

    def __init__(self, capacity: int = 4096, enabled: bool = True):
        self.enabled = enabled
        self.capacity = capacity
        self.records: List[dict] = []

    def record(self, action: str, actor: Optional[str], detail: Dict[str, Any], ok: bool) -> None:
        if not self.enabled:
            return
        self.records.append(
            {"ts": round(time.time(), 3), "action": action, "actor": actor or "anonymous",
             "detail": detail, "ok": ok}
        )
        if len(self.records) > self.capacity:
            self.records = self.records[-self.capacity:]
        log.info("audit action=%s actor=%s ok=%s", action, actor, ok)


class JobQueue:
    def __init__(self, ctx, audit: Optional[AuditLog] = None):
        self.ctx = ctx
        self.audit = audit or AuditLog()
        self._q: asyncio.Queue = asyncio.Queue()
        self._task: Optional[asyncio.Task] = None
        self._handlers: Dict[JobKind, Callable] = {
            JobKind.ADD_WORKER: self._add_worker,
            JobKind.REMOVE_WORKER: self._remove_worker,
            JobKind.INIT_WORKERS_FROM_CONFIG: self._init_workers,
            JobKind.REGISTER_MCP_SERVER: self._register_mcp,
            JobKind.ADD_TOKENIZER: self._add_tokenizer,
            JobKind.REMOVE_TOKENIZER: self._remove_tokenizer,
            JobKind.ADD_PLUGIN: self._add_plugin,
            JobKind.REMOVE_PLUGIN: self._remove_plugin,
        }

    def start(self) -> None:
        if self._task is None:
            self._task = asyncio.ensure_future(self._drain())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
            self._task = None

    async def submit(self, kind: JobKind, payload: Dict[str, Any], actor: Optional[str] = None) -> Any:
        job = Job(kind, payload, actor, asyncio.get_event_loop().create_future())
        await self._q.put(job)
        self.start()
        return await job.future

    async def _drain(self) -> None:
        while True:
            job = await self._q.get()
            handler = self._handlers.get(job.kind)
            try:
                result = await handler(job.payload) if handler else None
                self.audit.record(job.kind.value, job.actor, job.payload, True)
                if job.future and not job.future.done():
                    job.future.set_result(result)
            except asyncio.CancelledError:
                raise
            except Exception as exc:
                self.audit.record(job.kind.value, job.actor, job.payload, False)
                if job.future and not job.future.done():
                    job.future.set_exception(exc)

    # ---- handlers ----------------------------------------------------------
    async def _add_worker(self, p: Dict[str, Any]):
        """Routes through the worker-registration workflow (reference
        src/workflow/steps/mod.rs:91 — classify → detect → discover →
        create, with per-step retries)."""
        from .worker_workflow import make_engine, register_worker_via_workflow

        engine = getattr(self.ctx, "workflow_engine", None)
        if engine is None:
            engine = make_engine(self.ctx)
            self.ctx.workflow_engine = engine
        return await register_worker_via_workflow(engine, p)

    async def _remove_worker(self, p: Dict[str, Any]):
        w = self.ctx.worker_registry.remove_by_url(p["url"])
        if w is None:
            raise KeyError(f"worker {p['url']} not found")
        return w

    async def _init_workers(self, p: Dict[str, Any]):
        self.ctx.init_workers_from_config()
        return len(self.ctx.worker_registry)

    async def _register_mcp(self, p: Dict[str, Any]):
        from ..mcp.client import McpOrchestrator, McpServerConfig

        if self.ctx.mcp is None:
            self.ctx.mcp = McpOrchestrator()
        cfg = McpServerConfig(
            name=p["name"], transport=p.get("transport", "stdio"),
            command=p.get("command"), url=p.get("url"),
            allowed_tenants=p.get("allowed_tenants"),
        )
        return await self.ctx.mcp.register_server(cfg)

    async def _add_tokenizer(self, p: Dict[str, Any]):
        if self.ctx.tokenizer_registry is None:
            from ..tokenizer.registry import TokenizerRegistry

            self.ctx.tokenizer_registry = TokenizerRegistry()
        return self.ctx.tokenizer_registry.load(p["name"], p["path"], p.get("chat_template")).name

    async def _remove_tokenizer(self, p: Dict[str, Any]):
        if not self.ctx.tokenizer_registry or not self.ctx.tokenizer_registry.remove(p["name"]):
            raise KeyError(f"tokenizer {p['name']} not found")
        return True

    async def _add_plugin(self, p: Dict[str, Any]):
        return self.ctx.plugins.add_module(p["path"], p.get("name"))

    async def _remove_plugin(self, p: Dict[str, Any]):
        if not self.ctx.plugins.remove_module(p["module_uuid"]):
            raise KeyError("module not found")
        return True
