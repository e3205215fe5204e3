"""JobQueue + audit trail tests (reference workflow/job_queue.rs, auth audit.rs)."""
import pytest

from smg_amd.config import PolicyConfig, RouterConfig
from smg_amd.plugins import PluginManager
from smg_amd.server.app_context import AppContext
from smg_amd.server.jobs import JobKind, JobQueue


def make_ctx():
    cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
    ctx = AppContext(cfg)
    ctx.plugins = PluginManager()
    return ctx


def test_add_remove_worker_serialized(runner):
    async def run():
        ctx = make_ctx()
        q = JobQueue(ctx)
        w = await q.submit(JobKind.ADD_WORKER, {"url": "http://j1:8000", "model_id": "m"}, actor="admin")
        assert ctx.worker_registry.get_by_url("http://j1:8000") is not None
        await q.submit(JobKind.REMOVE_WORKER, {"url": "http://j1:8000"}, actor="admin")
        assert ctx.worker_registry.get_by_url("http://j1:8000") is None
        actions = [r["action"] for r in q.audit.records]
        assert actions == ["add_worker", "remove_worker"]
        assert all(r["actor"] == "admin" and r["ok"] for r in q.audit.records)
        await q.stop()

    runner(run())


def test_failed_job_audited_and_raises(runner):
    async def run():
        ctx = make_ctx()
        q = JobQueue(ctx)
        with pytest.raises(KeyError):
            await q.submit(JobKind.REMOVE_WORKER, {"url": "http://missing:1"})
        assert q.audit.records[-1]["ok"] is False
        await q.stop()

    runner(run())


def test_add_tokenizer_job(runner):
    async def run():
        ctx = make_ctx()
        q = JobQueue(ctx)
        name = await q.submit(JobKind.ADD_TOKENIZER, {"name": "mock-tok", "path": "mock"})
        assert ctx.tokenizer_registry.get("mock-tok") is not None
        await q.stop()

    runner(run())


def test_jobs_execute_in_order(runner):
    async def run():
        import asyncio

        ctx = make_ctx()
        q = JobQueue(ctx)
        futures = [
            asyncio.ensure_future(q.submit(JobKind.ADD_WORKER, {"url": f"http://o{i}:1", "model_id": "m"}))
            for i in range(5)
        ]
        await asyncio.gather(*futures)
        urls = [w.url for w in ctx.worker_registry.all()]
        assert urls == [f"http://o{i}:1" for i in range(5)]
        await q.stop()

    runner(run())
