"""Priority scheduler + tenant rate limit tests (reference:
model_gateway/tests scheduler_test.rs + rate_limiting suites)."""
import asyncio

import pytest

from smg_amd.rate_limit.tenant import RateLimitManager, TenantLimit, TenantRateLimitSettings
from smg_amd.scheduler.engine import CLASSES, PriorityScheduler, SchedulerConfig


def sched(**kw):
    queue_size = kw.pop("queue_size", 8)
    cfg = SchedulerConfig(min_slots=kw.pop("slots", 4),
                          queue_timeout_secs=kw.pop("timeout", 0.5), **kw)
    for c in cfg.classes.values():
        c.queue_size = queue_size
    return PriorityScheduler(cfg)


class TestScheduler:
    def test_admit_within_capacity(self, runner):
        async def run():
            s = sched(slots=4)
            assert await s.acquire("default")
            assert s.total_in_use() == 1
            s.release("default")
            assert s.total_in_use() == 0

        runner(run())

    def test_class_reservation_blocks_bulk(self, runner):
        async def run():
            s = sched(slots=10, timeout=0.1)
            # bulk may only use its own 10% slice
            assert s.class_limit("bulk") < s.class_limit("default") < s.class_limit("system")
            got = 0
            for _ in range(s.class_limit("bulk")):
                if await s.acquire("bulk"):
                    got += 1
            assert got == s.class_limit("bulk")
            assert not await s.acquire("bulk")  # exhausted its slice
            assert await s.acquire("system")  # higher class still admitted

        runner(run())

    def test_queue_then_dispatch_on_release(self, runner):
        async def run():
            s = sched(slots=1, timeout=2.0)
            assert await s.acquire("default")
            waiter = asyncio.ensure_future(s.acquire("default"))
            await asyncio.sleep(0.05)
            assert not waiter.done()
            s.release("default")
            assert await waiter

        runner(run())

    def test_priority_dispatch_order(self, runner):
        async def run():
            s = sched(slots=1, timeout=2.0)
            assert await s.acquire("system")
            bulk_w = asyncio.ensure_future(s.acquire("bulk"))
            await asyncio.sleep(0.02)
            inter_w = asyncio.ensure_future(s.acquire("interactive"))
            await asyncio.sleep(0.02)
            s.release("system")
            await asyncio.sleep(0.02)
            assert inter_w.done() and inter_w.result()  # interactive wins despite arriving later
            assert not bulk_w.done()
            s.release("interactive")
            await asyncio.wait_for(bulk_w, 2.0)

        runner(run())

    def test_class_queue_overflow_rejects(self, runner):
        async def run():
            s = sched(slots=1, queue_size=1, timeout=0.5)
            assert await s.acquire("bulk")
            w1 = asyncio.ensure_future(s.acquire("bulk"))  # fills bulk's queue
            await asyncio.sleep(0.02)
            assert not await s.acquire("bulk")  # fixed FIFO depth: reject
            assert s.rejected == 1
            s.release("bulk")
            assert await asyncio.wait_for(w1, 2.0)

        runner(run())

    def test_inflight_preemption_50ms_budget(self, runner):
        """engine.rs:193-232: a saturated higher-class arrival cancels the
        newest LOWER-class inflight request and takes its slot within the
        50 ms wait budget; the victim's admit() answers 429 preempted."""
        from aiohttp import web
        from aiohttp.test_utils import TestClient, TestServer

        async def run():
            s = sched(slots=1, timeout=2.0)

            async def slow_handler(request):
                await asyncio.sleep(30)
                return web.Response(text="done")

            async def admitted(request):
                return await s.admit(request, slow_handler)

            app = web.Application()
            app.router.add_get("/work", admitted)
            client = TestClient(TestServer(app))
            await client.start_server()
            try:
                victim = asyncio.ensure_future(
                    client.get("/work", headers={"x-smg-priority": "bulk"}))
                for _ in range(100):
                    if s.total_in_use() == 1:
                        break
                    await asyncio.sleep(0.01)
                assert s.stats()["inflight"] == 1
                t0 = asyncio.get_event_loop().time()
                assert await s.acquire("interactive")  # preempts the bulk body
                took = asyncio.get_event_loop().time() - t0
                assert took < 1.0  # well within budget+scheduling slack
                assert s.preempted_inflight == 1
                resp = await asyncio.wait_for(victim, 5.0)
                assert resp.status == 429
                assert resp.headers.get("x-smg-preempted") == "1"
                s.release("interactive")
            finally:
                await client.close()

        runner(run())

    def test_bulk_never_preempts(self, runner):
        async def run():
            s = sched(slots=1, timeout=0.1)
            assert await s.acquire("default")
            assert not await s.acquire("bulk")  # queues then times out
            assert s.preempted_inflight == 0

        runner(run())

    def test_capacity_shrink_clamps_priority_ordered(self):
        s = sched(slots=8)
        for c, cc in s.config.classes.items():
            cc.floor = {"system": 4, "interactive": 4, "default": 4, "bulk": 4}[c]
        s.apply_new_capacity(10)  # desired 16 > 10: clamp fills system->bulk
        assert s.reserved("system") == 4
        assert s.reserved("interactive") == 4
        assert s.reserved("default") == 2
        assert s.reserved("bulk") == 0
        s.apply_new_capacity(100)  # grow restores the full shares
        assert s.reserved("default") >= 4

    def test_yaml_class_config(self, tmp_path):
        p = tmp_path / "sched.yaml"
        p.write_text(
            "per_worker_concurrency: 8\n"
            "classes:\n  interactive: {floor: 3, share: 0.4, queue_size: 9}\n"
            "reservations:\n  bulk: 0.05\n"
            "tenants:\n  vip: interactive\n"
        )
        cfg = SchedulerConfig.from_yaml(str(p))
        assert cfg.classes["interactive"].floor == 3
        assert cfg.classes["interactive"].queue_size == 9
        assert cfg.classes["bulk"].share == 0.05
        assert cfg.tenant_classes["vip"] == "interactive"

    def test_classify(self):
        cfg = SchedulerConfig(tenant_classes={"vip": "interactive"})
        s = PriorityScheduler(cfg)
        assert s.classify("vip", None) == "interactive"
        assert s.classify("other", None) == "default"
        assert s.classify(None, "bulk") == "bulk"

    def test_capacity_recompute_from_registry(self):
        from smg_amd.workers.registry import WorkerRegistry
        from smg_amd.workers.worker import Worker

        reg = WorkerRegistry()
        s = PriorityScheduler(SchedulerConfig(per_worker_concurrency=10, min_slots=5), reg)
        assert s.capacity == max(5, 10)
        reg.register(Worker("http://a:1"))
        reg.register(Worker("http://b:2"))
        assert s.capacity == 20


class TestTenantRateLimit:
    def settings(self):
        return TenantRateLimitSettings(
            default=TenantLimit(requests_per_minute=5),
            tenants={"acme": TenantLimit(requests_per_minute=2, tokens_per_minute=100, max_concurrent=1)},
        )

    def test_request_limit(self):
        m = RateLimitManager(self.settings(), clock=lambda: 0)
        r1 = m.reserve("acme")
        m.settle(r1)
        r2 = m.reserve("acme")
        m.settle(r2)
        assert m.reserve("acme") is None  # 2/min exhausted

    def test_token_limit(self):
        m = RateLimitManager(self.settings(), clock=lambda: 0)
        r = m.reserve("acme", est_tokens=90)
        m.settle(r)
        assert m.reserve("acme", est_tokens=50) is None

    def test_concurrency_limit(self):
        m = RateLimitManager(self.settings(), clock=lambda: 0)
        r = m.reserve("acme")
        assert m.reserve("acme") is None  # max_concurrent 1
        m.settle(r)
        assert m.reserve("acme") is not None

    def test_epoch_rollover(self):
        now = [0.0]
        m = RateLimitManager(self.settings(), clock=lambda: now[0])
        m.settle(m.reserve("acme"))
        m.settle(m.reserve("acme"))
        assert m.reserve("acme") is None
        now[0] = 61.0
        assert m.reserve("acme") is not None

    def test_remote_usage_counts(self):
        m = RateLimitManager(self.settings(), clock=lambda: 0)
        m.observe_remote_usage("acme", 0, 80)
        assert m.reserve("acme", est_tokens=30) is None  # 80 remote + 30 > 100

    def test_settle_adjusts_actual_tokens(self):
        m = RateLimitManager(self.settings(), clock=lambda: 0)
        r = m.reserve("acme", est_tokens=50)
        m.settle(r, actual_tokens=10)
        assert m.reserve("acme", est_tokens=80) is not None
