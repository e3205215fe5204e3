"""Worker health + load monitoring (reference: model_gateway/src/worker/manager.rs
health loop, monitor.rs:146 WorkerMonitor / :207 WorkerLoadManager).

Two asyncio loops:
  * health loop: GET health endpoint per worker on an interval, flipping
    HealthState through the failure/success thresholds, optionally removing
    workers that stay unhealthy;
  * load loop: GET /get_loads per worker, updating queued/inflight tokens,
    KV utilization, generation throughput and DP-rank loads, then calling
    policy.update_loads.
"""
from __future__ import annotations

import asyncio
import logging
from typing import Optional

import aiohttp

from ..config import HealthCheckConfig
from .registry import WorkerRegistry
from .worker import HealthState, Worker

log = logging.getLogger("smg.worker.monitor")


class WorkerMonitor:
    def __init__(
        self,
        registry: WorkerRegistry,
        health_config: Optional[HealthCheckConfig] = None,
        load_interval_secs: float = 5.0,
        session: Optional[aiohttp.ClientSession] = None,
        policy_registry=None,
    ):
        self.registry = registry
        self.hc = health_config or HealthCheckConfig()
        self.load_interval_secs = load_interval_secs
        self._session = session
        self._tasks: list = []
        self._stopped = asyncio.Event()
        self.policy_registry = policy_registry

    async def start(self) -> None:
        self._stopped.clear()
        if self._session is None:
            self._session = aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=self.hc.timeout_secs)
            )
        if not self.hc.disable:
            self._tasks.append(asyncio.ensure_future(self._health_loop()))
        self._tasks.append(asyncio.ensure_future(self._load_loop()))

    async def stop(self) -> None:
        self._stopped.set()
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks.clear()
        if self._session is not None:
            await self._session.close()
            self._session = None

    # ---- health ----------------------------------------------------------
    async def check_worker_health(self, worker: Worker) -> bool:
        if worker.url.startswith(("sim://", "rccl://")):
            return True  # in-process / data-plane workers are healthy by construction
        try:
            async with self._session.get(worker.url + self.hc.endpoint) as resp:
                return 200 <= resp.status < 300
        except Exception:
            return False

    async def _health_loop(self) -> None:
        while not self._stopped.is_set():
            workers = self.registry.all()
            results = await asyncio.gather(*(self.check_worker_health(w) for w in workers))
            for w, ok in zip(workers, results):
                changed = w.observe_health(ok, self.hc.failure_threshold, self.hc.success_threshold)
                if changed:
                    log.info("worker %s health -> %s", w.url, w.health.value)
                    if (
                        w.health == HealthState.UNHEALTHY
                        and self.hc.remove_unhealthy_workers
                    ):
                        self.registry.remove(w.worker_id)
            try:
                await asyncio.wait_for(self._stopped.wait(), self.hc.check_interval_secs)
            except asyncio.TimeoutError:
                pass

    # ---- loads -----------------------------------------------------------
    async def poll_worker_loads(self, worker: Worker) -> None:
        if worker.url.startswith(("sim://", "rccl://")):
            return
        try:
            async with self._session.get(worker.url + "/get_loads") as resp:
                if resp.status != 200:
                    return
                data = await resp.json()
        except Exception:
            return
        self.apply_load_snapshot(worker, data)

    @staticmethod
    def apply_load_snapshot(worker: Worker, data: dict) -> None:
        """Engine load snapshot shape (reference SchedulerLoadSnapshot,
        crates/protocols/src/worker.rs:1214): num_queued_tokens/num_inflight_tokens
        or queued/inflight request counts, token_usage, gen_throughput, dp_loads."""
        if not isinstance(data, dict):
            return
        loads = data.get("loads") or data
        worker.queued_tokens = int(loads.get("num_queue_tokens", loads.get("queued_tokens", worker.queued_tokens)))
        worker.inflight_tokens = int(
            loads.get("num_inflight_tokens", loads.get("inflight_tokens", worker.inflight_tokens))
        )
        if "token_usage" in loads and loads["token_usage"] is not None:
            worker.token_usage = float(loads["token_usage"])
        if "gen_throughput" in loads and loads["gen_throughput"] is not None:
            worker.gen_throughput = float(loads["gen_throughput"])
        dp = loads.get("dp_loads")
        if isinstance(dp, list) and dp:
            worker.dp_size = len(dp)
            worker.dp_loads = [int(x) for x in dp]

    async def _load_loop(self) -> None:
        while not self._stopped.is_set():
            workers = self.registry.all()
            await asyncio.gather(*(self.poll_worker_loads(w) for w in workers))
            if self.policy_registry is not None:
                for p in self.policy_registry.all_policies():
                    p.update_loads(workers)
            try:
                await asyncio.wait_for(self._stopped.wait(), self.load_interval_secs)
            except asyncio.TimeoutError:
                pass
