"""Priority admission scheduler (reference: model_gateway/src/middleware/
scheduler/engine.rs (1,696 LoC) — 4 classes system/interactive/default/bulk,
SlotPool with per-class reservations recomputed from live worker capacity
(engine.rs:665-747), FIFO class queues, preemption budget (engine.rs:31),
AdmissionMode::from_config choosing scheduler vs legacy at startup
(server.rs:773)).

Slots = healthy_workers * per_worker_concurrency, recomputed on worker
events.  Each class reserves a fraction of slots; unused reservation spills
to lower classes.  When saturated, an arriving higher-class request preempts
the newest queued lower-class request (rejected 429) within the preemption
budget.
"""
from __future__ import annotations

import asyncio
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Dict, Optional

from aiohttp import web

from ..protocols.openai import error_body

CLASSES = ("system", "interactive", "default", "bulk")
CLASS_RANK = {c: i for i, c in enumerate(CLASSES)}  # lower = higher priority


@dataclass
class SchedulerConfig:
    per_worker_concurrency: int = 32
    min_slots: int = 8
    queue_size: int = 256
    queue_timeout_secs: float = 30.0
    preemption_budget_ms: float = 50.0
    # reserved slot fractions per class (spill-down allowed)
    reservations: Dict[str, float] = field(
        default_factory=lambda: {"system": 0.1, "interactive": 0.3, "default": 0.5, "bulk": 0.1}
    )
    default_class: str = "default"
    tenant_classes: Dict[str, str] = field(default_factory=dict)  # tenant -> class

    @classmethod
    def from_yaml(cls, path: Optional[str]) -> "SchedulerConfig":
        cfg = cls()
        if not path:
            return cfg
        import yaml

        with open(path) as f:
            data = yaml.safe_load(f) or {}
        for k in ("per_worker_concurrency", "min_slots", "queue_size", "queue_timeout_secs",
                  "preemption_budget_ms", "default_class"):
            if k in data:
                setattr(cfg, k, data[k])
        if "reservations" in data:
            cfg.reservations.update(data["reservations"])
        if "tenants" in data:
            for tenant, klass in data["tenants"].items():
                cfg.tenant_classes[tenant] = klass
        return cfg


class _Waiter:
    __slots__ = ("klass", "future", "enqueued_at")

    def __init__(self, klass: str):
        self.klass = klass
        self.future: asyncio.Future = asyncio.get_event_loop().create_future()
        self.enqueued_at = time.monotonic()


class PriorityScheduler:
    def __init__(self, config: SchedulerConfig, worker_registry=None, metrics=None):
        self.config = config
        self.registry = worker_registry
        self.metrics = metrics
        self._in_use: Dict[str, int] = {c: 0 for c in CLASSES}
        self._queues: Dict[str, deque] = {c: deque() for c in CLASSES}
        self._capacity = config.min_slots
        self.preempted = 0
        self.rejected = 0
        if worker_registry is not None:
            worker_registry.subscribe(lambda kind, w: self.recompute_capacity())
            self.recompute_capacity()

    # ---- capacity (engine.rs:665-747) --------------------------------------
    def recompute_capacity(self) -> None:
        if self.registry is None:
            return
        healthy = max(1, self.registry.healthy_count() or len(self.registry))
        self._capacity = max(self.config.min_slots, healthy * self.config.per_worker_concurrency)
        self._dispatch()

    @property
    def capacity(self) -> int:
        return self._capacity

    def class_limit(self, klass: str) -> int:
        """Slots this class may hold: its reservation plus everything reserved
        for LOWER classes (spill-down, never up)."""
        frac = 0.0
        for c in CLASSES:
            if CLASS_RANK[c] >= CLASS_RANK[klass]:
                frac += self.config.reservations.get(c, 0.0)
        return max(1, int(self._capacity * frac))

    def total_in_use(self) -> int:
        return sum(self._in_use.values())

    # ---- class resolution ---------------------------------------------------
    def classify(self, tenant_id: Optional[str], header_class: Optional[str]) -> str:
        if header_class in CLASSES:
            klass = header_class
        else:
            klass = self.config.tenant_classes.get(tenant_id or "", self.config.default_class)
        # a request may not claim a class above the configured maximum
        if CLASS_RANK[klass] < CLASS_RANK.get(self.config.default_class, 2) and (
            tenant_id not in self.config.tenant_classes and header_class not in CLASSES
        ):
            klass = self.config.default_class
        return klass

    # ---- acquire / release --------------------------------------------------
    def _can_admit(self, klass: str) -> bool:
        if self.total_in_use() >= self._capacity:
            return False
        held_at_or_above = sum(self._in_use[c] for c in CLASSES if CLASS_RANK[c] <= CLASS_RANK[klass])
        return held_at_or_above < self.class_limit(klass)

    async def acquire(self, klass: str) -> bool:
        if self._can_admit(klass):
            self._in_use[klass] += 1
            return True
        # saturated: try preempting a queued LOWER-class waiter to make queue room,
        # then wait our turn
        q = self._queues[klass]
        if sum(len(x) for x in self._queues.values()) >= self.config.queue_size:
            if not self._preempt_queued(klass):
                self.rejected += 1
                return False
        waiter = _Waiter(klass)
        q.append(waiter)
        try:
            await asyncio.wait_for(waiter.future, timeout=self.config.queue_timeout_secs)
            return True
        except asyncio.TimeoutError:
            try:
                q.remove(waiter)
            except ValueError:
                pass
            self.rejected += 1
            return False

    def _preempt_queued(self, klass: str) -> bool:
        """Reject the newest queued waiter of the lowest class below `klass`."""
        t0 = time.perf_counter()
        for c in reversed(CLASSES):
            if CLASS_RANK[c] <= CLASS_RANK[klass]:
                break
            q = self._queues[c]
            while q:
                if (time.perf_counter() - t0) * 1e3 > self.config.preemption_budget_ms:
                    return False
                victim = q.pop()
                if not victim.future.done():
                    victim.future.cancel()
                    self.preempted += 1
                    return True
        return False

    def release(self, klass: str) -> None:
        self._in_use[klass] = max(0, self._in_use[klass] - 1)
        self._dispatch()

    def _dispatch(self) -> None:
        """Grant freed slots to the highest-priority waiters first."""
        for c in CLASSES:
            q = self._queues[c]
            while q and self._can_admit(c):
                waiter = q.popleft()
                if waiter.future.done():
                    continue
                self._in_use[c] += 1
                waiter.future.set_result(True)

    # ---- aiohttp admission middleware hook ----------------------------------
    async def admit(self, request: web.Request, handler):
        klass = self.classify(request.get("tenant_id"), request.headers.get("x-smg-priority"))
        ok = await self.acquire(klass)
        if not ok:
            if self.metrics is not None and not self.metrics._null:
                self.metrics.scheduler_timeout.labels(klass).inc()
            return web.Response(
                status=429,
                body=error_body(f"admission rejected (class={klass})", 429, "rate_limit_error"),
                content_type="application/json",
                headers={"x-smg-class": klass},
            )
        if self.metrics is not None and not self.metrics._null:
            self.metrics.scheduler_admitted.labels(klass).inc()
        try:
            return await handler(request)
        finally:
            self.release(klass)

    def stats(self) -> Dict:
        return {
            "capacity": self._capacity,
            "in_use": dict(self._in_use),
            "queued": {c: len(q) for c, q in self._queues.items()},
            "preempted": self.preempted,
            "rejected": self.rejected,
        }
