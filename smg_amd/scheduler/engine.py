"""Priority admission scheduler (reference: model_gateway/src/middleware/
scheduler/engine.rs (1,696 LoC) — 4 classes system/interactive/default/bulk,
SlotPool with per-class reservations recomputed from live worker capacity
(engine.rs:665-747: desired = max(floor, ceil(share*capacity)), priority-
ordered clamp on shrink, grow-publishes-reservations-first), per-class FIFO
queues with fixed depths, INFLIGHT preemption with a 50 ms wait budget and
2 ms poll (engine.rs:30-36, find_preemption_victim :330), dispatcher wakeups
on release, and a metrics sampler task; AdmissionMode::from_config chooses
scheduler vs legacy at startup (server.rs:773)).

Python mapping notes: the asyncio event loop serializes slot-pool mutations,
so the reference's atomic publish-ordering rules collapse to plain code; the
dispatcher runs inline on release (a dedicated task buys nothing under the
GIL) while the SAMPLER stays a task.  Inflight preemption cancels the
victim's handler task; its admit() wrapper converts the cancellation into a
429 "preempted" response.
"""
from __future__ import annotations

import asyncio
import itertools
import math
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Dict, Optional

from aiohttp import web

from ..protocols.openai import error_body

CLASSES = ("system", "interactive", "default", "bulk")
CLASS_RANK = {c: i for i, c in enumerate(CLASSES)}  # lower = higher priority

PREEMPTION_WAIT_BUDGET_S = 0.050  # engine.rs:30
PREEMPTION_POLL_INTERVAL_S = 0.002  # engine.rs:33


@dataclass
class ClassConfig:
    floor: int = 0          # absolute reserved slots
    share: float = 0.0      # fraction of live capacity
    queue_size: int = 64    # per-class FIFO depth


@dataclass
class SchedulerConfig:
    per_worker_concurrency: int = 32
    min_slots: int = 8
    queue_timeout_secs: float = 30.0
    default_class: str = "default"
    tenant_classes: Dict[str, str] = field(default_factory=dict)  # tenant -> class
    classes: Dict[str, ClassConfig] = field(default_factory=lambda: {
        "system": ClassConfig(floor=1, share=0.10, queue_size=32),
        "interactive": ClassConfig(floor=1, share=0.30, queue_size=128),
        "default": ClassConfig(floor=1, share=0.50, queue_size=256),
        "bulk": ClassConfig(floor=0, share=0.10, queue_size=64),
    })
    enable_preemption: bool = True

    # back-compat alias consumed by older call sites/tests
    @property
    def queue_size(self) -> int:
        return sum(c.queue_size for c in self.classes.values())

    @classmethod
    def from_yaml(cls, path: Optional[str]) -> "SchedulerConfig":
        cfg = cls()
        if not path:
            return cfg
        import yaml

        with open(path) as f:
            data = yaml.safe_load(f) or {}
        for k in ("per_worker_concurrency", "min_slots", "queue_timeout_secs",
                  "default_class", "enable_preemption"):
            if k in data:
                setattr(cfg, k, data[k])
        # legacy shape: reservations: {class: fraction}
        for klass, frac in (data.get("reservations") or {}).items():
            if klass in cfg.classes:
                cfg.classes[klass].share = float(frac)
        for klass, sub in (data.get("classes") or {}).items():
            if klass in cfg.classes and isinstance(sub, dict):
                c = cfg.classes[klass]
                c.floor = int(sub.get("floor", c.floor))
                c.share = float(sub.get("share", c.share))
                c.queue_size = int(sub.get("queue_size", c.queue_size))
        if "queue_size" in data:  # legacy global depth -> default class
            cfg.classes["default"].queue_size = int(data["queue_size"])
        for tenant, klass in (data.get("tenants") or {}).items():
            cfg.tenant_classes[tenant] = klass
        return cfg


def desired_reservations(cfg: SchedulerConfig, capacity: int) -> Dict[str, int]:
    """engine.rs desired_reservations: max(floor, ceil(share*capacity)),
    capped per class at capacity."""
    out = {}
    for klass in CLASSES:
        c = cfg.classes.get(klass, ClassConfig())
        out[klass] = min(capacity, max(c.floor, math.ceil(c.share * capacity)))
    return out


def clamp_reservations(desired: Dict[str, int], capacity: int) -> Dict[str, int]:
    """engine.rs clamp_reservations_to_capacity: fill System -> Bulk; highest
    classes keep their seats under an extreme shrink, lowest yield first."""
    remaining = capacity
    out = {}
    for klass in CLASSES:
        give = min(desired[klass], remaining)
        out[klass] = give
        remaining -= give
    return out


class _Waiter:
    __slots__ = ("klass", "future", "enqueued_at")

    def __init__(self, klass: str):
        self.klass = klass
        self.future: asyncio.Future = asyncio.get_event_loop().create_future()
        self.enqueued_at = time.monotonic()


class _Inflight:
    __slots__ = ("id", "klass", "task", "started", "preempted")

    def __init__(self, id_, klass, task):
        self.id = id_
        self.klass = klass
        self.task = task
        self.started = time.monotonic()
        self.preempted = False


class PriorityScheduler:
    def __init__(self, config: SchedulerConfig, worker_registry=None, metrics=None):
        self.config = config
        self.registry = worker_registry
        self.metrics = metrics
        self._in_use: Dict[str, int] = {c: 0 for c in CLASSES}
        self._queues: Dict[str, deque] = {c: deque() for c in CLASSES}
        self._capacity = config.min_slots
        self._reserved: Dict[str, int] = clamp_reservations(
            desired_reservations(config, config.min_slots), config.min_slots)
        self._inflight: Dict[int, _Inflight] = {}
        self._ids = itertools.count(1)
        self.preempted_inflight = 0
        self.rejected = 0
        self._sampler_task: Optional[asyncio.Task] = None
        if worker_registry is not None:
            worker_registry.subscribe(lambda kind, w: self.recompute_capacity())
            self.recompute_capacity()

    # ---- capacity + live reservations (engine.rs:665-747) -------------------
    def recompute_capacity(self) -> None:
        if self.registry is None:
            return
        healthy = max(1, self.registry.healthy_count() or len(self.registry))
        self.apply_new_capacity(
            max(self.config.min_slots, healthy * self.config.per_worker_concurrency))

    def apply_new_capacity(self, new_capacity: int) -> None:
        if new_capacity == self._capacity:
            return
        desired = desired_reservations(self.config, new_capacity)
        if sum(desired.values()) > new_capacity:
            desired = clamp_reservations(desired, new_capacity)
        grow = new_capacity > self._capacity
        # reference publish-order rule (conservative transient) is automatic
        # here: the event loop serializes both writes
        self._reserved = desired
        self._capacity = new_capacity
        if grow:
            self._dispatch()  # release_notify on grow

    @property
    def capacity(self) -> int:
        return self._capacity

    def reserved(self, klass: str) -> int:
        return self._reserved.get(klass, 0)

    def class_limit(self, klass: str) -> int:
        """Slots available to `klass`: capacity minus what is reserved for
        classes ABOVE it (a class may spill into lower reservations but never
        upward into a higher class's floor)."""
        reserved_above = sum(
            self._reserved[c] for c in CLASSES if CLASS_RANK[c] < CLASS_RANK[klass])
        return max(1, self._capacity - reserved_above)

    def total_in_use(self) -> int:
        return sum(self._in_use.values())

    # ---- class resolution ---------------------------------------------------
    def classify(self, tenant_id: Optional[str], header_class: Optional[str]) -> str:
        if header_class in CLASSES:
            klass = header_class
        else:
            klass = self.config.tenant_classes.get(tenant_id or "", self.config.default_class)
        if CLASS_RANK[klass] < CLASS_RANK.get(self.config.default_class, 2) and (
            tenant_id not in self.config.tenant_classes and header_class not in CLASSES
        ):
            klass = self.config.default_class
        return klass

    # ---- acquire / release --------------------------------------------------
    def _can_admit(self, klass: str) -> bool:
        if self.total_in_use() >= self._capacity:
            return False
        held_at_or_above = sum(
            self._in_use[c] for c in CLASSES if CLASS_RANK[c] <= CLASS_RANK[klass])
        return held_at_or_above < self.class_limit(klass)

    def find_preemption_victim(self, waiter_class: str) -> Optional[_Inflight]:
        """engine.rs:330 — the NEWEST inflight request of the LOWEST class
        strictly below the waiter, not already being preempted."""
        for c in reversed(CLASSES):
            if CLASS_RANK[c] <= CLASS_RANK[waiter_class]:
                return None
            best = None
            for h in self._inflight.values():
                if h.klass == c and not h.preempted:
                    if best is None or h.started > best.started:
                        best = h
            if best is not None:
                return best
        return None

    async def _try_preempt_inflight(self, klass: str) -> bool:
        """Fire ONE cancel at the best victim, then wait up to the 50 ms
        budget (2 ms polls) for its slot to free (engine.rs:193-232)."""
        victim = self.find_preemption_victim(klass)
        if victim is None:
            return False
        victim.preempted = True
        victim.task.cancel()
        self.preempted_inflight += 1
        if self.metrics is not None and not self.metrics._null:
            self.metrics.scheduler_preempted.labels(victim.klass).inc()
        deadline = time.monotonic() + PREEMPTION_WAIT_BUDGET_S
        while time.monotonic() < deadline:
            if self._can_admit(klass):
                return True
            await asyncio.sleep(PREEMPTION_POLL_INTERVAL_S)
        return self._can_admit(klass)

    async def acquire(self, klass: str) -> bool:
        if self._can_admit(klass):
            self._in_use[klass] += 1
            return True
        if self.config.enable_preemption and klass != "bulk":
            if await self._try_preempt_inflight(klass) and self._can_admit(klass):
                self._in_use[klass] += 1
                return True
        q = self._queues[klass]
        if len(q) >= self.config.classes.get(klass, ClassConfig()).queue_size:
            # fixed per-class FIFO depth (engine.rs queue_for): overflow rejects
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
        except asyncio.CancelledError:
            try:
                q.remove(waiter)
            except ValueError:
                pass
            raise

    def release(self, klass: str) -> None:
        self._in_use[klass] = max(0, self._in_use[klass] - 1)
        self._dispatch()

    def _dispatch(self) -> None:
        """Grant freed slots to the highest-priority waiters first (the
        dispatcher; runs inline — the event loop is the serialization)."""
        for c in CLASSES:
            q = self._queues[c]
            while q and self._can_admit(c):
                waiter = q.popleft()
                if waiter.future.done():
                    continue
                self._in_use[c] += 1
                waiter.future.set_result(True)

    # ---- metrics sampler task (reference sampler task) ----------------------
    def start_sampler(self, interval_s: float = 1.0) -> None:
        if self._sampler_task is None or self._sampler_task.done():
            self._sampler_task = asyncio.ensure_future(self._sampler(interval_s))

    async def _sampler(self, interval_s: float) -> None:
        while True:
            if self.metrics is not None and not self.metrics._null:
                for c in CLASSES:
                    self.metrics.queue_depth.labels(c).set(len(self._queues[c]))
            await asyncio.sleep(interval_s)

    async def stop(self) -> None:
        if self._sampler_task is not None:
            self._sampler_task.cancel()
            try:
                await self._sampler_task
            except (asyncio.CancelledError, Exception):
                pass

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
        handle = _Inflight(next(self._ids), klass, asyncio.current_task())
        self._inflight[handle.id] = handle
        try:
            return await handler(request)
        except asyncio.CancelledError:
            if handle.preempted:
                # the victim's body was cancelled by a higher-class admission
                return web.Response(
                    status=429,
                    body=error_body(f"preempted by a higher-priority request (class={klass})",
                                    429, "rate_limit_error"),
                    content_type="application/json",
                    headers={"x-smg-class": klass, "x-smg-preempted": "1"},
                )
            raise
        finally:
            self._inflight.pop(handle.id, None)
            self.release(klass)

    def stats(self) -> Dict:
        return {
            "capacity": self._capacity,
            "reserved": dict(self._reserved),
            "in_use": dict(self._in_use),
            "queued": {c: len(q) for c, q in self._queues.items()},
            "inflight": len(self._inflight),
            "preempted_inflight": self.preempted_inflight,
            "rejected": self.rejected,
        }
