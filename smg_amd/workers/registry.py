"""Worker registry (reference: model_gateway/src/worker/registry.rs:92).

Central store of workers with secondary indexes: by model, by worker type, plus
per-model consistent-hash rings and an event stream for policy/monitor rewiring.
The reference uses DashMap + lock-free Arc<[T]> snapshots; here index rebuilds
produce immutable tuples so policies iterate a stable snapshot without locks,
and rebuilds are LAZY (dirty flag) so bulk registration of thousands of
workers is O(n), not O(n^2) over ring inserts.
"""
from __future__ import annotations

import asyncio
from typing import Callable, Dict, List, Optional, Tuple

from .hash_ring import HashRing
from .worker import Worker, WorkerType


class WorkerRegistry:
    def __init__(self) -> None:
        self._workers: Dict[int, Worker] = {}
        self._by_url: Dict[str, int] = {}
        self._by_model: Dict[str, Tuple[Worker, ...]] = {}
        self._by_type: Dict[WorkerType, Tuple[Worker, ...]] = {}
        self._alias_index: Dict[str, str] = {}  # alias -> model_id
        self._rings: Dict[str, HashRing] = {}
        self._all: Tuple[Worker, ...] = ()
        self._dirty = False
        self._listeners: List[Callable[[str, Worker], None]] = []
        self._change_event: Optional[asyncio.Event] = None
        # per-model retry overrides (reference registry.rs model_retry_configs)
        self._model_retry: Dict[str, object] = {}

    # ---- per-model retry configs ------------------------------------------
    def set_model_retry_config(self, model_id: str, cfg) -> None:
        self._model_retry[model_id] = cfg

    def get_model_retry_config(self, model_id):
        return self._model_retry.get(model_id)

    # ---- mutation --------------------------------------------------------
    def register(self, worker: Worker) -> Worker:
        if worker.url in self._by_url and self._workers.get(self._by_url[worker.url]) is not None:
            existing = self._workers[self._by_url[worker.url]]
            if existing.worker_type == worker.worker_type and existing.model_id == worker.model_id:
                return existing  # idempotent add
        self._workers[worker.worker_id] = worker
        self._by_url[worker.url] = worker.worker_id
        for alias in worker.model_aliases:
            self._alias_index[alias] = worker.model_id
        self._dirty = True
        self._emit("add", worker)
        return worker

    def remove(self, worker_id: int) -> Optional[Worker]:
        worker = self._workers.pop(worker_id, None)
        if worker is None:
            return None
        if self._by_url.get(worker.url) == worker_id:
            del self._by_url[worker.url]
        self._dirty = True
        self._emit("remove", worker)
        return worker

    def remove_by_url(self, url: str) -> Optional[Worker]:
        wid = self._by_url.get(url.rstrip("/"))
        return self.remove(wid) if wid is not None else None

    # ---- lookup ----------------------------------------------------------
    def get(self, worker_id: int) -> Optional[Worker]:
        return self._workers.get(worker_id)

    def get_by_url(self, url: str) -> Optional[Worker]:
        wid = self._by_url.get(url.rstrip("/"))
        return self._workers.get(wid) if wid is not None else None

    def _ensure(self) -> None:
        if self._dirty:
            self._rebuild()
            self._dirty = False

    def all(self) -> Tuple[Worker, ...]:
        self._ensure()
        return self._all

    def resolve_model(self, model_id: Optional[str]) -> Optional[str]:
        if model_id is None:
            return None
        return self._alias_index.get(model_id, model_id)

    def for_model(
        self,
        model_id: Optional[str],
        worker_type: Optional[WorkerType] = None,
        available_only: bool = True,
    ) -> List[Worker]:
        self._ensure()
        model_id = self.resolve_model(model_id)
        if model_id is not None and model_id in self._by_model:
            candidates = self._by_model[model_id]
        elif model_id is None:
            candidates = self._all
        else:
            candidates = ()
        out = []
        for w in candidates:
            if worker_type is not None and w.worker_type != worker_type:
                continue
            if available_only and not w.is_available():
                continue
            out.append(w)
        return out

    def by_type(self, worker_type: WorkerType, available_only: bool = True) -> List[Worker]:
        self._ensure()
        ws = self._by_type.get(worker_type, ())
        return [w for w in ws if not available_only or w.is_available()]

    def ring_for_model(self, model_id: Optional[str]) -> Optional[HashRing]:
        self._ensure()
        return self._rings.get(self.resolve_model(model_id) or "*")

    def models(self) -> List[str]:
        self._ensure()
        return sorted(self._by_model.keys())

    def healthy_count(self) -> int:
        self._ensure()
        return sum(1 for w in self._all if w.is_available())

    def __len__(self) -> int:
        return len(self._workers)

    # ---- events ----------------------------------------------------------
    def subscribe(self, fn: Callable[[str, Worker], None]) -> None:
        self._listeners.append(fn)

    def change_event(self) -> asyncio.Event:
        if self._change_event is None:
            self._change_event = asyncio.Event()
        return self._change_event

    def _emit(self, kind: str, worker: Worker) -> None:
        for fn in list(self._listeners):
            fn(kind, worker)
        if self._change_event is not None:
            self._change_event.set()

    # ---- internals -------------------------------------------------------
    def _rebuild(self) -> None:
        workers = tuple(sorted(self._workers.values(), key=lambda w: w.worker_id))
        self._all = workers
        by_model: Dict[str, List[Worker]] = {}
        by_type: Dict[WorkerType, List[Worker]] = {}
        for w in workers:
            by_model.setdefault(w.model_id, []).append(w)
            by_type.setdefault(w.worker_type, []).append(w)
        self._by_model = {m: tuple(v) for m, v in by_model.items()}
        self._by_type = {t: tuple(v) for t, v in by_type.items()}
        self._rings = {m: HashRing([w.url for w in v]) for m, v in self._by_model.items()}
        self._rings["*"] = HashRing([w.url for w in workers])
