"""KV-event monitor (reference: model_gateway/src/worker/kv_event_monitor.rs:48
— per-worker task subscribing the engine's SubscribeKvEvents stream, feeding
the per-model PositionalIndexer, learning the block size from the stream,
reconnecting with 100ms -> 30s backoff :346)."""
from __future__ import annotations

import asyncio
import logging
from typing import Dict, Optional

from ..kvindex.event_index import PositionalIndexer, compute_content_hashes

log = logging.getLogger("smg.worker.kvevents")


class KvEventMonitor:
    def __init__(self, registry, indexer: Optional[PositionalIndexer] = None, client_pool=None):
        self.registry = registry
        self.indexer = indexer or PositionalIndexer()
        self.client_pool = client_pool
        self._tasks: Dict[int, asyncio.Task] = {}
        self._stopped = asyncio.Event()

    async def start(self) -> None:
        self._stopped.clear()
        if self.client_pool is None:
            from ..grpc.client import ClientPool

            self.client_pool = ClientPool()
        self.registry.subscribe(self._on_worker_event)
        for w in self.registry.all():
            self._spawn(w)

    def _on_worker_event(self, kind: str, worker) -> None:
        if kind == "add":
            self._spawn(worker)
        elif kind == "remove":
            t = self._tasks.pop(worker.worker_id, None)
            if t:
                t.cancel()
            self.indexer.remove_worker(worker.model_id, worker.url)

    def _spawn(self, worker) -> None:
        if not worker.url.startswith("grpc") or worker.worker_id in self._tasks:
            return
        self._tasks[worker.worker_id] = asyncio.ensure_future(self._subscription_loop(worker))

    async def stop(self) -> None:
        self._stopped.set()
        for t in self._tasks.values():
            t.cancel()
        for t in self._tasks.values():
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks.clear()

    async def _subscription_loop(self, worker) -> None:
        backoff = 0.1
        while not self._stopped.is_set():
            try:
                client = self.client_pool.get(worker.url)
                async for batch in client.subscribe_kv_events():
                    backoff = 0.1
                    self._apply_batch(worker, batch)
            except asyncio.CancelledError:
                return
            except Exception as exc:
                log.debug("kv-event stream to %s dropped: %s", worker.url, exc)
            try:
                await asyncio.wait_for(self._stopped.wait(), backoff)
                return
            except asyncio.TimeoutError:
                backoff = min(backoff * 2, 30.0)

    def _apply_batch(self, worker, batch: dict) -> None:
        block_size = int(batch.get("block_size") or self.indexer.block_size)
        if block_size != self.indexer.block_size:
            # learn the engine's block size (reference kv_event_monitor.rs:314)
            self.indexer.block_size = block_size
        for ev in batch.get("events", []):
            etype = ev.get("type")
            if etype == "stored":
                tokens = ev.get("tokens")
                hashes = ev.get("block_hashes")
                if hashes is None and tokens is not None:
                    hashes = compute_content_hashes(tokens, block_size)
                if hashes:
                    self.indexer.apply_stored(worker.model_id, worker.url, hashes)
            elif etype == "removed":
                hashes = ev.get("block_hashes") or []
                if hashes:
                    self.indexer.apply_removed(worker.model_id, worker.url, hashes)
            elif etype == "all_cleared":
                self.indexer.remove_worker(worker.model_id, worker.url)
