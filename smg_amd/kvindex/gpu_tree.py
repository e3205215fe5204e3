"""GPU-resident tree wrapper (gfx950 kernels in csrc/gpu_tree.hip).

Single-request calls run as batch-of-1 kernel launches (~15 us); the
cache-aware policy's batched path (`match_and_insert_batch`) services a whole
arrival window in ONE launch: match walk + MRU-tenant reduction + min-load
decision + path insert, with intra-batch load increments done device-side.
Raises loudly when the extension or a GPU is missing — no silent CPU
fallback on a GPU box.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import numpy as np

from .pytree import MatchResult
from .slots import SlotMap


class GpuTokenTree:
    def __init__(
        self,
        page_size: int = 16,
        device: int = 0,
        capacity: int = 1 << 22,
        table_size: Optional[int] = None,
        max_pages: int = 1024,
        max_batch_reqs: int = 4096,
        max_batch_tokens: int = 1 << 22,
    ):
        from .. import _core  # torch-first guard lives in kvindex.__init__

        if _core.hip_device_count() <= 0:
            raise RuntimeError("GpuTokenTree requires an AMD GPU (hip_device_count==0)")
        if table_size is None:
            table_size = 1
            while table_size < capacity * 2:
                table_size <<= 1
        self._tree = _core.GpuTree(
            device=device,
            node_cap=capacity,
            table_size=table_size,
            page_size=page_size,
            max_pages=max_pages,
            max_batch_reqs=max_batch_reqs,
            max_batch_tokens=max_batch_tokens,
        )
        self.page_size = page_size
        self.slots = SlotMap()
        self.capacity = capacity
        self._approx_nodes = 0
        self._last_clock_check = 0.0

    # ---- low-level batch -------------------------------------------------
    def _run(
        self,
        token_lists: Sequence[Sequence[int]],
        healthy_mask: int,
        loads: List[int],
        processed: List[int],
        n_workers: int,
        cache_threshold: float = 0.3,
        imbalanced: bool = False,
        do_insert: bool = True,
        forced_tenant: int = -1,
        mode: int = 0,
    ):
        offsets = np.zeros(len(token_lists) + 1, dtype=np.uint32)
        for i, t in enumerate(token_lists):
            offsets[i + 1] = offsets[i] + len(t)
        flat = np.empty(int(offsets[-1]), dtype=np.uint32)
        for i, t in enumerate(token_lists):
            flat[offsets[i]: offsets[i + 1]] = np.asarray(t, dtype=np.uint32)
        sel, matched, tenant, depths = self._tree.run(
            flat,
            offsets,
            healthy_mask,
            loads,
            processed,
            n_workers,
            cache_threshold=cache_threshold,
            imbalanced=imbalanced,
            do_insert=do_insert,
            forced_tenant=forced_tenant,
            mode=mode,
        )
        self._approx_nodes += int(offsets[-1]) // self.page_size  # upper bound
        if mode == 1:
            return sel, matched, tenant, depths.reshape(len(token_lists), 64)
        return sel, matched, tenant

    # ---- tree API (parity with pytree) -----------------------------------
    def match(self, tokens: Sequence) -> MatchResult:
        sel, matched, tenant = self._run(
            [tokens], healthy_mask=(1 << 64) - 1, loads=[0] * 64, processed=[0] * 64,
            n_workers=64, do_insert=False,
        )
        slot = int(tenant[0])
        url = self.slots.url_of(slot) if slot != 0xFFFFFFFF else None
        return MatchResult(url, int(matched[0]), len(tokens))

    def insert(self, tokens: Sequence, tenant: str) -> int:
        slot = self.slots.slot_of(tenant)
        self._run(
            [tokens], healthy_mask=(1 << 64) - 1, loads=[0] * 64, processed=[0] * 64,
            n_workers=64, do_insert=True, forced_tenant=slot,
        )
        return len(tokens) - (len(tokens) % self.page_size)

    def match_and_insert(self, tokens: Sequence, choose_tenant):
        result = self.match(tokens)
        tenant = choose_tenant(result)
        if tenant is not None:
            self.insert(tokens, tenant)
        return result, tenant

    # ---- the one-launch batched decision (cache_aware fast path) ----------
    def match_and_insert_batch(
        self,
        token_lists: Sequence[Sequence[int]],
        urls: Sequence[str],
        candidates: Sequence[int],
        loads: Sequence[int],
        processed: Sequence[int],
        cache_threshold: float,
        min_load_idx: Optional[int],
        imbalanced: bool,
    ) -> List[Optional[int]]:
        if not token_lists:
            return []
        slot_of_idx = {i: self.slots.slot_of(urls[i]) for i in range(len(urls))}
        idx_of_slot = {s: i for i, s in slot_of_idx.items()}
        healthy_mask = 0
        loads64 = [1 << 20] * 64
        processed64 = [0] * 64
        for i in candidates:
            s = slot_of_idx[i]
            healthy_mask |= 1 << s
            loads64[s] = int(loads[i])
            processed64[s] = int(processed[i])
        sel, _matched, _tenant = self._run(
            token_lists,
            healthy_mask=healthy_mask,
            loads=loads64,
            processed=processed64,
            n_workers=64,
            cache_threshold=cache_threshold,
            imbalanced=imbalanced,
            do_insert=True,
        )
        return [idx_of_slot.get(int(s)) if int(s) >= 0 else None for s in sel]

    # ---- KV-event overlap scoring (mode 1/2 kernel paths) -----------------
    def match_worker_depths(self, token_lists):
        """[n, 64] matched token counts per worker slot (one kernel launch;
        reference event_tree.rs:571 find_matches semantics)."""
        _s, _m, _t, depths = self._run(
            token_lists, healthy_mask=(1 << 64) - 1, loads=[0] * 64, processed=[0] * 64,
            n_workers=64, do_insert=False, mode=1,
        )
        return depths

    def remove_path(self, tokens, tenant: str) -> None:
        """Clear `tenant`'s attribution along this token path (apply_removed)."""
        slot = self.slots.existing_slot(tenant)
        if slot is None:
            return
        self._run([tokens], healthy_mask=(1 << 64) - 1, loads=[0] * 64, processed=[0] * 64,
                  n_workers=64, do_insert=False, forced_tenant=slot, mode=2)

    # ---- maintenance -----------------------------------------------------
    def remove_tenant(self, tenant: str) -> None:
        slot = self.slots.existing_slot(tenant)
        if slot is not None:
            self._tree.remove_tenant(slot)
            self.slots.release(tenant)

    def evict(self, max_nodes: int) -> int:
        """LRU eviction to `max_nodes` that actually FREES pool memory
        (reference token_tree.rs eviction to --max-tree-size): progressively
        older->newer tenant-bit sweeps, each followed by a reclaim pass that
        tombstones table slots and refills the node free list, so a long-lived
        gateway keeps learning instead of silently saturating."""
        stats = self._tree.stats()
        live = int(stats["live_nodes"])
        # also reclaim when the bump allocator nears the pool cap even if the
        # live count is modest (nodes stranded by tenant removal / KV events)
        allocated = int(stats["allocated_nodes"])
        free = int(stats.get("free_nodes", 0))
        near_cap = allocated - free >= self.capacity - max(1024, self.capacity // 16)
        if live <= max_nodes and not near_cap:
            return 0
        clock = int(stats["clock"])
        reclaimed = int(self._tree.reclaim())  # whatever earlier sweeps freed
        for num in (2, 3, 7):  # clear oldest 1/4, then 1/2, then 7/8 of the clock range
            self._tree.evict_older(max(1, clock * num // 8))
            reclaimed += int(self._tree.reclaim())
            live = int(self._tree.stats()["live_nodes"])
            if live <= max_nodes:
                break
        else:
            if live > max_nodes:
                # pathological (all stamps current): last-resort full reset
                self._tree.clear()
                self._approx_nodes = 0
        return reclaimed

    def clear(self) -> None:
        self._tree.clear()
        self._approx_nodes = 0

    def stats(self) -> Dict:
        return self._tree.stats()

    def __len__(self) -> int:
        return int(self._tree.stats()["live_nodes"])

    @property
    def tenant_token_count(self) -> Dict[str, int]:
        stats = self._tree.stats()
        out = {}
        for url, slot in self.slots.url_to_slot.items():
            out[url] = int(stats["tenant_nodes"][slot]) * self.page_size
        return out


class GpuTextTree(GpuTokenTree):
    """Byte-paged text variant on the same device kernels."""

    def __init__(self, page_size: int = 8, device: int = 0, capacity: int = 1 << 21, **kw):
        super().__init__(page_size=page_size, device=device, capacity=capacity, **kw)

    @staticmethod
    def _bytes(text: str):
        return np.frombuffer(text.encode("utf-8", "ignore"), dtype=np.uint8).astype(np.uint32)

    def match_text(self, text: str) -> MatchResult:
        return self.match(self._bytes(text))

    def insert_text(self, text: str, tenant: str) -> int:
        return self.insert(self._bytes(text), tenant)

    def match_and_insert_text(self, text: str, choose_tenant):
        return self.match_and_insert(self._bytes(text), choose_tenant)
