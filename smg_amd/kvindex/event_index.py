"""Positional KV-event indexer (reference: crates/kv_index/src/event_tree.rs —
PositionalIndexer :365, apply_stored :415, apply_removed :492, find_matches
:571; content hashes compute_content_hash :144 / compute_request_content_hashes
:163 over LE token bytes, chunked by block_size with the partial tail dropped).

Engines report which KV blocks they hold (KvBlocksStored/Removed events); the
indexer maps (position, content_hash) -> worker bitset so cache-aware routing
can score overlap per worker without a gateway-side tree.  find_matches walks
a request's block hashes position by position and scores each worker by its
matched prefix length (jump_size pages at a time in the reference; exact walk
here, the GPU path batches it).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import xxhash


def compute_content_hashes(tokens: Sequence[int], block_size: int) -> List[int]:
    """Per-block chained content hashes: xxh3 over (prev_hash LE8 || token LE4
    bytes of the block).  Chaining matches engine-side prefix semantics: a
    block hash commits to its whole prefix."""
    n = (len(tokens) // block_size) * block_size
    out: List[int] = []
    prev = 0
    for i in range(0, n, block_size):
        h = xxhash.xxh3_64()
        h.update(prev.to_bytes(8, "little"))
        for t in tokens[i: i + block_size]:
            h.update(int(t & 0xFFFFFFFF).to_bytes(4, "little"))
        prev = h.intdigest()
        out.append(prev)
    return out


class PositionalIndexer:
    """Per-model index: position -> {content_hash -> worker bitset}."""

    MAX_POSITIONS = 8192

    def __init__(self, block_size: int = 16):
        self.block_size = block_size
        # model -> list[pos] of dict hash -> bitset
        self._index: Dict[str, List[Dict[int, int]]] = {}
        self._worker_slots: Dict[str, Dict[str, int]] = {}  # model -> url -> bit
        self._block_counts: Dict[str, Dict[str, int]] = {}  # model -> url -> blocks held

    # ---- worker slots ------------------------------------------------------
    def _slot(self, model: str, url: str) -> int:
        slots = self._worker_slots.setdefault(model, {})
        s = slots.get(url)
        if s is None:
            s = len(slots)
            slots[url] = s
        return s

    def has_events(self, model: str) -> bool:
        return bool(self._index.get(model))

    # ---- event application (reference apply_stored :415 / :492) -----------
    def apply_stored(self, model: str, url: str, hashes: Sequence[int], start_pos: int = 0) -> None:
        bit = 1 << self._slot(model, url)
        idx = self._index.setdefault(model, [])
        for i, h in enumerate(hashes):
            pos = start_pos + i
            if pos >= self.MAX_POSITIONS:
                break
            while len(idx) <= pos:
                idx.append({})
            idx[pos][h] = idx[pos].get(h, 0) | bit
        bc = self._block_counts.setdefault(model, {})
        bc[url] = bc.get(url, 0) + len(hashes)

    def apply_removed(self, model: str, url: str, hashes: Sequence[int]) -> None:
        slots = self._worker_slots.get(model, {})
        if url not in slots:
            return
        mask = ~(1 << slots[url])
        removed = 0
        idx = self._index.get(model, [])
        hs = set(hashes)
        for level in idx:
            for h in list(level.keys()):
                if h in hs:
                    level[h] &= mask
                    removed += 1
                    if level[h] == 0:
                        del level[h]
        bc = self._block_counts.setdefault(model, {})
        bc[url] = max(0, bc.get(url, 0) - removed)

    def remove_worker(self, model: str, url: str) -> None:
        slots = self._worker_slots.get(model, {})
        if url not in slots:
            return
        mask = ~(1 << slots[url])
        for level in self._index.get(model, []):
            for h in list(level.keys()):
                level[h] &= mask
                if level[h] == 0:
                    del level[h]
        self._block_counts.get(model, {}).pop(url, None)

    # ---- scoring (reference find_matches :571) -----------------------------
    def find_matches(self, model: str, tokens: Sequence[int]) -> Dict[str, int]:
        """Per-worker matched-block counts for this request's block hashes:
        a worker scores position p only if it matched every position < p
        (prefix semantics via the chained hashes)."""
        idx = self._index.get(model)
        if not idx:
            return {}
        hashes = compute_content_hashes(tokens, self.block_size)
        slots = self._worker_slots.get(model, {})
        alive = (1 << len(slots)) - 1
        scores_bits: List[int] = []
        live = alive
        for pos, h in enumerate(hashes):
            if pos >= len(idx) or live == 0:
                break
            live &= idx[pos].get(h, 0)
            scores_bits.append(live)
        out: Dict[str, int] = {}
        for url, s in slots.items():
            bit = 1 << s
            n = 0
            for b in scores_bits:
                if b & bit:
                    n += 1
                else:
                    break
            if n:
                out[url] = n * self.block_size
        return out

    def tree_size(self, model: str, url: str) -> int:
        return self._block_counts.get(model, {}).get(url, 0)

    def clear(self, model: Optional[str] = None) -> None:
        if model is None:
            self._index.clear()
            self._block_counts.clear()
        else:
            self._index.pop(model, None)
            self._block_counts.pop(model, None)
