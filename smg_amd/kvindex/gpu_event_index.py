"""GPU positional KV-event indexer (reference: event_tree.rs PositionalIndexer;
the "GPU hash-probe + bitset-popcount scoring kernel" of SURVEY §2.4).

Reuses the device prefix-index kernels: each 64-bit chained content hash is
fed as a 2-u32 "page", so (position, hash) entries live in the same
open-addressed device table with per-worker bitsets, and find_matches is one
mode-1 kernel launch returning per-worker matched depths for a whole batch.
API-compatible with kvindex.event_index.PositionalIndexer.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import numpy as np

from .chain_keys import chain_keys
from .event_index import compute_content_hashes
from .gpu_tree import GpuTokenTree


def _hashes_to_tokens(hashes: Sequence[int]) -> List[int]:
    out: List[int] = []
    for h in hashes:
        out.append(h & 0xFFFFFFFF)
        out.append((h >> 32) & 0xFFFFFFFF)
    return out


class GpuPositionalIndexer:
    def __init__(self, block_size: int = 16, device: int = 0, capacity: int = 1 << 21):
        self.block_size = block_size
        self._trees: Dict[str, GpuTokenTree] = {}
        self._device = device
        self._capacity = capacity
        self._has_events: Dict[str, bool] = {}
        self._block_counts: Dict[str, Dict[str, int]] = {}
        # chained content hash -> table key (for removal by value)
        self._hash_keys: Dict[str, Dict[int, int]] = {}

    def _tree(self, model: str) -> GpuTokenTree:
        t = self._trees.get(model)
        if t is None:
            t = GpuTokenTree(page_size=2, device=self._device, capacity=self._capacity)
            self._trees[model] = t
        return t

    def has_events(self, model: str) -> bool:
        return self._has_events.get(model, False)

    def apply_stored(self, model: str, url: str, hashes: Sequence[int], start_pos: int = 0) -> None:
        if start_pos != 0:
            # offset appends need the preceding prefix; engines report full
            # prefixes here (the monitor always applies from 0)
            return
        hash_tokens = _hashes_to_tokens(hashes)
        self._tree(model).insert(hash_tokens, url)
        keys = chain_keys(hash_tokens, 2)
        hk = self._hash_keys.setdefault(model, {})
        for h, k in zip(hashes, keys):
            hk[h] = k
        self._has_events[model] = True
        bc = self._block_counts.setdefault(model, {})
        bc[url] = bc.get(url, 0) + len(hashes)

    def apply_removed(self, model: str, url: str, hashes: Sequence[int]) -> None:
        tree = self._trees.get(model)
        if tree is None:
            return
        slot = tree.slots.existing_slot(url)
        if slot is None:
            return
        hk = self._hash_keys.get(model, {})
        keys = [hk[h] for h in hashes if h in hk]
        if keys:
            tree._tree.clear_entries(np.asarray(keys, dtype=np.uint64), slot)
        bc = self._block_counts.setdefault(model, {})
        bc[url] = max(0, bc.get(url, 0) - len(keys))

    def remove_worker(self, model: str, url: str) -> None:
        tree = self._trees.get(model)
        if tree is not None:
            tree.remove_tenant(url)
        self._block_counts.get(model, {}).pop(url, None)

    def find_matches(self, model: str, tokens: Sequence[int]) -> Dict[str, int]:
        return self.find_matches_batch(model, [tokens])[0]

    def find_matches_batch(self, model: str, token_lists: Sequence[Sequence[int]]) -> List[Dict[str, int]]:
        """Per-request {worker_url: matched_token_count} — ONE kernel launch."""
        tree = self._trees.get(model)
        if tree is None:
            return [{} for _ in token_lists]
        hash_tokens = [
            _hashes_to_tokens(compute_content_hashes(t, self.block_size)) for t in token_lists
        ]
        if not any(hash_tokens):
            return [{} for _ in token_lists]
        depths = tree.match_worker_depths(hash_tokens)  # [n, 64] in hash-token units
        out: List[Dict[str, int]] = []
        slot_urls = list(tree.slots.url_to_slot.items())
        for row in depths:
            scores: Dict[str, int] = {}
            for url, slot in slot_urls:
                blocks = int(row[slot]) // 2  # kernel reports hash-token units; 2 per block
                if blocks:
                    scores[url] = blocks * self.block_size
            out.append(scores)
        return out

    def tree_size(self, model: str, url: str) -> int:
        return self._block_counts.get(model, {}).get(url, 0)

    def clear(self, model: Optional[str] = None) -> None:
        if model is None:
            for t in self._trees.values():
                t.clear()
            self._block_counts.clear()
            self._has_events.clear()
        elif model in self._trees:
            self._trees[model].clear()
            self._block_counts.pop(model, None)
            self._has_events.pop(model, None)
