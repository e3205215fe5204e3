"""Cache-aware load-balancing policy (reference: model_gateway/src/policies/
cache_aware.rs:769 select_worker, :283 is_imbalanced, :986 select_worker_with_tokens,
:1076 select_worker_with_text).

Decision tree per request:
  1. imbalance triggers (KV overload ceiling / KV spread / count spread)
     -> shortest queue, still inserting the path for the chosen worker;
  2. token ids + a KV-event indexer -> event-driven overlap scoring;
  3. token ids -> paged radix tree match_and_insert: match_rate > cache_threshold
     routes to the matched tenant (if healthy), else min-load;
  4. text -> same over the byte tree.

Tree backend: GPU-resident gfx950 tree (smg_amd._core) when
PolicyConfig.gpu_tree and a device is present; the pure-host tree otherwise.
Batched arrivals flow through select_worker_batch which the GPU backend
services with one kernel launch.
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional, Sequence, Tuple

from ..config import PolicyConfig
from ..kvindex import make_text_tree, make_token_tree
from ..workers.worker import Worker
from .base import LoadBalancingPolicy, SelectWorkerInfo


def _min_load_idx(workers: Sequence[Worker], candidates: Sequence[int]) -> Optional[int]:
    # tie-break (load, processed_requests, idx): spreads load when decode
    # outpaces prefill (reference cache_aware.rs select_worker gather pass)
    best = None
    best_key = None
    for i in candidates:
        w = workers[i]
        key = (w.active_requests, w.processed_requests, i)
        if best_key is None or key < best_key:
            best, best_key = i, key
    return best


class CacheAwarePolicy(LoadBalancingPolicy):
    name = "cache_aware"

    def __init__(self, cfg: Optional[PolicyConfig] = None, indexer=None):
        self.cfg = cfg or PolicyConfig(name="cache_aware")
        self.token_trees: Dict[str, object] = {}
        self.text_trees: Dict[str, object] = {}
        self.indexer = indexer  # KV-event PositionalIndexer (event-driven mode)
        self._last_eviction = time.monotonic()
        # HA mesh hook: on_local_insert(model_id, tokens, tenant_url) publishes
        # tree deltas to replicas (reference mesh/adapters/tree_sync.rs:1-20)
        self.mesh_hook = None

    # ---- tree plumbing ---------------------------------------------------
    def _token_tree(self, model_id: str):
        tree = self.token_trees.get(model_id)
        if tree is None:
            tree = make_token_tree(
                page_size=self.cfg.block_size,
                gpu=self.cfg.gpu_tree,
                device=self.cfg.gpu_tree_device,
            )
            self.token_trees[model_id] = tree
        return tree

    def _text_tree(self, model_id: str):
        tree = self.text_trees.get(model_id)
        if tree is None:
            tree = make_text_tree(gpu=self.cfg.gpu_tree, device=self.cfg.gpu_tree_device)
            self.text_trees[model_id] = tree
        return tree

    # ---- imbalance triggers (reference :283) -----------------------------
    def is_imbalanced(self, workers: Sequence[Worker], candidates: Sequence[int]) -> bool:
        usages = [workers[i].token_usage for i in candidates if workers[i].token_usage is not None]
        if usages:
            max_u, min_u = max(usages), min(usages)
            if max_u > self.cfg.overload_token_usage_threshold:
                return True
            if max_u - min_u > self.cfg.balance_token_usage_threshold:
                return True
        loads = [workers[i].active_requests for i in candidates]
        if not loads:
            return False
        max_l, min_l = max(loads), min(loads)
        return (max_l - min_l) > self.cfg.balance_abs_threshold and max_l > min_l * self.cfg.balance_rel_threshold

    # ---- selection -------------------------------------------------------
    def needs_tokens(self) -> bool:
        return True

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        candidates = [i for i, w in enumerate(workers) if w.is_available()]
        if not candidates:
            return None
        model_id = info.model_id or "default"
        min_idx = _min_load_idx(workers, candidates)

        if self.is_imbalanced(workers, candidates):
            # shortest queue; still attribute the path so cache state tracks reality
            if min_idx is not None:
                url = workers[min_idx].url
                if info.tokens is not None:
                    self._token_tree(model_id).insert(list(info.tokens), url)
                elif info.text:
                    self._text_tree(model_id).insert_text(info.text, url)
            self._maybe_evict()
            return min_idx

        selected: Optional[int] = None
        if info.tokens is not None and self.indexer is not None and self.indexer.has_events(model_id):
            selected = self._select_event_driven(workers, candidates, info, model_id, min_idx)
        elif info.tokens is not None:
            selected = self._select_with_tokens(workers, candidates, info, model_id, min_idx)
        elif info.text:
            selected = self._select_with_text(workers, candidates, info, model_id, min_idx)
        else:
            selected = min_idx
        self._maybe_evict()
        return selected if selected is not None else (candidates[0] if candidates else None)

    def _choose(self, workers, candidates, min_idx, model_id=None, tokens=None):
        """Builds the choose_tenant closure shared by token and text paths."""
        picked: List[Optional[int]] = [None]

        def choose(result):
            if result.match_rate > self.cfg.cache_threshold and result.tenant is not None:
                for i in candidates:
                    if workers[i].url == result.tenant:
                        picked[0] = i
                        return workers[i].url
            picked[0] = min_idx
            return workers[min_idx].url if min_idx is not None else None

        return picked, choose

    def _select_with_tokens(self, workers, candidates, info, model_id, min_idx):
        tree = self._token_tree(model_id)
        picked, choose = self._choose(workers, candidates, min_idx)
        tree.match_and_insert(list(info.tokens), choose)
        if self.mesh_hook is not None and picked[0] is not None:
            self.mesh_hook(model_id, list(info.tokens), workers[picked[0]].url)
        return picked[0]

    def _select_with_text(self, workers, candidates, info, model_id, min_idx):
        tree = self._text_tree(model_id)
        picked, choose = self._choose(workers, candidates, min_idx)
        tree.match_and_insert_text(info.text, choose)
        return picked[0]

    def _select_event_driven(self, workers, candidates, info, model_id, min_idx):
        """Overlap scoring against engine-reported KV blocks (reference :890):
        max overlap, tie-break lower load then smaller tree, fallback min-load."""
        scores = self.indexer.find_matches(model_id, list(info.tokens))
        best, best_key = None, None
        for i in candidates:
            url = workers[i].url
            score = scores.get(url, 0)
            key = (-score, workers[i].active_requests, self.indexer.tree_size(model_id, url))
            if best_key is None or key < best_key:
                best, best_key = i, key
        if best is not None and scores.get(workers[best].url, 0) > 0:
            return best
        return min_idx

    # ---- batched path (GPU kernel services the whole batch) ---------------
    def select_worker_batch(
        self, workers: Sequence[Worker], infos: Sequence[SelectWorkerInfo]
    ) -> List[Optional[int]]:
        token_batches: List[Tuple[int, SelectWorkerInfo]] = []
        out: List[Optional[int]] = [None] * len(infos)
        for j, info in enumerate(infos):
            if info.tokens is not None:
                token_batches.append((j, info))
            else:
                out[j] = self.select_worker(workers, info)
        if not token_batches:
            return out
        model_id = token_batches[0][1].model_id or "default"
        tree = self._token_tree(model_id)
        if hasattr(tree, "match_and_insert_batch"):
            candidates = [i for i, w in enumerate(workers) if w.is_available()]
            min_idx = _min_load_idx(workers, candidates)
            imbalanced = self.is_imbalanced(workers, candidates)
            results = tree.match_and_insert_batch(
                [info.tokens for _, info in token_batches],  # ndarray passes through
                urls=[w.url for w in workers],
                candidates=candidates,
                loads=[w.active_requests for w in workers],
                processed=[w.processed_requests for w in workers],
                cache_threshold=self.cfg.cache_threshold,
                min_load_idx=min_idx,
                imbalanced=imbalanced,
            )
            for (j, _), sel in zip(token_batches, results):
                out[j] = sel
        else:
            for j, info in token_batches:
                out[j] = self.select_worker(workers, info)
        return out

    # ---- lifecycle -------------------------------------------------------
    def on_request_complete(self, worker: Worker, info: SelectWorkerInfo, success: bool) -> None:
        pass  # load guard decrements handled by the router's WorkerLoadGuard

    def on_worker_removed(self, worker: Worker) -> None:
        for tree in list(self.token_trees.values()) + list(self.text_trees.values()):
            tree.remove_tenant(worker.url)

    def _maybe_evict(self) -> None:
        now = time.monotonic()
        if now - self._last_eviction < self.cfg.eviction_interval_secs:
            return
        self._last_eviction = now
        for tree in list(self.token_trees.values()) + list(self.text_trees.values()):
            tree.evict(self.cfg.max_tree_size)

    def reset(self) -> None:
        self.token_trees.clear()
        self.text_trees.clear()
