"""The non-tree load-balancing policies.

Reference implementations: model_gateway/src/policies/{random,round_robin,
passthrough,power_of_two,least_load,prefix_hash,consistent_hashing,bucket,
manual,dp_min_token}.rs — behavior kept, code new.
"""
from __future__ import annotations

import hashlib
import random as _random
import time
from typing import Dict, Optional, Sequence

from ..config import PolicyConfig
from ..workers.worker import Worker
from .base import DPRankLoadPolicy, LoadBalancingPolicy, SelectWorkerInfo, filter_available


def _hash64(data: bytes) -> int:
    return int.from_bytes(hashlib.blake2b(data, digest_size=8).digest(), "little")


class RandomPolicy(LoadBalancingPolicy):
    name = "random"

    def __init__(self, seed: Optional[int] = None):
        self._rng = _random.Random(seed)

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        return self._rng.choice(avail) if avail else None


class RoundRobinPolicy(LoadBalancingPolicy):
    name = "round_robin"

    def __init__(self):
        self._counter = 0

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        idx = avail[self._counter % len(avail)]
        self._counter += 1
        return idx

    def reset(self) -> None:
        self._counter = 0


class PassthroughPolicy(LoadBalancingPolicy):
    """Single-worker passthrough: valid only with exactly one candidate
    (reference passthrough.rs)."""

    name = "passthrough"

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if len(avail) == 1:
            return avail[0]
        return avail[0] if avail else None


class PowerOfTwoPolicy(LoadBalancingPolicy):
    """Two random choices, pick the lighter (reference power_of_two.rs, using
    monitor-cached loads)."""

    name = "power_of_two"

    def __init__(self, seed: Optional[int] = None):
        self._rng = _random.Random(seed)

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        if len(avail) == 1:
            return avail[0]
        a, b = self._rng.sample(avail, 2)
        wa, wb = workers[a], workers[b]
        ka = (wa.active_requests + wa.inflight_tokens / 4096.0) * wa.cost
        kb = (wb.active_requests + wb.inflight_tokens / 4096.0) * wb.cost
        return a if ka <= kb else b


class LeastLoadPolicy(LoadBalancingPolicy):
    """Token-work scoring with an M/M/1 KV-pressure barrier
    (reference least_load.rs:26-50):

        score = (queued_tokens + inflight_tokens + est_prefill) / throughput
                + kv_weight * k / (1 - k)          where k = engine KV utilization
    """

    name = "least_load"

    def __init__(self, cfg: PolicyConfig):
        self.kv_pressure_weight = cfg.least_load_kv_pressure_weight
        self.default_throughput = cfg.least_load_default_throughput
        self.mean_prefill_tokens = cfg.least_load_mean_prefill_tokens

    def score(self, w: Worker, est_tokens: int) -> float:
        tput = w.gen_throughput or self.default_throughput
        tput = max(tput, 1.0)
        prefill = est_tokens if est_tokens > 0 else self.mean_prefill_tokens
        work = (w.queued_tokens + w.inflight_tokens + prefill) / tput
        k = min(max(w.token_usage or 0.0, 0.0), 0.999)
        return (work + self.kv_pressure_weight * k / (1.0 - k)) * w.cost

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        return min(avail, key=lambda i: self.score(workers[i], info.est_tokens))


class ConsistentHashingPolicy(LoadBalancingPolicy):
    """Hash-ring selection on routing key / text / request id
    (reference consistent_hashing.rs + hash_ring.rs)."""

    name = "consistent_hashing"

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        key = info.routing_key or info.text or info.request_id or ""
        h = _hash64(key.encode() if isinstance(key, str) else bytes(key))
        # rendezvous hashing over the candidate list keeps selection stable
        # under worker add/remove without a shared ring object
        best, best_score = None, -1
        for i in avail:
            s = _hash64(f"{workers[i].url}|{h}".encode())
            if s > best_score:
                best, best_score = i, s
        return best


class PrefixHashPolicy(LoadBalancingPolicy):
    """Hash of the first `prefix_token_count` tokens (or chars) pins a worker,
    with a load-factor escape to least-load (reference prefix_hash.rs)."""

    name = "prefix_hash"

    def __init__(self, cfg: PolicyConfig):
        self.prefix_token_count = cfg.prefix_token_count
        self.load_factor = cfg.prefix_hash_load_factor

    def needs_tokens(self) -> bool:
        return True

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        if info.tokens:
            prefix = bytes(
                b for t in info.tokens[: self.prefix_token_count] for b in int(t & 0xFFFFFFFF).to_bytes(4, "little")
            )
        elif info.text:
            prefix = info.text[: self.prefix_token_count * 4].encode()
        else:
            prefix = (info.routing_key or info.request_id or "").encode()
        h = _hash64(prefix)
        preferred = avail[h % len(avail)]
        loads = [workers[i].active_requests for i in avail]
        mean = sum(loads) / len(loads)
        if workers[preferred].active_requests <= max(self.load_factor * mean, mean + 1):
            return preferred
        return min(avail, key=lambda i: workers[i].active_requests)


class BucketPolicy(LoadBalancingPolicy):
    """Range-partition of the key hash space across workers with a
    least-load overflow valve (reference bucket.rs: contiguous key ranges
    per worker)."""

    name = "bucket"

    def __init__(self, cfg: Optional[PolicyConfig] = None):
        self.load_factor = cfg.prefix_hash_load_factor if cfg else 1.5

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        key = info.routing_key or info.tenant_id or info.text or info.request_id or ""
        h = _hash64(key.encode())
        span = (1 << 64) // len(avail)
        preferred = avail[min(h // span, len(avail) - 1)]
        loads = [workers[i].active_requests for i in avail]
        mean = sum(loads) / len(loads)
        if workers[preferred].active_requests <= max(self.load_factor * mean, mean + 2):
            return preferred
        return min(avail, key=lambda i: workers[i].active_requests)


class ManualPolicy(LoadBalancingPolicy):
    """Sticky routing-key -> worker assignment with idle eviction and
    configurable assignment of new keys (reference manual.rs:
    modes random | min_load | min_group)."""

    name = "manual"

    def __init__(self, cfg: PolicyConfig, seed: Optional[int] = None, clock=time.monotonic):
        self.max_idle_secs = cfg.max_idle_secs
        self.assignment_mode = cfg.assignment_mode
        self._assignments: Dict[str, str] = {}  # routing key -> worker url
        self._last_used: Dict[str, float] = {}
        self._rng = _random.Random(seed)
        self._clock = clock

    def select_worker(self, workers: Sequence[Worker], info: SelectWorkerInfo) -> Optional[int]:
        avail = filter_available(workers)
        if not avail:
            return None
        key = info.routing_key
        if not key:
            return min(avail, key=lambda i: workers[i].active_requests)
        self._evict_idle()
        url = self._assignments.get(key)
        if url is not None:
            for i in avail:
                if workers[i].url == url:
                    self._last_used[key] = self._clock()
                    return i
        # (re)assign
        if self.assignment_mode == "min_load":
            idx = min(avail, key=lambda i: workers[i].active_requests)
        elif self.assignment_mode == "min_group":
            counts = {workers[i].url: 0 for i in avail}
            for u in self._assignments.values():
                if u in counts:
                    counts[u] += 1
            idx = min(avail, key=lambda i: counts[workers[i].url])
        else:
            idx = self._rng.choice(avail)
        self._assignments[key] = workers[idx].url
        self._last_used[key] = self._clock()
        return idx

    def _evict_idle(self) -> None:
        now = self._clock()
        stale = [k for k, t in self._last_used.items() if now - t > self.max_idle_secs]
        for k in stale:
            self._assignments.pop(k, None)
            self._last_used.pop(k, None)

    def on_worker_removed(self, worker: Worker) -> None:
        stale = [k for k, u in self._assignments.items() if u == worker.url]
        for k in stale:
            self._assignments.pop(k, None)
            self._last_used.pop(k, None)

    def assignments(self) -> Dict[str, str]:
        return dict(self._assignments)


class MinimumTokensPolicy(DPRankLoadPolicy):
    """DP-rank selection: lowest-load rank, incremented at selection
    (reference dp_min_token.rs:12 -> WorkerLoadManager::
    select_and_increment_lowest_dp_load monitor.rs:164)."""

    def select_dp_rank(self, worker: Worker) -> Optional[int]:
        if not worker.dp_loads:
            return None
        rank = min(range(len(worker.dp_loads)), key=lambda r: worker.dp_loads[r])
        worker.dp_loads[rank] += 1
        return rank
