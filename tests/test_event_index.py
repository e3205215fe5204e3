"""PositionalIndexer + event-driven cache-aware routing tests (reference
event_tree.rs colocated tests + cache_aware.rs:890 select_worker_event_driven)."""
import pytest

from smg_amd.config import PolicyConfig
from smg_amd.kvindex.event_index import PositionalIndexer, compute_content_hashes
from smg_amd.policies import CacheAwarePolicy, SelectWorkerInfo
from smg_amd.workers.worker import Worker


def toks(n, base=0):
    return [base + i for i in range(n)]


class TestContentHashes:
    def test_chained_prefix_commitment(self):
        a = compute_content_hashes(toks(64), 16)
        b = compute_content_hashes(toks(64), 16)
        assert a == b and len(a) == 4
        # same block content at different prefix -> different hash (chained)
        c = compute_content_hashes(toks(16, base=1) + toks(48), 16)
        assert c[1] != a[1]

    def test_partial_tail_dropped(self):
        assert len(compute_content_hashes(toks(70), 16)) == 4


class TestIndexer:
    def test_stored_and_find(self):
        ix = PositionalIndexer(block_size=16)
        t = toks(64)
        ix.apply_stored("m", "http://w0", compute_content_hashes(t, 16))
        scores = ix.find_matches("m", t)
        assert scores == {"http://w0": 64}

    def test_prefix_semantics(self):
        ix = PositionalIndexer(block_size=16)
        t = toks(64)
        h = compute_content_hashes(t, 16)
        ix.apply_stored("m", "http://w0", h)
        ix.apply_stored("m", "http://w1", h[:2])  # only first 2 blocks
        q = ix.find_matches("m", t)
        assert q["http://w0"] == 64
        assert q["http://w1"] == 32

    def test_removed(self):
        ix = PositionalIndexer(block_size=16)
        t = toks(64)
        h = compute_content_hashes(t, 16)
        ix.apply_stored("m", "http://w0", h)
        ix.apply_removed("m", "http://w0", h[2:])
        assert ix.find_matches("m", t)["http://w0"] == 32

    def test_no_match_after_worker_removed(self):
        ix = PositionalIndexer(block_size=16)
        t = toks(32)
        ix.apply_stored("m", "http://w0", compute_content_hashes(t, 16))
        ix.remove_worker("m", "http://w0")
        assert ix.find_matches("m", t) == {}


class TestEventDrivenPolicy:
    def test_overlap_routing(self):
        ix = PositionalIndexer(block_size=16)
        cfg = PolicyConfig(name="cache_aware", gpu_tree=False, block_size=16)
        policy = CacheAwarePolicy(cfg, indexer=ix)
        workers = [Worker("http://w0", model_id="m"), Worker("http://w1", model_id="m")]
        t = toks(64)
        ix.apply_stored("m", "http://w1", compute_content_hashes(t, 16))
        sel = policy.select_worker(workers, SelectWorkerInfo(model_id="m", tokens=t))
        assert sel == 1

    def test_tie_break_by_load_then_blocks(self):
        ix = PositionalIndexer(block_size=16)
        cfg = PolicyConfig(name="cache_aware", gpu_tree=False, block_size=16)
        policy = CacheAwarePolicy(cfg, indexer=ix)
        workers = [Worker("http://w0", model_id="m"), Worker("http://w1", model_id="m")]
        t = toks(64)
        h = compute_content_hashes(t, 16)
        ix.apply_stored("m", "http://w0", h)
        ix.apply_stored("m", "http://w1", h)
        workers[0].active_requests = 9
        sel = policy.select_worker(workers, SelectWorkerInfo(model_id="m", tokens=t))
        assert sel == 1  # equal overlap, lower load wins

    def test_no_events_falls_back_to_tree(self):
        ix = PositionalIndexer(block_size=16)
        cfg = PolicyConfig(name="cache_aware", gpu_tree=False, block_size=16)
        policy = CacheAwarePolicy(cfg, indexer=ix)
        workers = [Worker("http://w0", model_id="m"), Worker("http://w1", model_id="m")]
        t = toks(64)
        a = policy.select_worker(workers, SelectWorkerInfo(model_id="m", tokens=t))
        b = policy.select_worker(workers, SelectWorkerInfo(model_id="m", tokens=t))
        assert a == b  # gateway-tree affinity


def test_kv_event_monitor_e2e(runner):
    """gRPC SubscribeKvEvents -> monitor -> indexer, against the mock engine."""
    import asyncio

    from smg_amd.grpc.servicer import serve_grpc_worker
    from smg_amd.mock.engine import SimConfig
    from smg_amd.workers.kv_event_monitor import KvEventMonitor
    from smg_amd.workers.registry import WorkerRegistry

    async def run():
        server, adapter, port = await serve_grpc_worker(port=0, sim_config=SimConfig(speedup=100.0))
        registry = WorkerRegistry()
        monitor = KvEventMonitor(registry)
        await monitor.start()
        registry.register(Worker(f"grpc://127.0.0.1:{port}", model_id="mock-model"))
        # drive a request through the sim so it stores KV blocks
        sim_req = adapter.engine.sim.submit(list(range(48)), 2)
        for _ in range(100):
            if adapter.engine.sim.total_generated >= 2:
                break
            await asyncio.sleep(0.02)
        await asyncio.sleep(0.3)  # let the event stream deliver
        scores = monitor.indexer.find_matches("mock-model", list(range(48)))
        await monitor.stop()
        await adapter.stop()
        server.stop(grace=None)
        assert scores, "indexer should have matched the stored prefix"

    runner(run())
