"""Pipelined (deferred-read) engine stepping and gateway ticks.

step_launch/step_finish must be observationally identical to step(): the
engine's control flow is host-deterministic (done is a token-count
condition), so deferring the D2H token reads one tick may delay WHEN events
are published but never change their content or order.

Reference analog: SGLang's overlapped scheduler (CPU scheduling runs while
the GPU executes the previous batch) — here as a lag-1 handle resolved by
the serving tick loop (smg_amd/routers/rccl_router.py).
"""
import random

import pytest

from smg_amd.config import PolicyConfig
from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig
from smg_amd.policies import CacheAwarePolicy
from smg_amd.routers.rccl_router import TickGateway
from smg_amd.workers.worker import Worker


def _cfg():
    cfg = TorchEngineConfig.tiny()
    cfg.max_slots = 8
    cfg.max_seq = 128
    return cfg


def _traffic(n=12, seed=7):
    rng = random.Random(seed)
    return [([rng.randrange(100) for _ in range(rng.randrange(8, 24))], rng.randrange(3, 9))
            for _ in range(n)]


def _run_sync(reqs):
    eng = TorchEngine(_cfg(), device="cpu")
    for i, (toks, mn) in enumerate(reqs):
        eng.submit(toks, mn, rid=i + 1)
    events = []
    for _ in range(200):
        eng.step()
        events.extend(eng.drain_events())
        if not eng.running and not eng.waiting:
            break
    return events


def _run_pipelined(reqs):
    eng = TorchEngine(_cfg(), device="cpu")
    for i, (toks, mn) in enumerate(reqs):
        eng.submit(toks, mn, rid=i + 1)
    events, prev = [], None
    for _ in range(200):
        launched = eng.step_launch()
        if prev is not None:
            eng.step_finish(prev)
            events.extend(eng.drain_events())
        prev = launched
        if not eng.running and not eng.waiting:
            break
    if prev is not None:
        eng.step_finish(prev)
        events.extend(eng.drain_events())
    return events


class TestStepLaunchFinish:
    def test_identical_event_stream(self):
        reqs = _traffic()
        assert _run_sync(reqs) == _run_pipelined(reqs)

    def test_handle_values_patch_generated(self):
        eng = TorchEngine(_cfg(), device="cpu")
        eng.submit([1, 2, 3, 4], 4, rid=9)
        done = []
        for _ in range(20):
            h = eng.step_launch()
            eng.step_finish(h)
            done.extend(eng.drain_events())
            if not eng.running and not eng.waiting:
                break
        toks = [t for rid, t, fl in done if rid == 9]
        assert len(toks) == 4
        assert all(isinstance(t, int) and t >= 0 for t in toks)  # placeholders patched

    def test_step_equals_launch_finish(self):
        a = TorchEngine(_cfg(), device="cpu")
        b = TorchEngine(_cfg(), device="cpu")
        for e in (a, b):
            e.submit([5, 6, 7], 3, rid=1)
        pa = a.step()
        pb = b.step_finish(b.step_launch())
        assert pa == pb
        assert a.drain_events() == b.drain_events()


def _gw(pipeline):
    eng = TorchEngine(_cfg(), device="cpu")
    w = Worker("rccl://rank-0", model_id="m", rccl_rank=0)
    pol = CacheAwarePolicy(PolicyConfig(name="cache_aware", block_size=16))
    seen = {}
    gw = TickGateway([w], pol, local_engine=eng, model_id="m", pipeline=pipeline,
                     on_event=lambda rid, tok, fl: seen.setdefault(rid, []).append((tok, fl)))
    return gw, seen


class TestPipelinedGateway:
    def test_same_streams_as_sync(self):
        reqs = _traffic(n=10, seed=3)
        streams = []
        for pipeline in (False, True):
            gw, seen = _gw(pipeline)
            for i, (toks, mn) in enumerate(reqs):
                gw.submit(toks, mn, rid=i + 1)
            for _ in range(300):
                gw.tick()
                if not gw.inflight and gw.pending_count == 0:
                    break
            gw._drain_pipeline()
            assert not gw.inflight
            assert gw.completed_total == len(reqs)
            streams.append(seen)
        assert streams[0] == streams[1]

    def test_drain_pipeline_counts_tail_completions(self):
        gw, _ = _gw(True)
        gw.submit([1, 2, 3], 2, rid=1)
        for _ in range(10):
            gw.tick()
            if gw.completed_total:
                break
        gw._drain_pipeline()
        assert gw.completed_total == 1
        assert gw._prev_handle is None


@pytest.mark.gpu
class TestPrefillGraph:
    def _run(self, graphs_env, prompts, suffixes=None):
        import os

        import torch
        from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig

        old = os.environ.pop("SMG_NO_PREFILL_GRAPH", None)
        if graphs_env:
            os.environ["SMG_NO_PREFILL_GRAPH"] = "1"
        try:
            cfg = TorchEngineConfig.tiny()
            cfg.max_slots = 16
            cfg.max_seq = 256
            cfg.prefill_group = 8
            eng = TorchEngine(cfg, device="cuda:0", graphs=True)
            outs = {}
            for wave in (prompts, suffixes or []):
                rids = [eng.submit(p, 6, rid=f"r{i}-{len(outs)}") for i, p in enumerate(wave)]
                for _ in range(80):
                    eng.step()
                    if not eng.running and not eng.waiting:
                        break
                for r in rids:
                    outs[r] = eng.collect(r)
            torch.cuda.synchronize()
            return outs
        finally:
            os.environ.pop("SMG_NO_PREFILL_GRAPH", None)
            if old is not None:
                os.environ["SMG_NO_PREFILL_GRAPH"] = old

    def test_graphed_prefill_matches_eager(self):
        """Batched prefill through the captured hipGraph must reproduce the
        eager launch stream exactly — including the prefix-cache-hit suffix
        wave (uniform start0 > 0, the merge path)."""
        import random

        rng = random.Random(11)
        base = [rng.randrange(100) for _ in range(64)]
        prompts = [base + [rng.randrange(100) for _ in range(16)] for _ in range(8)]
        suffixes = [base + [rng.randrange(100) for _ in range(16)] for _ in range(8)]
        a = self._run(False, prompts, suffixes)  # graphs on
        b = self._run(True, prompts, suffixes)   # SMG_NO_PREFILL_GRAPH=1
        assert a == b
