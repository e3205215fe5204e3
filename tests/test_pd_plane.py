"""PD disaggregation OVER THE RCCL PLANE: prefill on one rank, KV handoff as
an xGMI/gloo p2p tensor send, decode on another (MI355X-native equivalent of
the reference's engine-side Mooncake/NIXL KV transfer behind pd_router
bootstrap metadata — here the engine is ours, so the gateway orchestrates
the handoff itself through the lockstep plane)."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestEngineKvHandoff:
    """Engine-level export/import: splitting prefill and decode across two
    engines with identical weights must reproduce the single-engine stream."""

    def test_prefill_park_and_events(self):
        from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig

        eng = TorchEngine(TorchEngineConfig.tiny(), device="cpu")
        rid = eng.submit(list(range(24)), max_new_tokens=6, prefill_only=True)
        for _ in range(20):
            eng.step()
            evs = eng.drain_events()
            if evs:
                break
        flags = {f for _, _, f in evs}
        assert TorchEngine.PLEN_INFO in flags and TorchEngine.PREFILLED in flags
        plen = next(t for _, t, f in evs if f == TorchEngine.PLEN_INFO)
        assert plen == 24
        # parked: not finished, slot retained, no further decode
        assert not eng.finished(rid)
        eng.step()
        assert eng.drain_events() == []

    def test_handoff_matches_single_engine(self):
        from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig

        cfg = TorchEngineConfig.tiny()
        single = TorchEngine(cfg, device="cpu")
        prefill = TorchEngine(cfg, device="cpu")  # same seed => same weights
        decode = TorchEngine(cfg, device="cpu")
        prompt = list(range(40))
        # reference stream
        r0 = single.submit(prompt, max_new_tokens=6)
        while not single.finished(r0):
            single.step()
        ref = single.collect(r0)
        # PD: prefill leg
        rid = prefill.submit(prompt, max_new_tokens=6, rid="pd1", prefill_only=True)
        evs = []
        while not any(f == TorchEngine.PREFILLED for _, _, f in evs):
            prefill.step()
            evs += prefill.drain_events()
        plen = next(t for _, t, f in evs if f == TorchEngine.PLEN_INFO)
        first = next(t for _, t, f in evs if f == TorchEngine.PREFILLED)
        kv, plen2, first2 = prefill.export_kv("pd1")
        assert plen2 == plen and first2 == first
        assert kv.shape == prefill.kv_transfer_shape(plen)
        # prefill slot freed
        assert len(prefill._free_slots) == cfg.max_slots
        # decode leg
        assert decode.import_kv("pd1", kv, plen, first, max_new=6)
        toks = [first]
        while not decode.finished("pd1"):
            decode.step()
            toks += [t for _, t, f in decode.drain_events() if f in (0, 1)]
        assert toks == ref, (toks, ref)

    def test_import_kv_full_pool(self):
        from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig

        cfg = TorchEngineConfig.tiny()
        cfg.max_slots = 1
        eng = TorchEngine(cfg, device="cpu")
        eng.submit(list(range(8)), 50)
        eng.step()  # occupies the only slot (still decoding)
        import torch

        assert not eng.import_kv("x", torch.zeros(eng.kv_transfer_shape(8)), 8, 1, 2)  # pool full


def test_bench_pd_three_rank_gloo():
    """bench --pd over torchrun world 3 (gloo): rank1 prefill, ranks 0/2
    decode; KV hands off over the plane and completions flow end to end."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "3",
        "--master-addr", "127.0.0.1", "--master-port", "29585",
        "bench.py", "--gpus", "3", "--tiny", "--pd",
        "--steps", "2", "--warmup", "1", "--reqs-per-step", "4",
        "--concurrency", "6", "--prefix-len", "32", "--suffix-len", "8", "--max-new", "4",
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    result = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert result["value"] > 0
    assert "pd over rccl-xgmi" in result["config"]["parallelism"]
    assert "1p+2d" in result["config"]["parallelism"]


def test_rccl_pd_serving_three_rank_gloo():
    """`smg launch --connection-mode rccl --pd-disaggregation` world 3:
    chat served with the prefill leg on rank 1 and KV handoffs to decode
    ranks 0/2."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["SMG_TEST_PORT"] = "31899"
    env["SMG_TEST_PD"] = "1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "3",
        "--master-addr", "127.0.0.1", "--master-port", "29589",
        os.path.join(REPO, "tests", "rccl_serve_helper.py"),
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")][-1]
    results = json.loads(line[7:])
    assert results["completions"] == [4] * 6
    # the prefill rank handled every prefill leg; decode ranks produced tokens
    assert results["worker_processed"].get("rccl://rank-1", 0) >= 6
