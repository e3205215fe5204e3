"""EPD disaggregation OVER THE PLANE: pixels to an encode rank, vision
embeddings rank-to-rank into the decode engine's prefill (MI355X-native
equivalent of the reference's encode fleet + Mooncake/NIXL embedding
transport behind the gRPC EncodeStage)."""
import os
import subprocess
import sys

import pytest
import torch

from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig
from smg_amd.multimodal.encoder import ToyVisionEncoder

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestEmbeddedPrefill:
    def _cfg(self):
        return TorchEngineConfig.tiny()

    def test_mm_embed_changes_output_deterministically(self):
        cfg = self._cfg()
        enc = ToyVisionEncoder(cfg.d_model, image_size=64, patch=16, seed=5)
        g = torch.Generator().manual_seed(1)
        px = (torch.rand(3, 40, 52, generator=g) * 255).to(torch.uint8)
        emb = enc.encode(px)
        assert emb.shape == (16, cfg.d_model)
        prompt = list(range(24))
        outs = []
        for _ in range(2):
            eng = TorchEngine(cfg, device="cpu")
            r = eng.submit(prompt, 5, mm_embed=emb)
            while not eng.finished(r):
                eng.step()
            outs.append(eng.collect(r))
        assert outs[0] == outs[1] and len(outs[0]) == 5  # deterministic
        eng = TorchEngine(cfg, device="cpu")
        r = eng.submit(prompt, 5)
        while not eng.finished(r):
            eng.step()
        assert eng.collect(r) != outs[0]  # embeddings actually conditioned it

    def test_seq_accounting_includes_embeddings(self):
        cfg = self._cfg()
        eng = TorchEngine(cfg, device="cpu")
        emb = torch.zeros(16, cfg.d_model)
        r = eng.submit(list(range(10)), 3, mm_embed=emb)
        eng.step()
        req = eng._requests[r]
        assert eng._seq_len_host[req.slot] >= 16 + 10

    def test_accept_embed_pairs_either_order(self):
        cfg = self._cfg()
        emb = torch.ones(16, cfg.d_model) * 0.01
        # embed first, then submit
        a = TorchEngine(cfg, device="cpu")
        a.accept_embed(42, emb)
        ra = a.submit(list(range(12)), 4, rid=42)
        while not a.finished(ra):
            a.step()
        # submit queued, then embed arrives before the admitting step
        b = TorchEngine(cfg, device="cpu")
        rb = b.submit(list(range(12)), 4, rid=42)
        b.accept_embed(42, emb)
        while not b.finished(rb):
            b.step()
        assert a.collect(ra) == b.collect(rb)

    def test_mm_requests_skip_prefix_cache(self):
        cfg = self._cfg()
        eng = TorchEngine(cfg, device="cpu")
        emb = torch.zeros(16, cfg.d_model)
        prompt = list(range(40))
        r1 = eng.submit(prompt, 3, mm_embed=emb)
        while not eng.finished(r1):
            eng.step()
        hits0 = eng.prefix_cache_hits
        r2 = eng.submit(prompt, 3, mm_embed=emb)
        while not eng.finished(r2):
            eng.step()
        assert eng.prefix_cache_hits == hits0  # no (wrong) text-prefix restore


@pytest.mark.gpu
def test_embedded_prefill_gpu_matches_eager():
    """EPD embedded prefill on cuda:0: HIP-kernel engine vs torch-eager
    reference engine (same weights) agree on the greedy stream head."""
    cfg = TorchEngineConfig.tiny()
    cfg.dtype = "bfloat16"
    enc = ToyVisionEncoder(cfg.d_model, image_size=64, patch=16, seed=5,
                           device="cuda:0", dtype=torch.bfloat16)
    g = torch.Generator().manual_seed(3)
    px = (torch.rand(3, 40, 52, generator=g) * 255).to(torch.uint8)
    emb = enc.encode(px)
    prompt = list(range(24))
    eng = TorchEngine(cfg, device="cuda:0")
    ref = TorchEngine(cfg, device="cuda:0")
    ref._hip_attn = None
    ref._hip_fused = None
    r1 = eng.submit(prompt, 6, mm_embed=emb)
    r2 = ref.submit(prompt, 6, mm_embed=emb)
    while not eng.finished(r1):
        eng.step()
    while not ref.finished(r2):
        ref.step()
    a, b = eng.collect(r1), ref.collect(r2)
    assert len(a) == 6
    assert a[:2] == b[:2]  # bf16 kernel vs eager may diverge on near-ties


class TestGatewayEpd:
    def _world1_roundtrip(self, device: str):
        from smg_amd.config import PolicyConfig
        from smg_amd.policies import create_policy
        from smg_amd.routers.rccl_router import TickGateway
        from smg_amd.workers.worker import Worker

        cfg = TorchEngineConfig.tiny()
        if device != "cpu":
            cfg.dtype = "bfloat16"
        eng = TorchEngine(cfg, device=device)
        gw = TickGateway([Worker("rccl://rank-0", rccl_rank=0)],
                         create_policy(PolicyConfig(name="round_robin")),
                         local_engine=eng)
        g = torch.Generator().manual_seed(4)
        px = (torch.rand(3, 36, 44, generator=g) * 255).to(torch.uint8)
        seen = []
        gw.on_event = lambda rid, tok, fl: seen.append((rid, tok, fl))
        gw.submit(list(range(16)), 4, rid=9, pixels=px)
        for _ in range(60):
            gw.tick()
            if gw.completed_total:
                break
        gw._drain_pipeline()
        assert gw.completed_total == 1
        toks = [t for rid, t, _ in seen if rid == 9]
        assert len(toks) == 4 and all(t >= 0 for t in toks)
        return toks

    @pytest.mark.gpu
    def test_world1_epd_serving_gpu(self):
        """The full pixels->encode->embedded-prefill->decode serving loop on
        cuda:0 (hardware coverage of the gateway EPD wrapper)."""
        self._world1_roundtrip("cuda:0")

    def test_world1_local_encode_serving(self):
        """TickGateway EPD at world 1: pixels encode locally, the embedded
        request decodes on the local engine through the normal tick loop."""
        from smg_amd.config import PolicyConfig
        from smg_amd.policies import create_policy
        from smg_amd.routers.rccl_router import TickGateway
        from smg_amd.workers.worker import Worker

        cfg = TorchEngineConfig.tiny()
        eng = TorchEngine(cfg, device="cpu")
        gw = TickGateway([Worker("rccl://rank-0", rccl_rank=0)],
                         create_policy(PolicyConfig(name="round_robin")),
                         local_engine=eng)
        g = torch.Generator().manual_seed(4)
        px = (torch.rand(3, 36, 44, generator=g) * 255).to(torch.uint8)
        prompt = list(range(16))
        seen = []
        gw.on_event = lambda rid, tok, fl: seen.append((rid, tok, fl))
        gw.submit(prompt, 4, rid=9, pixels=px)
        for _ in range(60):
            gw.tick()
            if gw.completed_total:
                break
        gw._drain_pipeline()
        assert gw.completed_total == 1
        toks = [t for rid, t, _ in seen if rid == 9]
        # reference with the same default toy encoder
        enc = ToyVisionEncoder(cfg.d_model, image_size=64, patch=16)
        ref_eng = TorchEngine(cfg, device="cpu")
        r = ref_eng.submit(prompt, 4, mm_embed=enc.encode(px))
        while not ref_eng.finished(r):
            ref_eng.step()
        assert toks == ref_eng.collect(r)

    def test_epd_gateway_three_rank_gloo(self):
        """Full EPD serving topology over gloo: gateway + encode rank +
        decode rank, mixed text/multimodal traffic."""
        env = dict(os.environ, MASTER_ADDR="127.0.0.1")
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=3",
            "--master-addr", "127.0.0.1", "--master-port", "29617",
            os.path.join(REPO, "tests", "epd_gateway_helper.py"),
        ]
        out = subprocess.run(cmd, capture_output=True, text=True, timeout=300,
                             cwd=REPO, env=env)
        assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
        assert "EPD_GW_OK" in out.stdout, out.stdout[-2000:]


def test_epd_three_rank_gloo():
    """pixels(rank0) -> encode(rank1) -> embeddings(rank2) == single-engine."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=3",
        "--master-addr", "127.0.0.1", "--master-port", "29613",
        os.path.join(REPO, "tests", "epd_plane_helper.py"),
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "EPD_OK" in out.stdout, out.stdout[-2000:]
