"""Fused decode-attention kernel numerics vs plain PyTorch sdpa (fp32 reference)."""
import math

import pytest

torch = pytest.importorskip("torch")
core = pytest.importorskip("smg_amd._core")


@pytest.mark.gpu
class TestAttnDecodeKernel:
    @pytest.mark.parametrize("S,H,T,D,maxseq", [(8, 4, 33, 64, 64), (16, 16, 600, 128, 704), (3, 2, 1, 128, 32)])
    def test_matches_sdpa(self, S, H, T, D, maxseq):
        dev = "cuda:0"
        g = torch.Generator(device=dev).manual_seed(0)
        q = torch.randn(S, H, D, generator=g, device=dev, dtype=torch.float32).to(torch.bfloat16)
        k = torch.randn(S, H, maxseq, D, generator=g, device=dev, dtype=torch.float32).to(torch.bfloat16)
        v = torch.randn(S, H, maxseq, D, generator=g, device=dev, dtype=torch.float32).to(torch.bfloat16)
        pos = torch.randint(0, T, (S,), generator=g, device=dev, dtype=torch.int32)
        out = torch.zeros(S, H, D, device=dev, dtype=torch.bfloat16)
        core.attn_decode(
            q.contiguous().data_ptr(), k.contiguous().data_ptr(), v.contiguous().data_ptr(),
            pos.data_ptr(), out.data_ptr(), S, H, maxseq, D,
            1.0 / math.sqrt(D), torch.cuda.current_stream().cuda_stream,
        )
        torch.cuda.synchronize()
        # fp32 reference with the same inclusive-pos mask
        qf, kf, vf = q.float(), k.float(), v.float()
        kpos = torch.arange(maxseq, device=dev)
        mask = (kpos.unsqueeze(0) <= pos.unsqueeze(1).long()).unsqueeze(1).unsqueeze(1)
        ref = torch.nn.functional.scaled_dot_product_attention(
            qf.unsqueeze(2), kf, vf, attn_mask=mask
        ).squeeze(2)
        err = (out.float() - ref).abs().max().item()
        assert err < 0.05, f"max err {err}"

    def test_engine_decode_with_kernel_matches_sdpa_path(self):
        from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig

        cfg = TorchEngineConfig.tiny()
        cfg.dtype = "bfloat16"
        eng_hip = TorchEngine(cfg, device="cuda:0")
        assert eng_hip._hip_attn is not None, "hip decode attention must be active on GPU"
        eng_ref = TorchEngine(cfg, device="cuda:0")
        eng_ref._hip_attn = None  # force the sdpa path
        prompt = list(range(40))
        r1 = eng_hip.submit(prompt, max_new_tokens=8)
        r2 = eng_ref.submit(prompt, max_new_tokens=8)
        while not eng_hip.finished(r1):
            eng_hip.step()
        while not eng_ref.finished(r2):
            eng_ref.step()
        out1, out2 = eng_hip.collect(r1), eng_ref.collect(r2)
        # greedy streams from identical weights; bf16 kernel vs sdpa may
        # diverge on near-ties — require the first tokens to agree
        assert out1[:2] == out2[:2]
