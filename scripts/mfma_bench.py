#!/usr/bin/env python3
"""Microbench: hand-written MFMA rms_gemm vs torch (rms_norm + hipBLASLt mm)
on the decode projection shapes.  Run on a GPU box."""
import math
import sys
import time

import torch
import torch.nn.functional as F

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from smg_amd import _core


def bench(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    dev = "cuda:0"
    stream = torch.cuda.current_stream().cuda_stream
    for M, K, N, tag in [
        (520, 2048, 3072, "qkv (gqa)"),
        (520, 2048, 13056, "w13"),
        (520, 2048, 6144, "qkv (mha)"),
        (128, 2048, 3072, "qkv small-batch"),
    ]:
        g = torch.Generator(device=dev).manual_seed(0)
        a = (torch.randn(M, K, generator=g, device=dev) / math.sqrt(K)).to(torch.bfloat16)
        w = (torch.randn(K, N, generator=g, device=dev) / math.sqrt(K)).to(torch.bfloat16)
        gln = torch.ones(K, device=dev, dtype=torch.bfloat16)
        wt = (w.float() * gln.float().unsqueeze(1)).t().contiguous().to(torch.bfloat16)
        invrms = torch.zeros(M, device=dev, dtype=torch.float32)
        out = torch.zeros(M, N, device=dev, dtype=torch.bfloat16)

        def mfma():
            _core.row_invrms(a.data_ptr(), invrms.data_ptr(), M, K, 1e-5, stream)
            _core.rms_gemm(a.data_ptr(), wt.data_ptr(), invrms.data_ptr(), out.data_ptr(),
                           M, K, N, stream)

        def blaslt():
            F.rms_norm(a, (K,), weight=gln, eps=1e-5) @ w

        def blaslt_raw():
            torch.mm(a, w)

        t_m = bench(mfma)
        t_b = bench(blaslt)
        t_r = bench(blaslt_raw)
        gflop = 2 * M * K * N / 1e9
        print(f"{tag:18s} M{M} K{K} N{N}: mfma {t_m:7.1f}us ({gflop/t_m*1e3:6.0f} TF) | "
              f"rms+blaslt {t_b:7.1f}us ({gflop/t_b*1e3:6.0f} TF) | mm-only {t_r:7.1f}us",
              flush=True)


if __name__ == "__main__":
    sys.exit(main())
