"""Decode-attention kernel microbench at bench shapes (no graphs, no engine)
— safe under rocprofv3 --pmc, and the A/B harness for kernel iterations.

Usage: python scripts/attn_bench.py [--slots 520] [--t 592] [--iters 50]
       [--kv8] [--split N]
Prints per-call ms, streamed GB, achieved TB/s vs the 8 TB/s HBM3E peak.
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from smg_amd import _core as core  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--slots", type=int, default=520)
    p.add_argument("--kvh", type=int, default=4)
    p.add_argument("--heads", type=int, default=16)
    p.add_argument("--t", type=int, default=592, help="live KV window per slot")
    p.add_argument("--max-seq", type=int, default=640)
    p.add_argument("--hd", type=int, default=128)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--kv8", action="store_true")
    p.add_argument("--split", type=int, default=1)
    p.add_argument("--v2", action="store_true", help="v9 kernel (dot2 + two-tile rounds)")
    p.add_argument("--check", action="store_true", help="compare v2 vs v1 output")
    args = p.parse_args()

    dev = "cuda:0"
    S, KVH, H, T, MS, HD = args.slots, args.kvh, args.heads, args.t, args.max_seq, args.hd
    kdt = torch.float8_e4m3fn if args.kv8 else torch.bfloat16
    torch.manual_seed(0)
    k = torch.randn(S, KVH, MS, HD, device=dev, dtype=torch.bfloat16).to(kdt)
    v = torch.randn(S, KVH, MS, HD, device=dev, dtype=torch.bfloat16).to(kdt)
    q = torch.randn(S, H, HD, device=dev, dtype=torch.bfloat16)
    pos = torch.full((S,), T - 1, device=dev, dtype=torch.int32)
    out = torch.empty(S, H, HD, device=dev, dtype=torch.bfloat16)
    stream = torch.cuda.current_stream().cuda_stream
    scale = HD ** -0.5

    def call():
        if args.split > 1:
            part = torch.empty(S, KVH, args.split, (H // KVH), HD, device=dev, dtype=torch.float32)
            ml = torch.empty(S, KVH, args.split, (H // KVH), 2, device=dev, dtype=torch.float32)
            core.attn_decode_split(q.data_ptr(), k.data_ptr(), v.data_ptr(), pos.data_ptr(),
                                   out.data_ptr(), part.data_ptr(), ml.data_ptr(),
                                   S, H, KVH, args.split, MS, HD, scale, stream,
                                   1 if args.kv8 else 0)
        elif args.v2:
            core.attn_decode2(q.data_ptr(), k.data_ptr(), v.data_ptr(), pos.data_ptr(),
                              out.data_ptr(), S, H, MS, HD, scale, stream,
                              1 if args.kv8 else 0, KVH)
        else:
            core.attn_decode(q.data_ptr(), k.data_ptr(), v.data_ptr(), pos.data_ptr(),
                             out.data_ptr(), S, H, MS, HD, scale, stream,
                             1 if args.kv8 else 0, KVH)

    if args.check:
        ref = torch.empty_like(out)
        core.attn_decode(q.data_ptr(), k.data_ptr(), v.data_ptr(), pos.data_ptr(),
                         ref.data_ptr(), S, H, MS, HD, scale, stream, 1 if args.kv8 else 0, KVH)
        core.attn_decode2(q.data_ptr(), k.data_ptr(), v.data_ptr(), pos.data_ptr(),
                          out.data_ptr(), S, H, MS, HD, scale, stream, 1 if args.kv8 else 0, KVH)
        torch.cuda.synchronize()
        err = (out.float() - ref.float()).abs().max().item()
        print(f"v2-vs-v1 max abs err: {err:.5f}")
    for _ in range(5):
        call()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(args.iters):
        call()
    t1.record()
    torch.cuda.synchronize()
    ms = t0.elapsed_time(t1) / args.iters
    bytes_per = S * T * 2 * KVH * HD * (1 if args.kv8 else 2)
    tbs = bytes_per / (ms * 1e-3) / 1e12
    print(f"slots={S} kvh={KVH} T={T} hd={HD} kv8={args.kv8} split={args.split}: "
          f"{ms*1e3:.1f} us/call  {bytes_per/1e6:.0f} MB streamed  "
          f"{tbs:.2f} TB/s ({tbs/8*100:.0f}% of 8 TB/s peak)")


if __name__ == "__main__":
    main()
