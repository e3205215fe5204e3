#!/usr/bin/env python3
"""Determine mfma_f32_16x16x32_bf16's true A/B fragment k-ordering by
running the probe kernel against candidate layouts (GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from smg_amd import _core


def run_probe(A, B, a_map, b_map):
    """A [16,32], B [32,16] f32; maps: (lane, e) -> (i, k) / (k, j)."""
    dev = "cuda:0"
    af = torch.zeros(64, 8, dtype=torch.bfloat16, device=dev)
    bf = torch.zeros(64, 8, dtype=torch.bfloat16, device=dev)
    At = torch.from_numpy(A).to(dev)
    Bt = torch.from_numpy(B).to(dev)
    for lane in range(64):
        for e in range(8):
            i, k = a_map(lane, e)
            af[lane, e] = At[i, k].to(torch.bfloat16)
            k2, j = b_map(lane, e)
            bf[lane, e] = Bt[k2, j].to(torch.bfloat16)
    out = torch.zeros(64, 4, dtype=torch.float32, device=dev)
    _core.mfma_probe(af.contiguous().data_ptr(), bf.contiguous().data_ptr(),
                     out.data_ptr(), torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    # C map: col = lane&15, row = (lane>>4)*4 + reg
    D = np.zeros((16, 16), dtype=np.float32)
    o = out.cpu().numpy()
    for lane in range(64):
        for r in range(4):
            D[(lane >> 4) * 4 + r, lane & 15] = o[lane, r]
    return D


def main():
    rng = np.random.default_rng(0)
    A = (rng.standard_normal((16, 32)) / 4).astype(np.float32)
    B = (rng.standard_normal((32, 16)) / 4).astype(np.float32)
    # bf16-round the reference inputs
    Ab = torch.from_numpy(A).to(torch.bfloat16).float().numpy()
    Bb = torch.from_numpy(B).to(torch.bfloat16).float().numpy()
    ref = Ab @ Bb

    candidates = {
        "L1 k=8*(l>>4)+e":        (lambda l, e: (l & 15, 8 * (l >> 4) + e),
                                   lambda l, e: (8 * (l >> 4) + e, l & 15)),
        "L2 two-K16 blocks":      (lambda l, e: (l & 15, 4 * (l >> 4) + (e % 4) + 16 * (e // 4)),
                                   lambda l, e: (4 * (l >> 4) + (e % 4) + 16 * (e // 4), l & 15)),
        "L3 k=(l>>4)+4*e":        (lambda l, e: (l & 15, (l >> 4) + 4 * e),
                                   lambda l, e: ((l >> 4) + 4 * e, l & 15)),
        "L4 k=2*(l>>4)+e%2+8*(e//2)": (lambda l, e: (l & 15, 2 * (l >> 4) + (e % 2) + 8 * (e // 2)),
                                   lambda l, e: (2 * (l >> 4) + (e % 2) + 8 * (e // 2), l & 15)),
    }
    # also probe the C-map alternative (row/col swap)
    for name, (am, bm) in candidates.items():
        D = run_probe(A, B, am, bm)
        err = np.abs(D - ref).max()
        errT = np.abs(D.T - ref).max()
        print(f"{name:30s} err={err:.4f}  errT(swapped C)={errT:.4f}", flush=True)


if __name__ == "__main__":
    main()
