#!/usr/bin/env python3
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from smg_amd import _core

dev = "cuda:0"
stream = torch.cuda.current_stream().cuda_stream

def run(a, wt, invrms, M, K, N):
    out = torch.zeros(M, N, device=dev, dtype=torch.bfloat16)
    _core.rms_gemm(a.data_ptr(), wt.data_ptr(), invrms.data_ptr(), out.data_ptr(), M, K, N, stream)
    torch.cuda.synchronize()
    return out

# P1: ones — every output must be exactly K
M, K, N = 64, 512, 128
a = torch.ones(M, K, device=dev, dtype=torch.bfloat16)
wt = torch.ones(N, K, device=dev, dtype=torch.bfloat16)
inv = torch.ones(M, device=dev, dtype=torch.float32)
out = run(a, wt, inv, M, K, N)
print("P1 ones: expect", K, "got min", out.min().item(), "max", out.max().item())

# P1b: K-chunk probe: A=1 only in k-octet c -> out = 8 for all
for c in [0, 1, 7, 8, 63]:
    a2 = torch.zeros(M, K, device=dev, dtype=torch.bfloat16); a2[:, c*8:(c+1)*8] = 1
    o = run(a2, wt, inv, M, K, N)
    u = o.float().unique()
    print(f"P1b octet {c}: unique={u.tolist()[:6]}")

# P2: random vs torch bf16 matmul (same rounding class)
g = torch.Generator(device=dev).manual_seed(0)
a = (torch.randn(M, K, generator=g, device=dev) / 22).to(torch.bfloat16)
wt = (torch.randn(N, K, generator=g, device=dev) / 22).to(torch.bfloat16)
out = run(a, wt, inv, M, K, N)
ref = (a @ wt.t().contiguous())
err = (out.float() - ref.float()).abs()
print("P2 vs torch-bf16-mm: max", err.max().item(), "at", [x.item() for x in (err.argmax()//N, err.argmax()%N)])
row_err = err.max(dim=1).values
col_err = err.max(dim=0).values
print("P2 worst rows:", row_err.topk(4).indices.tolist(), "worst cols:", col_err.topk(4).indices.tolist())
print("P2 row-err by row%16 mean:", [round(row_err[r::16].mean().item(), 5) for r in range(0, 16, 4)])
print("P2 col-err by col%32 mean:", [round(col_err[c::32].mean().item(), 5) for c in range(0, 32, 8)])
