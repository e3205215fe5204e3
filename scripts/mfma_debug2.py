import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
import torch.nn.functional as F
from smg_amd import _core
dev = "cuda:0"; stream = torch.cuda.current_stream().cuda_stream
M, K, N = 520, 2048, 3072
g = torch.Generator(device=dev).manual_seed(0)
a = (torch.randn(M, K, generator=g, device=dev) / math.sqrt(K)).to(torch.bfloat16)
w = (torch.randn(K, N, generator=g, device=dev) / math.sqrt(K)).to(torch.bfloat16)
gln = (1 + 0.1 * torch.randn(K, generator=g, device=dev)).to(torch.bfloat16)
wt = (w.float() * gln.float().unsqueeze(1)).t().contiguous().to(torch.bfloat16)
invrms = torch.zeros(M, device=dev, dtype=torch.float32)
out = torch.zeros(M, N, device=dev, dtype=torch.bfloat16)
_core.row_invrms(a.data_ptr(), invrms.data_ptr(), M, K, 1e-5, stream)
_core.rms_gemm(a.data_ptr(), wt.data_ptr(), invrms.data_ptr(), out.data_ptr(), M, K, N, stream)
torch.cuda.synchronize()
ref32 = F.rms_norm(a.float(), (K,), weight=gln.float(), eps=1e-5) @ w.float()
tb = ((a @ wt.t().contiguous()).float() * invrms[:, None]).to(torch.bfloat16)
e1 = (out.float() - ref32).abs().max().item()
e2 = (tb.float() - ref32).abs().max().item()
e3 = (out.float() - tb.float()).abs().max().item()
print(f"e1 kernel-vs-fp32 {e1:.5f}  e2 torchbf16-vs-fp32 {e2:.5f}  e3 kernel-vs-torchbf16 {e3:.6f}")
print("out std", ref32.std().item())
