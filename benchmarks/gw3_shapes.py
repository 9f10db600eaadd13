"""Probe library GEMM vs our kernel for the gW3 shape [128,S]x[S,20]."""
import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from dfno_amd import _ext
ext = _ext.get(required=True)
S = 64*64*64*30
gz = torch.randn(1, 128, S, device="cuda")
x = torch.randn(1, 20, S, device="cuda")

def t(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.time()-t0)/n*1000

print("ours(glds3 24,2):", round(t(lambda: ext.channel_mix_bwd_w(gz, x, False)), 3), "ms")
print("matmul:", round(t(lambda: torch.matmul(gz[0], x[0].t())), 3), "ms")
print("einsum:", round(t(lambda: torch.einsum('os,is->oi', gz[0], x[0])), 3), "ms")
g2 = gz[0].t().contiguous()   # [S,128]
x2 = x[0].t().contiguous()    # [S,20]
print("matmul_tn:", round(t(lambda: torch.matmul(g2.t(), x2)), 3), "ms")
