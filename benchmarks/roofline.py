"""Streaming roofline on this box: read / read+write / copy at several sizes."""
import time, torch
def t(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.time()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.time()-t0)/n
GB=1e9
a = torch.randn(1310720*256, device="cuda")     # 1.34 GB
b = torch.empty_like(a)
dt = t(lambda: b.copy_(a));  print(f"copy 1.3GB r+w: {2*a.numel()*4/GB/dt:.2f} GB/s")
dt = t(lambda: a.sum());     print(f"reduce 1.3GB read: {a.numel()*4/GB/dt:.2f} GB/s")
dt = t(lambda: torch.add(a, 1.0, out=b)); print(f"add-scalar r+w: {2*a.numel()*4/GB/dt:.2f} GB/s")
c = torch.randn(64, 5242880, device="cuda")  # strided-row read: 64 rows of 20MB
dt = t(lambda: c.sum(dim=1)); print(f"rowwise reduce: {c.numel()*4/GB/dt:.2f} GB/s")
