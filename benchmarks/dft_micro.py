import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from dfno_amd import _ext
ext = _ext.get(required=True)
x = torch.randn(1, 20, 64, 64, 64, 30, device="cuda")
xz = torch.randn(1, 20, 64, 64, 64, 8, device="cuda", dtype=torch.complex64)
for _ in range(8):
    ext.dft_rfft_trunc(x, 5, 8)
    ext.dft_c2c(xz, 4, 64, 12, 12, True, 1.0)
torch.cuda.synchronize()
print("done")
