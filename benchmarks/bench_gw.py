import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from dfno_amd import _ext
ext = _ext.get(required=True)
S = 64*64*64*30
gz = torch.randn(1, 20, S, device="cuda")
x = torch.randn(1, 20, S, device="cuda")
for _ in range(3):
    ext.channel_mix_bwd_w(gz, x, False)
torch.cuda.synchronize()
t0 = time.time()
for _ in range(10):
    ext.channel_mix_bwd_w(gz, x, False)
torch.cuda.synchronize()
print("ms:", (time.time()-t0)*100)
