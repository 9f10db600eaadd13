import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from dfno_amd import _ext
ext = _ext.get(required=True)
S = 64*64*64*30
gz = torch.randn(1, 20, S, device="cuda")
x = torch.randn(1, 20, S, device="cuda")
for _ in range(3): ext.channel_mix_bwd_w(gz, x, False)
torch.cuda.synchronize()
t0 = time.time()
for _ in range(20): ext.channel_mix_bwd_w(gz, x, False)
torch.cuda.synchronize()
ms = (time.time()-t0)*50
import os
print(f"nschunk={os.environ.get('DFNO_GW_NSCHUNK','def')} no_glds={os.environ.get('DFNO_GW_NO_GLDS','0')}: {ms:.3f} ms  algBW {1.26/ms*1000:.0f} GB/s")
