"""128^3 capability check: N>64 dims run through the torch.fft fallback."""
import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import dfno_amd as dfno
from dfno_amd.partition import Partition

P = Partition((0,), (1, 1, 1, 1, 1, 1))
model = dfno.DistributedFNONd(P, [1, 2, 128, 128, 128, 1], 10, 12, (16, 16, 16, 4),
                              num_blocks=2, device="cuda", dtype=torch.float32)
x = torch.randn(1, 2, 128, 128, 128, 1, device="cuda")
crit = dfno.DistributedRelativeLpLoss(P)
tgt = torch.randn(1, 1, 128, 128, 128, 10, device="cuda")
for it in range(2):
    t0 = time.time()
    y = model(x)
    loss = crit(y, tgt)
    loss.backward()
    model.zero_grad(set_to_none=True)
    torch.cuda.synchronize()
    print(f"iter {it}: {time.time()-t0:.2f}s loss={loss.item():.4f} "
          f"peakGB={torch.cuda.max_memory_allocated()/2**30:.1f}")
assert torch.isfinite(loss)
print("128^3 OK")
