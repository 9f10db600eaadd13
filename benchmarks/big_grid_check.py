"""Large-grid capability check: 128^3 (and optionally 256^3) fwd+bwd with the
NATIVE big-N transforms (no torch.fft in the hot path — VERDICT.md r1 item 2).

  python benchmarks/big_grid_check.py [--grid 128] [--blocks 2] [--width 12]

256^3 at width 20 is the BASELINE.json config #4 memory-capability point
(8-GPU model-parallel in production; a single 288 GB MI355X holds the
1/8-shard footprint several times over, so --grid 256 on one GPU doubles as
the memory-budget check).
"""
import argparse
import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import dfno_amd as dfno
from dfno_amd import dispatch
from dfno_amd.partition import Partition

p = argparse.ArgumentParser()
p.add_argument("--grid", type=int, default=128)
p.add_argument("--blocks", type=int, default=2)
p.add_argument("--width", type=int, default=12)
p.add_argument("--out-t", type=int, default=10)
p.add_argument("--modes", type=int, nargs="+", default=None)
args = p.parse_args()
g = args.grid
modes = tuple(args.modes) if args.modes else (16, 16, 16, 4)

P = Partition((0,), (1, 1, 1, 1, 1, 1))
model = dfno.DistributedFNONd(P, [1, 2, g, g, g, 1], args.out_t, args.width,
                              modes, num_blocks=args.blocks, device="cuda",
                              dtype=torch.float32)
x = torch.randn(1, 2, g, g, g, 1, device="cuda")
crit = dfno.DistributedRelativeLpLoss(P)
tgt = torch.randn(1, 1, g, g, g, args.out_t, device="cuda")
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
# the whole point: no op left the native kernels for eager torch.fft
dispatch.assert_all_native()
print(f"{g}^3 OK (all hot ops native)")
