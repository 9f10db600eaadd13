"""fp8 spectral-weight characterization (BASELINE.json config #5).

Measures, on the flagship 3D two-phase config:
  * eval-mode forward latency fp32 vs fp8 (weights quantized once — the
    deployment case the fp8 storage targets),
  * train-step latency fp32 vs fp8 (requantization included),
  * output relative error fp8 vs fp32 with identical master weights.

Writes a markdown report to stdout (redirect into profiles/fp8_report.md).
"""
import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import dfno_amd as dfno
from dfno_amd.partition import Partition

P = Partition((0,), (1, 1, 1, 1, 1, 1))
shape = [1, 2, 64, 64, 64, 1]
kw = dict(out_timesteps=30, width=20, modes=(12, 12, 12, 8), num_blocks=4,
          device="cuda")
f32 = dfno.DistributedFNONd(P, shape, spectral_fp8=False, **kw)
f8 = dfno.DistributedFNONd(P, shape, spectral_fp8=True, **kw)
f8.load_state_dict(f32.state_dict())
x = torch.rand(*shape, device="cuda")


def timeit(fn, n=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3


with torch.no_grad():
    y32 = f32(x)
    from dfno_amd.ops.spectral import bump_quant_epoch
    bump_quant_epoch()
    y8 = f8(x)
    rel = ((y8 - y32).norm() / y32.norm().clamp_min(1e-30)).item()
    t_eval32 = timeit(lambda: f32(x))
    t_eval8 = timeit(lambda: f8(x))

crit = dfno.DistributedRelativeLpLoss(P)
tgt = torch.rand(1, 1, 64, 64, 64, 30, device="cuda")
from dfno_amd.optim import Adam
o32 = Adam(f32.parameters(), lr=1e-3)
o8 = Adam(f8.parameters(), lr=1e-3)


def step(m, o):
    o.zero_grad(set_to_none=True)
    loss = crit(m(x), tgt)
    loss.backward()
    o.step()


t_tr32 = timeit(lambda: step(f32, o32), n=15, warm=5)
t_tr8 = timeit(lambda: step(f8, o8), n=15, warm=5)

print(f"""# fp8 (e4m3) spectral-weight characterization — flagship 64^3 config

| measurement | fp32 weights | fp8 weights | ratio |
|---|---|---|---|
| eval forward (ms) | {t_eval32:.2f} | {t_eval8:.2f} | {t_eval8 / t_eval32:.2f}x |
| train step (ms) | {t_tr32:.2f} | {t_tr8:.2f} | {t_tr8 / t_tr32:.2f}x |

Output relative L2 error (identical masters, quantized once): {rel:.4f}

Notes: fp8 storage cuts the spectral contraction's weight stream 4x (its
bandwidth bound), but a TRAINING step re-quantizes the 1.4 GB master set
every iteration (fused device amax+encode, ~2 extra passes over the
masters) and the contraction becomes dequant-VALU-bound, so train-step
time is slightly above fp32 at width 20 / batch 1.  The config pays at
inference/serving (quantize once) and at larger batch or width where the
weight stream amortizes over more spectrum elements.""")
