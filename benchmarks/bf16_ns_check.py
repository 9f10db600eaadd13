"""bf16 compute-config characterization on the 2D+time Navier-Stokes shape
(BASELINE.json config #2: NS FNO 64x64, T_in 10 -> T_out 40, width 20,
modes (4,4,4), 1 MI355X).

Measures train-step and eval-forward latency bf16 vs fp32 and the output
agreement, plus the flagship 3D shape for reference.  Redirect stdout into
profiles/bf16_report.md.
"""
import sys, time
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import dfno_amd as dfno
from dfno_amd.partition import Partition
from dfno_amd.optim import Adam


def timeit(fn, n=30, warm=8):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3


def characterize(name, pdim, in_shape, out_t, width, modes, blocks, batch_out):
    P = Partition((0,), tuple([1] * pdim))
    kw = dict(out_timesteps=out_t, width=width, modes=modes,
              num_blocks=blocks, device="cuda")
    f32 = dfno.DistributedFNONd(P, in_shape, **kw)
    b16 = dfno.DistributedFNONd(P, in_shape, dtype=torch.bfloat16, **kw)
    sd = {k: (v.bfloat16() if v.dtype == torch.float32 else v)
          for k, v in f32.state_dict().items()}
    b16.load_state_dict(sd)
    x = torch.rand(*in_shape, device="cuda")
    with torch.no_grad():
        y32 = f32(x)
        y16 = b16(x.bfloat16())
        rel = ((y16.float() - y32).norm() / y32.norm().clamp_min(1e-30)).item()
        e32 = timeit(lambda: f32(x))
        e16 = timeit(lambda: b16(x.bfloat16()))
    crit = dfno.DistributedRelativeLpLoss(P)
    tgt32 = torch.rand(*batch_out, device="cuda")
    tgt16 = tgt32.bfloat16()
    x16 = x.bfloat16()
    o32 = Adam(f32.parameters(), lr=1e-3)
    o16 = Adam(b16.parameters(), lr=1e-3)

    def step(m, o, xx, tt):
        o.zero_grad(set_to_none=True)
        loss = crit(m(xx), tt)
        loss.backward()
        o.step()

    t32 = timeit(lambda: step(f32, o32, x, tgt32), n=20, warm=6)
    t16 = timeit(lambda: step(b16, o16, x16, tgt16), n=20, warm=6)
    print(f"| {name} | {e32:.2f} | {e16:.2f} | {t32:.2f} | {t16:.2f} | {rel:.4f} |")


print("""# bf16 compute-config characterization

| config | eval fp32 (ms) | eval bf16 (ms) | step fp32 (ms) | step bf16 (ms) | fwd rel err |
|---|---|---|---|---|---|""")
characterize("NS 2D+time 64x64, T 10->40, w20, m(4,4,4)", 5,
             [1, 1, 64, 64, 10], 40, 20, (4, 4, 4), 4,
             [1, 1, 64, 64, 40])
characterize("two-phase 3D 64^3x30, w20, m(12,12,12,8)", 6,
             [1, 2, 64, 64, 64, 1], 30, 20, (12, 12, 12, 8), 4,
             [1, 1, 64, 64, 64, 30])
print("""
bf16 = bf16 activation storage + fp32 arithmetic everywhere: native
pointwise kernels (csrc/bf16.hip) with MFMA grad-W, bf16-IO transforms
(incl. the raw-bf16 glds DMA r2c), bf16-IO fused heads and the one-kernel
trunk mix backward; the spectral core stays complex64.  At the flagship
this config now runs AHEAD of fp32 (BASELINE.md).""")
