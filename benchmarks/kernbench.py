"""Microbenchmarks of the dfno_amd HIP kernels at flagship shapes.

Prints achieved GB/s (algorithmic bytes / wall) per kernel so regressions
against the ~6.3 TB/s HBM roofline are visible.  Dev tool, single GPU.
"""

import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from dfno_amd import _ext  # noqa: E402

ext = _ext.get(required=True)


def bench(fn, *args, iters=20, warmup=5):
    for _ in range(warmup):
        fn(*args)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn(*args)
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def report(name, dt, bytes_):
    print(f"{name:44s} {dt*1e3:8.3f} ms   {bytes_/dt/1e12:6.2f} TB/s  ({bytes_/1e9:.2f} GB)")


def main():
    S = 64 * 64 * 64 * 30
    dev = "cuda"

    # channel mix fwd (block linear 20->20, no act)
    x = torch.randn(1, 20, S, device=dev)
    W = torch.randn(20, 20, device=dev)
    b0 = torch.empty(0, device=dev)
    dt = bench(lambda: ext.channel_mix_fwd(x, W, b0, False))
    report("channel_mix_fwd 20->20", dt, (20 + 20) * S * 4)

    # channel mix fwd 2->20 + gelu + z
    x2 = torch.randn(1, 2, S, device=dev)
    W2 = torch.randn(20, 2, device=dev)
    b2 = torch.randn(20, device=dev)
    dt = bench(lambda: ext.channel_mix_fwd(x2, W2, b2, True))
    report("channel_mix_fwd 2->20 gelu(+z)", dt, (2 + 40) * S * 4)

    # transposed (grad-x of 20->20)
    gz = torch.randn(1, 20, S, device=dev)
    dt = bench(lambda: ext.channel_mix_fwd_t(gz, W))
    report("channel_mix_fwd_t 20->20", dt, 40 * S * 4)

    # grad-W reduction 20x20
    dt = bench(lambda: ext.channel_mix_bwd_w(gz, x, False))
    report("channel_mix_bwd_w 20x20xS", dt, 40 * S * 4)

    # grad-W reduction 128x20 (proj head gW3)
    gz3 = torch.randn(1, 128, S, device=dev)
    dt = bench(lambda: ext.channel_mix_bwd_w(gz3, x, False))
    report("channel_mix_bwd_w 128x20xS", dt, (128 + 20) * S * 4)

    # proj head fwd 20->128->1
    W3 = torch.randn(128, 20, device=dev) / 20
    b3 = torch.randn(128, device=dev)
    W4 = torch.randn(1, 128, device=dev) / 128
    b4 = torch.randn(1, device=dev)
    dt = bench(lambda: ext.proj_head_fwd(x, W3, b3, W4, b4))
    report("proj_head_fwd 20->128->1", dt, 21 * S * 4)

    # proj head bwd
    gy = torch.randn(1, 1, S, device=dev)
    dt = bench(lambda: ext.proj_head_bwd(gy, x, W3, b3, W4))
    report("proj_head_bwd (gz3 write)", dt, (1 + 20 + 128) * S * 4)

    # fully-fused head backward (no gz3 in HBM; reads x+gy, writes gx)
    dt = bench(lambda: ext.proj_head_bwd_fused(gy, x, W3, b3, W4))
    report("proj_head_bwd_fused", dt, (20 + 1 + 20) * S * 4)

    # fused trunk mix backward (gz as an LDS tile; gz streamed = res grad)
    zmix = torch.randn(1, 20, S, device=dev)
    gymix = torch.randn(1, 20, S, device=dev)
    Wmix = torch.randn(20, 20, device=dev) / 20
    dt = bench(lambda: ext.channel_mix_bwd_fused(gymix, zmix, x, Wmix,
                                                 False, True))
    report("channel_mix_bwd_fused (+gz out)", dt, (3 * 20 + 2 * 20) * S * 4)

    # add gelu
    a = torch.randn(1, 20, S, device=dev)
    bb = torch.randn(1, 20, S, device=dev)
    dt = bench(lambda: ext.add_gelu_fwd(a, bb))
    report("add_gelu (y+z out)", dt, 80 * S * 4)

    # gelu bwd
    dt = bench(lambda: ext.gelu_bwd(a, bb))
    report("gelu_bwd", dt, 60 * S * 4)

    # spectral corners (two-phase block shapes, serial)
    F_ = (24, 24, 24, 8)
    xs = torch.randn(1, 20, *F_, device=dev, dtype=torch.complex64)
    ys = torch.zeros(1, 20, *F_, device=dev, dtype=torch.complex64)
    ws, starts = [], []
    m = (12, 12, 12, 8)
    for c in range(8):
        st = [(0 if (c >> d) & 1 == 0 else F_[d] - m[d]) for d in range(3)] + [0]
        ws.append(torch.randn(20, 20, *m, device=dev, dtype=torch.complex64) / 400)
        starts.append(st)
    wbytes = sum(w.numel() * 8 for w in ws)
    xbytes = xs.numel() * 8
    dt = bench(lambda: ext.spectral_corners_fwd(xs, ws, ys, starts))
    report("spectral_corners_fwd (8 corners)", dt, wbytes + 2 * xbytes)

    dt = bench(lambda: ext.spectral_corners_bwd_x(ys, ws, xs, starts))
    report("spectral_corners_bwd_x (8 corners)", dt, wbytes + 2 * xbytes)

    # reference: raw copy bandwidth
    src = torch.randn(200_000_000 // 4, device=dev)
    dst = torch.empty_like(src)
    dt = bench(lambda: dst.copy_(src))
    report("torch copy_ 200MB (r+w)", dt, 2 * src.numel() * 4)


if __name__ == "__main__":
    if "--dft" not in sys.argv:
        main()


def dft_bench():
    """DFT kernels at flagship shapes."""
    # t-dim rfft: [1,20,64,64,64,30] -> m 8
    x = torch.randn(1, 20, 64, 64, 64, 30, device="cuda")
    dt = bench(lambda: ext.dft_rfft_trunc(x, 5, 8))
    report("dft_rfft_trunc t30->8", dt, (30 + 16) * 5242880 * 4)
    # z-dim c2c: [1,20,64,64,64,8] c64 -> 24
    xz = torch.randn(1, 20, 64, 64, 64, 8, device="cuda", dtype=torch.complex64)
    dt = bench(lambda: ext.dft_c2c(xz, 4, 64, 12, 12, True, 1.0))
    report("dft_c2c z64->24 analysis", dt, (64 + 24) * 8 * 20 * 64 * 64 * 8)
    dtS = bench(lambda: ext.dft_c2c(ext.dft_c2c(xz, 4, 64, 12, 12, True, 1.0), 4, 64, 12, 12, False, 1.0/64))
    report("dft_c2c z analysis+synthesis", dtS, 2 * (64 + 24) * 8 * 20 * 64 * 64 * 8)
    # y-dim c2c on truncated: [1,20,64,64,24,8]
    xy = torch.randn(1, 20, 64, 64, 24, 8, device="cuda", dtype=torch.complex64)
    dt = bench(lambda: ext.dft_c2c(xy, 3, 64, 12, 12, True, 1.0))
    report("dft_c2c y64->24 analysis", dt, (64 + 24) * 8 * 20 * 64 * 24 * 8)
    # irfft
    yt = torch.randn(1, 20, 64, 64, 64, 8, device="cuda", dtype=torch.complex64)
    dt = bench(lambda: ext.dft_pad_irfft(yt, 5, 30, 8))
    report("dft_pad_irfft 8->30", dt, (16 + 30) * 5242880 * 4)
    # torch reference: full rfft
    dt = bench(lambda: torch.fft.rfft(x, dim=5))
    report("torch rfft t30 (no trunc)", dt, (30 + 32) * 5242880 * 4)
    dt = bench(lambda: torch.fft.fft(xz, dim=4))
    report("torch fft z64 (no trunc)", dt, 2 * 64 * 8 * 20 * 64 * 64 * 8)


if "__main__" == __name__ and "--dft" in sys.argv:
    dft_bench()
