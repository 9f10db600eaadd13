"""Correctness + bandwidth check of the glds3 grad-W kernel on MI355X."""
import sys, time
from pathlib import Path
import torch
sys.path.insert(0, "/root/repo")
from dfno_amd import _ext
ext = _ext.get(required=True)

torch.manual_seed(0)
fails = 0
# correctness matrix: clamp paths (O % 4*OW != 0), padding (I < ICAP), B>1,
# bias on/off, odd S (non-vec fallback)
for (B, O, I, S) in [(1, 20, 20, 262144), (1, 19, 17, 262144), (2, 8, 8, 65536),
                     (1, 32, 32, 131072), (1, 24, 24, 131072), (1, 5, 3, 65536),
                     (1, 20, 20, 262147), (1, 16, 8, 131072), (3, 7, 25, 4096),
                     (1, 20, 20, 1024)]:
    for bias in (False, True):
        gz = torch.randn(B, O, S, device="cuda")
        x = torch.randn(B, I, S, device="cuda")
        gW, gb = ext.channel_mix_bwd_w(gz, x, bias)
        ref = torch.einsum("bos,bis->oi", gz, x)
        err = (gW - ref).abs().max() / ref.abs().max()
        ok = err < 1e-4
        if bias:
            rb = gz.sum(dim=(0, 2))
            erb = (gb - rb).abs().max() / rb.abs().max()
            ok = ok and erb < 1e-4
        if not ok:
            fails += 1
            print(f"FAIL B={B} O={O} I={I} S={S} bias={bias} err={err:.2e}")
print("correctness fails:", fails)

# perf on the three flagship call-site shapes
for (B, O, I, S, tag) in [(1, 20, 20, 64*64*64*30, "Wres/linres"),
                          (1, 20, 20, 64*64*64*15, "halfS"),
                          (1, 8, 8, 64*64*64*30, "small")]:
    gz = torch.randn(B, O, S, device="cuda")
    x = torch.randn(B, I, S, device="cuda")
    for _ in range(3):
        ext.channel_mix_bwd_w(gz, x, False)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(20):
        ext.channel_mix_bwd_w(gz, x, False)
    torch.cuda.synchronize()
    ms = (time.time() - t0) * 1000 / 20
    OW = 4 if I <= 8 else 2
    o_tiles = (O + 4 * OW - 1) // (4 * OW)
    gb_moved = (o_tiles * I * S * B + (O + 4 * OW - 1) // (4 * OW) * 4 * OW * S * B) * 4 / 1e9
    print(f"{tag}: {ms:.3f} ms  algBW~{(O+I)*S*B*4/1e9/ms*1000:.2f} GB/s  "
          f"actualBW~{gb_moved/ms*1000:.2f} GB/s")
