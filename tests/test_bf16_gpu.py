"""bf16 compute-config tests (BASELINE.json config #2).

Native bf16-storage kernels (csrc/bf16.hip: fp32 arithmetic, bf16 IO) are
checked against plain fp32 torch references at bf16 tolerances, and the full
bf16 model fwd+bwd is compared against the fp32 model with identical weights.
"""

import pytest
import torch
import torch.nn.functional as F

import dfno_amd as dfno

pytestmark = pytest.mark.gpu

BT = 2 ** -7   # bf16 mantissa step: tolerance scale


def _ext():
    from dfno_amd import _ext as e
    return e.get(required=True)


@pytest.mark.parametrize("I,O,act,bias", [
    (20, 20, True, True),    # block channel mix
    (20, 128, True, True),   # projection lift
    (128, 1, False, True),   # projection head (ores path)
    (2, 20, True, True),     # channel lift
])
def test_bf16_channel_mix_fwd(I, O, act, bias):
    ext = _ext()
    torch.manual_seed(0)
    B, S = 2, 1024
    x = (torch.rand(B, I, S, device="cuda") - 0.5)
    W = (torch.rand(O, I, device="cuda") - 0.5) / I
    b = (torch.rand(O, device="cuda") - 0.5) if bias else None

    xb, Wb = x.bfloat16(), W.bfloat16()
    bb = b.bfloat16() if bias else torch.empty(0, dtype=torch.bfloat16, device="cuda")
    empty = torch.empty(0, dtype=torch.bfloat16, device="cuda")
    y, z = ext.bf16_channel_mix(xb, Wb, bb, act, False, act, empty)

    ref = torch.einsum("oi,bis->bos", Wb.float(), xb.float())
    if bias:
        ref = ref + bb.float().view(1, -1, 1)
    zr = ref
    if act:
        ref = F.gelu(ref)
    assert torch.allclose(y.float(), ref, rtol=BT, atol=BT), \
        f"max {(y.float()-ref).abs().max()}"
    if act:
        assert torch.allclose(z.float(), zr, rtol=BT, atol=BT)


def test_bf16_channel_mix_transposed():
    ext = _ext()
    torch.manual_seed(1)
    B, I, O, S = 1, 20, 128, 2048
    gz = (torch.rand(B, O, S, device="cuda") - 0.5).bfloat16()
    W = ((torch.rand(O, I, device="cuda") - 0.5) / I).bfloat16()
    empty = torch.empty(0, dtype=torch.bfloat16, device="cuda")
    gx, _ = ext.bf16_channel_mix(gz, W, empty, False, True, False, empty)
    ref = torch.einsum("oi,bos->bis", W.float(), gz.float())
    assert torch.allclose(gx.float(), ref, rtol=BT, atol=BT)


def test_bf16_grad_w():
    # S % 256 != 0 -> the LDS-tiled VALU kernel (the MFMA path's fallback)
    ext = _ext()
    torch.manual_seed(2)
    B, I, O, S = 2, 20, 24, 4224
    gz = (torch.rand(B, O, S, device="cuda") - 0.5).bfloat16()
    x = (torch.rand(B, I, S, device="cuda") - 0.5).bfloat16()
    gW, gb = ext.bf16_channel_mix_bwd_w(gz, x, True)
    refW = torch.einsum("bos,bis->oi", gz.float(), x.float())
    refb = gz.float().sum(dim=(0, 2))
    assert torch.allclose(gW, refW, rtol=2e-3, atol=refW.abs().max() * 2e-3), \
        f"max {(gW-refW).abs().max()}"
    assert torch.allclose(gb, refb, rtol=2e-3, atol=refb.abs().max() * 2e-3)


@pytest.mark.parametrize("I,O,bias,S", [
    (20, 20, False, 4096),    # block residual grad-W (NPAIR=1)
    (20, 24, True, 4096),
    (20, 128, True, 7936),    # projection lift grad-W (NPAIR=4, ragged tiles)
    (2, 20, True, 4096),      # channel-lift grad-W (ragged i-tile)
])
def test_bf16_grad_w_mfma(I, O, bias, S):
    """The MFMA grad-W path (v_mfma_f32_16x16x32_bf16) against einsum.
    Asymmetric inputs so an operand/output transpose cannot pass."""
    ext = _ext()
    torch.manual_seed(7)
    B = 1
    gz = (torch.rand(B, O, S, device="cuda") - 0.3).bfloat16()
    x = (torch.rand(B, I, S, device="cuda") - 0.7).bfloat16()
    gW, gb = ext.bf16_channel_mix_bwd_w(gz, x, bias)
    refW = torch.einsum("bos,bis->oi", gz.float(), x.float())
    assert torch.allclose(gW, refW, rtol=2e-3, atol=refW.abs().max() * 2e-3), \
        f"max {(gW-refW).abs().max()} of {refW.abs().max()}"
    if bias:
        refb = gz.float().sum(dim=(0, 2))
        assert torch.allclose(gb, refb, rtol=2e-3, atol=refb.abs().max() * 2e-3), \
            f"bias max {(gb-refb).abs().max()}"


def test_bf16_gelu_and_add_gelu():
    ext = _ext()
    torch.manual_seed(3)
    a = (torch.randn(4096 * 3, device="cuda")).bfloat16()
    b = (torch.randn(4096 * 3, device="cuda")).bfloat16()
    y = ext.bf16_gelu_fwd(a)
    assert torch.allclose(y.float(), F.gelu(a.float()), rtol=BT, atol=BT)
    y2, z2 = ext.bf16_add_gelu(a, b)
    zf = a.float() + b.float()
    assert torch.allclose(z2.float(), zf, rtol=BT, atol=BT)
    assert torch.allclose(y2.float(), F.gelu(torch.tensor(z2.float())), rtol=BT, atol=BT)
    gy = torch.randn_like(a).bfloat16()
    gz = ext.bf16_gelu_bwd(gy, a)
    zfl = a.float()
    import math
    ref = gy.float() * (0.5 * (1 + torch.erf(zfl / math.sqrt(2.0)))
                        + zfl * torch.exp(-0.5 * zfl * zfl) / math.sqrt(2 * math.pi))
    assert torch.allclose(gz.float(), ref, rtol=BT, atol=BT)


@pytest.mark.parametrize("shape,m", [
    ((1, 4, 9, 9, 30), 8),     # flagship t-dim
    ((2, 3, 6, 64), 12),
])
def test_bf16_io_rfft_and_irfft(shape, m):
    """bf16-IO transform kernels: bf16 in -> c64 spectrum -> bf16 out,
    against the fp32 composition on the same (bf16-rounded) values."""
    from dfno_amd.ops.fft import rfft_trunc, pad_irfft, _t_rfft_trunc, _t_pad_irfft
    torch.manual_seed(9)
    N = shape[-1]
    xb = torch.randn(*shape, device="cuda").bfloat16().requires_grad_(True)
    y = rfft_trunc(xb, -1, m)
    assert y.dtype == torch.complex64
    xr = xb.detach().float().requires_grad_(True)
    yr = _t_rfft_trunc(xr, -1, m)
    assert torch.allclose(y, yr, rtol=1e-4, atol=1e-4), \
        f"r2c fwd {(y-yr).abs().max()}"
    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g)
    assert torch.allclose(xb.grad.float(), xr.grad, rtol=BT, atol=BT), \
        f"r2c bwd {(xb.grad.float()-xr.grad).abs().max()}"

    # inverse with bf16 output
    n_half = N // 2 + 1
    z = torch.randn(*shape[:-1], m, dtype=torch.complex64,
                    device="cuda").requires_grad_(True)
    w = pad_irfft(z, -1, n_half, N, m, out_dtype=torch.bfloat16)
    assert w.dtype == torch.bfloat16
    zr = z.detach().clone().requires_grad_(True)
    wr = _t_pad_irfft(zr, -1, n_half, N, m)
    assert torch.allclose(w.float(), wr, rtol=BT, atol=BT), \
        f"c2r fwd {(w.float()-wr).abs().max()}"
    g2 = torch.randn_like(w)
    w.backward(g2)
    wr.backward(g2.float())
    # grads through a bf16-rounded gx: bf16 tolerance
    assert torch.allclose(z.grad, zr.grad, rtol=BT, atol=BT * 4), \
        f"c2r bwd {(z.grad-zr.grad).abs().max()}"


def test_bf16_model_matches_fp32():
    """Full bf16 model fwd+bwd vs the fp32 model with the SAME weights
    (the 2D+time NS shape of BASELINE config #2, small extents)."""
    torch.manual_seed(4)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1))
    in_shape = [1, 1, 32, 32, 8]
    f32 = dfno.DistributedFNONd(P_x, in_shape, 16, 16, (4, 4, 4),
                                num_blocks=2, device=torch.device("cuda"),
                                dtype=torch.float32)
    b16 = dfno.DistributedFNONd(P_x, in_shape, 16, 16, (4, 4, 4),
                                num_blocks=2, device=torch.device("cuda"),
                                dtype=torch.bfloat16)
    sd = {}
    for k, v in f32.state_dict().items():
        sd[k] = v.bfloat16() if v.dtype == torch.float32 else v
    b16.load_state_dict(sd)

    x = torch.rand(*in_shape, device="cuda")
    y32 = f32(x)
    y16 = b16(x.bfloat16())
    assert y16.dtype == torch.bfloat16
    scale = y32.abs().max().clamp_min(1.0)
    err = (y16.float() - y32).abs().max() / scale
    assert err < 0.06, f"bf16 fwd rel err {err}"

    # backward: loss gradients stay correlated with fp32
    t = torch.rand_like(y32)
    crit32 = dfno.DistributedRelativeLpLoss(P_x)
    crit16 = dfno.DistributedRelativeLpLoss(P_x)
    l32 = crit32(y32, t)
    l16 = crit16(y16, t.bfloat16())
    assert abs(l16.item() - l32.item()) / abs(l32.item()) < 0.05
    l32.backward()
    l16.backward()
    for (n, p32), (_, p16) in zip(f32.named_parameters(), b16.named_parameters()):
        if p32.grad is None or p32.grad.numel() == 0:
            continue
        g32 = p32.grad.flatten()
        g16 = p16.grad.flatten()
        if g32.is_complex():
            g32 = torch.view_as_real(g32).flatten()
            g16 = torch.view_as_real(g16).flatten()
        g32 = g32.float()
        g16 = g16.float()
        if g32.norm() < 1e-12:
            continue
        cos = torch.dot(g16, g32) / (g16.norm() * g32.norm()).clamp_min(1e-30)
        assert cos > 0.98, f"{n}: grad cosine {cos}"


def test_bf16_model_all_native():
    """The bf16 flagship shape must not fall back to eager torch anywhere
    (other than the documented composed-head routing)."""
    from dfno_amd import dispatch
    torch.manual_seed(5)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    model = dfno.DistributedFNONd(P_x, [1, 2, 16, 16, 16, 1], 8, 20,
                                  (4, 4, 4, 3), num_blocks=1,
                                  device=torch.device("cuda"),
                                  dtype=torch.bfloat16)
    x = torch.rand(1, 2, 16, 16, 16, 1, device="cuda", dtype=torch.bfloat16)
    dispatch.reset_fallbacks()
    y = model(x)
    y.float().square().sum().backward()
    torch.cuda.synchronize()
    dispatch.assert_all_native()


import pytest as _pytest


@_pytest.mark.parametrize("B,O2,S", [(1, 1, 4099), (2, 2, 8192)])
def test_bf16_proj_head_fused(B, O2, S):
    """bf16-IO proj head (fwd kernel + fused backward) vs the fp32
    composition at the flagship head shape."""
    import torch.nn.functional as F
    from dfno_amd.ops import proj_head
    torch.manual_seed(11)
    I, M = 20, 128
    x = torch.randn(B, I, S, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    W3 = (torch.randn(M, I, device="cuda", dtype=torch.bfloat16) / I
          ).requires_grad_(True)
    b3 = torch.randn(M, device="cuda", dtype=torch.bfloat16).requires_grad_(True)
    W4 = (torch.randn(O2, M, device="cuda", dtype=torch.bfloat16) / M
          ).requires_grad_(True)
    b4 = torch.randn(O2, device="cuda", dtype=torch.bfloat16).requires_grad_(True)
    y = proj_head(x, W3, b3, W4, b4)
    assert y.dtype == torch.bfloat16
    gy = torch.randn_like(y)
    y.backward(gy)

    xr = x.detach().float().requires_grad_(True)
    W3r = W3.detach().float().requires_grad_(True)
    b3r = b3.detach().float().requires_grad_(True)
    W4r = W4.detach().float().requires_grad_(True)
    b4r = b4.detach().float().requires_grad_(True)
    h = F.gelu(torch.einsum("mi,bis->bms", W3r, xr) + b3r.view(1, -1, 1))
    yr = torch.einsum("om,bms->bos", W4r, h) + b4r.view(1, -1, 1)
    yr.backward(gy.float())
    assert torch.allclose(y.float(), yr, rtol=2e-2, atol=2e-2), \
        f"fwd {(y.float()-yr).abs().max()}"
    for a, b in [(x, xr), (b3, b3r), (b4, b4r)]:
        assert torch.allclose(a.grad.float(), b.grad, rtol=5e-2, atol=5e-2), \
            f"grad {(a.grad.float()-b.grad).abs().max()}"
    # weight grads accumulate over S: looser tolerance
    for a, b in [(W3, W3r), (W4, W4r)]:
        assert torch.allclose(a.grad.float(), b.grad, rtol=5e-2, atol=5e-1), \
            f"wgrad {(a.grad.float()-b.grad).abs().max()}"


def test_bf16_lift_head_fused():
    """bf16-IO lift head (lane-per-k kernels) vs the fp32 composition."""
    import torch.nn.functional as F
    from dfno_amd.ops import lift_head
    torch.manual_seed(12)
    B, C, Wd, Tn, S = 1, 4, 20, 30, 1000
    x = torch.randn(B, C, S, 1, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    W1 = torch.randn(Tn, 1, device="cuda", dtype=torch.bfloat16).requires_grad_(True)
    b1 = torch.randn(Tn, device="cuda", dtype=torch.bfloat16).requires_grad_(True)
    W2 = (torch.randn(Wd, C, device="cuda", dtype=torch.bfloat16) / C
          ).requires_grad_(True)
    b2 = torch.randn(Wd, device="cuda", dtype=torch.bfloat16).requires_grad_(True)
    y = lift_head(x, W1, b1, W2, b2)
    assert y.dtype == torch.bfloat16
    gy = torch.randn_like(y)
    y.backward(gy)

    xr = x.detach().float().requires_grad_(True)
    W1r = W1.detach().float().requires_grad_(True)
    b1r = b1.detach().float().requires_grad_(True)
    W2r = W2.detach().float().requires_grad_(True)
    b2r = b2.detach().float().requires_grad_(True)
    h = F.gelu(torch.einsum("to,bcso->bcst", W1r, xr) + b1r.view(1, 1, 1, -1))
    yr = F.gelu(torch.einsum("wc,bcst->bwst", W2r, h) + b2r.view(1, -1, 1, 1))
    yr.backward(gy.float())
    assert torch.allclose(y.float(), yr, rtol=2e-2, atol=2e-2), \
        f"fwd {(y.float()-yr).abs().max()}"
    assert torch.allclose(x.grad.float(), xr.grad, rtol=5e-2, atol=5e-2), \
        f"gx {(x.grad.float()-xr.grad).abs().max()}"
    for a, b in [(W1, W1r), (b1, b1r), (W2, W2r), (b2, b2r)]:
        assert torch.allclose(a.grad.float(), b.grad, rtol=5e-2, atol=5e-1), \
            f"wgrad {(a.grad.float()-b.grad).abs().max()}"


def test_bf16_rfft_adj_accumulate():
    """bf16 c2r adjoint with the packed-bf16 accumulate operand (the bf16
    stash path) vs unfused adjoint + add."""
    from dfno_amd import _ext
    ext = _ext.get(required=True)
    torch.manual_seed(13)
    for shape, n in [((2, 6, 64, 30), 30), ((1, 3, 17, 31), 31)]:
        m = 8
        gy = torch.randn(*shape[:-1], m, dtype=torch.complex64, device="cuda")
        acc = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
        fused = ext.dft_rfft_trunc_adj_bf16(gy, len(shape) - 1, n, acc)
        base = ext.dft_rfft_trunc_adj_bf16(
            gy, len(shape) - 1, n, torch.empty(0, dtype=torch.bfloat16,
                                               device="cuda"))
        ref = (base.float() + acc.float()).bfloat16()
        assert fused.dtype == torch.bfloat16
        assert torch.allclose(fused.float(), ref.float(), rtol=2e-2,
                              atol=2e-2), \
            f"{(fused.float()-ref.float()).abs().max()}"


def test_bf16_model_stash_grads():
    """bf16 serial model grads with the stash active match fp32 (the
    residual-grad add now folds into the bf16 c2r adjoint writeback)."""
    import dfno_amd as dfno
    torch.manual_seed(21)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    in_shape = [1, 2, 8, 8, 32, 1]
    ref = dfno.DistributedFNONd(P_x, in_shape, 16, 8, (3, 3, 8, 6),
                                num_blocks=2, device=torch.device("cuda"))
    bf = dfno.DistributedFNONd(P_x, in_shape, 16, 8, (3, 3, 8, 6),
                               num_blocks=2, device=torch.device("cuda"),
                               dtype=torch.bfloat16)
    sd = {k: (v.bfloat16() if v.is_floating_point() and v.dtype == torch.float32
              else v) for k, v in ref.state_dict().items()}
    bf.load_state_dict(sd)
    x = torch.rand(*in_shape, device="cuda")
    yr = ref(x)
    yb = bf(x.bfloat16())
    yr.square().mean().backward()
    yb.float().square().mean().backward()
    for (n, pr), (_, pb) in zip(ref.named_parameters(), bf.named_parameters()):
        if pr.grad is None or pr.grad.numel() == 0:
            continue
        a, b = pr.grad.float(), pb.grad.float()
        if a.is_complex():
            a, b = torch.view_as_real(a), torch.view_as_real(b)
        scale = a.abs().max().clamp_min(1e-6)
        assert (a - b).abs().max() / scale < 0.12, \
            f"{n}: rel {(a-b).abs().max()/scale}"
