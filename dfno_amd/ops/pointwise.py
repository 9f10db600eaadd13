"""Pointwise linear (channel/time mixing) and fused GELU ops.

MI355X-native implementation of the reference's einsum-based pointwise linear
(`torch.einsum(self.eqn, W, x)` at /root/reference/dfno/dfno.py:62) and the
separate `F.gelu` passes (/root/reference/dfno/dfno.py:291,335,338,350).

These ops are HBM-bandwidth-bound on MI355X (arithmetic intensity 2*I*O /
(4*(I+O)) flop/byte is ~5-9 for this model's widths, below the 25 flop/byte
fp32 ridge at 6.3 TB/s), so the native kernels win by FUSING: one kernel does
contraction + bias + GELU, touching the activation once instead of the
reference's three passes (einsum, +=bias, gelu).

Dispatch: CUDA tensors require the HIP extension (fail loudly — no silent
eager fallback on a GPU box); CPU uses pure-torch reference math.  The
channel-contraction kernel covers the shapes this model uses; exotic shapes
raise on GPU only if the extension is missing, otherwise fall back to a
library GEMM (rocBLAS einsum), which is an allowed library path.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F

from .. import _ext
from ..dispatch import note_fallback


def _native_dt(t: torch.Tensor, op: str = "channel_mix") -> bool:
    """fp32/fp64 native-kernel gate; bf16 is handled by the dedicated bf16
    branch at each call site BEFORE this check.  Anything else takes the
    composed-torch path (still on GPU via rocBLAS/eager) with a warn-once."""
    ok = t.dtype in (torch.float32, torch.float64)
    if not ok and t.is_cuda and t.dtype != torch.bfloat16:
        note_fallback(op, f"dtype {t.dtype} has no native kernel")
    return ok


_EMPTY_BF16 = {}


def _ebf(device):
    t = _EMPTY_BF16.get(device)
    if t is None:
        t = torch.empty(0, dtype=torch.bfloat16, device=device)
        _EMPTY_BF16[device] = t
    return t


def _bf16_mix_ok(x3: torch.Tensor, I: int, O: int, op: str) -> bool:
    """bf16-storage channel-mix kernel coverage (csrc/bf16.hip)."""
    if not (x3.is_cuda and x3.dtype == torch.bfloat16):
        return False
    S = x3.shape[2] if x3.dim() == 3 else 0
    ok = S % 8 == 0 and (I <= 32 or O <= 24)
    if not ok:
        note_fallback(op, f"bf16 shape I={I} O={O} S={S} outside kernel range")
    return ok

__all__ = ["linear_nd", "add_gelu", "gelu", "linear_res_gelu"]

_SQRT_2 = math.sqrt(2.0)
_INV_SQRT_2PI = 1.0 / math.sqrt(2.0 * math.pi)


def _gelu_grad(z: torch.Tensor) -> torch.Tensor:
    """d/dz gelu(z) for the exact (erf) gelu, matching F.gelu default."""
    return 0.5 * (1.0 + torch.erf(z / _SQRT_2)) + z * torch.exp(-0.5 * z * z) * _INV_SQRT_2PI


# ---------------------------------------------------------------------------
# channel-contraction (dim=1) fused linear+bias+gelu
# ---------------------------------------------------------------------------

class _ChannelMixFn(torch.autograd.Function):
    """y[b,o,s] = act( sum_i W[o,i] x[b,i,s] + bias[o] ) with x [B,I,S]."""

    @staticmethod
    def forward(ctx, x, W, b, act: bool):
        B, I = x.shape[0], x.shape[1]
        S = x.numel() // max(B * I, 1)
        x3 = x.reshape(B, I, S)
        if x.is_cuda and _native_dt(x):
            ext = _ext.get(required=True)
            y3, z3 = ext.channel_mix_fwd(x3, W, b if b is not None else torch.empty(0, dtype=x.dtype, device=x.device), act)
        elif _bf16_mix_ok(x3, I, W.shape[0], "channel_mix"):
            ext = _ext.get(required=True)
            y3, z3 = ext.bf16_channel_mix(
                x3.contiguous(), W.contiguous(),
                b.contiguous() if b is not None else _ebf(x.device),
                act, False, act, _ebf(x.device))
        else:
            z3 = torch.einsum("oi,bis->bos", W, x3)
            if b is not None:
                z3 = z3 + b.view(1, -1, 1)
            y3 = F.gelu(z3) if act else z3
        ctx.save_for_backward(x3, W, z3 if act else torch.empty(0))
        ctx.act = act
        ctx.has_bias = b is not None
        ctx.x_shape = tuple(x.shape)
        return y3

    @staticmethod
    def backward(ctx, gy):
        x3, W, z3 = ctx.saved_tensors
        act = ctx.act
        gy = gy.contiguous()
        if (act and gy.is_cuda and gy.dtype in (torch.float32, torch.bfloat16)
                and x3.shape[1] == 20 and W.shape[0] == 20
                and gy.dtype == x3.dtype):
            # trunk 20x20: one kernel, gz never materialized (mix_bwd.hip)
            ext = _ext.get(required=True)
            gx, gW, gb, _ = ext.channel_mix_bwd_fused(
                gy, z3.contiguous(), x3.contiguous(),
                W.contiguous().float(), ctx.has_bias, False)
            return (gx.reshape(ctx.x_shape), gW.to(W.dtype),
                    gb.to(W.dtype) if ctx.has_bias else None, None)
        is_bf16 = gy.is_cuda and gy.dtype == torch.bfloat16
        bf16_ok = is_bf16 and _bf16_mix_ok(gy.reshape(x3.shape[0], W.shape[0], -1),
                                           W.shape[0], x3.shape[1], "channel_mix_bwd")
        if act:
            if gy.is_cuda and _native_dt(gy):
                ext = _ext.get(required=True)
                gz = ext.gelu_bwd(gy, z3)
            elif is_bf16 and gy.numel() % 8 == 0:
                ext = _ext.get(required=True)
                gz = ext.bf16_gelu_bwd(gy, z3.contiguous())
            else:
                gz = gy * _gelu_grad(z3)
        else:
            gz = gy
        # grad x: contraction with W^T
        if gy.is_cuda and _native_dt(gy):
            ext = _ext.get(required=True)
            gx = ext.channel_mix_fwd_t(gz, W)  # sum_o W[o,i] gz[b,o,s]
        elif bf16_ok:
            ext = _ext.get(required=True)
            gx, _ = ext.bf16_channel_mix(gz.contiguous(), W.contiguous(),
                                         _ebf(gy.device), False, True, False,
                                         _ebf(gy.device))
        else:
            gx = torch.einsum("oi,bos->bis", W, gz)
        # grad W / b
        if gy.is_cuda and _native_dt(gy) and x3.shape[1] <= 32:
            ext = _ext.get(required=True)
            gW, gb = ext.channel_mix_bwd_w(gz.contiguous(), x3, ctx.has_bias)
            if not ctx.has_bias:
                gb = None
        elif (is_bf16 and x3.shape[1] <= 32 and W.shape[0] <= 128
              and x3.shape[2] % 128 == 0):
            ext = _ext.get(required=True)
            gW, gb = ext.bf16_channel_mix_bwd_w(gz.contiguous(), x3.contiguous(),
                                                ctx.has_bias)
            gW = gW.to(W.dtype)
            gb = gb.to(W.dtype) if ctx.has_bias else None
        else:
            gW = torch.einsum("bos,bis->oi", gz, x3)
            gb = gz.sum(dim=(0, 2)) if ctx.has_bias else None
        return gx.reshape(ctx.x_shape), gW, gb, None


def _linear_lastdim(x: torch.Tensor, W: torch.Tensor, b: Optional[torch.Tensor],
                    act: bool) -> torch.Tensor:
    """Contraction over the trailing (time) dim via a library GEMM:
    y[..., o] = sum_t x[..., t] W[o, t] (+ b[o]), optionally gelu."""
    y = torch.matmul(x, W.t())
    if b is not None:
        y = y + b.reshape(*([1] * (y.dim() - 1)), -1)
    if act:
        y = F.gelu(y)
    return y


def linear_nd(x: torch.Tensor, W: torch.Tensor, b: Optional[torch.Tensor],
              dim: int, activation: Optional[str] = None) -> torch.Tensor:
    """Linear layer along tensor dim ``dim``: out_dim = W.shape[0],
    contraction over W.shape[1].  Optionally fused exact GELU.

    Bias ``b`` may be shaped [O] or broadcastable [1,..,O,..,1] (the
    reference stores it broadcast-shaped, dfno.py:29-35).
    """
    nd = x.dim()
    d = dim % nd
    act = activation == "gelu"
    b_flat = b.reshape(-1) if b is not None else None

    if d == nd - 1:
        return _linear_lastdim(x, W, b_flat, act)

    if d == 1:
        y3 = _ChannelMixFn.apply(x, W, b_flat, act)
        out_shape = list(x.shape)
        out_shape[1] = W.shape[0]
        return y3.reshape(out_shape)

    # general dim: move to 1, recurse
    xm = x.movedim(d, 1).contiguous()
    ym = _ChannelMixFn.apply(xm, W, b_flat, act)
    out_shape = list(xm.shape)
    out_shape[1] = W.shape[0]
    return ym.reshape(out_shape).movedim(1, d).contiguous()


# ---------------------------------------------------------------------------
# fused residual-linear epilogue: gelu(W @ x + res)  (block epilogue with the
# pass-through linear folded in — the reference computes y0 = linear(x) at
# block start and gelu(y0 + y) at the end, dfno.py:244,291; fusing removes
# the y0 materialization entirely)
# ---------------------------------------------------------------------------

class _LinearResGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, W, res):
        B, I = x.shape[0], x.shape[1]
        S = x.numel() // max(B * I, 1)
        x3 = x.reshape(B, I, S)
        r3 = res.reshape(B, W.shape[0], S)
        if x.is_cuda and _native_dt(x):
            ext = _ext.get(required=True)
            y3, z3 = ext.linear_res_gelu_fwd(x3.contiguous(), W, r3.contiguous())
        elif _bf16_mix_ok(x3, x3.shape[1], W.shape[0], "linear_res_gelu"):
            ext = _ext.get(required=True)
            y3, z3 = ext.bf16_channel_mix(x3.contiguous(), W.contiguous(),
                                          _ebf(x.device), True, False, True,
                                          r3.contiguous())
        else:
            z3 = torch.einsum("oi,bis->bos", W, x3) + r3
            y3 = F.gelu(z3)
        ctx.save_for_backward(x3, W, z3)
        ctx.x_shape = tuple(x.shape)
        ctx.res_shape = tuple(res.shape)
        return y3

    @staticmethod
    def backward(ctx, gy):
        x3, W, z3 = ctx.saved_tensors
        gy = gy.contiguous()
        if (gy.is_cuda and gy.dtype in (torch.float32, torch.bfloat16)
                and x3.shape[1] == 20 and W.shape[0] == 20
                and gy.dtype == x3.dtype):
            # trunk 20x20: one kernel; gz comes back as the residual grad
            ext = _ext.get(required=True)
            gx, gW, _, gz = ext.channel_mix_bwd_fused(
                gy, z3.contiguous(), x3.contiguous(),
                W.contiguous().float(), False, True)
            return (gx.reshape(ctx.x_shape), gW.to(W.dtype),
                    gz.reshape(ctx.res_shape))
        if gy.is_cuda and _native_dt(gy):
            ext = _ext.get(required=True)
            gz = ext.gelu_bwd(gy, z3)
            gx = ext.channel_mix_fwd_t(gz, W)
            if x3.shape[1] <= 32:
                gW, _ = ext.channel_mix_bwd_w(gz, x3.contiguous(), False)
            else:
                gW = torch.einsum("bos,bis->oi", gz, x3)
        elif _bf16_mix_ok(x3, W.shape[0], x3.shape[1], "linear_res_gelu_bwd"):
            ext = _ext.get(required=True)
            gz = ext.bf16_gelu_bwd(gy, z3.contiguous())
            gx, _ = ext.bf16_channel_mix(gz, W.contiguous(), _ebf(gy.device),
                                         False, True, False, _ebf(gy.device))
            if (x3.shape[1] <= 32 and W.shape[0] <= 128
                    and x3.shape[2] % 128 == 0):
                gW, _ = ext.bf16_channel_mix_bwd_w(gz, x3.contiguous(), False)
                gW = gW.to(W.dtype)
            else:
                gW = torch.einsum("bos,bis->oi", gz, x3)
        else:
            gz = gy * _gelu_grad(z3)
            gx = torch.einsum("oi,bos->bis", W, gz)
            gW = torch.einsum("bos,bis->oi", gz, x3)
        return gx.reshape(ctx.x_shape), gW, gz.reshape(ctx.res_shape)


def linear_res_gelu(x: torch.Tensor, W: torch.Tensor, res: torch.Tensor) -> torch.Tensor:
    """gelu(W @_channel x + res): the block's residual path in one pass."""
    out = _LinearResGeluFn.apply(x, W, res)
    shape = list(x.shape)
    shape[1] = W.shape[0]
    return out.reshape(shape)


# ---------------------------------------------------------------------------
# fused residual-add + gelu (block epilogue, reference dfno.py:291)
# ---------------------------------------------------------------------------

class _AddGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, bt):
        if a.is_cuda and _native_dt(a):
            ext = _ext.get(required=True)
            y, z = ext.add_gelu_fwd(a.contiguous(), bt.contiguous())
        elif a.is_cuda and a.dtype == torch.bfloat16 and a.numel() % 8 == 0:
            ext = _ext.get(required=True)
            y, z = ext.bf16_add_gelu(a.contiguous(), bt.contiguous())
        else:
            z = a + bt
            y = F.gelu(z)
        ctx.save_for_backward(z)
        return y

    @staticmethod
    def backward(ctx, gy):
        (z,) = ctx.saved_tensors
        gy = gy.contiguous()
        if gy.is_cuda and _native_dt(gy):
            ext = _ext.get(required=True)
            gz = ext.gelu_bwd(gy, z)
        elif gy.is_cuda and gy.dtype == torch.bfloat16 and gy.numel() % 8 == 0:
            ext = _ext.get(required=True)
            gz = ext.bf16_gelu_bwd(gy, z.contiguous())
        else:
            gz = gy * _gelu_grad(z)
        return gz, gz


def add_gelu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """gelu(a + b), one fused pass on GPU."""
    return _AddGeluFn.apply(a, b)


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda and _native_dt(x):
            ext = _ext.get(required=True)
            y = ext.gelu_fwd(x.contiguous())
        elif x.is_cuda and x.dtype == torch.bfloat16 and x.numel() % 8 == 0:
            ext = _ext.get(required=True)
            y = ext.bf16_gelu_fwd(x.contiguous())
        else:
            y = F.gelu(x)
        ctx.save_for_backward(x)
        return y

    @staticmethod
    def backward(ctx, gy):
        (x,) = ctx.saved_tensors
        gy = gy.contiguous()
        if gy.is_cuda and _native_dt(gy):
            ext = _ext.get(required=True)
            return ext.gelu_bwd(gy, x)
        if gy.is_cuda and gy.dtype == torch.bfloat16 and gy.numel() % 8 == 0:
            ext = _ext.get(required=True)
            return ext.bf16_gelu_bwd(gy, x.contiguous())
        return gy * _gelu_grad(x)


def gelu(x: torch.Tensor) -> torch.Tensor:
    return _GeluFn.apply(x)
