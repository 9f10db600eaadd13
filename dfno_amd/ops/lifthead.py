"""Fused FNO lift head op (see csrc/lift_head.hip).

Applies linear1 (time lift, T_in == 1) + GELU + linear2 (channel lift) +
GELU in one kernel.  CPU / unsupported shapes use the composed path in the
model instead.
"""

from __future__ import annotations

import torch

from .. import _ext

__all__ = ["lift_head", "lift_head_supported"]


def lift_head_supported(x, t_in, t_out, c_in, width) -> bool:
    return (x.is_cuda
            and x.dtype in (torch.float32, torch.float64, torch.bfloat16)
            and t_in == 1 and t_out <= 32 and c_in <= 4 and width <= 24)


class _LiftHeadFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x3, W1, b1, W2, b2):
        ext = _ext.get(required=True)
        out = ext.lift_head_fwd(x3, W1.contiguous(), b1.contiguous(),
                                W2.contiguous(), b2.contiguous())
        ctx.save_for_backward(x3, W1, b1, W2, b2)
        return out

    @staticmethod
    def backward(ctx, gy):
        x3, W1, b1, W2, b2 = ctx.saved_tensors
        ext = _ext.get(required=True)
        gx, gW1, gb1, gW2, gb2 = ext.lift_head_bwd(
            gy.contiguous(), x3, W1.contiguous(), b1.contiguous(),
            W2.contiguous(), b2.contiguous())
        # bf16 activations: weight grads come back fp32-accumulated
        return (gx, gW1.to(W1.dtype), gb1.to(b1.dtype), gW2.to(W2.dtype),
                gb2.to(b2.dtype))


def lift_head(x, W1, b1, W2, b2):
    """x: [B, C, *sp, 1] -> [B, width, *sp, T_out], fully fused."""
    B, C = x.shape[0], x.shape[1]
    sp = x.shape[2:-1]
    S = 1
    for d in sp:
        S *= d
    x3 = x.reshape(B, C, S).contiguous()
    out = _LiftHeadFn.apply(x3, W1, b1.reshape(-1), W2, b2.reshape(-1))
    return out.reshape(B, W2.shape[0], *sp, W1.shape[0])
