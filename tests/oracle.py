"""Independent serial oracle implementing the reference's FNO math directly
with plain torch einsum/fft ops (mirrors /root/reference/dfno/dfno.py
semantics), used to validate dfno_amd's fused ops and bookkeeping.

Operates on a world_size-1 dfno_amd.DistributedFNONd instance, reading its
parameters, so outputs/grads are directly comparable.
"""

import torch
import torch.nn.functional as F

from dfno_amd.utils import alphabet


def oracle_linear(model_lin, x, activation=None):
    dim = model_lin.dim
    nd = x.dim()
    x_chars = alphabet(nd, as_array=True)
    y_chars = alphabet(nd, as_array=True)
    x_chars[dim] = "i"
    y_chars[dim] = "o"
    eqn = f"oi,{''.join(x_chars)}->{''.join(y_chars)}"
    y = torch.einsum(eqn, model_lin.W, x)
    if model_lin.bias:
        y = y + model_lin.b
    if activation == "gelu":
        y = F.gelu(y)
    return y


def oracle_block(block, x):
    y0 = oracle_linear(block.linear, x)

    saved_shapes = {}
    outermost = block.dim_m[-1]
    x = torch.fft.rfft(x, dim=outermost)
    saved_shapes[outermost] = list(x.shape)
    x = block.restrict(x, outermost)
    for dim in reversed(block.dim_m[:-1]):
        x = torch.fft.fft(x, dim=dim)
        saved_shapes[dim] = list(x.shape)
        x = block.restrict(x, dim)
    for dim in reversed(block.dim_y):
        x = torch.fft.fft(x, dim=dim)
        saved_shapes[dim] = list(x.shape)
        x = block.restrict(x, dim)

    # spectral corner einsum exactly as the reference writes it
    nd = x.dim()
    w_chars = alphabet(nd, as_array=True)
    x_chars = alphabet(nd, as_array=True)
    y_chars = alphabet(nd, as_array=True)
    w_chars[0] = "i"
    w_chars[1] = "o"
    x_chars[1] = "i"
    y_chars[1] = "o"
    eqn = f"{''.join(x_chars)},{''.join(w_chars)}->{''.join(y_chars)}"

    y = 0 * x.clone()
    for w, sl in zip(block.weights, block.slices):
        y[tuple(sl)] = torch.einsum(eqn, x[tuple(sl)], w)

    for dim in block.dim_y:
        y = block.zeropad(y, dim, saved_shapes[dim])
        y = torch.fft.ifft(y, dim=dim)
    for dim in block.dim_m[:-1]:
        y = block.zeropad(y, dim, saved_shapes[dim])
        y = torch.fft.ifft(y, dim=dim)
    y = block.zeropad(y, outermost, saved_shapes[outermost])
    y = torch.fft.irfft(y, n=block.in_shape[-1], dim=outermost)

    return F.gelu(y0 + y)


def oracle_fno(model, x):
    x = F.gelu(oracle_linear(model.linear1, x))
    x = F.gelu(oracle_linear(model.linear2, x))
    for block in model.blocks:
        x = oracle_block(block, x)
    x = F.gelu(oracle_linear(model.linear3, x))
    x = oracle_linear(model.linear4, x)
    return x
