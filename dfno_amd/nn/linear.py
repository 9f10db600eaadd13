"""BroadcastedLinear: root-stored weights, broadcast each forward.

Reference counterpart: /root/reference/dfno/dfno.py:17-65.  Semantics kept:
weights (out x in) + bias live only on the root rank of ``P_x`` (zero-volume
placeholders elsewhere, so per-rank checkpoints shard identically to the
reference); both are re-broadcast every forward through an autograd Broadcast
whose adjoint sum-reduces the gradients back to root.  The contraction runs
through the fused MI355X pointwise kernel (ops.linear_nd) instead of a bare
einsum, with optional fused GELU.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

from ..comm import Broadcast
from ..partition import Partition, create_root_partition, zero_volume_tensor
from ..ops import linear_nd
from ..timing import comm_region

__all__ = ["BroadcastedLinear"]


class BroadcastedLinear(nn.Module):

    def __init__(self, P_x: Partition, in_features: int, out_features: int,
                 dim: int = -1, bias: bool = True,
                 device=torch.device("cpu"), dtype=torch.float32):
        super().__init__()

        self.P_x = P_x
        self.in_features = in_features
        self.out_features = out_features
        self.scale = 1 / np.sqrt(in_features * out_features)
        self.bias = bias
        self.dim = dim

        # bias stored broadcast-shaped, as the reference does (dfno.py:29-30)
        self.b_shape = [1] * P_x.dim
        self.b_shape[dim] = out_features

        self.P_root = create_root_partition(P_x)
        if self.P_root.active:
            self.W = nn.Parameter(torch.empty(out_features, in_features, device=device, dtype=dtype))
            self.b = nn.Parameter(torch.zeros(*self.b_shape, device=device, dtype=dtype))
            torch.nn.init.kaiming_uniform_(self.W, a=np.sqrt(5))
        else:
            self.W = nn.Parameter(zero_volume_tensor(device=device))
            self.b = nn.Parameter(zero_volume_tensor(device=device))

        self.W_bcast = Broadcast(self.P_root, P_x)
        self.b_bcast = Broadcast(self.P_root, P_x)

        self.dt_comm = 0.0

    def forward(self, x: torch.Tensor, activation: str = None) -> torch.Tensor:
        self.dt_comm = 0.0

        with comm_region() as r:
            W = self.W_bcast(self.W)
            # bias-less linears keep the parameter (state-dict parity) but
            # never ship it: a dead broadcast would still run its adjoint
            # reduce through the comm ordering chain
            b = self.b_bcast(self.b) if self.bias else None
        self.dt_comm += r.host_dt

        return linear_nd(x, W, b, self.dim, activation)
