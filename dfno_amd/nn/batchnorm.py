"""DistributedBatchNorm over a block-distributed feature tensor.

Reference counterpart: DistDL's ``dnn.DistributedBatchNorm`` as constructed
by the model (/root/reference/dfno/dfno.py:325-326 — note the reference
comments it out of ``forward``, dfno.py:340,346; it exists here for API and
checkpoint-key parity and as a usable module).

Statistics are computed over (batch, *spatial, time) with a sum allreduce
over the partition (RCCL allreduce over xGMI); gamma/beta are ordinary local
parameters replicated on every rank (feature dim is rarely partitioned; if it
is, this module raises).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..comm import AllReduceSum
from ..partition import Partition

__all__ = ["DistributedBatchNorm"]


class DistributedBatchNorm(nn.Module):

    def __init__(self, P_x: Partition, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, affine: bool = True):
        super().__init__()
        self.P_x = P_x
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.affine = affine
        if int(P_x.shape[1]) != 1:
            raise NotImplementedError("feature-dim partitioning is not supported")
        if affine:
            self.weight = nn.Parameter(torch.ones(num_features))
            self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))
        self._ar = AllReduceSum(P_x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # Zero-volume ranks (1-D empty placeholder) still participate in the
        # packed allreduce with zeroed sums so the partition never desyncs.
        zero_volume = x.numel() == 0
        C = self.num_features if zero_volume else x.shape[1]
        red_dims = [d for d in range(x.dim()) if d != 1]
        if self.training:
            if zero_volume:
                cnt = torch.zeros(1, device=x.device, dtype=x.dtype)
                s = torch.zeros(C, device=x.device, dtype=x.dtype)
                ss = torch.zeros(C, device=x.device, dtype=x.dtype)
            else:
                cnt = torch.tensor([x.numel() / max(C, 1)], device=x.device, dtype=x.dtype)
                s = torch.sum(x, dim=red_dims)
                ss = torch.sum(x * x, dim=red_dims)
            packed = torch.cat([s, ss, cnt])
            packed = self._ar(packed)
            n = packed[-1].clamp_min(1.0)
            mean = packed[:C] / n
            var = packed[C:2 * C] / n - mean * mean
            with torch.no_grad():
                self.num_batches_tracked += 1
                m = self.momentum
                self.running_mean.mul_(1 - m).add_(m * mean.detach().to(self.running_mean.dtype))
                unbiased = var.detach() * (n / (n - 1.0).clamp_min(1.0))
                self.running_var.mul_(1 - m).add_(m * unbiased.to(self.running_var.dtype))
        else:
            mean = self.running_mean.to(x.dtype).to(x.device)
            var = self.running_var.to(x.dtype).to(x.device)

        if zero_volume:
            return x

        shape = [1, C] + [1] * (x.dim() - 2)
        xh = (x - mean.reshape(shape)) / torch.sqrt(var.reshape(shape) + self.eps)
        if self.affine:
            xh = xh * self.weight.reshape(shape).to(x.dtype) + self.bias.reshape(shape).to(x.dtype)
        return xh
