"""Distributed losses.

Reference counterparts: ``DistributedRelativeLpLoss``
(/root/reference/dfno/loss.py:8-35) and DistDL's ``DistributedMSELoss``
(used at /root/reference/dfno/dfno.py:374).  Per-rank partial reductions are
sum-reduced to the root partition (RCCL reduce over xGMI; adjoint broadcast),
the root finishes the scalar math, and ZeroVolumeCorrectorFunction gives the
non-root ranks a graph-connected scalar so ``loss.backward()`` runs SPMD on
every rank.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..comm import SumReduce, ZeroVolumeCorrectorFunction
from ..partition import Partition, create_root_partition

__all__ = ["DistributedRelativeLpLoss", "DistributedMSELoss"]


class DistributedRelativeLpLoss(nn.Module):

    def __init__(self, P_x: Partition, p: int = 2):
        super().__init__()
        self.P_x = P_x
        self.p = p
        self.P_0 = create_root_partition(P_x)
        self.sr0 = SumReduce(P_x, self.P_0)
        self.sr1 = SumReduce(P_x, self.P_0)

    def forward(self, y_hat: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        batch_size = y_hat.shape[0] if y_hat.dim() > 0 and y_hat.numel() > 0 else 0
        if y_hat.dtype == torch.bfloat16:
            # bf16 sums over millions of elements lose the loss signal;
            # accumulate the norms in fp32 (grad flows back through the cast)
            y_hat = y_hat.float()
            y = y.float()
        if batch_size > 0:
            y_hat_flat = y_hat.reshape(batch_size, -1)
            y_flat = y.reshape(batch_size, -1)
            num = torch.sum(torch.pow(torch.abs(y_hat_flat - y_flat), self.p), dim=1)
            denom = torch.sum(torch.pow(torch.abs(y_flat), self.p), dim=1)
        else:
            num = y_hat.reshape(0)
            denom = y.reshape(0)

        num_global = self.sr0(num)
        denom_global = self.sr1(denom)

        if self.P_0.active:
            num_global = torch.pow(num_global, 1.0 / self.p)
            denom_global = torch.pow(denom_global, 1.0 / self.p)
            out = torch.mean(num_global / denom_global)
            return ZeroVolumeCorrectorFunction.apply(out)
        # non-root: zero-volume in -> graph-connected scalar 0 out
        return ZeroVolumeCorrectorFunction.apply(num_global + denom_global)


class DistributedMSELoss(nn.Module):
    """Global mean-squared error over a block-distributed tensor."""

    def __init__(self, P_x: Partition):
        super().__init__()
        self.P_x = P_x
        self.P_0 = create_root_partition(P_x)
        self.sr = SumReduce(P_x, self.P_0)

    def forward(self, y_hat: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        sq = torch.sum((y_hat - y) ** 2).reshape(1)
        n = torch.tensor([float(y.numel())], device=y.device, dtype=sq.dtype)
        packed = torch.cat([sq, n])
        total = self.sr(packed)
        if self.P_0.active:
            out = total[0] / total[1].clamp_min(1.0)
            return ZeroVolumeCorrectorFunction.apply(out)
        return ZeroVolumeCorrectorFunction.apply(total)
