"""DistributedFNOBlock: pencil-decomposed distributed N-D FFT + spectral conv.

Reference counterpart: /root/reference/dfno/dfno.py:67-291.  The algorithm is
kept exactly: transform dims split into leading
``n0 = ceil(n/2)`` and trailing ``n1 = floor(n/2)`` halves; partition ``P_m``
localizes the trailing dims (serial rfft/fft along them), ``P_y`` the leading
dims; four Repartitions per block rotate which dims are local; per-dim mode
truncation happens immediately after each dim's FFT so the R2/R3 exchanges
move only the truncated spectrum.

MI355X-native differences:
* Repartitions are RCCL grouped send/recv plans over xGMI with analytically
  precomputed global shapes (no lazy shape inference on the hot path).
* The corner-block spectral contraction is the fused HIP kernel
  (ops.spectral_conv) — no ``0*x.clone()`` materialization, no per-corner
  slicing copies.
* The residual ``gelu(y0 + y)`` epilogue is one fused kernel.
* ``irfft`` is given the explicit output length (fixes the odd-T bug class
  the reference has).
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np
import torch
import torch.nn as nn

from ..comm import Repartition
from ..partition import Partition, compute_distribution_info
from ..ops import spectral_conv, add_gelu, linear_res_gelu, rfft_trunc, fft_trunc, pad_ifft, pad_irfft
from ..timing import comm_region
from .linear import BroadcastedLinear

__all__ = ["DistributedFNOBlock"]


class DistributedFNOBlock(nn.Module):

    def __init__(self, P_x: Partition, in_shape, modes,
                 device=torch.device("cpu"), dtype=torch.float32):
        super().__init__()

        self.P_x = P_x
        self.in_shape = [int(s) for s in in_shape]
        self.width = self.in_shape[1]
        self.modes = [int(m) for m in modes]
        self.n = P_x.dim - 2
        self.device = device
        self.dtype = dtype
        self.dtype_complex = torch.complex64 if dtype == torch.float32 else torch.complex128

        # ---- FFT pencil partitions (reference dfno.py:82-97) -------------
        shape_m = P_x.shape.copy()
        shape_y = P_x.shape.copy()

        n0 = int(np.ceil(self.n / 2))
        n1 = int(np.floor(self.n / 2))
        shape_m[2 + n0:] = 1
        shape_m[2:2 + n1] *= P_x.shape[2 + n0:]
        shape_y[2:2 + n0] = 1
        shape_y[2 + n0:] *= P_x.shape[2:2 + n1]

        self.dim_m = list(range(2 + n0, P_x.dim))   # trailing dims (incl. time)
        self.dim_y = list(range(2, 2 + n0))         # leading dims

        self.P_m = P_x.create_cartesian_topology_partition([int(s) for s in shape_m])
        self.P_y = P_x.create_cartesian_topology_partition([int(s) for s in shape_y])

        # ---- per-dim mode truncation bookkeeping (dfno.py:104-111) -------
        self.restrict_prefixes: Dict[int, int] = {}
        self.restrict_suffixes: Dict[int, int] = {}
        for dim in [*self.dim_m, *self.dim_y]:
            mode = self.modes[dim - 2]
            self.restrict_prefixes[dim] = mode
            if dim != self.dim_m[-1]:
                self.restrict_suffixes[dim] = mode

        # global shape of the spectrum after the dim_m truncations only
        # (what R2/R3 move), and of the fully truncated spectrum (weights).
        self.trunc_m_shape = list(self.in_shape)
        for dim in self.dim_m:
            self.trunc_m_shape[dim] = self.restrict_prefixes[dim] + self.restrict_suffixes.get(dim, 0)
        fft_shape = list(self.trunc_m_shape)
        for dim in self.dim_y:
            fft_shape[dim] = self.restrict_prefixes[dim] + self.restrict_suffixes.get(dim, 0)
        self.fft_shape = fft_shape

        # ---- repartition plans with analytic global shapes ---------------
        self.R1 = Repartition(self.P_x, self.P_m, global_shape=self.in_shape)
        self.R2 = Repartition(self.P_m, self.P_y, global_shape=self.trunc_m_shape)
        self.R3 = Repartition(self.P_y, self.P_m, global_shape=self.trunc_m_shape)
        self.R4 = Repartition(self.P_m, self.P_x, global_shape=self.in_shape)

        # ---- frequency-sharded spectral weights (dfno.py:113-161) --------
        # Corners of the truncated-spectrum hypercube: 2^(n-1) corners (the
        # rfft time dim only has low modes); each rank owns the intersection
        # of each corner with its local P_y block.
        self.scale = 1 / (self.width * self.width)
        info = compute_distribution_info(self.P_y, fft_shape)

        self.weights = nn.ParameterList([])
        self.corner_bounds: List[List[Tuple[int, int]]] = []
        self.slices = []  # reference-compatible: local slice list per corner
        # bookkeeping for checkpoint/equivalence tooling:
        self.corner_ids: List[int] = []          # which global corner each weight is
        self.corner_shapes: List[List[int]] = [] # full (global) extents of each corner
        self.corner_local_in_corner: List[List[Tuple[int, int]]] = []  # shard within corner

        for i in range(2 ** (self.n - 1) if self.P_y.active else 0):
            s = bin(i)[2:].zfill(self.n)
            bounds = []
            gbounds = []   # corner box in global fft_shape coordinates
            for j, digit in enumerate(s):
                dim = self.P_y.dim - j - 1
                mode = self.modes[dim - 2]
                start = info["start"][dim]
                stop = info["stop"][dim]
                dim_size = fft_shape[dim]
                if digit == "0":
                    bounds.append((max(0, start) - start, min(mode, stop) - start))
                    gbounds.append((0, mode))
                else:
                    bounds.append((max(dim_size - mode, start) - start,
                                   min(dim_size, stop) - start))
                    gbounds.append((dim_size - mode, dim_size))
            bounds = list(reversed(bounds))
            gbounds = list(reversed(gbounds))
            if any(b - a <= 0 for a, b in bounds):
                continue
            w = nn.Parameter(
                self.scale * torch.rand(self.width, self.width,
                                        *[b - a for a, b in bounds],
                                        device=device, dtype=self.dtype_complex))
            self.weights.append(w)
            self.corner_bounds.append(bounds)
            self.slices.append([slice(None), slice(None)] + [slice(a, b) for a, b in bounds])
            self.corner_ids.append(i)
            self.corner_shapes.append([hi - lo for lo, hi in gbounds])
            # this rank's shard within the corner box (corner-local coords)
            loc = []
            for d, (glo, ghi) in zip(range(2, self.P_y.dim), gbounds):
                start, stop = info["start"][d], info["stop"][d]
                loc.append((max(start, glo) - glo, min(stop, ghi) - glo))
            self.corner_local_in_corner.append(loc)

        # linear pass-through (residual path), no bias (dfno.py:174)
        self.linear = BroadcastedLinear(self.P_x, self.width, self.width,
                                        bias=False, dim=1, device=device, dtype=dtype)

        self.dt_comm = 0.0

    # ---- mode truncation (reference dfno.py:178-239, bugs fixed) ---------
    def restrict(self, x: torch.Tensor, dim: int) -> torch.Tensor:
        """Keep the low (and, for non-rfft dims, high) frequency slabs."""
        if dim not in self.restrict_prefixes and dim not in self.restrict_suffixes:
            return x
        pieces = []
        sl = [slice(None)] * x.dim()
        if dim in self.restrict_prefixes:
            sl[dim] = slice(None, self.restrict_prefixes[dim])
            pieces.append(x[tuple(sl)])
        if dim in self.restrict_suffixes:
            sl[dim] = slice(-self.restrict_suffixes[dim], None)
            pieces.append(x[tuple(sl)])
        if len(pieces) == 1:
            return pieces[0]
        return torch.cat(pieces, dim=dim)

    def zeropad(self, y: torch.Tensor, dim: int, target_shape: List[int]) -> torch.Tensor:
        """Scatter the kept slabs back into a zero spectrum of target_shape."""
        if dim not in self.restrict_prefixes and dim not in self.restrict_suffixes:
            return y
        pad_shape = list(target_shape)
        pad_shape[dim] -= y.shape[dim]
        if pad_shape[dim] < 1:
            return y
        if any(s < 1 for d, s in enumerate(pad_shape) if d != dim):
            # empty local block: nothing to pad
            return y
        pieces = []
        sl = [slice(None)] * y.dim()
        if dim in self.restrict_prefixes:
            sl[dim] = slice(None, self.restrict_prefixes[dim])
            pieces.append(y[tuple(sl)])
        pieces.append(torch.zeros(pad_shape, dtype=y.dtype, layout=y.layout, device=y.device))
        if dim in self.restrict_suffixes:
            sl[dim] = slice(-self.restrict_suffixes[dim], None)
            pieces.append(y[tuple(sl)])
        return torch.cat(pieces, dim=dim)

    # ---- forward (reference dfno.py:241-291) ------------------------------
    # Each transform runs through the fused truncated-DFT ops (ops/fft.py):
    # transform + mode truncation / zero-padding + inverse scale in a single
    # strided-native kernel per dim (hand-written gfx950 DFT on GPU, a
    # torch.fft composition on CPU).
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        self.dt_comm = 0.0

        # The residual pass-through linear (reference computes y0 = linear(x)
        # up front, dfno.py:244) is deferred and fused into the epilogue as
        # gelu(W x_in + y) — no y0 tensor is materialized.  Its weight is
        # broadcast here so the collective order matches the reference.
        x_in = x
        with comm_region() as r:
            W_res = self.linear.W_bcast(self.linear.W)
            b_res = self.linear.b_bcast(self.linear.b)  # unused (bias=False);
        self.linear.dt_comm = r.host_dt                 # broadcast parity
        del b_res

        with comm_region() as r:
            x = self.R1(x)
        self.dt_comm += r.host_dt

        saved_last: Dict[int, int] = {}   # pre-truncation extent per dim
        outermost = self.dim_m[-1]
        if x.numel() > 0:
            n_t = x.shape[outermost]
            saved_last[outermost] = n_t // 2 + 1
            x = rfft_trunc(x, outermost, self.restrict_prefixes[outermost])
            for dim in reversed(self.dim_m[:-1]):
                saved_last[dim] = x.shape[dim]
                x = fft_trunc(x, dim, self.restrict_prefixes[dim],
                              self.restrict_suffixes.get(dim, 0))
        else:
            x = torch.empty(0, dtype=self.dtype_complex, device=x.device)

        with comm_region() as r:
            x = self.R2(x)
        self.dt_comm += r.host_dt

        if x.numel() > 0:
            for dim in reversed(self.dim_y):
                saved_last[dim] = x.shape[dim]
                x = fft_trunc(x, dim, self.restrict_prefixes[dim],
                              self.restrict_suffixes.get(dim, 0))

            y = spectral_conv(x, list(self.weights), self.corner_bounds, self.width)

            for dim in self.dim_y:
                y = pad_ifft(y, dim, saved_last[dim],
                             self.restrict_prefixes[dim],
                             self.restrict_suffixes.get(dim, 0))
        else:
            y = x

        with comm_region() as r:
            y = self.R3(y)
        self.dt_comm += r.host_dt

        if y.numel() > 0:
            for dim in self.dim_m[:-1]:
                y = pad_ifft(y, dim, saved_last[dim],
                             self.restrict_prefixes[dim],
                             self.restrict_suffixes.get(dim, 0))
            y = pad_irfft(y, outermost, saved_last[outermost],
                          self.in_shape[-1], self.restrict_prefixes[outermost])
        else:
            y = torch.empty(0, dtype=self.dtype, device=y.device)

        with comm_region() as r:
            y = self.R4(y)
        self.dt_comm += r.host_dt

        return linear_res_gelu(x_in, W_res, y)
