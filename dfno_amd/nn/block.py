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

import os

from ..comm import (Repartition, _get_plan, repartition_complete,
                    repartition_issue)
from ..partition import Partition, compute_distribution_info
from ..ops import spectral_conv, add_gelu, linear_res_gelu, rfft_trunc, fft_trunc, pad_ifft, pad_irfft
from ..timing import comm_region
from .linear import BroadcastedLinear

__all__ = ["DistributedFNOBlock"]


class DistributedFNOBlock(nn.Module):

    def __init__(self, P_x: Partition, in_shape, modes,
                 device=torch.device("cpu"), dtype=torch.float32,
                 spectral_fp8: bool = False):
        super().__init__()

        self.P_x = P_x
        # fp8 (e4m3) storage for the corner weights in the contraction
        # (BASELINE.json config #5); masters stay complex64
        self.spectral_fp8 = spectral_fp8
        self.in_shape = [int(s) for s in in_shape]
        self.width = self.in_shape[1]
        self.modes = [int(m) for m in modes]
        self.n = P_x.dim - 2
        self.device = device
        self.dtype = dtype
        # bf16 models keep the spectral path in fp32/complex64 (the reference
        # semantic; transforms upcast at entry, downcast after the inverse)
        self.dtype_complex = (torch.complex128 if dtype == torch.float64
                              else torch.complex64)

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

    # ---- per-phase transform helpers (shared by both forward paths) ------
    def _fwd_m(self, x: torch.Tensor, saved: Dict[int, int]) -> torch.Tensor:
        """rfft + truncation along the trailing (P_m-local) dims."""
        if x.numel() == 0:
            # requires_grad in grad mode so the downstream comm ops keep a
            # rank-uniform backward graph (their adjoints must run on EVERY
            # rank, including ones whose local block is empty)
            return torch.empty(0, dtype=self.dtype_complex, device=x.device,
                               requires_grad=torch.is_grad_enabled())
        outermost = self.dim_m[-1]
        if len(self.dim_m) == 2:
            # fused (z,t) plane transform: both trailing dims in one kernel
            from ..ops.fft import zt_enabled, zt_fwd, zt_native_ok
            z_dim = self.dim_m[0]
            mzl = self.restrict_prefixes[z_dim]
            mzh = self.restrict_suffixes.get(z_dim, 0)
            mt = self.restrict_prefixes[outermost]
            if zt_enabled() and zt_native_ok(x, mzl, mzh, mt):
                saved[outermost] = x.shape[outermost] // 2 + 1
                saved[z_dim] = x.shape[z_dim]
                return zt_fwd(x, mzl, mzh, mt)
        if x.dtype == torch.bfloat16:
            from ..ops.fft import rfft_bf16_native_ok
            if not rfft_bf16_native_ok(x, outermost,
                                       self.restrict_prefixes[outermost]):
                # no bf16-IO kernel for this shape: boundary-cast to fp32
                # (the spectral path runs fp32/c64 either way)
                x = x.float()
        saved[outermost] = x.shape[outermost] // 2 + 1
        x = rfft_trunc(x, outermost, self.restrict_prefixes[outermost])
        for dim in reversed(self.dim_m[:-1]):
            saved[dim] = x.shape[dim]
            x = fft_trunc(x, dim, self.restrict_prefixes[dim],
                          self.restrict_suffixes.get(dim, 0))
        return x

    def _fwd_y(self, x: torch.Tensor, saved: Dict[int, int]) -> torch.Tensor:
        """fft + truncation along the leading (P_y-local) dims."""
        if x.numel() == 0:
            return x
        for dim in reversed(self.dim_y):
            saved[dim] = x.shape[dim]
            x = fft_trunc(x, dim, self.restrict_prefixes[dim],
                          self.restrict_suffixes.get(dim, 0))
        return x

    def _inv_y(self, y: torch.Tensor, saved: Dict[int, int]) -> torch.Tensor:
        if y.numel() == 0:
            return y
        for dim in self.dim_y:
            y = pad_ifft(y, dim, saved[dim],
                         self.restrict_prefixes[dim],
                         self.restrict_suffixes.get(dim, 0))
        return y

    def _inv_m(self, y: torch.Tensor, saved: Dict[int, int]) -> torch.Tensor:
        if y.numel() == 0:
            return torch.empty(0, dtype=self.dtype, device=y.device,
                               requires_grad=torch.is_grad_enabled())
        outermost = self.dim_m[-1]
        if len(self.dim_m) == 2 and y.is_cuda and y.dtype == torch.complex64:
            from ..ops.fft import zt_enabled, zt_inv
            z_dim = self.dim_m[0]
            mzl = self.restrict_prefixes[z_dim]
            mzh = self.restrict_suffixes.get(z_dim, 0)
            Z, T = saved[z_dim], self.in_shape[-1]
            if (zt_enabled() and Z <= 64 and T <= 64
                    and mzl + mzh <= min(48, Z)
                    and y.shape[-1] <= 32 and y.shape[-2] == mzl + mzh):
                return zt_inv(y, Z, T, mzl, mzh, out_dtype=self.dtype)
        for dim in self.dim_m[:-1]:
            y = pad_ifft(y, dim, saved[dim],
                         self.restrict_prefixes[dim],
                         self.restrict_suffixes.get(dim, 0))
        y = pad_irfft(y, outermost, saved[outermost],
                      self.in_shape[-1], self.restrict_prefixes[outermost],
                      out_dtype=self.dtype)
        # (bf16 models: the c2r kernel emits bf16 directly; the exchange in
        # R4 then moves bf16.  Anything else lands here as a no-op.)
        return y.to(self.dtype) if y.dtype != self.dtype else y

    # ---- pipelined-chunk policy ------------------------------------------
    def _pipeline_chunks(self) -> int:
        """Channel-chunk count for the overlapped pencil chain.

        All inputs are GLOBAL (partition geometry, width, env), so every
        rank takes the same path — required for matched collectives.
        0 disables.  Default: 2 chunks when any repartition is a real
        exchange, else 1 (no point pipelining identities).
        """
        if getattr(self, "_nch", None) is not None:
            return self._nch
        env = os.environ.get("DFNO_PIPELINE_CHUNKS")
        nch = int(env) if env else 2
        if nch > 1:
            any_exchange = not (
                _get_plan(self.P_x, self.P_m, tuple(self.in_shape)).is_identity
                and _get_plan(self.P_m, self.P_y, tuple(self.trunc_m_shape)).is_identity
                and _get_plan(self.P_y, self.P_m, tuple(self.trunc_m_shape)).is_identity
                and _get_plan(self.P_m, self.P_x, tuple(self.in_shape)).is_identity)
            if not any_exchange or self.width < nch:
                nch = 1
        self._nch = max(1, nch)
        return self._nch

    def _chunk_sizes(self, nch: int):
        w = self.width
        base, rem = divmod(w, nch)
        return [base + (1 if i < rem else 0) for i in range(nch)]

    @staticmethod
    def _with_c(shape, c):
        s = list(shape)
        s[1] = c
        return s

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
            W_res = self.linear.W_bcast(self.linear.W)  # residual weight only
        self.linear.dt_comm = r.host_dt                 # (bias=False)

        nch = self._pipeline_chunks()
        if nch > 1:
            y = self._forward_pipelined(x, nch)
        else:
            y, x_in = self._forward_seq(x)

        return linear_res_gelu(x_in, W_res, y)

    def _forward_seq(self, x: torch.Tensor):
        """Returns (chain output, epilogue input).  The epilogue input is
        either ``x`` itself or its StashGradFn alias when the residual
        input-gradient fuses into the rfft adjoint (ops/fft.py stash)."""
        from ..ops.fft import (StashGradFn, new_stash_key, rfft_trunc_stash,
                               stash_fusable)

        saved: Dict[int, int] = {}   # pre-truncation extent per dim
        x_orig = x
        with comm_region() as r:
            x = self.R1(x)
        self.dt_comm += r.host_dt

        x_epi = x_orig
        outermost = self.dim_m[-1]
        zt_stash = False
        if (x is x_orig and len(self.dim_m) == 2 and x.is_cuda
                and x.numel() > 0 and x.dtype == torch.float32
                and torch.is_grad_enabled() and x.requires_grad):
            from ..ops.fft import zt_enabled, zt_fwd, zt_native_ok
            z_dim = self.dim_m[0]
            mzl = self.restrict_prefixes[z_dim]
            mzh = self.restrict_suffixes.get(z_dim, 0)
            mt = self.restrict_prefixes[outermost]
            if zt_enabled() and zt_native_ok(x, mzl, mzh, mt):
                # fused (z,t) transform + residual-grad stash in its adjoint
                key = new_stash_key()
                saved[outermost] = x.shape[outermost] // 2 + 1
                saved[z_dim] = x.shape[z_dim]
                xm, tok = zt_fwd(x, mzl, mzh, mt, stash_key=key)
                x_epi = StashGradFn.apply(x_orig, tok, key)
                x = xm
                zt_stash = True
        if zt_stash:
            pass
        elif (x is x_orig
                and stash_fusable(x, outermost, self.restrict_prefixes[outermost])):
            # R1 is an identity here: the chain's input-grad producer is the
            # rfft adjoint — fold the epilogue's gradient into its writeback
            key = new_stash_key()
            saved[outermost] = x.shape[outermost] // 2 + 1
            xm, tok = rfft_trunc_stash(x, outermost,
                                       self.restrict_prefixes[outermost], key)
            for dim in reversed(self.dim_m[:-1]):
                saved[dim] = xm.shape[dim]
                xm = fft_trunc(xm, dim, self.restrict_prefixes[dim],
                               self.restrict_suffixes.get(dim, 0))
            x_epi = StashGradFn.apply(x_orig, tok, key)
            x = xm
        else:
            x = self._fwd_m(x, saved)

        with comm_region() as r:
            x = self.R2(x)
        self.dt_comm += r.host_dt

        if x.numel() > 0:
            x = self._fwd_y(x, saved)
            y = spectral_conv(x, list(self.weights), self.corner_bounds,
                              self.width, fp8=self.spectral_fp8)
            y = self._inv_y(y, saved)
        else:
            y = x

        with comm_region() as r:
            y = self.R3(y)
        self.dt_comm += r.host_dt

        y = self._inv_m(y, saved)

        with comm_region() as r:
            y = self.R4(y)
        self.dt_comm += r.host_dt
        return y, x_epi

    def _forward_pipelined(self, x: torch.Tensor, nch: int) -> torch.Tensor:
        """Channel-chunked software pipeline over the pencil chain.

        All chunk exchanges of a phase are ISSUED before any chunk's
        transforms run (comm/compute overlap, SURVEY.md K9 / VERDICT.md
        round-1 item 1): chunk c's transforms cover chunks c+1..'s transfers
        — on RCCL via the NCCL stream, on gloo via its background threads.
        Every rank takes identical chunk/collective order (the chunk policy
        is a function of global config only).
        """
        sizes = self._chunk_sizes(nch)
        saved: Dict[int, int] = {}

        xs = list(x.split(sizes, dim=1)) if x.numel() > 0 else [x] * nch

        # phase 1: R1 for every chunk up-front, then transform-as-received
        with comm_region() as r:
            h1 = [repartition_issue(self.R1, xc.contiguous(),
                                    self._with_c(self.in_shape, c))
                  for xc, c in zip(xs, sizes)]
        self.dt_comm += r.host_dt

        h2 = []
        for h, c in zip(h1, sizes):
            with comm_region() as r:
                xc = repartition_complete(h)
            self.dt_comm += r.host_dt
            xc = self._fwd_m(xc, saved)
            with comm_region() as r:
                h2.append(repartition_issue(self.R2, xc,
                                            self._with_c(self.trunc_m_shape, c)))
            self.dt_comm += r.host_dt

        zs = []
        for h in h2:
            with comm_region() as r:
                xc = repartition_complete(h)
            self.dt_comm += r.host_dt
            zs.append(self._fwd_y(xc, saved))

        if all(z.numel() == 0 for z in zs):
            x = zs[0]
        else:
            x = torch.cat(zs, dim=1)

        # spectral contraction mixes channels: needs the full spectrum
        if x.numel() > 0:
            y = spectral_conv(x, list(self.weights), self.corner_bounds,
                              self.width, fp8=self.spectral_fp8)
        else:
            y = x

        ys = list(y.split(sizes, dim=1)) if y.numel() > 0 else [y] * nch

        h3 = []
        for yc, c in zip(ys, sizes):
            yc = self._inv_y(yc.contiguous() if yc.numel() else yc, saved)
            with comm_region() as r:
                h3.append(repartition_issue(self.R3, yc,
                                            self._with_c(self.trunc_m_shape, c)))
            self.dt_comm += r.host_dt

        h4 = []
        for h, c in zip(h3, sizes):
            with comm_region() as r:
                yc = repartition_complete(h)
            self.dt_comm += r.host_dt
            yc = self._inv_m(yc, saved)
            with comm_region() as r:
                h4.append(repartition_issue(self.R4, yc,
                                            self._with_c(self.in_shape, c)))
            self.dt_comm += r.host_dt

        outs = []
        for h in h4:
            with comm_region() as r:
                outs.append(repartition_complete(h))
            self.dt_comm += r.host_dt

        if all(o.numel() == 0 for o in outs):
            return outs[0]
        return torch.cat(outs, dim=1)
