"""Taylor-remainder gradient testing engine.

Semantics follow the reference's tests/gradient_test.py:40-132 (the
distributed replacement for ``torch.autograd.gradcheck``): for each named
parameter p, with loss f(p) = 0.5 ||net(x1) - net(x0)||^2,
verify |f(p0 + h dp) - f(p0)| = O(h) and
|f(p0 + h dp) - f(p0) - h <dp, g>| = O(h^2) as h halves, by fitting the
log-log slope.

Distributed-correct extension (the reference fudges this with an ad-hoc
``/P_x.size`` scaling of the O(h^2) fit, gradient_test.py:120 — see
SURVEY.md section 4): here the scalar loss, its perturbed values and the
directional gradient are all summed over the partition with an allreduce, so
the Taylor identity holds globally and the clean slopes 1.0 / 2.0 are
recovered at any world size.  Parameters with no gradient or zero volume on
this rank are locally inactive but still participate in the global test.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Generator, List, Optional, Tuple

import numpy as np
import torch
import torch.nn as nn

from .partition import Partition

__all__ = ["GradientTestResult", "gradient_test"]


@dataclass
class GradientTestResult:
    name: str
    active: bool
    converged: Tuple[bool, bool]
    convergence: Tuple[List[float], List[float]]
    steps: List[float]
    polyfit: Tuple[List[float], List[float]]

    def __str__(self):
        c0 = [f"{x:.2e}" for x in self.convergence[0]]
        c1 = [f"{x:.2e}" for x in self.convergence[1]]
        steps = [f"{x:.2e}" for x in self.steps]
        s = f"==== {self.name} ====\n"
        s += f"active: {self.active}\n"
        s += "converged:\n"
        s += f"\tO(h):   {self.converged[0]}\n"
        s += f"\tO(h^2): {self.converged[1]}\n"
        s += f"convergence:\n\tO(h)   err = {c0}\n\tO(h^2) err = {c1}\n"
        s += f"steps:\n\t h = {steps}\n"
        if self.active and len(self.polyfit[0]):
            s += f"polyfit:\n\tO(h)   slope = {self.polyfit[0][0]:.3f}\n"
            s += f"\tO(h^2) slope = {self.polyfit[1][0]:.3f}"
        else:
            s += "polyfit: N/A"
        return s


def _global_sum(v: float, P: Optional[Partition]) -> float:
    if P is None:
        return v
    return P.allreduce_scalar(v, op="sum")


def gradient_test(f: nn.Module,
                  input_shape: List[int],
                  max_iter: int = 10,
                  dtype: torch.dtype = torch.float64,
                  P: Optional[Partition] = None,
                  seed: int = 0) -> Generator[GradientTestResult, None, None]:
    """Yield a GradientTestResult per named parameter of ``f``.

    ``input_shape`` is this rank's LOCAL input shape.  Pass ``P`` (the model
    partition) to run the globally-summed distributed variant.
    """

    rank = max(P.rank, 0) if P is not None else 0

    def inner(p, pidx: int) -> GradientTestResult:
        # ``p`` may be None on ranks that do not hold this (sharded) param;
        # they still run every evaluation so the collectives stay matched.
        gen = torch.Generator().manual_seed(seed * 7919 + pidx * 131 + rank)

        def loss(x, y):
            if p is not None and p.grad is not None:
                p.grad.zero_()
            return 0.5 * torch.norm(f(x) - y) ** 2

        p_init = p.data if p is not None else None

        if p is not None:
            p0 = 1 + torch.rand(*p.shape, dtype=p.dtype, generator=gen)
            dp = 1e-3 * (1 + torch.rand(*p.shape, dtype=p.dtype, generator=gen))
            p.data = p0

        x0 = 1 + torch.rand(*input_shape, dtype=dtype, generator=gen)
        x1 = 1 + torch.rand(*input_shape, dtype=dtype, generator=gen)

        with torch.no_grad():
            y0 = f(x0)

        f0t = loss(x1, y0)
        f0t.backward()

        locally_active = p is not None
        gdx = 0.0
        if p is not None:
            try:
                g0 = p.grad.detach()
                if g0.nelement() == 0:
                    locally_active = False
                elif g0.is_complex():
                    # torch convention: grad is conj-Wirtinger; the
                    # first-order change under p -> p + h*dp is
                    # h * Re(<conj(g), dp>)
                    gdx = float(torch.vdot(g0.flatten(), dp.flatten().to(g0.dtype)).real)
                else:
                    gdx = float(torch.dot(dp.flatten().to(g0.dtype), g0.flatten()))
            except AttributeError:
                locally_active = False

        f0 = _global_sum(float(f0t.detach()), P)
        gdx = _global_sum(gdx, P)
        active = locally_active if P is None else (abs(gdx) > 0)

        err1, err2, hs = [], [], []
        h = 1.0
        for _ in range(max_iter):
            if p is not None:
                p.data = p0 + h * dp
            with torch.no_grad():
                fk = float(loss(x1, y0).detach())
            fk = _global_sum(fk, P)
            if active:
                err1.append(abs(fk - f0))
                err2.append(abs(fk - f0 - h * gdx))
            hs.append(h)
            h = h / 2

        p1, p2 = [], []
        c1, c2 = False, False
        if active and len(err1):
            # drop points that have hit fp64 roundoff (the Taylor remainder
            # floors at ~eps * f0, flattening the log-log tail)
            floor = 1e-11 * max(abs(f0), 1.0)

            def fit(errs):
                pts = [(h, e) for h, e in zip(hs, errs) if e > floor]
                if len(pts) < 4:
                    return None  # everything at machine precision: converged
                hh, ee = zip(*pts)
                return np.polyfit(np.log10(hh), np.log10(ee), 1)

            f1 = fit(err1)
            f2 = fit(err2)
            p1 = list(f1) if f1 is not None else []
            p2 = list(f2) if f2 is not None else []
            # O(h) passes for slope >= ~1 (a vanishing linear term makes the
            # first-order error superconverge at slope 2, which still
            # satisfies the O(h) bound); O(h^2) requires slope ~2.
            c1 = True if f1 is None else bool(0.9 <= f1[0] <= 2.2)
            c2 = True if f2 is None else bool(np.isclose(f2[0], 2.0, rtol=0.1))

        if p is not None:
            p.data = p_init
        return GradientTestResult("", active, (c1, c2), (err1, err2), hs, (list(p1), list(p2)))

    # Iterate a GLOBALLY agreed parameter-name list: with sharded parameter
    # lists (e.g. frequency-sharded spectral weights on a folded P_y) ranks
    # hold different subsets, and a per-rank loop would desynchronize the
    # collectives inside f.
    params = dict(f.named_parameters())
    names = list(params.keys())
    if P is not None and P.size > 1:
        import torch.distributed as dist
        from .partition import is_distributed
        if is_distributed():
            gathered: list = [None] * dist.get_world_size(P.group)
            dist.all_gather_object(gathered, names, group=P.group)
            seen = []
            for lst in gathered:
                for n in lst:
                    if n not in seen:
                        seen.append(n)
            names = seen

    for i, name in enumerate(names):
        gt = inner(params.get(name), i)
        gt.name = name
        yield gt
