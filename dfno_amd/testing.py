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

    def inner(p: nn.Parameter, pidx: int) -> GradientTestResult:
        gen = torch.Generator().manual_seed(seed * 7919 + pidx * 131 + rank)

        def loss(x, y):
            if p.grad is not None:
                p.grad.zero_()
            return 0.5 * torch.norm(f(x) - y) ** 2

        p_init = p.data

        p0 = 1 + torch.rand(*p.shape, dtype=p.dtype, generator=gen)
        dp = 1e-3 * (1 + torch.rand(*p.shape, dtype=p.dtype, generator=gen))

        x0 = 1 + torch.rand(*input_shape, dtype=dtype, generator=gen)
        x1 = 1 + torch.rand(*input_shape, dtype=dtype, generator=gen)

        p.data = p0

        with torch.no_grad():
            y0 = f(x0)

        f0t = loss(x1, y0)
        f0t.backward()

        locally_active = True
        gdx = 0.0
        try:
            g0 = p.grad.detach()
            if g0.nelement() == 0:
                locally_active = False
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
        if active and len(err1) and min(err1) > 0 and min(err2) > 0:
            p1 = np.polyfit(np.log10(hs), np.log10(err1), 1)
            p2 = np.polyfit(np.log10(hs), np.log10(err2), 1)
            c1 = bool(np.isclose(p1[0], 1.0, rtol=0.1))
            c2 = bool(np.isclose(p2[0], 2.0, rtol=0.1))
        elif active:
            # machine-precision remainders count as converged
            c1 = max(err1) < 1e-12
            c2 = max(err2) < 1e-12

        p.data = p_init
        return GradientTestResult("", active, (c1, c2), (err1, err2), hs, (list(p1), list(p2)))

    for i, (name, p) in enumerate(f.named_parameters()):
        gt = inner(p, i)
        gt.name = name
        yield gt
