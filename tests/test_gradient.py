"""Taylor-remainder gradient tests (reference: tests/gradient_test*.py).

Tier structure mirrors the reference's drivers:
sequential torch control -> broadcast-linear unit -> full model, serial and
on a 4-rank CPU partition in fp64.
"""

import pytest
import torch
import torch.nn as nn

from dist_utils import run_dist

import dfno_amd as dfno
from dfno_amd.testing import gradient_test


def test_engine_on_sequential_linear():
    # control: plain torch module validates the engine itself
    torch.manual_seed(0)
    f = nn.Sequential(nn.Linear(6, 8, dtype=torch.float64),
                      nn.Tanh(),
                      nn.Linear(8, 4, dtype=torch.float64))
    results = list(gradient_test(f, [3, 6]))
    assert len(results) == 4
    for r in results:
        assert r.active
        assert r.converged[0], f"{r.name}: O(h) failed\n{r}"
        assert r.converged[1], f"{r.name}: O(h^2) failed\n{r}"


def test_broadcasted_linear_serial():
    torch.manual_seed(1)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1))
    f = dfno.BroadcastedLinear(P_x, 5, 7, dim=1, dtype=torch.float64)
    for r in gradient_test(f, [2, 5, 9]):
        assert r.active, r.name
        assert r.converged[0] and r.converged[1], f"{r.name}\n{r}"


def test_fno_serial_fp64():
    torch.manual_seed(2)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1))
    f = dfno.DistributedFNONd(P_x, [2, 2, 8, 8, 3], 4, 6, (3, 3, 2),
                              num_blocks=1, dtype=torch.float64)
    checked = 0
    for r in gradient_test(f, [2, 2, 8, 8, 3], max_iter=8):
        if not r.active:
            continue  # unused params (bn, unused bias) have no gradient
        checked += 1
        assert r.converged[0], f"{r.name}: O(h)\n{r}"
        assert r.converged[1], f"{r.name}: O(h^2)\n{r}"
    assert checked >= 10


def _fno_dist_gradient_body(rank, world):
    import dfno_amd as dfno
    from dfno_amd.partition import Partition, compute_distribution_info
    from dfno_amd.testing import gradient_test

    torch.manual_seed(3 + rank)
    P_x = Partition(tuple(range(world)), (1, 1, 2, 2, 1))
    gshape = [2, 2, 8, 8, 3]
    f = dfno.DistributedFNONd(P_x, gshape, 4, 6, (3, 3, 2),
                              num_blocks=1, dtype=torch.float64)
    local_shape = compute_distribution_info(P_x, gshape)["shape"]

    checked = 0
    for r in gradient_test(f, local_shape, max_iter=8, P=P_x):
        if not r.active:
            continue
        checked += 1
        assert r.converged[0], f"rank {rank} {r.name}: O(h)\n{r}"
        assert r.converged[1], f"rank {rank} {r.name}: O(h^2)\n{r}"
    assert checked >= 5, f"only {checked} active params on rank {rank}"


def test_fno_distributed_gradient_4rank():
    # the reference's gradient_test_dfno.py analog: 4-rank CPU partition,
    # small shapes, fp64 — with globally-summed Taylor quantities
    run_dist(_fno_dist_gradient_body, 4)
