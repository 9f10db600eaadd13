"""Unit tests for partition math and balanced decomposition (serial)."""

import numpy as np
import pytest
import torch

from dfno_amd.partition import (
    Partition,
    _balanced_splits,
    block_bounds,
    compute_distribution_info,
    compute_subtensor_shapes_balanced,
    create_root_partition,
    create_standard_partitions,
)


def test_balanced_splits():
    assert _balanced_splits(10, 4) == [(0, 3), (3, 6), (6, 8), (8, 10)]
    assert _balanced_splits(8, 4) == [(0, 2), (2, 4), (4, 6), (6, 8)]
    assert _balanced_splits(3, 4) == [(0, 1), (1, 2), (2, 3), (3, 3)]
    assert _balanced_splits(0, 2) == [(0, 0), (0, 0)]


def test_subtensor_shapes():
    shapes = compute_subtensor_shapes_balanced([6, 7], (2, 3))
    assert shapes.shape == (2, 3, 2)
    # dim0: 3+3, dim1: 3+2+2
    assert list(shapes[0, 0]) == [3, 3]
    assert list(shapes[1, 2]) == [3, 2]
    total = 0
    for i in range(2):
        for j in range(3):
            total += shapes[i, j, 0] * shapes[i, j, 1]
    assert total == 42


def test_partition_serial():
    P = Partition((0,), (1, 1, 1))
    assert P.active and P.rank == 0
    assert tuple(P.index) == (0, 0, 0)
    info = compute_distribution_info(P, [4, 5, 6])
    assert info["shape"] == [4, 5, 6]
    assert info["start"] == [0, 0, 0]


def test_block_bounds_cover():
    P = Partition((0,), (1,))  # serial world; bounds computed for any rank of a virtual partition

    class FakeP:
        shape = np.array([2, 3])

        @staticmethod
        def rank_to_index(r):
            return tuple(int(i) for i in np.unravel_index(r, (2, 3)))

    g = [5, 7]
    seen = np.zeros(g, dtype=int)
    for r in range(6):
        b = block_bounds(FakeP, g, r)
        seen[b[0][0]:b[0][1], b[1][0]:b[1][1]] += 1
    assert (seen == 1).all()


def test_standard_partitions_serial():
    P_world, P_x, P_root = create_standard_partitions((1, 1, 1, 1))
    assert P_x.size == 1 and P_x.active
    assert P_root.size == 1 and P_root.active


def test_oversubscribed_raises():
    with pytest.raises(ValueError):
        create_standard_partitions((1, 2, 2))
