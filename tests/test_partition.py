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


# ---------------------------------------------------------------------------
# packed-repartition descriptor property test (CPU emulation of the
# csrc/pack.hip index walk vs direct slicing; hypothesis-driven shapes)
# ---------------------------------------------------------------------------

from hypothesis import given, settings, strategies as st


def _emulate_pack(x_words, rec):
    """Replicate copy_boxes_kernel's mixed-radix walk for one descriptor."""
    flat_off, tens_off, numel, nd = rec[0], rec[1], rec[2], rec[3]
    mdims = rec[4:12]
    mstrs = rec[12:20]
    out = torch.empty(numel, dtype=x_words.dtype)
    for e in range(numel):
        rem, off = e, tens_off
        for k in range(7, 0, -1):
            if k < nd:
                idx = rem % mdims[k]
                rem //= mdims[k]
                off += idx * mstrs[k]
        off += rem * mstrs[0]
        out[e] = x_words[off]
    return out, flat_off


@settings(max_examples=25, deadline=None)
@given(st.data())
def test_box_record_matches_slicing(data):
    """_box_record's merged-dim descriptors address exactly the elements
    direct slicing produces, for random shapes/boxes/word widths."""
    from dfno_amd.comm import _box_record

    ndim = data.draw(st.integers(2, 5))
    shape = tuple(data.draw(st.integers(1, 6)) for _ in range(ndim))
    box = []
    for d in shape:
        a = data.draw(st.integers(0, d - 1))
        b = data.draw(st.integers(a + 1, d))
        box.append((a, b))
    wpe = data.draw(st.sampled_from([1, 2]))

    torch.manual_seed(0)
    x = torch.randn(*shape, 2)[..., :wpe].contiguous()   # [..., wpe] words
    x_words = x.reshape(-1)
    rec, n = _box_record(shape, box, wpe, flat_off=0)
    assert len(rec) == 20
    got, _ = _emulate_pack(x_words, rec)
    ref = x[tuple(slice(a, b) for a, b in box)].reshape(-1)
    assert n == ref.numel()
    assert torch.equal(got, ref)


@settings(max_examples=100, deadline=None)
@given(st.integers(0, 200), st.integers(1, 16))
def test_balanced_splits_tile_exactly(n, p):
    """The balanced split must cover [0, n) contiguously with p pieces
    whose sizes differ by at most one — every repartition plan, block
    bound and shard shape derives from this."""
    from dfno_amd.partition import _balanced_splits

    s = _balanced_splits(n, p)
    assert len(s) == p
    cur = 0
    sizes = []
    for a, b in s:
        assert a == cur and b >= a
        sizes.append(b - a)
        cur = b
    assert cur == n
    assert max(sizes) - min(sizes) <= 1
