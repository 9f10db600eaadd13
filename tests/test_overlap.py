"""Comm/compute overlap of the split-phase repartition.

Two checks:
* simulated-latency hiding: with an artificial per-exchange transport delay,
  the issue-all-then-complete pipeline finishes in ~(T + k*C) wall versus the
  sequential ~k*(T + C) — the overlap mechanism measurably hides comm under
  compute on gloo's background-progress transport (the RCCL analogue is the
  NCCL-stream overlap on GPU).
* the channel-chunked block pipeline matches the sequential path bitwise for
  odd chunk splits (width not divisible by the chunk count).
"""

import os
import time

import numpy as np
import pytest
import torch

from dist_utils import run_dist


def _overlap_body(rank, world, T, C, k):
    import torch.distributed as dist
    import dfno_amd as dfno
    from dfno_amd.partition import Partition, compute_distribution_info
    from dfno_amd.comm import repartition_issue, repartition_complete

    # wrap the transport with a completion delay of T seconds after ISSUE:
    # models a slow link whose transfer progresses in the background
    real_batch = dist.batch_isend_irecv

    class _SlowReq:
        def __init__(self, req, deadline):
            self.req = req
            self.deadline = deadline

        def wait(self):
            self.req.wait()
            now = time.monotonic()
            if now < self.deadline:
                time.sleep(self.deadline - now)

    def slow_batch(ops):
        deadline = time.monotonic() + T
        return [_SlowReq(r, deadline) for r in real_batch(ops)]

    dist.batch_isend_irecv = slow_batch
    try:
        P_a = Partition((0, 1), (1, 2))
        P_b = Partition((0, 1), (2, 1))
        gshape = [8, 2 * world]
        R = dfno.Repartition(P_a, P_b, global_shape=gshape)
        info = compute_distribution_info(P_a, gshape)
        xs = [torch.randn(*info["shape"]) for _ in range(k)]

        def compute(y):
            time.sleep(C)          # compute stand-in
            return y * 2.0

        # sequential: exchange chunk -> compute chunk
        dist.barrier()
        t0 = time.monotonic()
        seq = [compute(R(x)) for x in xs]
        t_seq = time.monotonic() - t0

        # pipelined: issue ALL chunk exchanges, then complete+compute in order
        dist.barrier()
        t0 = time.monotonic()
        hs = [repartition_issue(R, x, gshape) for x in xs]
        pip = [compute(repartition_complete(h)) for h in hs]
        t_pip = time.monotonic() - t0

        for a, b in zip(seq, pip):
            assert torch.equal(a, b)
        # ideal: t_seq ~ k*(T+C), t_pip ~ T + k*C; generous margin for CI
        assert t_pip < 0.8 * t_seq, f"no overlap: pipelined {t_pip:.3f}s vs sequential {t_seq:.3f}s"
    finally:
        dist.batch_isend_irecv = real_batch


def test_pipeline_hides_simulated_latency():
    run_dist(_overlap_body, 2, 0.12, 0.06, 4)


def _odd_chunks_body(rank, world, nch):
    os.environ["DFNO_PIPELINE_CHUNKS"] = str(nch)
    import dfno_amd as dfno
    from dfno_amd.partition import Partition, compute_distribution_info

    in_shape = [1, 2, 8, 8, 6, 1]
    out_t = 8
    width = 6   # not divisible by nch=4 -> uneven chunk sizes
    modes = (3, 3, 2, 2)
    pshape = (1, 1, 2, 1, 2, 1)   # z-partitioned: real R1/R4 exchanges

    torch.manual_seed(3 + rank)
    P_x = Partition(tuple(range(world)), pshape)
    model = dfno.DistributedFNONd(P_x, in_shape, out_t, width, modes,
                                  num_blocks=1, dtype=torch.float64)

    gen = torch.Generator().manual_seed(7)
    gx = torch.rand(*in_shape, generator=gen, dtype=torch.float64)
    info = compute_distribution_info(P_x, in_shape)
    x_local = gx[info["slice"]].clone()

    y_pip = model(x_local)

    # same model, pipeline disabled
    for blk in model.blocks:
        blk._nch = 1
    y_seq = model(x_local)

    assert torch.allclose(y_pip, y_seq, rtol=1e-12, atol=1e-12)

    # gradients too
    for blk in model.blocks:
        blk._nch = None
    os.environ["DFNO_PIPELINE_CHUNKS"] = str(nch)
    loss = model(x_local).square().sum()
    from dfno_amd.comm import AllReduceSum
    loss = AllReduceSum(P_x)(loss.reshape(1)).sum()
    loss.backward()
    g_pip = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}

    model.zero_grad()
    for blk in model.blocks:
        blk._nch = 1
    loss = model(x_local).square().sum()
    loss = AllReduceSum(P_x)(loss.reshape(1)).sum()
    loss.backward()
    for n, p in model.named_parameters():
        if p.grad is None:
            assert n not in g_pip
            continue
        assert torch.allclose(g_pip[n], p.grad, rtol=1e-12, atol=1e-12), n


def test_pipelined_chunks_match_sequential_odd_split():
    run_dist(_odd_chunks_body, 4, 4)
