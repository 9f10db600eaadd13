"""Store-heartbeat rank-failure detection (dfno_amd/health.py)."""

import time

import pytest
import torch.distributed as dist

from dist_utils import run_dist


def _healthy(rank, world):
    from dfno_amd.health import HeartbeatMonitor

    with HeartbeatMonitor(interval=0.2, timeout=10.0) as hb:
        dist.barrier()
        t0 = time.monotonic()
        while time.monotonic() - t0 < 1.5:
            hb.check()          # must never raise while everyone beats
            time.sleep(0.1)
        dist.barrier()


def test_heartbeat_healthy():
    run_dist(_healthy, 2)


def _one_dies(rank, world):
    from dfno_amd.health import HeartbeatMonitor, RankFailure

    hb = HeartbeatMonitor(interval=0.2, timeout=2.0).start()
    dist.barrier()
    if rank == 1:
        hb.stop()               # simulated death: heartbeats stop
        time.sleep(4.0)         # stay alive so gloo teardown stays clean
        return
    t0 = time.monotonic()
    with pytest.raises(RankFailure) as exc:
        while time.monotonic() - t0 < 20.0:
            hb.check()
            time.sleep(0.1)
    assert exc.value.dead_ranks == [1]
    hb.stop()


def test_heartbeat_detects_dead_rank():
    run_dist(_one_dies, 2)
