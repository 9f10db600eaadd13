"""Rank-failure detection via store heartbeats (docs/ROADMAP.md item 8).

The reference has no failure handling beyond the launcher's teardown: a
dead rank leaves the others blocked inside the next collective until the
job-level timeout.  ``HeartbeatMonitor`` gives every rank a cheap way to
notice a peer's death FIRST and fail fast with a diagnosable error:

- each rank runs a daemon thread that bumps a per-rank counter key in
  the process group's store (TCPStore or FileStore) every ``interval``
  seconds;
- ``check()`` (call it between steps) compares every peer's counter
  against the last value seen; a
  peer whose counter has not advanced for ``timeout`` seconds is reported
  dead and a ``RankFailure`` is raised locally — BEFORE the next
  collective deadlocks on it.

This is detection, not elasticity: the intended reaction is a clean abort
(and a restart from the checkpoint/resume path in training/two_phase).
Uses only ``torch.distributed``'s store API — no extra connections, works
under gloo and RCCL alike.

Usage:
    hb = HeartbeatMonitor(interval=2.0, timeout=30.0)
    hb.start()
    for step in ...:
        hb.check()        # raises RankFailure naming the dead ranks
        ...
    hb.stop()
"""

from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional

import torch.distributed as dist

__all__ = ["HeartbeatMonitor", "RankFailure"]


class RankFailure(RuntimeError):
    """A peer's heartbeat stalled past the timeout."""

    def __init__(self, dead_ranks: List[int], stalled_s: float):
        self.dead_ranks = dead_ranks
        super().__init__(
            f"rank(s) {dead_ranks} missed heartbeats for >= {stalled_s:.1f}s "
            f"(presumed dead); aborting before the next collective hangs")


def _default_store():
    # the process group's store; works for both init methods
    return dist.distributed_c10d._get_default_store()


class HeartbeatMonitor:
    def __init__(self, interval: float = 2.0, timeout: float = 30.0,
                 store=None, rank: Optional[int] = None,
                 world_size: Optional[int] = None, prefix: str = "dfno_hb"):
        if not dist.is_initialized() and (store is None or rank is None
                                          or world_size is None):
            raise RuntimeError("HeartbeatMonitor needs an initialized "
                               "process group (or explicit store/rank/world)")
        self.interval = float(interval)
        self.timeout = float(timeout)
        self.store = store if store is not None else _default_store()
        self.rank = rank if rank is not None else dist.get_rank()
        self.world = (world_size if world_size is not None
                      else dist.get_world_size())
        self.prefix = prefix
        self._beat = 0
        self._seen: Dict[int, int] = {}          # peer -> last counter
        self._seen_at: Dict[int, float] = {}     # peer -> time of last change
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()

    def _key(self, r: int) -> str:
        return f"{self.prefix}/{r}"

    # -- producer side ------------------------------------------------------
    def _pump(self):
        while not self._stop.wait(self.interval):
            self._beat += 1
            self.store.set(self._key(self.rank), str(self._beat))

    def start(self) -> "HeartbeatMonitor":
        self.store.set(self._key(self.rank), "0")
        now = time.monotonic()
        for r in range(self.world):
            self._seen[r] = -1
            self._seen_at[r] = now
        self._thread = threading.Thread(target=self._pump, daemon=True,
                                        name="dfno-heartbeat")
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2 * self.interval)
            self._thread = None

    # -- consumer side ------------------------------------------------------
    def check(self) -> None:
        """Raise RankFailure if any peer's counter stalled past timeout."""
        now = time.monotonic()
        dead: List[int] = []
        with self._lock:
            for r in range(self.world):
                if r == self.rank:
                    continue
                key = self._key(r)
                try:
                    # non-blocking existence probe where the store supports
                    # it (FileStore.get would otherwise BLOCK on a missing
                    # key until the store timeout)
                    if hasattr(self.store, "check") and not self.store.check([key]):
                        continue    # not started yet; timeout still applies
                    cur = int(self.store.get(key))
                except Exception:
                    cur = self._seen.get(r, -1)
                if cur != self._seen.get(r, -1):
                    self._seen[r] = cur
                    self._seen_at[r] = now
                elif now - self._seen_at[r] >= self.timeout:
                    dead.append(r)
        if dead:
            raise RankFailure(dead, self.timeout)

    def __enter__(self) -> "HeartbeatMonitor":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
