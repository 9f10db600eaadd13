"""Communication timing: host launch-side wall clock + CUDA-event device time.

The reference accumulates ``time.time()`` deltas around each collective
(/root/reference/dfno/dfno.py:51-60) — honest under host-blocking MPI, but on
RCCL the collectives are stream-enqueued, so a host timer measures launch
overhead only.  Every comm call site here is wrapped in :func:`comm_region`,
which always measures host wall time (kept as the reference-compatible
``dt_comm`` fields, explicitly *launch-side* on GPU) and, when event timing
is enabled, additionally brackets the region with CUDA events on the current
stream.  ``device_seconds()`` (after a sync) is then the honest device-side
communication time.
"""

from __future__ import annotations

import time

import torch

__all__ = ["comm_region", "enable_event_timing", "disable_event_timing",
           "device_seconds", "reset_events"]

_pairs = []          # (start_event, end_event) on the recording stream
_enabled = False


def enable_event_timing() -> None:
    """Start recording CUDA event pairs around every comm region."""
    global _enabled
    _pairs.clear()
    _enabled = True


def disable_event_timing() -> None:
    global _enabled
    _enabled = False
    _pairs.clear()


def reset_events() -> None:
    _pairs.clear()


def device_seconds() -> float:
    """Total device-side seconds across recorded comm regions.

    Synchronizes the device (call it OUTSIDE any timed section).
    """
    if not _pairs:
        return 0.0
    torch.cuda.synchronize()
    total_ms = 0.0
    for s, e in _pairs:
        total_ms += s.elapsed_time(e)
    return total_ms / 1e3


class _Region:
    __slots__ = ("t0", "host_dt", "_start")

    def __enter__(self):
        self.t0 = time.time()
        if _enabled and torch.cuda.is_available():
            self._start = torch.cuda.Event(enable_timing=True)
            self._start.record()
        else:
            self._start = None
        return self

    def __exit__(self, *exc):
        self.host_dt = time.time() - self.t0
        if self._start is not None:
            end = torch.cuda.Event(enable_timing=True)
            end.record()
            _pairs.append((self._start, end))
        return False


def comm_region() -> _Region:
    """Context manager timing one communication call.

    ``region.host_dt`` after exit = host wall time (launch-side on GPU).
    When event timing is enabled, the region is also bracketed by CUDA
    events summed by :func:`device_seconds`.
    """
    return _Region()
