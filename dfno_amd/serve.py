"""hipGraph-captured inference path (docs/ROADMAP.md item 7).

Small per-GPU grids (the Navier-Stokes 2D+time config, or the flagship
sharded 8 ways) leave the eval forward launch-bound: ~120 kernel launches
whose device time is a few ms while each launch costs host-side microseconds.
``GraphedEval`` records the whole eval forward once into a HIP graph
(``torch.cuda.CUDAGraph`` is hipGraph on ROCm) and replays it with ONE
launch per request.

Scope: single-process inference with fixed input shape/dtype (the serving
case).  Training steps are not captured — the autograd graph and the
collectives' dynamic plan execution are not capture-safe.  Capture runs on
a side stream per the CUDAGraph contract; the model must be on CUDA and in
eval mode (enforced here).

Usage:
    ge = GraphedEval(model, example_input)    # captures once
    y = ge(x)                                 # one graph launch
"""

from __future__ import annotations

import torch

__all__ = ["GraphedEval"]


class GraphedEval:
    """Replayable hipGraph capture of ``model(x)`` for a fixed shape.

    The input is copied into a static capture buffer and the returned
    tensor is a fresh copy of the static output (safe to hold across
    subsequent calls).
    """

    def __init__(self, model: torch.nn.Module, example: torch.Tensor,
                 warmup: int = 2):
        if not example.is_cuda:
            raise ValueError("GraphedEval: CUDA input required")
        self.model = model.eval()
        self._static_in = example.detach().clone()

        # warmup on a side stream (allocator + one-time lazy init must not
        # happen inside capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(max(warmup, 1)):
                out = self.model(self._static_in)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        self._graph = torch.cuda.CUDAGraph()
        with torch.no_grad(), torch.cuda.graph(self._graph):
            self._static_out = self.model(self._static_in)

    @torch.no_grad()
    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if x.shape != self._static_in.shape or x.dtype != self._static_in.dtype:
            raise ValueError(
                f"GraphedEval: expected {tuple(self._static_in.shape)} "
                f"{self._static_in.dtype}, got {tuple(x.shape)} {x.dtype}")
        self._static_in.copy_(x)
        self._graph.replay()
        return self._static_out.clone()
