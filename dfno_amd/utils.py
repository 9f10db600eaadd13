"""Utility layer: env/device selection, normalisation, GPU memory profiling.

MI355X-native counterpart of /root/reference/dfno/utils.py.  Differences by
design: device binding is ``torch.cuda.set_device`` (HIP) — no CuPy context is
needed because RCCL communicators are created per device —
and GPU memory polling uses ``torch.cuda.memory_stats`` / ``rocm-smi``
instead of ``nvidia-smi``.
"""

from __future__ import annotations

import os
import subprocess as sp
import time
from contextlib import nullcontext
from typing import List

import numpy as np
import torch

__all__ = [
    "alphabet",
    "generate_batch_indices",
    "get_env",
    "get_gpu_memory",
    "profile_gpu_memory",
    "unit_guassian_normalize",
    "unit_gaussian_normalize",
    "unit_gaussian_denormalize",
]


def alphabet(n: int, as_array: bool = False):
    """einsum subscript generator 'a', 'b', ... (reference utils.py:85-88)."""
    array = [chr(i + 97) for i in range(n)]
    if as_array:
        return array
    return "".join(array)


def get_env(P, num_gpus: int = None):
    """Device selection for rank ``P.rank``.

    Returns ``(use_cuda, cuda_aware, device_ordinal, device, ctx)`` with the
    same tuple shape as the reference (utils.py:42-55) so scripts port
    unchanged; ``ctx`` is always a nullcontext (RCCL needs no CuPy device
    context) and ``cuda_aware`` reports True whenever RCCL is in use (device
    buffers always travel directly over xGMI).
    """
    use_cuda = torch.cuda.is_available() and ("USE_CPU" not in os.environ)
    if num_gpus is None:
        num_gpus = torch.cuda.device_count() if use_cuda else 1
    rank = max(P.rank, 0)
    device_ordinal = rank % max(num_gpus, 1)
    if use_cuda:
        local = int(os.environ.get("LOCAL_RANK", device_ordinal))
        device_ordinal = local % torch.cuda.device_count()
        torch.cuda.set_device(device_ordinal)
        device = torch.device(f"cuda:{device_ordinal}")
    else:
        device = torch.device("cpu")
    cuda_aware = use_cuda  # RCCL passes device pointers natively
    return use_cuda, cuda_aware, device_ordinal, device, nullcontext()


def get_gpu_memory() -> List[int]:
    """Per-GPU used VRAM in MiB via rocm-smi (reference used nvidia-smi)."""
    try:
        out = sp.check_output(
            ["rocm-smi", "--showmeminfo", "vram", "--csv"], stderr=sp.STDOUT
        ).decode()
    except (sp.CalledProcessError, FileNotFoundError):
        # fall back to torch allocator stats for the current device
        if torch.cuda.is_available():
            return [int(torch.cuda.memory_allocated() // (1024 * 1024))]
        return []
    vals = []
    for line in out.splitlines()[1:]:
        parts = line.split(",")
        if len(parts) >= 3 and parts[0].startswith("card"):
            try:
                vals.append(int(parts[2]) // (1024 * 1024))
            except ValueError:
                pass
    return vals


def profile_gpu_memory(outfile, dt: float = 1.0):
    """Poll GPU memory to a CSV forever (daemon-process target; reference
    utils.py:28-40)."""
    t0 = time.time()
    with open(outfile, "w") as f:
        while True:
            muv = get_gpu_memory()
            f.write(f"{time.time() - t0}, " + ", ".join(str(m) for m in muv) + "\n")
            f.flush()
            time.sleep(dt)


def generate_batch_indices(P, n: int, batch_size: int, shuffle: bool = False,
                           seed: int = 0):
    """Yield (start, stop) index ranges covering ``n`` samples.

    Restores the helper the reference calls but never defines
    (experiment_navier_stokes.py:130,157 — SURVEY.md 2.5).  The range order
    is derived from a fixed seed so every rank of ``P`` iterates the same
    batches (ranks are seeded differently for weights, so relying on the
    global torch RNG would desynchronize the collectives).
    """
    bounds = [(a, min(a + batch_size, n)) for a in range(0, n, batch_size)]
    if shuffle:
        rng = np.random.RandomState(seed)
        order = rng.permutation(len(bounds))
        bounds = [bounds[i] for i in order]
    return bounds


def unit_gaussian_normalize(x):
    """Per-location batch normalisation (reference utils.py:90-94)."""
    mu = torch.mean(x, 0).unsqueeze(0)
    std = torch.std(x, 0).unsqueeze(0)
    out = (x - mu) / (std + 1e-6)
    return out, mu, std


# The reference exports this under a typo'd name (utils.py:90); keep the alias
# for API compatibility.
unit_guassian_normalize = unit_gaussian_normalize


def unit_gaussian_denormalize(x, mu, std):
    return x * (std + 1e-6) + mu
