"""In-tree HIP extension loader for gfx950.

Builds ``dfno_amd/csrc/*.hip`` into ``dfno_amd/_build/dfno_hip.so`` (in-tree so
the built artifact travels to GPU boxes with the repo snapshot), via
``torch.utils.cpp_extension`` driving hipcc with ``PYTORCH_ROCM_ARCH=gfx950``.

Policy: on a CUDA/HIP device the extension is REQUIRED — ops raise rather than
silently falling back to eager PyTorch (so a GPU run that passes is running
the native kernels).  On CPU the pure-torch reference paths are used.
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

_PKG_DIR = Path(__file__).resolve().parent
_CSRC = _PKG_DIR / "csrc"
_BUILD = _PKG_DIR / "_build"
_EXT_NAME = "dfno_hip"

_ext = None
_load_error = None

SOURCES = [
    _CSRC / "bindings.cpp",
    _CSRC / "pointwise.hip",
    _CSRC / "spectral.hip",
    _CSRC / "proj_head.hip",
    _CSRC / "dft.hip",
    _CSRC / "lift_head.hip",
    _CSRC / "pack.hip",
    _CSRC / "bf16.hip",
    _CSRC / "dft2d.hip",
    _CSRC / "mix_bwd.hip",
]


def _find_prebuilt():
    for suffix in (".so",):
        cand = _BUILD / f"{_EXT_NAME}{suffix}"
        if cand.exists():
            srcs_mtime = max((s.stat().st_mtime for s in SOURCES if s.exists()), default=0.0)
            if cand.stat().st_mtime >= srcs_mtime:
                return cand
    return None


def _import_so(path: Path):
    import importlib.util

    spec = importlib.util.spec_from_file_location(_EXT_NAME, str(path))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def build(verbose: bool = False):
    """Compile the HIP extension for gfx950 into the in-tree build dir."""
    global _ext, _load_error
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    _BUILD.mkdir(exist_ok=True)
    from torch.utils.cpp_extension import load

    mod = load(
        name=_EXT_NAME,
        sources=[str(s) for s in SOURCES],
        build_directory=str(_BUILD),
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
    )
    _ext = mod
    _load_error = None
    return mod


def get(required: bool = False):
    """Return the extension module, importing the prebuilt .so if present.

    required=True (any GPU-tensor call site): raise loudly when missing.
    """
    global _ext, _load_error
    if _ext is not None:
        return _ext
    pre = _find_prebuilt()
    if pre is not None:
        try:
            _ext = _import_so(pre)
            return _ext
        except Exception as e:  # corrupt/stale build — try rebuilding
            _load_error = e
    try:
        return build()
    except Exception as e:
        _load_error = e
        if required:
            raise RuntimeError(
                f"dfno_amd HIP extension is required on GPU but failed to load/build: {e}"
            ) from e
        return None
