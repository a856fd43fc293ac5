"""Loader for the gemscore HIP extension (mpi4dl_amd/csrc).

Policy: on a GPU machine the native extension is REQUIRED — a missing
.so raises instead of silently falling back to eager PyTorch, so a
"working" GPU run always means the CDNA4 kernels ran. On CPU-only
machines (unit tests, shape inference) the torch fallback paths are
used and this module reports unavailable.

Build:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

from __future__ import annotations

import os

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        from .. import _gemscore  # built in-tree next to the package

        _ext = _gemscore
    except ImportError as e:
        _ext = None
        if torch.cuda.is_available() and os.environ.get("MPI4DL_ALLOW_EAGER") != "1":
            raise RuntimeError(
                "mpi4dl_amd._gemscore HIP extension not built but a GPU is "
                "present - run `PYTORCH_ROCM_ARCH=gfx950 python setup.py "
                "build_ext --inplace` (set MPI4DL_ALLOW_EAGER=1 to force the "
                f"slow eager fallback). Import error: {e}"
            ) from e
    return _ext


def ext():
    return _load()


def available() -> bool:
    return _load() is not None
