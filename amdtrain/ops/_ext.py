"""Loader for the in-tree compiled HIP extension ``amdtrain._C``.

The .so is built in-tree (``python setup.py build_ext --inplace`` /
``__graft_entry__.build()``) so it travels with the repo snapshot to the GPU
box.  Policy:

  * extension present           -> used for every GPU op
  * extension absent, CPU only  -> fine; torch reference paths serve the op
  * extension absent, GPU box   -> RuntimeError at first GPU-op dispatch
                                   (set AMDTRAIN_ALLOW_EAGER=1 to override)
  * AMDTRAIN_DISABLE_EXT=1      -> force torch paths (A/B benchmarking)
"""

from __future__ import annotations

import os
from typing import Optional

import torch

_EXT = None
_TRIED = False


def _load() -> Optional[object]:
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    if os.environ.get("AMDTRAIN_DISABLE_EXT") == "1":
        _EXT = None
        return None
    try:
        from amdtrain import _C  # type: ignore
        _EXT = _C
    except ImportError:
        _EXT = None
    return _EXT


def ext() -> Optional[object]:
    return _load()


def ext_available() -> bool:
    return _load() is not None


def require_ext():
    """Return the extension; on a GPU box raise loudly if it is missing."""
    e = _load()
    if e is not None:
        return e
    if torch.cuda.is_available() and os.environ.get("AMDTRAIN_ALLOW_EAGER") != "1":
        raise RuntimeError(
            "amdtrain._C HIP extension is not built but a GPU is present. "
            "Build it in-tree with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950). Set AMDTRAIN_ALLOW_EAGER=1 only to "
            "deliberately benchmark the plain-PyTorch path."
        )
    return None


def use_ext_for(x: torch.Tensor) -> bool:
    """True when ``x`` is a GPU tensor and the extension should serve it."""
    if not x.is_cuda:
        return False
    return require_ext() is not None
