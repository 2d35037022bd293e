"""Checkpoint save/load with the reference's dict schema.

The reference writes ``{'epoch': epoch+1, 'arch': str, 'state_dict': ...,
'best_acc1': float}`` to ``checkpoint.pth.tar`` and copies to
``model_best.pth.tar`` when the top-1 improved (distributed.py:218-225,
327-330).  Optimizer state is NOT part of the schema (resume is via
``--start-epoch``).  We keep that exact schema (BASELINE.json north-star)
and add an optional richer resume path on top.
"""

from __future__ import annotations

import os
import shutil
from typing import Any, Dict, Optional

import torch


def save_checkpoint(state: Dict[str, Any], is_best: bool,
                    filename: str = "checkpoint.pth.tar",
                    best_filename: str = "model_best.pth.tar") -> None:
    """Write ``state`` to ``filename``; copy to ``best_filename`` on best."""
    torch.save(state, filename)
    if is_best:
        shutil.copyfile(filename, best_filename)


def make_checkpoint_state(epoch: int, arch: str, model,
                          best_acc1: float) -> Dict[str, Any]:
    """Build the reference-schema checkpoint dict.

    ``model`` may be a bare module or a wrapper exposing ``.module``
    (our NativeDDP / ScatterGatherDataParallel do, matching the reference's
    ``model.module.state_dict()`` at distributed.py:222).
    """
    module = getattr(model, "module", model)
    return {
        "epoch": epoch + 1,
        "arch": arch,
        "state_dict": module.state_dict(),
        "best_acc1": best_acc1,
    }


def load_checkpoint(path: str, model=None,
                    map_location: Optional[str] = "cpu") -> Dict[str, Any]:
    """Load a reference-schema checkpoint; optionally restore into ``model``."""
    if not os.path.isfile(path):
        raise FileNotFoundError(f"no checkpoint found at '{path}'")
    # the schema is plain tensors/scalars: prefer the pickle-safe load and
    # only fall back for externally produced checkpoints with custom classes
    try:
        state = torch.load(path, map_location=map_location, weights_only=True)
    except Exception:
        state = torch.load(path, map_location=map_location, weights_only=False)
    if model is not None:
        module = getattr(model, "module", model)
        sd = state["state_dict"]
        # tolerate checkpoints written from a wrapped model ("module." prefix)
        if any(k.startswith("module.") for k in sd):
            sd = {k[len("module."):]: v for k, v in sd.items()}
        module.load_state_dict(sd)
    return state
