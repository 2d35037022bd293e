"""Seeding / determinism controls.

Parity with the reference's seed block (distributed.py:116-124): seed Python
and torch RNGs and flip the backend into deterministic mode (with the same
"you may see unexpected behavior" caveat the reference warns about —
deterministic MIOpen algorithms are slower).  When no seed is given we keep
the reference's fast path: ``benchmark = True`` (distributed.py:158).
"""

from __future__ import annotations

import random
import warnings
from typing import Optional

import torch
import torch.backends.cudnn as cudnn  # maps to MIOpen on ROCm


def set_seed(seed: Optional[int], warn: bool = True) -> None:
    if seed is None:
        cudnn.benchmark = True
        return
    random.seed(seed)
    torch.manual_seed(seed)
    cudnn.deterministic = True
    cudnn.benchmark = False
    if warn:
        warnings.warn(
            "You have chosen to seed training. This will turn on the "
            "deterministic MIOpen mode, which can slow down training "
            "considerably. You may see unexpected behavior when restarting "
            "from checkpoints."
        )
