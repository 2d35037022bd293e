"""Hand-written gfx950 (CDNA4) HIP kernels + CPU reference implementations.

The compiled extension ``amdtrain._C`` (built in-tree from ``ops/csrc`` by
``setup.py build_ext --inplace`` with ``--offload-arch=gfx950``) carries the
MI355X compute path.  On CPU (or with ``AMDTRAIN_DISABLE_EXT=1``) every op
falls back to a plain PyTorch implementation — that is the numerics reference
the kernel tests compare against.

On a GPU box, ops raise if the extension is missing unless
``AMDTRAIN_ALLOW_EAGER=1`` — no silent eager fallback.
"""

from . import functional  # noqa: F401
from . import fused  # noqa: F401
from .sgd import FusedSGD  # noqa: F401
from .cross_entropy import CrossEntropyLoss  # noqa: F401

__all__ = ["functional", "fused", "FusedSGD", "CrossEntropyLoss"]
