"""CrossEntropyLoss module over the fused HIP kernel.

Drop-in for ``nn.CrossEntropyLoss`` as the reference uses it
(distributed.py:151,251): mean reduction over a [B, C] logits batch.
Forward computes per-row logsumexp + NLL in one kernel (fp32 accumulation
even for bf16 logits); backward is the fused ``softmax - onehot`` kernel
(SURVEY §2c).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .functional import cross_entropy


class CrossEntropyLoss(nn.Module):
    def __init__(self):
        super().__init__()

    def forward(self, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        return cross_entropy(logits, target)
