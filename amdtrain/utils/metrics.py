"""Top-k classification accuracy.

Behavioral parity with the reference's ``accuracy`` helper
(distributed.py:381-395): returns a list of [1]-shaped float tensors, one per
requested k, each ``100 * (#samples whose top-k predictions contain the
label) / batch``.

On MI355X the [B,1000] top-5 selection runs as a hand-written HIP kernel
(amdtrain/ops/csrc/elementwise.hip) — this module dispatches to it when the
extension is loaded and the input lives on the GPU.
"""

from __future__ import annotations

from typing import Iterable, List

import torch


def accuracy(output: torch.Tensor, target: torch.Tensor,
             topk: Iterable[int] = (1,)) -> List[torch.Tensor]:
    """Compute top-k accuracies of ``output`` logits against ``target`` labels."""
    topk = tuple(topk)
    maxk = max(topk)

    if output.is_cuda:
        from ..ops import functional as F
        if F.ext_available():
            counts = F.topk_correct_counts(output, target, topk)  # [len(topk)] fp32
            batch = output.size(0)
            return [counts[i].reshape(1) * (100.0 / batch) for i in range(len(topk))]

    with torch.no_grad():
        batch = target.size(0)
        _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)  # [B, maxk]
        correct = pred.eq(target.reshape(-1, 1))  # [B, maxk] bool
        res = []
        for k in topk:
            correct_k = correct[:, :k].any(dim=1).float().sum(0, keepdim=True)
            res.append(correct_k * (100.0 / batch))
        return res
