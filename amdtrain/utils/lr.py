"""Learning-rate schedule.

Behavioral parity with the reference step decay (distributed.py:374-378):
``lr = base_lr * 0.1 ** (epoch // 30)`` applied by mutating the optimizer's
param groups in place.
"""

from __future__ import annotations


def adjust_learning_rate(optimizer, epoch: int, base_lr: float,
                         decay: float = 0.1, step: int = 30) -> float:
    """Step-decay the LR and write it into every param group; returns the LR."""
    lr = base_lr * (decay ** (epoch // step))
    for group in optimizer.param_groups:
        group["lr"] = lr
    return lr
