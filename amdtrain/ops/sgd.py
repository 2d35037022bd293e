"""Fused multi-tensor SGD with momentum + weight decay.

The reference's optimizer step launches one elementwise CUDA kernel per
parameter tensor (161 tensors for ResNet-50; torch.optim.SGD, reference
distributed.py:153-156,269) plus another 161 for ``zero_grad``.  Here the
whole update is one multi-tensor HIP launch (ops/csrc/sgd.hip) with the
grad-zeroing folded in (SURVEY §2c "fuse into one multi-tensor HIP kernel").

Semantics match ``torch.optim.SGD`` (momentum buffer ``b = mu*b + g + wd*p``,
``p -= lr*b``; first step ``b = g + wd*p``) for the reference's configuration
(dampening=0).  Non-zero dampening falls back to the per-tensor loop.
"""

from __future__ import annotations

from typing import List

import torch
from torch.optim import Optimizer

from ._ext import ext_available, require_ext


class FusedSGD(Optimizer):
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        if lr < 0.0:
            raise ValueError(f"invalid lr {lr}")
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("nesterov requires momentum > 0 and dampening = 0")
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None, zero_grad: bool = False):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params: List[torch.Tensor] = []
            grads: List[torch.Tensor] = []
            bufs: List[torch.Tensor] = []
            firsts: List[bool] = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                first = "momentum_buffer" not in state
                if first and group["momentum"] != 0:
                    state["momentum_buffer"] = torch.zeros_like(
                        p, memory_format=torch.preserve_format)
                params.append(p)
                grads.append(p.grad)
                bufs.append(state["momentum_buffer"] if group["momentum"] != 0
                            else p.grad)  # unused when momentum==0
                firsts.append(first)

            if not params:
                continue

            fused_ok = (params[0].is_cuda and ext_available()
                        and group["dampening"] == 0
                        and all(f == firsts[0] for f in firsts))
            if fused_ok:
                require_ext().multi_tensor_sgd(
                    params, grads, bufs,
                    float(group["lr"]), float(group["momentum"]),
                    float(group["weight_decay"]), bool(group["nesterov"]),
                    bool(firsts[0] and group["momentum"] != 0),
                    bool(zero_grad))
            else:
                self._step_loop(group, params, grads, firsts, zero_grad)
        return loss

    def _step_loop(self, group, params, grads, firsts, zero_grad):
        mu, damp = group["momentum"], group["dampening"]
        wd, lr, nesterov = group["weight_decay"], group["lr"], group["nesterov"]
        for p, g, first in zip(params, grads, firsts):
            if wd != 0:
                g = g.add(p, alpha=wd)
            if mu != 0:
                buf = self.state[p]["momentum_buffer"]
                if first:
                    buf.copy_(g)
                else:
                    buf.mul_(mu).add_(g, alpha=1 - damp)
                g = g.add(buf, alpha=mu) if nesterov else buf
            p.add_(g, alpha=-lr)
            if zero_grad:
                p.grad.zero_()
