"""Model-facing fused NHWC ops: BatchNorm(+add)(+ReLU), pooling.

The reference runs BN and ReLU as separate cuDNN/elementwise kernels after
every conv (53 BN layers in ResNet-50) — on MI355X that is pure
HBM traffic, so normalization, the residual add and the activation are fused
into single NHWC HIP kernels (ops/csrc/batchnorm.hip, pool.hip).  Stats are
accumulated in fp32 regardless of the activation dtype (bf16 path).

CPU fallback composes plain torch ops (autograd handled by torch); GPU path
uses custom autograd.Functions over the extension.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ._ext import require_ext, use_ext_for
from .conv import GradCell

__all__ = ["batch_norm", "bn_add_relu", "max_pool_3x3_s2", "global_avg_pool"]


class _BNFunction(torch.autograd.Function):
    """Fused BN(+add)(+ReLU) training-mode forward/backward on NHWC GPU
    tensors.  When the producing conv attached per-block statistics
    partials (conv epilogue fusion), the reduce pass over x is skipped."""

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                momentum, eps, relu, addend, parts, res_cell=None):
        e = require_ext()
        if parts is not None:
            y, mean, invstd, scale, shift, mask = \
                e.batch_norm_fwd_train_from_parts(
                    x, parts, weight, bias, running_mean, running_var,
                    float(momentum), float(eps), bool(relu), addend)
        else:
            y, mean, invstd, scale, shift, mask = e.batch_norm_fwd_train(
                x, weight, bias, running_mean, running_var,
                float(momentum), float(eps), bool(relu), addend)
        # scale/shift ([C] fp32) let the bwd recompute the relu mask as
        # scale*x+shift>0 for addend-free BNs; block-tail BNs save the
        # fwd-computed 1-bit relu mask instead (bwd skips the y read)
        ctx.save_for_backward(x, y, weight, mean, invstd, scale, shift,
                              mask)
        ctx.relu = bool(relu)
        ctx.has_addend = addend is not None
        # block-tail BN grad mailbox (see batch_norm below): our backward
        # folds a rerouted shortcut gradient into the reduce as a second
        # linear go operand, replacing the eager CUDAFunctor_add at the
        # output's AccumulateGrad.
        ctx.res_cell = res_cell
        return y

    @staticmethod
    def backward(ctx, grad_out):
        x, y, weight, mean, invstd, scale, shift, mask = ctx.saved_tensors
        e = require_ext()
        go2 = None
        cell = getattr(ctx, "res_cell", None)
        if cell is not None and cell.g is not None:
            go2 = cell.g.contiguous(memory_format=torch.channels_last)
            cell.g = None
        grad_x, grad_w, grad_b, ghat = e.batch_norm_bwd(
            x, grad_out.contiguous(memory_format=torch.channels_last),
            y, weight, mean, invstd, ctx.relu, ctx.has_addend,
            scale, shift, go2, mask)
        grad_addend = ghat if ctx.has_addend else None
        return (grad_x, grad_w, grad_b, None, None, None, None, None,
                grad_addend, None, None)


def _bn_torch(x: torch.Tensor, bn, relu: bool,
              addend: Optional[torch.Tensor]) -> torch.Tensor:
    """Plain-torch reference path (CPU, or GPU with ext disabled)."""
    # match nn.BatchNorm2d bookkeeping (num_batches_tracked handled by caller)
    y = F.batch_norm(
        x, bn.running_mean, bn.running_var, bn.weight, bn.bias,
        bn.training or not bn.track_running_stats,
        bn.momentum if bn.momentum is not None else 0.0, bn.eps)
    if addend is not None:
        y = y + addend
    if relu:
        y = F.relu(y, inplace=True)
    return y


def batch_norm(x: torch.Tensor, bn, relu: bool = False,
               addend: Optional[torch.Tensor] = None) -> torch.Tensor:
    """BatchNorm2d through module ``bn``'s parameters/buffers, optionally fused
    with a residual add and/or ReLU."""
    if bn.training and bn.track_running_stats and bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    C = x.size(1)
    if use_ext_for(x) and (C & (C - 1)) == 0:  # kernels take pow2 C only
        parts = getattr(x, "_amdtrain_bn_stats", None) if bn.training else None
        xc = x.contiguous(memory_format=torch.channels_last)
        ac = None if addend is None else addend.contiguous(memory_format=torch.channels_last)
        if bn.training:
            # block-tail BNs publish a mailbox the NEXT residual block can
            # reroute its shortcut gradient into (ResidualGradTap)
            cell = GradCell() if (relu and ac is not None) else None
            out = _BNFunction.apply(xc, bn.weight, bn.bias, bn.running_mean,
                                    bn.running_var, bn.momentum, bn.eps,
                                    relu, ac, parts, cell)
            if cell is not None:
                out._amdtrain_next_cell = cell
            return out
        return require_ext().batch_norm_fwd_eval(
            xc, bn.weight, bn.bias, bn.running_mean, bn.running_var,
            float(bn.eps), bool(relu), ac)
    return _bn_torch(x, bn, relu, addend)


def bn_add_relu(x: torch.Tensor, bn, addend: torch.Tensor) -> torch.Tensor:
    """relu(bn(x) + addend) — the residual-block tail, one kernel on GPU."""
    return batch_norm(x, bn, relu=True, addend=addend)


class _MaxPool3x3S2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        e = require_ext()
        y, idx = e.max_pool_3x3_s2_fwd(x)
        ctx.save_for_backward(idx)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, grad_out):
        (idx,) = ctx.saved_tensors
        e = require_ext()
        n, c, h, w = ctx.in_shape
        gx = e.max_pool_3x3_s2_bwd(
            grad_out.contiguous(memory_format=torch.channels_last), idx,
            int(h), int(w))
        return gx


def max_pool_3x3_s2(x: torch.Tensor) -> torch.Tensor:
    """3x3 stride-2 pad-1 max pool (the ResNet stem pool)."""
    if use_ext_for(x):
        return _MaxPool3x3S2.apply(x.contiguous(memory_format=torch.channels_last))
    return F.max_pool2d(x, kernel_size=3, stride=2, padding=1)


class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        e = require_ext()
        y = e.global_avg_pool_fwd(x)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, grad_out):
        e = require_ext()
        n, c, h, w = ctx.in_shape
        return e.global_avg_pool_bwd(grad_out.contiguous(), int(h), int(w))


def global_avg_pool(x: torch.Tensor) -> torch.Tensor:
    """Global average pool to [N,C,1,1] (AdaptiveAvgPool2d((1,1)) parity)."""
    if use_ext_for(x):
        return _GlobalAvgPool.apply(x.contiguous(memory_format=torch.channels_last))
    return F.adaptive_avg_pool2d(x, (1, 1))
