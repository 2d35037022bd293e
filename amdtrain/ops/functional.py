"""Functional wrappers over the gfx950 HIP kernels, with CPU references.

Each op has (a) a plain-PyTorch reference implementation (serves CPU tests and
is the numerics baseline the GPU kernels are verified against) and (b) a
dispatch to ``amdtrain._C`` for GPU tensors.  Autograd integration lives here
as ``torch.autograd.Function`` subclasses.

Kernel inventory follows SURVEY.md §2c (the per-step kernel set the reference
launches through cuDNN/cuBLAS/apex/horovod).
"""

from __future__ import annotations

from typing import List, Sequence

import torch
import torch.nn.functional as F

from ._ext import ext, ext_available, require_ext, use_ext_for

__all__ = [
    "ext", "ext_available", "require_ext", "use_ext_for",
    "cross_entropy", "topk_correct_counts", "normalize_u8",
    "multi_tensor_scale_check", "multi_tensor_cast",
]


# --------------------------------------------------------------------------
# Cross-entropy (log-softmax + NLL), mean reduction — reference
# nn.CrossEntropyLoss at distributed.py:151, launched per step.
# --------------------------------------------------------------------------

class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor):
        e = require_ext()
        losses, lse = e.cross_entropy_fwd(logits, target)
        ctx.save_for_backward(logits, target, lse)
        return losses.mean()

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        logits, target, lse = ctx.saved_tensors
        e = require_ext()
        # upstream grad stays on device: no host sync per backward
        grad_logits = e.cross_entropy_bwd(logits, target, lse,
                                          grad_out.reshape(1))
        return grad_logits, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy loss over the batch (fp32 accumulation)."""
    if use_ext_for(logits):
        return _CrossEntropy.apply(logits, target)
    return F.cross_entropy(logits.float(), target)


# --------------------------------------------------------------------------
# Top-k accuracy ranks (reference accuracy(), distributed.py:381-395).
# --------------------------------------------------------------------------

def topk_ranks(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Rank of the true class in the sorted (desc, first-occurrence) logits.

    rank r means the label would appear at position r in ``topk`` output,
    so correct@k == (rank < k).
    """
    if use_ext_for(logits):
        return require_ext().topk_ranks(logits, target)
    with torch.no_grad():
        tv = logits.gather(1, target.reshape(-1, 1))  # [B,1]
        greater = (logits > tv).sum(dim=1)
        # ties broken by index order (torch.topk returns lower index first)
        idx = torch.arange(logits.size(1), device=logits.device)
        tie_before = ((logits == tv) & (idx.unsqueeze(0) < target.reshape(-1, 1))).sum(dim=1)
        return (greater + tie_before).to(torch.int32)


def topk_correct_counts(logits: torch.Tensor, target: torch.Tensor,
                        ks: Sequence[int]) -> torch.Tensor:
    """[len(ks)] fp32 tensor of #correct-within-top-k."""
    ranks = topk_ranks(logits, target)
    return torch.stack([(ranks < k).sum().float() for k in ks])


# --------------------------------------------------------------------------
# On-GPU input normalization (the apex-style prefetcher's
# float-cast + sub_(mean) + div_(std), apex_distributed.py:119-122,157-158).
# --------------------------------------------------------------------------

# ImageNet stats scaled to the uint8 [0,255] domain, as in the reference
# prefetcher (apex_distributed.py:119-122).
IMAGENET_MEAN_255 = (0.485 * 255, 0.456 * 255, 0.406 * 255)
IMAGENET_STD_255 = (0.229 * 255, 0.224 * 255, 0.225 * 255)


def normalize_u8(x: torch.Tensor, dtype: torch.dtype = torch.float32,
                 mean: Sequence[float] = IMAGENET_MEAN_255,
                 std: Sequence[float] = IMAGENET_STD_255) -> torch.Tensor:
    """uint8 NCHW(channels_last) [N,3,H,W] -> normalized float NCHW(channels_last).

    One fused HIP kernel on GPU (cast + (x-mean)/std, NHWC-vectorized).
    """
    assert x.dtype == torch.uint8 and x.dim() == 4 and x.size(1) == len(mean)
    if use_ext_for(x):
        xc = x.contiguous(memory_format=torch.channels_last)
        code = {torch.float32: 0, torch.bfloat16: 1, torch.float16: 2}[dtype]
        return require_ext().normalize_u8(xc, list(mean), list(std), code)
    m = torch.tensor(mean, dtype=torch.float32, device=x.device).reshape(1, -1, 1, 1)
    s = torch.tensor(std, dtype=torch.float32, device=x.device).reshape(1, -1, 1, 1)
    return ((x.float() - m) / s).to(dtype)


# --------------------------------------------------------------------------
# Multi-tensor loss-scale / unscale + inf check and dtype casts
# (apex amp_C equivalents, apex_distributed.py:216,328-329; horovod fp16
# gradient compression, horovod_distributed.py:159).
# --------------------------------------------------------------------------

def multi_tensor_scale_check(tensors: List[torch.Tensor], scale: float,
                             found_inf: torch.Tensor) -> None:
    """In-place: t *= scale for each tensor; set found_inf[0]=1 on inf/nan."""
    if tensors and tensors[0].is_cuda and ext_available():
        require_ext().multi_tensor_scale_check(tensors, scale, found_inf)
        return
    for t in tensors:
        t.mul_(scale)
        if not torch.isfinite(t).all():
            found_inf.fill_(1.0)


def multi_tensor_cast(src: List[torch.Tensor], dst: List[torch.Tensor]) -> None:
    """dst[i].copy_(src[i]) with dtype conversion, one fused launch on GPU."""
    if src and src[0].is_cuda and ext_available():
        require_ext().multi_tensor_cast(src, dst)
        return
    for s, d in zip(src, dst):
        d.copy_(s)
