"""Hand-written 1x1 convolution over the gfx950 MFMA GEMM kernels.

A 1x1 conv on NHWC data is exactly a GEMM over rows R = N*H*W
(SURVEY §2c): forward C[R,Cout] = X[R,Cin] x W[Cout,Cin]^T, dgrad uses the
pre-transposed weight, wgrad is the TN GEMM with split-M accumulation.
Strided (downsample) 1x1 convs gather/scatter the even rows around the same
GEMMs.

Dispatch: ``AmdConv2d`` routes all eligible convs (GPU + extension + bf16
path) to the hand-written kernels; see the class docstring for the matrix.
profiles/ holds the per-shape measurements behind the defaults.
"""

from __future__ import annotations

import os

import torch
import torch.nn as nn

from ._ext import ext_available, require_ext


def _rows(x: torch.Tensor) -> torch.Tensor:
    """channels_last NCHW -> [N*H*W, C] view (no copy)."""
    n, c, h, w = x.shape
    return x.permute(0, 2, 3, 1).reshape(n * h * w, c)


def _fuse_stats_enabled() -> bool:
    return os.environ.get("AMDTRAIN_FUSE_BNSTATS", "1") == "1"


def _wgrad2_enabled() -> bool:
    """tn2_wgrad (wgrad.hip): tr-read TN core — single launch for all 9
    conv3x3 taps, no atomics, bitwise deterministic.  Default on;
    AMDTRAIN_WGRAD2=0 falls back to the round-1 per-tap kernels for A/B."""
    return os.environ.get("AMDTRAIN_WGRAD2", "1") == "1"


class GradCell:
    """Shared mailbox between ResidualGradTap and the producing block-tail
    _BNFunction (fused.py).  A plain class (NOT a dict):
    torch.amp.custom_fwd(cast_inputs=...) deep-copies dict arguments while
    casting, which would silently disconnect the mailbox."""
    __slots__ = ("armed", "g")

    def __init__(self):
        self.armed = False
        self.g = None


class ResidualGradTap(torch.autograd.Function):
    """Reroutes a residual block's identity-shortcut gradient into the
    PRODUCING block-tail BN's backward (a second, linearly-read go operand
    of its reduce kernel) instead of an eager 2-read+1-write tensor add at
    the shared tensor's AccumulateGrad.

    The tap wraps the ADDEND input of block b's bn_add_relu; the cell is
    created and attached by block b-1's tail BN, whose backward runs
    strictly AFTER every node of block b (it is upstream of all of them).
    A first attempt fused this add into conv1's dgrad epilogue instead:
    measured SLOWER (the fragment-shaped 2-byte epilogue reads cost ~5x
    the eager add) — the BN reduce reads the extra operand with the same
    16 B vectorized stream as its other inputs.  Safety latch: the grad is
    stashed ONLY when the producing BN armed the cell; any fallback keeps
    plain autograd accumulation.
    """

    @staticmethod
    def forward(ctx, z: torch.Tensor, cell: "GradCell"):
        ctx.cell = cell
        return z.view_as(z)

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        if ctx.cell.armed:
            ctx.cell.g = grad
            return None, None
        return grad, None


class _Conv1x1(torch.autograd.Function):
    """1x1 conv; forward also emits per-block BN-statistics partials
    (non-differentiable 2nd output) for the fused conv->BN pipeline when
    stride == 1 and fusion is enabled.  An optional grad_cell (see
    ResidualGradTap) fuses the residual-branch gradient into the dgrad."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, stride: int):
        e = require_ext()
        n, cin, h, w = x.shape
        cout = weight.shape[0]
        xc = x.contiguous(memory_format=torch.channels_last)
        x2d = _rows(xc)
        w2d = weight.reshape(cout, cin)
        stats = None
        if stride > 1:
            # even-row gather happens inside the A staging (no copy); the
            # full x2d is saved — it is the same tensor the sibling conv of
            # the residual block already saves, so no extra memory
            y2d = e.gemm_bt_strided(x2d, w2d, n, h, w, stride)
            ho = (h + stride - 1) // stride
            wo = (w + stride - 1) // stride
        elif _fuse_stats_enabled():
            y2d, stats = e.gemm_bt_stats(x2d, w2d)
            ho, wo = h, w
        else:
            y2d = e.gemm_bt(x2d, w2d, False)
            ho, wo = h, w
        ctx.save_for_backward(x2d, w2d)
        ctx.meta = (n, cin, h, w, stride, ho, wo, cout)
        y = y2d.view(n, ho, wo, cout).permute(0, 3, 1, 2)
        if stats is None:
            stats = torch.empty(0, device=x.device)
        ctx.mark_non_differentiable(stats)
        return y, stats

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, grad_y: torch.Tensor, _grad_stats=None):
        e = require_ext()
        x2d, w2d = ctx.saved_tensors
        n, cin, h, w, stride, ho, wo, cout = ctx.meta
        gy = grad_y.contiguous(memory_format=torch.channels_last)
        gy2d = _rows(gy).to(torch.bfloat16)
        # dgrad: dX = dY x W  (BT form with pre-transposed weight)
        wT = e.transpose_2d(w2d)                      # [Cin, Cout]
        dx2d = e.gemm_bt(gy2d, wT, False)             # [R_sub, Cin] bf16
        if stride > 1:
            # fused zero+scatter (one write pass; stride-2 only in ResNet)
            dx = e.scatter_rows_x2(dx2d, n, h, w, ho, wo)
            dw = e.tn2_wgrad(gy2d, x2d, 1, n, h, w, stride, 1) \
                if _wgrad2_enabled() \
                else e.gemm_tn_strided(gy2d, x2d, n, h, w, stride)
        else:
            dx = dx2d.view(n, ho, wo, cin).permute(0, 3, 1, 2)
            dw = e.tn2_wgrad(gy2d, x2d) if _wgrad2_enabled() \
                else e.gemm_tn(gy2d, x2d, 0)
        return dx, dw.reshape(cout, cin, 1, 1), None


def conv1x1_mfma(x: torch.Tensor, weight: torch.Tensor,
                 stride: int = 1) -> torch.Tensor:
    y, stats = _Conv1x1.apply(x, weight, stride)
    if stats.numel():
        y._amdtrain_bn_stats = stats
    return y


class _Conv3x3(torch.autograd.Function):
    """3x3 pad-1 conv (stride 1/2) over the implicit-GEMM gfx950 kernels."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, stride: int):
        e = require_ext()
        n, cin, h, w = x.shape
        cout = weight.shape[0]
        xc = x.contiguous(memory_format=torch.channels_last)
        x2d = _rows(xc)
        # channels_last weight [Cout,Cin,3,3] is [Cout][kh][kw][Cin] in memory
        w2d = weight.contiguous(memory_format=torch.channels_last) \
            .permute(0, 2, 3, 1).reshape(cout, 9 * cin)
        if _fuse_stats_enabled():
            y2d, stats = e.conv3x3_fwd_stats(x2d, n, h, w, stride, w2d)
        else:
            y2d = e.conv3x3_fwd(x2d, n, h, w, stride, w2d)
            stats = torch.empty(0, device=x.device)
        ctx.save_for_backward(x2d, w2d)
        ctx.meta = (n, cin, h, w, stride, cout)
        ho = (h + 2 - 3) // stride + 1
        wo = (w + 2 - 3) // stride + 1
        y = y2d.view(n, ho, wo, cout).permute(0, 3, 1, 2)
        ctx.mark_non_differentiable(stats)
        return y, stats

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, grad_y: torch.Tensor, _grad_stats=None):
        e = require_ext()
        x2d, w2d = ctx.saved_tensors
        n, cin, h, w, stride, cout = ctx.meta
        gy = grad_y.contiguous(memory_format=torch.channels_last)
        gy2d = _rows(gy).to(torch.bfloat16)
        dx2d = e.conv3x3_dgrad(gy2d, n, h, w, stride, w2d)
        dx = dx2d.view(n, h, w, cin).permute(0, 3, 1, 2)
        # [Cout, 9*Cin] f32 — single-launch tap-gather TN (wgrad.hip)
        dw2d = e.tn2_wgrad(gy2d, x2d, 9, n, h, w, stride, 2) \
            if _wgrad2_enabled() \
            else e.conv3x3_wgrad(gy2d, x2d, n, h, w, stride)
        # back to [Cout,Cin,3,3] (channels_last layout of the weight)
        dw = dw2d.view(cout, 3, 3, cin).permute(0, 3, 1, 2) \
            .contiguous(memory_format=torch.channels_last)
        return dx, dw, None


def conv3x3_mfma(x: torch.Tensor, weight: torch.Tensor,
                 stride: int = 1) -> torch.Tensor:
    y, stats = _Conv3x3.apply(x, weight, stride)
    if stats.numel():
        y._amdtrain_bn_stats = stats
    return y


class _ConvGrouped3x3(torch.autograd.Function):
    """Grouped 3x3 conv (ResNeXt) as a BLOCK-DIAGONALIZED dense conv over
    the in-house implicit-GEMM kernels.

    The grouped weight [Cout, S, 3, 3] (S = Cin/groups) is expanded to a
    dense [Cout, 9*Cin] with zeros off the group diagonal, then the
    standard conv3x3 fwd/dgrad/wgrad kernels run at dense speed (~640 TF)
    — MIOpen's grouped path measured ~4x slower end-to-end on
    resnext50_32x4d.  Numerically exact: zeros cannot contaminate fwd or
    dgrad, and wgrad is an outer product, so slicing the group diagonal
    of the dense dW recovers the grouped gradient bit-for-bit.
    """

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, stride: int,
                groups: int):
        e = require_ext()
        n, cin, h, w = x.shape
        cout = weight.shape[0]
        sg = cin // groups
        xc = x.contiguous(memory_format=torch.channels_last)
        x2d = _rows(xc)
        # dense [G, Sout, 3, 3, G, Sin] with the grouped weight on the
        # (g, :, :, :, g, :) diagonal
        sout = cout // groups
        wg = weight.contiguous(memory_format=torch.channels_last) \
            .permute(0, 2, 3, 1).reshape(groups, sout, 3, 3, sg)
        dense = torch.zeros(groups, sout, 3, 3, groups, sg,
                            dtype=wg.dtype, device=wg.device)
        gi = torch.arange(groups, device=wg.device)
        dense[gi, :, :, :, gi, :] = wg
        w2d = dense.reshape(cout, 3, 3, cin).reshape(cout, 9 * cin)
        # group-diagonal banding: skip the provably-zero K-steps (the
        # 128-channel window of each n-tile) when alignment allows
        banded = (cin == cout and cin % 128 == 0)
        if _fuse_stats_enabled():
            y2d, stats = e.conv3x3_fwd_stats(x2d, n, h, w, stride, w2d,
                                             banded)
        else:
            y2d = e.conv3x3_fwd(x2d, n, h, w, stride, w2d)
            stats = torch.empty(0, device=x.device)
        ctx.save_for_backward(x2d, w2d)
        ctx.meta = (n, cin, h, w, stride, cout, groups, banded)
        ho = (h + 2 - 3) // stride + 1
        wo = (w + 2 - 3) // stride + 1
        y = y2d.view(n, ho, wo, cout).permute(0, 3, 1, 2)
        ctx.mark_non_differentiable(stats)
        return y, stats

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, grad_y: torch.Tensor, _grad_stats=None):
        e = require_ext()
        x2d, w2d = ctx.saved_tensors
        n, cin, h, w, stride, cout, groups, banded = ctx.meta
        sg = cin // groups
        sout = cout // groups
        gy = grad_y.contiguous(memory_format=torch.channels_last)
        gy2d = _rows(gy).to(torch.bfloat16)
        dx2d = e.conv3x3_dgrad(gy2d, n, h, w, stride, w2d, banded)
        dx = dx2d.view(n, h, w, cin).permute(0, 3, 1, 2)
        if banded and _wgrad2_enabled():
            # compact band [Cout, 9, 128]: extract the group diagonal of
            # each 128-channel window
            dwb = e.tn2_wgrad_banded(gy2d, x2d, n, h, w, stride) \
                .view(cout // 128, 128 // sg, sg, 9, 128 // sg, sg)
            bi = torch.arange(cout // 128, device=dwb.device)[:, None]
            gi = torch.arange(128 // sg, device=dwb.device)[None, :]
            dw = dwb[bi, gi, :, :, gi, :].reshape(cout, 9, sg) \
                .view(cout, 3, 3, sg).permute(0, 3, 1, 2) \
                .contiguous(memory_format=torch.channels_last)
            return dx, dw, None, None
        dw2d = e.tn2_wgrad(gy2d, x2d, 9, n, h, w, stride, 2) \
            if _wgrad2_enabled() \
            else e.conv3x3_wgrad(gy2d, x2d, n, h, w, stride)
        # slice the group diagonal of the dense dW
        dwd = dw2d.view(groups, sout, 3, 3, groups, sg)
        gi = torch.arange(groups, device=dwd.device)
        dw = dwd[gi, :, :, :, gi, :].permute(0, 1, 4, 2, 3) \
            .reshape(cout, sg, 3, 3) \
            .contiguous(memory_format=torch.channels_last)
        return dx, dw, None, None


def conv3x3_grouped_mfma(x: torch.Tensor, weight: torch.Tensor,
                         stride: int, groups: int) -> torch.Tensor:
    y, stats = _ConvGrouped3x3.apply(x, weight, stride, groups)
    if stats.numel():
        y._amdtrain_bn_stats = stats
    return y


class _ConvGeneric(torch.autograd.Function):
    """Generic implicit-GEMM conv (element-gather im2col) — serves the 7x7
    stem and any other odd configuration.  Input gradient gathers dY
    through the transposed-conv map (conv_generic_dgrad) when needed."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, stride: int,
                pad: int):
        e = require_ext()
        n, cin, h, w = x.shape
        cout, _, kh, kw = weight.shape
        xc = x.contiguous(memory_format=torch.channels_last)
        x2d = _rows(xc)
        w4 = weight.contiguous(memory_format=torch.channels_last) \
            .permute(0, 2, 3, 1)                       # [Cout, KH, KW, Cin]
        if cin < 8:
            # stem fast path: pad channels to 8 so every 16 B staging unit
            # is one tap — the kernel's vectorized tap-gather (C8) route
            x2d = torch.nn.functional.pad(x2d, (0, 8 - cin))
            w4 = torch.nn.functional.pad(w4, (0, 8 - cin))
            cin_k = 8
        else:
            cin_k = cin
        k = kh * kw * cin_k
        kpad = (k + 31) // 32 * 32
        w2 = w4.reshape(cout, k)
        if kpad != k:
            w2 = torch.nn.functional.pad(w2, (0, kpad - k))
        y2d = e.conv_generic_fwd(x2d, n, h, w, kh, kw, stride, pad, w2)
        ctx.save_for_backward(x2d, weight)
        ctx.meta = (n, cin, h, w, kh, kw, stride, pad, cout, k, cin_k)
        ho = (h + 2 * pad - kh) // stride + 1
        wo = (w + 2 * pad - kw) // stride + 1
        return y2d.view(n, ho, wo, cout).permute(0, 3, 1, 2)

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, grad_y: torch.Tensor):
        e = require_ext()
        x2d, weight = ctx.saved_tensors
        n, cin, h, w, kh, kw, stride, pad, cout, k, cin_k = ctx.meta
        gy2d = _rows(grad_y.contiguous(memory_format=torch.channels_last)) \
            .to(torch.bfloat16)
        dx = None
        if ctx.needs_input_grad[0]:
            # weight permuted [Cin, KH*KW*Cout] + K2 padding
            k2 = kh * kw * cout
            k2p = (k2 + 31) // 32 * 32
            w2p = weight.to(torch.bfloat16).permute(1, 2, 3, 0) \
                .reshape(cin, k2)
            if k2p != k2:
                w2p = torch.nn.functional.pad(w2p, (0, k2p - k2))
            dx2d = e.conv_generic_dgrad(gy2d, w2p.contiguous(), n, h, w,
                                        kh, kw, stride, pad)
            dx = dx2d.view(n, h, w, cin).permute(0, 3, 1, 2)
        if _wgrad2_enabled() and cout % 8 == 0:
            # tap-gather TN core; x2d is already channel-padded (fwd)
            dwp = e.tn2_wgrad(gy2d, x2d, kh * kw, n, h, w, stride, 2,
                              kh, kw, pad)               # [Cout, taps*cin_k]
            dw = dwp.view(cout, kh * kw, cin_k)[:, :, :cin] \
                .view(cout, kh, kw, cin).permute(0, 3, 1, 2) \
                .contiguous(memory_format=torch.channels_last)
        else:
            dw2 = e.conv_generic_wgrad(gy2d, x2d, n, h, w, kh, kw, stride,
                                       pad)
            dw = dw2[:, :k].view(cout, kh * kw, cin_k)[:, :, :cin] \
                .reshape(cout, kh, kw, cin).permute(0, 3, 1, 2) \
                .contiguous(memory_format=torch.channels_last)
        return dx, dw, None, None


def conv_stem_mfma(x: torch.Tensor, weight: torch.Tensor, stride: int,
                   pad: int) -> torch.Tensor:
    return _ConvGeneric.apply(x, weight, stride, pad)


def _conv1x1_env_default() -> str:
    return os.environ.get("AMDTRAIN_CONV1X1", "custom")


class AmdConv2d(nn.Conv2d):
    """nn.Conv2d dispatching to the hand-written gfx950 kernels.

    State dict / init are identical to nn.Conv2d.  On-GPU bf16 paths:
    1x1 -> MFMA GEMM (gemm.hip), 3x3 pad-1 s1/s2 -> implicit GEMM
    (conv3x3.hip), anything else without an input gradient (the 7x7
    stem) -> generic element-gather implicit GEMM (conv_stem.hip).
    Grouped/dilated convs and fp32 inference fall through to MIOpen.
    Env overrides AMDTRAIN_CONV1X1 / AMDTRAIN_CONV3X3 / AMDTRAIN_CONVSTEM
    = "miopen" for A/B measurement.
    """

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # MFMA path is bf16: eligible under autocast or with bf16 inputs;
        # fp32 inference/training falls through to MIOpen
        bf16_ok = (x.dtype == torch.bfloat16
                   or torch.is_autocast_enabled("cuda"))
        if (x.is_cuda and bf16_ok and ext_available()
                and self.bias is None and self.dilation == (1, 1)
                and self.groups > 1):
            # grouped 3x3 (ResNeXt): block-diagonalized dense path
            if (self.kernel_size == (3, 3) and self.padding == (1, 1)
                    and self.stride[0] in (1, 2)
                    and self.in_channels % 32 == 0
                    and self.out_channels % 16 == 0
                    and self.in_channels % self.groups == 0
                    and os.environ.get("AMDTRAIN_CONV3X3G", "custom")
                    == "custom"):
                return conv3x3_grouped_mfma(x, self.weight, self.stride[0],
                                            self.groups)
            return super().forward(x)
        if (x.is_cuda and bf16_ok and ext_available()
                and self.bias is None and self.groups == 1
                and self.dilation == (1, 1)):
            ch_ok = (self.in_channels % 32 == 0
                     and self.out_channels % 16 == 0)
            if (ch_ok and self.kernel_size == (1, 1)
                    and self.padding == (0, 0)
                    and _conv1x1_env_default() == "custom"):
                return conv1x1_mfma(x, self.weight, self.stride[0])
            if (ch_ok and self.kernel_size == (3, 3)
                    and self.padding == (1, 1)
                    and self.stride[0] in (1, 2)
                    and os.environ.get("AMDTRAIN_CONV3X3", "custom")
                    == "custom"):
                return conv3x3_mfma(x, self.weight, self.stride[0])
            if (self.out_channels % 16 == 0
                    and self.stride[0] == self.stride[1]
                    and self.padding[0] == self.padding[1]
                    and os.environ.get("AMDTRAIN_CONVSTEM", "custom")
                    == "custom"):
                # generic element-gather path (stem + odd shapes; full
                # fwd/dgrad/wgrad coverage)
                return conv_stem_mfma(x, self.weight, self.stride[0],
                                      self.padding[0])
        return super().forward(x)
