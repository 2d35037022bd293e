"""Hand-written FC classifier layer over the gfx950 MFMA GEMM kernels.

The reference's one remaining library GEMM is the [B,2048]x[2048,1000]
classifier (SURVEY §2c "Linear / GEMM (cuBLAS)"; reference forward at
distributed.py:250).  This replaces it with the in-tree kernels:
  * forward: gemm_bt with a fused per-column fp32 bias epilogue
  * dgrad:   gemm_bt over the transposed weight, with the 1000-class K
             padded to 1024 (gemm_bt needs K % 32 == 0)
  * wgrad:   tn2_wgrad (the tr-read TN core in wgrad.hip)
  * dbias:   column sum (tiny: [B,1000] fp32 reduce)
``AmdLinear`` is a state_dict-compatible nn.Linear that dispatches to this
path on GPU bf16/autocast and falls back to F.linear otherwise.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import ext_available, require_ext


class _LinearFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, bias):
        e = require_ext()
        xc = x.contiguous()
        w = weight.contiguous()
        y = e.gemm_bt(xc, w, False, None, bias)
        ctx.save_for_backward(xc, w)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, grad_y: torch.Tensor):
        e = require_ext()
        x, w = ctx.saved_tensors
        gy = grad_y.contiguous().to(torch.bfloat16)
        n_out = w.shape[0]
        npad = (n_out + 31) // 32 * 32
        gyp = gy if npad == n_out else F.pad(gy, (0, npad - n_out))
        wT = e.transpose_2d(w)                       # [in, out]
        wTp = wT if npad == n_out else F.pad(wT, (0, npad - n_out))
        dx = e.gemm_bt(gyp, wTp, False)              # [B, in] bf16
        dw = e.tn2_wgrad(gy, x)                      # [out, in] fp32
        db = gy.float().sum(dim=0) if ctx.has_bias else None
        return dx, dw, db


class AmdLinear(nn.Linear):
    """nn.Linear dispatching to the hand-written gfx950 GEMM path.

    State dict / init identical to nn.Linear.  GPU + extension + bf16 (or
    autocast) + in_features % 32 == 0 takes the custom kernels; anything
    else falls through to F.linear.
    """

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        bf16_ok = (x.dtype == torch.bfloat16
                   or torch.is_autocast_enabled("cuda"))
        if (x.is_cuda and bf16_ok and ext_available()
                and x.dim() == 2 and self.in_features % 32 == 0
                and self.out_features % 8 == 0):
            return _LinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)
