"""Apex-style automatic mixed precision for MI355X.

MI355X-native equivalent of the Apex AMP machinery the reference invokes
(``amp.initialize(model, optimizer)`` at apex_distributed.py:216 and
``with amp.scale_loss(loss, optimizer) as scaled: scaled.backward()`` at
apex_distributed.py:328-329; SURVEY §2b "Apex AMP C++/CUDA").

Two opt levels, mirroring Apex's:
  * O1 — autocast patching: the forward runs under ``torch.autocast`` in the
    chosen half dtype (bf16 by default on gfx950 — native MFMA rate, no loss
    scaling required; fp16 selectable), params/grads stay fp32.
  * O2 — half model + fp32 master weights: parameters are cast to the half
    dtype (BatchNorm kept fp32), the optimizer steps fp32 master copies, and
    the post-step master->model cast plus the grad unscale + inf/nan check
    run as fused multi-tensor HIP kernels (ops/csrc/elementwise.hip).

Dynamic loss scaling matches Apex defaults: init 2**16, backoff x0.5 on
inf/nan (step skipped), growth x2 every 2000 clean steps.
"""

from __future__ import annotations

import contextlib
from typing import List

import torch

from ..ops import functional as OF


class DynamicLossScaler:
    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000,
                 enabled: bool = True):
        self.scale = init_scale if enabled else 1.0
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.enabled = enabled
        self._growth_tracker = 0

    def update(self, found_inf: bool) -> None:
        if not self.enabled:
            return
        if found_inf:
            self.scale = max(self.scale * self.backoff_factor, 1.0)
            self._growth_tracker = 0
        else:
            self._growth_tracker += 1
            if self._growth_tracker >= self.growth_interval:
                self.scale *= self.growth_factor
                self._growth_tracker = 0


class _AmpHandle:
    """Per-(model, optimizer) AMP state installed by ``initialize``."""

    def __init__(self, optimizer, opt_level: str, dtype: torch.dtype,
                 scaler: DynamicLossScaler):
        self.optimizer = optimizer
        self.opt_level = opt_level
        self.dtype = dtype
        self.scaler = scaler
        self.found_inf = False
        self.steps_skipped = 0
        # O2 master-weight state
        self.model_params: List[torch.nn.Parameter] = []
        self.master_params: List[torch.Tensor] = []
        # set by scale_loss when it has already copied half grads into the
        # fp32 masters (so the unscale runs at fp32); step() then skips the
        # cast it would otherwise do
        self.masters_have_grads = False

    @torch.no_grad()
    def resync_masters(self) -> None:
        """Re-copy model half weights into the fp32 masters.  Needed after
        any external overwrite of the model weights (e.g. the DDP rank-0
        broadcast, which runs after ``initialize`` created the masters)."""
        for mp, hp in zip(self.master_params, self.model_params):
            mp.data.copy_(hp.data.to(mp.dtype))


def _needs_scaling(dtype: torch.dtype) -> bool:
    return dtype == torch.float16  # bf16 has fp32's exponent range


def initialize(model: torch.nn.Module, optimizer: torch.optim.Optimizer,
               opt_level: str = "O1", dtype: torch.dtype = torch.bfloat16,
               loss_scale: str = "dynamic", init_scale: float = 2.0 ** 16):
    """Apex ``amp.initialize`` parity.  Returns (model, optimizer); the
    optimizer gains ``_amp_handle`` and its ``step`` is wrapped to honor
    inf/nan skips (O1-fp16/O2) and the O2 master-weight flow."""
    if opt_level not in ("O0", "O1", "O2"):
        raise ValueError(f"unsupported opt_level {opt_level}")

    enabled = opt_level != "O0" and (_needs_scaling(dtype)
                                     if loss_scale == "dynamic" else False)
    scaler = DynamicLossScaler(init_scale=init_scale, enabled=enabled)
    handle = _AmpHandle(optimizer, opt_level, dtype, scaler)

    if opt_level == "O2":
        # half the model (BatchNorm stays fp32 for stable statistics)
        def _half(m: torch.nn.Module):
            for child in m.children():
                _half(child)
            if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                return
            for name, p in m.named_parameters(recurse=False):
                p.data = p.data.to(dtype)
        _half(model)
        # fp32 master copies, one per trainable half param
        for group in optimizer.param_groups:
            new_params = []
            for p in group["params"]:
                if p.dtype == dtype and p.requires_grad:
                    master = p.detach().clone().float()
                    master.requires_grad_(True)
                    handle.model_params.append(p)
                    handle.master_params.append(master)
                    new_params.append(master)
                else:
                    new_params.append(p)
            group["params"] = new_params

    optimizer._amp_handle = handle
    _wrap_step(optimizer, handle)
    model._amp_handle = handle
    if opt_level == "O1":
        _wrap_forward_autocast(model, dtype)
    return model, optimizer


def _wrap_forward_autocast(model: torch.nn.Module, dtype: torch.dtype) -> None:
    inner = model.forward

    def forward(*args, **kwargs):
        with torch.autocast("cuda", dtype=dtype,
                            enabled=torch.cuda.is_available()):
            return inner(*args, **kwargs)
    model.forward = forward


def _wrap_step(optimizer, handle: _AmpHandle) -> None:
    inner_step = optimizer.step
    inner_zero = optimizer.zero_grad

    def zero_grad(set_to_none: bool = False):
        inner_zero(set_to_none=set_to_none)
        if handle.opt_level == "O2":
            with torch.no_grad():
                for hp in handle.model_params:
                    if hp.grad is not None:
                        if set_to_none:
                            hp.grad = None
                        else:
                            hp.grad.zero_()
    optimizer.zero_grad = zero_grad

    def step(closure=None, **kw):
        if handle.found_inf:
            handle.steps_skipped += 1
            handle.found_inf = False
            handle.masters_have_grads = False
            return None  # Apex behavior: skip the step after overflow
        if handle.opt_level == "O2" and handle.master_params:
            if not handle.masters_have_grads:
                _populate_master_grads(handle)
            handle.masters_have_grads = False
            out = inner_step(closure, **kw) if closure is not None else inner_step(**kw)
            # master fp32 -> model half
            with torch.no_grad():
                OF.multi_tensor_cast(handle.master_params, handle.model_params)
            return out
        return inner_step(closure, **kw) if closure is not None else inner_step(**kw)

    optimizer.step = step


def _populate_master_grads(handle: _AmpHandle) -> None:
    """Copy the model's half grads into the fp32 master grads (one fused
    multi-tensor cast).  Unscaling then happens at fp32 precision, so
    dividing by 2**16 cannot flush small fp16 gradients to zero."""
    src, dst = [], []
    for mp, hp in zip(handle.master_params, handle.model_params):
        if hp.grad is None:
            mp.grad = None
            continue
        if mp.grad is None:
            mp.grad = torch.empty_like(mp)
        src.append(hp.grad)
        dst.append(mp.grad)
    OF.multi_tensor_cast(src, dst)


@contextlib.contextmanager
def scale_loss(loss: torch.Tensor, optimizer):
    """``with amp.scale_loss(loss, optimizer) as scaled: scaled.backward()``
    (apex_distributed.py:328-329).  Scales the loss up before backward,
    unscales gradients + checks inf/nan after, and updates the dynamic scale.

    In the O2 path the half model grads are first copied into the fp32
    master grads and the unscale + inf check runs on those (Apex O2
    semantics — unscaling in fp16 would reintroduce the underflow loss
    scaling exists to prevent); step() consumes the populated masters.
    """
    handle: _AmpHandle = optimizer._amp_handle
    scaler = handle.scaler
    yield loss * scaler.scale if scaler.scale != 1.0 else loss

    # after backward: unscale grads in-place + detect overflow
    if handle.opt_level == "O2" and handle.master_params:
        _populate_master_grads(handle)
        handle.masters_have_grads = True
        grads = [mp.grad for mp in handle.master_params if mp.grad is not None]
        # any non-master (fp32, e.g. BatchNorm) params still need unscaling
        masters = set(id(mp) for mp in handle.master_params)
        grads += [p.grad for g in optimizer.param_groups for p in g["params"]
                  if p.grad is not None and id(p) not in masters]
    else:
        grads = [p.grad for g in optimizer.param_groups for p in g["params"]
                 if p.grad is not None]
    if not grads:
        handle.found_inf = False
        return
    dev = grads[0].device
    found = torch.zeros(1, dtype=torch.float32, device=dev)
    if scaler.scale != 1.0 or scaler.enabled:
        OF.multi_tensor_scale_check(grads, 1.0 / scaler.scale, found)
        found_inf = bool(found.item())
    else:
        found_inf = False
    handle.found_inf = found_inf
    scaler.update(found_inf)
