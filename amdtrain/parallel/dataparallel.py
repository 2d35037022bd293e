"""Single-process multi-GPU scatter/gather data parallelism.

MI355X-native equivalent of ``nn.DataParallel`` as the reference uses it
(dataparallel.py:138, forward at :246; SURVEY §2b "nn.DataParallel
primitives"): per step — replicate parameters to every GPU (P2P copies over
xGMI), scatter the batch along dim 0, run one forward per GPU on Python
threads, gather logits to the output device, and reduce gradients back to
the master replica during backward.

This launch style is kept for capability parity; it is documented-slow
(3.48x vs DDP in the reference's own benchmark) because of the
per-step replicate/gather and the GIL-serialized threaded launch.  The
replicate/reduce copies are coalesced into one flat buffer per dtype per
device so each step issues a few large xGMI P2P transfers instead of
hundreds of small ones.

Implementation notes:
  * replica[0] IS the master module when it already lives on devices[0]
    (torch semantics: BN running stats update through replica 0).
  * gradient reduction to the master is queued as an autograd final
    callback from the gather node, so ``loss.backward()`` leaves master
    ``param.grad`` populated exactly like the reference path.
"""

from __future__ import annotations

import copy
import threading
from typing import Dict, List, Optional, Sequence

import torch
import torch.nn as nn


class _Gather(torch.autograd.Function):
    """Concatenate per-device outputs on the output device; backward splits
    the gradient and routes each slice back to its source device."""

    @staticmethod
    def forward(ctx, output_device: torch.device, *chunks: torch.Tensor):
        ctx.devices = [c.device for c in chunks]
        ctx.sizes = [c.size(0) for c in chunks]
        return torch.cat([c.to(output_device) for c in chunks], dim=0)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        grads = []
        off = 0
        for dev, n in zip(ctx.devices, ctx.sizes):
            grads.append(grad_out[off:off + n].to(dev))
            off += n
        return (None, *grads)


class ScatterGatherDataParallel(nn.Module):
    def __init__(self, module: nn.Module, device_ids: Sequence[int],
                 output_device: Optional[int] = None):
        super().__init__()
        self.devices = [torch.device(f"cuda:{d}") for d in device_ids]
        self.output_device = torch.device(
            f"cuda:{output_device if output_device is not None else device_ids[0]}")
        self.module = module
        self._replicas: List[nn.Module] = []
        self._master_params = [p for p in module.parameters() if p.requires_grad]
        self._build_replicas()

    # -- replica management ------------------------------------------------

    def _build_replicas(self) -> None:
        master_dev = next(self.module.parameters()).device
        for dev in self.devices:
            if dev == master_dev:
                self._replicas.append(self.module)
            else:
                rep = copy.deepcopy(self.module).to(dev)
                for p in rep.parameters():
                    p.requires_grad_(True)
                self._replicas.append(rep)

    @torch.no_grad()
    def _sync_replicas(self) -> None:
        """Coalesced master->replica parameter+buffer copy (one flat P2P
        transfer per dtype per device — the 'replicate broadcast')."""
        srcs = list(self.module.state_dict().values())
        by_dtype: Dict[torch.dtype, List[torch.Tensor]] = {}
        for t in srcs:
            if isinstance(t, torch.Tensor) and t.numel():
                by_dtype.setdefault(t.dtype, []).append(t)
        flats = {dt: torch.cat([t.reshape(-1) for t in ts])
                 for dt, ts in by_dtype.items()}
        for rep in self._replicas:
            if rep is self.module:
                continue
            dsts = list(rep.state_dict().values())
            dst_by_dtype: Dict[torch.dtype, List[torch.Tensor]] = {}
            for t in dsts:
                if isinstance(t, torch.Tensor) and t.numel():
                    dst_by_dtype.setdefault(t.dtype, []).append(t)
            for dt, ts in dst_by_dtype.items():
                flat = flats[dt].to(ts[0].device)
                off = 0
                for t in ts:
                    t.reshape(-1).copy_(flat[off:off + t.numel()])
                    off += t.numel()

    @torch.no_grad()
    def _reduce_gradients(self) -> None:
        """Sum replica gradients into master param.grad (coalesced P2P)."""
        master_dev = next(self.module.parameters()).device
        master = self._master_params
        acc = [None] * len(master)
        for rep in self._replicas:
            if rep is self.module:
                continue
            reps = [p for p in rep.parameters() if p.requires_grad]
            grads = [p.grad for p in reps]
            if all(g is None for g in grads):
                continue
            flat = torch.cat([
                (g if g is not None else torch.zeros_like(p)).reshape(-1)
                for g, p in zip(grads, reps)]).to(master_dev)
            off = 0
            for i, p in enumerate(master):
                n = p.numel()
                piece = flat[off:off + n].view_as(p)
                acc[i] = piece if acc[i] is None else acc[i] + piece
                off += n
            for p in reps:
                p.grad = None
        for p, extra in zip(master, acc):
            if extra is None:
                continue
            if p.grad is None:
                p.grad = extra.clone()
            else:
                p.grad.add_(extra)

    # -- forward -----------------------------------------------------------

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if len(self.devices) == 1:
            return self.module(x.to(self.devices[0]))

        self._sync_replicas()
        chunks = x.chunk(len(self.devices), dim=0)
        chunks = [c.to(dev) for c, dev in zip(chunks, self.devices)]

        outputs: List[Optional[torch.Tensor]] = [None] * len(chunks)
        errors: List[Optional[BaseException]] = [None] * len(chunks)
        grad_on = torch.is_grad_enabled()
        ac_on = torch.is_autocast_enabled("cuda")
        ac_dtype = torch.get_autocast_dtype("cuda") if ac_on else None

        def worker(i: int) -> None:
            try:  # grad/autocast modes are thread-local: re-apply the caller's
                dev = self.devices[i]
                with torch.set_grad_enabled(grad_on), torch.cuda.device(dev):
                    if ac_on:
                        with torch.autocast("cuda", dtype=ac_dtype):
                            outputs[i] = self._replicas[i](chunks[i])
                    else:
                        outputs[i] = self._replicas[i](chunks[i])
            except BaseException as e:  # re-raised on the main thread
                errors[i] = e

        threads = [threading.Thread(target=worker, args=(i,))
                   for i in range(len(chunks))]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        for e in errors:
            if e is not None:
                raise e

        out = _Gather.apply(self.output_device, *outputs)
        if torch.is_grad_enabled() and out.requires_grad:
            out.register_hook(self._queue_grad_reduction)
        return out

    def _queue_grad_reduction(self, grad):
        torch.autograd.Variable._execution_engine.queue_callback(
            self._reduce_gradients)
        return grad

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, state_dict, strict: bool = True):
        return self.module.load_state_dict(state_dict, strict=strict)
