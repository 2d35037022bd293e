"""NativeDDP — multi-process data parallelism over the xGMI-tuned reducer.

MI355X-native equivalent of ``torch.nn.parallel.DistributedDataParallel`` as
the reference wraps it (distributed.py:147-148): one process per GPU, rank-0
parameter/buffer broadcast at construction, bucketed gradient all-reduce
overlapped with backward.  Gradient sync completes (stream-ordered) before
``optimizer.step()`` thanks to the reducer's autograd final callback — no
explicit ``finish()`` call is needed, same usage contract as torch DDP.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..comm.collectives import broadcast_module_state
from ..comm.init import is_distributed
from .reducer import BucketedReducer


class NativeDDP(nn.Module):
    def __init__(self, module: nn.Module,
                 bucket_cap_mb: float = 50.0,
                 broadcast_buffers: bool = True,
                 compression: str = "none",
                 process_group=None):
        super().__init__()
        self.module = module
        self.broadcast_buffers = broadcast_buffers
        # replicas start identical: rank-0 broadcast of params + buffers
        # (torch-DDP construction broadcast)
        broadcast_module_state(module, src=0)
        amp_handle = getattr(module, "_amp_handle", None)
        if amp_handle is not None and amp_handle.master_params:
            # O2 masters were cloned before the broadcast — re-sync them
            amp_handle.resync_masters()
        self.reducer = BucketedReducer(
            list(module.parameters()), bucket_cap_mb=bucket_cap_mb,
            compression=compression, process_group=process_group)
        self._flat_buffers = self._flatten_buffers(module)

    @staticmethod
    @torch.no_grad()
    def _flatten_buffers(module):
        """Re-home same-dtype module buffers (BN running stats etc.) as views
        into one flat tensor per dtype, so the per-forward rank-0 broadcast
        is a single RCCL collective with no pack/unpack copies."""
        groups = {}
        for mod in module.modules():
            for name, buf in list(mod._buffers.items()):
                if buf is None or buf.numel() == 0:
                    continue
                groups.setdefault(buf.dtype, []).append((mod, name, buf))
        flats = []
        for dtype, items in groups.items():
            flat = torch.empty(sum(b.numel() for _, _, b in items),
                               dtype=dtype, device=items[0][2].device)
            off = 0
            for mod, name, buf in items:
                n = buf.numel()
                view = flat[off:off + n].view_as(buf)
                view.copy_(buf)
                mod._buffers[name] = view
                off += n
            flats.append(flat)
        return flats

    def forward(self, *args, **kwargs):
        if self.training and self.broadcast_buffers and is_distributed():
            # rank-0 buffer broadcast each forward (torch-DDP semantics for
            # BN running stats) — one flat broadcast per dtype
            for flat in self._flat_buffers:
                torch.distributed.broadcast(flat, src=0)
        return self.module(*args, **kwargs)

    def zero_grad(self, set_to_none: bool = False):  # type: ignore[override]
        # grads are bucket views — zero the flat buffers instead of detaching
        self.reducer.zero_grad()

    def no_sync(self):
        return self.reducer.no_sync()

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, state_dict, strict: bool = True):
        return self.module.load_state_dict(state_dict, strict=strict)
