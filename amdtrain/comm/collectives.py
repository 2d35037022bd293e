"""Metric/state collectives over RCCL.

The reference pays a per-iteration latency tax of 1 ``dist.barrier()`` + 3
separate 1-element all-reduces for (loss, acc1, acc5)
(distributed.py:256-260).  On xGMI those are four serialized ring latencies.
``MetricReducer`` keeps the same semantics (synchronized, mean-reduced
metrics every iteration) but folds them into ONE fused k-element all-reduce
(SURVEY §5 "comm backend" plan item iv).

``reduce_mean`` is kept as the verbatim-parity helper (distributed.py:105-109).
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from .init import is_distributed


def reduce_mean(tensor: torch.Tensor, nprocs: Optional[int] = None) -> torch.Tensor:
    """clone -> all_reduce(SUM) -> /nprocs (reference distributed.py:105-109)."""
    if not is_distributed():
        return tensor.clone()
    nprocs = nprocs or dist.get_world_size()
    rt = tensor.clone()
    dist.all_reduce(rt, op=dist.ReduceOp.SUM)
    rt.div_(nprocs)
    return rt


def barrier() -> None:
    if is_distributed():
        dist.barrier()


class MetricReducer:
    """Fused mean-reduction of k scalar metrics in one all-reduce.

    Replaces the reference's barrier + 3 scalar all-reduces per iteration
    with a single k-element all-reduce (the all-reduce itself is the sync
    point).  Values stay on ``device`` to avoid host round-trips until
    ``.items()`` is called.
    """

    def __init__(self, k: int, device: torch.device):
        self.k = k
        self.buf = torch.zeros(k, dtype=torch.float32, device=device)

    def reduce(self, values: Sequence[torch.Tensor]) -> torch.Tensor:
        """values: k scalar tensors -> [k] tensor of global means."""
        assert len(values) == self.k
        for i, v in enumerate(values):
            self.buf[i] = v.detach().reshape(())
        if is_distributed():
            dist.all_reduce(self.buf, op=dist.ReduceOp.SUM)
            self.buf.div_(dist.get_world_size())
        return self.buf

    def items(self) -> List[float]:
        return self.buf.tolist()  # one D2H sync for all k metrics


@torch.no_grad()
def broadcast_module_state(module: torch.nn.Module, src: int = 0) -> None:
    """Broadcast all parameters AND buffers from ``src`` so replicas start
    identical (torch-DDP construction broadcast / hvd.broadcast_parameters
    parity).  Tensors are coalesced into one flat buffer per
    dtype to issue few large RCCL broadcasts instead of hundreds of tiny ones.
    """
    if not is_distributed():
        return
    tensors = [t for t in module.state_dict().values()
               if isinstance(t, torch.Tensor) and t.numel() > 0]
    _broadcast_coalesced(tensors, src)


def _broadcast_coalesced(tensors: List[torch.Tensor], src: int,
                         bucket_bytes: int = 256 << 20) -> None:
    by_dtype = {}
    for t in tensors:
        by_dtype.setdefault(t.dtype, []).append(t)
    rank = dist.get_rank()
    for dtype, group in by_dtype.items():
        flat = torch.empty(sum(t.numel() for t in group), dtype=dtype,
                           device=group[0].device)
        off = 0
        for t in group:
            flat[off:off + t.numel()].copy_(t.reshape(-1))
            off += t.numel()
        dist.broadcast(flat, src=src)
        if rank != src:
            off = 0
            for t in group:
                t.reshape(-1).copy_(flat[off:off + t.numel()])
                off += t.numel()


@torch.no_grad()
def broadcast_optimizer_state(optimizer: torch.optim.Optimizer,
                              src: int = 0,
                              device: Optional[torch.device] = None) -> None:
    """hvd.broadcast_optimizer_state parity (horovod_distributed.py:158):
    rank ``src``'s optimizer state dict replaces every other rank's.

    The structured state dict (hyperparameters + any lazily-created tensor
    state such as momentum buffers) is shipped once via an object broadcast —
    this happens a single time at startup, before the hot loop, so the
    pickle path is fine; steady-state gradient traffic stays on RCCL.
    """
    if not is_distributed():
        return
    rank = dist.get_rank()
    payload = [optimizer.state_dict() if rank == src else None]
    dist.broadcast_object_list(payload, src=src)
    if rank != src:
        sd = payload[0]
        if device is not None:
            for st in sd.get("state", {}).values():
                for k, v in st.items():
                    if isinstance(v, torch.Tensor):
                        st[k] = v.to(device)
        optimizer.load_state_dict(sd)
