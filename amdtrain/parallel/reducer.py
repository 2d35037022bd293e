"""Bucketed gradient all-reduce overlapped with backward, tuned for xGMI.

MI355X-native replacement for the torch DDP C++ reducer the reference invokes
implicitly through ``DistributedDataParallel`` + ``loss.backward()``
(reference distributed.py:147-148, 268; SURVEY §2b "torch DDP C++ reducer").

Design (xGMI-first, SURVEY §5 comm-backend plan):
  * Gradients live as VIEWS into per-bucket flat buffers — autograd
    accumulates directly into the bucket, so there is no flatten copy before
    the collective and no scatter copy after (zero-copy bucketing).
  * Buckets are filled in REVERSE parameter order (the order backward
    produces gradients), so the first all-reduce launches while most of
    backward is still running.
  * Each full bucket triggers one async RCCL all-reduce.  ProcessGroupNCCL
    runs the collective on its own HIP stream ordered after the compute
    stream, so communication overlaps the remaining backward.  On the 8-GPU
    MI355X node each GPU has 7 point-to-point xGMI links (~153 GB/s each);
    ring all-reduce is per-link bound, so the default bucket is sized large
    (50 MB) to amortize per-collective latency while still splitting the
    ResNet-50 102 MB gradient set into overlappable pieces.
  * Averaging uses RCCL ReduceOp.AVG on GPU (one fused op); SUM + divide on
    gloo (CPU tests).
  * Optional gradient compression ("bf16"/"fp16"): the bucket is cast to the
    half dtype before the collective and back after — the Horovod
    fp16-compression path (horovod_distributed.py:159) at half the xGMI bytes.
  * End-of-backward is detected with an autograd final callback, so user code
    needs no explicit ``finish()`` call: by the time ``optimizer.step()``
    runs, the compute stream has been made to wait on every collective.
"""

from __future__ import annotations

import contextlib
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist


class _Bucket:
    __slots__ = ("params", "flat", "comm_buf", "ready", "work", "index")

    def __init__(self, index: int, params: List[torch.nn.Parameter],
                 flat: torch.Tensor, comm_buf: Optional[torch.Tensor]):
        self.index = index
        self.params = params
        self.flat = flat
        self.comm_buf = comm_buf  # half-precision staging when compressing
        self.ready = 0
        self.work = None


class BucketedReducer:
    def __init__(self, params: Sequence[torch.nn.Parameter],
                 bucket_cap_mb: float = 50.0,
                 compression: str = "none",
                 process_group=None,
                 average: bool = True):
        assert compression in ("none", "bf16", "fp16")
        self.pg = process_group
        self.compression = compression
        self.average = average
        self.params = [p for p in params if p.requires_grad]
        self._grad_accum = False  # no_sync mode
        self._cb_queued = False
        self._launch_order: List[_Bucket] = []
        # observability: how many bucket all-reduces were ENQUEUED while
        # backward was still running (vs flushed by the final callback) —
        # the compute/communication overlap the design promises.  Readable
        # after each backward; asserted by tests/test_reducer.py.
        self.last_overlap_launches = 0
        self._build_buckets(bucket_cap_mb)
        self._attach_hooks()

    # -- construction ------------------------------------------------------

    def _build_buckets(self, cap_mb: float) -> None:
        cap = int(cap_mb * 1024 * 1024)
        comm_dtype = {"none": None, "bf16": torch.bfloat16,
                      "fp16": torch.float16}[self.compression]
        self.buckets: List[_Bucket] = []
        self.param_to_bucket: Dict[torch.nn.Parameter, _Bucket] = {}

        # group in reverse parameter order, split by (device, dtype)
        groups: List[List[torch.nn.Parameter]] = []
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        cur_key = None
        for p in reversed(self.params):
            key = (p.device, p.dtype)
            nbytes = p.numel() * p.element_size()
            if cur and (key != cur_key or cur_bytes + nbytes > cap):
                groups.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
            cur_key = key
        if cur:
            groups.append(cur)

        for i, group in enumerate(groups):
            numel = sum(p.numel() for p in group)
            flat = torch.zeros(numel, dtype=group[0].dtype,
                               device=group[0].device)
            comm_buf = None
            if comm_dtype is not None and group[0].dtype != comm_dtype:
                comm_buf = torch.empty(numel, dtype=comm_dtype,
                                       device=group[0].device)
            bucket = _Bucket(i, group, flat, comm_buf)
            off = 0
            for p in group:
                n = p.numel()
                p.grad = flat[off:off + n].view_as(p)
                off += n
                self.param_to_bucket[p] = bucket
            self.buckets.append(bucket)

    def _attach_hooks(self) -> None:
        self._hooks = []
        for p in self.params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready))

    # -- per-iteration machinery ------------------------------------------

    def _on_grad_ready(self, p: torch.nn.Parameter) -> None:
        if self._grad_accum:
            return
        if not self._cb_queued:
            torch.autograd.Variable._execution_engine.queue_callback(
                self._final_callback)
            self._cb_queued = True
            self.last_overlap_launches = 0
        b = self.param_to_bucket[p]
        b.ready += 1
        if b.ready == len(b.params):
            self._launch(b)
            if b.work is not None:
                self.last_overlap_launches += 1

    def _world(self) -> int:
        if not (dist.is_available() and dist.is_initialized()):
            return 1
        return dist.get_world_size(self.pg)

    def _launch(self, b: _Bucket) -> None:
        world = self._world()
        if world == 1:
            return
        buf = b.flat
        if b.comm_buf is not None:
            b.comm_buf.copy_(b.flat)  # fp32 -> bf16/fp16 compress
            buf = b.comm_buf
        backend_nccl = dist.get_backend(self.pg) == "nccl"
        if self.average and backend_nccl:
            op = dist.ReduceOp.AVG
        else:
            op = dist.ReduceOp.SUM
        b.work = dist.all_reduce(buf, op=op, group=self.pg, async_op=True)
        self._launch_order.append(b)

    def _final_callback(self) -> None:
        """Autograd end-of-backward: flush stragglers, wait all collectives,
        decompress, divide where the backend lacks AVG."""
        self._cb_queued = False
        world = self._world()
        for b in self.buckets:
            if b.work is None and b.ready > 0 and world > 1:
                self._launch(b)  # partially-ready bucket (unused params)
            b.ready = 0
        need_div = (self.average and world > 1
                    and dist.get_backend(self.pg) != "nccl")
        for b in self._launch_order:
            if b.work is not None:
                b.work.wait()
                b.work = None
            if b.comm_buf is not None:
                if need_div:
                    b.comm_buf.div_(world)
                b.flat.copy_(b.comm_buf)  # decompress
            elif need_div:
                b.flat.div_(world)
        self._launch_order.clear()

    # -- public API --------------------------------------------------------

    @contextlib.contextmanager
    def no_sync(self):
        """Skip gradient synchronization (gradient accumulation steps)."""
        prev = self._grad_accum
        self._grad_accum = True
        try:
            yield
        finally:
            self._grad_accum = prev

    def zero_grad(self) -> None:
        """Zero all bucket buffers (== zeroing every param.grad view) and
        reset per-iteration state (robust to an aborted backward)."""
        for b in self.buckets:
            b.flat.zero_()
            b.ready = 0
            b.work = None
        self._launch_order.clear()
        self._cb_queued = False

    def grad_buffers(self) -> List[torch.Tensor]:
        return [b.flat for b in self.buckets]

    def detach(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks = []
