"""Shared train/validate loops (L2) used by all six entrypoints.

Replaces the reference's six copy-pasted ``train``/``validate`` functions
(e.g. distributed.py:228-324) with one parameterized pair, preserving:
  * batch_time / data_time / losses / top1 / top5 AverageMeters and the
    ProgressMeter line every ``--print-freq`` batches,
  * per-iteration synchronized metric reduction — but fused into ONE
    4-element all-reduce instead of the reference's barrier + 3 scalar
    all-reduces (distributed.py:256-260),
  * the final ``* Acc@1 ... Acc@5 ...`` summary line (distributed.py:321-322).

Mixed precision: when an Apex-style handle is installed on the optimizer
(``parallel.amp.initialize``), backward runs through ``amp.scale_loss``;
otherwise an autocast context is applied per ``args.dtype``.
"""

from __future__ import annotations

import contextlib
import dataclasses
import time
from typing import Optional

import torch

from ..comm.collectives import MetricReducer
from ..parallel import amp as amp_mod
from ..utils.meters import AverageMeter, ProgressMeter
from ..utils.metrics import accuracy


@dataclasses.dataclass
class TrainState:
    device: torch.device
    world_size: int = 1
    rank: int = 0
    reduce_metrics: bool = True      # slurm-style sets False
    channels_last: bool = True
    autocast_dtype: Optional[torch.dtype] = None  # None => fp32
    data_dtype: Optional[torch.dtype] = None      # activation input dtype
    print_freq: int = 10
    max_steps: int = 0               # 0 = full epoch

    def input_dtype(self) -> torch.dtype:
        return self.data_dtype or self.autocast_dtype or torch.float32


def _autocast(state: TrainState, model):
    if getattr(model, "_amp_handle", None) is not None:
        # amp.initialize already wrapped forward (O1) or halved the model (O2)
        return contextlib.nullcontext()
    if state.autocast_dtype is not None and state.device.type == "cuda":
        return torch.autocast("cuda", dtype=state.autocast_dtype)
    return contextlib.nullcontext()


def _to_device(batch, state: TrainState):
    images, target = batch
    if images.device != state.device:
        images = images.to(state.device, non_blocking=True)
        target = target.to(state.device, non_blocking=True)
    if state.channels_last and images.dim() == 4 and images.device.type == "cuda":
        images = images.contiguous(memory_format=torch.channels_last)
    if images.dtype == torch.uint8:
        from ..ops import functional as OF
        images = OF.normalize_u8(images, dtype=state.input_dtype())
    return images, target


def _zero_grad(model, optimizer) -> None:
    if hasattr(model, "reducer"):
        model.zero_grad()          # NativeDDP: zero the bucket buffers
    elif hasattr(optimizer, "reducer"):
        optimizer.zero_grad()      # Horovod-style DistributedOptimizer
    else:
        optimizer.zero_grad(set_to_none=False)


def _step(optimizer) -> None:
    optimizer.step()


def train(loader, model, criterion, optimizer, epoch: int,
          state: TrainState) -> float:
    batch_time = AverageMeter("Time", ":6.3f")
    data_time = AverageMeter("Data", ":6.3f")
    losses = AverageMeter("Loss", ":.4e")
    top1 = AverageMeter("Acc@1", ":6.2f")
    top5 = AverageMeter("Acc@5", ":6.2f")
    try:
        nb = len(loader)
    except TypeError:
        nb = 0
    progress = ProgressMeter(nb, [batch_time, data_time, losses, top1, top5],
                             prefix=f"Epoch: [{epoch}]")
    reducer = MetricReducer(3, state.device) \
        if (state.reduce_metrics and state.world_size > 1) else None

    model.train()
    amp_handle = getattr(optimizer, "_amp_handle", None)

    end = time.time()
    for i, batch in enumerate(loader):
        if state.max_steps and i >= state.max_steps:
            break
        images, target = _to_device(batch, state)
        data_time.update(time.time() - end)

        with _autocast(state, model):
            output = model(images)
        loss = criterion(output, target)

        with torch.no_grad():
            acc1, acc5 = accuracy(output.float(), target, topk=(1, 5))
        if reducer is not None:
            vals = reducer.reduce([loss.detach(), acc1[0], acc5[0]])
            lv, a1, a5 = reducer.items()
        else:
            lv, a1, a5 = loss.item(), acc1.item(), acc5.item()
        n = images.size(0)
        losses.update(lv, n)
        top1.update(a1, n)
        top5.update(a5, n)

        _zero_grad(model, optimizer)
        if amp_handle is not None:
            with amp_mod.scale_loss(loss, optimizer) as scaled:
                scaled.backward()
        else:
            loss.backward()
        _step(optimizer)

        batch_time.update(time.time() - end)
        end = time.time()

        if i % state.print_freq == 0 and state.rank == 0:
            progress.display(i)
    return losses.avg


def validate(loader, model, criterion, state: TrainState,
             prefix: str = "Test: ") -> float:
    batch_time = AverageMeter("Time", ":6.3f")
    losses = AverageMeter("Loss", ":.4e")
    top1 = AverageMeter("Acc@1", ":6.2f")
    top5 = AverageMeter("Acc@5", ":6.2f")
    try:
        nb = len(loader)
    except TypeError:
        nb = 0
    progress = ProgressMeter(nb, [batch_time, losses, top1, top5],
                             prefix=prefix)
    reducer = MetricReducer(3, state.device) \
        if (state.reduce_metrics and state.world_size > 1) else None

    model.eval()
    with torch.no_grad():
        end = time.time()
        for i, batch in enumerate(loader):
            if state.max_steps and i >= state.max_steps:
                break
            images, target = _to_device(batch, state)
            with _autocast(state, model):
                output = model(images)
            loss = criterion(output, target)
            acc1, acc5 = accuracy(output.float(), target, topk=(1, 5))
            if reducer is not None:
                reducer.reduce([loss.detach(), acc1[0], acc5[0]])
                lv, a1, a5 = reducer.items()
            else:
                lv, a1, a5 = loss.item(), acc1.item(), acc5.item()
            n = images.size(0)
            losses.update(lv, n)
            top1.update(a1, n)
            top5.update(a5, n)
            batch_time.update(time.time() - end)
            end = time.time()
            if i % state.print_freq == 0 and state.rank == 0:
                progress.display(i)
    if state.rank == 0:
        print(f" * Acc@1 {top1.avg:.3f} Acc@5 {top5.avg:.3f}", flush=True)
    return top1.avg
