"""Shared per-GPU worker used by the distributed entrypoints.

The reference's six scripts are one vertical copy-paste each;
here the shared 90% lives once and each entrypoint contributes only its
launch/bootstrap + wiring quirks (which are preserved deliberately, e.g.
the apex variant's unsharded validation set).
"""

from __future__ import annotations

import dataclasses
import os
import time
from typing import Optional

import torch

from .. import comm
from ..data.build import build_loaders
from ..data.prefetcher import CudaPrefetcher
from ..engine.loops import TrainState, train, validate
from ..models import build_model
from ..ops import CrossEntropyLoss, FusedSGD
from ..parallel import NativeDDP, DistributedOptimizer, Compression, amp
from ..utils import (adjust_learning_rate, load_checkpoint, save_checkpoint,
                     set_seed)
from ..utils.checkpoint import make_checkpoint_state
from ..utils.csvlog import EpochTimer


@dataclasses.dataclass
class StyleConfig:
    name: str
    wrap: str = "ddp"              # ddp | horovod | none
    amp_level: Optional[str] = None  # None | "O1" | "O2"
    compression: str = "none"
    shard_val: bool = True          # apex ref quirk: False
    reduce_metrics: bool = True     # slurm ref quirk: False
    ckpt_rank0_only: bool = True    # slurm ref quirk: False
    use_prefetcher: bool = False    # apex style: True


_DTYPES = {"fp32": None, "bf16": torch.bfloat16, "fp16": torch.float16}


def select_device(local_rank: int) -> torch.device:
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        return torch.device(f"cuda:{local_rank}")
    return torch.device("cpu")


def run_worker(local_rank: int, nprocs: int, args, style: StyleConfig,
               global_rank: Optional[int] = None) -> float:
    """Train/evaluate on one rank.  Returns best top-1."""
    rank = global_rank if global_rank is not None else local_rank
    device = select_device(local_rank)
    set_seed(args.seed)

    model = build_model(args.arch)
    model = model.to(device)
    channels_last = (device.type == "cuda"
                     and not getattr(args, "no_channels_last", False))
    if channels_last:
        model = model.to(memory_format=torch.channels_last)

    # per-GPU slice of the node batch (reference distributed.py:146)
    world = comm.get_world_size()
    args.batch_size = max(1, args.batch_size // max(nprocs, 1))

    criterion = CrossEntropyLoss().to(device)
    optimizer = FusedSGD(model.parameters(), args.lr,
                         momentum=args.momentum,
                         weight_decay=args.weight_decay)

    autocast_dtype = _DTYPES[args.dtype] if device.type == "cuda" else None

    if style.amp_level:
        model, optimizer = amp.initialize(
            model, optimizer, opt_level=style.amp_level,
            dtype=autocast_dtype or torch.bfloat16)
        autocast_dtype = None  # amp handle drives precision now

    if style.wrap == "ddp" and world > 1:
        model = NativeDDP(model, bucket_cap_mb=args.bucket_mb,
                          compression=style.compression)
    elif style.wrap == "horovod":
        comm.broadcast_module_state(model, src=0)
        comm.broadcast_optimizer_state(optimizer, src=0, device=device)
        optimizer = DistributedOptimizer(
            optimizer, model.named_parameters(),
            compression=style.compression, fusion_mb=args.bucket_mb)

    resume_best = 0.0
    if args.resume:
        ck = load_checkpoint(args.resume, model)
        args.start_epoch = ck.get("epoch", args.start_epoch)
        # restore the running best so the first post-resume epoch doesn't
        # unconditionally overwrite model_best.pth.tar
        resume_best = float(ck.get("best_acc1", 0.0))
    elif args.pretrained and os.path.isfile("checkpoint.pth.tar"):
        # no model-zoo download in this environment: --pretrained loads the
        # local checkpoint (reference loads torchvision zoo weights here)
        load_checkpoint("checkpoint.pth.tar", model)

    train_loader, val_loader, train_sampler, _ = build_loaders(
        args, world_size=world, rank=rank,
        distributed=(world > 1), distributed_val=style.shard_val)

    # activation input dtype: the half dtype for O2 (model weights are
    # half), the autocast dtype otherwise
    data_dtype = None
    if style.amp_level == "O2" and device.type == "cuda":
        data_dtype = _DTYPES[args.dtype] or torch.bfloat16
    state = TrainState(
        device=device, world_size=world, rank=rank,
        reduce_metrics=style.reduce_metrics,
        channels_last=channels_last,
        autocast_dtype=autocast_dtype,
        data_dtype=data_dtype,
        print_freq=args.print_freq,
        max_steps=args.max_steps,
    )

    if args.evaluate:
        return validate(_wrap_loader(val_loader, style, state), model,
                        criterion, state)

    timer = EpochTimer(args.epoch_csv or None) if rank == 0 else EpochTimer(None)
    best_acc1 = resume_best
    for epoch in range(args.start_epoch, args.epochs):
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)
        adjust_learning_rate(optimizer, epoch, args.lr)
        timer.start()
        train(_wrap_loader(train_loader, style, state), model, criterion,
              optimizer, epoch, state)
        acc1 = validate(_wrap_loader(val_loader, style, state), model,
                        criterion, state)
        timer.stop(epoch)

        is_best = acc1 > best_acc1
        best_acc1 = max(acc1, best_acc1)
        if rank == 0 or not style.ckpt_rank0_only:
            save_checkpoint(
                make_checkpoint_state(epoch, args.arch, model, best_acc1),
                is_best)
    timer.close()
    return best_acc1


def _wrap_loader(loader, style: StyleConfig, state: TrainState):
    # a CudaPrefetcher consumes its loader iterator, so a fresh one is built
    # for every epoch/pass (matching the reference, which constructs
    # data_prefetcher(loader) at the top of train()/validate(),
    # apex_distributed.py:302,357)
    if style.use_prefetcher and state.device.type == "cuda":
        return CudaPrefetcher(loader, device=state.device,
                              dtype=state.input_dtype(),
                              channels_last=state.channels_last)
    return loader
