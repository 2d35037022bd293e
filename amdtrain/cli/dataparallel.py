"""Single-process scatter/gather DataParallel entrypoint (reference dataparallel.py).

One Python process drives all GPUs through
``ScatterGatherDataParallel`` (xGMI P2P scatter/replicate/gather).  Kept for
capability parity with the reference's slowest style: full-node batch in one
loader (``shuffle=True``, no sampler, reference :165-169), local-only
metrics, CSV epoch-time log (:188,207-213), checkpoint every epoch without a
rank guard (:215-221).
"""

from __future__ import annotations

import torch

from ..config import base_parser
from ..data.build import build_loaders
from ..engine.loops import TrainState, train, validate
from ..models import build_model
from ..ops import CrossEntropyLoss, FusedSGD
from ..parallel import ScatterGatherDataParallel
from ..utils import adjust_learning_rate, save_checkpoint, set_seed
from ..utils.checkpoint import make_checkpoint_state
from ..utils.csvlog import EpochTimer


def parse_args(argv=None):
    p = base_parser("MI355X single-process scatter/gather DataParallel training")
    p.add_argument("--gpus", default="", type=str,
                   help="comma-separated device ids (default: all visible)")
    return p.parse_args(argv)


def main(argv=None) -> float:
    args = parse_args(argv)
    set_seed(args.seed)

    if torch.cuda.is_available():
        gpus = [int(g) for g in args.gpus.split(",") if g != ""] or \
            list(range(torch.cuda.device_count()))
        device = torch.device(f"cuda:{gpus[0]}")
        torch.cuda.set_device(device)
    else:
        gpus = []
        device = torch.device("cpu")

    model = build_model(args.arch).to(device)
    channels_last = device.type == "cuda" and not args.no_channels_last
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    if len(gpus) > 1:
        model = ScatterGatherDataParallel(model, gpus, output_device=gpus[0])

    criterion = CrossEntropyLoss().to(device)
    optimizer = FusedSGD(model.parameters(), args.lr, momentum=args.momentum,
                         weight_decay=args.weight_decay)

    # full-node batch, plain shuffled loader (reference :165-169)
    train_loader, val_loader, _, _ = build_loaders(args, world_size=1, rank=0,
                                                   distributed=False)
    state = TrainState(
        device=device, world_size=1, rank=0, reduce_metrics=False,
        channels_last=channels_last,
        autocast_dtype={"fp32": None, "bf16": torch.bfloat16,
                        "fp16": torch.float16}[args.dtype]
        if device.type == "cuda" else None,
        print_freq=args.print_freq, max_steps=args.max_steps)

    if args.evaluate:
        return validate(val_loader, model, criterion, state)

    timer = EpochTimer(args.epoch_csv or "dataparallel_epochs.csv")
    best_acc1 = 0.0
    for epoch in range(args.start_epoch, args.epochs):
        adjust_learning_rate(optimizer, epoch, args.lr)
        timer.start()
        train(train_loader, model, criterion, optimizer, epoch, state)
        acc1 = validate(val_loader, model, criterion, state)
        timer.stop(epoch)
        is_best = acc1 > best_acc1
        best_acc1 = max(acc1, best_acc1)
        save_checkpoint(make_checkpoint_state(epoch, args.arch, model,
                                              best_acc1), is_best)
    timer.close()
    return best_acc1


if __name__ == "__main__":
    main()
