"""Shared argparse surface — the reference CLI, preserved.

Every flag of the reference scripts (distributed.py:25-102, shared across
all six) plus MI355X-specific extensions.  Launch-style-specific flags
(--local_rank, --dist-file) are added by the individual entrypoints.
"""

from __future__ import annotations

import argparse

from .models import model_names


def base_parser(description: str) -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description=description)
    names = model_names()
    p.add_argument("--data", metavar="DIR", default="",
                   help="path to dataset (ImageFolder layout); synthetic "
                        "data is used when absent")
    p.add_argument("-a", "--arch", metavar="ARCH", default="resnet18",
                   choices=names,
                   help="model architecture: " + " | ".join(names) +
                        " (default: resnet18)")
    p.add_argument("-j", "--workers", default=4, type=int, metavar="N",
                   help="number of data loading workers (default: 4; honored "
                        "— the reference exposes -j 4 but hardcodes "
                        "num_workers=2 in its DataLoaders, "
                        "distributed.py:178; we deliberately honor the flag)")
    p.add_argument("--epochs", default=90, type=int, metavar="N",
                   help="number of total epochs to run")
    p.add_argument("--start-epoch", default=0, type=int, metavar="N",
                   help="manual epoch number (useful on restarts)")
    p.add_argument("-b", "--batch-size", default=3200, type=int, metavar="N",
                   help="total batch size of all GPUs on the node "
                        "(default: 3200)")
    p.add_argument("--lr", "--learning-rate", default=0.1, type=float,
                   metavar="LR", help="initial learning rate", dest="lr")
    p.add_argument("--momentum", default=0.9, type=float, metavar="M",
                   help="momentum")
    p.add_argument("--wd", "--weight-decay", default=1e-4, type=float,
                   metavar="W", help="weight decay (default: 1e-4)",
                   dest="weight_decay")
    p.add_argument("-p", "--print-freq", default=10, type=int, metavar="N",
                   help="print frequency (default: 10)")
    p.add_argument("-e", "--evaluate", dest="evaluate", action="store_true",
                   help="evaluate model on validation set")
    p.add_argument("--pretrained", dest="pretrained", action="store_true",
                   help="use pre-trained model (loads checkpoint.pth.tar)")
    p.add_argument("--seed", default=None, type=int,
                   help="seed for initializing training")
    p.add_argument("--resume", default="", type=str, metavar="PATH",
                   help="path to checkpoint to resume from")

    # MI355X-native extensions
    p.add_argument("--synthetic", action="store_true",
                   help="force the synthetic dataset")
    p.add_argument("--synthetic-train-size", default=0, type=int)
    p.add_argument("--synthetic-val-size", default=0, type=int)
    p.add_argument("--image-size", default=224, type=int)
    p.add_argument("--dtype", default="bf16",
                   choices=["fp32", "bf16", "fp16"],
                   help="compute dtype for the autocast path")
    p.add_argument("--bucket-mb", default=50.0, type=float,
                   help="gradient all-reduce bucket size (xGMI-tuned)")
    p.add_argument("--no-channels-last", action="store_true",
                   help="disable NHWC (channels_last) layout")
    p.add_argument("--no-prefetcher", action="store_true",
                   help="disable the side-stream GPU prefetcher")
    p.add_argument("--epoch-csv", default="", type=str,
                   help="append per-epoch wall-clock seconds to this CSV")
    p.add_argument("--max-steps", default=0, type=int,
                   help="cap steps per epoch (quick runs / tests)")
    return p
