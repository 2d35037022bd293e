"""Loader construction shared by all entrypoints.

Builds the train/val DataLoaders with the reference's knobs
(pin_memory=True, num_workers, DistributedSampler for train and — when
``distributed_val`` — val too; distributed.py:166-195).  Falls back to the
synthetic dataset when ``--data`` doesn't exist or ``--synthetic`` is set
(there is no network for real ImageNet in the benchmark environment;
BASELINE.json runs on synthetic data).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch
from torch.utils.data import DataLoader

from .folder import ImageFolder
from .sampler import DistributedSampler
from .synthetic import SyntheticImageNet
from .transforms import train_transforms, val_transforms


def build_datasets(args):
    use_synth = getattr(args, "synthetic", False) or not (
        args.data and os.path.isdir(os.path.join(args.data, "train")))
    if use_synth:
        n_train = getattr(args, "synthetic_train_size", 0) or 1_281_167
        n_val = getattr(args, "synthetic_val_size", 0) or 50_000
        image_size = getattr(args, "image_size", 224)
        train = SyntheticImageNet(n_train, image_size=image_size, seed=1)
        val = SyntheticImageNet(n_val, image_size=image_size, seed=2)
        return train, val
    gpu_norm = getattr(args, "gpu_normalize", True)
    img = getattr(args, "image_size", 224)
    # reference crop/resize ratio (224/256, distributed.py:182-189) scales
    # with --image-size
    resize = max(img + 1, int(round(img * 256 / 224)))
    train = ImageFolder(os.path.join(args.data, "train"),
                        train_transforms(size=img, gpu_normalize=gpu_norm))
    val = ImageFolder(os.path.join(args.data, "val"),
                      val_transforms(size=img, resize=resize,
                                     gpu_normalize=gpu_norm))
    return train, val


def build_loaders(args, world_size: int = 1, rank: int = 0,
                  distributed: bool = True, distributed_val: bool = True
                  ) -> Tuple[DataLoader, DataLoader,
                             Optional[DistributedSampler],
                             Optional[DistributedSampler]]:
    train_set, val_set = build_datasets(args)
    train_sampler = val_sampler = None
    if distributed and world_size > 1:
        train_sampler = DistributedSampler(train_set, world_size, rank,
                                           shuffle=True)
        if distributed_val:
            # sharded validation = distributed evaluation
            # (reference distributed.py:190-195; README.md:586-662)
            val_sampler = DistributedSampler(val_set, world_size, rank,
                                             shuffle=False)
    train_loader = DataLoader(
        train_set, batch_size=args.batch_size,
        shuffle=(train_sampler is None), sampler=train_sampler,
        num_workers=args.workers, pin_memory=True, drop_last=False,
        persistent_workers=args.workers > 0)
    val_loader = DataLoader(
        val_set, batch_size=args.batch_size, shuffle=False,
        sampler=val_sampler, num_workers=args.workers, pin_memory=True,
        persistent_workers=args.workers > 0)
    return train_loader, val_loader, train_sampler, val_sampler
