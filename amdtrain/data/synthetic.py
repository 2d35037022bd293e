"""Synthetic ImageNet-shaped dataset.

Stands in for the reference's ``ImageFolder`` train/val sets
(distributed.py:166-189) when no dataset is on disk (BASELINE.json: the
benchmark runs on synthetic 3x224x224 random data with random-init weights).

Samples are generated deterministically per index (seeded generator), so
every rank/worker sees the same data for a given index — which keeps
sampler-sharding tests exact.  Images are uint8 (the prefetcher does the
on-GPU cast + normalize, like the reference's apex prefetcher,
apex_distributed.py:119-122,157-158); set ``normalized=True`` to get
float CHW tensors directly (plain-loader path).
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset


class SyntheticImageNet(Dataset):
    def __init__(self, length: int = 1_281_167, num_classes: int = 1000,
                 image_size: int = 224, normalized: bool = False,
                 seed: int = 0, dtype: torch.dtype = torch.float32):
        self.length = length
        self.num_classes = num_classes
        self.image_size = image_size
        self.normalized = normalized
        self.seed = seed
        self.dtype = dtype

    def __len__(self) -> int:
        return self.length

    def __getitem__(self, index: int):
        g = torch.Generator().manual_seed(self.seed * 0x9E3779B1 + index)
        target = int(torch.randint(0, self.num_classes, (1,), generator=g))
        if self.normalized:
            img = torch.randn(3, self.image_size, self.image_size,
                              generator=g, dtype=torch.float32).to(self.dtype)
        else:
            img = torch.randint(0, 256, (3, self.image_size, self.image_size),
                                generator=g, dtype=torch.uint8)
        return img, target
