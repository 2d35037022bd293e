"""Distributed sampler with the reference's sharding semantics.

Own implementation of the ``torch.utils.data.distributed.DistributedSampler``
behavior the reference relies on (distributed.py:174-175,190-195 for train
AND val sharding — distributed evaluation; horovod_distributed.py:182-199
passes explicit num_replicas/rank): pad-to-divisible round-robin sharding,
epoch-seeded shuffle via ``set_epoch``.
"""

from __future__ import annotations

import math
from typing import Iterator, Optional

import torch
import torch.distributed as dist
from torch.utils.data import Sampler


class DistributedSampler(Sampler):
    def __init__(self, dataset, num_replicas: Optional[int] = None,
                 rank: Optional[int] = None, shuffle: bool = True,
                 seed: int = 0, drop_last: bool = False):
        if num_replicas is None:
            num_replicas = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        if not (0 <= rank < num_replicas):
            raise ValueError(f"rank {rank} out of range for {num_replicas} replicas")
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(dataset)
        if drop_last and n % num_replicas:
            self.num_samples = n // num_replicas
        else:
            self.num_samples = math.ceil(n / num_replicas)
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch: int) -> None:
        """Reshuffle seed per epoch (reference sampler.set_epoch,
        distributed.py:202-203)."""
        self.epoch = epoch

    def __iter__(self) -> Iterator[int]:
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                reps = math.ceil(pad / max(len(indices), 1))
                indices += (indices * reps)[:pad]
        else:
            indices = indices[:self.total_size]
        assert len(indices) == self.total_size
        shard = indices[self.rank:self.total_size:self.num_replicas]
        assert len(shard) == self.num_samples
        return iter(shard)

    def __len__(self) -> int:
        return self.num_samples
