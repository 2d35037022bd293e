from .synthetic import SyntheticImageNet
from .sampler import DistributedSampler
from .prefetcher import CudaPrefetcher
from .build import build_loaders

__all__ = ["SyntheticImageNet", "DistributedSampler", "CudaPrefetcher",
           "build_loaders"]
