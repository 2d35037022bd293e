from .reducer import BucketedReducer
from .ddp import NativeDDP
from .dataparallel import ScatterGatherDataParallel
from .horovod_style import DistributedOptimizer, Compression
from . import amp

__all__ = [
    "BucketedReducer",
    "NativeDDP",
    "ScatterGatherDataParallel",
    "DistributedOptimizer",
    "Compression",
    "amp",
]
