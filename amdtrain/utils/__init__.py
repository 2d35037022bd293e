from .meters import AverageMeter, ProgressMeter
from .metrics import accuracy
from .lr import adjust_learning_rate
from .checkpoint import save_checkpoint, load_checkpoint
from .seed import set_seed

__all__ = [
    "AverageMeter",
    "ProgressMeter",
    "accuracy",
    "adjust_learning_rate",
    "save_checkpoint",
    "load_checkpoint",
    "set_seed",
]
