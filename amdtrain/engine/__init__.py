from .loops import train, validate, TrainState

__all__ = ["train", "validate", "TrainState"]
