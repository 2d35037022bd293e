from .resnet import (
    ResNet,
    BasicBlock,
    Bottleneck,
    build_model,
    model_names,
    resnet18,
    resnet34,
    resnet50,
    resnet101,
    resnet152,
)

__all__ = [
    "ResNet", "BasicBlock", "Bottleneck", "build_model", "model_names",
    "resnet18", "resnet34", "resnet50", "resnet101", "resnet152",
]
