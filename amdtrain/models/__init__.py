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
    resnext50_32x4d,
    wide_resnet50_2,
)

__all__ = [
    "ResNet", "BasicBlock", "Bottleneck", "build_model", "model_names",
    "resnet18", "resnet34", "resnet50", "resnet101", "resnet152",
    "resnext50_32x4d", "wide_resnet50_2",
]
