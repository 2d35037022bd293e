"""Native ResNet family for ImageNet, written for MI355X execution.

The reference builds any torchvision classification model by name
(``torchvision.models.__dict__[args.arch]()``, reference distributed.py:21-23,
134-139; default ``resnet18``; the published benchmark uses the ResNet family).
torchvision is not a dependency of this framework — the architectures are
implemented here from the original papers (He et al. 2015, and the
"ResNet v1.5" stride-on-3x3 variant that torchvision/NVIDIA use), with
torchvision-compatible ``state_dict`` key names so reference checkpoints load
unchanged (``conv1``, ``bn1``, ``layerN.M.convK/bnK``, ``downsample.0/1``,
``fc``).

MI355X-first details:
  * The module tree holds parameters; the hot math runs through
    ``amdtrain.ops.functional`` which dispatches to hand-written gfx950 HIP
    kernels (fused BN+ReLU, pooling) when the extension is loaded and the
    tensors live on GPU, and to plain PyTorch ops otherwise (CPU tests).
  * NHWC (channels_last) is the native layout on gfx950 — ``.to(memory_format=
    torch.channels_last)`` the module + inputs; all custom kernels assume NHWC.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional, Type, Union

import torch
import torch.nn as nn

from ..ops import fused as OF


import os

from ..ops.conv import AmdConv2d, GradCell, ResidualGradTap  # noqa: F401
from ..ops.linear import AmdLinear


def _residual_fuse_enabled() -> bool:
    """Fuse the identity-shortcut gradient into conv1's dgrad epilogue
    (AMDTRAIN_RESFUSE=0 restores plain autograd accumulation)."""
    return os.environ.get("AMDTRAIN_RESFUSE", "1") == "1"


def conv3x3(in_planes: int, out_planes: int, stride: int = 1,
            groups: int = 1) -> nn.Conv2d:
    return AmdConv2d(in_planes, out_planes, kernel_size=3, stride=stride,
                     padding=1, groups=groups, bias=False)


def conv1x1(in_planes: int, out_planes: int, stride: int = 1) -> nn.Conv2d:
    return AmdConv2d(in_planes, out_planes, kernel_size=1, stride=stride,
                     bias=False)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d whose forward can fuse the trailing ReLU.

    Parameter/buffer names are identical to ``nn.BatchNorm2d`` so state_dicts
    interoperate.  ``forward_relu`` is the fused entrypoint used inside the
    residual blocks; it routes through ``amdtrain.ops.fused.batch_norm`` which
    selects the hand-written NHWC HIP kernel on GPU.
    """

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # type: ignore[override]
        return OF.batch_norm(x, self, relu=False)

    def forward_relu(self, x: torch.Tensor) -> torch.Tensor:
        return OF.batch_norm(x, self, relu=True)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = FusedBatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)  # kept for module-tree parity
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = FusedBatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        cell = None
        if (self.training and x.is_cuda and torch.is_grad_enabled()
                and _residual_fuse_enabled()):
            cell = getattr(x, "_amdtrain_next_cell", None)
        out = self.bn1.forward_relu(self.conv1(x))
        out = self.conv2(out)
        if cell is not None:
            cell.armed = True
            tapped = ResidualGradTap.apply(x, cell)
            identity = self.downsample(tapped)                 if self.downsample is not None else tapped
        elif self.downsample is not None:
            identity = self.downsample(x)
        return OF.bn_add_relu(out, self.bn2, identity)


class Bottleneck(nn.Module):
    # ResNet v1.5: stride lives on the 3x3 conv (torchvision behavior),
    # which is what the published ResNet-50 throughput numbers assume.
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None, groups: int = 1,
                 base_width: int = 64):
        super().__init__()
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = FusedBatchNorm2d(width)
        self.conv2 = conv3x3(width, width, stride, groups)
        self.bn2 = FusedBatchNorm2d(width)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = FusedBatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        cell = None
        if (self.training and x.is_cuda and torch.is_grad_enabled()
                and _residual_fuse_enabled()):
            # reroute the shortcut-branch gradient into the PRODUCING
            # block-tail BN's backward (mailbox attached to x by that BN) —
            # kills the eager add at x's AccumulateGrad.  Works for both
            # identity shortcuts (tap feeds bn3's addend directly) and
            # downsample shortcuts (tap feeds the downsample conv, whose
            # backward runs before the producing BN's).
            cell = getattr(x, "_amdtrain_next_cell", None)
        out = self.bn1.forward_relu(self.conv1(x))
        out = self.bn2.forward_relu(self.conv2(out))
        out = self.conv3(out)
        if cell is not None:
            cell.armed = True
            tapped = ResidualGradTap.apply(x, cell)
            identity = self.downsample(tapped)                 if self.downsample is not None else tapped
        elif self.downsample is not None:
            identity = self.downsample(x)
        return OF.bn_add_relu(out, self.bn3, identity)


class ResNet(nn.Module):
    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], num_classes: int = 1000,
                 zero_init_residual: bool = False, groups: int = 1,
                 width_per_group: int = 64):
        super().__init__()
        self.groups = groups
        self.base_width = width_per_group
        self.inplanes = 64
        self.conv1 = AmdConv2d(3, 64, kernel_size=7, stride=2, padding=3,
                               bias=False)
        self.bn1 = FusedBatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = AmdLinear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.constant_(m.bn3.weight, 0)
                elif isinstance(m, BasicBlock):
                    nn.init.constant_(m.bn2.weight, 0)

    def _make_layer(self, block, planes: int, blocks: int,
                    stride: int = 1) -> nn.Sequential:
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                FusedBatchNorm2d(planes * block.expansion),
            )
        kw = {}
        if block is Bottleneck:
            kw = dict(groups=self.groups, base_width=self.base_width)
        layers = [block(self.inplanes, planes, stride, downsample, **kw)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes, **kw))
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.bn1.forward_relu(self.conv1(x))
        x = OF.max_pool_3x3_s2(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = OF.global_avg_pool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def resnet18(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes=num_classes, **kw)


def resnet34(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes=num_classes, **kw)


def resnet50(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes, **kw)


def resnet101(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes=num_classes, **kw)


def resnet152(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes=num_classes, **kw)


def resnext50_32x4d(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes,
                  groups=32, width_per_group=4, **kw)


def wide_resnet50_2(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes,
                  width_per_group=128, **kw)


_REGISTRY: Dict[str, Callable[..., nn.Module]] = {
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "resnet101": resnet101,
    "resnet152": resnet152,
    "resnext50_32x4d": resnext50_32x4d,
    "wide_resnet50_2": wide_resnet50_2,
}


def model_names() -> List[str]:
    """Sorted architecture names, for the ``-a/--arch`` choices list
    (reference distributed.py:21-23)."""
    return sorted(_REGISTRY)


def build_model(arch: str, num_classes: int = 1000, **kw) -> nn.Module:
    """``torchvision.models.__dict__[arch]()`` equivalent (distributed.py:134-139)."""
    if arch not in _REGISTRY:
        raise ValueError(f"unknown arch '{arch}'; choices: {model_names()}")
    return _REGISTRY[arch](num_classes=num_classes, **kw)
