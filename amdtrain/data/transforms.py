"""Minimal PIL-based image transforms (torchvision-free).

The reference composes RandomResizedCrop(224) + RandomHorizontalFlip +
ToTensor + Normalize for train and Resize(256) + CenterCrop(224) + ToTensor
+ Normalize for val (distributed.py:166-189).  torchvision is not a
dependency here, so the same five transforms are implemented over PIL +
torch directly.
"""

from __future__ import annotations

import math
import random
from typing import Sequence

import numpy as np
import torch


class Compose:
    def __init__(self, transforms):
        self.transforms = list(transforms)

    def __call__(self, img):
        for t in self.transforms:
            img = t(img)
        return img


class Resize:
    """Resize the SHORT side to ``size`` keeping aspect ratio (torchvision
    Resize(int) semantics)."""

    def __init__(self, size: int):
        self.size = size

    def __call__(self, img):
        w, h = img.size
        if w <= h:
            nw, nh = self.size, max(1, round(h * self.size / w))
        else:
            nh, nw = self.size, max(1, round(w * self.size / h))
        return img.resize((nw, nh), resample=2)  # BILINEAR


class CenterCrop:
    def __init__(self, size: int):
        self.size = size

    def __call__(self, img):
        w, h = img.size
        left = (w - self.size) // 2
        top = (h - self.size) // 2
        return img.crop((left, top, left + self.size, top + self.size))


class RandomHorizontalFlip:
    def __init__(self, p: float = 0.5):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            return img.transpose(0)  # FLIP_LEFT_RIGHT
        return img


class RandomResizedCrop:
    """Random area (0.08..1.0) + aspect (3/4..4/3) crop resized to ``size``."""

    def __init__(self, size: int, scale=(0.08, 1.0),
                 ratio=(3.0 / 4.0, 4.0 / 3.0)):
        self.size = size
        self.scale = scale
        self.ratio = ratio

    def __call__(self, img):
        w, h = img.size
        area = w * h
        for _ in range(10):
            target_area = random.uniform(*self.scale) * area
            log_ratio = (math.log(self.ratio[0]), math.log(self.ratio[1]))
            aspect = math.exp(random.uniform(*log_ratio))
            cw = int(round(math.sqrt(target_area * aspect)))
            ch = int(round(math.sqrt(target_area / aspect)))
            if 0 < cw <= w and 0 < ch <= h:
                left = random.randint(0, w - cw)
                top = random.randint(0, h - ch)
                img = img.crop((left, top, left + cw, top + ch))
                return img.resize((self.size, self.size), resample=2)
        # fallback: center crop
        img = Resize(self.size)(img)
        return CenterCrop(self.size)(img)


class ToTensor:
    """PIL RGB -> float32 CHW in [0,1]."""

    def __call__(self, img) -> torch.Tensor:
        arr = np.asarray(img.convert("RGB"), dtype=np.uint8)  # HWC
        t = torch.from_numpy(arr.copy()).permute(2, 0, 1).contiguous()
        return t.float().div_(255.0)


class ToUint8Tensor:
    """PIL RGB -> uint8 CHW (for the GPU-side normalize path)."""

    def __call__(self, img) -> torch.Tensor:
        arr = np.asarray(img.convert("RGB"), dtype=np.uint8)
        return torch.from_numpy(arr.copy()).permute(2, 0, 1).contiguous()


class Normalize:
    def __init__(self, mean: Sequence[float], std: Sequence[float]):
        self.mean = torch.tensor(mean).reshape(-1, 1, 1)
        self.std = torch.tensor(std).reshape(-1, 1, 1)

    def __call__(self, t: torch.Tensor) -> torch.Tensor:
        return (t - self.mean) / self.std


IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)


def train_transforms(size: int = 224, gpu_normalize: bool = False) -> Compose:
    """Reference train pipeline (distributed.py:166-173)."""
    tail = [ToUint8Tensor()] if gpu_normalize else \
        [ToTensor(), Normalize(IMAGENET_MEAN, IMAGENET_STD)]
    return Compose([RandomResizedCrop(size), RandomHorizontalFlip()] + tail)


def val_transforms(size: int = 224, resize: int = 256,
                   gpu_normalize: bool = False) -> Compose:
    """Reference val pipeline (distributed.py:182-189)."""
    tail = [ToUint8Tensor()] if gpu_normalize else \
        [ToTensor(), Normalize(IMAGENET_MEAN, IMAGENET_STD)]
    return Compose([Resize(resize), CenterCrop(size)] + tail)
