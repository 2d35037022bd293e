"""ImageFolder-compatible dataset (torchvision-free, PIL-backed).

Same directory contract as the ``ImageFolder`` the reference trains from
(distributed.py:166-173): ``root/<class_name>/<image file>``, classes sorted
alphabetically and mapped to contiguous indices.
"""

from __future__ import annotations

import os
from typing import Callable, List, Optional, Tuple

from torch.utils.data import Dataset

IMG_EXTENSIONS = (".jpg", ".jpeg", ".png", ".ppm", ".bmp", ".pgm", ".tif",
                  ".tiff", ".webp")


class ImageFolder(Dataset):
    def __init__(self, root: str, transform: Optional[Callable] = None):
        self.root = root
        self.transform = transform
        self.classes = sorted(
            d.name for d in os.scandir(root) if d.is_dir())
        if not self.classes:
            raise FileNotFoundError(f"no class directories under {root}")
        self.class_to_idx = {c: i for i, c in enumerate(self.classes)}
        self.samples: List[Tuple[str, int]] = []
        for c in self.classes:
            cdir = os.path.join(root, c)
            for dirpath, _, filenames in sorted(os.walk(cdir)):
                for fn in sorted(filenames):
                    if fn.lower().endswith(IMG_EXTENSIONS):
                        self.samples.append(
                            (os.path.join(dirpath, fn), self.class_to_idx[c]))
        if not self.samples:
            raise FileNotFoundError(f"no images found under {root}")

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, index: int):
        from PIL import Image
        path, target = self.samples[index]
        with Image.open(path) as img:
            img = img.convert("RGB")
            if self.transform is not None:
                img = self.transform(img)
        return img, target
