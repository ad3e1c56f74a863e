"""Vision preprocessing helpers. Self-contained (no torchvision dependency):
tensor conversion/normalization/cropping implemented on torch + PIL directly.
Parity: reference data/vision/common.py plus the torchvision transform pipelines
it composes (mnist.py:85-100)."""
from __future__ import annotations

from typing import Optional, Sequence

import numpy as np
import torch


class ImagePreprocessor:
    def __init__(self, transform):
        self.transform = transform

    def preprocess(self, img):
        return self.transform(img)

    def preprocess_batch(self, img_batch):
        return torch.stack([self.preprocess(img) for img in img_batch])


def lift_transform(transform):
    def apply(examples):
        examples["image"] = [transform(image) for image in examples["image"]]
        return examples

    return apply


def channels_to_last(img: torch.Tensor) -> torch.Tensor:
    return img.permute(1, 2, 0).contiguous()


def to_tensor(img) -> torch.Tensor:
    """PIL image / ndarray (H, W[, C]) uint8 -> float tensor (C, H, W) in [0, 1]."""
    arr = np.asarray(img)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    t = torch.from_numpy(np.ascontiguousarray(arr))
    if t.dtype == torch.uint8:
        t = t.float().div_(255.0)
    else:
        t = t.float()
    return t.permute(2, 0, 1).contiguous()


class Normalize:
    def __init__(self, mean: Sequence[float], std: Sequence[float]):
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        return (img - self.mean) / self.std


class RandomCrop:
    """Random spatial crop on a PIL image or (C, H, W) tensor."""

    def __init__(self, size: int):
        self.size = size

    def __call__(self, img):
        if isinstance(img, torch.Tensor):
            _, h, w = img.shape
            top = int(torch.randint(0, h - self.size + 1, (1,)))
            left = int(torch.randint(0, w - self.size + 1, (1,)))
            return img[:, top: top + self.size, left: left + self.size]
        w, h = img.size
        top = int(torch.randint(0, h - self.size + 1, (1,)))
        left = int(torch.randint(0, w - self.size + 1, (1,)))
        return img.crop((left, top, left + self.size, top + self.size))


class Compose:
    def __init__(self, transforms):
        self.transforms = list(transforms)

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


def center_crop_resize(img, crop_size: int, size: int):
    """Center-crop to the shorter side scaled to ``crop_size`` then bicubic-resize to
    ``size`` x ``size`` (the deepmind ImageNet eval pipeline)."""
    from PIL import Image

    w, h = img.size
    short = min(w, h)
    scale = crop_size / short
    img = img.resize((round(w * scale), round(h * scale)), Image.BICUBIC)
    w, h = img.size
    left = (w - crop_size) // 2
    top = (h - crop_size) // 2
    img = img.crop((left, top, left + crop_size, top + crop_size))
    if size != crop_size:
        img = img.resize((size, size), Image.BICUBIC)
    return img
