"""ImageNet inference-side preprocessor: center-crop 256 -> bicubic 224 + ImageNet
normalization + channels-last (parity: reference data/vision/imagenet.py; implemented
on PIL + torch directly, no torchvision)."""
from __future__ import annotations

from perceiver_amd.data.vision.common import (
    Compose,
    ImagePreprocessor,
    Normalize,
    center_crop_resize,
    channels_to_last,
    to_tensor,
)

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)


def imagenet_valid_transform(crop_size: int, size: int, channels_last: bool):
    transforms = [
        lambda img: center_crop_resize(img.convert("RGB") if hasattr(img, "convert") else img,
                                       crop_size, size),
        to_tensor,
        Normalize(IMAGENET_MEAN, IMAGENET_STD),
    ]
    if channels_last:
        transforms.append(channels_to_last)
    return Compose(transforms)


class ImageNetPreprocessor(ImagePreprocessor):
    def __init__(self, crop_size: int = 256, size: int = 224, channels_last: bool = True):
        super().__init__(imagenet_valid_transform(crop_size, size, channels_last))
