"""MNIST data module: 🤗 mnist + normalize(0.5, 0.5), channels-last, optional random
crop; batch dict {"image", "label"}. Parity: reference data/vision/mnist.py."""
from __future__ import annotations

import os
from typing import Optional

from torch.utils.data import DataLoader

from perceiver_amd.data.text.common import Hparams
from perceiver_amd.data.vision.common import (
    Compose,
    ImagePreprocessor,
    Normalize,
    RandomCrop,
    channels_to_last,
    lift_transform,
    to_tensor,
)


def mnist_transform(normalize: bool = True, channels_last: bool = True,
                    random_crop: Optional[int] = None):
    transform_list = []
    if random_crop is not None:
        transform_list.append(RandomCrop(random_crop))
    transform_list.append(to_tensor)
    if normalize:
        transform_list.append(Normalize(mean=(0.5,), std=(0.5,)))
    if channels_last:
        transform_list.append(channels_to_last)
    return Compose(transform_list)


class MNISTPreprocessor(ImagePreprocessor):
    def __init__(self, normalize: bool = True, channels_last: bool = True):
        super().__init__(mnist_transform(normalize, channels_last))


class MNISTDataModule:
    def __init__(
        self,
        dataset_dir: str = os.path.join(".cache", "mnist"),
        normalize: bool = True,
        channels_last: bool = True,
        random_crop: Optional[int] = None,
        batch_size: int = 64,
        num_workers: int = 3,
        pin_memory: bool = True,
        shuffle: bool = True,
    ):
        self.hparams = Hparams(dataset_dir=dataset_dir, normalize=normalize,
                               channels_last=channels_last, random_crop=random_crop,
                               batch_size=batch_size, num_workers=num_workers,
                               pin_memory=pin_memory, shuffle=shuffle)
        self.channels_last = channels_last
        self.tf_train = mnist_transform(normalize, channels_last, random_crop=random_crop)
        self.tf_valid = mnist_transform(normalize, channels_last, random_crop=None)
        self.ds_train = None
        self.ds_valid = None

    @property
    def num_classes(self):
        return 10

    @property
    def image_shape(self):
        return (28, 28, 1) if self.hparams.channels_last else (1, 28, 28)

    def load_dataset(self, split: Optional[str] = None):
        from datasets import load_dataset

        return load_dataset("mnist", split=split, cache_dir=self.hparams.dataset_dir)

    def prepare_data(self) -> None:
        self.load_dataset()

    def setup(self, stage: Optional[str] = None) -> None:
        self.ds_train = self.load_dataset(split="train")
        self.ds_train.set_transform(lift_transform(self.tf_train))
        self.ds_valid = self.load_dataset(split="test")
        self.ds_valid.set_transform(lift_transform(self.tf_valid))

    def train_dataloader(self):
        return DataLoader(self.ds_train, shuffle=self.hparams.shuffle,
                          batch_size=self.hparams.batch_size,
                          num_workers=self.hparams.num_workers,
                          pin_memory=self.hparams.pin_memory)

    def val_dataloader(self):
        return DataLoader(self.ds_valid, shuffle=False, batch_size=self.hparams.batch_size,
                          num_workers=self.hparams.num_workers,
                          pin_memory=self.hparams.pin_memory)
