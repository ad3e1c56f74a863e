"""IMDb data module: clf uses train/test; mlm uses the unsupervised split
(parity: reference data/text/imdb.py)."""
from __future__ import annotations

import os
from typing import Any

from perceiver_amd.data.text.common import Task, TextDataModule


class ImdbDataModule(TextDataModule):
    def __init__(self, *args: Any, dataset_dir: str = os.path.join(".cache", "imdb"), **kwargs: Any):
        super().__init__(dataset_dir, *args, **kwargs)

    @property
    def num_classes(self):
        return 2

    def load_source_dataset(self):
        from datasets import DatasetDict, load_dataset

        dataset = load_dataset("imdb", "plain_text", cache_dir=self.hparams.dataset_dir)
        if self.hparams.task == Task.clf:
            ds_train, ds_valid = dataset["train"], dataset["test"]
        else:
            ds_train = dataset["unsupervised"].remove_columns("label")
            ds_valid = dataset["test"].remove_columns("label")
        return DatasetDict(train=ds_train, valid=ds_valid)
