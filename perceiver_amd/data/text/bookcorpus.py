"""BookCorpus / BookCorpusOpen data modules (parity: reference
data/text/bookcorpus.py, bookcorpusopen.py)."""
from __future__ import annotations

import os
from typing import Any, Union

from perceiver_amd.data.text.common import TextDataModule


class BookCorpusDataModule(TextDataModule):
    def __init__(self, *args: Any, dataset_dir: str = os.path.join(".cache", "bookcorpus"),
                 source_train_size: Union[float, int, None] = None,
                 source_valid_size: Union[float, int, None] = 0.02,
                 preproc_batch_size: int = 10000, **kwargs: Any):
        super().__init__(dataset_dir, *args, preproc_batch_size=preproc_batch_size,
                         source_train_size=source_train_size,
                         source_valid_size=source_valid_size, **kwargs)

    def load_source_dataset(self):
        from datasets import load_dataset

        dataset = load_dataset("bookcorpus", "plain_text", split="train",
                               cache_dir=self.hparams.dataset_dir)
        return self._train_valid_split(dataset, self.hparams["source_train_size"],
                                       self.hparams["source_valid_size"])


class BookCorpusOpenDataModule(TextDataModule):
    def __init__(self, *args: Any, dataset_dir: str = os.path.join(".cache", "bookcorpusopen"),
                 source_train_size: Union[float, int, None] = None,
                 source_valid_size: Union[float, int, None] = 0.02,
                 preproc_batch_size: int = 10, **kwargs: Any):
        super().__init__(dataset_dir, *args, preproc_batch_size=preproc_batch_size,
                         source_train_size=source_train_size,
                         source_valid_size=source_valid_size, **kwargs)

    def load_source_dataset(self):
        from datasets import DatasetDict, load_dataset

        dataset = load_dataset("bookcorpusopen", "plain_text", split="train",
                               cache_dir=self.hparams.dataset_dir)
        dataset = self._train_valid_split(dataset, self.hparams["source_train_size"],
                                          self.hparams["source_valid_size"])
        return DatasetDict(train=dataset["train"].remove_columns(["title"]),
                           valid=dataset["valid"].remove_columns(["title"]))
