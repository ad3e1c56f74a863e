"""Wikipedia (20220301.en) data module (parity: reference data/text/wikipedia.py)."""
from __future__ import annotations

import os
from typing import Any, Union

from perceiver_amd.data.text.common import TextDataModule


class WikipediaDataModule(TextDataModule):
    def __init__(self, *args: Any, dataset_dir: str = os.path.join(".cache", "wikipedia"),
                 source_train_size: Union[float, int, None] = None,
                 source_valid_size: Union[float, int, None] = 0.02, **kwargs: Any):
        super().__init__(dataset_dir, *args, source_train_size=source_train_size,
                         source_valid_size=source_valid_size, **kwargs)

    def load_source_dataset(self):
        from datasets import DatasetDict, load_dataset

        dataset = load_dataset("wikipedia", "20220301.en", split="train",
                               cache_dir=self.hparams.dataset_dir)
        dataset = self._train_valid_split(dataset, self.hparams["source_train_size"],
                                          self.hparams["source_valid_size"])
        cols = ["id", "url", "title"]
        return DatasetDict(train=dataset["train"].remove_columns(cols),
                           valid=dataset["valid"].remove_columns(cols))
