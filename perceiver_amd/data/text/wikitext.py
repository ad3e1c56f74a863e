"""WikiText-103-raw data module (parity: reference data/text/wikitext.py)."""
from __future__ import annotations

import os
from typing import Any

from perceiver_amd.data.text.common import TextDataModule


class WikiTextDataModule(TextDataModule):
    def __init__(self, *args: Any, dataset_dir: str = os.path.join(".cache", "wikitext"), **kwargs: Any):
        super().__init__(dataset_dir, *args, **kwargs)

    def load_source_dataset(self):
        from datasets import DatasetDict, load_dataset

        dataset = load_dataset("wikitext", "wikitext-103-raw-v1", cache_dir=self.hparams.dataset_dir)
        return DatasetDict(train=dataset["train"], valid=dataset["validation"])
