"""Text data modules: consistent preprocessing (tokenize -> chunk -> optional static
masking, cached on disk keyed by a hyperparameter hash) and task-specific loading.

Behavioral parity with reference data/text/common.py (Task enum, preproc pipeline,
batch contract, RandomShiftDataset, CLMDataset) without PyTorch Lightning: data
modules are plain classes with prepare_data/setup/train_dataloader/val_dataloader.
"""
from __future__ import annotations

import hashlib
import os
from enum import Enum
from itertools import chain
from typing import Optional, Sequence

import torch
from torch.utils.data import DataLoader

from perceiver_amd.data.text.collator import (
    DefaultCollator,
    RandomTruncateCollator,
    TokenMaskingCollator,
    WordMaskingCollator,
)
from perceiver_amd.data.text.utils import PerceiverTokenizerUtil

os.environ.setdefault("TOKENIZERS_PARALLELISM", "false")

PERCEIVER_TOKENIZERS = ["krasserm/perceiver-io-mlm", "deepmind/language-perceiver"]


def resolve_tokenizer(tokenizer):
    """Accepts a 🤗 hub name OR a tokenizer instance (offline-friendly: the
    byte-level PerceiverTokenizer constructs without network access)."""
    if isinstance(tokenizer, str):
        from transformers import AutoTokenizer

        return AutoTokenizer.from_pretrained(tokenizer, verbose=False)
    return tokenizer


class Hparams(dict):
    """Attribute-access hyperparameter store (stand-in for Lightning's hparams)."""

    def __getattr__(self, name):
        try:
            return self[name]
        except KeyError:
            raise AttributeError(name) from None


class Task(Enum):
    mlm = 0
    clm = 1
    clf = 2


class TextPreprocessor:
    """Inference-side text -> (input_ids, pad_mask) preprocessing."""

    def __init__(self, tokenizer, max_seq_len: int, add_special_tokens: bool):
        self.tokenizer = resolve_tokenizer(tokenizer)
        self.max_seq_len = max_seq_len
        self.add_special_tokens = add_special_tokens

    def preprocess(self, text):
        xs, pad_mask = self.preprocess_batch([text])
        return xs[0], pad_mask[0]

    def preprocess_batch(self, text_batch):
        result = self.tokenizer(
            text_batch,
            padding=self.tokenizer.pad_token is not None,
            truncation=True,
            add_special_tokens=self.add_special_tokens,
            return_token_type_ids=False,
            return_attention_mask=True,
            max_length=self.max_seq_len,
            return_tensors="pt",
        )
        return result["input_ids"], ~result["attention_mask"].type(torch.bool)


class TextDataModule:
    def __init__(
        self,
        dataset_dir: str,
        tokenizer: str,
        max_seq_len: int,
        task: Task = Task.mlm,
        mask_prob: float = 0.15,
        mask_words: bool = True,
        static_masking: bool = False,
        add_special_tokens: bool = False,
        add_eos_token: bool = False,
        padding_side: Optional[str] = None,
        random_train_shift: bool = False,
        random_valid_shift: bool = False,
        random_train_truncation: bool = False,
        random_valid_truncation: bool = False,
        random_min_seq_len: int = 16,
        preproc_batch_size: int = 1000,
        preproc_workers: Optional[int] = None,
        batch_size: int = 64,
        valid_batch_size: Optional[int] = None,
        num_workers: int = 3,
        pin_memory: bool = True,
        **extra_hparams,
    ):
        self.hparams = Hparams(
            dataset_dir=dataset_dir, tokenizer=tokenizer, max_seq_len=max_seq_len, task=task,
            mask_prob=mask_prob, mask_words=mask_words, static_masking=static_masking,
            add_special_tokens=add_special_tokens, add_eos_token=add_eos_token,
            padding_side=padding_side, random_train_shift=random_train_shift,
            random_valid_shift=random_valid_shift, random_train_truncation=random_train_truncation,
            random_valid_truncation=random_valid_truncation, random_min_seq_len=random_min_seq_len,
            preproc_batch_size=preproc_batch_size, preproc_workers=preproc_workers,
            batch_size=batch_size, valid_batch_size=valid_batch_size, num_workers=num_workers,
            pin_memory=pin_memory, **extra_hparams,
        )

        if static_masking and not mask_words:
            raise ValueError("static_masking=true is only supported for mask_words=true")

        self.tokenizer = resolve_tokenizer(tokenizer)
        if padding_side is not None:
            self.tokenizer.padding_side = padding_side

        # the PerceiverTokenizer is not a fast tokenizer: word ids come from
        # whitespace boundaries instead of encoding.word_ids()
        from transformers import PerceiverTokenizer

        self.perceiver_tokenizer_configured = (
            tokenizer in PERCEIVER_TOKENIZERS or isinstance(self.tokenizer, PerceiverTokenizer)
        )
        if self.perceiver_tokenizer_configured:
            self.perceiver_tokenizer_util = PerceiverTokenizerUtil(self.tokenizer)

        if task == Task.mlm and not static_masking:
            if mask_words:
                self.collator = WordMaskingCollator(tokenizer=self.tokenizer, mask_prob=mask_prob)
            else:
                self.collator = TokenMaskingCollator(tokenizer=self.tokenizer, mask_prob=mask_prob)
        else:
            self.collator = DefaultCollator(tokenizer=self.tokenizer, max_seq_len=max_seq_len)

        self.ds_train = None
        self.ds_valid = None

    # ---------------------------------------------------------------- properties
    @property
    def valid_batch_size(self):
        return self.hparams.valid_batch_size or self.hparams.batch_size

    @property
    def vocab_size(self):
        return self.tokenizer.vocab_size

    @property
    def max_seq_len(self):
        return self.hparams.max_seq_len

    @property
    def random_shift(self):
        return self.hparams.random_train_shift or self.hparams.random_valid_shift

    @property
    def preproc_workers(self):
        if self.hparams.preproc_workers is not None:
            return self.hparams.preproc_workers
        return max(1, self.hparams.num_workers)

    @property
    def preproc_dir(self):
        h = hashlib.new("md5")
        h.update(self.preproc_dir_hash_input().encode())
        return os.path.join(self.hparams.dataset_dir, "preproc", h.hexdigest())

    def preproc_dir_hash_input(self) -> str:
        hp = self.hparams
        hash_input = f"{hp.tokenizer}-{self.max_seq_len}-{hp.task.name}-{self.random_shift}"
        if hp.task == Task.mlm and hp.static_masking:
            hash_input = f"{hash_input}-{hp.mask_words}-{hp.mask_prob}"
        if hp.add_special_tokens:
            hash_input = f"{hash_input}-st"
        if hp.add_eos_token:
            hash_input = f"{hash_input}-eos"
        if hp.get("source_train_size") is not None:
            hash_input = f"{hash_input}-ts-{hp['source_train_size']}"
        if hp.get("source_valid_size") is not None:
            hash_input = f"{hash_input}-vs-{hp['source_valid_size']}"
        return hash_input

    # ---------------------------------------------------------------- lifecycle
    def prepare_data(self) -> None:
        if not os.path.exists(self.preproc_dir):
            dataset = self.load_source_dataset()
            dataset = self._prepare_dataset(dataset)
            dataset.save_to_disk(self.preproc_dir)

    def setup(self, stage=None):
        dataset = self.load_prepared_dataset()
        self.ds_train = dataset["train"]
        self.ds_valid = dataset["valid"]

        if self.hparams.task in (Task.clm, Task.mlm):
            if self.hparams.random_train_shift:
                self.ds_train = RandomShiftDataset(self.ds_train)
            if self.hparams.random_valid_shift:
                self.ds_valid = RandomShiftDataset(self.ds_valid)

        if self.hparams.task == Task.clm:
            self.ds_train = CLMDataset(self.ds_train)
            self.ds_valid = CLMDataset(self.ds_valid)

    def _loader(self, ds, shuffle, batch_size, random_truncation):
        collator = self.collator
        if random_truncation:
            collator = RandomTruncateCollator(collator, self.hparams.random_min_seq_len)
        return DataLoader(
            ds, shuffle=shuffle, collate_fn=collator, batch_size=batch_size,
            num_workers=self.hparams.num_workers, pin_memory=self.hparams.pin_memory,
        )

    def train_dataloader(self):
        return self._loader(self.ds_train, True, self.hparams.batch_size,
                            self.hparams.random_train_truncation)

    def val_dataloader(self):
        return self._loader(self.ds_valid, False, self.valid_batch_size,
                            self.hparams.random_valid_truncation)

    def text_preprocessor(self) -> TextPreprocessor:
        preproc = TextPreprocessor(
            tokenizer=self.hparams.tokenizer,
            max_seq_len=self.hparams.max_seq_len,
            add_special_tokens=self.hparams.add_special_tokens,
        )
        if self.hparams.padding_side is not None:
            preproc.tokenizer.padding_side = self.hparams.padding_side
        return preproc

    # ---------------------------------------------------------------- preprocessing
    def load_source_dataset(self):
        """Return a DatasetDict with keys 'train' and 'valid'."""
        raise NotImplementedError

    def load_prepared_dataset(self):
        from datasets import DatasetDict

        return DatasetDict.load_from_disk(self.preproc_dir)

    def _prepare_dataset(self, dataset):
        if self.hparams.task == Task.clm:
            dataset = self._tokenize_dataset(dataset, return_word_ids=False)
            # +1 so CLMDataset can shift input/labels by one
            dataset = self._chunk_dataset(dataset, chunk_size=self.max_seq_len + 1,
                                          include_keys=["input_ids"])
        elif self.hparams.task == Task.mlm:
            dataset = self._tokenize_dataset(dataset, return_word_ids=True)
            dataset = self._chunk_dataset(dataset, chunk_size=self.max_seq_len)
            if self.hparams.static_masking:
                dataset = self._mask_dataset(dataset)
        else:  # Task.clf
            assert "label" in dataset["train"].column_names
            assert "label" in dataset["valid"].column_names
            dataset = self._tokenize_dataset(dataset, max_length=self.max_seq_len,
                                             truncation=True, return_word_ids=False)
        return dataset

    def _tokenize_dataset(self, dataset, padding=False, truncation=False, max_length=None,
                          return_word_ids=True):
        from datasets import DatasetDict

        def tokenize(examples):
            if self.hparams.add_eos_token:
                examples["text"] = [t + self.tokenizer.eos_token for t in examples["text"]]
            encoding = self.tokenizer(
                examples["text"], padding=padding, truncation=truncation, max_length=max_length,
                add_special_tokens=self.hparams.add_special_tokens,
                return_token_type_ids=False, return_attention_mask=False,
            )
            if return_word_ids:
                if self.perceiver_tokenizer_configured:
                    encoding["word_ids"] = [
                        self.perceiver_tokenizer_util.word_ids(ids) for ids in encoding["input_ids"]
                    ]
                else:
                    encoding["word_ids"] = [encoding.word_ids(i) for i in range(len(encoding["input_ids"]))]
            return encoding

        result = DatasetDict()
        for key in dataset.keys():
            result[key] = dataset[key].map(
                tokenize, batched=True, batch_size=self.hparams.preproc_batch_size,
                num_proc=self.preproc_workers, remove_columns=["text"],
                load_from_cache_file=False, desc="Running tokenizer on dataset",
            )
        return result

    def _chunk_dataset(self, dataset, chunk_size: int,
                       include_keys: Sequence[str] = ("input_ids", "word_ids"),
                       remove_keys: Sequence[str] = ()):
        from datasets import DatasetDict

        def chunk(*args):
            chained = {k: list(chain(*args[i])) for i, k in enumerate(include_keys)}
            chained_len = len(chained[include_keys[0]])
            if chained_len >= chunk_size:
                chained_len = (chained_len // chunk_size) * chunk_size
            return {k: [t[i: i + chunk_size] for i in range(0, chained_len, chunk_size)]
                    for k, t in chained.items()}

        result = DatasetDict()
        for key in dataset.keys():
            result[key] = dataset[key].map(
                chunk, batched=True, batch_size=self.hparams.preproc_batch_size,
                num_proc=self.preproc_workers, input_columns=list(include_keys),
                remove_columns=list(remove_keys), load_from_cache_file=False,
                desc=f"Split dataset into chunks of size {chunk_size}",
            )
        return result

    def _mask_dataset(self, dataset):
        from datasets import DatasetDict

        wmc = WordMaskingCollator(tokenizer=self.tokenizer, mask_prob=self.hparams.mask_prob)

        result = DatasetDict()
        for key in dataset.keys():
            result[key] = dataset[key].map(
                wmc.mask_words_1, batched=False, num_proc=self.preproc_workers,
                load_from_cache_file=False, desc="Mask words in dataset",
            )
        return result

    def _train_valid_split(self, dataset, train_size, test_size):
        from datasets import DatasetDict

        dataset = dataset.train_test_split(train_size=train_size, test_size=test_size,
                                           shuffle=not self.random_shift)
        return DatasetDict(train=dataset["train"], valid=dataset["test"])


class RandomShiftDataset(torch.utils.data.Dataset):
    """Concatenates each record with its successor at a random offset (data
    augmentation for chunked LM corpora)."""

    def __init__(self, dataset):
        self.dataset = dataset

    def __getitem__(self, idx):
        example_1 = self.dataset[idx]
        example_2 = self.dataset[idx + 1]
        result = {}
        shift = None
        for key in example_1.keys():
            record_1, record_2 = example_1[key], example_2[key]
            if shift is None:
                shift = int(torch.randint(len(record_1), (1,)))
            result[key] = record_1[shift:] + record_2[:shift]
        return result

    def __len__(self):
        return len(self.dataset) - 1


class CLMDataset(torch.utils.data.Dataset):
    """Shift-by-one input/label pairs from (max_seq_len + 1)-sized chunks."""

    def __init__(self, dataset):
        self.dataset = dataset

    def __getitem__(self, idx):
        record = self.dataset[idx]["input_ids"]
        return {"input_ids": record[:-1], "label_ids": record[1:]}

    def __len__(self):
        return len(self.dataset)
