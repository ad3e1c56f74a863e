"""C4 streaming data module: window-shuffled streaming 🤗 dataset, sharded across
ranks with split_dataset_by_node (the one explicit torch.distributed data-path call,
reference data/text/c4.py:57-164), EOS-joined token stream chunked to
max_seq_len (+1) with optional random chunk length; collator pads then shifts
input/labels by one."""
from __future__ import annotations

import os
from typing import Optional

import torch
from torch.utils.data import DataLoader

from perceiver_amd.data.text.collator import Collator
from perceiver_amd.data.text.common import Hparams, TextPreprocessor
from perceiver_amd.parallel import get_rank, get_world_size

os.environ.setdefault("TOKENIZERS_PARALLELISM", "false")
os.environ.setdefault("TRANSFORMERS_NO_ADVISORY_WARNINGS", "true")


class C4DataModule:
    def __init__(
        self,
        tokenizer: str,
        max_seq_len: int,
        min_seq_len: Optional[int] = None,
        batch_size: int = 4,
        shuffle_window_seed: int = 0,
        shuffle_window_size: int = 10000,
        concat_batch_size: int = 16,
        num_train_workers: int = 2,
        num_valid_workers: int = 1,
        padding_side: Optional[str] = None,
        pin_memory: bool = True,
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
    ):
        from transformers import AutoTokenizer, PreTrainedTokenizerBase

        if isinstance(tokenizer, PreTrainedTokenizerBase):
            # instance accepted for offline tests, mirroring resolve_tokenizer
            self._tokenizer_obj = tokenizer
            tokenizer = type(tokenizer).__name__
        else:
            self._tokenizer_obj = None
        self.hparams = Hparams(
            tokenizer=tokenizer, max_seq_len=max_seq_len, min_seq_len=min_seq_len,
            batch_size=batch_size, shuffle_window_seed=shuffle_window_seed,
            shuffle_window_size=shuffle_window_size, concat_batch_size=concat_batch_size,
            num_train_workers=num_train_workers, num_valid_workers=num_valid_workers,
            padding_side=padding_side, pin_memory=pin_memory, rank=rank, world_size=world_size,
        )
        self.tokenizer = (self._tokenizer_obj if self._tokenizer_obj is not None
                          else AutoTokenizer.from_pretrained(tokenizer, verbose=False))
        self.collator = C4Collator(self.tokenizer)
        if padding_side is not None:
            self.tokenizer.padding_side = padding_side
        self.ds_train = None
        self.ds_valid = None

    @property
    def vocab_size(self):
        return self.tokenizer.vocab_size

    @property
    def max_seq_len(self):
        return self.hparams.max_seq_len

    @property
    def rank(self):
        return get_rank() if self.hparams.rank is None else self.hparams.rank

    @property
    def world_size(self):
        return get_world_size() if self.hparams.world_size is None else self.hparams.world_size

    def prepare_data(self):
        pass  # streaming: nothing to prepare

    def text_preprocessor(self) -> TextPreprocessor:
        return TextPreprocessor(tokenizer=self.hparams.tokenizer,
                                max_seq_len=self.hparams.max_seq_len, add_special_tokens=False)

    def _create_dataset(self, split):
        from datasets import load_dataset
        from datasets.distributed import split_dataset_by_node

        dataset = load_dataset("c4", "en", split=split, streaming=True)
        dataset = dataset.shuffle(seed=self.hparams.shuffle_window_seed,
                                  buffer_size=self.hparams.shuffle_window_size)
        return split_dataset_by_node(dataset, rank=self.rank, world_size=self.world_size)

    def _tokenize_fn(self):
        def tokenize(examples):
            return self.tokenizer(
                examples["text"], padding=False, truncation=False, max_length=None,
                add_special_tokens=False, return_token_type_ids=False,
                return_attention_mask=False,
            )
        return tokenize

    def _next_chunk_len(self, min_seq_len):
        """Chunk length incl. the shift token: fixed, or random in
        [min, max] for variable-length training."""
        if min_seq_len is None:
            return self.hparams.max_seq_len + 1
        return int(torch.randint(min_seq_len, self.hparams.max_seq_len + 1, size=(1,))) + 1

    def _chunk_fn(self, min_seq_len):
        eos = self.tokenizer.eos_token_id

        def chunk(examples):
            # join the documents of this map-batch into one EOS-separated
            # stream and cut it into (variable-length) training chunks; a
            # trailing partial chunk is dropped with its batch
            chunks = []
            current = []
            want = self._next_chunk_len(min_seq_len)
            for doc in examples["input_ids"]:
                for token_id in (*doc, eos):
                    current.append(token_id)
                    if len(current) == want:
                        chunks.append(current)
                        current = []
                        want = self._next_chunk_len(min_seq_len)
            if not chunks:
                return []
            examples["input_ids"] = chunks
            return examples
        return chunk

    def _create_pipeline(self, dataset, min_seq_len=None):
        tokenized = dataset.map(self._tokenize_fn(), batched=True,
                                remove_columns=["text", "timestamp", "url"])
        return tokenized.map(self._chunk_fn(min_seq_len), batched=True,
                             batch_size=self.hparams.concat_batch_size)

    def setup(self, stage=None):
        self.ds_train = self._create_pipeline(self._create_dataset("train"),
                                              min_seq_len=self.hparams.min_seq_len)
        self.ds_valid = self._create_pipeline(self._create_dataset("validation"), min_seq_len=None)

    def train_dataloader(self):
        return DataLoader(self.ds_train, collate_fn=self.collator,
                          batch_size=self.hparams.batch_size,
                          num_workers=self.hparams.num_train_workers,
                          pin_memory=self.hparams.pin_memory)

    def val_dataloader(self):
        return DataLoader(self.ds_valid, collate_fn=self.collator,
                          batch_size=self.hparams.batch_size,
                          num_workers=self.hparams.num_valid_workers,
                          pin_memory=self.hparams.pin_memory)


class C4Collator(Collator):
    """Pad the chunked streams, then shift by one for next-token prediction."""

    def __init__(self, tokenizer):
        self.tokenizer = tokenizer

    def collate(self, examples):
        batch = self.tokenizer.pad(examples, return_attention_mask=True, return_tensors="pt")
        window = batch["input_ids"]
        batch["labels"] = window[..., 1:]
        batch["input_ids"] = window[..., :-1]
        batch["attention_mask"] = batch["attention_mask"][..., :-1]
        return batch
