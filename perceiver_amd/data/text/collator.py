"""Batch collators for text tasks. Batch contract (parity with reference
data/text/collator.py): calling a collator returns ``(labels, input_ids, pad_mask)``
with ``pad_mask=True`` at padding positions."""
from __future__ import annotations

from collections import defaultdict
from typing import Optional

import numpy as np
import torch
from transformers import (
    DataCollatorForLanguageModeling,
    DataCollatorWithPadding,
    DefaultDataCollator,
    PreTrainedTokenizerFast,
)
from transformers.utils import PaddingStrategy


class Collator:
    def collate(self, examples):
        raise NotImplementedError

    def __call__(self, examples):
        result = self.collate(examples)
        return result["labels"], result["input_ids"], ~result["attention_mask"].type(torch.bool)


class RandomTruncateCollator(Collator):
    """Randomly right-truncates each batch down to >= min_seq_len tokens."""

    def __init__(self, collator: Collator, min_seq_len: int):
        self.collator = collator
        self.min_seq_len = min_seq_len

    def collate(self, examples):
        result = self.collator.collate(examples)
        seq_len = result["input_ids"].shape[1]
        if seq_len <= self.min_seq_len:
            return result
        drop = int(torch.randint(1, seq_len - self.min_seq_len + 1, size=(1,)))
        for key in ("labels", "input_ids", "attention_mask"):
            result[key] = result[key][:, :-drop]
        return result


class DefaultCollator(Collator):
    """Pads/truncates via tokenizer.prepare_for_model; passes through label keys."""

    label_keys = ("label", "labels")

    def __init__(self, tokenizer: PreTrainedTokenizerFast, max_seq_len: Optional[int] = None):
        self.collator = DefaultDataCollator()
        self.tokenizer = tokenizer
        self.max_seq_len = max_seq_len

    def collate(self, examples):
        cur_length = max(len(e["input_ids"]) for e in examples)
        max_length = min(cur_length, self.max_seq_len) if self.max_seq_len else cur_length
        return self.collator([self._prepare(e, max_length) for e in examples])

    def _prepare(self, example, max_length):
        prepared = self._prepare_sequence(example["input_ids"], max_length)
        if "label_ids" in example:
            prepared["label_ids"] = self._prepare_sequence(example["label_ids"], max_length)["input_ids"]
        for key in self.label_keys:
            if key in example:
                prepared[key] = example[key]
        return prepared

    def _prepare_sequence(self, sequence, max_length):
        return self.tokenizer.prepare_for_model(
            sequence,
            add_special_tokens=False,
            return_token_type_ids=False,
            padding=False if self.tokenizer.pad_token is None else PaddingStrategy.MAX_LENGTH,
            max_length=max_length,
            truncation=True,
        )


class WordMaskingCollator(Collator):
    """Whole-word masking with the 80/10/10 mask/random/keep split applied per word
    (all tokens of a selected word get the same treatment)."""

    def __init__(self, tokenizer: PreTrainedTokenizerFast, mask_prob: float = 0.15):
        self.collator = DataCollatorWithPadding(tokenizer)
        self.mask_token_id = tokenizer.mask_token_id
        self.vocab_size = tokenizer.vocab_size
        self.mask_prob = mask_prob

    def collate(self, examples):
        return self.collator(self.mask_words(examples))

    def mask_words(self, examples):
        for example in examples:
            self.mask_words_1(example)
        return examples

    def mask_words_1(self, example):
        # mutates its argument (used both dynamically per batch and for static masking)
        word_ids = example.pop("word_ids")
        input_ids = example["input_ids"]
        labels = [-100] * len(input_ids)

        mapping = defaultdict(list)
        current_word_index = -1
        current_word_id = None
        for idx, word_id in enumerate(word_ids):
            if word_id is not None:
                if word_id != current_word_id:
                    current_word_id = word_id
                    current_word_index += 1
                mapping[current_word_index].append(idx)

        mask = np.random.binomial(1, self.mask_prob, len(mapping))
        for word_index in np.where(mask)[0]:
            rand_nr = np.random.rand(2)
            for idx in mapping[word_index]:
                labels[idx] = input_ids[idx]
                if rand_nr[0] < 0.8:
                    input_ids[idx] = self.mask_token_id        # 80%: mask token(s)
                elif rand_nr[1] < 0.5:
                    input_ids[idx] = np.random.randint(self.vocab_size)  # 10%: random
                # else 10%: unchanged

        example["labels"] = labels
        return example


class TokenMaskingCollator(Collator):
    def __init__(self, tokenizer: PreTrainedTokenizerFast, mask_prob: float = 0.15):
        self.collator = DataCollatorForLanguageModeling(tokenizer, mlm_probability=mask_prob)

    def collate(self, examples):
        return self.collator(examples)
