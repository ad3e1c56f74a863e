"""Batch collators for the text tasks.

Every collator is a callable over tokenized examples producing the batch
contract the task models train on: ``(labels, input_ids, pad_mask)`` with
``pad_mask == True`` at padding positions (the attention-mask inversion
happens here, once). Masking behavior mirrors the reference's
data/text/collator.py: whole-word masking selects words at ``mask_prob`` and
applies the BERT-style 80/10/10 mask/random/keep split per WORD (every token
of a selected word receives the same treatment).
"""
from __future__ import annotations

from typing import Dict, List, Optional

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
    """Base: subclasses implement ``collate`` returning a 🤗 batch dict."""

    def collate(self, examples):
        raise NotImplementedError

    def __call__(self, examples):
        batch = self.collate(examples)
        pad_mask = ~batch["attention_mask"].type(torch.bool)
        return batch["labels"], batch["input_ids"], pad_mask


class RandomTruncateCollator(Collator):
    """Wraps a collator and randomly right-truncates each batch.

    The surviving length is uniform over [min_seq_len, seq_len); used to make
    the MLM robust to variable-length inference inputs.
    """

    def __init__(self, collator: Collator, min_seq_len: int):
        self.inner = collator
        self.min_seq_len = min_seq_len

    @property
    def collator(self):  # underlying collator (API parity)
        return self.inner

    def collate(self, examples):
        batch = self.inner.collate(examples)
        seq_len = batch["input_ids"].shape[1]
        if seq_len > self.min_seq_len:
            drop = int(torch.randint(1, seq_len - self.min_seq_len + 1, size=(1,)))
            for key in ("labels", "input_ids", "attention_mask"):
                batch[key] = batch[key][:, :-drop]
        return batch


class DefaultCollator(Collator):
    """Pad/truncate through ``tokenizer.prepare_for_model``.

    ``add_special_tokens`` stays off — the chunked training records already
    carry their special tokens, and re-adding would duplicate them. Label
    keys (classification targets, CLM label_ids) pass through.
    """

    label_keys = ("label", "labels")

    def __init__(self, tokenizer: PreTrainedTokenizerFast, max_seq_len: Optional[int] = None):
        self.collator = DefaultDataCollator()
        self.tokenizer = tokenizer
        self.max_seq_len = max_seq_len

    def collate(self, examples):
        longest = max(len(e["input_ids"]) for e in examples)
        target = min(longest, self.max_seq_len) if self.max_seq_len else longest
        return self.collator([self._prepare(e, target) for e in examples])

    def _pad_or_truncate(self, sequence, target_len):
        return self.tokenizer.prepare_for_model(
            sequence,
            add_special_tokens=False,
            return_token_type_ids=False,
            padding=False if self.tokenizer.pad_token is None else PaddingStrategy.MAX_LENGTH,
            max_length=target_len,
            truncation=True,
        )

    def _prepare(self, example, target_len):
        out = self._pad_or_truncate(example["input_ids"], target_len)
        if "label_ids" in example:
            out["label_ids"] = self._pad_or_truncate(example["label_ids"], target_len)["input_ids"]
        for key in self.label_keys:
            if key in example:
                out[key] = example[key]
        return out


def word_spans(word_ids: List[Optional[int]]) -> List[List[int]]:
    """Token-index groups per word: consecutive equal non-None ids form one
    word; None (special tokens) breaks and never joins a word."""
    spans: List[List[int]] = []
    previous = object()
    for idx, wid in enumerate(word_ids):
        if wid is None:
            previous = None
            continue
        if wid != previous:
            spans.append([])
            previous = wid
        spans[-1].append(idx)
    return spans


class WordMaskingCollator(Collator):
    """Whole-word masking (80/10/10 applied per word, not per token)."""

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

    def mask_words_1(self, example: Dict) -> Dict:
        """Mutates ``example`` in place (the same routine serves dynamic
        per-batch masking and one-shot static masking)."""
        spans = word_spans(example.pop("word_ids"))
        input_ids = example["input_ids"]
        labels = [-100] * len(input_ids)

        selected = np.random.binomial(1, self.mask_prob, len(spans))
        for word_index in np.where(selected)[0]:
            u_mask, u_rand = np.random.rand(2)
            for idx in spans[word_index]:
                labels[idx] = input_ids[idx]
                if u_mask < 0.8:
                    input_ids[idx] = self.mask_token_id    # 80%: mask token(s)
                elif u_rand < 0.5:
                    # 10%: an independent random token per position
                    input_ids[idx] = np.random.randint(self.vocab_size)
                # remaining 10%: tokens kept unchanged

        example["labels"] = labels
        return example


class TokenMaskingCollator(Collator):
    """Per-token masking via 🤗's standard MLM collator."""

    def __init__(self, tokenizer: PreTrainedTokenizerFast, mask_prob: float = 0.15):
        self.collator = DataCollatorForLanguageModeling(tokenizer, mlm_probability=mask_prob)

    def collate(self, examples):
        return self.collator(examples)
