"""Adapters and query providers — the task-facing edges of every Perceiver.

Three small families:

* **Query providers** produce the cross-attention query array: the learned
  latent array of encoders, the learned output queries of most decoders, or
  (optical flow) the adapted input itself.
* **Input adapters** map raw task input (token ids, pixels, patch features)
  to the generic (B, L, C) array the encoder cross-attends into, attaching
  position information — learned absolute embeddings, Fourier codes, or (for
  Perceiver-AR) rotary frequency codes via the ``RotarySupport`` mixin.
* **Output adapters** map decoded query vectors to task output (class
  logits, vocabulary logits, flow vectors).

The class surface and the parameter names (``_query``, ``txt_embedding``,
``pos_embedding``, ``frq_pos_encoding``, ``linear``, ``bias``) are the
checkpoint contract shared with the reference (model/core/adapter.py:8-150);
the implementations are this repo's own.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from perceiver_amd.core.position import FrequencyPositionEncoding, positions


# ---------------------------------------------------------------- query side
class QueryProvider:
    """Protocol for producers of cross-attention query input."""

    @property
    def num_query_channels(self) -> int:
        raise NotImplementedError

    def __call__(self, x=None):
        raise NotImplementedError


class TrainableQueryProvider(nn.Module, QueryProvider):
    """A learned (num_queries, num_channels) array, normal-initialized.

    Serves as the encoder's latent array and as the output-query array of the
    classification/MLM decoders. The optional input is ignored — the queries
    are input-independent.
    """

    def __init__(self, num_queries: int, num_query_channels: int, init_scale: float = 0.02):
        super().__init__()
        self._query = nn.Parameter(torch.empty(num_queries, num_query_channels))
        nn.init.normal_(self._query, mean=0.0, std=init_scale)

    @property
    def num_query_channels(self) -> int:
        return self._query.shape[-1]

    def forward(self, x=None):
        return self._query.unsqueeze(0)


# ---------------------------------------------------------------- input side
class InputAdapter(nn.Module):
    """Base: declares how many channels the adapted input carries."""

    def __init__(self, num_input_channels: int, *args, **kwargs):
        super().__init__()
        self._num_input_channels = num_input_channels

    @property
    def num_input_channels(self) -> int:
        return self._num_input_channels


class RotarySupport(InputAdapter):
    """Mixin for Perceiver-AR adapters: alongside the adapted input, emit the
    rotary frequency codes for the given absolute positions (the attention
    layers rotate q/k with them)."""

    def __init__(self, rotated_channels_per_head: int, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.frq_pos_encoding = FrequencyPositionEncoding(dim=rotated_channels_per_head)

    def forward(self, x, abs_pos=None):
        if abs_pos is None:
            abs_pos = positions(*x.shape, device=x.device)
        return super().forward(x, abs_pos), self.frq_pos_encoding(abs_pos)


class TokenInputAdapter(InputAdapter):
    """Token embedding, optionally plus a learned absolute position embedding.

    Position handling is right-aligned: when the input is shorter than the
    supplied position codes (cached decoding passes full-context positions),
    the trailing codes apply.
    """

    def __init__(self, vocab_size: int, max_seq_len: int, num_input_channels: int,
                 abs_pos_emb: bool = True):
        super().__init__(num_input_channels)
        self._max_seq_len = max_seq_len
        self._abs_pos_emb = abs_pos_emb
        self.txt_embedding = nn.Embedding(vocab_size, num_input_channels)
        if abs_pos_emb:
            self.pos_embedding = nn.Embedding(max_seq_len, num_input_channels)

    @property
    def vocab_size(self) -> int:
        return self.txt_embedding.num_embeddings

    @property
    def max_seq_len(self) -> int:
        return self._max_seq_len

    def _aligned_positions(self, x, abs_pos: Optional[torch.Tensor]) -> torch.Tensor:
        if abs_pos is None:
            return positions(*x.shape, device=x.device)
        if x.shape[1] < abs_pos.shape[1]:
            return abs_pos[:, -x.shape[1]:]
        return abs_pos

    def forward(self, x, abs_pos: Optional[torch.Tensor] = None):
        embedded = self.txt_embedding(x)
        if self._abs_pos_emb:
            embedded = embedded + self.pos_embedding(self._aligned_positions(x, abs_pos))
        return embedded


class TokenInputAdapterWithRotarySupport(RotarySupport, TokenInputAdapter):
    """Token adapter for Perceiver-AR: embeds tokens and emits rotary codes."""

    def __init__(self, rotated_channels_per_head: int, vocab_size: int, max_seq_len: int,
                 num_input_channels: int, abs_pos_emb: bool):
        super().__init__(
            rotated_channels_per_head=rotated_channels_per_head,
            vocab_size=vocab_size,
            max_seq_len=max_seq_len,
            num_input_channels=num_input_channels,
            abs_pos_emb=abs_pos_emb,
        )

    def forward(self, x, abs_pos=None):
        return super().forward(x, abs_pos)


# --------------------------------------------------------------- output side
class OutputAdapter(nn.Module):
    """Base marker: maps decoder cross-attention output to task output."""


class ClassificationOutputAdapter(OutputAdapter):
    """Linear head to class logits; single-query decoders squeeze the query dim."""

    def __init__(self, num_classes: int, num_output_query_channels: int):
        super().__init__()
        self.linear = nn.Linear(num_output_query_channels, num_classes)

    def forward(self, x):
        return self.linear(x).squeeze(dim=1)


class TiedTokenOutputAdapter(OutputAdapter):
    """Vocabulary logits through the (shared) token embedding: x @ emb.Tᵀ + bias.

    The embedding is an argument of ``forward``, not a submodule — weights
    stay tied to the input adapter by construction. On GPU this is the fused
    norm + tied-logits GEMM.
    """

    def __init__(self, vocab_size: int, emb_bias: bool = True):
        super().__init__()
        self._emb_bias = emb_bias
        if emb_bias:
            self.bias = nn.Parameter(torch.zeros(vocab_size))

    def forward(self, x, txt_embedding: nn.Embedding):
        logits = x @ txt_embedding.weight.t()
        return logits + self.bias if self._emb_bias else logits
