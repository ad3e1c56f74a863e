"""Input/output adapters and query providers.

Parity with /root/reference/perceiver/model/core/adapter.py:8-150 (InputAdapter,
RotarySupport, OutputAdapter, ClassificationOutputAdapter, QueryProvider,
TrainableQueryProvider, TokenInputAdapter(+WithRotarySupport), TiedTokenOutputAdapter).
State-dict key names (_query, txt_embedding, pos_embedding, frq_pos_encoding, linear,
bias) match the reference checkpoint layout.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from perceiver_amd.core.position import FrequencyPositionEncoding, positions


class InputAdapter(nn.Module):
    """Transforms and position-encodes task-specific input to generic encoder input."""

    def __init__(self, num_input_channels: int, *args, **kwargs):
        super().__init__()
        self._num_input_channels = num_input_channels

    @property
    def num_input_channels(self) -> int:
        return self._num_input_channels


class RotarySupport(InputAdapter):
    """Mixin: additionally emits a frequency position encoding for rotary embeddings."""

    def __init__(self, rotated_channels_per_head: int, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.frq_pos_encoding = FrequencyPositionEncoding(dim=rotated_channels_per_head)

    def forward(self, x, abs_pos=None):
        if abs_pos is None:
            abs_pos = positions(*x.shape, device=x.device)
        return super().forward(x, abs_pos), self.frq_pos_encoding(abs_pos)


class OutputAdapter(nn.Module):
    """Transforms generic decoder cross-attention output to task-specific output."""


class ClassificationOutputAdapter(OutputAdapter):
    def __init__(self, num_classes: int, num_output_query_channels: int):
        super().__init__()
        self.linear = nn.Linear(num_output_query_channels, num_classes)

    def forward(self, x):
        return self.linear(x).squeeze(dim=1)


class QueryProvider:
    """Provider of cross-attention query input."""

    @property
    def num_query_channels(self) -> int:
        raise NotImplementedError

    def __call__(self, x=None):
        raise NotImplementedError


class TrainableQueryProvider(nn.Module, QueryProvider):
    """Learned query array — the latent array of encoders and the output query of
    most decoders."""

    def __init__(self, num_queries: int, num_query_channels: int, init_scale: float = 0.02):
        super().__init__()
        self._query = nn.Parameter(torch.empty(num_queries, num_query_channels))
        with torch.no_grad():
            self._query.normal_(0.0, init_scale)

    @property
    def num_query_channels(self) -> int:
        return self._query.shape[-1]

    def forward(self, x=None):
        return self._query.unsqueeze(0)


class TokenInputAdapter(InputAdapter):
    """Token embedding + optional learned absolute position embedding. For inputs
    shorter than the supplied position codes, the right-most codes are used
    (right alignment, reference adapter.py:105-114)."""

    def __init__(self, vocab_size: int, max_seq_len: int, num_input_channels: int, abs_pos_emb: bool = True):
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

    def forward(self, x, abs_pos: Optional[torch.Tensor] = None):
        if not self._abs_pos_emb:
            return self.txt_embedding(x)
        if abs_pos is None:
            abs_pos = positions(*x.shape, device=x.device)
        elif x.shape[1] < abs_pos.shape[1]:
            abs_pos = abs_pos[:, -x.shape[1]:]
        return self.txt_embedding(x) + self.pos_embedding(abs_pos)


class TokenInputAdapterWithRotarySupport(RotarySupport, TokenInputAdapter):
    def __init__(
        self,
        rotated_channels_per_head: int,
        vocab_size: int,
        max_seq_len: int,
        num_input_channels: int,
        abs_pos_emb: bool,
    ):
        super().__init__(
            rotated_channels_per_head=rotated_channels_per_head,
            vocab_size=vocab_size,
            max_seq_len=max_seq_len,
            num_input_channels=num_input_channels,
            abs_pos_emb=abs_pos_emb,
        )

    def forward(self, x, abs_pos=None):
        return super().forward(x, abs_pos)


class TiedTokenOutputAdapter(OutputAdapter):
    """logits = x @ txt_embedding.weight.T (+ bias). The embedding is passed at call
    time so weights stay tied. On GPU this is the fused norm+tied-logits GEMM
    (SURVEY.md §2.3 K8)."""

    def __init__(self, vocab_size: int, emb_bias: bool = True):
        super().__init__()
        self._emb_bias = emb_bias
        if emb_bias:
            self.bias = nn.Parameter(torch.zeros(vocab_size))

    def forward(self, x, txt_embedding: nn.Embedding):
        result = torch.matmul(x, txt_embedding.weight.t())
        if self._emb_bias:
            result = result + self.bias
        return result
