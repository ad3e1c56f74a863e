"""Masked-language-model Perceiver IO backend.

The encoder compresses the (masked) token sequence into the latent array; the
decoder queries it with ``max_seq_len`` learned output queries and projects to
vocabulary logits. Two heads exist, selected by
``decoder.num_output_query_channels``:

* ``None`` → tied head: output queries live in the encoder's input-channel
  space and logits come from the shared token embedding
  (``TiedTokenOutputAdapter``, x @ emb.Tᵀ) — the deepmind/language-perceiver
  layout.
* an int → untied head: a separate Linear projects query channels to vocab.

Behavioral contract mirrored from the reference MLM backend
(/root/reference/perceiver/model/text/mlm/backend.py:18-89); state-dict keys
are checkpoint-compatible.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch.nn as nn

from perceiver_amd.core import (
    DecoderConfig,
    OutputAdapter,
    PerceiverDecoder,
    PerceiverIO,
    PerceiverIOConfig,
    TiedTokenOutputAdapter,
)
from perceiver_amd.models.builders import assemble_decoder, latent_kwargs, learned_queries
from perceiver_amd.models.text.common import TextEncoder, TextEncoderConfig


@dataclass
class TextDecoderConfig(DecoderConfig):
    num_output_query_channels: Optional[int] = None
    vocab_size: int = 10003
    max_seq_len: int = 512

    def base_kwargs(self, exclude=("freeze", "num_output_query_channels", "vocab_size", "max_seq_len")):
        return super().base_kwargs(exclude=exclude)


MaskedLanguageModelConfig = PerceiverIOConfig[TextEncoderConfig, TextDecoderConfig]


class TokenOutputAdapter(OutputAdapter):
    """Untied logits head: its own Linear from query channels to vocab."""

    def __init__(self, vocab_size: int, num_output_query_channels: int):
        super().__init__()
        self.linear = nn.Linear(num_output_query_channels, vocab_size)

    def forward(self, x):
        return self.linear(x).squeeze(dim=1)


def _token_logits_decoder(config: MaskedLanguageModelConfig) -> PerceiverDecoder:
    dc = config.decoder
    tied = dc.num_output_query_channels is None
    query_channels = config.encoder.num_input_channels if tied else dc.num_output_query_channels
    head = (TiedTokenOutputAdapter(vocab_size=dc.vocab_size) if tied
            else TokenOutputAdapter(vocab_size=dc.vocab_size,
                                    num_output_query_channels=dc.num_output_query_channels))
    queries = learned_queries(dc.max_seq_len, query_channels, dc.init_scale)
    return assemble_decoder(head, queries, config, dc)


class MaskedLanguageModel(PerceiverIO):
    """TextEncoder + token-logits decoder (tied or untied)."""

    def __init__(self, config: MaskedLanguageModelConfig):
        super().__init__(
            TextEncoder(
                config.encoder,
                num_latents=config.num_latents,
                num_latent_channels=config.num_latent_channels,
                **latent_kwargs(config),
            ),
            _token_logits_decoder(config),
        )
        self.config = config

    @property
    def _tied(self) -> bool:
        return isinstance(self.decoder.output_adapter, TiedTokenOutputAdapter)

    def forward(self, x_masked, pad_mask=None):
        seq_len = x_masked.shape[1]
        latents = self.encoder(x_masked, pad_mask)
        if self._tied:
            # the tied adapter receives the token embedding at call time
            logits = self.decoder(latents, txt_embedding=self.encoder.input_adapter.txt_embedding)
        else:
            logits = self.decoder(latents)
        # the decoder always produces max_seq_len rows; crop to the input length
        return logits[:, :seq_len, :]


# alias kept for API parity with the reference
TextDecoder = PerceiverDecoder
