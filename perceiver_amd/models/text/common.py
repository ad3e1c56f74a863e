"""The text encoder shared by the MLM and the text classifier.

``TextEncoder`` is a ``PerceiverEncoder`` whose input adapter embeds token ids
(plus learned absolute positions) — the piece both text tasks share, and the
piece transfer learning moves between them: a classifier built on a trained
MLM encoder loads these weights unchanged.

Behavioral contract mirrored from the reference
(/root/reference/perceiver/model/text/common/backend.py:8-40); the config
fields carried on top of ``EncoderConfig`` (vocab/seq-len/embedding width and
the ``params`` checkpoint pointer) are the CLI/checkpoint schema.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from perceiver_amd.core import EncoderConfig, PerceiverEncoder, TokenInputAdapter
from perceiver_amd.core.utils import freeze


@dataclass
class TextEncoderConfig(EncoderConfig):
    vocab_size: int = 10003
    max_seq_len: int = 256
    num_input_channels: int = 64
    params: Optional[str] = None

    def base_kwargs(self, exclude=("freeze", "vocab_size", "max_seq_len", "num_input_channels", "params")):
        return super().base_kwargs(exclude=exclude)


def _token_adapter(config: TextEncoderConfig) -> TokenInputAdapter:
    """Token-id embedding + learned absolute positions, sized by the config."""
    return TokenInputAdapter(
        vocab_size=config.vocab_size,
        max_seq_len=config.max_seq_len,
        num_input_channels=config.num_input_channels,
    )


class TextEncoder(PerceiverEncoder):
    """PerceiverEncoder over embedded tokens; optionally frozen for transfer."""

    def __init__(
        self,
        config: TextEncoderConfig,
        num_latents: int,
        num_latent_channels: int,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
    ):
        build_kwargs = dict(config.base_kwargs())
        build_kwargs.update(
            input_adapter=_token_adapter(config),
            num_latents=num_latents,
            num_latent_channels=num_latent_channels,
            activation_checkpointing=activation_checkpointing,
            activation_offloading=activation_offloading,
        )
        super().__init__(**build_kwargs)
        if config.freeze:
            freeze(self)
