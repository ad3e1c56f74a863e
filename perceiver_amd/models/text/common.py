"""Text encoder shared by MLM and text classification.

Parity: /root/reference/perceiver/model/text/common/backend.py:8-40.
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


class TextEncoder(PerceiverEncoder):
    def __init__(
        self,
        config: TextEncoderConfig,
        num_latents: int,
        num_latent_channels: int,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
    ):
        input_adapter = TokenInputAdapter(
            vocab_size=config.vocab_size,
            max_seq_len=config.max_seq_len,
            num_input_channels=config.num_input_channels,
        )
        super().__init__(
            input_adapter=input_adapter,
            num_latents=num_latents,
            num_latent_channels=num_latent_channels,
            activation_checkpointing=activation_checkpointing,
            activation_offloading=activation_offloading,
            **config.base_kwargs(),
        )
        if config.freeze:
            freeze(self)
