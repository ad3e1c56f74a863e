"""Optical flow (Perceiver IO): temporal-concat patch projection + Fourier encodings;
the decoder queries the latents with the *adapted input* (identity query provider)
and maps to per-pixel (dx, dy) flow.

Parity: /root/reference/perceiver/model/vision/optical_flow/backend.py:22-137.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import torch
import torch.nn as nn

from perceiver_amd.core import (
    DecoderConfig,
    EncoderConfig,
    FourierPositionEncoding,
    InputAdapter,
    OutputAdapter,
    PerceiverDecoder,
    PerceiverEncoder,
    PerceiverIO,
    PerceiverIOConfig,
    QueryProvider,
)


@dataclass
class OpticalFlowEncoderConfig(EncoderConfig):
    image_shape: Tuple[int, int] = (368, 496)
    num_patch_input_channels: int = 27
    num_patch_hidden_channels: int = 64
    num_frequency_bands: int = 64

    def base_kwargs(
        self,
        exclude=("freeze", "image_shape", "num_patch_input_channels", "num_patch_hidden_channels", "num_frequency_bands"),
    ):
        return super().base_kwargs(exclude=exclude)


@dataclass
class OpticalFlowDecoderConfig(DecoderConfig):
    image_shape: Tuple[int, int] = (368, 496)
    rescale_factor: float = 100.0

    def base_kwargs(self, exclude=("freeze", "image_shape", "rescale_factor")):
        return super().base_kwargs(exclude=exclude)


OpticalFlowConfig = PerceiverIOConfig[OpticalFlowEncoderConfig, OpticalFlowDecoderConfig]


class OpticalFlowInputAdapter(InputAdapter):
    """(b, t, c, h, w) -> concat temporal frames along channels -> Linear patch
    projection -> flatten + Fourier position-encoding concat."""

    def __init__(
        self,
        image_shape: Tuple[int, int],
        num_patch_input_channels: int,
        num_patch_hidden_channels: int,
        num_frequency_bands: int,
    ):
        position_encoding = FourierPositionEncoding(input_shape=image_shape, num_frequency_bands=num_frequency_bands)
        super().__init__(num_patch_hidden_channels + position_encoding.num_position_encoding_channels())
        self.linear = nn.Linear(num_patch_input_channels * 2, num_patch_hidden_channels)
        self.position_encoding = position_encoding

    def forward(self, x):
        b, t, c, h, w = x.shape
        # b t c h w -> b h w (t c)
        x = x.permute(0, 3, 4, 1, 2).reshape(b, h, w, t * c)
        x = self.linear(x)
        x = x.flatten(1, 2)
        return torch.cat([x, self.position_encoding(b).to(x.dtype)], dim=-1)


class OpticalFlowOutputAdapter(OutputAdapter):
    def __init__(
        self,
        image_shape: Tuple[int, int],
        num_output_query_channels: int,
        num_output_image_channels: int = 2,
        rescale_factor: float = 100.0,
    ):
        super().__init__()
        self.image_shape = tuple(image_shape)
        self.rescale_factor = rescale_factor
        self.linear = nn.Linear(num_output_query_channels, num_output_image_channels)

    def forward(self, x):
        x = self.linear(x) / self.rescale_factor
        b, _, c = x.shape
        return x.view(b, self.image_shape[0], self.image_shape[1], c)


class OpticalFlowQueryProvider(nn.Module, QueryProvider):
    """Identity query provider: the decoder queries with the adapted input."""

    def __init__(self, num_query_channels: int):
        super().__init__()
        self._num_query_channels = num_query_channels

    @property
    def num_query_channels(self):
        return self._num_query_channels

    def forward(self, x):
        assert x.shape[-1] == self.num_query_channels
        return x


class OpticalFlow(PerceiverIO):
    def __init__(self, config: OpticalFlowConfig):
        input_adapter = OpticalFlowInputAdapter(
            image_shape=config.encoder.image_shape,
            num_patch_input_channels=config.encoder.num_patch_input_channels,
            num_patch_hidden_channels=config.encoder.num_patch_hidden_channels,
            num_frequency_bands=config.encoder.num_frequency_bands,
        )
        encoder_kwargs = config.encoder.base_kwargs()
        if encoder_kwargs["num_cross_attention_qk_channels"] is None:
            encoder_kwargs["num_cross_attention_qk_channels"] = input_adapter.num_input_channels
        if encoder_kwargs["num_cross_attention_v_channels"] is None:
            encoder_kwargs["num_cross_attention_v_channels"] = input_adapter.num_input_channels

        encoder = PerceiverEncoder(
            input_adapter=input_adapter,
            num_latents=config.num_latents,
            num_latent_channels=config.num_latent_channels,
            activation_checkpointing=config.activation_checkpointing,
            activation_offloading=config.activation_offloading,
            **encoder_kwargs,
        )
        output_adapter = OpticalFlowOutputAdapter(
            image_shape=config.decoder.image_shape,
            num_output_query_channels=input_adapter.num_input_channels,
            rescale_factor=config.decoder.rescale_factor,
        )
        output_query_provider = OpticalFlowQueryProvider(num_query_channels=input_adapter.num_input_channels)
        decoder = PerceiverDecoder(
            output_adapter=output_adapter,
            output_query_provider=output_query_provider,
            num_latent_channels=config.num_latent_channels,
            activation_checkpointing=config.activation_checkpointing,
            activation_offloading=config.activation_offloading,
            **config.decoder.base_kwargs(),
        )
        super().__init__(encoder, decoder)
        self.config = config

    def forward(self, x: torch.Tensor):
        x_latent, x_adapted = self.encoder(x, return_adapted_input=True)
        return self.decoder(x_latent, x_adapted=x_adapted)
