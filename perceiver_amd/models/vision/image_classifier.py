"""Image-classification Perceiver IO backend.

Images enter as raw channels-last pixels: the input adapter flattens the
spatial grid and appends a fixed Fourier position code per pixel (sin/cos over
``num_frequency_bands`` frequencies per spatial dim, plus the raw coordinate),
giving a (B, H*W, C_pixel + C_pos) K/V array of ~50k rows at 224². The encoder
cross-attends a few hundred latents into that array; a single learned query
decodes the class logits.

Behavioral contract mirrored from the reference image-classifier backend
(/root/reference/perceiver/model/vision/image_classifier/backend.py:21-96),
including its default of widening the cross-attention qk channels to the
adapter width when unset. State-dict layout is checkpoint-compatible. On GPU
the Fourier-encode + concat prologue is the K2 fusion target.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import torch

from perceiver_amd.core import (
    ClassificationDecoderConfig,
    EncoderConfig,
    FourierPositionEncoding,
    InputAdapter,
    PerceiverEncoder,
    PerceiverIO,
    PerceiverIOConfig,
)
from perceiver_amd.models.builders import classification_decoder, latent_kwargs


@dataclass
class ImageEncoderConfig(EncoderConfig):
    image_shape: Tuple[int, int, int] = (224, 224, 3)
    num_frequency_bands: int = 32

    def base_kwargs(self, exclude=("freeze", "image_shape", "num_frequency_bands")):
        return super().base_kwargs(exclude=exclude)


ImageClassifierConfig = PerceiverIOConfig[ImageEncoderConfig, ClassificationDecoderConfig]


class ImageInputAdapter(InputAdapter):
    """Pixel-flattening adapter with concatenated Fourier position codes."""

    def __init__(self, image_shape: Tuple[int, ...], num_frequency_bands: int):
        spatial, channels = tuple(image_shape[:-1]), image_shape[-1]
        pos = FourierPositionEncoding(input_shape=spatial, num_frequency_bands=num_frequency_bands)
        super().__init__(num_input_channels=channels + pos.num_position_encoding_channels())
        self.image_shape = tuple(image_shape)
        self.position_encoding = pos

    def forward(self, x):
        got = tuple(x.shape[1:])
        if got != self.image_shape:
            raise ValueError(f"Input vision shape {got} different from required shape {self.image_shape}")
        flat = x.flatten(1, -2)  # (B, H*W, C) from channels-last input
        codes = self.position_encoding(x.shape[0]).to(flat.dtype)
        return torch.cat([flat, codes], dim=-1)


def _image_encoder(config: ImageClassifierConfig) -> PerceiverEncoder:
    adapter = ImageInputAdapter(
        image_shape=config.encoder.image_shape,
        num_frequency_bands=config.encoder.num_frequency_bands,
    )
    kwargs = config.encoder.base_kwargs()
    # reference default: qk width follows the (position-encoding-heavy) adapter
    # channels unless the config pins it
    if kwargs.get("num_cross_attention_qk_channels") is None:
        kwargs["num_cross_attention_qk_channels"] = adapter.num_input_channels
    return PerceiverEncoder(
        input_adapter=adapter,
        num_latents=config.num_latents,
        num_latent_channels=config.num_latent_channels,
        **latent_kwargs(config),
        **kwargs,
    )


class ImageClassifier(PerceiverIO):
    """Fourier-encoded pixel encoder + single-query classification decoder."""

    def __init__(self, config: ImageClassifierConfig):
        super().__init__(_image_encoder(config), classification_decoder(config, num_queries=1))
        self.config = config

    def forward(self, x, pad_mask=None):
        return self.decoder(self.encoder(x, pad_mask=pad_mask))


# alias kept for API parity with the reference
ImageEncoder = PerceiverEncoder
