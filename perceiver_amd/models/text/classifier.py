"""Sequence-classification Perceiver IO backend.

A byte/token sequence runs through the shared ``TextEncoder`` (the same
encoder class the MLM trains — transfer learning loads its weights directly),
and a small classification head decodes the latent array from one or more
learned output queries into class logits.

Behavioral contract mirrored from the reference's text classifier backend
(/root/reference/perceiver/model/text/classifier/backend.py:15-46); the
state-dict layout (``encoder.*`` / ``decoder.*``) is checkpoint-compatible.
"""
from __future__ import annotations

from perceiver_amd.core import ClassificationDecoderConfig, PerceiverIO, PerceiverIOConfig
from perceiver_amd.models.builders import classification_decoder, latent_kwargs
from perceiver_amd.models.text.common import TextEncoder, TextEncoderConfig

TextClassifierConfig = PerceiverIOConfig[TextEncoderConfig, ClassificationDecoderConfig]


class TextClassifier(PerceiverIO):
    """TextEncoder + classification decoder."""

    def __init__(self, config: TextClassifierConfig):
        super().__init__(
            TextEncoder(
                config.encoder,
                num_latents=config.num_latents,
                num_latent_channels=config.num_latent_channels,
                **latent_kwargs(config),
            ),
            classification_decoder(config),
        )
        self.config = config

    def forward(self, x, pad_mask=None):
        return self.decoder(self.encoder(x, pad_mask=pad_mask))
