"""Flagship model configurations — the named configs of BASELINE.json / BASELINE.md.

These mirror the reference's CLI set_defaults (e.g. the 201M-parameter Perceiver IO
MLM of /root/reference/perceiver/scripts/text/mlm.py:16-39) and are what bench.py and
the examples instantiate.
"""
from __future__ import annotations

from perceiver_amd.core import ClassificationDecoderConfig
from perceiver_amd.models.text.clm import CausalLanguageModelConfig
from perceiver_amd.models.text.common import TextEncoderConfig
from perceiver_amd.models.text.mlm import MaskedLanguageModelConfig, TextDecoderConfig
from perceiver_amd.models.vision.image_classifier import ImageClassifierConfig, ImageEncoderConfig
from perceiver_amd.models.vision.optical_flow import (
    OpticalFlowConfig,
    OpticalFlowDecoderConfig,
    OpticalFlowEncoderConfig,
)


def mlm_flagship(vocab_size: int = 262, max_seq_len: int = 2048,
                 num_latents: int = 512) -> MaskedLanguageModelConfig:
    """Perceiver IO MLM flagship: 1280-channel latents, 26 self-attention layers,
    qk 256 / v 1280, UTF-8 bytes seq 2048. BASELINE.json names 512 latents (the
    reference CLI default is 256; same architecture otherwise)."""
    return MaskedLanguageModelConfig(
        encoder=TextEncoderConfig(
            vocab_size=vocab_size,
            max_seq_len=max_seq_len,
            num_input_channels=768,
            num_cross_attention_layers=1,
            num_cross_attention_qk_channels=256,
            num_cross_attention_v_channels=1280,
            num_cross_attention_heads=8,
            num_self_attention_qk_channels=256,
            num_self_attention_v_channels=1280,
            num_self_attention_heads=8,
            num_self_attention_layers_per_block=26,
            num_self_attention_blocks=1,
            dropout=0.1,
        ),
        decoder=TextDecoderConfig(
            vocab_size=vocab_size,
            max_seq_len=max_seq_len,
            num_cross_attention_qk_channels=256,
            num_cross_attention_v_channels=768,
            num_cross_attention_heads=8,
            cross_attention_residual=False,
            dropout=0.1,
        ),
        num_latents=num_latents,
        num_latent_channels=1280,
    )


def image_classifier_flagship(num_classes: int = 1000, image_shape=(224, 224, 3)) -> ImageClassifierConfig:
    """Perceiver IO image classifier: 512 latents x 1024, 8 blocks x 6 layers,
    64 Fourier bands (~50k KV rows at 224^2)."""
    return ImageClassifierConfig(
        encoder=ImageEncoderConfig(
            image_shape=tuple(image_shape),
            num_frequency_bands=64,
            num_cross_attention_layers=1,
            num_cross_attention_heads=1,
            num_self_attention_heads=8,
            num_self_attention_layers_per_block=6,
            num_self_attention_blocks=8,
            dropout=0.1,
        ),
        decoder=ClassificationDecoderConfig(
            num_classes=num_classes,
            num_output_query_channels=1024,
            num_cross_attention_heads=1,
            dropout=0.1,
        ),
        num_latents=512,
        num_latent_channels=1024,
    )


def optical_flow_flagship() -> OpticalFlowConfig:
    """Perceiver IO optical flow: 368x496 frame pairs (~182k KV rows), 2048 latents x 512,
    24 self-attention layers."""
    return OpticalFlowConfig(
        encoder=OpticalFlowEncoderConfig(
            image_shape=(368, 496),
            num_patch_input_channels=27,
            num_patch_hidden_channels=64,
            num_frequency_bands=64,
            num_cross_attention_heads=1,
            num_self_attention_heads=16,
            num_self_attention_layers_per_block=24,
            num_self_attention_blocks=1,
        ),
        decoder=OpticalFlowDecoderConfig(image_shape=(368, 496), num_cross_attention_heads=1),
        num_latents=2048,
        num_latent_channels=512,
    )


def clm_flagship(vocab_size: int = 262) -> CausalLanguageModelConfig:
    """Perceiver-AR causal LM for the 8192-ctx / 1024-latent decode benchmark."""
    return CausalLanguageModelConfig(
        vocab_size=vocab_size,
        max_seq_len=8192,
        max_latents=1024,
        num_channels=1024,
        num_heads=8,
        num_self_attention_layers=12,
        cross_attention_dropout=0.5,
        abs_pos_emb=False,
    )


def clm_wikitext() -> CausalLanguageModelConfig:
    """30.7M Perceiver-AR of the WikiText-103 bytes example: ctx 4096, 512 latents,
    512 channels, 9 layers."""
    return CausalLanguageModelConfig(
        vocab_size=262,
        max_seq_len=4096,
        max_latents=512,
        num_channels=512,
        num_heads=8,
        num_self_attention_layers=8,
        cross_attention_dropout=0.5,
    )
