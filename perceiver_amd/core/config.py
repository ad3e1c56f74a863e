"""Model configuration dataclasses.

These dataclasses are the configuration *schema* of the framework: their
field names/defaults are what the CLI exposes as flags, what checkpoints
embed as hyperparameters, and what the 🤗 wrappers serialize into
config.json — so names, ordering and defaults deliberately match the
reference's model/core/config.py:5-100 (they are the compatibility contract;
a renamed field would strand every existing checkpoint and config file).

Layout conventions shared by all attention configs:

* ``num_*_qk_channels`` / ``num_*_v_channels`` may differ from the model
  width and from each other (the MLM flagship runs qk 256 / v 1280); ``None``
  means "follow the query input width".
* ``base_kwargs()`` projects a subclass's fields down to what the underlying
  module constructor accepts, dropping schema-only fields like ``freeze`` —
  each subclass extends the exclusion list with its own extras.
"""
from __future__ import annotations

from dataclasses import asdict, dataclass, fields
from typing import Generic, Optional, TypeVar


def _base_kwargs(config, base_class, exclude):
    """Fields of ``base_class`` present on ``config``, minus ``exclude``."""
    keep = {f.name for f in fields(base_class)}.difference(exclude)
    return {name: value for name, value in asdict(config).items() if name in keep}


@dataclass
class EncoderConfig:
    """PerceiverEncoder hyperparameters: one (possibly repeated) cross-attention
    into the input plus a stack of latent self-attention blocks.

    ``num_cross_attention_layers`` > 1 re-applies cross-attention between
    self-attention blocks; the ``first_*_shared`` flags control whether the
    first layer/block shares weights with the repeats. ``freeze`` stops
    gradients for the whole encoder (transfer learning).
    """

    num_cross_attention_heads: int = 8
    num_cross_attention_qk_channels: Optional[int] = None
    num_cross_attention_v_channels: Optional[int] = None
    num_cross_attention_layers: int = 1
    first_cross_attention_layer_shared: bool = False
    cross_attention_widening_factor: int = 1
    num_self_attention_heads: int = 8
    num_self_attention_qk_channels: Optional[int] = None
    num_self_attention_v_channels: Optional[int] = None
    num_self_attention_layers_per_block: int = 8
    num_self_attention_blocks: int = 1
    first_self_attention_block_shared: bool = True
    self_attention_widening_factor: int = 1
    dropout: float = 0.0
    init_scale: float = 0.02
    freeze: bool = False

    def base_kwargs(self, exclude=("freeze",)):
        return _base_kwargs(self, EncoderConfig, exclude)


@dataclass
class DecoderConfig:
    """PerceiverDecoder hyperparameters: one output-query cross-attention over
    the latents. ``cross_attention_residual=False`` is used by heads whose
    query array lives in a different space than the output (MLM, flow)."""

    num_cross_attention_heads: int = 8
    num_cross_attention_qk_channels: Optional[int] = None
    num_cross_attention_v_channels: Optional[int] = None
    cross_attention_widening_factor: int = 1
    cross_attention_residual: bool = True
    dropout: float = 0.0
    init_scale: float = 0.02
    freeze: bool = False

    def base_kwargs(self, exclude=("freeze",)):
        return _base_kwargs(self, DecoderConfig, exclude)


@dataclass
class ClassificationDecoderConfig(DecoderConfig):
    """Decoder schema for classification heads (learned queries → logits)."""

    num_output_queries: int = 1
    num_output_query_channels: int = 256
    num_classes: int = 100


E = TypeVar("E", bound=EncoderConfig)
D = TypeVar("D", bound=DecoderConfig)


@dataclass
class PerceiverIOConfig(Generic[E, D]):
    """A full Perceiver IO = encoder schema + decoder schema + latent array
    geometry + activation-checkpointing flags."""

    encoder: E
    decoder: D
    num_latents: int
    num_latent_channels: int
    activation_checkpointing: bool = False
    activation_offloading: bool = False


@dataclass
class PerceiverARConfig:
    """Perceiver-AR core schema: causal cross-attention from the latent window
    into the (gradient-free) prefix, then causal latent self-attention.

    ``cross_attention_dropout`` drops a fraction of prefix *positions* during
    training (not elements); ``num_self_attention_rotary_layers`` bounds how
    many leading self-attention layers apply rotary embeddings (-1 = all);
    ``max_heads_parallel`` caps concurrently-computed heads in the eager
    fallback path to bound attention-matrix memory.
    """

    num_heads: int = 8
    max_heads_parallel: Optional[int] = None
    num_self_attention_layers: int = 8
    num_self_attention_rotary_layers: int = 1
    self_attention_widening_factor: int = 4
    cross_attention_widening_factor: int = 4
    cross_attention_dropout: float = 0.5
    post_attention_dropout: float = 0.0
    residual_dropout: float = 0.0
    activation_checkpointing: bool = False
    activation_offloading: bool = False

    def base_kwargs(self, exclude=()):
        return _base_kwargs(self, PerceiverARConfig, exclude)


@dataclass
class CausalSequenceModelConfig(PerceiverARConfig):
    """Perceiver-AR with a token vocabulary: adds the sequence/window geometry
    (``max_seq_len`` context, ``max_latents`` window), embedding width
    (``num_channels``), and output-head flags (final LayerNorm, logit bias,
    learned absolute positions vs rotary-only)."""

    vocab_size: int = 262
    max_seq_len: int = 4096
    max_latents: int = 512
    num_channels: int = 512
    output_norm: bool = False
    output_bias: bool = True
    abs_pos_emb: bool = True
    init_scale: float = 0.02

    @classmethod
    def create(cls, **kwargs):
        """Construct from a superset dict (e.g. a 🤗 ``model_config`` round
        trip), ignoring unknown keys."""
        known = {f.name for f in fields(cls)}
        return cls(**{k: v for k, v in kwargs.items() if k in known})
