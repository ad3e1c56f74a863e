"""🤗 face of the masked language model.

Registers the model with ``AutoModelForMaskedLM`` (fill-mask pipeline) and
carries the conversion utilities in both directions: training checkpoints →
``save_pretrained`` directories, and the published
``deepmind/language-perceiver`` weights → this framework's module layout
(the parity target pinned at 201,108,230 parameters).

Behavioral contract mirrored from the reference's text/mlm/huggingface.py
and text/common/huggingface.py.
"""
from __future__ import annotations

from typing import Optional

import torch
from transformers import AutoConfig, AutoModelForMaskedLM, AutoTokenizer, PretrainedConfig, PreTrainedModel
from transformers.modeling_outputs import MaskedLMOutput

from perceiver_amd.models.hf_base import (
    copy_cross_attention_layer_params,
    copy_latent_provider_params,
    copy_param,
    copy_params,
    copy_self_attention_block_params,
)
from perceiver_amd.models.hf_registry import (
    BackendConfigMixin,
    save_with_tokenizer,
    wrap_lit_checkpoint,
)
from perceiver_amd.models.text.common import TextEncoder, TextEncoderConfig
from perceiver_amd.models.text.mlm import (
    MaskedLanguageModel,
    MaskedLanguageModelConfig,
    PerceiverDecoder,
    TextDecoderConfig,
)


class PerceiverMaskedLanguageModelConfig(BackendConfigMixin, PretrainedConfig):
    model_type = "perceiver-io-masked-language-model"
    backend_config_class = MaskedLanguageModelConfig

    def __init__(self, backend_config=None, **kwargs):
        # explicit __init__: transformers 5.x wraps configs without one
        # in a kwargs-only guard that would swallow backend_config
        super().__init__(backend_config, **kwargs)

    @classmethod
    def default_backend_config(cls):
        return MaskedLanguageModelConfig(
            TextEncoderConfig(), TextDecoderConfig(),
            num_latents=512, num_latent_channels=512,
        )

    @classmethod
    def decode_backend_config(cls, model_config):
        flat = dict(model_config)
        return MaskedLanguageModelConfig(
            encoder=TextEncoderConfig(**flat.pop("encoder")),
            decoder=TextDecoderConfig(**flat.pop("decoder")),
            **flat,
        )


class PerceiverMaskedLanguageModel(PreTrainedModel):
    config_class = PerceiverMaskedLanguageModelConfig

    def __init__(self, config: PerceiverMaskedLanguageModelConfig):
        super().__init__(config)
        self.backend_model = MaskedLanguageModel(config.backend_config)
        self.post_init()

    @staticmethod
    def from_checkpoint(ckpt_path):
        from perceiver_amd.train.lit import LitMaskedLanguageModel

        return wrap_lit_checkpoint(LitMaskedLanguageModel, PerceiverMaskedLanguageModel,
                                   ckpt_path, is_decoder=False)

    def forward(self, input_ids: torch.LongTensor,
                attention_mask: Optional[torch.FloatTensor] = None,
                labels: Optional[torch.LongTensor] = None):
        if labels is not None:
            raise ValueError("Loss computation from labels not supported yet")
        pad_mask = None if attention_mask is None else ~attention_mask.type(torch.bool)
        return MaskedLMOutput(logits=self.backend_model(input_ids, pad_mask=pad_mask))


AutoConfig.register(PerceiverMaskedLanguageModelConfig.model_type, PerceiverMaskedLanguageModelConfig)
AutoModelForMaskedLM.register(PerceiverMaskedLanguageModelConfig, PerceiverMaskedLanguageModel)


# ------------------------------------------------------------------ conversion
def copy_text_encoder_params(src, tgt: TextEncoder):
    """transformers PerceiverModel -> TextEncoder (latents, CA, SA, embeddings)."""
    copy_cross_attention_layer_params(src.encoder.cross_attention, tgt.cross_attn_1, query_residual=True)
    copy_self_attention_block_params(src.encoder.self_attends, tgt.self_attn_1)
    copy_latent_provider_params(src, tgt)
    copy_params(src.input_preprocessor.embeddings, tgt.input_adapter.txt_embedding)
    copy_params(src.input_preprocessor.position_embeddings, tgt.input_adapter.pos_embedding)


def copy_text_decoder_params(src, tgt: PerceiverDecoder):
    """transformers PerceiverForMaskedLM decoder -> tied-logits PerceiverDecoder."""
    copy_cross_attention_layer_params(
        src.perceiver.decoder.decoding_cross_attention, tgt.cross_attn, query_residual=False
    )
    copy_param(src.perceiver.decoder.output_position_encodings.position_embeddings,
               tgt.output_query_provider._query)
    copy_param(src.embedding_decoder.bias, tgt.output_adapter.bias)


def convert_config(config) -> MaskedLanguageModelConfig:
    """transformers PerceiverConfig -> the backend dataclass tree.

    Field-by-field translation of the deepmind checkpoint hyperparameters;
    the decoder is the tied-head variant with no cross-attention residual
    (matching the published architecture).
    """
    assert config.hidden_act == "gelu"
    # transformers 5.x guards global access to tie_word_embeddings on some configs
    assert config.to_dict().get("tie_word_embeddings", True)

    shared = dict(
        vocab_size=config.vocab_size,
        max_seq_len=config.max_position_embeddings,
        num_cross_attention_qk_channels=config.qk_channels,
        num_cross_attention_heads=config.num_cross_attention_heads,
        cross_attention_widening_factor=config.cross_attention_widening_factor,
        dropout=config.attention_probs_dropout_prob,
        init_scale=config.initializer_range,
    )
    encoder = TextEncoderConfig(
        num_input_channels=config.d_model,
        num_cross_attention_v_channels=config.v_channels,
        num_self_attention_qk_channels=config.qk_channels,
        num_self_attention_v_channels=config.v_channels,
        num_self_attention_heads=config.num_self_attention_heads,
        num_self_attention_layers_per_block=config.num_self_attends_per_block,
        num_self_attention_blocks=config.num_blocks,
        self_attention_widening_factor=config.self_attention_widening_factor,
        **shared,
    )
    decoder = TextDecoderConfig(
        num_cross_attention_v_channels=config.d_model,
        cross_attention_residual=False,
        **shared,
    )
    return MaskedLanguageModelConfig(
        encoder, decoder,
        num_latents=config.num_latents, num_latent_channels=config.d_latents,
    )


def convert_checkpoint(save_dir, ckpt_url, tokenizer_name, **kwargs):
    """Training .ckpt -> persistent 🤗 directory with tokenizer."""
    save_with_tokenizer(PerceiverMaskedLanguageModel.from_checkpoint(ckpt_url),
                        tokenizer_name, save_dir, **kwargs)


def convert_model(save_dir, source_repo_id="deepmind/language-perceiver", **kwargs):
    """Published transformers PerceiverForMaskedLM -> this framework's format."""
    import transformers

    source = transformers.PerceiverForMaskedLM.from_pretrained(source_repo_id)
    target = PerceiverMaskedLanguageModel(
        PerceiverMaskedLanguageModelConfig(convert_config(source.config)))
    copy_text_encoder_params(source.perceiver, target.backend_model.encoder)
    copy_text_decoder_params(source, target.backend_model.decoder)
    save_with_tokenizer(target, source_repo_id, save_dir, **kwargs)
