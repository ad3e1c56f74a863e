"""🤗 face of the causal language model.

Registers the Perceiver-AR CLM with ``AutoModelForCausalLM`` (and therefore
the text-generation pipeline), and converts training checkpoints into
``save_pretrained`` directories with a LEFT-padding tokenizer — Perceiver-AR
requires left padding so the latent window stays right-aligned.

Behavioral contract mirrored from the reference
(/root/reference/perceiver/model/text/clm/huggingface.py); the generation
machinery itself lives in models/hf_base.py.
"""
from __future__ import annotations

from transformers import AutoConfig, AutoModelForCausalLM, PretrainedConfig

from perceiver_amd.models.hf_base import PerceiverCausalSequenceModel
from perceiver_amd.models.hf_registry import (
    BackendConfigMixin,
    save_with_tokenizer,
    wrap_lit_checkpoint,
)
from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig


class PerceiverCausalLanguageModelConfig(BackendConfigMixin, PretrainedConfig):
    model_type = "perceiver-ar-causal-language-model"
    backend_config_class = CausalLanguageModelConfig

    def __init__(self, backend_config=None, **kwargs):
        # explicit __init__: transformers 5.x wraps configs without one
        # in a kwargs-only guard that would swallow backend_config
        super().__init__(backend_config, **kwargs)


class PerceiverCausalLanguageModel(PerceiverCausalSequenceModel):
    config_class = PerceiverCausalLanguageModelConfig

    def __init__(self, config: PerceiverCausalLanguageModelConfig, **kwargs):
        super().__init__(config)
        # "backend_model" kwarg = zero-copy wrap of an existing (training)
        # model for in-training sample generation
        self.backend_model = kwargs.get("backend_model") or CausalLanguageModel(config.backend_config)
        self.post_init()

    @staticmethod
    def from_checkpoint(ckpt_path):
        from perceiver_amd.train.lit import LitCausalLanguageModel

        return wrap_lit_checkpoint(LitCausalLanguageModel, PerceiverCausalLanguageModel,
                                   ckpt_path, is_decoder=True)


AutoConfig.register(PerceiverCausalLanguageModelConfig.model_type, PerceiverCausalLanguageModelConfig)
AutoModelForCausalLM.register(PerceiverCausalLanguageModelConfig, PerceiverCausalLanguageModel)


def convert_checkpoint(save_dir, ckpt_url, tokenizer_name, **kwargs):
    """Training .ckpt -> persistent 🤗 dir (left-padding tokenizer included)."""
    save_with_tokenizer(PerceiverCausalLanguageModel.from_checkpoint(ckpt_url),
                        tokenizer_name, save_dir, padding_side="left", **kwargs)
