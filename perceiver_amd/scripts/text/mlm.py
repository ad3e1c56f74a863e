"""Masked-LM training CLI with the 201M flagship defaults
(parity: reference scripts/text/mlm.py).

    python -m perceiver_amd.scripts.text.mlm fit --data.batch_size 32 ...
"""
from __future__ import annotations

from perceiver_amd.data.text import ImdbDataModule, Task
from perceiver_amd.models.text.common import TextEncoderConfig
from perceiver_amd.models.text.mlm import TextDecoderConfig
from perceiver_amd.scripts.cli import CLI, build_dataclass
from perceiver_amd.train.lit import LitMaskedLanguageModel

DEFAULTS = {
    "model": {
        "num_latents": 256,
        "num_latent_channels": 1280,
        "encoder": {
            "num_input_channels": 768,
            "num_cross_attention_layers": 1,
            "num_cross_attention_qk_channels": 256,
            "num_cross_attention_v_channels": 1280,
            "num_cross_attention_heads": 8,
            "num_self_attention_qk_channels": 256,
            "num_self_attention_v_channels": 1280,
            "num_self_attention_heads": 8,
            "num_self_attention_layers_per_block": 26,
            "num_self_attention_blocks": 1,
            "dropout": 0.1,
        },
        "decoder": {
            "num_cross_attention_qk_channels": 256,
            "num_cross_attention_v_channels": 768,
            "num_cross_attention_heads": 8,
            "cross_attention_residual": False,
            "dropout": 0.1,
        },
        "num_predictions": 5,
        "masked_samples": None,
    },
    "data": {"task": Task.mlm, "tokenizer": "deepmind/language-perceiver", "max_seq_len": 2048},
    "trainer": {"out_dir": "logs/mlm"},
    "optimizer": {"lr": 2e-4, "lr_schedule": "constant", "warmup_steps": 1000},
}


def link(cfg, dm):
    cfg["model"]["encoder"]["vocab_size"] = dm.vocab_size
    cfg["model"]["decoder"]["vocab_size"] = dm.vocab_size
    cfg["model"]["encoder"]["max_seq_len"] = dm.max_seq_len
    cfg["model"]["decoder"]["max_seq_len"] = dm.max_seq_len


def build_model(model_cfg, dm):
    encoder = build_dataclass(TextEncoderConfig, model_cfg.get("encoder", {}))
    decoder = build_dataclass(TextDecoderConfig, model_cfg.get("decoder", {}))
    extra = {k: v for k, v in model_cfg.items() if k not in ("encoder", "decoder")}
    return LitMaskedLanguageModel(encoder, decoder, **extra)


if __name__ == "__main__":
    CLI(LitMaskedLanguageModel, ImdbDataModule, DEFAULTS, build_model, link)
