"""Text classifier training CLI (parity: reference scripts/text/classifier.py)."""
from __future__ import annotations

from perceiver_amd.core import ClassificationDecoderConfig
from perceiver_amd.data.text import ImdbDataModule, Task
from perceiver_amd.models.text.common import TextEncoderConfig
from perceiver_amd.scripts.cli import CLI, build_dataclass
from perceiver_amd.train.lit import LitTextClassifier

DEFAULTS = {
    "model": {
        "num_latents": 256,
        "num_latent_channels": 1280,
        "encoder": {
            "num_input_channels": 768,
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
        "decoder": {"num_output_query_channels": 256, "dropout": 0.1},
    },
    "data": {"task": Task.clf, "tokenizer": "deepmind/language-perceiver", "max_seq_len": 2048},
    "trainer": {"out_dir": "logs/txt_clf"},
    "optimizer": {"lr": 2e-4, "lr_schedule": "constant", "warmup_steps": 500},
}


def link(cfg, dm):
    cfg["model"]["encoder"]["vocab_size"] = dm.vocab_size
    cfg["model"]["encoder"]["max_seq_len"] = dm.max_seq_len
    cfg["model"]["decoder"]["num_classes"] = dm.num_classes


def build_model(model_cfg, dm):
    encoder = build_dataclass(TextEncoderConfig, model_cfg.get("encoder", {}))
    decoder = build_dataclass(ClassificationDecoderConfig, model_cfg.get("decoder", {}))
    extra = {k: v for k, v in model_cfg.items() if k not in ("encoder", "decoder")}
    return LitTextClassifier(encoder, decoder, **extra)


if __name__ == "__main__":
    CLI(LitTextClassifier, ImdbDataModule, DEFAULTS, build_model, link)
