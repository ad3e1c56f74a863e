"""Causal-LM training CLI (parity: reference scripts/text/clm.py).

    python -m perceiver_amd.scripts.text.clm fit --data.batch_size 24 ...
"""
from __future__ import annotations

from perceiver_amd.data.text import Task, WikiTextDataModule
from perceiver_amd.scripts.cli import CLI
from perceiver_amd.train.lit import LitCausalLanguageModel

DEFAULTS = {
    "model": {
        "max_latents": 512,
        "num_channels": 512,
        "num_heads": 8,
        "num_self_attention_layers": 8,
        "cross_attention_dropout": 0.5,
    },
    "data": {"task": Task.clm, "tokenizer": "deepmind/language-perceiver",
             "max_seq_len": 4096, "padding_side": "left"},
    "trainer": {"out_dir": "logs/clm"},
    "optimizer": {"lr": 2e-4, "lr_schedule": "cosine", "warmup_steps": 200},
}


def link(cfg, dm):
    cfg["model"]["vocab_size"] = dm.vocab_size
    cfg["model"]["max_seq_len"] = dm.max_seq_len


def build_model(model_cfg, dm):
    return LitCausalLanguageModel(**model_cfg)


if __name__ == "__main__":
    CLI(LitCausalLanguageModel, WikiTextDataModule, DEFAULTS, build_model, link)
