"""Symbolic-audio training CLI (parity: reference scripts/audio/symbolic.py)."""
from __future__ import annotations

from perceiver_amd.data.audio import GiantMidiPianoDataModule
from perceiver_amd.scripts.cli import CLI
from perceiver_amd.train.lit import LitSymbolicAudioModel

DEFAULTS = {
    "model": {
        "max_latents": 512,
        "num_channels": 512,
        "num_heads": 8,
        "num_self_attention_layers": 12,
        "cross_attention_dropout": 0.5,
        "post_attention_dropout": 0.0,
        "output_norm": True,
        "output_bias": False,
        "abs_pos_emb": False,
    },
    "data": {"max_seq_len": 2048},
    "trainer": {"out_dir": "logs/sam"},
    "optimizer": {"lr": 2e-4, "lr_schedule": "cosine", "warmup_steps": 500},
}


def link(cfg, dm):
    cfg["model"]["vocab_size"] = dm.vocab_size
    cfg["model"]["max_seq_len"] = dm.max_seq_len


def build_model(model_cfg, dm):
    return LitSymbolicAudioModel(**model_cfg)


if __name__ == "__main__":
    CLI(LitSymbolicAudioModel, GiantMidiPianoDataModule, DEFAULTS, build_model, link)
