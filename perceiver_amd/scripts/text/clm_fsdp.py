"""FSDP variant of the causal-LM trainer (parity: reference scripts/text/clm_fsdp.py,
which trains the 455M C4 model with transformer_auto_wrap_policy over the attention
layers). On MI355X the 288 GB HBM per GPU makes sharding unnecessary at reference
scales — DDP (perceiver_amd.parallel) is the default — but the entry point wraps the
model in torch FSDP (fully_shard is NOT required; classic FSDP works over RCCL) for
much larger configs.

    python -m torch.distributed.run --nproc-per-node 8 --master-addr 127.0.0.1 \
        -m perceiver_amd.scripts.text.clm_fsdp fit --data.batch_size 32 ...
"""
from __future__ import annotations

import functools

import torch

from perceiver_amd.core.modules import CrossAttentionLayer, SelfAttentionLayer
from perceiver_amd.data.text import C4DataModule
from perceiver_amd.scripts.cli import CLI
from perceiver_amd.train.lit import LitCausalLanguageModel

DEFAULTS = {
    "model": {
        "max_latents": 512,
        "num_channels": 1280,
        "num_heads": 10,
        "num_self_attention_layers": 20,
        "cross_attention_dropout": 0.5,
        "activation_checkpointing": True,
    },
    "data": {"tokenizer": "xlnet-base-cased", "max_seq_len": 1024, "padding_side": "left"},
    "trainer": {"out_dir": "logs/clm_fsdp", "grad_clip": 0.5},
    "optimizer": {"lr": 2e-4, "lr_schedule": "cosine", "warmup_steps": 1000},
}


def shard_perceiver_ar(model, device_type: str = None):
    """Shard a PerceiverAR-family model per attention layer with ``fully_shard``
    (FSDP2) — the same layer granularity as the reference's
    transformer_auto_wrap_policy over its attention layer classes
    (reference scripts/text/clm_fsdp.py:15-23). Parameters become DTensors over a
    1-D mesh of the full world; per-layer units overlap reshard/all-gather with
    compute. Works over RCCL on GPU and gloo on CPU (the 2-rank CPU test)."""
    import torch.distributed as dist
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.fsdp import fully_shard

    if device_type is None:
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
    mesh = init_device_mesh(device_type, (dist.get_world_size(),))
    for m in model.modules():
        if isinstance(m, (CrossAttentionLayer, SelfAttentionLayer)):
            fully_shard(m, mesh=mesh)
    fully_shard(model, mesh=mesh)
    return model


class LitCausalLanguageModelFSDP(LitCausalLanguageModel):
    """Wraps the backend with per-attention-layer fully_shard (FSDP2)."""

    def wrap_fsdp(self):
        if torch.cuda.is_available():
            self.model = self.model.to(torch.device("cuda", torch.cuda.current_device()))
        shard_perceiver_ar(self.model)
        return self


def link(cfg, dm):
    cfg["model"]["vocab_size"] = dm.vocab_size
    cfg["model"]["max_seq_len"] = dm.max_seq_len


def build_model(model_cfg, dm):
    task = LitCausalLanguageModelFSDP(**model_cfg)
    import torch.distributed as dist

    if dist.is_initialized():
        task.wrap_fsdp()
    return task


if __name__ == "__main__":
    CLI(LitCausalLanguageModelFSDP, C4DataModule, DEFAULTS, build_model, link)
