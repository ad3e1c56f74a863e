"""Subprocess entry point for distributed CPU tests (fresh interpreter per rank).

Usage: python tests/_dist_worker.py <name> <rank> <world> <port> <out_dir>
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn.functional as F


def make_clm_model(seed=0):
    from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig

    torch.manual_seed(seed)
    cfg = CausalLanguageModelConfig(
        vocab_size=50, max_seq_len=16, max_latents=8, num_channels=24, num_heads=4,
        num_self_attention_layers=2, cross_attention_dropout=0.0,
    )
    return CausalLanguageModel(cfg)


def reducer_worker(rank, world, out_dir):
    from perceiver_amd.parallel import BucketedGradReducer

    model = make_clm_model()  # same seed on all ranks -> identical weights
    reducer = BucketedGradReducer(model, bucket_cap_mb=0.05)

    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randint(0, 50, (2, 16))
    out = model(x, prefix_len=8)
    loss = F.cross_entropy(out.logits.flatten(0, 1), x[:, 8:].flatten())
    loss.backward()
    reducer.finalize()

    grads = {n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None}
    torch.save(grads, os.path.join(out_dir, f"rank{rank}.pt"))


def trainer_worker(rank, world, out_dir):
    """Two-rank training-loop smoke test: train a tiny CLM for 3 steps with the
    native Trainer and dump final weights (must be identical across ranks)."""
    from perceiver_amd.train.trainer import Trainer, TrainConfig

    model = make_clm_model()
    torch.manual_seed(200 + rank)
    xs = [torch.randint(0, 50, (2, 16)) for _ in range(3)]

    def batches():
        for x in xs:
            yield {"x": x, "prefix_len": 8, "labels": x[:, 8:]}

    def step_fn(model, batch):
        out = model(batch["x"], prefix_len=batch["prefix_len"])
        return F.cross_entropy(out.logits.flatten(0, 1), batch["labels"].flatten())

    trainer = Trainer(TrainConfig(max_steps=3, log_every=100, out_dir=os.path.join(out_dir, "run")))
    trainer.fit_steps(model, batches(), step_fn)
    torch.save(model.state_dict(), os.path.join(out_dir, f"rank{rank}_weights.pt"))


def fsdp_worker(rank, world, out_dir):
    """FSDP path (the scripts/text/clm_fsdp.py sharding layout) on gloo/CPU: shard
    a tiny CLM per attention layer with fully_shard, run 2 optimizer steps on
    identical data, dump the reassembled full state dict (must match across
    ranks)."""
    from torch.distributed.checkpoint.state_dict import StateDictOptions, get_model_state_dict

    from perceiver_amd.scripts.text.clm_fsdp import shard_perceiver_ar

    model = make_clm_model()
    shard_perceiver_ar(model, device_type="cpu")
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)

    torch.manual_seed(7)  # same data on all ranks -> parameter trajectories identical
    losses = []
    for _ in range(2):
        x = torch.randint(0, 50, (2, 16))
        out = model(x, prefix_len=8)
        loss = F.cross_entropy(out.logits.flatten(0, 1), x[:, 8:].flatten())
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        losses.append(loss.item())

    full = get_model_state_dict(model, options=StateDictOptions(full_state_dict=True))
    torch.save({"state_dict": full, "losses": losses},
               os.path.join(out_dir, f"rank{rank}_fsdp.pt"))


def accum_worker(rank, world, out_dir):
    """Gradient accumulation under the bucketed reducer: 2 ranks x 2 micro
    batches; the boundary reduction must yield grads equal to the mean over all
    4 micro-batch gradients (checked against a single-process reference by the
    test)."""
    from perceiver_amd.parallel import BucketedGradReducer

    model = make_clm_model()
    reducer = BucketedGradReducer(model, bucket_cap_mb=0.05)

    for micro in range(2):
        torch.manual_seed(300 + rank * 2 + micro)  # distinct data per micro-batch
        x = torch.randint(0, 50, (2, 16))
        boundary = micro == 1
        reducer.set_sync(boundary)
        out = model(x, prefix_len=8)
        loss = F.cross_entropy(out.logits.flatten(0, 1), x[:, 8:].flatten()) / 2
        loss.backward()
    reducer.finalize()

    grads = {n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None}
    torch.save(grads, os.path.join(out_dir, f"rank{rank}_accum.pt"))


MODES = {"reducer": reducer_worker, "trainer": trainer_worker, "fsdp": fsdp_worker,
         "accum": accum_worker}

if __name__ == "__main__":
    name, rank, world, port, out_dir = sys.argv[1:6]
    rank, world = int(rank), int(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = port
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        MODES[name](rank, world, out_dir)
    finally:
        dist.destroy_process_group()
