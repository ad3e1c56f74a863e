"""GPU sanity for the fp16 (autocast + GradScaler) trainer path.

The CLI maps Lightning's ``--trainer.precision 16`` onto
``TrainConfig(precision="fp16")``; this exercises that path end-to-end on a
real GPU: scaled backward, unscale+clip, optimizer step, finite losses, and
weights actually moving.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fp16_trainer_steps_and_updates():
    from perceiver_amd.train.trainer import TrainConfig, Trainer

    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.GELU(), torch.nn.Linear(128, 8)
    )
    w0 = model[0].weight.detach().clone()

    cfg = TrainConfig(max_steps=8, lr=1e-3, precision="fp16", grad_clip=1.0,
                      log_every=1000, out_dir="/tmp/fp16_run", tensorboard=False)
    trainer = Trainer(cfg)
    assert trainer.scaler.is_enabled()

    g = torch.Generator().manual_seed(1)
    batches = [(torch.randn(32, 64, generator=g), torch.randint(0, 8, (32,), generator=g))
               for _ in range(8)]

    losses = []

    def step_fn(m, batch):
        x, y = batch
        loss = torch.nn.functional.cross_entropy(m(x), y)
        losses.append(float(loss.detach()))
        return loss

    trainer.fit_steps(model, batches, step_fn)

    assert len(losses) == 8
    assert all(torch.isfinite(torch.tensor(losses))), losses
    # fp16 autocast really ran and the (fp32) master weights moved
    assert not torch.allclose(model[0].weight.detach().cpu(), w0)


def test_fp16_scaler_recovers_from_overflow():
    """An inf gradient must be skipped by the scaler, not poison the weights."""
    from perceiver_amd.train.trainer import TrainConfig, Trainer

    torch.manual_seed(0)
    model = torch.nn.Linear(16, 4)
    cfg = TrainConfig(max_steps=3, lr=1e-3, precision="fp16",
                      log_every=1000, out_dir="/tmp/fp16_run2", tensorboard=False)
    trainer = Trainer(cfg)

    def step_fn(m, batch):
        x, scale = batch
        return (m(x) * scale).float().pow(2).mean()

    batches = [(torch.randn(8, 16), 1.0),
               (torch.randn(8, 16), 1e30),   # overflows fp16 grads
               (torch.randn(8, 16), 1.0)]
    trainer.fit_steps(model, batches, step_fn)
    assert torch.isfinite(model.weight.detach()).all()
