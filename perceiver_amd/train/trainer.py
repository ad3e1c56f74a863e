"""Native MI355X training loop — replaces PyTorch Lightning's Trainer.

One process per GPU over RCCL/xGMI (torch.distributed "nccl" backend on ROCm),
native bucketed gradient all-reduce overlapped with backward
(perceiver_amd.parallel.BucketedGradReducer), bf16 autocast compute with fp32
master weights, step-interval LR scheduling, rank-0 checkpointing in the
Lightning-compatible ``.ckpt`` layout ({"state_dict", "hyper_parameters",
"global_step", ...}), and JSONL metric logging.

Two APIs:
  - ``fit(task, datamodule)``: Lightning-style, where ``task`` is a
    perceiver_amd.train.lit wrapper with training_step/validation_step;
  - ``fit_steps(model, batches, step_fn)``: low-level step loop used by tests
    and custom drivers.
"""
from __future__ import annotations

import json
import math
import os
import time
from dataclasses import dataclass, field
from typing import Callable, Iterable, Optional

import torch
import torch.distributed as dist

from perceiver_amd.parallel import (
    BucketedGradReducer,
    get_rank,
    get_world_size,
    init_distributed_from_env,
    is_main_process,
)
from perceiver_amd.train.lrs import ConstantWithWarmupLR, CosineWithWarmupLR


@dataclass
class TrainConfig:
    max_steps: Optional[int] = None
    max_epochs: Optional[int] = None
    lr: float = 2e-4
    weight_decay: float = 0.01
    betas: tuple = (0.9, 0.999)
    optimizer: str = "adamw"         # "adamw" | "master_adamw" | "lamb"
    precision: str = "bf16"          # "bf16" | "fp16" (autocast+GradScaler) | "fp32"
    grad_clip: Optional[float] = None
    accumulate_grad_batches: int = 1
    bucket_cap_mb: float = 50.0
    lr_schedule: str = "constant"    # "constant" | "cosine" | "none"
    warmup_steps: int = 0
    min_lr_fraction: float = 0.0
    log_every: int = 50
    val_every_steps: Optional[int] = None
    ckpt_every_steps: Optional[int] = None
    out_dir: str = "logs/run"
    tensorboard: bool = True         # scalar event files under out_dir/tb
    save_top_k: int = 1
    monitor: str = "val_loss"
    seed: Optional[int] = None


class Trainer:
    def __init__(self, config: TrainConfig):
        self.cfg = config
        if torch.cuda.is_available():
            from perceiver_amd.utils.tunableop import arm_tunableop

            arm_tunableop(int(os.environ.get("RANK", "0")))
        self.local_rank = init_distributed_from_env()
        self.device = torch.device("cuda", self.local_rank) if torch.cuda.is_available() else torch.device("cpu")
        self.global_step = 0
        self.epoch = 0
        self._best_monitor = math.inf
        self._log_file = None
        self._tb = None
        if is_main_process():
            os.makedirs(os.path.join(self.cfg.out_dir, "checkpoints"), exist_ok=True)
            self._log_file = open(os.path.join(self.cfg.out_dir, "metrics.jsonl"), "a")
            if self.cfg.tensorboard:
                from perceiver_amd.utils.tensorboard import ScalarWriter

                self._tb = ScalarWriter(os.path.join(self.cfg.out_dir, "tb"))
        if config.seed is not None:
            torch.manual_seed(config.seed + get_rank())
        # fp16 needs loss scaling; for bf16/fp32 the scaler is a no-op passthrough
        self.scaler = torch.amp.GradScaler(
            "cuda", enabled=(self.cfg.precision == "fp16" and self.device.type == "cuda"))

    # ------------------------------------------------------------------ helpers
    def _autocast(self):
        dtype = {"bf16": torch.bfloat16, "fp16": torch.float16}.get(self.cfg.precision)
        enabled = dtype is not None and self.device.type == "cuda"
        return torch.autocast(self.device.type, dtype=dtype or torch.bfloat16, enabled=enabled)

    def _optimizer_step(self, optimizer, model_params):
        """Clip (through the scaler when fp16) and step."""
        if self.scaler.is_enabled():
            self.scaler.unscale_(optimizer)
        if self.cfg.grad_clip and not getattr(optimizer, "max_grad_norm", 0.0):
            torch.nn.utils.clip_grad_norm_(model_params, self.cfg.grad_clip)
        self.scaler.step(optimizer)
        self.scaler.update()

    def _make_optimizer(self, model):
        params = [p for p in model.parameters() if p.requires_grad]
        kw = dict(lr=self.cfg.lr, weight_decay=self.cfg.weight_decay, betas=self.cfg.betas)
        if self.cfg.optimizer == "lamb":
            from perceiver_amd.train.optim import Lamb

            return Lamb(params, **kw)
        if self.cfg.optimizer == "master_adamw":
            from perceiver_amd.train.optim import MasterAdamW

            return MasterAdamW(params, max_grad_norm=self.cfg.grad_clip or 0.0, **kw)
        return torch.optim.AdamW(params, foreach=True, **kw)

    def _make_scheduler(self, opt):
        if self.cfg.lr_schedule == "cosine":
            total = self.cfg.max_steps or 10000
            return CosineWithWarmupLR(opt, training_steps=total, warmup_steps=self.cfg.warmup_steps,
                                      min_fraction=self.cfg.min_lr_fraction)
        if self.cfg.lr_schedule == "constant":
            return ConstantWithWarmupLR(opt, warmup_steps=self.cfg.warmup_steps)
        return None

    def log_metrics(self, metrics: dict, step: Optional[int] = None):
        at = self.global_step if step is None else step
        if self._log_file is not None:
            rec = {"step": at, "time": time.time(), **metrics}
            self._log_file.write(json.dumps(rec) + "\n")
            self._log_file.flush()
        if self._tb is not None:
            for key, value in metrics.items():
                if isinstance(value, (int, float)):
                    self._tb.add_scalar(key, value, at)
            self._tb.flush()

    def _reduce_mean(self, value: torch.Tensor) -> torch.Tensor:
        value = value.detach()
        if dist.is_initialized():
            value = value.clone()
            dist.all_reduce(value)
            value /= get_world_size()
        return value

    # ------------------------------------------------------------------ low-level
    def fit_steps(self, model: torch.nn.Module, batches: Iterable, step_fn: Callable,
                  optimizer=None, scheduler=None):
        """Run the step loop over ``batches``; ``step_fn(model, batch) -> loss``."""
        model = model.to(self.device)
        model.train()
        optimizer = optimizer or self._make_optimizer(model)
        scheduler = scheduler if scheduler is not None else self._make_scheduler(optimizer)

        reducer = None
        if get_world_size() > 1:
            reducer = BucketedGradReducer(model, bucket_cap_mb=self.cfg.bucket_cap_mb)

        accum = max(1, self.cfg.accumulate_grad_batches)
        micro = 0
        for batch in batches:
            batch = _move(batch, self.device)
            if micro == 0:
                # the reducer's grads are bucket views: zero through it
                if reducer is not None:
                    reducer.zero_grad()
                else:
                    optimizer.zero_grad(set_to_none=True)
            boundary = (micro + 1) == accum
            if reducer is not None:
                reducer.set_sync(boundary)
            with self._autocast():
                loss = step_fn(model, batch)
            self.scaler.scale(loss / accum).backward()
            micro += 1
            if not boundary:
                continue
            micro = 0
            if reducer is not None:
                reducer.finalize()
            self._optimizer_step(optimizer, model.parameters())
            if scheduler is not None:
                scheduler.step()
            self.global_step += 1

            if self.global_step % self.cfg.log_every == 0:
                self.log_metrics({"train_loss": float(self._reduce_mean(loss)),
                                  "lr": optimizer.param_groups[0]["lr"]})
            if self.cfg.max_steps and self.global_step >= self.cfg.max_steps:
                break

        if micro > 0 and not (self.cfg.max_steps and self.global_step >= self.cfg.max_steps):
            # flush a trailing partial accumulation window (see fit())
            if reducer is not None:
                reducer.reduce_now()
            self._optimizer_step(optimizer, model.parameters())
            if scheduler is not None:
                scheduler.step()
            self.global_step += 1

        if reducer is not None:
            reducer.remove()
        return model

    # ------------------------------------------------------------------ task API
    def fit(self, task, datamodule=None, train_loader=None, val_loader=None, ckpt_path: Optional[str] = None):
        """Lightning-style fit of a perceiver_amd.train.lit task wrapper."""
        task = task.to(self.device)
        optimizer = self._make_optimizer(task)
        scheduler = self._make_scheduler(optimizer)

        if ckpt_path is not None:
            self._restore(task, optimizer, scheduler, ckpt_path)

        if datamodule is not None:
            datamodule.prepare_data()
            datamodule.setup("fit")
            train_loader = datamodule.train_dataloader()
            val_loader = datamodule.val_dataloader() if hasattr(datamodule, "val_dataloader") else None
        self.datamodule = datamodule

        reducer = None
        if get_world_size() > 1:
            reducer = BucketedGradReducer(task, bucket_cap_mb=self.cfg.bucket_cap_mb)

        task.trainer = self
        done = False
        max_epochs = self.cfg.max_epochs or (1 if self.cfg.max_steps is None else 10**9)
        while not done and self.epoch < max_epochs:
            task.train()
            sampler = getattr(train_loader, "sampler", None)
            if hasattr(sampler, "set_epoch"):
                sampler.set_epoch(self.epoch)
            accum = max(1, self.cfg.accumulate_grad_batches)
            micro = 0
            for batch in train_loader:
                batch = _move(batch, self.device)
                if micro == 0:
                    # the reducer's grads are bucket views: zero through it
                    if reducer is not None:
                        reducer.zero_grad()
                    else:
                        optimizer.zero_grad(set_to_none=True)
                boundary = (micro + 1) == accum
                if reducer is not None:
                    reducer.set_sync(boundary)
                with self._autocast():
                    loss = task.training_step(batch, self.global_step)
                self.scaler.scale(loss / accum).backward()
                micro += 1
                if not boundary:
                    continue
                micro = 0
                if reducer is not None:
                    reducer.finalize()
                self._optimizer_step(optimizer, task.parameters())
                if scheduler is not None:
                    scheduler.step()
                self.global_step += 1

                if self.global_step % self.cfg.log_every == 0:
                    self.log_metrics({"train_loss": float(self._reduce_mean(loss)),
                                      "lr": optimizer.param_groups[0]["lr"]})
                if self.cfg.val_every_steps and self.global_step % self.cfg.val_every_steps == 0 and val_loader:
                    self._validate(task, val_loader, optimizer, scheduler)
                if self.cfg.ckpt_every_steps and self.global_step % self.cfg.ckpt_every_steps == 0:
                    self._save_checkpoint(task, optimizer, scheduler, {})
                if self.cfg.max_steps and self.global_step >= self.cfg.max_steps:
                    done = True
                    break
            if micro > 0 and not done:
                # trailing partial accumulation window at epoch end: its grads
                # are real — flush them instead of silently discarding
                if reducer is not None:
                    reducer.reduce_now()
                self._optimizer_step(optimizer, task.parameters())
                if scheduler is not None:
                    scheduler.step()
                self.global_step += 1
            self.epoch += 1
            if not done and val_loader is not None:
                self._validate(task, val_loader, optimizer, scheduler)

        if val_loader is not None:
            metrics = self._validate(task, val_loader, optimizer, scheduler)
        else:
            metrics = {}
            self._save_checkpoint(task, optimizer, scheduler, metrics)

        if reducer is not None:
            reducer.remove()
        return task

    @torch.no_grad()
    def _validate(self, task, val_loader, optimizer=None, scheduler=None):
        task.eval()
        totals, count = {}, 0
        for batch in val_loader:
            batch = _move(batch, self.device)
            with self._autocast():
                out = task.validation_step(batch)
            for k, v in out.items():
                totals[k] = totals.get(k, 0.0) + float(v)
            count += 1
        task.train()
        metrics = {k: v / max(count, 1) for k, v in totals.items()}
        # sync_dist: mean across ranks
        if dist.is_initialized() and metrics:
            t = torch.tensor([metrics[k] for k in sorted(metrics)], device=self.device
                             if dist.get_backend() == "nccl" else "cpu")
            dist.all_reduce(t)
            t /= get_world_size()
            metrics = {k: float(t[i]) for i, k in enumerate(sorted(metrics))}
        self.log_metrics(metrics)
        if hasattr(task, "on_validation_end"):
            task.on_validation_end(self, metrics)
        self._save_checkpoint(task, optimizer, scheduler, metrics)
        return metrics

    # ------------------------------------------------------------------ checkpoint
    def _save_checkpoint(self, task, optimizer, scheduler, metrics: dict):
        if not is_main_process():
            return
        monitor_val = metrics.get(self.cfg.monitor)
        ckpt = {
            "state_dict": task.state_dict(),
            "hyper_parameters": getattr(task, "hparams", {}),
            "global_step": self.global_step,
            "epoch": self.epoch,
            "optimizer_states": [optimizer.state_dict()] if optimizer else [],
            "lr_schedulers": [scheduler.state_dict()] if scheduler else [],
            "metrics": metrics,
        }
        ckpt_dir = os.path.join(self.cfg.out_dir, "checkpoints")
        os.makedirs(ckpt_dir, exist_ok=True)
        if monitor_val is not None:
            name = f"epoch={self.epoch:03d}-{self.cfg.monitor}={monitor_val:.3f}.ckpt"
            if monitor_val < self._best_monitor:
                self._best_monitor = monitor_val
                torch.save(ckpt, os.path.join(ckpt_dir, "best.ckpt"))
        else:
            name = f"step={self.global_step}.ckpt"
        torch.save(ckpt, os.path.join(ckpt_dir, name))
        torch.save(ckpt, os.path.join(ckpt_dir, "last.ckpt"))

    def _restore(self, task, optimizer, scheduler, ckpt_path: str):
        ckpt = torch.load(ckpt_path, map_location="cpu", weights_only=False)
        task.load_state_dict(ckpt["state_dict"])
        self.global_step = ckpt.get("global_step", 0)
        self.epoch = ckpt.get("epoch", 0)
        if optimizer is not None and ckpt.get("optimizer_states"):
            optimizer.load_state_dict(ckpt["optimizer_states"][0])
        if scheduler is not None and ckpt.get("lr_schedulers"):
            scheduler.load_state_dict(ckpt["lr_schedulers"][0])


def _move(batch, device):
    if torch.is_tensor(batch):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, dict):
        return {k: _move(v, device) for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        return type(batch)(_move(v, device) for v in batch)
    return batch
