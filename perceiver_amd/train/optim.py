"""Optimizers for pure-bf16 training on MI355X.

MasterAdamW keeps bf16 model parameters (no autocast cast traffic: the eager-baseline
profile showed ~23 ms/step of bf16<->fp32 copy kernels, profiles/r01_mlm_eager_baseline.md)
while doing the AdamW update in fp32 against a master copy held in optimizer state —
fp32 master weights + moments, bf16 weights/grads, decoupled weight decay.
Implemented with torch._foreach_* (single C++ dispatch per op across all params).
"""
from __future__ import annotations

import math
from typing import Iterable

import torch


class MasterAdamW(torch.optim.Optimizer):
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8, weight_decay: float = 0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, raw_grads, grads, masters, exp_avgs, exp_avg_sqs, steps = [], [], [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["master"] = p.detach().to(torch.float32, copy=True)
                    state["exp_avg"] = torch.zeros_like(state["master"])
                    state["exp_avg_sq"] = torch.zeros_like(state["master"])
                    state["grad32"] = torch.empty_like(state["master"])
                state["step"] += 1
                params.append(p)
                raw_grads.append(p.grad)
                grads.append(state["grad32"])
                masters.append(state["master"])
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
                steps.append(state["step"])

            if not params:
                continue

            # single fused multi-tensor cast instead of one cast kernel per tensor
            torch._foreach_copy_(grads, raw_grads)

            beta1, beta2 = group["betas"]
            lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]

            # decoupled weight decay on the master weights
            if wd != 0:
                torch._foreach_mul_(masters, 1.0 - lr * wd)

            torch._foreach_lerp_(exp_avgs, grads, 1.0 - beta1)
            torch._foreach_mul_(exp_avg_sqs, beta2)
            torch._foreach_addcmul_(exp_avg_sqs, grads, grads, 1.0 - beta2)

            # bias correction (per-tensor step counts may differ across groups)
            step_sizes = [lr / (1.0 - beta1 ** t) for t in steps]
            bc2 = [math.sqrt(1.0 - beta2 ** t) for t in steps]

            denoms = torch._foreach_sqrt(exp_avg_sqs)
            torch._foreach_div_(denoms, bc2)
            torch._foreach_add_(denoms, eps)

            updates = torch._foreach_div(exp_avgs, denoms)
            torch._foreach_mul_(updates, step_sizes)
            torch._foreach_sub_(masters, updates)

            # write back to the live (possibly bf16) parameters (fused multi-tensor)
            torch._foreach_copy_(params, masters)

        return loss


def convert_to_bf16_training(model: torch.nn.Module) -> torch.nn.Module:
    """Cast parameters/buffers to bf16 for cast-free training, keeping position-
    encoding buffers in fp32 (RoPE/Fourier tables need fp32 phase precision; the
    rotary rotation computes in fp32 and casts back — see core/position.py)."""
    from perceiver_amd.core.position import FourierPositionEncoding, FrequencyPositionEncoding

    model = model.to(torch.bfloat16)
    for mod in model.modules():
        if isinstance(mod, (FourierPositionEncoding, FrequencyPositionEncoding)):
            mod.float()
    return model
