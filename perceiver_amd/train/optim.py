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
    """On CUDA with the HIP extension, the whole update runs as: one fused
    multi-tensor grad cast into a flat fp32 buffer, ONE single-pass adamw_step
    kernel over the flat master/m/v/grad buffers, one fused write-back to the
    bf16 parameters. Falls back to torch._foreach_* elsewhere."""

    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8, weight_decay: float = 0.01,
                 max_grad_norm: float = 0.0):
        """``max_grad_norm``: 0 disables; otherwise global-norm gradient clipping
        is applied INSIDE the step on the fp32 gradients — on the fused path one
        norm + one scale kernel over the flat buffer with no host sync, instead
        of torch's per-parameter foreach chain before the step."""
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self._flat = None  # lazy per-group flat state

    def _build_flat(self, group):
        params = [p for p in group["params"] if p.requires_grad]
        n = sum(p.numel() for p in params)
        dev = params[0].device
        flat = {
            "params": params,
            "master": torch.empty(n, dtype=torch.float32, device=dev),
            "m": torch.zeros(n, dtype=torch.float32, device=dev),
            "v": torch.zeros(n, dtype=torch.float32, device=dev),
            "g": torch.empty(n, dtype=torch.float32, device=dev),
            "step": 0,
        }
        views_master, views_g = [], []
        off = 0
        for p in params:
            k = p.numel()
            mv = flat["master"][off: off + k].view_as(p)
            mv.copy_(p.detach().to(torch.float32))
            views_master.append(mv)
            views_g.append(flat["g"][off: off + k].view_as(p))
            off += k
        flat["views_master"] = views_master
        flat["views_g"] = views_g
        return flat

    def _fused_available(self, group):
        from perceiver_amd.ops import hip

        p0 = group["params"][0]
        return p0.is_cuda and hip.is_available()

    def _step_fused(self, group):
        from perceiver_amd.ops import hip

        if self._flat is None:
            self._flat = {}
        gid = id(group)
        if gid not in self._flat:
            self._flat[gid] = self._build_flat(group)
        fl = self._flat[gid]
        fl["step"] += 1
        grads = [p.grad for p in fl["params"]]
        assert all(g is not None for g in grads), "missing gradient in fused AdamW step"
        torch._foreach_copy_(fl["views_g"], grads)
        if self.max_grad_norm > 0:
            scale = (self.max_grad_norm / (fl["g"].norm() + 1e-6)).clamp(max=1.0)
            fl["g"].mul_(scale)
        beta1, beta2 = group["betas"]
        hip.ext().adamw_step(fl["master"], fl["m"], fl["v"], fl["g"],
                             group["lr"], beta1, beta2, group["eps"],
                             group["weight_decay"], fl["step"])
        torch._foreach_copy_(fl["params"], fl["views_master"])

    def state_dict(self):
        """Include the fused-path flat state (master weights, moments, step) —
        it lives outside ``self.state`` and would otherwise silently reset on
        resume (losing both the Adam moments and the fp32 master precision)."""
        sd = super().state_dict()
        if self._flat:
            fused = {}
            for gi, group in enumerate(self.param_groups):
                fl = self._flat.get(id(group))
                if fl is not None:
                    fused[gi] = {"master": fl["master"], "m": fl["m"], "v": fl["v"],
                                 "step": fl["step"]}
            sd["fused_flat"] = fused
        return sd

    def load_state_dict(self, state_dict):
        state_dict = dict(state_dict)
        fused = state_dict.pop("fused_flat", None)
        super().load_state_dict(state_dict)
        if fused:
            self._flat = {}
            for gi, group in enumerate(self.param_groups):
                rec = fused.get(gi, fused.get(str(gi)))
                if rec is None:
                    continue
                fl = self._build_flat(group)
                fl["master"].copy_(rec["master"].to(fl["master"].device))
                fl["m"].copy_(rec["m"].to(fl["m"].device))
                fl["v"].copy_(rec["v"].to(fl["v"].device))
                fl["step"] = int(rec["step"])
                # restore the live parameters from the fp32 masters
                with torch.no_grad():
                    torch._foreach_copy_(fl["params"], fl["views_master"])
                self._flat[id(group)] = fl

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            if self._fused_available(group):
                self._step_fused(group)
                continue
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
            if self.max_grad_norm > 0:
                total = torch.linalg.vector_norm(
                    torch.stack([torch.linalg.vector_norm(g) for g in grads]))
                scale = (self.max_grad_norm / (total + 1e-6)).clamp(max=1.0)
                torch._foreach_mul_(grads, scale)

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


class Lamb(torch.optim.Optimizer):
    """LAMB (layer-wise adaptive moments, You et al. 2020) — the optimizer the
    reference's MLM example trains with (its CLI exposes it via torch_optimizer,
    reference scripts/cli.py:1-47; not installable offline, so implemented here).
    AdamW-style moments with decoupled weight decay, update scaled per parameter
    tensor by the trust ratio ||p|| / ||update||."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999), eps: float = 1e-6,
                 weight_decay: float = 0.01, clamp_trust: float = 10.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        clamp_trust=clamp_trust)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(g)
                    state["exp_avg_sq"] = torch.zeros_like(g)
                state["step"] += 1
                t = state["step"]
                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(beta1).add_(g, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                m_hat = m / (1 - beta1 ** t)
                v_hat = v / (1 - beta2 ** t)
                update = m_hat / (v_hat.sqrt() + group["eps"])
                if group["weight_decay"] != 0:
                    update = update + group["weight_decay"] * p.float()
                p_norm = p.float().norm()
                u_norm = update.norm()
                trust = torch.where(
                    (p_norm > 0) & (u_norm > 0),
                    (p_norm / u_norm).clamp(max=group["clamp_trust"]),
                    torch.ones_like(p_norm),
                )
                p.add_((update * (-group["lr"] * trust)).to(p.dtype))
        return loss
