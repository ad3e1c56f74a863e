"""Autograd binding for the fused residual dropout-add kernel.

``out = residual + dropout(x, p)`` in one memory pass; the dropout mask is
regenerated in the backward from the saved (seed, index) counter hash, so
nothing besides the seed is kept alive.
"""
from __future__ import annotations

import torch

from perceiver_amd.ops import hip


class DropoutAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, p: float):
        seed = int(torch.randint(0, 2**62, (1,)).item())
        out = hip.ext().dropout_add_fwd(x.contiguous(), res.contiguous(), float(p), seed)
        ctx.p = float(p)
        ctx.seed = seed
        return out

    @staticmethod
    def backward(ctx, dy):
        dx = hip.ext().dropout_add_bwd(dy, ctx.p, ctx.seed)
        return dx, dy, None


def dropout_add(x, res, p: float):
    return DropoutAdd.apply(x, res, p)


def can_use_dropout_add(x, res, p: float, training: bool) -> bool:
    return (training and p > 0.0 and hip.is_available()
            and x.is_cuda and x.dtype == torch.bfloat16
            and res.dtype == torch.bfloat16 and x.numel() % 8 == 0
            and x.shape == res.shape)
