"""Fused bias+GELU autograd op (K6 epilogue piece): y = gelu(x + b) in one pass on
GPU; bias gradient = column-sum of the pre-activation gradient."""
from __future__ import annotations

import torch

from perceiver_amd.ops import hip


class GeluBias(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        y = hip.ext().gelu_bias_fwd(x, bias)
        ctx.save_for_backward(x, bias if bias is not None else torch.tensor([]))
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        b = bias if ctx.has_bias else None
        dx = hip.ext().gelu_bias_bwd(x, b, dy.contiguous())
        db = None
        if ctx.has_bias and ctx.needs_input_grad[1]:
            db = dx.float().sum(dim=tuple(range(dx.dim() - 1))).to(dx.dtype)
        return dx, db


def can_fuse_gelu_bias(x: torch.Tensor) -> bool:
    return (
        x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 8 == 0 and hip.is_available()
    )
