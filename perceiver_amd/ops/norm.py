"""Fused bf16 LayerNorm: drop-in nn.LayerNorm subclass dispatching to the CDNA4
kernel (ops/csrc/layer_norm.hip) for bf16 CUDA inputs; eager elsewhere. State-dict
layout identical to nn.LayerNorm."""
from __future__ import annotations

import torch
import torch.nn as nn

from perceiver_amd.ops import hip


class _FusedLayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        y, mean, rstd = hip.ext().ln_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        needs_dwdb = ctx.needs_input_grad[1] or ctx.needs_input_grad[2]
        dx, dw, db = hip.ext().ln_bwd(dy, x, weight, mean, rstd, needs_dwdb)
        return dx, dw if ctx.needs_input_grad[1] else None, \
            db if ctx.needs_input_grad[2] else None, None


class LayerNorm(nn.LayerNorm):
    def forward(self, x):
        if (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and self.weight is not None
            and self.weight.dtype == torch.bfloat16
            and len(self.normalized_shape) == 1
            and self.normalized_shape[0] <= 2048
            and hip.is_available()
        ):
            return _FusedLayerNormFn.apply(x.contiguous(), self.weight, self.bias, self.eps)
        return super().forward(x)
