"""Linear with a fast bias-gradient path.

For tall activations (hundreds of thousands of rows in the big-KV encoder
cross-attentions), torch's generic reduce computes the bias gradient far off
the bandwidth roofline. ``PerceiverLinear`` keeps nn.Linear's state-dict and
forward exactly, but its backward computes ``db`` with the coalesced
column-sum kernel (ops/csrc/colsum.hip); ``dx``/``dw`` stay on hipBLASLt
through torch.matmul (the same GEMMs torch's own backward runs).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from perceiver_amd.ops import hip


class _ColsumLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        return F.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = dy.matmul(weight)
        if ctx.needs_input_grad[1]:
            dw = dy2.t().matmul(x.reshape(-1, x.shape[-1]))
        if ctx.needs_input_grad[2]:
            db = hip.ext().colsum_bf16(dy2).to(dy.dtype)
        return dx, dw, db


class PerceiverLinear(nn.Linear):
    """nn.Linear whose bias grad uses the column-sum kernel on tall inputs."""

    _MIN_ROWS = 8192  # below this torch's reduce is fine

    def forward(self, x):
        if (self.bias is not None and x.is_cuda and x.dtype == torch.bfloat16
                and self.weight.dtype == torch.bfloat16
                and self.out_features <= 2048
                and x.numel() >= self._MIN_ROWS * x.shape[-1]
                and not torch.is_autocast_enabled()
                and hip.is_available()):
            return _ColsumLinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)
