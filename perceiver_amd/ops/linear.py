"""Linear with fast bias-gradient and custom-GEMM paths.

For tall activations (hundreds of thousands of rows in the big-KV encoder
cross-attentions), torch's generic reduce computes the bias gradient far off
the bandwidth roofline. ``PerceiverLinear`` keeps nn.Linear's state-dict and
forward exactly, but its backward computes ``db`` with the coalesced
column-sum kernel (ops/csrc/colsum.hip), and the forward GEMM plus the
data-gradient GEMM route through the deep-pipeline CDNA4 kernel
(ops/csrc/gemm_bt.hip) when the shape matches its tiling — both operands of
``y = x @ w^T`` are K-contiguous in nn.Linear's native layout, and the dgrad
``dx = dy @ w`` reuses the same kernel with a (cheap, 2-4 MB) transposed
weight copy. ``dw`` stays on hipBLASLt through torch.matmul.

Set PERCEIVER_NO_CUSTOM_GEMM=1 to keep every GEMM on hipBLASLt (A/B lever).
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from perceiver_amd.ops import hip

_NO_CUSTOM_GEMM = os.environ.get("PERCEIVER_NO_CUSTOM_GEMM", "") == "1"
# Plain GEMMs stay on hipBLASLt by default: with the shipped TunableOp table
# it runs ~850-1100 TF/s on the flagship shapes and the custom kernel
# measures ~0.9x of that (same-box A/B: routing dgrad through gemm_bt +
# transpose cost 1.3 ms/step on the MLM flagship). The custom kernel earns
# its keep where hipBLASLt cannot follow: the fused GEMM+bias+GELU epilogue
# (ops/gelu.py LinearGeluBias), which stays on by default.
_GEMM_FWD = os.environ.get("PERCEIVER_GEMM_FWD", "0") == "1" and not _NO_CUSTOM_GEMM
_GEMM_DGRAD = os.environ.get("PERCEIVER_GEMM_DGRAD", "0") == "1" and not _NO_CUSTOM_GEMM


def _gemm_bt_ok(M: int, N: int, K: int) -> bool:
    return bool(hip.ext().gemm_bt_applicable(M, N, K))


def dgrad_matmul(d2, weight, lead_shape):
    """dx = d @ w for a 2-D upstream grad d2, reshaped to (*lead_shape, K).
    Routes through the custom B^T GEMM with a (cheap) transposed-weight copy
    when the inner dim is large enough to win (measured: 1.07x at 1280,
    parity at 1792, loses below ~1024)."""
    K = weight.shape[1]
    if _GEMM_DGRAD and d2.shape[1] >= 1024 and _gemm_bt_ok(d2.shape[0], K, d2.shape[1]):
        dx = hip.ext().gemm_bt(d2, _transposed(weight), None)
    else:
        dx = d2.matmul(weight)
    return dx.view(*lead_shape, K)


def _transposed(weight: torch.Tensor) -> torch.Tensor:
    """Transposed copy via the LDS-tiled kernel, cached per tensor version —
    weight-shared blocks hit the same weight several times per backward, and
    the version bump from the optimizer's in-place update invalidates."""
    cached = getattr(weight, "_perceiver_t_cache", None)
    if cached is not None and cached[0] == weight._version:
        return cached[1]
    wt = hip.ext().transpose_bf16(weight.detach())
    weight._perceiver_t_cache = (weight._version, wt)
    return wt


class _ColsumLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        lead = x.shape[:-1]
        M, K, N = x.numel() // x.shape[-1], x.shape[-1], weight.shape[0]
        if _GEMM_FWD and _gemm_bt_ok(M, N, K):
            x2 = x.reshape(M, K)
            return hip.ext().gemm_bt(x2, weight, bias).view(*lead, N)
        return F.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = dgrad_matmul(dy2, weight, dy.shape[:-1])
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
