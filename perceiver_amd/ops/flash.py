"""Autograd binding for the fused CDNA4 flash attention kernels."""
from __future__ import annotations

from typing import Optional

import torch

from perceiver_amd.ops import hip


def _to_bf16(t: torch.Tensor) -> torch.Tensor:
    return t if t.dtype == torch.bfloat16 else t.to(torch.bfloat16)


class FlashAttention(torch.autograd.Function):
    """out = softmax(mask(q k^T)) v with online softmax on device.

    q arrives pre-scaled. pad_mask: (B, Lk) bool, True = padding. causal uses the
    right-aligned convention (j > Lk - Nq + i masked). Saves (q, k, v, out, lse)
    for the recompute-based backward.
    """

    @staticmethod
    def forward(ctx, q, k, v, pad_mask: Optional[torch.Tensor], causal: bool,
                dropout_p: float, training: bool):
        # last-dim contiguity is enough (the kernels take batch/head/seq strides);
        # head-transposed views and preallocated cache buffers pass zero-copy
        q, k, v = _to_bf16(q), _to_bf16(k), _to_bf16(v)
        p = float(dropout_p) if training else 0.0
        seed = int(torch.randint(0, 2**62, (1,)).item()) if p > 0 else 0
        out, lse = hip.ext().flash_fwd(q, k, v, pad_mask, causal, p, seed)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.pad_mask = pad_mask
        ctx.causal = causal
        ctx.dropout = (p, seed)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        p, seed = ctx.dropout
        dq, dk, dv = hip.ext().flash_bwd(
            _to_bf16(dout), q, k, v, out, lse, ctx.pad_mask, ctx.causal, p, seed
        )
        return dq, dk, dv, None, None, None, None
