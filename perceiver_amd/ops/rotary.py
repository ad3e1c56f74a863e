"""Fused rotary dispatch: one HIP kernel instead of the
~6-op host chain; the backward is the same rotation with negated sin."""
from __future__ import annotations

import torch


def can_use_fused(pos_enc: torch.Tensor) -> bool:
    from perceiver_amd.ops import hip

    return (hip.is_available() and hip._load() is not None
            and pos_enc.dtype == torch.float32)


class _RotaryFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, t, frq, rot):
        from perceiver_amd.ops import hip

        ctx.save_for_backward(frq)
        ctx.rot = rot
        return hip.ext().rotary_apply(t, frq, rot, False)

    @staticmethod
    def backward(ctx, grad):
        from perceiver_amd.ops import hip

        (frq,) = ctx.saved_tensors
        g = grad.to(torch.bfloat16) if grad.dtype != torch.bfloat16 else grad
        dt = hip.ext().rotary_apply(g.contiguous(), frq, ctx.rot, True)
        return dt, None, None


def fused_rotate(t: torch.Tensor, frq: torch.Tensor, rot: int) -> torch.Tensor:
    """t: (B,H,N,D) bf16 (arbitrary strides); frq: (FB,N,rot) fp32."""
    return _RotaryFn.apply(t, frq, rot)
