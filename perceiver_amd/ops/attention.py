"""Functional attention core with MI355X dispatch.

All Perceiver attention variants (encoder cross-attention, latent self-attention,
Perceiver-AR causal cross-attention, decoder cross-attention, cached decode) funnel
through :func:`scaled_dot_attention`. On a gfx950 GPU with the in-tree HIP extension
loaded, the fused flash-style MFMA kernel runs; elsewhere (CPU tests, numerics
references) a plain PyTorch composition with identical semantics runs.

Semantics (parity with /root/reference/perceiver/model/core/modules.py:90-170):
  - q arrives pre-scaled by (Dqk)^-0.5 (the module does the scaling),
  - pad_mask: (B, Lk) bool, True = padding -> masked with -finfo.max,
  - causal: mask = triu(Lk - Lq + 1), i.e. q/k right-aligned when lengths differ,
  - softmax over the key axis, optional dropout on the probabilities,
  - output = probs @ v.
"""
from __future__ import annotations

from typing import Optional

import torch


def eager_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    pad_mask: Optional[torch.Tensor] = None,
    causal: bool = False,
    dropout_p: float = 0.0,
    training: bool = False,
    max_heads_parallel: Optional[int] = None,
) -> torch.Tensor:
    """Reference composition. q, k, v: (B, H, Nq/Lk, D). Returns (B, H, Nq, Dv)."""
    nq, lk = q.shape[-2], k.shape[-2]

    if pad_mask is not None:
        pad_mask = pad_mask[:, None, None, :]

    causal_mask = None
    if causal:
        causal_mask = torch.ones((nq, lk), device=q.device, dtype=torch.bool).triu(lk - nq + 1)

    chunk = max_heads_parallel or q.shape[1]
    outs = []
    for qc, kc, vc in zip(q.split(chunk, dim=1), k.split(chunk, dim=1), v.split(chunk, dim=1)):
        scores = torch.matmul(qc, kc.transpose(-2, -1))
        neg = -torch.finfo(scores.dtype).max
        if pad_mask is not None:
            scores = scores.masked_fill(pad_mask, neg)
        if causal_mask is not None:
            scores = scores.masked_fill(causal_mask, neg)
        probs = scores.softmax(dim=-1)
        if dropout_p > 0.0:
            probs = torch.nn.functional.dropout(probs, p=dropout_p, training=training)
        outs.append(torch.matmul(probs, vc))
    return torch.cat(outs, dim=1)


def _hip_available() -> bool:
    from perceiver_amd.ops import hip

    return hip.is_available()


def scaled_dot_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    pad_mask: Optional[torch.Tensor] = None,
    causal: bool = False,
    dropout_p: float = 0.0,
    training: bool = False,
    max_heads_parallel: Optional[int] = None,
) -> torch.Tensor:
    """Dispatching attention core. Shapes as in :func:`eager_attention`.

    Batch-1 queries (learned latent/output-query arrays) broadcast against
    batch-N keys, matching the reference's einsum broadcasting."""
    if q.shape[0] != k.shape[0]:
        if q.shape[0] == 1:
            q = q.expand(k.shape[0], *q.shape[1:])
        elif k.shape[0] == 1:
            k = k.expand(q.shape[0], *k.shape[1:])
            v = v.expand(q.shape[0], *v.shape[1:])
        else:
            raise ValueError(f"incompatible batch sizes {q.shape[0]} vs {k.shape[0]}")
    if q.is_cuda:
        from perceiver_amd.ops import hip

        if hip.can_use_flash(q, k, v, dropout_p=dropout_p, training=training):
            return hip.flash_attention(q, k, v, pad_mask=pad_mask, causal=causal,
                                       dropout_p=dropout_p, training=training)
    return eager_attention(
        q, k, v, pad_mask=pad_mask, causal=causal, dropout_p=dropout_p,
        training=training, max_heads_parallel=max_heads_parallel,
    )
