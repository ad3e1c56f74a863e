"""Compute estimator for Perceiver-AR scaling studies (parity: reference
examples/scaling/clm/flops.py): training FLOPs modeled as a decoder-only
transformer over the latent positions plus the cross-attention extra scaled by
(prefix/latents) * (1 - cross_attention_dropout)."""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ComputeEstimator:
    vocab_size: int
    max_seq_len: int
    max_latents: int
    num_channels: int
    num_layers: int
    cross_attention_dropout: float = 0.5
    widening_factor: int = 4

    @property
    def prefix_len(self) -> int:
        return self.max_seq_len - self.max_latents

    def params(self) -> int:
        """Parameter count excluding embeddings (Chinchilla convention)."""
        d = self.num_channels
        per_layer = 4 * d * d + 2 * self.widening_factor * d * d
        return self.num_layers * per_layer

    def embedding_params(self) -> int:
        return self.vocab_size * self.num_channels

    def forward_flops_per_seq(self) -> float:
        """Dense-compute estimate for one training sequence."""
        d = self.num_channels
        n = self.max_latents
        # self-attention stack over latents (= decoder-only transformer slice)
        proj = 2 * n * (4 * d * d)                      # q,k,v,o projections
        attn = 2 * n * n * d * 2                        # QK^T + PV (causal halves in practice)
        mlp = 2 * n * (2 * self.widening_factor * d * d)
        per_layer = proj + attn + mlp
        total = self.num_layers * per_layer
        # cross-attention extra: latents also attend to the (dropout-thinned) prefix
        kept_prefix = self.prefix_len * (1.0 - self.cross_attention_dropout)
        total += 2 * n * kept_prefix * d * 2            # QK^T + PV against the prefix
        total += 2 * kept_prefix * (2 * d * d)          # prefix k/v projections
        # logits
        total += 2 * n * d * self.vocab_size
        return float(total)

    def train_flops(self, num_sequences: int) -> float:
        """fwd + bwd ~= 3x fwd."""
        return 3.0 * self.forward_flops_per_seq() * num_sequences
