"""Position encodings: absolute positions, rotary (RoPE), inverse-frequency and Fourier.

Semantics match /root/reference/perceiver/model/core/position.py:9-138 (positions with
left-pad shift, RotaryPositionEmbedding with right_align slicing and interleaved
rotate-half, FrequencyPositionEncoding 10000^(-2i/d) outer product repeated x2,
FourierPositionEncoding sin/cos bands + raw coords). On the MI355X compute path the
rotary rotation is applied inside the fused attention kernels from cos/sin tables;
this module is the host-side/reference implementation and the table builder.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn


def positions(
    b: int,
    n: int,
    shift: Optional[torch.Tensor] = None,
    device: Optional[torch.device] = None,
) -> torch.Tensor:
    """Absolute positions 0..n-1 per batch element, shifted left by ``shift`` (left-pad
    correction) and clamped at 0. ``shift`` has shape (b, 1)."""
    pos = torch.arange(n, device=device).unsqueeze(0).expand(b, n)
    if shift is not None:
        if shift.shape != (b, 1):
            raise ValueError(f"shift must have shape {(b, 1)} but has shape {tuple(shift.shape)}")
        pos = pos - shift
    return pos.clamp(min=0)


def rotate_half_interleaved(x: torch.Tensor) -> torch.Tensor:
    """[x1, x2, x3, x4, ...] -> [-x2, x1, -x4, x3, ...] along the last dim."""
    x = x.unflatten(-1, (-1, 2))
    x1, x2 = x.unbind(dim=-1)
    return torch.stack((-x2, x1), dim=-1).flatten(-2)


class RotaryPositionEmbedding:
    """Holds per-position frequency encodings of shape (b, 1, n, c) and rotates the
    first ``rotate_dim`` channels of q/k. Not an nn.Module (no parameters).

    ``right_align=True`` slices the encoding from the right — required when q (latents)
    is shorter than k (prefix + latents) in Perceiver AR.
    """

    def __init__(self, frq_pos_enc: torch.Tensor, right_align: bool = False):
        # (b, n, c) -> (b, 1, n, c): broadcast over heads
        self.frq_pos_enc = frq_pos_enc.unsqueeze(1)
        self.rotate_dim = frq_pos_enc.shape[-1]
        self.right_align = right_align

    def rotate(self, t: torch.Tensor) -> torch.Tensor:
        seq_len = t.shape[-2]
        if self.right_align:
            pos_enc = self.frq_pos_enc[..., -seq_len:, :]
        else:
            pos_enc = self.frq_pos_enc[..., :seq_len, :]

        if t.is_cuda and t.dtype == torch.bfloat16 and t.dim() == 4:
            from perceiver_amd.ops import rotary

            if rotary.can_use_fused(pos_enc):
                return rotary.fused_rotate(t, pos_enc.squeeze(1), self.rotate_dim)

        t_rot, t_pass = t[..., : self.rotate_dim], t[..., self.rotate_dim :]
        # rotation in the encoding dtype (fp32 tables under bf16 training), result
        # cast back to t's dtype so the bf16 fused-kernel path stays bf16
        t_rot = (t_rot * pos_enc.cos() + rotate_half_interleaved(t_rot) * pos_enc.sin()).to(t.dtype)
        return torch.cat((t_rot, t_pass), dim=-1)


class FrequencyPositionEncoding(nn.Module):
    """phi[b, n, 2i] = phi[b, n, 2i+1] = pos[b, n] * 10000^(-2i/dim)."""

    def __init__(self, dim: int):
        super().__init__()
        inv_freq = 1.0 / (10000 ** (torch.arange(0, dim, 2).float() / dim))
        self.register_buffer("inv_freq", inv_freq)

    def forward(self, abs_pos: torch.Tensor) -> torch.Tensor:
        # outer product (b, n) x (f,) -> (b, n, f), then interleave-repeat x2
        pos_enc = abs_pos.to(self.inv_freq.dtype).unsqueeze(-1) * self.inv_freq
        return pos_enc.repeat_interleave(2, dim=-1)


class FourierPositionEncoding(nn.Module):
    """Precomputed Fourier features of an nd grid: raw coords in [-1, 1] plus
    sin/cos at ``num_frequency_bands`` frequencies linspace(1, max_freq/2) per dim,
    flattened to (prod(shape), C_pos) with C_pos = len(shape) * (2*bands + 1)."""

    def __init__(self, input_shape: Tuple[int, ...], num_frequency_bands: int):
        super().__init__()
        self.input_shape = tuple(input_shape)
        self.num_frequency_bands = num_frequency_bands

        coords = [torch.linspace(-1.0, 1.0, steps=s) for s in self.input_shape]
        pos = torch.stack(torch.meshgrid(*coords, indexing="ij"), dim=len(self.input_shape))

        enc = [pos]
        grids = []
        for i, max_freq in enumerate(self.input_shape):
            freqs = torch.linspace(1.0, max_freq / 2.0, num_frequency_bands)
            grids.append(pos[..., i : i + 1] * freqs)
        enc.extend(torch.sin(math.pi * g) for g in grids)
        enc.extend(torch.cos(math.pi * g) for g in grids)
        enc = torch.cat(enc, dim=-1).flatten(0, len(self.input_shape) - 1)
        self.register_buffer("position_encoding", enc)

    def num_position_encoding_channels(self, include_positions: bool = True) -> int:
        return len(self.input_shape) * (2 * self.num_frequency_bands + include_positions)

    def forward(self, b: int) -> torch.Tensor:
        return self.position_encoding.unsqueeze(0).expand(b, *self.position_encoding.shape)
