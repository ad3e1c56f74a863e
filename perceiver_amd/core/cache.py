"""Preallocated KV cache for autoregressive decoding.

The legacy protocol concatenates (k, v) tuples each step — an O(cache) reallocating
copy per layer per token. StaticKVCache preallocates (B, capacity, C) buffers once
and appends in place; the flash kernels read the live prefix as a strided view
(batch stride = capacity * C), so no copies happen on the decode path. Sliding-
window truncation shifts in place (only active once the window is full).
"""
from __future__ import annotations

from typing import List, Tuple, Union

import torch


class StaticKVCache:
    def __init__(self, batch: int, capacity: int, k_channels: int, v_channels: int,
                 device=None, dtype=torch.float32):
        self.k_buf = torch.empty(batch, capacity, k_channels, device=device, dtype=dtype)
        self.v_buf = torch.empty(batch, capacity, v_channels, device=device, dtype=dtype)
        self.length = 0
        # pre_rotated: keys are stored ALREADY rotary-rotated at their absolute
        # positions, so attention skips the O(cache) re-rotation every step.
        # Scores are identical (RoPE depends on position differences only), but
        # baked-in rotations are incompatible with window sliding — only the
        # graph decoder (which forbids truncation) sets this.
        self.pre_rotated = False

    @property
    def capacity(self) -> int:
        return self.k_buf.shape[1]

    def enable_graph_append(self, len_dev: torch.Tensor):
        """Switch to device-indexed appends for hipGraph capture: the write index
        lives in ``len_dev`` (1-elem int64 device tensor) instead of the host
        ``length`` counter, and ``append`` returns FULL-capacity views (callers
        mask the tail ``j > len`` themselves). No host sync anywhere on the path,
        so the step can be stream-captured and replayed."""
        self._len_dev = len_dev

    def disable_graph_append(self):
        self._len_dev = None

    def append(self, k_new: torch.Tensor, v_new: torch.Tensor):
        """Write new keys/values in place; returns (k, v) views over the full live
        prefix (length includes the new entries)."""
        if getattr(self, "_len_dev", None) is not None:
            if k_new.shape[1] != 1:
                raise RuntimeError("graph-mode append is single-token only")
            self.k_buf.index_copy_(1, self._len_dev, k_new)
            self.v_buf.index_copy_(1, self._len_dev, v_new)
            return self.k_buf, self.v_buf
        n = k_new.shape[1]
        if self.length + n > self.capacity:
            raise RuntimeError(
                f"KV cache overflow: {self.length}+{n} > capacity {self.capacity}"
            )
        self.k_buf[:, self.length: self.length + n].copy_(k_new)
        self.v_buf[:, self.length: self.length + n].copy_(v_new)
        self.length += n
        return self.k_buf[:, : self.length], self.v_buf[:, : self.length]

    def reset(self):
        self.length = 0

    def truncate_front_to(self, max_len: int):
        """Keep only the last ``max_len`` entries (window slide)."""
        if self.length > max_len:
            drop = self.length - max_len
            self.k_buf[:, :max_len].copy_(self.k_buf[:, drop: self.length].clone())
            self.v_buf[:, :max_len].copy_(self.v_buf[:, drop: self.length].clone())
            self.length = max_len

    def index_select_batch(self, idx: torch.Tensor) -> "StaticKVCache":
        """Beam-search reorder."""
        out = StaticKVCache.__new__(StaticKVCache)
        out.k_buf = self.k_buf.index_select(0, idx)
        out.v_buf = self.v_buf.index_select(0, idx)
        out.length = self.length
        out.pre_rotated = self.pre_rotated
        out._len_dev = None
        return out


KVCacheEntry = Union[StaticKVCache, Tuple[torch.Tensor, torch.Tensor]]


def cache_len(entry: KVCacheEntry) -> int:
    if isinstance(entry, StaticKVCache):
        return entry.length
    return entry[0].shape[1]


def allocate_kv_cache(model, batch: int, device=None, dtype=torch.float32,
                      ca_capacity=None) -> List[StaticKVCache]:
    """Preallocate the [cross-attention, *self-attention] cache list for a
    PerceiverAR-family ``model``.

    ``ca_capacity`` bounds the cross-attention cache below ``max_seq_len`` —
    the graph decoder reads the FULL capacity (masked) every step, so sizing
    the cache to the actual sequence need (bucketed, see hf_base.generate)
    cuts the dead-row KV traffic of short decodes.
    """
    def chans(layer):
        if not hasattr(layer, "num_qk_channels"):  # activation-checkpoint wrapper
            layer = layer.module
        return layer.num_qk_channels, layer.num_v_channels

    cap = model.max_seq_len if ca_capacity is None else min(int(ca_capacity), model.max_seq_len)
    qk, vc = chans(model.cross_attention)
    caches = [StaticKVCache(batch, cap, qk, vc, device=device, dtype=dtype)]
    for layer in model.self_attention:
        qk, vc = chans(layer)
        caches.append(StaticKVCache(batch, model.max_latents, qk, vc, device=device, dtype=dtype))
    return caches
