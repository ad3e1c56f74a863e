"""Perceiver / Perceiver IO / Perceiver AR core modules, MI355X-native.

Functional parity with /root/reference/perceiver/model/core/modules.py (see SURVEY.md
§2.1 for the per-component contract), re-designed for the MI355X stack:

  - every attention variant funnels through ``perceiver_amd.ops.scaled_dot_attention``,
    which dispatches to the fused CDNA4 flash kernels on GPU (K1/K3/K4/K7/K9 of
    SURVEY.md §2.3) and to a plain PyTorch composition on CPU;
  - activation checkpointing uses ``torch.utils.checkpoint`` (no fairscale);
  - state-dict key layout follows the reference checkpoint format
    (cross_attn_1/self_attn_1/..., latent_provider._query); self-attention layers
    store one merged ``qkv_proj`` parameter (rows [q|k|v]) and load legacy split
    q_proj/k_proj/v_proj keys through a pre-hook.

KV caches are (k, v) tuples of shape (B, L, num_qk_channels)/(B, L, num_v_channels),
concatenated along the sequence dim *before* the head split
(reference modules.py:117-121).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from perceiver_amd.core.position import RotaryPositionEmbedding
from perceiver_amd.ops.linear import PerceiverLinear
from perceiver_amd.core.utils import ModuleOutput, Residual, init_parameters
from perceiver_amd.core.adapter import (
    InputAdapter,
    OutputAdapter,
    QueryProvider,
    RotarySupport,
    TiedTokenOutputAdapter,
    TokenInputAdapterWithRotarySupport,
    TrainableQueryProvider,
)
from perceiver_amd.core.config import CausalSequenceModelConfig
from perceiver_amd.core.position import positions
from perceiver_amd.core.cache import StaticKVCache, cache_len
from perceiver_amd.ops.attention import scaled_dot_attention
from perceiver_amd.ops.norm import LayerNorm

KVCache = Tuple[torch.Tensor, torch.Tensor]


class MultiHeadAttention(nn.Module):
    """The single attention primitive used by every Perceiver variant.

    Independently configurable ``num_qk_channels`` / ``num_v_channels`` (both may
    differ from the model dim), optional rotary rotation of q and/or k, boolean pad
    mask (True = padding), causal masking with right-aligned q/k when lengths differ,
    and a (k, v) cache concatenated along the sequence dim before the head split.
    Parity: reference modules.py:23-170.
    """

    def __init__(
        self,
        num_heads: int,
        num_q_input_channels: int,
        num_kv_input_channels: int,
        num_qk_channels: Optional[int] = None,
        num_v_channels: Optional[int] = None,
        num_output_channels: Optional[int] = None,
        max_heads_parallel: Optional[int] = None,
        causal_attention: bool = False,
        dropout: float = 0.0,
        qkv_bias: bool = True,
        out_bias: bool = True,
        merged_qkv: bool = False,
    ):
        super().__init__()

        if num_qk_channels is None:
            num_qk_channels = num_q_input_channels
        if num_v_channels is None:
            num_v_channels = num_qk_channels
        if num_output_channels is None:
            num_output_channels = num_q_input_channels

        if num_qk_channels % num_heads != 0:
            raise ValueError("num_qk_channels must be divisible by num_heads")
        if num_v_channels % num_heads != 0:
            raise ValueError("num_v_channels must be divisible by num_heads")

        self.dp_scale = (num_qk_channels // num_heads) ** -0.5
        self.num_heads = num_heads
        self.num_qk_channels = num_qk_channels
        self.num_v_channels = num_v_channels
        self.causal_attention = causal_attention
        self.attention_dropout = dropout
        self.max_heads_parallel = max_heads_parallel

        if merged_qkv and num_q_input_channels == num_kv_input_channels:
            # self-attention: ONE merged projection parameter — one wide GEMM per
            # forward instead of three skinny ones plus a per-call weight concat.
            # qkv_proj.weight rows are [q | k | v] in the reference's per-matrix
            # order (modules.py:50-55); the load hook accepts legacy checkpoints
            # with split q_proj/k_proj/v_proj keys.
            self.qkv_proj = PerceiverLinear(
                num_kv_input_channels, 2 * num_qk_channels + num_v_channels, bias=qkv_bias
            )
            self._register_load_state_dict_pre_hook(
                MultiHeadAttention._merge_sd_hook, with_module=True
            )
        else:
            self.q_proj = PerceiverLinear(num_q_input_channels, num_qk_channels, bias=qkv_bias)
            self.k_proj = PerceiverLinear(num_kv_input_channels, num_qk_channels, bias=qkv_bias)
            self.v_proj = PerceiverLinear(num_kv_input_channels, num_v_channels, bias=qkv_bias)
        self.o_proj = PerceiverLinear(num_v_channels, num_output_channels, bias=out_bias)
        # kept as a module for state-dict/layout parity; the dispatch path applies
        # dropout functionally inside the attention core
        self.dropout = nn.Dropout(dropout)

    @staticmethod
    def _merge_sd_hook(module, state_dict, prefix, *args):
        """Accept split q/k/v keys and fold them into the merged projection."""
        if prefix + "q_proj.weight" in state_dict:
            state_dict[prefix + "qkv_proj.weight"] = torch.cat(
                [state_dict.pop(prefix + n + "_proj.weight") for n in ("q", "k", "v")], dim=0
            )
        if prefix + "q_proj.bias" in state_dict:
            state_dict[prefix + "qkv_proj.bias"] = torch.cat(
                [state_dict.pop(prefix + n + "_proj.bias") for n in ("q", "k", "v")], dim=0
            )

    def _split_heads(self, x: torch.Tensor) -> torch.Tensor:
        b, n, _ = x.shape
        return x.view(b, n, self.num_heads, -1).transpose(1, 2)

    def _merge_heads(self, x: torch.Tensor) -> torch.Tensor:
        b, h, n, c = x.shape
        return x.transpose(1, 2).reshape(b, n, h * c)

    def forward(
        self,
        x_q: torch.Tensor,
        x_kv: torch.Tensor,
        pad_mask: Optional[torch.Tensor] = None,
        rot_pos_emb_q: Optional[RotaryPositionEmbedding] = None,
        rot_pos_emb_k: Optional[RotaryPositionEmbedding] = None,
        kv_cache: Optional[KVCache] = None,
    ) -> ModuleOutput:
        qk, vc = self.num_qk_channels, self.num_v_channels
        if hasattr(self, "qkv_proj"):
            if x_q is x_kv:
                qkv = self.qkv_proj(x_q)
                q, k, v = qkv.split([qk, qk, vc], dim=-1)
            else:
                # same tensors, different inputs (e.g. cached decode through a
                # SelfAttention layer): sliced views of the merged parameter
                w, bias = self.qkv_proj.weight, self.qkv_proj.bias
                q = torch.nn.functional.linear(x_q, w[:qk], None if bias is None else bias[:qk])
                k = torch.nn.functional.linear(x_kv, w[qk: 2 * qk], None if bias is None else bias[qk: 2 * qk])
                v = torch.nn.functional.linear(x_kv, w[2 * qk:], None if bias is None else bias[2 * qk:])
        else:
            q = self.q_proj(x_q)
            k = self.k_proj(x_kv)
            v = self.v_proj(x_kv)

        pre_rotate = (
            isinstance(kv_cache, StaticKVCache)
            and kv_cache.pre_rotated
            and rot_pos_emb_k is not None
        )
        if pre_rotate:
            # rotate only the NEW rows (right-aligned table rows are exactly
            # their positions) and store them rotated — the cached prefix keeps
            # its baked-in rotation instead of being re-rotated every step
            k = self._merge_heads(rot_pos_emb_k.rotate(self._split_heads(k)))

        if kv_cache is not None:
            if isinstance(kv_cache, StaticKVCache):
                # in-place append; k/v become strided views over the live prefix
                k, v = kv_cache.append(k, v)
            else:
                k_cache, v_cache = kv_cache
                k = torch.cat([k_cache, k], dim=1)
                v = torch.cat([v_cache, v], dim=1)
                kv_cache = (k, v)

        q = self._split_heads(q) * self.dp_scale
        k = self._split_heads(k)
        v = self._split_heads(v)

        if rot_pos_emb_q is not None:
            q = rot_pos_emb_q.rotate(q)
        if rot_pos_emb_k is not None and not pre_rotate:
            k = rot_pos_emb_k.rotate(k)

        o = scaled_dot_attention(
            q,
            k,
            v,
            pad_mask=pad_mask,
            causal=self.causal_attention,
            dropout_p=self.attention_dropout,
            training=self.training,
            max_heads_parallel=self.max_heads_parallel,
        )
        o = self.o_proj(self._merge_heads(o))
        return ModuleOutput(last_hidden_state=o, kv_cache=kv_cache)


class CrossAttention(nn.Module):
    """Pre-LN cross-attention. With ``x_kv_prefix`` the KV sequence is
    cat([kv_norm(prefix), q_norm(x_q)]) so the queries attend to themselves at the end
    of the KV sequence (Perceiver AR). Parity: reference modules.py:173-230."""

    def __init__(
        self,
        num_heads: int,
        num_q_input_channels: int,
        num_kv_input_channels: int,
        num_qk_channels: Optional[int] = None,
        num_v_channels: Optional[int] = None,
        max_heads_parallel: Optional[int] = None,
        causal_attention: bool = False,
        dropout: float = 0.0,
        qkv_bias: bool = True,
        out_bias: bool = True,
    ):
        super().__init__()
        self.q_norm = LayerNorm(num_q_input_channels)
        self.kv_norm = LayerNorm(num_kv_input_channels)
        self.attention = MultiHeadAttention(
            num_heads=num_heads,
            num_q_input_channels=num_q_input_channels,
            num_kv_input_channels=num_kv_input_channels,
            num_qk_channels=num_qk_channels,
            num_v_channels=num_v_channels,
            max_heads_parallel=max_heads_parallel,
            causal_attention=causal_attention,
            dropout=dropout,
            qkv_bias=qkv_bias,
            out_bias=out_bias,
        )

    def forward(
        self,
        x_q: torch.Tensor,
        x_kv: Optional[torch.Tensor] = None,
        x_kv_prefix: Optional[torch.Tensor] = None,
        pad_mask: Optional[torch.Tensor] = None,
        rot_pos_emb_q: Optional[RotaryPositionEmbedding] = None,
        rot_pos_emb_k: Optional[RotaryPositionEmbedding] = None,
        kv_cache: Optional[KVCache] = None,
    ) -> ModuleOutput:
        x_q = self.q_norm(x_q)
        if x_kv is None:
            x_kv = torch.cat([self.kv_norm(x_kv_prefix), x_q], dim=1)
        else:
            x_kv = self.kv_norm(x_kv)
        return self.attention(
            x_q, x_kv, pad_mask=pad_mask, rot_pos_emb_q=rot_pos_emb_q,
            rot_pos_emb_k=rot_pos_emb_k, kv_cache=kv_cache,
        )


class SelfAttention(nn.Module):
    """Pre-LN self-attention (shared norm, same rotary on q and k).
    Parity: reference modules.py:233-278."""

    def __init__(
        self,
        num_heads: int,
        num_channels: int,
        num_qk_channels: Optional[int] = None,
        num_v_channels: Optional[int] = None,
        max_heads_parallel: Optional[int] = None,
        causal_attention: bool = False,
        dropout: float = 0.0,
        qkv_bias: bool = True,
        out_bias: bool = True,
    ):
        super().__init__()
        self.norm = LayerNorm(num_channels)
        self.attention = MultiHeadAttention(
            num_heads=num_heads,
            num_q_input_channels=num_channels,
            num_kv_input_channels=num_channels,
            num_qk_channels=num_qk_channels,
            num_v_channels=num_v_channels,
            max_heads_parallel=max_heads_parallel,
            causal_attention=causal_attention,
            dropout=dropout,
            qkv_bias=qkv_bias,
            out_bias=out_bias,
            merged_qkv=True,
        )

    def forward(
        self,
        x: torch.Tensor,
        pad_mask: Optional[torch.Tensor] = None,
        rot_pos_emb: Optional[RotaryPositionEmbedding] = None,
        kv_cache: Optional[KVCache] = None,
    ) -> ModuleOutput:
        x = self.norm(x)
        return self.attention(
            x, x, pad_mask=pad_mask, rot_pos_emb_q=rot_pos_emb,
            rot_pos_emb_k=rot_pos_emb, kv_cache=kv_cache,
        )


class AbstractAttentionLayer(nn.Sequential):
    """[attention, MLP] pair; threads kv_cache through the attention only."""

    def empty_kv_cache(self, x: torch.Tensor) -> KVCache:
        k_cache = torch.empty(x.shape[0], 0, self.num_qk_channels, dtype=x.dtype, device=x.device)
        v_cache = torch.empty(x.shape[0], 0, self.num_v_channels, dtype=x.dtype, device=x.device)
        return k_cache, v_cache

    def forward(self, *args, kv_cache: Optional[KVCache] = None, **kwargs) -> ModuleOutput:
        attn_output = self[0](*args, kv_cache=kv_cache, **kwargs)
        mlp_output = self[1](attn_output.last_hidden_state)
        return ModuleOutput(last_hidden_state=mlp_output.last_hidden_state, kv_cache=attn_output.kv_cache)


class CrossAttentionLayer(AbstractAttentionLayer):
    def __init__(
        self,
        num_heads: int,
        num_q_input_channels: int,
        num_kv_input_channels: int,
        num_qk_channels: Optional[int] = None,
        num_v_channels: Optional[int] = None,
        max_heads_parallel: Optional[int] = None,
        causal_attention: bool = False,
        widening_factor: int = 1,
        dropout: float = 0.0,
        residual_dropout: float = 0.0,
        attention_residual: bool = True,
        qkv_bias: bool = True,
        out_bias: bool = True,
        mlp_bias: bool = True,
    ):
        cross_attn = CrossAttention(
            num_heads=num_heads,
            num_q_input_channels=num_q_input_channels,
            num_kv_input_channels=num_kv_input_channels,
            num_qk_channels=num_qk_channels,
            num_v_channels=num_v_channels,
            max_heads_parallel=max_heads_parallel,
            causal_attention=causal_attention,
            dropout=dropout,
            qkv_bias=qkv_bias,
            out_bias=out_bias,
        )
        self.num_qk_channels = cross_attn.attention.num_qk_channels
        self.num_v_channels = cross_attn.attention.num_v_channels
        super().__init__(
            Residual(cross_attn, residual_dropout) if attention_residual else cross_attn,
            Residual(MLP(num_q_input_channels, widening_factor, bias=mlp_bias), residual_dropout),
        )


class SelfAttentionLayer(AbstractAttentionLayer):
    def __init__(
        self,
        num_heads: int,
        num_channels: int,
        num_qk_channels: Optional[int] = None,
        num_v_channels: Optional[int] = None,
        max_heads_parallel: Optional[int] = None,
        causal_attention: bool = False,
        widening_factor: int = 1,
        dropout: float = 0.0,
        residual_dropout: float = 0.0,
        qkv_bias: bool = True,
        out_bias: bool = True,
        mlp_bias: bool = True,
    ):
        self_attn = SelfAttention(
            num_heads=num_heads,
            num_channels=num_channels,
            num_qk_channels=num_qk_channels,
            num_v_channels=num_v_channels,
            max_heads_parallel=max_heads_parallel,
            causal_attention=causal_attention,
            dropout=dropout,
            qkv_bias=qkv_bias,
            out_bias=out_bias,
        )
        self.num_qk_channels = self_attn.attention.num_qk_channels
        self.num_v_channels = self_attn.attention.num_v_channels
        super().__init__(
            Residual(self_attn, residual_dropout),
            Residual(MLP(num_channels, widening_factor, bias=mlp_bias), residual_dropout),
        )


def activation_checkpoint_wrapper(layer: AbstractAttentionLayer, offload_to_cpu: bool = False):
    """Per-layer activation checkpointing via torch.utils.checkpoint (the reference
    uses fairscale's checkpoint_wrapper, modules.py:933-956; on MI355X with 288 GB
    HBM this is rarely needed — kept for config parity and the very large KV configs).

    Wraps the layer's forward; kv_cache is not supported under checkpointing.
    ``offload_to_cpu`` additionally parks the checkpoint's saved inputs in host
    memory (torch.autograd.graph.save_on_cpu — the fairscale offload_to_cpu
    equivalent).
    """
    import torch.utils.checkpoint as cp

    class _Checkpointed(nn.Module):
        def __init__(self, inner):
            super().__init__()
            self.module = inner

        def empty_kv_cache(self, x):
            return self.module.empty_kv_cache(x)

        def forward(self, *args, kv_cache=None, **kwargs):
            if kv_cache is not None or not torch.is_grad_enabled():
                return self.module(*args, kv_cache=kv_cache, **kwargs)

            def run(*tensors):
                out = self.module(*args, kv_cache=None, **kwargs)
                return out.last_hidden_state

            if offload_to_cpu:
                pin = torch.cuda.is_available()
                with torch.autograd.graph.save_on_cpu(pin_memory=pin):
                    hidden = cp.checkpoint(run, *args, use_reentrant=False)
            else:
                hidden = cp.checkpoint(run, *args, use_reentrant=False)
            return ModuleOutput(last_hidden_state=hidden, kv_cache=None)

    return _Checkpointed(layer)


class SelfAttentionBlock(nn.Sequential):
    """N stacked self-attention layers; rotary limited to the first
    ``num_rotary_layers`` (-1 = all); per-layer kv-cache list protocol.
    Parity: reference modules.py:370-441."""

    def __init__(
        self,
        num_layers: int,
        num_heads: int,
        num_channels: int,
        num_qk_channels: Optional[int] = None,
        num_v_channels: Optional[int] = None,
        num_rotary_layers: int = 1,
        max_heads_parallel: Optional[int] = None,
        causal_attention: bool = False,
        widening_factor: int = 1,
        dropout: float = 0.0,
        residual_dropout: float = 0.0,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
        qkv_bias: bool = True,
        out_bias: bool = True,
        mlp_bias: bool = True,
    ):
        layers = [
            SelfAttentionLayer(
                num_heads=num_heads,
                num_channels=num_channels,
                num_qk_channels=num_qk_channels,
                num_v_channels=num_v_channels,
                max_heads_parallel=max_heads_parallel,
                causal_attention=causal_attention,
                widening_factor=widening_factor,
                dropout=dropout,
                residual_dropout=residual_dropout,
                qkv_bias=qkv_bias,
                out_bias=out_bias,
                mlp_bias=mlp_bias,
            )
            for _ in range(num_layers)
        ]
        if activation_checkpointing:
            layers = [activation_checkpoint_wrapper(l, offload_to_cpu=activation_offloading) for l in layers]
        self.num_rotary_layers = num_rotary_layers
        super().__init__(*layers)

    def forward(
        self,
        x: torch.Tensor,
        pad_mask: Optional[torch.Tensor] = None,
        rot_pos_emb: Optional[RotaryPositionEmbedding] = None,
        kv_cache: Optional[List[KVCache]] = None,
    ) -> ModuleOutput:
        if kv_cache is None:
            kv_cache_updated = None
        else:
            if len(kv_cache) == 0:
                kv_cache = [layer.empty_kv_cache(x) for layer in self]
            kv_cache_updated = []

        for i, layer in enumerate(self):
            use_rot = i < self.num_rotary_layers or self.num_rotary_layers == -1
            out = layer(
                x,
                pad_mask=pad_mask,
                rot_pos_emb=rot_pos_emb if use_rot else None,
                kv_cache=None if kv_cache is None else kv_cache[i],
            )
            x = out.last_hidden_state
            if kv_cache_updated is not None:
                kv_cache_updated.append(out.kv_cache)

        return ModuleOutput(last_hidden_state=x, kv_cache=kv_cache_updated)


class MLP(nn.Sequential):
    """LayerNorm -> Linear(x widening) -> GELU -> Linear.
    On GPU in bf16 the LayerNorm is the fused HIP LN and the widening Linear's bias
    add fuses into the GELU kernel (the GEMM itself stays on hipBLASLt)."""

    def __init__(self, num_channels: int, widening_factor: int, bias: bool = True):
        super().__init__(
            LayerNorm(num_channels),
            PerceiverLinear(num_channels, widening_factor * num_channels, bias=bias),
            nn.GELU(),
            PerceiverLinear(widening_factor * num_channels, num_channels, bias=bias),
        )

    def forward(self, x: torch.Tensor) -> ModuleOutput:
        from perceiver_amd.ops.gelu import (GeluBias, LinearGeluBias, can_fuse_gelu_bias,
                                            can_fuse_linear_gelu)

        if can_fuse_gelu_bias(x):
            h = self[0](x)
            if self[1].bias is not None and can_fuse_linear_gelu(h, self[1].weight):
                # one kernel: GEMM + bias + GELU (epilogue-fused)
                h = LinearGeluBias.apply(h, self[1].weight, self[1].bias)
            else:
                h = torch.nn.functional.linear(h, self[1].weight)  # bias folds into GELU
                h = GeluBias.apply(h.contiguous(), self[1].bias)
            h = self[3](h)
            return ModuleOutput(last_hidden_state=h)
        return ModuleOutput(last_hidden_state=super().forward(x))


class PerceiverEncoder(nn.Module):
    """Latent cross-attention + iterated self-attention blocks with optional weight
    sharing between repeats. Parity: reference modules.py:457-607."""

    def __init__(
        self,
        input_adapter: InputAdapter,
        num_latents: int,
        num_latent_channels: int,
        num_cross_attention_heads: int = 4,
        num_cross_attention_qk_channels: Optional[int] = None,
        num_cross_attention_v_channels: Optional[int] = None,
        num_cross_attention_layers: int = 1,
        first_cross_attention_layer_shared: bool = False,
        cross_attention_widening_factor: int = 1,
        num_self_attention_heads: int = 4,
        num_self_attention_qk_channels: Optional[int] = None,
        num_self_attention_v_channels: Optional[int] = None,
        num_self_attention_layers_per_block: int = 6,
        num_self_attention_blocks: int = 1,
        first_self_attention_block_shared: bool = True,
        self_attention_widening_factor: int = 1,
        dropout: float = 0.0,
        residual_dropout: float = 0.0,
        init_scale: float = 0.02,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
    ):
        super().__init__()

        if num_cross_attention_layers <= 0:
            raise ValueError("num_cross_attention_layers must be > 0")
        if num_self_attention_blocks <= 0:
            raise ValueError("num_self_attention_blocks must be > 0")
        if num_cross_attention_layers > num_self_attention_blocks:
            raise ValueError("num_cross_attention_layers must be <= num_self_attention_blocks")

        self.latent_provider = TrainableQueryProvider(num_latents, num_latent_channels, init_scale=init_scale)
        self.input_adapter = input_adapter
        self.num_cross_attention_layers = num_cross_attention_layers
        self.num_self_attention_blocks = num_self_attention_blocks
        self.first_cross_attention_layer_shared = first_cross_attention_layer_shared
        self.first_self_attention_block_shared = first_self_attention_block_shared

        def make_cross_attn():
            layer = CrossAttentionLayer(
                num_heads=num_cross_attention_heads,
                num_q_input_channels=num_latent_channels,
                num_kv_input_channels=input_adapter.num_input_channels,
                num_qk_channels=num_cross_attention_qk_channels,
                num_v_channels=num_cross_attention_v_channels,
                widening_factor=cross_attention_widening_factor,
                dropout=dropout,
                residual_dropout=residual_dropout,
            )
            if activation_checkpointing:
                layer = activation_checkpoint_wrapper(layer, offload_to_cpu=activation_offloading)
            return layer

        def make_self_attn():
            return SelfAttentionBlock(
                num_layers=num_self_attention_layers_per_block,
                num_heads=num_self_attention_heads,
                num_channels=num_latent_channels,
                num_qk_channels=num_self_attention_qk_channels,
                num_v_channels=num_self_attention_v_channels,
                widening_factor=self_attention_widening_factor,
                dropout=dropout,
                residual_dropout=residual_dropout,
                activation_checkpointing=activation_checkpointing,
                activation_offloading=activation_offloading,
            )

        self.cross_attn_1 = make_cross_attn()
        self.self_attn_1 = make_self_attn()
        if self.extra_cross_attention_layer:
            self.cross_attn_n = make_cross_attn()
        if self.extra_self_attention_block:
            self.self_attn_n = make_self_attn()

        with torch.no_grad():
            init_parameters(self, init_scale)

    @property
    def extra_cross_attention_layer(self) -> bool:
        return self.num_cross_attention_layers > 1 and not self.first_cross_attention_layer_shared

    @property
    def extra_self_attention_block(self) -> bool:
        return self.num_self_attention_blocks > 1 and not self.first_self_attention_block_shared

    def forward(self, x, pad_mask=None, return_adapted_input: bool = False):
        b = x.shape[0]

        x_adapted = self.input_adapter(x)
        x_latent = self.latent_provider()

        x_latent = self.cross_attn_1(x_latent, x_adapted, pad_mask=pad_mask).last_hidden_state
        x_latent = self.self_attn_1(x_latent).last_hidden_state

        cross_attn_n = self.cross_attn_n if self.extra_cross_attention_layer else self.cross_attn_1
        self_attn_n = self.self_attn_n if self.extra_self_attention_block else self.self_attn_1

        for i in range(1, self.num_self_attention_blocks):
            if i < self.num_cross_attention_layers:
                x_latent = cross_attn_n(x_latent, x_adapted, pad_mask=pad_mask).last_hidden_state
            x_latent = self_attn_n(x_latent).last_hidden_state

        if return_adapted_input:
            return x_latent, x_adapted
        return x_latent


class PerceiverDecoder(nn.Module):
    """Output-query cross-attention over the latent array + task output adapter.
    Parity: reference modules.py:610-675."""

    def __init__(
        self,
        output_adapter: OutputAdapter,
        output_query_provider: QueryProvider,
        num_latent_channels: int,
        num_cross_attention_heads: int = 4,
        num_cross_attention_qk_channels: Optional[int] = None,
        num_cross_attention_v_channels: Optional[int] = None,
        cross_attention_widening_factor: int = 1,
        cross_attention_residual: bool = True,
        dropout: float = 0.0,
        init_scale: float = 0.02,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
    ):
        super().__init__()
        self.output_query_provider = output_query_provider
        self.output_adapter = output_adapter

        cross_attn = CrossAttentionLayer(
            num_heads=num_cross_attention_heads,
            num_q_input_channels=output_query_provider.num_query_channels,
            num_kv_input_channels=num_latent_channels,
            num_qk_channels=num_cross_attention_qk_channels,
            num_v_channels=num_cross_attention_v_channels,
            widening_factor=cross_attention_widening_factor,
            attention_residual=cross_attention_residual,
            dropout=dropout,
        )
        if activation_checkpointing:
            cross_attn = activation_checkpoint_wrapper(cross_attn, offload_to_cpu=activation_offloading)
        self.cross_attn = cross_attn

        with torch.no_grad():
            init_parameters(self, init_scale)

    def forward(self, x_latent, x_adapted=None, **kwargs):
        output_query = self.output_query_provider(x_adapted)
        output = self.cross_attn(output_query, x_latent).last_hidden_state
        return self.output_adapter(output, **kwargs)


class PerceiverIO(nn.Sequential):
    def __init__(self, encoder: PerceiverEncoder, decoder: PerceiverDecoder):
        super().__init__(encoder, decoder)

    @property
    def encoder(self) -> PerceiverEncoder:
        return self[0]

    @property
    def decoder(self) -> PerceiverDecoder:
        return self[1]


class PerceiverAR(nn.Module):
    """Causal cross-attention autoregressive model (https://arxiv.org/abs/2202.07765).

    The input splits into a gradient-free K/V prefix and ``n - prefix_len`` latents;
    latents attend causally to prefix + themselves (x_kv_prefix mode, right-aligned
    rotary) followed by a causal self-attention block over the latents.
    Three-state kv_cache protocol: None (no caching) / [] (initialize) /
    populated [ca_cache, *sa_caches]. Parity: reference modules.py:691-871.
    """

    def __init__(
        self,
        input_adapter: RotarySupport,
        num_heads: int = 8,
        max_heads_parallel: Optional[int] = None,
        num_self_attention_layers: int = 6,
        num_self_attention_rotary_layers: int = 1,
        self_attention_widening_factor: int = 4,
        cross_attention_widening_factor: int = 4,
        cross_attention_dropout: float = 0.5,
        post_attention_dropout: float = 0.0,
        residual_dropout: float = 0.0,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
    ):
        super().__init__()

        cross_attn = CrossAttentionLayer(
            num_heads=num_heads,
            num_q_input_channels=input_adapter.num_input_channels,
            num_kv_input_channels=input_adapter.num_input_channels,
            max_heads_parallel=max_heads_parallel,
            causal_attention=True,
            widening_factor=cross_attention_widening_factor,
            dropout=post_attention_dropout,
            residual_dropout=residual_dropout,
            qkv_bias=False,
            out_bias=True,
            mlp_bias=False,
        )
        if activation_checkpointing:
            cross_attn = activation_checkpoint_wrapper(cross_attn, offload_to_cpu=activation_offloading)

        self.input_adapter = input_adapter
        self.cross_attention_dropout = cross_attention_dropout
        self.cross_attention = cross_attn
        self.self_attention = SelfAttentionBlock(
            num_layers=num_self_attention_layers,
            num_heads=num_heads,
            num_channels=input_adapter.num_input_channels,
            causal_attention=True,
            widening_factor=self_attention_widening_factor,
            dropout=post_attention_dropout,
            residual_dropout=residual_dropout,
            num_rotary_layers=num_self_attention_rotary_layers,
            activation_checkpointing=activation_checkpointing,
            activation_offloading=activation_offloading,
            qkv_bias=False,
            out_bias=False,
            mlp_bias=False,
        )

    def forward(
        self,
        x: torch.Tensor,
        prefix_len: int,
        pad_mask: Optional[torch.Tensor] = None,
        kv_cache: Optional[List[KVCache]] = None,
    ) -> ModuleOutput:
        # left-pad correction for absolute positions (caller guarantees left padding)
        shift = None if pad_mask is None else pad_mask.sum(dim=1, keepdim=True)

        cache_fresh = kv_cache is not None and (
            len(kv_cache) == 0 or cache_len(kv_cache[0]) == 0
        )
        if kv_cache is None or cache_fresh:
            b, n = x.shape
        else:
            b = x.shape[0]
            n = cache_len(kv_cache[0]) + x.shape[1]

        if not 0 <= prefix_len < n:
            raise ValueError(f"prefix_len ({prefix_len}) out of valid range [0..{n})")

        x, frq_pos_enc = self.input_adapter(x, abs_pos=positions(b, n, shift=shift, device=x.device))

        if kv_cache is None or cache_fresh:
            x_latent, x_prefix = x[:, prefix_len:], x[:, :prefix_len]
        else:
            # cached decode: every new token is a latent
            x_latent, x_prefix = x, x[:, :0]

        frq_pos_enc_latent = frq_pos_enc[:, prefix_len:]
        frq_pos_enc_prefix = frq_pos_enc[:, :prefix_len]

        if pad_mask is not None:
            pad_mask_latent = pad_mask[:, prefix_len:]
            pad_mask_prefix = pad_mask[:, :prefix_len]

        if self.training and prefix_len > 0 and self.cross_attention_dropout > 0.0:
            if kv_cache is not None:
                raise ValueError("cross-attention dropout not supported with caching")
            # drop a random fraction of *prefix positions* (training-time regularizer;
            # host-side index select — the kernels see the thinned prefix)
            rand = torch.rand(b, prefix_len, device=x.device)
            keep = prefix_len - int(prefix_len * self.cross_attention_dropout)
            keep_indices = rand.topk(keep, dim=-1).indices
            keep_mask = torch.zeros_like(rand, dtype=torch.bool).scatter_(1, keep_indices, True)

            x_prefix = x_prefix[keep_mask].view(b, keep, -1)
            frq_pos_enc_prefix = frq_pos_enc_prefix[keep_mask].view(b, keep, -1)
            if pad_mask is not None:
                pad_mask_prefix = pad_mask_prefix[keep_mask].view(b, keep)

        frq_pos_enc_q = frq_pos_enc_latent
        frq_pos_enc_k = torch.cat([frq_pos_enc_prefix, frq_pos_enc_latent], dim=1)

        if pad_mask is not None:
            pad_mask = torch.cat([pad_mask_prefix, pad_mask_latent], dim=1)

        if kv_cache is None:
            ca_kv_cache, sa_kv_cache, kv_cache_updated = None, None, None
        elif len(kv_cache) == 0:
            ca_kv_cache, sa_kv_cache, kv_cache_updated = self.cross_attention.empty_kv_cache(x_latent), [], []
        else:
            # populated tuples, or preallocated StaticKVCache objects (possibly fresh)
            ca_kv_cache, *sa_kv_cache = kv_cache
            kv_cache_updated = []

        ca_output = self.cross_attention(
            x_latent,
            x_kv_prefix=x_prefix,
            pad_mask=pad_mask,
            rot_pos_emb_q=RotaryPositionEmbedding(frq_pos_enc_q, right_align=True),
            rot_pos_emb_k=RotaryPositionEmbedding(frq_pos_enc_k, right_align=True),
            kv_cache=ca_kv_cache,
        )
        if kv_cache_updated is not None:
            kv_cache_updated.append(ca_output.kv_cache)

        sa_output = self.self_attention(
            ca_output.last_hidden_state,
            rot_pos_emb=RotaryPositionEmbedding(frq_pos_enc_latent, right_align=True),
            kv_cache=sa_kv_cache,
        )
        if kv_cache_updated is not None:
            kv_cache_updated.extend(sa_output.kv_cache)

        return ModuleOutput(last_hidden_state=sa_output.last_hidden_state, kv_cache=kv_cache_updated)


class CausalSequenceModel(PerceiverAR):
    """PerceiverAR + token input adapter with rotary support + optional output
    LayerNorm + tied-embedding logits. Parity: reference modules.py:874-930."""

    def __init__(self, config: CausalSequenceModelConfig):
        num_rotated_channels = config.num_channels // config.num_heads
        if config.abs_pos_emb:
            # rotary on the first 50% of per-head channels only
            num_rotated_channels //= 2

        input_adapter = TokenInputAdapterWithRotarySupport(
            rotated_channels_per_head=num_rotated_channels,
            vocab_size=config.vocab_size,
            max_seq_len=config.max_seq_len,
            num_input_channels=config.num_channels,
            abs_pos_emb=config.abs_pos_emb,
        )
        super().__init__(input_adapter=input_adapter, **config.base_kwargs())
        self.config = config

        if config.output_norm:
            self.out_norm = LayerNorm(config.num_channels)

        self.output_adapter = TiedTokenOutputAdapter(vocab_size=config.vocab_size, emb_bias=config.output_bias)

        with torch.no_grad():
            init_parameters(self, config.init_scale)

    @property
    def max_seq_len(self) -> int:
        return self.input_adapter.max_seq_len

    @property
    def max_latents(self) -> int:
        return self.config.max_latents

    @property
    def max_prefix_len(self) -> int:
        return self.max_seq_len - self.max_latents

    def forward(
        self,
        x: torch.Tensor,
        prefix_len: int,
        pad_mask: Optional[torch.Tensor] = None,
        kv_cache: Optional[List[KVCache]] = None,
    ) -> ModuleOutput:
        if prefix_len > self.max_prefix_len:
            raise ValueError(f"prefix_len ({prefix_len}) exceeds max_prefix_len ({self.max_prefix_len})")

        output = super().forward(x, prefix_len=prefix_len, pad_mask=pad_mask, kv_cache=kv_cache)
        if self.config.output_norm:
            output.last_hidden_state = self.out_norm(output.last_hidden_state)
        output.logits = self.output_adapter(output.last_hidden_state, txt_embedding=self.input_adapter.txt_embedding)
        return output
