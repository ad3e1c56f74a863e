"""KV-cache equivalence tests: incremental cached decoding (one token at a time) must
match a single full forward pass. This pins the tricky causal-mask right-alignment,
left-pad shift and three-state cache protocol (contract of reference
tests/kv_cache_test.py, see SURVEY.md §4 category 1)."""
import pytest
import torch

from perceiver_amd.core import (
    CausalSequenceModelConfig,
    CausalSequenceModel,
    CrossAttentionLayer,
    RotaryPositionEmbedding,
    SelfAttentionBlock,
    FrequencyPositionEncoding,
    positions,
)

B, N, C, HEADS = 2, 10, 32, 4
ROT = C // HEADS


def make_rot(n, b=B, shift=None, right_align=True):
    frq = FrequencyPositionEncoding(dim=ROT)
    enc = frq(positions(b, n, shift=shift))
    return RotaryPositionEmbedding(enc, right_align=right_align)


@torch.no_grad()
def test_self_attention_block_cached_decode_matches_full():
    torch.manual_seed(0)
    block = SelfAttentionBlock(
        num_layers=3, num_heads=HEADS, num_channels=C, causal_attention=True,
        num_rotary_layers=-1,
    ).eval()

    x = torch.randn(B, N, C)
    rot = make_rot(N)
    full = block(x, rot_pos_emb=rot).last_hidden_state

    kv_cache = []
    outs = []
    for i in range(N):
        rot_i = RotaryPositionEmbedding(rot.frq_pos_enc[:, 0, : i + 1], right_align=True)
        out = block(x[:, i : i + 1], rot_pos_emb=rot_i, kv_cache=kv_cache)
        kv_cache = out.kv_cache
        outs.append(out.last_hidden_state)
    incremental = torch.cat(outs, dim=1)

    assert torch.allclose(full, incremental, atol=1e-5)


@torch.no_grad()
def test_causal_cross_attention_layer_cached_decode_matches_full():
    torch.manual_seed(1)
    prefix_len = 4
    layer = CrossAttentionLayer(
        num_heads=HEADS, num_q_input_channels=C, num_kv_input_channels=C,
        causal_attention=True,
    ).eval()

    # left-padded batch: first row has 2 pads
    pad_mask = torch.zeros(B, N, dtype=torch.bool)
    pad_mask[0, :2] = True
    shift = pad_mask.sum(dim=1, keepdim=True)

    x = torch.randn(B, N, C)
    rot_full = make_rot(N, shift=shift)
    x_prefix, x_latent = x[:, :prefix_len], x[:, prefix_len:]
    rot_q = RotaryPositionEmbedding(rot_full.frq_pos_enc[:, 0, prefix_len:], right_align=True)

    full = layer(
        x_latent, x_kv_prefix=x_prefix, pad_mask=pad_mask,
        rot_pos_emb_q=rot_q, rot_pos_emb_k=rot_full,
    ).last_hidden_state

    # incremental: feed latents one at a time with cache
    kv_cache = layer.empty_kv_cache(x_latent)
    outs = []
    first = True
    for i in range(prefix_len, N):
        x_in = x[:, i : i + 1]
        n_seen = i + 1
        rot_q_i = RotaryPositionEmbedding(rot_full.frq_pos_enc[:, 0, i : i + 1], right_align=True)
        rot_k_i = RotaryPositionEmbedding(rot_full.frq_pos_enc[:, 0, :n_seen], right_align=True)
        if first:
            # first step carries the prefix
            out = layer(
                x_in, x_kv_prefix=x_prefix, pad_mask=pad_mask[:, :n_seen],
                rot_pos_emb_q=rot_q_i, rot_pos_emb_k=rot_k_i, kv_cache=kv_cache,
            )
            first = False
        else:
            out = layer(
                x_in, x_kv_prefix=x_in[:, :0], pad_mask=pad_mask[:, :n_seen],
                rot_pos_emb_q=rot_q_i, rot_pos_emb_k=rot_k_i, kv_cache=kv_cache,
            )
        kv_cache = out.kv_cache
        outs.append(out.last_hidden_state)
    incremental = torch.cat(outs, dim=1)

    assert torch.allclose(full, incremental, atol=1e-5)


@torch.no_grad()
def test_causal_sequence_model_cached_decode_matches_full():
    torch.manual_seed(2)
    vocab, seq, latents = 110, 12, 6
    cfg = CausalSequenceModelConfig(
        vocab_size=vocab, max_seq_len=seq, max_latents=latents, num_channels=C,
        num_heads=HEADS, num_self_attention_layers=2, cross_attention_dropout=0.0,
    )
    model = CausalSequenceModel(cfg).eval()

    x = torch.randint(0, vocab, (B, seq))
    prefix_len = seq - latents
    full = model(x, prefix_len=prefix_len).logits

    kv_cache = []
    outs = []
    for i in range(latents):
        n_in = prefix_len + 1 if i == 0 else 1
        x_in = x[:, : prefix_len + 1] if i == 0 else x[:, prefix_len + i : prefix_len + i + 1]
        out = model(x_in, prefix_len=prefix_len if i == 0 else 0, kv_cache=kv_cache)
        kv_cache = out.kv_cache
        outs.append(out.logits)
    incremental = torch.cat(outs, dim=1)

    assert torch.allclose(full, incremental, atol=1e-4)


@torch.no_grad()
def test_causal_sequence_model_cached_decode_matches_full_with_left_padding():
    torch.manual_seed(3)
    vocab, seq, latents = 110, 12, 6
    cfg = CausalSequenceModelConfig(
        vocab_size=vocab, max_seq_len=seq, max_latents=latents, num_channels=C,
        num_heads=HEADS, num_self_attention_layers=2, cross_attention_dropout=0.0,
        abs_pos_emb=False,
    )
    model = CausalSequenceModel(cfg).eval()

    x = torch.randint(0, vocab, (B, seq))
    pad_mask = torch.zeros(B, seq, dtype=torch.bool)
    pad_mask[0, :3] = True

    prefix_len = seq - latents
    full = model(x, prefix_len=prefix_len, pad_mask=pad_mask).logits

    kv_cache = []
    outs = []
    for i in range(latents):
        x_in = x[:, : prefix_len + 1] if i == 0 else x[:, prefix_len + i : prefix_len + i + 1]
        n_seen = prefix_len + i + 1
        out = model(
            x_in, prefix_len=prefix_len if i == 0 else 0,
            pad_mask=pad_mask[:, :n_seen], kv_cache=kv_cache,
        )
        kv_cache = out.kv_cache
        outs.append(out.logits)
    incremental = torch.cat(outs, dim=1)

    assert torch.allclose(full, incremental, atol=1e-4)


def test_allocate_kv_cache_bucketed_capacity():
    """ca_capacity bounds the cross-attention cache (bucketed decode graphs);
    self-attention caches keep max_latents."""
    import torch

    from perceiver_amd.core.cache import allocate_kv_cache
    from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig

    model = CausalLanguageModel(CausalLanguageModelConfig(
        vocab_size=32, max_seq_len=8192, max_latents=64, num_channels=32,
        num_heads=4, num_self_attention_layers=2))
    caches = allocate_kv_cache(model, 2, ca_capacity=2048)
    assert caches[0].capacity == 2048
    assert all(c.capacity == 64 for c in caches[1:])
    # never above max_seq_len
    caches = allocate_kv_cache(model, 2, ca_capacity=1 << 20)
    assert caches[0].capacity == 8192
