"""Construction/forward smoke tests and attention-semantics unit tests for the core
modules (contract of SURVEY.md §4 categories 3/5)."""
import pytest
import torch

from perceiver_amd.core import (
    ClassificationOutputAdapter,
    CrossAttentionLayer,
    FourierPositionEncoding,
    FrequencyPositionEncoding,
    InputAdapter,
    MultiHeadAttention,
    PerceiverDecoder,
    PerceiverEncoder,
    PerceiverIO,
    RotaryPositionEmbedding,
    TrainableQueryProvider,
    positions,
)
from perceiver_amd.ops.attention import eager_attention


class _IdentityAdapter(InputAdapter):
    def forward(self, x):
        return x


def test_mha_shapes_and_asymmetric_channels():
    mha = MultiHeadAttention(
        num_heads=4, num_q_input_channels=64, num_kv_input_channels=48,
        num_qk_channels=32, num_v_channels=80, num_output_channels=64,
    )
    out = mha(torch.randn(2, 5, 64), torch.randn(2, 9, 48))
    assert out.last_hidden_state.shape == (2, 5, 64)
    assert out.kv_cache is None


def test_mha_pad_mask_ignores_padding_positions():
    torch.manual_seed(0)
    mha = MultiHeadAttention(num_heads=2, num_q_input_channels=16, num_kv_input_channels=16).eval()
    x_q, x_kv = torch.randn(1, 3, 16), torch.randn(1, 6, 16)
    pad = torch.zeros(1, 6, dtype=torch.bool)
    pad[0, 4:] = True
    with torch.no_grad():
        o1 = mha(x_q, x_kv, pad_mask=pad).last_hidden_state
        x_kv2 = x_kv.clone()
        x_kv2[0, 4:] = 123.0  # padded content must not matter
        o2 = mha(x_q, x_kv2, pad_mask=pad).last_hidden_state
    assert torch.allclose(o1, o2, atol=1e-6)


def test_causal_mask_right_alignment():
    """With Nq < Lk, query i attends to keys 0..(Lk-Nq+i)."""
    torch.manual_seed(0)
    q = torch.randn(1, 1, 3, 8)
    k = torch.randn(1, 1, 5, 8)
    v = torch.randn(1, 1, 5, 8)
    out = eager_attention(q, k, v, causal=True)
    # last query attends to everything; first query only to keys 0..2
    k2, v2 = k.clone(), v.clone()
    k2[0, 0, 3:], v2[0, 0, 3:] = 99.0, 99.0
    out2 = eager_attention(q, k2, v2, causal=True)
    assert torch.allclose(out[0, 0, 0], out2[0, 0, 0], atol=1e-6)
    assert not torch.allclose(out[0, 0, 2], out2[0, 0, 2], atol=1e-3)


def test_mha_head_chunking_equivalent():
    torch.manual_seed(1)
    q, k, v = torch.randn(2, 8, 4, 16), torch.randn(2, 8, 9, 16), torch.randn(2, 8, 9, 16)
    full = eager_attention(q, k, v)
    chunked = eager_attention(q, k, v, max_heads_parallel=3)
    assert torch.allclose(full, chunked, atol=1e-6)


def _encoder(num_cross_layers=1, num_blocks=1, shared_ca=False, shared_sa=True):
    return PerceiverEncoder(
        input_adapter=_IdentityAdapter(num_input_channels=32),
        num_latents=8,
        num_latent_channels=24,
        num_cross_attention_heads=2,
        num_cross_attention_layers=num_cross_layers,
        num_self_attention_heads=2,
        num_self_attention_layers_per_block=2,
        num_self_attention_blocks=num_blocks,
        first_cross_attention_layer_shared=shared_ca,
        first_self_attention_block_shared=shared_sa,
    )


def test_encoder_forward_and_weight_sharing():
    enc = _encoder(num_cross_layers=2, num_blocks=3, shared_ca=False, shared_sa=False)
    assert enc.extra_cross_attention_layer and enc.extra_self_attention_block
    out = enc(torch.randn(2, 11, 32))
    assert out.shape == (2, 8, 24)

    enc_shared = _encoder(num_cross_layers=2, num_blocks=3, shared_ca=True, shared_sa=True)
    assert not enc_shared.extra_cross_attention_layer and not enc_shared.extra_self_attention_block
    assert enc_shared(torch.randn(2, 11, 32)).shape == (2, 8, 24)


def test_encoder_validation_errors():
    with pytest.raises(ValueError):
        _encoder(num_cross_layers=2, num_blocks=1)


def test_perceiver_io_classifier_shape():
    enc = _encoder()
    dec = PerceiverDecoder(
        output_adapter=ClassificationOutputAdapter(num_classes=5, num_output_query_channels=16),
        output_query_provider=TrainableQueryProvider(num_queries=1, num_query_channels=16),
        num_latent_channels=24,
        num_cross_attention_heads=2,
    )
    model = PerceiverIO(enc, dec)
    logits = model.decoder(model.encoder(torch.randn(2, 11, 32)))
    assert logits.shape == (2, 5)


def test_decoder_without_attention_residual():
    dec = PerceiverDecoder(
        output_adapter=ClassificationOutputAdapter(num_classes=5, num_output_query_channels=16),
        output_query_provider=TrainableQueryProvider(num_queries=3, num_query_channels=16),
        num_latent_channels=24,
        cross_attention_residual=False,
    )
    out = dec.cross_attn(torch.randn(2, 3, 16), torch.randn(2, 8, 24))
    assert out.last_hidden_state.shape == (2, 3, 16)


def test_positions_shift_and_clamp():
    shift = torch.tensor([[0], [3]])
    pos = positions(2, 5, shift=shift)
    assert pos[0].tolist() == [0, 1, 2, 3, 4]
    assert pos[1].tolist() == [0, 0, 0, 0, 1]


def test_fourier_position_encoding_channels():
    enc = FourierPositionEncoding(input_shape=(7, 9), num_frequency_bands=4)
    assert enc.num_position_encoding_channels() == 2 * (2 * 4 + 1)
    out = enc(b=3)
    assert out.shape == (3, 63, 18)


def test_frequency_position_encoding_interleave():
    frq = FrequencyPositionEncoding(dim=8)
    enc = frq(positions(1, 4))
    assert enc.shape == (1, 4, 8)
    # pairs are repeated
    assert torch.equal(enc[..., 0::2], enc[..., 1::2])


def test_rotary_right_align_slices_from_right():
    frq = FrequencyPositionEncoding(dim=8)
    enc = frq(positions(1, 10))
    rot = RotaryPositionEmbedding(enc, right_align=True)
    t = torch.randn(1, 2, 4, 8)  # seq 4 < 10
    out_r = rot.rotate(t)
    rot_l = RotaryPositionEmbedding(enc[:, -4:], right_align=False)
    assert torch.allclose(out_r, rot_l.rotate(t), atol=1e-6)


def test_rotary_partial_rotation_passthrough():
    frq = FrequencyPositionEncoding(dim=4)
    enc = frq(positions(1, 4))
    rot = RotaryPositionEmbedding(enc)
    t = torch.randn(1, 1, 4, 12)
    out = rot.rotate(t)
    assert torch.equal(out[..., 4:], t[..., 4:])
    assert not torch.allclose(out[..., :4], t[..., :4])


def test_activation_checkpoint_offload_cpu():
    """offload_to_cpu path: forward+backward through a checkpointed+offloaded
    layer produces the same gradients as the plain layer."""
    import copy

    from perceiver_amd.core.modules import SelfAttentionLayer, activation_checkpoint_wrapper

    torch.manual_seed(0)
    layer = SelfAttentionLayer(num_heads=2, num_channels=16, widening_factor=1)
    wrapped = activation_checkpoint_wrapper(copy.deepcopy(layer), offload_to_cpu=True)

    x1 = torch.randn(2, 6, 16, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    layer(x1).last_hidden_state.sum().backward()
    wrapped(x2).last_hidden_state.sum().backward()
    torch.testing.assert_close(x1.grad, x2.grad)
    for p1, p2 in zip(layer.parameters(), wrapped.parameters()):
        torch.testing.assert_close(p1.grad, p2.grad)
