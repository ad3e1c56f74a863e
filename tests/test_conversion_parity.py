"""Golden-model conversion parity (SURVEY.md §4 category 2, offline): build a
random-init ``transformers`` Perceiver locally, convert its weights with this
framework's copy utilities, and require logits allclose — the same contract the
reference pins against deepmind/language-perceiver, without network access."""
import pytest
import torch

transformers = pytest.importorskip("transformers")
from transformers import PerceiverConfig  # noqa: E402


def _tiny_perceiver_config(**kw):
    return PerceiverConfig(
        num_latents=16,
        d_latents=32,
        d_model=24,
        num_blocks=1,
        num_self_attends_per_block=2,
        num_self_attention_heads=2,
        num_cross_attention_heads=2,
        qk_channels=256,  # transformers hardcodes the MaskedLM decoder qk to 8*32
        v_channels=32,
        cross_attention_widening_factor=1,
        self_attention_widening_factor=1,
        attention_probs_dropout_prob=0.0,
        max_position_embeddings=48,
        vocab_size=60,
        **kw,
    )


def test_mlm_conversion_matches_transformers():
    from perceiver_amd.models.text.mlm import MaskedLanguageModel
    from perceiver_amd.models.text.mlm_hf import (
        PerceiverMaskedLanguageModelConfig,
        PerceiverMaskedLanguageModel,
        convert_config,
        copy_text_decoder_params,
        copy_text_encoder_params,
    )

    torch.manual_seed(0)
    src = transformers.PerceiverForMaskedLM(_tiny_perceiver_config()).eval()

    tgt_config = PerceiverMaskedLanguageModelConfig(convert_config(src.config))
    tgt = PerceiverMaskedLanguageModel(tgt_config).eval()
    copy_text_encoder_params(src.perceiver, tgt.backend_model.encoder)
    copy_text_decoder_params(src, tgt.backend_model.decoder)

    x = torch.randint(0, 60, (2, 48))
    mask = torch.ones(2, 48, dtype=torch.long)
    with torch.no_grad():
        ref = src(inputs=x, attention_mask=mask).logits
        got = tgt(input_ids=x, attention_mask=mask).logits
    assert got.shape == ref.shape
    assert torch.allclose(got, ref, atol=1e-4, rtol=1e-4), \
        f"max err {(got - ref).abs().max().item()}"


def test_mlm_conversion_param_count_matches():
    """The converted model must hold exactly the source's relevant parameters
    (the reference asserts 201,108,230 for the full-size model)."""
    from perceiver_amd.models.text.mlm_hf import (
        PerceiverMaskedLanguageModelConfig,
        PerceiverMaskedLanguageModel,
        convert_config,
    )

    src = transformers.PerceiverForMaskedLM(_tiny_perceiver_config())
    tgt = PerceiverMaskedLanguageModel(PerceiverMaskedLanguageModelConfig(convert_config(src.config)))
    n_src = sum(p.numel() for p in src.parameters())
    n_tgt = sum(p.numel() for p in tgt.parameters())
    assert n_src == n_tgt, (n_src, n_tgt)


def test_flagship_mlm_param_count_is_201_108_230():
    """Flagship-scale architecture parity: at the reference CLI defaults
    (256 latents x 1280ch, 26 SA layers, qk 256 / v 1280, vocab 262, seq 2048)
    the model must have exactly the deepmind/language-perceiver parameter count
    the reference pins (tests/masked_language_model_convert_test.py:12).
    Built on the meta device — no 800 MB allocation."""
    from perceiver_amd.models.flagship import mlm_flagship
    from perceiver_amd.models.text.mlm import MaskedLanguageModel

    with torch.device("meta"):
        model = MaskedLanguageModel(mlm_flagship(num_latents=256))
    assert sum(p.numel() for p in model.parameters()) == 201_108_230
