"""Construction/forward smoke tests for every task backend model (SURVEY.md §4
category 5), on tiny configs."""
import torch

from perceiver_amd.core import ClassificationDecoderConfig
from perceiver_amd.models.audio.symbolic import SymbolicAudioModel, SymbolicAudioModelConfig
from perceiver_amd.models.text.classifier import TextClassifier, TextClassifierConfig
from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig
from perceiver_amd.models.text.common import TextEncoderConfig
from perceiver_amd.models.text.mlm import MaskedLanguageModel, MaskedLanguageModelConfig, TextDecoderConfig
from perceiver_amd.models.vision.image_classifier import ImageClassifier, ImageClassifierConfig, ImageEncoderConfig
from perceiver_amd.models.vision.optical_flow import (
    OpticalFlow,
    OpticalFlowConfig,
    OpticalFlowDecoderConfig,
    OpticalFlowEncoderConfig,
)

VOCAB, SEQ = 50, 16


def _text_encoder_cfg(**kw):
    return TextEncoderConfig(
        vocab_size=VOCAB, max_seq_len=SEQ, num_input_channels=24,
        num_cross_attention_heads=2, num_self_attention_heads=2,
        num_self_attention_layers_per_block=2, **kw,
    )


def test_masked_language_model_tied_and_untied():
    for out_q in (None, 20):
        cfg = MaskedLanguageModelConfig(
            encoder=_text_encoder_cfg(),
            decoder=TextDecoderConfig(vocab_size=VOCAB, max_seq_len=SEQ, num_output_query_channels=out_q,
                                      num_cross_attention_heads=2),
            num_latents=8, num_latent_channels=24,
        )
        model = MaskedLanguageModel(cfg).eval()
        x = torch.randint(0, VOCAB, (2, 12))
        pad = torch.zeros(2, 12, dtype=torch.bool)
        logits = model(x, pad)
        assert logits.shape == (2, 12, VOCAB)


def test_text_classifier():
    cfg = TextClassifierConfig(
        encoder=_text_encoder_cfg(),
        decoder=ClassificationDecoderConfig(num_classes=3, num_output_query_channels=16,
                                            num_cross_attention_heads=2),
        num_latents=8, num_latent_channels=24,
    )
    model = TextClassifier(cfg).eval()
    logits = model(torch.randint(0, VOCAB, (2, SEQ)))
    assert logits.shape == (2, 3)


def test_causal_language_model_train_loss_decreases_smoke():
    cfg = CausalLanguageModelConfig(
        vocab_size=VOCAB, max_seq_len=SEQ, max_latents=8, num_channels=24, num_heads=4,
        num_self_attention_layers=2, cross_attention_dropout=0.5,
    )
    model = CausalLanguageModel(cfg)
    x = torch.randint(0, VOCAB, (2, SEQ))
    out = model(x, prefix_len=SEQ - 8)
    assert out.logits.shape == (2, 8, VOCAB)
    # training-mode cross-attention dropout path
    model.train()
    out = model(x, prefix_len=SEQ - 8)
    assert out.logits.shape == (2, 8, VOCAB)


def test_symbolic_audio_model():
    cfg = SymbolicAudioModelConfig(
        vocab_size=VOCAB, max_seq_len=SEQ, max_latents=8, num_channels=24, num_heads=4,
        num_self_attention_layers=2, output_norm=True, output_bias=False, abs_pos_emb=False,
    )
    model = SymbolicAudioModel(cfg).eval()
    out = model(torch.randint(0, VOCAB, (2, SEQ)), prefix_len=SEQ - 8)
    assert out.logits.shape == (2, 8, VOCAB)


def test_image_classifier():
    cfg = ImageClassifierConfig(
        encoder=ImageEncoderConfig(
            image_shape=(8, 8, 1), num_frequency_bands=4,
            num_cross_attention_heads=1, num_self_attention_heads=2,
            num_self_attention_layers_per_block=2, num_self_attention_blocks=2,
        ),
        decoder=ClassificationDecoderConfig(num_classes=10, num_output_query_channels=16,
                                            num_cross_attention_heads=1),
        num_latents=8, num_latent_channels=16,
    )
    model = ImageClassifier(cfg).eval()
    logits = model(torch.randn(2, 8, 8, 1))
    assert logits.shape == (2, 10)
    # qk channels default to adapter channels
    assert model.encoder.cross_attn_1.num_qk_channels == model.encoder.input_adapter.num_input_channels


def test_optical_flow():
    cfg = OpticalFlowConfig(
        encoder=OpticalFlowEncoderConfig(
            image_shape=(8, 12), num_patch_input_channels=5, num_patch_hidden_channels=8,
            num_frequency_bands=2, num_cross_attention_heads=1, num_self_attention_heads=2,
            num_self_attention_layers_per_block=2,
        ),
        decoder=OpticalFlowDecoderConfig(image_shape=(8, 12), num_cross_attention_heads=1),
        num_latents=8, num_latent_channels=16,
    )
    model = OpticalFlow(cfg).eval()
    flow = model(torch.randn(2, 2, 5, 8, 12))
    assert flow.shape == (2, 8, 12, 2)
