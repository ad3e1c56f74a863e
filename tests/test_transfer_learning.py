"""Transfer-learning checkpoint chains (reference text/classifier/lightning.py):
a text classifier initialized from an MLM encoder checkpoint must carry the MLM
encoder weights; `freeze` must stop encoder gradients."""
import torch

from perceiver_amd.core import ClassificationDecoderConfig
from perceiver_amd.models.text.common import TextEncoderConfig
from perceiver_amd.models.text.mlm import TextDecoderConfig
from perceiver_amd.train.lit import LitMaskedLanguageModel, LitTextClassifier


def _enc_cfg(**kw):
    return TextEncoderConfig(vocab_size=50, max_seq_len=32, num_input_channels=24,
                             num_cross_attention_heads=2, num_self_attention_heads=2,
                             num_self_attention_layers_per_block=2, **kw)


def test_classifier_initializes_encoder_from_mlm_checkpoint(tmp_path):
    torch.manual_seed(0)
    mlm = LitMaskedLanguageModel(
        _enc_cfg(), TextDecoderConfig(vocab_size=50, max_seq_len=32, num_cross_attention_heads=2),
        num_latents=8, num_latent_channels=24,
    )
    ckpt_path = tmp_path / "mlm.ckpt"
    torch.save({"state_dict": mlm.state_dict(), "hyper_parameters": dict(mlm.hparams)}, ckpt_path)

    clf = LitTextClassifier(
        _enc_cfg(params=str(ckpt_path)),
        ClassificationDecoderConfig(num_classes=2, num_output_query_channels=16,
                                    num_cross_attention_heads=2),
        num_latents=8, num_latent_channels=24,
    )
    for (n1, p1), (n2, p2) in zip(mlm.model.encoder.named_parameters(),
                                  clf.model.encoder.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2), f"encoder weight not transferred: {n1}"


def test_frozen_encoder_has_no_grads(tmp_path):
    clf = LitTextClassifier(
        _enc_cfg(freeze=True),
        ClassificationDecoderConfig(num_classes=2, num_output_query_channels=16,
                                    num_cross_attention_heads=2),
        num_latents=8, num_latent_channels=24,
    )
    x = torch.randint(0, 50, (2, 32))
    y = torch.randint(0, 2, (2,))
    pad = torch.zeros(2, 32, dtype=torch.bool)
    loss, _ = clf.step((y, x, pad))
    loss.backward()
    assert all(not p.requires_grad for p in clf.model.encoder.parameters())
    assert any(p.grad is not None for p in clf.model.decoder.parameters())


def test_full_model_params_chain(tmp_path):
    torch.manual_seed(1)
    a = LitTextClassifier(
        _enc_cfg(), ClassificationDecoderConfig(num_classes=2, num_output_query_channels=16,
                                                num_cross_attention_heads=2),
        num_latents=8, num_latent_channels=24,
    )
    path = tmp_path / "clf.ckpt"
    torch.save({"state_dict": a.state_dict(), "hyper_parameters": dict(a.hparams)}, path)

    b = LitTextClassifier(
        _enc_cfg(), ClassificationDecoderConfig(num_classes=2, num_output_query_channels=16,
                                                num_cross_attention_heads=2),
        num_latents=8, num_latent_channels=24, params=str(path),
    )
    for p1, p2 in zip(a.model.parameters(), b.model.parameters()):
        assert torch.equal(p1, p2)
