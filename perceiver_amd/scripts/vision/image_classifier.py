"""Image classifier training CLI (parity: reference scripts/vision/image_classifier.py;
defaults = the reference MNIST example architecture)."""
from __future__ import annotations

from perceiver_amd.core import ClassificationDecoderConfig
from perceiver_amd.data.vision import MNISTDataModule
from perceiver_amd.models.vision.image_classifier import ImageEncoderConfig
from perceiver_amd.scripts.cli import CLI, build_dataclass
from perceiver_amd.train.lit import LitImageClassifier

DEFAULTS = {
    "model": {
        "num_latents": 512,
        "num_latent_channels": 1024,
        "encoder": {
            "num_frequency_bands": 64,
            "num_cross_attention_layers": 1,
            "num_cross_attention_heads": 1,
            "num_self_attention_heads": 8,
            "num_self_attention_layers_per_block": 6,
            "num_self_attention_blocks": 8,
            "dropout": 0.1,
        },
        "decoder": {"num_output_query_channels": 1024, "num_cross_attention_heads": 1,
                    "dropout": 0.1},
    },
    "data": {},
    "trainer": {"out_dir": "logs/img_clf"},
    "optimizer": {"lr": 1e-3, "lr_schedule": "constant", "warmup_steps": 500},
}


def link(cfg, dm):
    cfg["model"]["encoder"]["image_shape"] = tuple(dm.image_shape)
    cfg["model"]["decoder"]["num_classes"] = dm.num_classes


def build_model(model_cfg, dm):
    encoder = build_dataclass(ImageEncoderConfig, model_cfg.get("encoder", {}))
    decoder = build_dataclass(ClassificationDecoderConfig, model_cfg.get("decoder", {}))
    extra = {k: v for k, v in model_cfg.items() if k not in ("encoder", "decoder")}
    return LitImageClassifier(encoder, decoder, **extra)


if __name__ == "__main__":
    CLI(LitImageClassifier, MNISTDataModule, DEFAULTS, build_model, link)
