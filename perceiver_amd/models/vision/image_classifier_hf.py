"""🤗 wrapper for the image classifier: AutoModelForImageClassification +
AutoImageProcessor registration, MNIST/deepmind conversion.
Parity: reference vision/image_classifier/huggingface.py. The input processor is
self-contained (PIL + torch; this environment's transformers PerceiverImageProcessor
needs torchvision): center-crop/resize/normalize with channels-last and grayscale
options.
"""
from __future__ import annotations

from typing import Any, List, Optional, Union

import numpy as np
import torch
from transformers import (
    AutoConfig,
    AutoModelForImageClassification,
    PretrainedConfig,
    PreTrainedModel,
)
from transformers.image_processing_utils import BaseImageProcessor, BatchFeature
from transformers.modeling_outputs import ImageClassifierOutput

from perceiver_amd.core import ClassificationDecoderConfig
from perceiver_amd.data.vision.common import Normalize, center_crop_resize, to_tensor
from perceiver_amd.models.hf_registry import BackendConfigMixin, wrap_lit_checkpoint
from perceiver_amd.models.hf_base import (
    copy_classification_decoder_params,
    copy_cross_attention_layer_params,
    copy_latent_provider_params,
    copy_self_attention_block_params,
)
from perceiver_amd.models.vision.image_classifier import (
    ImageClassifier,
    ImageClassifierConfig,
    ImageEncoderConfig,
    PerceiverEncoder,
)


class PerceiverImageClassifierConfig(BackendConfigMixin, PretrainedConfig):
    model_type = "perceiver-io-image-classifier"
    backend_config_class = ImageClassifierConfig

    def __init__(self, backend_config=None, **kwargs):
        # explicit __init__: transformers 5.x wraps configs without one
        # in a kwargs-only guard that would swallow backend_config
        super().__init__(backend_config, **kwargs)

    @classmethod
    def default_backend_config(cls):
        return ImageClassifierConfig(
            ImageEncoderConfig(), ClassificationDecoderConfig(),
            num_latents=512, num_latent_channels=512,
        )

    @classmethod
    def decode_backend_config(cls, model_config):
        flat = dict(model_config)
        config = ImageClassifierConfig(
            encoder=ImageEncoderConfig(**flat.pop("encoder")),
            decoder=ClassificationDecoderConfig(**flat.pop("decoder")),
            **flat,
        )
        config.encoder.image_shape = tuple(config.encoder.image_shape)
        return config


class PerceiverImageClassifierInputProcessor(BaseImageProcessor):
    """Center-crop -> resize -> normalize -> channels-last, with optional grayscale
    single-channel mode (MNIST)."""

    model_input_names = ["pixel_values"]

    def __init__(self, channels_last: bool = True, single_channel: bool = False,
                 do_center_crop: bool = True, crop_size: int = 256,
                 do_resize: bool = True, size: int = 224,
                 do_normalize: bool = True, image_mean=None, image_std=None, **kwargs):
        super().__init__(**kwargs)
        self.channels_last = channels_last
        self.single_channel = single_channel
        self.do_center_crop = do_center_crop
        self.crop_size = crop_size
        self.do_resize = do_resize
        self.size = size
        self.do_normalize = do_normalize
        self.image_mean = image_mean if image_mean is not None else [0.485, 0.456, 0.406]
        self.image_std = image_std if image_std is not None else [0.229, 0.224, 0.225]

    def grayscale(self, images):
        from PIL import Image

        if isinstance(images, Image.Image):
            return np.array(images.convert("L"))
        if isinstance(images, List):
            return [self.grayscale(im) for im in images]
        return images

    def _process_one(self, img) -> torch.Tensor:
        from PIL import Image

        if isinstance(img, np.ndarray):
            img_t = to_tensor(img)
        elif isinstance(img, Image.Image):
            if self.do_center_crop or self.do_resize:
                size = self.size if self.do_resize else self.crop_size
                crop = self.crop_size if self.do_center_crop else size
                img = center_crop_resize(img.convert("RGB"), crop, size)
            img_t = to_tensor(img)
        elif isinstance(img, torch.Tensor):
            img_t = img.float()
            if img_t.ndim == 2:
                img_t = img_t[None]
        else:
            raise ValueError(f"unsupported image type {type(img)}")

        if self.do_normalize:
            mean = self.image_mean if isinstance(self.image_mean, (list, tuple)) else [self.image_mean] * img_t.shape[0]
            std = self.image_std if isinstance(self.image_std, (list, tuple)) else [self.image_std] * img_t.shape[0]
            img_t = Normalize(mean[: img_t.shape[0]], std[: img_t.shape[0]])(img_t)
        return img_t

    def preprocess(self, images, return_tensors=None, **kwargs) -> BatchFeature:
        if self.single_channel:
            images = self.grayscale(images)
        if not isinstance(images, list):
            images = [images]
        pixel_values = torch.stack([self._process_one(im) for im in images])
        if self.channels_last:
            pixel_values = pixel_values.permute(0, 2, 3, 1).contiguous()
        return BatchFeature(data={"pixel_values": pixel_values}, tensor_type=return_tensors)


class PerceiverImageClassifier(PreTrainedModel):
    config_class = PerceiverImageClassifierConfig

    def __init__(self, config: PerceiverImageClassifierConfig):
        super().__init__(config)
        self.backend_model = ImageClassifier(config.backend_config)
        self.post_init()

    @staticmethod
    def from_checkpoint(ckpt_path):
        from perceiver_amd.train.lit import LitImageClassifier

        return wrap_lit_checkpoint(LitImageClassifier, PerceiverImageClassifier,
                                   ckpt_path, is_decoder=False)

    def forward(self, inputs: Optional[torch.Tensor] = None,
                labels: Optional[torch.LongTensor] = None,
                pixel_values: Optional[torch.Tensor] = None, **kwargs: Any):
        if labels is not None:
            raise ValueError("Loss computation from labels not supported yet")
        if inputs is None and pixel_values is None:
            raise ValueError("Either inputs or pixel_values must be defined")
        if inputs is None:
            inputs = pixel_values
        logits = self.backend_model(inputs)
        return ImageClassifierOutput(logits=logits)


AutoConfig.register(PerceiverImageClassifierConfig.model_type, PerceiverImageClassifierConfig)
AutoModelForImageClassification.register(PerceiverImageClassifierConfig, PerceiverImageClassifier)
try:  # AutoImageProcessor requires torchvision in transformers 5.x; registration is optional
    from transformers import AutoImageProcessor

    AutoImageProcessor.register(PerceiverImageClassifierConfig,
                                slow_image_processor_class=PerceiverImageClassifierInputProcessor)
except ImportError:
    pass


# ------------------------------------------------------------------ conversion
def convert_checkpoint(save_dir, ckpt_url, image_processor, id2label=None, label2id=None, **kwargs):
    image_processor.save_pretrained(save_dir, **kwargs)
    model = PerceiverImageClassifier.from_checkpoint(ckpt_url)
    if id2label is not None:
        model.config.id2label = id2label
    if label2id is not None:
        model.config.label2id = label2id
    model.save_pretrained(save_dir, **kwargs)


def convert_mnist_classifier_checkpoint(save_dir, ckpt_url, **kwargs):
    image_processor = PerceiverImageClassifierInputProcessor(
        single_channel=True, do_center_crop=False, do_resize=False,
        image_mean=0.5, image_std=0.5,
    )
    convert_checkpoint(
        save_dir=save_dir, ckpt_url=ckpt_url, image_processor=image_processor,
        id2label={i: i for i in range(10)}, label2id={i: i for i in range(10)}, **kwargs,
    )


def convert_config(config) -> ImageClassifierConfig:
    """transformers PerceiverConfig -> ImageClassifierConfig."""
    assert config.hidden_act == "gelu"
    encoder_config = ImageEncoderConfig(
        image_shape=(224, 224, 3),
        num_frequency_bands=64,
        num_cross_attention_heads=config.num_cross_attention_heads,
        num_self_attention_heads=config.num_self_attention_heads,
        num_self_attention_layers_per_block=config.num_self_attends_per_block,
        num_self_attention_blocks=config.num_blocks,
        dropout=config.attention_probs_dropout_prob,
        init_scale=config.initializer_range,
    )
    decoder_config = ClassificationDecoderConfig(
        num_classes=config.num_labels,
        num_output_query_channels=config.d_latents,
        num_cross_attention_heads=config.num_cross_attention_heads,
        cross_attention_residual=True,
        dropout=config.attention_probs_dropout_prob,
        init_scale=config.initializer_range,
    )
    return ImageClassifierConfig(
        encoder_config, decoder_config,
        num_latents=config.num_latents, num_latent_channels=config.d_latents,
    )


def copy_image_encoder_params(src, tgt: PerceiverEncoder):
    copy_cross_attention_layer_params(src.encoder.cross_attention, tgt.cross_attn_1, query_residual=True)
    copy_self_attention_block_params(src.encoder.self_attends, tgt.self_attn_1)
    copy_latent_provider_params(src, tgt)


def convert_model(save_dir, source_repo_id="deepmind/vision-perceiver-fourier", **kwargs):
    """transformers PerceiverForImageClassificationFourier -> persistent wrapper."""
    import transformers

    src_model = transformers.PerceiverForImageClassificationFourier.from_pretrained(source_repo_id)
    tgt_config = PerceiverImageClassifierConfig(
        convert_config(src_model.config),
        id2label=src_model.config.id2label, label2id=src_model.config.label2id,
    )
    tgt_model = PerceiverImageClassifier(tgt_config)
    copy_image_encoder_params(src_model.perceiver, tgt_model.backend_model.encoder)
    copy_classification_decoder_params(src_model.perceiver, tgt_model.backend_model.decoder)
    tgt_model.save_pretrained(save_dir, **kwargs)

    processor = PerceiverImageClassifierInputProcessor(channels_last=True)
    processor.save_pretrained(save_dir, **kwargs)
