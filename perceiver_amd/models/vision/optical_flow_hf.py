"""🤗 wrapper + custom registered pipeline task "optical-flow": preprocess (patch
tiling) -> micro-batched forward -> distance-weighted recombination, optional HSV
render. Parity: reference vision/optical_flow/huggingface.py."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple, Union

import numpy as np
import torch
from transformers import AutoConfig, Pipeline, PretrainedConfig, PreTrainedModel
from transformers.modeling_outputs import ModelOutput
from transformers.pipelines import PIPELINE_REGISTRY

from perceiver_amd.data.vision.optical_flow import OpticalFlowProcessor, render_optical_flow
from perceiver_amd.models.hf_registry import BackendConfigMixin
from perceiver_amd.models.hf_base import (
    copy_cross_attention_layer_params,
    copy_latent_provider_params,
    copy_params,
    copy_self_attention_block_params,
)
from perceiver_amd.core import PerceiverDecoder, PerceiverEncoder
from perceiver_amd.models.vision.optical_flow import (
    OpticalFlow,
    OpticalFlowConfig,
    OpticalFlowDecoderConfig,
    OpticalFlowEncoderConfig,
)


class OpticalFlowPerceiverConfig(BackendConfigMixin, PretrainedConfig):
    model_type = "perceiver-io-optical-flow"
    backend_config_class = OpticalFlowConfig

    def __init__(self, backend_config=None, **kwargs):
        # explicit __init__: transformers 5.x wraps configs without one
        # in a kwargs-only guard that would swallow backend_config
        super().__init__(backend_config, **kwargs)

    @classmethod
    def default_backend_config(cls):
        return OpticalFlowConfig(
            OpticalFlowEncoderConfig(), OpticalFlowDecoderConfig(),
            num_latents=512, num_latent_channels=512,
        )

    @classmethod
    def decode_backend_config(cls, model_config):
        flat = dict(model_config)
        cfg = OpticalFlowConfig(
            encoder=OpticalFlowEncoderConfig(**flat.pop("encoder")),
            decoder=OpticalFlowDecoderConfig(**flat.pop("decoder")),
            **flat,
        )
        cfg.encoder.image_shape = tuple(cfg.encoder.image_shape)
        cfg.decoder.image_shape = tuple(cfg.decoder.image_shape)
        return cfg


@dataclass
class OpticalFlowPerceiverOutput(ModelOutput):
    logits: torch.FloatTensor = None


class OpticalFlowPerceiver(PreTrainedModel):
    config_class = OpticalFlowPerceiverConfig

    def __init__(self, config: OpticalFlowPerceiverConfig):
        super().__init__(config)
        self.backend_model = OpticalFlow(config.backend_config)
        self.post_init()

    def forward(self, inputs: torch.Tensor):
        return OpticalFlowPerceiverOutput(logits=self.backend_model(inputs))


ImagePair = Union[Tuple[np.ndarray, np.ndarray], Tuple[torch.Tensor, torch.Tensor]]


class OpticalFlowPipeline(Pipeline):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.patch_size = self.model.config.backend_config.encoder.image_shape
        self.processor = OpticalFlowProcessor(patch_size=self.patch_size)

    def _sanitize_parameters(self, **kwargs):
        forward_kwargs = {}
        postprocess_kwargs = {}
        if "micro_batch_size" in kwargs:
            forward_kwargs["micro_batch_size"] = kwargs["micro_batch_size"]
        if "render" in kwargs:
            postprocess_kwargs["render"] = kwargs["render"]
        return {}, forward_kwargs, postprocess_kwargs

    def preprocess(self, image_pair: ImagePair, **kwargs):
        try:
            from PIL import Image

            if isinstance(image_pair[0], Image.Image):
                image_pair = (np.array(image_pair[0]), np.array(image_pair[1]))
        except ImportError:
            pass
        return {
            "input_features": self.processor.preprocess(image_pair),
            "input_image_shape": image_pair[0].shape,
        }

    def _forward(self, inputs, micro_batch_size=1, **kwargs):
        input_features = inputs["input_features"]
        output_tensors = []
        for i in range(0, input_features.shape[0], micro_batch_size):
            micro_batch = input_features[i: i + micro_batch_size]
            output_tensors.append(self.model(micro_batch).logits)
        model_output = OpticalFlowPerceiverOutput(logits=torch.concat(output_tensors, dim=0))
        model_output["input_image_shape"] = inputs["input_image_shape"]
        return model_output

    def postprocess(self, model_output, render=False, **kwargs):
        optical_flow = self.processor.postprocess(model_output.logits.float(),
                                                  img_shape=model_output.input_image_shape)
        optical_flow = optical_flow[0].numpy()
        if render:
            return render_optical_flow(optical_flow)
        return optical_flow


AutoConfig.register(OpticalFlowPerceiverConfig.model_type, OpticalFlowPerceiverConfig)
PIPELINE_REGISTRY.register_pipeline(
    "optical-flow", pipeline_class=OpticalFlowPipeline, pt_model=OpticalFlowPerceiver,
)


# ------------------------------------------------------------------ conversion
def convert_config(config) -> OpticalFlowConfig:
    """transformers PerceiverConfig (optical-flow) -> the backend dataclass tree."""
    assert config.hidden_act == "gelu"
    shared = dict(
        num_cross_attention_heads=config.num_cross_attention_heads,
        dropout=config.attention_probs_dropout_prob,
        init_scale=config.initializer_range,
    )
    encoder = OpticalFlowEncoderConfig(
        num_self_attention_heads=config.num_self_attention_heads,
        num_self_attention_layers_per_block=config.num_self_attends_per_block,
        num_self_attention_blocks=config.num_blocks,
        **shared,
    )
    return OpticalFlowConfig(encoder, OpticalFlowDecoderConfig(**shared),
                             num_latents=config.num_latents, num_latent_channels=config.d_latents)


def copy_flow_encoder_params(src, tgt: PerceiverEncoder):
    copy_cross_attention_layer_params(src.encoder.cross_attention, tgt.cross_attn_1, query_residual=True)
    copy_self_attention_block_params(src.encoder.self_attends, tgt.self_attn_1)
    copy_latent_provider_params(src, tgt)
    copy_params(src.input_preprocessor.conv_after_patches, tgt.input_adapter.linear)


def copy_flow_decoder_params(src, tgt: PerceiverDecoder):
    copy_cross_attention_layer_params(src.decoder.decoder.decoding_cross_attention,
                                      tgt.cross_attn, query_residual=False)
    copy_params(src.decoder.decoder.final_layer, tgt.output_adapter.linear)


def convert_model(save_dir, source_repo_id="deepmind/optical-flow-perceiver", **kwargs):
    """transformers PerceiverForOpticalFlow -> persistent OpticalFlowPerceiver."""
    import transformers

    src_model = transformers.PerceiverForOpticalFlow.from_pretrained(source_repo_id)
    tgt_config = OpticalFlowPerceiverConfig(convert_config(src_model.config))
    tgt_model = OpticalFlowPerceiver(tgt_config)
    copy_flow_encoder_params(src_model.perceiver, tgt_model.backend_model.encoder)
    copy_flow_decoder_params(src_model.perceiver, tgt_model.backend_model.decoder)
    tgt_model.save_pretrained(save_dir, **kwargs)
