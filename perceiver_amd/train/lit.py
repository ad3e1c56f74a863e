"""Task training wrappers — the Lightning-module layer of the reference
(perceiver/model/*/lightning.py), rebuilt on the native Trainer.

Each wrapper holds the backend model as ``self.model`` (state-dict keys therefore
match the reference's ``model.*`` checkpoint layout), stores its constructor
arguments as ``hparams`` (embedded in checkpoints so ``load_from_checkpoint``
reconstructs the model without external config), and implements
``training_step(batch, step) -> loss`` / ``validation_step(batch) -> metrics``.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from perceiver_amd.core import DecoderConfig, EncoderConfig, PerceiverIOConfig
from perceiver_amd.data.text.common import Hparams


def is_checkpoint(path: str) -> bool:
    return os.path.splitext(path)[1] == ".ckpt"


class TrainableModule(nn.Module):
    """Minimal Lightning-module stand-in: hparams capture + checkpoint IO."""

    def save_hyperparameters(self, values: dict, ignore=("self", "__class__", "kwargs")):
        self.hparams = Hparams({k: v for k, v in values.items() if k not in ignore})

    @property
    def backend_model(self):
        return self.model

    @classmethod
    def load_from_checkpoint(cls, ckpt_path: str, map_location="cpu", **override):
        ckpt = torch.load(ckpt_path, map_location=map_location, weights_only=False)
        hparams = dict(ckpt.get("hyper_parameters", {}))
        hparams.update(override)
        obj = cls(**hparams)
        obj.load_state_dict(ckpt["state_dict"])
        return obj

    def training_step(self, batch, step: int):
        raise NotImplementedError

    @torch.no_grad()
    def validation_step(self, batch) -> dict:
        raise NotImplementedError


# ---------------------------------------------------------------------- bases
class LitPerceiverIO(TrainableModule):
    def __init__(
        self,
        encoder: EncoderConfig,
        decoder: DecoderConfig,
        num_latents: int,
        num_latent_channels: int,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
        params: Optional[str] = None,
        **extra,
    ):
        super().__init__()
        self.save_hyperparameters(dict(
            encoder=encoder, decoder=decoder, num_latents=num_latents,
            num_latent_channels=num_latent_channels,
            activation_checkpointing=activation_checkpointing,
            activation_offloading=activation_offloading, params=params, **extra,
        ))

    @classmethod
    def create(cls, config: PerceiverIOConfig, *args: Any, **kwargs: Any):
        return cls(
            config.encoder,
            config.decoder,
            *args,
            num_latents=config.num_latents,
            num_latent_channels=config.num_latent_channels,
            activation_checkpointing=config.activation_checkpointing,
            activation_offloading=config.activation_offloading,
            **kwargs,
        )

    def _maybe_load_params(self, hf_cls=None):
        params = self.hparams.get("params")
        if params is None:
            return
        if is_checkpoint(params):
            wrapper = type(self).load_from_checkpoint(params, params=None)
            self.model.load_state_dict(wrapper.model.state_dict())
        elif hf_cls is not None:
            wrapper = hf_cls.from_pretrained(params)
            self.model.load_state_dict(wrapper.backend_model.state_dict())


class LitClassifier(LitPerceiverIO):
    def __init__(self, *args: Any, **kwargs: Any):
        super().__init__(*args, **kwargs)
        self.loss = nn.CrossEntropyLoss()

    def step(self, batch):
        raise NotImplementedError

    def loss_acc(self, logits, y):
        loss = self.loss(logits, y)
        acc = (logits.argmax(dim=-1) == y).float().mean()
        return loss, acc

    def training_step(self, batch, step):
        loss, acc = self.step(batch)
        return loss

    @torch.no_grad()
    def validation_step(self, batch):
        loss, acc = self.step(batch)
        return {"val_loss": float(loss), "val_acc": float(acc)}


class LitCausalSequenceModel(TrainableModule):
    def __init__(
        self,
        vocab_size: int,
        max_seq_len: int,
        max_latents: int = 512,
        num_channels: int = 512,
        num_heads: int = 8,
        max_heads_parallel: Optional[int] = None,
        num_self_attention_layers: int = 6,
        num_self_attention_rotary_layers: int = 1,
        self_attention_widening_factor: int = 4,
        cross_attention_widening_factor: int = 4,
        cross_attention_dropout: float = 0.5,
        post_attention_dropout: float = 0.0,
        residual_dropout: float = 0.0,
        output_norm: bool = False,
        output_bias: bool = True,
        abs_pos_emb: bool = True,
        init_scale: float = 0.02,
        activation_checkpointing: bool = False,
        activation_offloading: bool = False,
        validation_sample_prompt: Optional[str] = None,
        validation_sample_record: Optional[int] = None,
        params: Optional[str] = None,
        **kwargs: Any,
    ):
        super().__init__()
        self.save_hyperparameters(dict(
            vocab_size=vocab_size, max_seq_len=max_seq_len, max_latents=max_latents,
            num_channels=num_channels, num_heads=num_heads, max_heads_parallel=max_heads_parallel,
            num_self_attention_layers=num_self_attention_layers,
            num_self_attention_rotary_layers=num_self_attention_rotary_layers,
            self_attention_widening_factor=self_attention_widening_factor,
            cross_attention_widening_factor=cross_attention_widening_factor,
            cross_attention_dropout=cross_attention_dropout,
            post_attention_dropout=post_attention_dropout, residual_dropout=residual_dropout,
            output_norm=output_norm, output_bias=output_bias, abs_pos_emb=abs_pos_emb,
            init_scale=init_scale, activation_checkpointing=activation_checkpointing,
            activation_offloading=activation_offloading,
            validation_sample_prompt=validation_sample_prompt,
            validation_sample_record=validation_sample_record, params=params,
        ))
        self.loss = nn.CrossEntropyLoss()

    @property
    def backend_model(self):
        return self.model

    def forward(self, x, prefix_len, pad_mask=None):
        return self.model(x, prefix_len=prefix_len, pad_mask=pad_mask)

    def step(self, batch):
        labels, x, pad_mask = batch
        labels = labels.clone()
        labels[pad_mask] = -100

        seq_len = x.shape[1]
        max_lat = self.hparams.max_latents
        if seq_len < max_lat:
            raise ValueError(f"Training sequence length must be at least {max_lat} (= max_latents)")

        logits = self(x, prefix_len=seq_len - max_lat, pad_mask=pad_mask).logits
        labels = labels[:, -logits.shape[1]:]
        return self.loss(logits.flatten(0, 1).float(), labels.flatten())

    def training_step(self, batch, step):
        return self.step(batch)

    @torch.no_grad()
    def validation_step(self, batch):
        return {"val_loss": float(self.step(batch))}

    @torch.no_grad()
    def on_validation_end(self, trainer, metrics):
        """Qualitative sampling at validation end (parity: reference
        text/clm/lightning.py:54-100): top-k continuation of a fixed prompt or a
        validation record, logged to the metrics stream (rank zero)."""
        from perceiver_amd.parallel import is_main_process

        prompt_text = self.hparams.get("validation_sample_prompt")
        if not is_main_process() or prompt_text is None:
            return
        dm = getattr(trainer, "datamodule", None)
        if dm is None or not hasattr(dm, "text_preprocessor"):
            return
        try:
            preproc = dm.text_preprocessor()
            ids, _ = preproc.preprocess(prompt_text)
            device = next(self.parameters()).device
            hgf = self.to_hgf_model()
            out = hgf.generate(input_ids=ids[None].to(device), num_latents=1,
                               max_new_tokens=64, do_sample=True, top_k=10)
            text = preproc.tokenizer.decode(out[0].tolist())
            trainer.log_metrics({"sample_generation": text})
        except Exception as e:  # sampling must never kill training
            trainer.log_metrics({"sample_generation_error": str(e)})


# ---------------------------------------------------------------------- text
class LitMaskedLanguageModel(LitPerceiverIO):
    def __init__(self, encoder, decoder, num_predictions: int = 3,
                 masked_samples: Optional[List[str]] = None, **kwargs: Any):
        super().__init__(encoder, decoder, num_predictions=num_predictions,
                         masked_samples=masked_samples, **kwargs)
        from perceiver_amd.models.text.mlm import MaskedLanguageModel, MaskedLanguageModelConfig

        self.loss = nn.CrossEntropyLoss()
        self.model = MaskedLanguageModel(MaskedLanguageModelConfig(
            encoder=encoder, decoder=decoder,
            num_latents=self.hparams.num_latents,
            num_latent_channels=self.hparams.num_latent_channels,
            activation_checkpointing=self.hparams.activation_checkpointing,
            activation_offloading=self.hparams.activation_offloading,
        ))
        self._maybe_load_params(_lazy_hf("mlm"))

    def forward(self, x, pad_mask=None):
        return self.model(x, pad_mask)

    def step(self, batch):
        labels, x, pad_mask = batch
        logits = self(x, pad_mask)
        return self.loss(logits.flatten(0, 1).float(), labels.flatten())

    def training_step(self, batch, step):
        return self.step(batch)

    @torch.no_grad()
    def validation_step(self, batch):
        return {"val_loss": float(self.step(batch))}

    @torch.no_grad()
    def on_validation_end(self, trainer, metrics):
        """Mask-filling table at validation end (parity: reference
        text/mlm/lightning.py:77-94), rank zero, logged to the metrics stream."""
        from perceiver_amd.parallel import is_main_process

        samples = self.hparams.get("masked_samples")
        if not is_main_process() or not samples:
            return
        dm = getattr(trainer, "datamodule", None)
        if dm is None or not hasattr(dm, "text_preprocessor"):
            return
        try:
            from perceiver_amd.models.text.mlm_utils import MaskFiller

            filler = MaskFiller(dm.text_preprocessor())
            device = next(self.parameters()).device
            masked, preds = filler.fill(self.model, list(samples),
                                        int(self.hparams.get("num_predictions", 3)),
                                        device=device)
            trainer.log_metrics({"masked_samples": masked, "predictions": preds})
        except Exception as e:
            trainer.log_metrics({"mask_filling_error": str(e)})


class LitTextClassifier(LitClassifier):
    def __init__(self, encoder, decoder, **kwargs: Any):
        super().__init__(encoder, decoder, **kwargs)
        from perceiver_amd.models.text.classifier import TextClassifier, TextClassifierConfig

        self.model = TextClassifier(TextClassifierConfig(
            encoder=encoder, decoder=decoder,
            num_latents=self.hparams.num_latents,
            num_latent_channels=self.hparams.num_latent_channels,
            activation_checkpointing=self.hparams.activation_checkpointing,
            activation_offloading=self.hparams.activation_offloading,
        ))
        # transfer learning: full-model ckpt via `params`, or an MLM-encoder
        # checkpoint via `encoder.params` (reference text/classifier/lightning.py)
        params = self.hparams.get("params")
        if params is not None and is_checkpoint(params):
            wrapper = LitTextClassifier.load_from_checkpoint(params, params=None)
            self.model.load_state_dict(wrapper.model.state_dict())
        elif encoder.params is not None and is_checkpoint(encoder.params):
            mlm = LitMaskedLanguageModel.load_from_checkpoint(encoder.params, params=None)
            self.model.encoder.load_state_dict(mlm.model.encoder.state_dict())

    def forward(self, x, pad_mask=None):
        return self.model(x, pad_mask)

    def step(self, batch):
        y, x, pad_mask = batch
        return self.loss_acc(self(x, pad_mask), y)


class LitCausalLanguageModel(LitCausalSequenceModel):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig

        self.model = CausalLanguageModel(CausalLanguageModelConfig.create(**self.hparams))
        params = self.hparams.get("params")
        if params is not None and is_checkpoint(params):
            wrapper = LitCausalLanguageModel.load_from_checkpoint(params, params=None)
            self.model.load_state_dict(wrapper.model.state_dict())

    @classmethod
    def create(cls, config, **kwargs):
        from dataclasses import asdict

        return cls(**asdict(config), **kwargs)

    def to_hgf_model(self):
        """Zero-copy wrap as the 🤗 causal-LM model for in-training generation."""
        from perceiver_amd.models.text.clm_hf import (
            PerceiverCausalLanguageModel,
            PerceiverCausalLanguageModelConfig,
        )

        config = PerceiverCausalLanguageModelConfig(self.model.config)
        return PerceiverCausalLanguageModel(config, backend_model=self.model)


# ---------------------------------------------------------------------- vision
class LitImageClassifier(LitClassifier):
    def __init__(self, encoder, decoder, **kwargs: Any):
        super().__init__(encoder, decoder, **kwargs)
        from perceiver_amd.models.vision.image_classifier import ImageClassifier, ImageClassifierConfig

        self.model = ImageClassifier(ImageClassifierConfig(
            encoder=encoder, decoder=decoder,
            num_latents=self.hparams.num_latents,
            num_latent_channels=self.hparams.num_latent_channels,
            activation_checkpointing=self.hparams.activation_checkpointing,
            activation_offloading=self.hparams.activation_offloading,
        ))
        self._maybe_load_params(_lazy_hf("img"))

    def forward(self, x):
        return self.model(x)

    def step(self, batch):
        return self.loss_acc(self(batch["image"]), batch["label"])


# ---------------------------------------------------------------------- audio
class LitSymbolicAudioModel(LitCausalSequenceModel):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        from perceiver_amd.models.audio.symbolic import SymbolicAudioModel, SymbolicAudioModelConfig

        self.model = SymbolicAudioModel(SymbolicAudioModelConfig.create(**self.hparams))
        params = self.hparams.get("params")
        if params is not None and is_checkpoint(params):
            wrapper = LitSymbolicAudioModel.load_from_checkpoint(params, params=None)
            self.model.load_state_dict(wrapper.model.state_dict())

    @classmethod
    def create(cls, config, **kwargs):
        from dataclasses import asdict

        return cls(**asdict(config), **kwargs)


def _lazy_hf(kind: str):
    try:
        if kind == "mlm":
            from perceiver_amd.models.text.mlm_hf import PerceiverMaskedLanguageModel

            return PerceiverMaskedLanguageModel
        if kind == "img":
            from perceiver_amd.models.vision.image_classifier_hf import PerceiverImageClassifier

            return PerceiverImageClassifier
    except ImportError:
        return None
    return None
