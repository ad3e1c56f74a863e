"""🤗 wrapper + registered pipeline "symbolic-audio-generation": MIDI prompt ->
encode_midi -> generate (sliding latent/prefix schedule) -> decode_midi -> optional
fluidsynth WAV rendering. Parity: reference audio/symbolic/huggingface.py, built on
the self-contained MIDI layer (perceiver_amd.data.audio.midi.MidiFile instead of
pretty_midi)."""
from __future__ import annotations

import enum
import os
import subprocess
import tempfile
from dataclasses import asdict
from typing import Any, Dict, Optional

import torch
from transformers import AutoConfig, AutoModelForCausalLM, Pipeline, PretrainedConfig
from transformers.pipelines import PIPELINE_REGISTRY
from transformers.utils import ModelOutput

from perceiver_amd.data.audio.midi import MidiFile
from perceiver_amd.data.audio.midi_processor import decode_midi, encode_midi
from perceiver_amd.models.audio.symbolic import SymbolicAudioModel, SymbolicAudioModelConfig
from perceiver_amd.models.hf_base import (
    PerceiverCausalSequenceModel,
    PerceiverCausalSequenceModelOutput,
)
from perceiver_amd.models.hf_registry import BackendConfigMixin


class ReturnType(enum.Enum):
    TENSORS = 0
    AUDIO = 1


class PerceiverSymbolicAudioModelConfig(BackendConfigMixin, PretrainedConfig):
    model_type = "perceiver-ar-symbolic-audio-model"
    backend_config_class = SymbolicAudioModelConfig

    def __init__(self, backend_config=None, **kwargs):
        # explicit __init__: transformers 5.x wraps configs without one
        # in a kwargs-only guard that would swallow backend_config
        super().__init__(backend_config, **kwargs)


class PerceiverSymbolicAudioModel(PerceiverCausalSequenceModel):
    config_class = PerceiverSymbolicAudioModelConfig

    def __init__(self, config: PerceiverSymbolicAudioModelConfig, **kwargs):
        super().__init__(config)
        self.backend_model = kwargs.get("backend_model") or SymbolicAudioModel(config.backend_config)
        self.post_init()

    @staticmethod
    def from_checkpoint(ckpt_path):
        from perceiver_amd.models.hf_registry import wrap_lit_checkpoint
        from perceiver_amd.train.lit import LitSymbolicAudioModel

        return wrap_lit_checkpoint(LitSymbolicAudioModel, PerceiverSymbolicAudioModel,
                                   ckpt_path, is_decoder=True)


class SymbolicAudioPipeline(Pipeline):
    """Task "symbolic-audio-generation": MidiFile prompt -> generated MidiFile or
    rendered WAV bytes (requires fluidsynth)."""

    def _sanitize_parameters(self, max_prompt_length=None, return_full_audio=True,
                             return_type=ReturnType.AUDIO, render=False, sf2_path=None,
                             **generate_kwargs):
        preprocess_params = {}
        forward_params = generate_kwargs
        postprocess_params = {}

        if max_prompt_length is not None:
            if max_prompt_length == 0:
                raise ValueError("max_prompt_length must be > 0")
            preprocess_params["max_prompt_length"] = max_prompt_length

        if render and not self._can_render_midi_files():
            raise ValueError("Rendering requires the library `fluidsynth` to be installed.")
        postprocess_params["render"] = render

        if sf2_path is not None and not os.path.exists(sf2_path):
            raise ValueError(f"Provided sf2_path=`{sf2_path}` does not exist")
        postprocess_params["sf2_path"] = sf2_path
        postprocess_params["return_full_audio"] = return_full_audio
        postprocess_params["return_type"] = return_type
        return preprocess_params, forward_params, postprocess_params

    @staticmethod
    def _can_render_midi_files() -> bool:
        try:
            subprocess.check_output(["which", "fluidsynth"])
            return True
        except subprocess.CalledProcessError:
            return False

    def preprocess(self, prompt_midi: MidiFile, max_prompt_length: int = None, **kwargs):
        encoded_input = torch.tensor(encode_midi(prompt_midi))
        if max_prompt_length is not None:
            encoded_input = encoded_input[:max_prompt_length]
        return {"input_features": torch.unsqueeze(encoded_input, dim=0)}

    def _forward(self, model_inputs: Dict, **generate_kwargs) -> ModelOutput:
        input_features = model_inputs["input_features"]
        generated = self.model.generate(input_ids=input_features, **generate_kwargs)
        model_output = PerceiverCausalSequenceModelOutput(logits=generated)
        model_output["prompt_length"] = input_features.shape[1]
        return model_output

    def postprocess(self, model_outputs: ModelOutput, return_type=ReturnType.AUDIO,
                    return_full_audio=True, render=False, sf2_path=None, **kwargs) -> Any:
        generated_sequence = model_outputs["logits"][0].cpu().numpy().tolist()
        prompt_length = model_outputs["prompt_length"]

        if return_type == ReturnType.TENSORS:
            return {"generated_token_ids":
                    generated_sequence if return_full_audio else generated_sequence[prompt_length:]}

        if return_type == ReturnType.AUDIO:
            sequence = generated_sequence if return_full_audio else generated_sequence[prompt_length:]
            midi = decode_midi(sequence)
            if not render:
                return {"generated_audio_midi": midi}

            with tempfile.TemporaryDirectory() as tmp_dir:
                midi_file = os.path.join(tmp_dir, "generated_audio.mid")
                wav_file = os.path.join(tmp_dir, "generated_audio.wav")
                midi.write(midi_file)
                cmd = ["fluidsynth", "-F", wav_file]
                if sf2_path is not None:
                    cmd.append(sf2_path)
                cmd.append(midi_file)
                subprocess.run(cmd)
                with open(wav_file, "rb") as f:
                    generated_wav = f.read()
                return {"generated_audio_wav": generated_wav}

        raise ValueError(f"Invalid return_type={return_type}")


AutoConfig.register(PerceiverSymbolicAudioModelConfig.model_type, PerceiverSymbolicAudioModelConfig)
AutoModelForCausalLM.register(PerceiverSymbolicAudioModelConfig, PerceiverSymbolicAudioModel)
PIPELINE_REGISTRY.register_pipeline(
    "symbolic-audio-generation",
    pipeline_class=SymbolicAudioPipeline,
    pt_model=PerceiverSymbolicAudioModel,
)


def convert_checkpoint(save_dir, ckpt_url, **kwargs):
    """LitSymbolicAudioModel .ckpt -> persistent PerceiverSymbolicAudioModel dir."""
    model = PerceiverSymbolicAudioModel.from_checkpoint(ckpt_url)
    model.save_pretrained(save_dir, **kwargs)
