"""Pipeline integration tests with locally-built (random-init) models — the offline
analog of the reference's download-based pipeline tests (SURVEY.md §4 category 4):
optical-flow patch-tiled pipeline and symbolic-audio-generation MIDI round trip."""
import numpy as np
import pytest
import torch
from transformers import pipeline

from perceiver_amd.data.audio.midi import Instrument, MidiFile, Note


@pytest.fixture(scope="module")
def flow_pipeline():
    from perceiver_amd.models.vision.optical_flow import (
        OpticalFlowConfig,
        OpticalFlowDecoderConfig,
        OpticalFlowEncoderConfig,
    )
    from perceiver_amd.models.vision.optical_flow_hf import (
        OpticalFlowPerceiver,
        OpticalFlowPerceiverConfig,
        OpticalFlowPipeline,
    )

    # patch (48, 64) with the pipeline's default min-overlap 20 -> strides (28, 44);
    # the reference grid requires test images < 2x stride beyond the patch
    cfg = OpticalFlowConfig(
        encoder=OpticalFlowEncoderConfig(image_shape=(48, 64), num_patch_input_channels=27,
                                         num_patch_hidden_channels=16, num_frequency_bands=4,
                                         num_cross_attention_heads=1, num_self_attention_heads=2,
                                         num_self_attention_layers_per_block=1),
        decoder=OpticalFlowDecoderConfig(image_shape=(48, 64), num_cross_attention_heads=1),
        num_latents=16, num_latent_channels=32,
    )
    model = OpticalFlowPerceiver(OpticalFlowPerceiverConfig(cfg)).eval()
    return OpticalFlowPipeline(model=model)


def test_optical_flow_pipeline(flow_pipeline):
    img1 = np.random.randint(0, 255, (54, 86, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (54, 86, 3), dtype=np.uint8)
    flow = flow_pipeline((img1, img2), micro_batch_size=2)
    assert flow.shape == (54, 86, 2)
    assert np.isfinite(flow).all()


def test_optical_flow_pipeline_batch(flow_pipeline):
    img = np.random.randint(0, 255, (54, 86, 3), dtype=np.uint8)
    flows = flow_pipeline([(img, img), (img, img)])
    assert len(flows) == 2
    assert flows[0].shape == (54, 86, 2)


@pytest.fixture(scope="module")
def audio_pipeline():
    from perceiver_amd.models.audio.symbolic import SymbolicAudioModelConfig
    from perceiver_amd.models.audio.symbolic_hf import (
        PerceiverSymbolicAudioModel,
        PerceiverSymbolicAudioModelConfig,
        SymbolicAudioPipeline,
    )

    cfg = SymbolicAudioModelConfig(
        vocab_size=389, max_seq_len=64, max_latents=16, num_channels=32, num_heads=4,
        num_self_attention_layers=1, cross_attention_dropout=0.0,
        output_norm=True, output_bias=False, abs_pos_emb=False,
    )
    model = PerceiverSymbolicAudioModel(PerceiverSymbolicAudioModelConfig(cfg)).eval()
    return SymbolicAudioPipeline(model=model)


def _prompt_midi():
    m = MidiFile()
    inst = Instrument()
    inst.notes = [Note(80, 60 + i, 0.2 * i, 0.2 * i + 0.15) for i in range(8)]
    m.instruments.append(inst)
    return m


def test_symbolic_audio_pipeline_returns_midi(audio_pipeline):
    from perceiver_amd.models.audio.symbolic_hf import ReturnType

    out = audio_pipeline(_prompt_midi(), max_new_tokens=8, num_latents=8,
                         return_type=ReturnType.AUDIO)
    assert "generated_audio_midi" in out
    assert isinstance(out["generated_audio_midi"], MidiFile)


def test_symbolic_audio_pipeline_returns_tokens(audio_pipeline):
    from perceiver_amd.models.audio.symbolic_hf import ReturnType

    out = audio_pipeline(_prompt_midi(), max_new_tokens=8, num_latents=8,
                         return_type=ReturnType.TENSORS, return_full_audio=False)
    toks = out["generated_token_ids"]
    assert len(toks) == 8
    assert all(0 <= t < 389 for t in toks)


def test_symbolic_audio_pipeline_max_prompt_length(audio_pipeline):
    from perceiver_amd.models.audio.symbolic_hf import ReturnType

    out = audio_pipeline(_prompt_midi(), max_prompt_length=10, max_new_tokens=4,
                         num_latents=4, return_type=ReturnType.TENSORS)
    assert len(out["generated_token_ids"]) == 14


# --------------------------------------------------------- text generation
@pytest.fixture(scope="module")
def text_generator():
    """transformers text-generation pipeline over the registered CLM wrapper
    (contract of reference tests/causal_language_model_pipeline_test.py:35-62,
    strategies x use_cache — offline PerceiverTokenizer, random-init weights)."""
    import torch
    from transformers import PerceiverTokenizer, pipeline

    from perceiver_amd.models.text.clm import CausalLanguageModelConfig
    from perceiver_amd.models.text.clm_hf import (
        PerceiverCausalLanguageModel,
        PerceiverCausalLanguageModelConfig,
    )

    torch.manual_seed(0)
    cfg = CausalLanguageModelConfig(vocab_size=262, max_seq_len=64, max_latents=16,
                                    num_channels=32, num_heads=4,
                                    num_self_attention_layers=2,
                                    cross_attention_dropout=0.0)
    model = PerceiverCausalLanguageModel(PerceiverCausalLanguageModelConfig(cfg)).eval()
    tok = PerceiverTokenizer(padding_side="left")
    return pipeline("text-generation", model=model, tokenizer=tok, device="cpu")


@pytest.mark.parametrize("use_cache", [True, False])
@pytest.mark.parametrize("kwargs", [
    {},                                             # greedy
    {"do_sample": True, "top_k": 5},                # top-k sampling
    {"do_sample": True, "top_p": 0.9},              # nucleus
    {"num_beams": 2},                               # beam search
    {"penalty_alpha": 0.6, "top_k": 4},             # contrastive
])
def test_text_generation_pipeline_strategies(text_generator, kwargs, use_cache):
    if not use_cache and ("penalty_alpha" in kwargs):
        pytest.skip("contrastive search requires the cache")
    out = text_generator("a quick brown fox", max_new_tokens=6, num_latents=4,
                         use_cache=use_cache, **kwargs)
    text = out[0]["generated_text"]
    assert isinstance(text, str)
    assert text.startswith("a quick brown fox")
