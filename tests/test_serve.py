"""Serving endpoint (perceiver_amd.serve): /health and /generate over the
native generation stack, exercised in-process with the starlette TestClient."""
import pytest
import torch
from fastapi.testclient import TestClient
from transformers import PerceiverTokenizer

from perceiver_amd.models.text.clm import CausalLanguageModelConfig
from perceiver_amd.models.text.clm_hf import (
    PerceiverCausalLanguageModel,
    PerceiverCausalLanguageModelConfig,
)
from perceiver_amd.serve import create_app


@pytest.fixture(scope="module")
def client():
    torch.manual_seed(0)
    cfg = CausalLanguageModelConfig(vocab_size=262, max_seq_len=128, max_latents=32,
                                    num_channels=32, num_heads=4,
                                    num_self_attention_layers=2,
                                    cross_attention_dropout=0.0)
    model = PerceiverCausalLanguageModel(PerceiverCausalLanguageModelConfig(cfg)).eval()
    return TestClient(create_app(model, PerceiverTokenizer()))


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "ok" and body["max_seq_len"] == 128


def test_generate_greedy_deterministic(client):
    req = {"prompt": "a quick brown fox jumps", "max_new_tokens": 8, "num_latents": 4}
    r1 = client.post("/generate", json=req)
    r2 = client.post("/generate", json=req)
    assert r1.status_code == 200
    b1, b2 = r1.json(), r2.json()
    assert b1["generated_tokens"] == 8
    assert b1["prompt_tokens"] == len("a quick brown fox jumps")  # byte tokenizer
    assert b1["text"] == b2["text"]
    assert isinstance(b1["text"], str)


def test_generate_strategies(client):
    for extra in ({"do_sample": True, "top_k": 5},
                  {"num_beams": 2},
                  {"penalty_alpha": 0.6, "top_k": 4}):
        r = client.post("/generate", json={"prompt": "hello world", "max_new_tokens": 6,
                                           "num_latents": 4, **extra})
        assert r.status_code == 200, (extra, r.text)
        assert r.json()["generated_tokens"] == 6


def test_generate_clamps_max_new_tokens(client):
    r = client.post("/generate", json={"prompt": "x", "max_new_tokens": 500,
                                       "num_latents": 1})
    assert r.status_code == 200
    assert r.json()["generated_tokens"] <= 31  # max_latents - 1


def test_generate_overlong_prompt_truncated(client):
    # prompt longer than max_seq_len (128): server truncates to the tail
    # instead of letting generate() raise (unhandled 500)
    r = client.post("/generate", json={"prompt": "z" * 300, "max_new_tokens": 4,
                                       "num_latents": 8})
    assert r.status_code == 200
    assert r.json()["generated_tokens"] == 4


def test_generate_empty_prompt_is_400(client):
    r = client.post("/generate", json={"prompt": "", "max_new_tokens": 4})
    assert r.status_code == 400
