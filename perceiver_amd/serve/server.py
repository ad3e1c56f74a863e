"""Minimal production-serving endpoint for Perceiver-AR text generation.

FastAPI app over the 🤗 causal wrapper: on GPU the decode path is the
hipGraph-replayed ``GraphedDecoder`` (one replay per token, pre-rotated static
caches); on CPU it falls back to the native ``generate`` loop. The reference
has no serving layer — this is MI355X-native deployment machinery.

Run:
    python -m perceiver_amd.serve --model <save_pretrained dir> --port 8000
    curl -s localhost:8000/generate -d '{"prompt": "hello", "max_new_tokens": 32}'
"""
from __future__ import annotations

import threading
from typing import Optional

import torch
from fastapi import FastAPI, HTTPException
from pydantic import BaseModel, Field


class GenerateRequest(BaseModel):
    prompt: str
    max_new_tokens: int = Field(default=64, ge=1, le=4096)
    num_latents: int = Field(default=64, ge=1)
    do_sample: bool = False
    temperature: float = Field(default=1.0, gt=0)
    top_k: Optional[int] = Field(default=None, ge=1)
    top_p: Optional[float] = Field(default=None, gt=0, le=1)
    num_beams: int = Field(default=1, ge=1)
    penalty_alpha: Optional[float] = Field(default=None, ge=0, le=1)


class GenerateResponse(BaseModel):
    text: str
    prompt_tokens: int
    generated_tokens: int


def create_app(model, tokenizer) -> FastAPI:
    """``model``: a PerceiverCausalSequenceModel subclass (eval mode);
    ``tokenizer``: any 🤗 tokenizer (PerceiverTokenizer for byte models)."""
    app = FastAPI(title="perceiver-mi355x", version="0.1")
    lock = threading.Lock()  # one generation at a time per worker
    device = next(model.parameters()).device

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(device),
                "max_seq_len": model.backend_model.max_seq_len,
                "max_latents": model.backend_model.max_latents}

    @app.post("/generate", response_model=GenerateResponse)
    def generate(req: GenerateRequest):
        enc = tokenizer(req.prompt, return_tensors="pt", add_special_tokens=False)
        ids = enc["input_ids"].to(device)
        if ids.shape[1] == 0:
            raise HTTPException(status_code=400, detail="prompt tokenized to 0 tokens")
        backend = model.backend_model
        # keep the prompt inside the model's context window (last tokens win)
        if ids.shape[1] >= backend.max_seq_len:
            ids = ids[:, -(backend.max_seq_len - 1):]
        n_prompt = ids.shape[1]
        # clamp latents into the valid range: enough that the prefix fits
        # max_prefix_len, at most the prompt length / max_latents
        lo = max(1, n_prompt - backend.max_prefix_len)
        num_latents = max(lo, min(req.num_latents, n_prompt, backend.max_latents))
        max_new = min(req.max_new_tokens, backend.max_latents - 1)
        with lock, torch.no_grad():
            out = model.generate(
                input_ids=ids,
                num_latents=num_latents,
                max_new_tokens=max_new,
                do_sample=req.do_sample,
                temperature=req.temperature,
                top_k=req.top_k,
                top_p=req.top_p,
                num_beams=req.num_beams,
                penalty_alpha=req.penalty_alpha,
            )
        new_tokens = out[0, n_prompt:]
        return GenerateResponse(
            text=tokenizer.decode(new_tokens, skip_special_tokens=True),
            prompt_tokens=n_prompt,
            generated_tokens=int(new_tokens.shape[0]),
        )

    return app
