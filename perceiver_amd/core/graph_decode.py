"""hipGraph-captured single-token decode for Perceiver-AR.

Small-batch decode is kernel-launch-bound: one cached token step of the flagship
(1 CA + 12 SA layers) issues ~150 small kernels whose launch latency dwarfs their
execution. This module re-expresses the cached decode step of
``CausalSequenceModel`` with every dynamic quantity in a device buffer — cache
lengths, write indices, the query position, the sampled token — so the whole step
(embed → CA → SA stack → logits → argmax → cache/length update) can be captured
once into a hipGraph and replayed per token with a single launch.

Design (vs. the host-driven path in ``PerceiverAR.forward``):

- KV caches read at FULL capacity every step; a per-step pad mask
  ``arange(capacity) > len`` masks the dead tail inside the flash kernel (same
  -inf fill as real padding). Lengths become device counters incremented
  in-graph; ``StaticKVCache.enable_graph_append`` switches the in-place append
  to ``index_copy_`` with the device index.
- Rotary tables are precomputed at full capacity (row i = absolute position i
  for the CA cache, ``prefix0 + i`` for the SA caches, both fixed for a given
  prefill). Only the query row is dynamic: it is regenerated each step from the
  device position counter through the model's own ``FrequencyPositionEncoding``,
  and for the self-attention stack written into the last row of the table
  (``right_align=True`` makes the 1-row query read exactly that row).
- Greedy selection, the token round-trip and the output record
  (``index_copy_`` at a device step counter) all stay on the GPU: a decode of N
  tokens is N graph replays with zero host synchronisation.

The un-captured step (``_step``) runs eagerly on any device, which is what the
CPU numerics test compares against the host-driven cached forward. The reference
has no equivalent (its cached decode is host-driven Python per token,
reference modules.py:820-871); this is MI355X-native serving machinery.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from perceiver_amd.core.cache import StaticKVCache
from perceiver_amd.core.position import RotaryPositionEmbedding


class GraphedDecoder:
    """Owns the static buffers and the (optionally hipGraph-captured) decode step
    for a ``CausalSequenceModel``-family model driven with ``StaticKVCache``s.

    Usage::

        caches = allocate_kv_cache(model, batch, device=dev, dtype=dtype)
        gd = GraphedDecoder(model, caches)
        gd.prefill(prompt, prefix_len=prompt.shape[1] - 1)   # normal forward
        tokens = gd.decode(n)                                # n graph replays
    """

    def __init__(self, model, caches: List[StaticKVCache], use_graph: bool = True,
                 do_sample: bool = False, temperature: float = 1.0,
                 top_k: Optional[int] = None):
        """``do_sample``: in-graph ancestral sampling (temperature scaling +
        optional top-k truncation + exponential-race Gumbel trick). torch's CUDA
        Philox state is graph-safe, so replays draw fresh randomness with zero
        host round-trips, same as the greedy path."""
        self.model = model
        self.caches = caches
        self.do_sample = do_sample
        self.temperature = temperature
        self.top_k = top_k
        p = next(model.parameters())
        self.device, self.dtype = p.device, p.dtype
        self.use_graph = use_graph and p.device.type == "cuda"
        self.batch = caches[0].k_buf.shape[0]
        self.ca_cap = caches[0].capacity
        self.sa_cap = caches[1].capacity if len(caches) > 1 else 0

        # full-capacity reads include not-yet-written rows: they are masked, but
        # 0-prob × garbage-NaN would still poison P@V — keep the tails finite
        for c in caches:
            c.k_buf.zero_()
            c.v_buf.zero_()

        dev = self.device
        self.tok = torch.zeros(self.batch, 1, dtype=torch.long, device=dev)
        # device counters: current cache lengths (= index the next token is
        # written at) and the output-record cursor
        self.ca_len = torch.zeros(1, dtype=torch.long, device=dev)
        self.sa_len = torch.zeros(1, dtype=torch.long, device=dev)
        self.step_idx = torch.zeros(1, dtype=torch.long, device=dev)
        self.ar_ca = torch.arange(self.ca_cap, device=dev).unsqueeze(0)
        self.ar_sa = torch.arange(self.sa_cap, device=dev).unsqueeze(0)
        self.out_buf = torch.zeros(self.batch, self.sa_cap or self.ca_cap,
                                   dtype=torch.long, device=dev)
        self._graph = None

    # ------------------------------------------------------------------ prefill

    @torch.no_grad()
    def prefill(self, prompt: torch.Tensor, prefix_len: int) -> torch.Tensor:
        """Run the normal host-driven cached forward over the prompt, seed the
        device counters/buffers, and return the first greedy token."""
        for c in self.caches:
            c.disable_graph_append()
            c.reset()
            # store keys pre-rotated at their absolute positions: the decode
            # step then never touches cached rows again (no O(cache) re-rotate)
            c.pre_rotated = True
        out = self.model(prompt, prefix_len=prefix_len, kv_cache=self.caches)
        tok = self._select(out.logits[:, -1, :].float())

        n0 = prompt.shape[1]
        prefix0 = prefix_len
        self.tok.copy_(tok)
        self.ca_len.fill_(n0)
        self.sa_len.fill_(n0 - prefix0)
        self.step_idx.fill_(0)
        self.ca_len_host = n0
        self.sa_len_host = n0 - prefix0
        self.caches[0].enable_graph_append(self.ca_len)
        for c in self.caches[1:]:
            c.enable_graph_append(self.sa_len)
        return tok

    def _select(self, scores: torch.Tensor) -> torch.Tensor:
        """Greedy or Gumbel-max sampled token from (batch, vocab) fp32 scores.
        Capture-safe: torch's CUDA Philox state is graph-aware, and the
        exponential_/log Gumbel trick avoids multinomial's sync."""
        if not self.do_sample:
            return scores.argmax(-1, keepdim=True)
        if self.temperature != 1.0:
            scores = scores / self.temperature
        if self.top_k is not None and self.top_k > 0:
            kth = torch.topk(scores, min(self.top_k, scores.shape[-1]))[0][..., -1, None]
            scores = scores.masked_fill(scores < kth, float("-inf"))
        g = torch.empty_like(scores).exponential_().log().neg_()
        return (scores + g).argmax(-1, keepdim=True)

    # --------------------------------------------------------------------- step

    def _step(self):
        """One single-token decode step, everything dynamic read from device
        buffers. Mirrors the cached branch of ``PerceiverAR.forward`` +
        ``CausalSequenceModel.forward`` (modules.py) with full-capacity masked
        attention instead of host-length views."""
        m = self.model
        pos_q = self.ca_len.view(1, 1).expand(self.batch, 1)
        x, frq_q = m.input_adapter(self.tok, abs_pos=pos_q)

        # cross-attention: new token is the single latent; cached K/V at full
        # capacity, tail j > ca_len masked (the new token lands AT index ca_len).
        # Caches are pre-rotated, so BOTH rotary tables are just the current
        # position's row: q uses it directly, k uses it to rotate the new row
        # before it is appended.
        rot_now = RotaryPositionEmbedding(frq_q, right_align=True)
        ca_mask = (self.ar_ca > self.ca_len).expand(self.batch, -1)
        ca_out = m.cross_attention(
            x,
            x_kv_prefix=x[:, :0],
            pad_mask=ca_mask,
            rot_pos_emb_q=rot_now,
            rot_pos_emb_k=rot_now,
            kv_cache=self.caches[0],
        )

        sa_mask = (self.ar_sa > self.sa_len).expand(self.batch, -1)
        sa_out = m.self_attention(
            ca_out.last_hidden_state,
            pad_mask=sa_mask,
            rot_pos_emb=rot_now,
            kv_cache=list(self.caches[1:]),
        )

        h = sa_out.last_hidden_state
        if m.config.output_norm:
            h = m.out_norm(h)
        logits = m.output_adapter(h, txt_embedding=m.input_adapter.txt_embedding)

        # token selection + bookkeeping, all on-device: the graph is self-advancing
        self.last_logits = logits  # graph-pool tensor: valid until the next replay
        self.out_buf.index_copy_(1, self.step_idx, self.tok)
        self.tok.copy_(self._select(logits[:, -1, :].float()))
        self.ca_len.add_(1)
        self.sa_len.add_(1)
        self.step_idx.add_(1)

    def _capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):  # allocator/dispatch warmup — real decode steps
                self._step()
                self.ca_len_host += 1
                self.sa_len_host += 1
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._step()

    @torch.no_grad()
    def decode(self, n: int) -> torch.Tensor:
        """Generate ``n`` greedy tokens after ``prefill``; returns (batch, n).
        The first call captures the step graph (2 warmup steps run eagerly and
        count toward ``n``); later calls are pure replays."""
        if self.ca_len_host + n > self.ca_cap or self.sa_len_host + n > self.sa_cap:
            raise RuntimeError("decode would overflow the KV cache capacity")
        done = 0
        if self.use_graph and self._graph is None:
            if n < 2:
                raise RuntimeError("first decode() call needs n >= 2 to capture")
            self._capture()
            done = 2
        for _ in range(n - done):
            if self.use_graph:
                self._graph.replay()
            else:
                self._step()
            self.ca_len_host += 1
            self.sa_len_host += 1
        # out_buf rows 0..n-1 hold the tokens EMITTED by each step's predecessor:
        # row i is the token fed INTO step i, i.e. the model's (i)'th generated
        # token counting the prefill argmax as the 0'th. Together with the final
        # self.tok this yields the n generated tokens after the prefill token.
        first = self.out_buf[:, 1:n]
        return torch.cat([first, self.tok], dim=1)
