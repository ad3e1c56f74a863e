"""GPU numerics tests: every HIP kernel against a plain PyTorch fp32 reference of
the same op (SURVEY.md §4 — the kernel-parity analog of the reference's golden
tests). Marked gpu; run on an MI355X via `pytest -m gpu`."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from perceiver_amd.ops import hip

    return hip.ext()


def _rand_qkv(b, h, nq, lk, d, dv, device, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    q = torch.randn(b, h, nq, d, generator=g) * (d ** -0.25)
    k = torch.randn(b, h, lk, d, generator=g) * (d ** -0.25)
    v = torch.randn(b, h, lk, dv, generator=g)
    return q.to(device), k.to(device), v.to(device)


def _eager_ref(q, k, v, pad_mask=None, causal=False):
    from perceiver_amd.ops.attention import eager_attention

    return eager_attention(q.float(), k.float(), v.float(), pad_mask=pad_mask, causal=causal)


SHAPES = [
    # (b, h, nq, lk, d, dv) — the Perceiver regimes
    (2, 8, 64, 256, 32, 160),    # MLM encoder cross-attn shape class
    (2, 8, 128, 128, 32, 160),   # MLM latent self-attn
    (2, 8, 256, 64, 32, 96),     # MLM decoder (long Q short KV)
    (2, 4, 96, 96, 64, 64),
    (1, 8, 128, 320, 128, 128),  # AR class
    (1, 1, 64, 500, 261, 261),   # img-classifier class (odd D, odd Lk)
    (1, 2, 40, 70, 48, 80),      # odd row counts
]


@pytest.mark.parametrize("shape", SHAPES)
def test_flash_fwd_matches_eager(shape):
    b, h, nq, lk, d, dv = shape
    device = "cuda"
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, device)
    ref = _eager_ref(q, k, v)
    out, lse = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), None, False, 0.0, 0)
    assert out.shape == ref.shape
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


@pytest.mark.parametrize("shape", [(2, 8, 128, 128, 32, 160), (1, 8, 96, 320, 128, 128),
                                   (2, 2, 70, 70, 48, 48)])
def test_flash_fwd_causal_right_aligned(shape):
    b, h, nq, lk, d, dv = shape
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=1)
    ref = _eager_ref(q, k, v, causal=True)
    out, _ = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), None, True, 0.0, 0)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_flash_fwd_pad_mask():
    b, h, nq, lk, d, dv = 2, 4, 32, 100, 32, 64
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=2)
    pad = torch.zeros(b, lk, dtype=torch.bool, device="cuda")
    pad[0, 60:] = True
    pad[1, :17] = True
    ref = _eager_ref(q, k, v, pad_mask=pad)
    out, _ = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), pad, False, 0.0, 0)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"

    # padded key content must not matter
    k2, v2 = k.clone(), v.clone()
    k2[0, :, 60:] = 7.0
    v2[0, :, 60:] = -3.0
    out2, _ = _ext().flash_fwd(q.bfloat16(), k2.bfloat16(), v2.bfloat16(), pad, False, 0.0, 0)
    assert torch.allclose(out.float(), out2.float(), atol=1e-5)


@pytest.mark.parametrize("shape,causal", [
    ((2, 4, 64, 128, 32, 160), False),
    ((2, 4, 64, 128, 32, 160), True),
    ((1, 2, 96, 96, 64, 64), True),
    ((1, 8, 64, 192, 128, 128), False),
    ((1, 1, 48, 200, 261, 261), False),
])
def test_flash_bwd_matches_autograd(shape, causal):
    b, h, nq, lk, d, dv = shape
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=3)

    qf = q.float().requires_grad_()
    kf = k.float().requires_grad_()
    vf = v.float().requires_grad_()
    ref = _eager_ref(qf, kf, vf, causal=causal)
    gout = torch.randn_like(ref)
    ref.backward(gout)

    qb, kb, vb = q.bfloat16(), k.bfloat16(), v.bfloat16()
    out, lse = _ext().flash_fwd(qb, kb, vb, None, causal, 0.0, 0)
    dq, dk, dv_ = _ext().flash_bwd(gout.bfloat16(), qb, kb, vb, out, lse, None, causal, 0.0, 0)

    for got, want, name in [(dq, qf.grad, "dq"), (dk, kf.grad, "dk"), (dv_, vf.grad, "dv")]:
        err = (got.float() - want).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        assert err / scale < 5e-2, f"{name} rel err {err/scale:.4f} (abs {err:.4f})"


def test_flash_autograd_function_end_to_end():
    from perceiver_amd.ops.flash import FlashAttention

    b, h, nq, lk, d, dv = 2, 4, 64, 128, 32, 160
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=4)
    qb = q.bfloat16().requires_grad_()
    kb = k.bfloat16().requires_grad_()
    vb = v.bfloat16().requires_grad_()
    out = FlashAttention.apply(qb, kb, vb, None, False, 0.0, False)
    out.sum().backward()
    assert qb.grad is not None and kb.grad is not None and vb.grad is not None
    assert torch.isfinite(qb.grad.float()).all()


def test_gelu_bias_matches_torch():
    x = torch.randn(1000, 512, device="cuda").bfloat16()
    b = torch.randn(512, device="cuda").bfloat16()
    y = _ext().gelu_bias_fwd(x, b)
    ref = torch.nn.functional.gelu(x.float() + b.float())
    assert torch.allclose(y.float(), ref, atol=2e-2, rtol=2e-2)

    dy = torch.randn_like(x)
    dx = _ext().gelu_bias_bwd(x, b, dy)
    xf = (x.float() + b.float()).requires_grad_()
    torch.nn.functional.gelu(xf).backward(dy.float())
    assert torch.allclose(dx.float(), xf.grad, atol=2e-2, rtol=2e-2)


def test_model_forward_uses_flash_and_matches_cpu():
    """Tiny MLM forward on GPU (bf16, flash path) vs CPU fp32 eager reference."""
    from perceiver_amd.models.text.common import TextEncoderConfig
    from perceiver_amd.models.text.mlm import MaskedLanguageModel, MaskedLanguageModelConfig, TextDecoderConfig

    torch.manual_seed(0)
    cfg = MaskedLanguageModelConfig(
        encoder=TextEncoderConfig(vocab_size=262, max_seq_len=256, num_input_channels=64,
                                  num_cross_attention_heads=4, num_self_attention_heads=4,
                                  num_self_attention_layers_per_block=2, dropout=0.0),
        decoder=TextDecoderConfig(vocab_size=262, max_seq_len=256, num_cross_attention_heads=4, dropout=0.0),
        num_latents=32, num_latent_channels=64,
    )
    model = MaskedLanguageModel(cfg).eval()
    x = torch.randint(0, 262, (2, 256))
    with torch.no_grad():
        ref = model(x, torch.zeros(2, 256, dtype=torch.bool))
        gm = model.cuda()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            got = gm(x.cuda(), torch.zeros(2, 256, dtype=torch.bool, device="cuda"))
    assert got.shape == ref.shape, f"shape mismatch {got.shape} vs {ref.shape}"
    err = (got.float().cpu() - ref).abs().max().item()
    assert err < 0.25, f"GPU/CPU logits diverge: {err}"


def test_fused_layer_norm_matches_torch():
    for C in (512, 768, 1280, 322):
        x = torch.randn(400, C, device="cuda")
        w = torch.randn(C, device="cuda").abs() + 0.5
        b = torch.randn(C, device="cuda")
        ref = torch.nn.functional.layer_norm(x, (C,), w, b, 1e-5)
        if C % 8:  # kernel requires the vector path; module falls back for others
            continue
        y, mean, rstd = _ext().ln_fwd(x.bfloat16(), w.bfloat16(), b.bfloat16(), 1e-5)
        assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2), C

        xf = x.clone().requires_grad_()
        wf = w.clone().requires_grad_()
        bf = b.clone().requires_grad_()
        out = torch.nn.functional.layer_norm(xf, (C,), wf, bf, 1e-5)
        dy = torch.randn_like(out)
        out.backward(dy)
        dx, dw, db = _ext().ln_bwd(dy.bfloat16(), x.bfloat16(), w.bfloat16(), mean, rstd, True)
        assert torch.allclose(dx.float(), xf.grad, atol=5e-2, rtol=5e-2), C
        # dw/db are 400-term reductions of bf16 products: scale the absolute
        # budget by the reduction magnitude instead of a flat atol=0.5 that
        # could hide a real bug at small magnitudes
        for got, want in ((dw.float(), wf.grad), (db.float(), bf.grad)):
            scale = want.abs().mean().clamp(min=1.0)
            err = (got - want).abs().max().item()
            assert err < 0.04 * float(scale) + 0.08, (C, err, float(scale))


def test_layer_norm_module_dispatch():
    from perceiver_amd.ops.norm import LayerNorm

    ln = LayerNorm(768).cuda().bfloat16()
    x = torch.randn(64, 768, device="cuda").bfloat16().requires_grad_()
    y = ln(x)
    y.sum().backward()
    ref = torch.nn.functional.layer_norm(x.float().detach(), (768,),
                                         ln.weight.float(), ln.bias.float(), ln.eps)
    assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2)
    assert x.grad is not None and ln.weight.grad is not None


def test_flash_dropout_statistics_and_backward():
    """Dropout rate ~p, kept entries scaled by 1/(1-p); fwd/bwd masks agree."""
    b, h, nq, lk, d, dv = 2, 4, 64, 256, 32, 64
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=5)
    qb, kb, vb = q.bfloat16(), k.bfloat16(), v.bfloat16()
    p, seed = 0.3, 12345

    out_nd, lse = _ext().flash_fwd(qb, kb, vb, None, False, 0.0, 0)
    out_d, lse_d = _ext().flash_fwd(qb, kb, vb, None, False, p, seed)
    # expectation preserved: mean over many keys approx equal
    rel = (out_d.float().mean(-1) - out_nd.float().mean(-1)).abs().mean().item()
    assert rel < 0.2, rel
    # deterministic for same seed
    out_d2, _ = _ext().flash_fwd(qb, kb, vb, None, False, p, seed)
    assert torch.equal(out_d, out_d2)
    # different for different seed
    out_d3, _ = _ext().flash_fwd(qb, kb, vb, None, False, p, seed + 1)
    assert not torch.equal(out_d, out_d3)
    # backward runs and is finite
    gout = torch.randn_like(out_d)
    dq, dk, dv_ = _ext().flash_bwd(gout, qb, kb, vb, out_d, lse_d, None, False, p, seed)
    for t in (dq, dk, dv_):
        assert torch.isfinite(t.float()).all()


def test_flash_dropout_grad_matches_eager_with_same_mask():
    """Cross-check dS math: compare against an eager composition that uses the
    kernel's own dropout mask (recovered by probing with identity V)."""
    b, h, nq, lk, d = 1, 1, 32, 64, 32
    q, k, v = _rand_qkv(b, h, nq, lk, d, lk, "cuda", seed=6)
    qb, kb = q.bfloat16(), k.bfloat16()
    p, seed = 0.25, 777
    # identity V -> out rows are the (normalized, dropped, rescaled... ) probs
    v_eye = torch.eye(lk, device="cuda").expand(b, h, lk, lk).contiguous().bfloat16()
    probs_dropped, _ = _ext().flash_fwd(qb, kb, v_eye, None, False, p, seed)
    probs_full, _ = _ext().flash_fwd(qb, kb, v_eye, None, False, 0.0, 0)
    mask = (probs_dropped.float() > 0) | (probs_full.float() == 0)
    keep_rate = mask.float().mean().item()
    assert abs(keep_rate - (1 - p)) < 0.05, keep_rate
    # dropped+rescaled probs match full probs / (1-p) where kept
    kept = probs_dropped.float()[mask]
    ref = (probs_full.float() / (1 - p))[mask]
    assert torch.allclose(kept, ref, atol=3e-2, rtol=3e-2)


def test_master_adamw_matches_fp32_adamw():
    from perceiver_amd.train.optim import MasterAdamW

    torch.manual_seed(0)
    w32 = torch.randn(100, 64, device="cuda")
    wbf = w32.bfloat16()
    p32 = w32.clone().requires_grad_()
    pbf = torch.nn.Parameter(wbf.clone())
    opt32 = torch.optim.AdamW([p32], lr=1e-2, weight_decay=0.01)
    optbf = MasterAdamW([pbf], lr=1e-2, weight_decay=0.01)
    for i in range(5):
        g = torch.randn_like(w32)
        p32.grad = g.clone()
        pbf.grad = g.bfloat16()
        opt32.step()
        optbf.step()
    err = (p32.detach() - pbf.detach().float()).abs().max().item()
    assert err < 2e-2, err


def test_fused_adamw_matches_reference_adamw():
    from perceiver_amd.train.optim import MasterAdamW

    torch.manual_seed(0)
    shapes = [(100, 64), (37,), (8, 8, 8)]
    w32 = [torch.randn(*s, device="cuda") for s in shapes]
    p32 = [w.clone().requires_grad_() for w in w32]
    pbf = [torch.nn.Parameter(w.bfloat16().clone()) for w in w32]
    opt32 = torch.optim.AdamW(p32, lr=1e-2, weight_decay=0.01)
    optbf = MasterAdamW(pbf, lr=1e-2, weight_decay=0.01)
    for _ in range(5):
        gs = [torch.randn_like(w) for w in w32]
        for p, g in zip(p32, gs):
            p.grad = g.clone()
        for p, g in zip(pbf, gs):
            p.grad = g.bfloat16()
        opt32.step()
        optbf.step()
    for a, b in zip(p32, pbf):
        # params live in bf16: tolerance = ~2 bf16 ulps at |w|~2 (the fused kernel
        # itself matches the fp32 update to ~2e-7; see tools/dbg_adamw.py)
        err = (a.detach() - b.detach().float()).abs().max().item()
        assert err < 4e-2, err


def test_mlp_gelu_bias_fusion_matches_eager():
    from perceiver_amd.core.modules import MLP

    torch.manual_seed(0)
    mlp = MLP(128, widening_factor=4).cuda().bfloat16()
    x = torch.randn(64, 128, device="cuda").bfloat16().requires_grad_()
    out = mlp(x).last_hidden_state
    out.sum().backward()
    gx = x.grad.clone()
    gb = mlp[1].bias.grad.clone()

    # eager reference (fp32)
    import torch.nn.functional as F

    xf = x.detach().float().requires_grad_()
    h = F.layer_norm(xf, (128,), mlp[0].weight.float(), mlp[0].bias.float(), mlp[0].eps)
    h = F.gelu(F.linear(h, mlp[1].weight.float(), mlp[1].bias.float()))
    ref = F.linear(h, mlp[3].weight.float(), mlp[3].bias.float())
    ref.sum().backward()
    assert torch.allclose(out.float(), ref, atol=0.1, rtol=0.1)
    rel = (gx.float() - xf.grad).abs().max() / (xf.grad.abs().max() + 1e-6)
    assert rel < 0.1, rel


@pytest.mark.gpu
def test_fused_adamw_in_step_clipping():
    """Fused-path in-step clipping equals external clip + unclipped fused step."""
    import copy

    from perceiver_amd.train.optim import MasterAdamW

    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    m1 = torch.nn.Sequential(torch.nn.Linear(64, 128), torch.nn.Linear(128, 32)).to(dev, torch.bfloat16)
    m2 = copy.deepcopy(m1)
    o1 = MasterAdamW(m1.parameters(), lr=1e-2, max_grad_norm=0.5)
    o2 = MasterAdamW(m2.parameters(), lr=1e-2)

    for _ in range(3):
        x = torch.randn(8, 64, device=dev, dtype=torch.bfloat16)
        for m, o, clip in ((m1, o1, False), (m2, o2, True)):
            o.zero_grad(set_to_none=True)
            (m(x).float().square().mean() * 37).backward()
            if clip:
                torch.nn.utils.clip_grad_norm_(m.parameters(), 0.5)
            o.step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        # clip order differs (flat fp32 copy vs per-param bf16): bf16-rounding-level tolerance
        torch.testing.assert_close(p1.float(), p2.float(), rtol=0, atol=4e-2)


@pytest.mark.gpu
def test_rccl_reducer_single_rank():
    """RCCL (nccl backend) process-group init + the bucketed reducer on CUDA
    tensors: world_size 1 makes all-reduce a no-op numerically but exercises the
    exact code path the 8-GPU bench runs (init, bucket build, async all_reduce,
    wait, unflatten)."""
    import os

    import torch.distributed as dist

    from perceiver_amd.parallel import BucketedGradReducer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29612")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        dev = torch.device("cuda:0")
        m = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.Linear(64, 8)).to(dev, torch.bfloat16)
        reducer = BucketedGradReducer(m, bucket_cap_mb=0.01)
        x = torch.randn(4, 64, device=dev, dtype=torch.bfloat16)
        loss = m(x).float().square().mean()
        loss.backward()
        reducer.finalize()
        for p in m.parameters():
            assert p.grad is not None and torch.isfinite(p.grad.float()).all()
    finally:
        dist.destroy_process_group()


@pytest.mark.gpu
@pytest.mark.parametrize("shape,rot,right", [
    ((2, 8, 64, 128), 64, True),     # AR self-attn class (rotate half the head)
    ((2, 8, 33, 128), 128, True),    # full-dim rotation, odd rows
    ((1, 4, 16, 96), 32, False),
])
def test_fused_rotary_matches_host(shape, rot, right):
    from perceiver_amd.core.position import FrequencyPositionEncoding, RotaryPositionEmbedding

    torch.manual_seed(0)
    b, h, n, d = shape
    dev = torch.device("cuda:0")
    frq_mod = FrequencyPositionEncoding(dim=rot).to(dev)
    pos = torch.arange(n + 5, device=dev).unsqueeze(0).expand(b, n + 5)
    emb = RotaryPositionEmbedding(frq_mod(pos), right_align=right)

    t = torch.randn(b, h, n, d, device=dev, dtype=torch.bfloat16, requires_grad=True)
    t_host = t.detach().clone().requires_grad_(True)

    out = emb.rotate(t)  # dispatches to the fused kernel on bf16 CUDA input
    # host reference: force the eager path by upcasting the table through cat
    from perceiver_amd.core.position import rotate_half_interleaved
    pos_enc = emb.frq_pos_enc[..., -n:, :] if right else emb.frq_pos_enc[..., :n, :]
    tr, tp = t_host[..., :rot], t_host[..., rot:]
    ref = torch.cat(((tr * pos_enc.cos() + rotate_half_interleaved(tr) * pos_enc.sin())
                     .to(t.dtype), tp), dim=-1)

    assert torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2), \
        (out.float() - ref.float()).abs().max().item()

    g = torch.randn_like(out)
    out.backward(g)
    ref.backward(g)
    assert torch.allclose(t.grad.float(), t_host.grad.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.gpu
def test_training_convergence_soak():
    """End-to-end learning check on GPU: a mid-size MLM (pure bf16 + fused
    master-AdamW + flash/LN/GELU kernels) must memorize a fixed synthetic
    mapping — loss falling well below its starting point proves the backward
    kernels produce usable gradients, not just allclose-passing ones."""
    from perceiver_amd.models.text.common import TextEncoderConfig
    from perceiver_amd.models.text.mlm import (
        MaskedLanguageModel,
        MaskedLanguageModelConfig,
        TextDecoderConfig,
    )
    from perceiver_amd.train.optim import MasterAdamW, convert_to_bf16_training

    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    cfg = MaskedLanguageModelConfig(
        encoder=TextEncoderConfig(vocab_size=262, max_seq_len=512, num_input_channels=256,
                                  num_cross_attention_heads=8, num_self_attention_heads=8,
                                  num_self_attention_layers_per_block=4, dropout=0.0),
        decoder=TextDecoderConfig(vocab_size=262, max_seq_len=512, num_cross_attention_heads=8,
                                  dropout=0.0),
        num_latents=64, num_latent_channels=256,
    )
    model = convert_to_bf16_training(MaskedLanguageModel(cfg).to(dev)).train()
    opt = MasterAdamW(model.parameters(), lr=3e-3, weight_decay=0.0)

    x = torch.randint(6, 262, (8, 512), device=dev)
    pad = torch.zeros(8, 512, dtype=torch.bool, device=dev)
    labels = torch.randint(6, 262, (8, 512), device=dev)  # fixed -> memorizable

    losses = []
    for _ in range(150):
        logits = model(x, pad)
        loss = torch.nn.functional.cross_entropy(logits.flatten(0, 1).float(), labels.flatten())
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert all(torch.isfinite(torch.tensor(losses))), losses[-5:]
    assert losses[-1] < losses[0] * 0.5, (losses[0], losses[-1])


@pytest.mark.gpu
def test_fused_adamw_state_roundtrip():
    """Checkpoint-resume of the fused optimizer: moments + fp32 masters must
    survive state_dict/load_state_dict — a resumed run takes the same next step
    as the uninterrupted one."""
    import copy

    from perceiver_amd.train.optim import MasterAdamW

    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    m = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.Linear(64, 16)).to(dev, torch.bfloat16)
    opt = MasterAdamW(m.parameters(), lr=1e-2)
    xs = [torch.randn(4, 64, device=dev, dtype=torch.bfloat16) for _ in range(3)]
    for x in xs[:2]:
        opt.zero_grad(set_to_none=True)
        (m(x).float().square().mean() * 11).backward()
        opt.step()

    # snapshot, then resume into a fresh model+optimizer
    model_sd = copy.deepcopy(m.state_dict())
    opt_sd = copy.deepcopy(opt.state_dict())
    m2 = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.Linear(64, 16)).to(dev, torch.bfloat16)
    m2.load_state_dict(model_sd)
    opt2 = MasterAdamW(m2.parameters(), lr=1e-2)
    opt2.load_state_dict(opt_sd)

    for mm, oo in ((m, opt), (m2, opt2)):
        oo.zero_grad(set_to_none=True)
        (mm(xs[2]).float().square().mean() * 11).backward()
        oo.step()
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


# ---- deep-pipelined forward (flash_fwd_pipe.hip, round 2) ----
# engages when Nq >= 128 and (D, Dv) hits an exact template; parity vs the
# same fp32 eager reference as the v3 kernel tests above

PIPE_SHAPES = [
    (2, 8, 128, 512, 32, 160),    # mlm-sa class
    (1, 8, 128, 2048, 32, 160),   # mlm-ca (long KV)
    (1, 8, 256, 512, 32, 96),     # mlm-dec
    (2, 4, 128, 130, 32, 64),     # ragged KV tail
    (1, 8, 128, 320, 128, 128),   # AR class
    (1, 2, 128, 448, 64, 64),
    (1, 2, 160, 100, 32, 32),     # Nq tail (160 = 128 + 32), Lk < 2 tiles
]


@pytest.mark.parametrize("shape", PIPE_SHAPES)
def test_flash_fwd_pipe_matches_eager(shape):
    b, h, nq, lk, d, dv = shape
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=7)
    ref = _eager_ref(q, k, v)
    out, lse = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), None, False, 0.0, 0)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"
    # lse must be finite where rows are live (the backward consumes it)
    assert torch.isfinite(lse).all()


@pytest.mark.parametrize("shape", [(1, 8, 256, 256, 128, 128),
                                   (2, 2, 128, 192, 64, 64),
                                   (1, 2, 192, 200, 32, 160)])
def test_flash_fwd_pipe_causal_right_aligned(shape):
    b, h, nq, lk, d, dv = shape
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=8)
    ref = _eager_ref(q, k, v, causal=True)
    out, _ = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), None, True, 0.0, 0)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_flash_fwd_pipe_pad_mask():
    b, h, nq, lk, d, dv = 2, 4, 128, 200, 32, 64
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=9)
    pad = torch.zeros(b, lk, dtype=torch.bool, device="cuda")
    pad[0, 150:] = True
    pad[1, :33] = True
    ref = _eager_ref(q, k, v, pad_mask=pad)
    out, _ = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), pad, False, 0.0, 0)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"

    # padded key content must not leak into the output
    k2, v2 = k.clone(), v.clone()
    k2[0, :, 150:] = 7.0
    v2[0, :, 150:] = -3.0
    out2, _ = _ext().flash_fwd(q.bfloat16(), k2.bfloat16(), v2.bfloat16(), pad, False, 0.0, 0)
    assert torch.allclose(out.float(), out2.float(), atol=1e-5)


def test_flash_fwd_pipe_dropout_statistics():
    # dropout is applied to P in-kernel; at p=0.5 the output magnitude stays
    # calibrated (E[out] = out_nodrop) and two seeds give different outputs
    b, h, nq, lk, d, dv = 1, 4, 128, 512, 32, 64
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=10)
    qb, kb, vb = q.bfloat16(), k.bfloat16(), v.bfloat16()
    base, _ = _ext().flash_fwd(qb, kb, vb, None, False, 0.0, 0)
    o1, _ = _ext().flash_fwd(qb, kb, vb, None, False, 0.5, 1234)
    o2, _ = _ext().flash_fwd(qb, kb, vb, None, False, 0.5, 5678)
    assert not torch.allclose(o1, o2)
    # average of many seeds approaches the no-dropout output
    acc = torch.zeros_like(base, dtype=torch.float32)
    for s in range(24):
        oi, _ = _ext().flash_fwd(qb, kb, vb, None, False, 0.5, 1000 + s)
        acc += oi.float()
    mean = acc / 24
    err = (mean - base.float()).abs().mean().item()
    assert err < 0.15, err


def test_dropout_add_fused_matches_semantics():
    # fused out = res + dropout(x): gradient wrt res is exactly dy; gradient
    # wrt x uses the SAME regenerated mask as the forward; seed-mean approaches
    # the p=0 sum
    from perceiver_amd.ops.dropadd import dropout_add

    torch.manual_seed(0)
    x = torch.randn(64, 1280, device="cuda").bfloat16().requires_grad_()
    r = torch.randn(64, 1280, device="cuda").bfloat16().requires_grad_()
    out = dropout_add(x, r, 0.5)
    dy = torch.randn_like(out)
    out.backward(dy)
    assert torch.equal(r.grad, dy)  # residual grad is the incoming grad
    # mask consistency: dx nonzero exactly where the forward kept x
    kept_fwd = (out - r).detach() != 0
    kept_bwd = x.grad != 0
    dy_nonzero = dy != 0
    x_nonzero = x.detach() != 0
    both = kept_fwd & kept_bwd & dy_nonzero & x_nonzero
    agree = (kept_fwd == kept_bwd) | ~(dy_nonzero & x_nonzero)
    assert agree.float().mean().item() > 0.999, agree.float().mean().item()
    assert both.any()

    # statistical calibration: mean over seeds ~= x + r
    acc = torch.zeros_like(out, dtype=torch.float32)
    for _ in range(32):
        acc += dropout_add(x.detach(), r.detach(), 0.5).float()
    err = (acc / 32 - (x.detach().float() + r.detach().float())).abs().mean().item()
    assert err < 0.2, err


def test_residual_module_uses_fused_path_in_training():
    from perceiver_amd.core.utils import ModuleOutput, Residual

    class Body(torch.nn.Module):
        def forward(self, x):
            return ModuleOutput(last_hidden_state=x * 2)

    res = Residual(Body(), dropout=0.1).cuda().train()
    x = torch.randn(8, 1280, device="cuda").bfloat16()
    out = res(x).last_hidden_state
    assert out.dtype == torch.bfloat16 and out.shape == x.shape
    res.eval()
    out_eval = res(x).last_hidden_state
    assert torch.allclose(out_eval.float(), (x * 3).float(), atol=1e-2, rtol=1e-2)


def test_colsum_matches_torch_sum():
    # the kernel emits bf16 directly (single final rounding, same as the old
    # fp32-out + .to(bf16) pair); compare against the identically-rounded
    # torch sum with tolerance for fp32 accumulation-order ulps
    for rows, C in [(401408, 261), (16384, 1792), (1000, 7)]:
        x = (torch.randn(rows, C) * 0.5).bfloat16().cuda()
        got = _ext().colsum_bf16(x)
        assert got.dtype == torch.bfloat16
        want = x.float().sum(0).bfloat16().float()
        torch.testing.assert_close(got.float(), want, atol=4.0, rtol=1e-2)


def test_perceiver_linear_bias_grad_matches_torch():
    from perceiver_amd.ops.linear import PerceiverLinear

    torch.manual_seed(0)
    lin = PerceiverLinear(64, 32).cuda().bfloat16()
    ref = torch.nn.Linear(64, 32).cuda().bfloat16()
    ref.load_state_dict(lin.state_dict())
    x = torch.randn(16384, 64, device="cuda").bfloat16()
    ya = lin(x); yb = ref(x)
    assert torch.equal(ya, yb)  # same forward kernel
    dy = torch.randn_like(ya)
    ya.backward(dy); yb.backward(dy)
    torch.testing.assert_close(lin.bias.grad.float(), ref.bias.grad.float(),
                               atol=2.0, rtol=2e-2)
    torch.testing.assert_close(lin.weight.grad.float(), ref.weight.grad.float(),
                               atol=2.0, rtol=2e-2)


def test_flash_fwd_pipe_kv_split_matches_eager():
    # tiny grid (B*H = 1) with long KV forces the gridDim.z KV-split + merge
    b, h, nq, lk, d, dv = 1, 1, 128, 4096, 64, 64
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=11)
    ref = _eager_ref(q, k, v)
    out, _ = _ext().flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), None, False, 0.0, 0)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_flash_bwd_q_split_matches_autograd():
    # the decoder-backward shape class: many queries, few keys, tiny B*H —
    # dkv takes the gridDim.z Q-split with fp32 partial accumulation
    b, h, nq, lk, d, dv = 1, 1, 2048, 128, 32, 64
    q, k, v = _rand_qkv(b, h, nq, lk, d, dv, "cuda", seed=12)
    qf = q.float().requires_grad_()
    kf = k.float().requires_grad_()
    vf = v.float().requires_grad_()
    ref = _eager_ref(qf, kf, vf)
    gout = torch.randn_like(ref)
    ref.backward(gout)

    qb, kb, vb = q.bfloat16(), k.bfloat16(), v.bfloat16()
    out, lse = _ext().flash_fwd(qb, kb, vb, None, False, 0.0, 0)
    dq, dk, dv_ = _ext().flash_bwd(gout.bfloat16(), qb, kb, vb, out, lse, None, False, 0.0, 0)
    for got, want, name in [(dq, qf.grad, "dq"), (dk, kf.grad, "dk"), (dv_, vf.grad, "dv")]:
        err = (got.float() - want).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        assert err / scale < 5e-2, f"{name} rel err {err/scale:.4f}"


def test_flash_padded_288_class_matches_autograd():
    # the img/flow D=261/322 class runs via zero-padding to the 288 template
    for d in (261, 200):
        b, h, nq, lk = 1, 1, 192, 640
        q, k, v = _rand_qkv(b, h, nq, lk, d, d, "cuda", seed=13)
        qf = q.float().requires_grad_()
        kf = k.float().requires_grad_()
        vf = v.float().requires_grad_()
        ref = _eager_ref(qf, kf, vf)
        gout = torch.randn_like(ref)
        ref.backward(gout)
        qb, kb, vb = q.bfloat16(), k.bfloat16(), v.bfloat16()
        out, lse = _ext().flash_fwd(qb, kb, vb, None, False, 0.0, 0)
        assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2), d
        dq, dk, dv_ = _ext().flash_bwd(gout.bfloat16(), qb, kb, vb, out, lse, None, False, 0.0, 0)
        for got, want in [(dq, qf.grad), (dk, kf.grad), (dv_, vf.grad)]:
            err = (got.float() - want).abs().max().item()
            scale = want.abs().max().item() + 1e-6
            assert err / scale < 5e-2, (d, err / scale)
