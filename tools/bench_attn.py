"""Standalone flash-attention kernel microbench: times fwd/bwd on the Perceiver
flagship shapes and prints TFLOP/s (algorithmic 2*N*L*(D+Dv) per direction-pass).

Usage (GPU box): python tools/bench_attn.py [--iters 50]
"""
import argparse
import sys
import time

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SHAPES = [
    # name, b, h, nq, lk, d, dv, causal
    ("mlm-sa", 32, 8, 512, 512, 32, 160, False),
    ("mlm-ca", 32, 8, 512, 2048, 32, 160, False),
    ("mlm-dec", 32, 8, 2048, 512, 32, 96, False),
    ("ar-sa", 8, 8, 1024, 1024, 128, 128, True),
    ("ar-ca", 8, 8, 1024, 8192, 128, 128, True),
    ("img-ca", 8, 1, 512, 50176, 261, 261, False),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--only", default=None, help="comma-separated shape names")
    args = ap.parse_args()
    shapes = SHAPES if args.only is None else [s for s in SHAPES if s[0] in args.only.split(",")]

    from perceiver_amd.ops import hip

    ext = hip.ext()
    dev = "cuda"
    results = {}
    for name, b, h, nq, lk, d, dv, causal in shapes:
        q = (torch.randn(b, h, nq, d, device=dev) * (d ** -0.5)).bfloat16()
        k = torch.randn(b, h, lk, d, device=dev).bfloat16()
        v = torch.randn(b, h, lk, dv, device=dev).bfloat16()

        out, lse = ext.flash_fwd(q, k, v, None, causal, 0.0, 0)
        gout = torch.randn_like(out)

        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            out, lse = ext.flash_fwd(q, k, v, None, causal, 0.0, 0)
        torch.cuda.synchronize()
        fwd_ms = (time.perf_counter() - t0) / args.iters * 1e3

        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            dq, dk, dv_ = ext.flash_bwd(gout, q, k, v, out, lse, None, causal, 0.0, 0)
        torch.cuda.synchronize()
        bwd_ms = (time.perf_counter() - t0) / args.iters * 1e3

        causal_frac = 1.0 if not causal else (1 - max(0, (lk - nq)) / lk) * 0.5 + max(0, lk - nq) / lk
        flops_fwd = 2 * b * h * nq * lk * (d + dv) * causal_frac
        flops_bwd = flops_fwd * 2.5  # recompute S + 4 matmul-equivalents vs 2
        results[name] = (fwd_ms, flops_fwd / fwd_ms / 1e9, bwd_ms, flops_bwd / bwd_ms / 1e9)
        print(f"{name:8s} fwd {fwd_ms:7.3f} ms ({results[name][1]:6.1f} TF/s)   "
              f"bwd {bwd_ms:7.3f} ms ({results[name][3]:6.1f} TF/s)")

    # eager comparison on the training-critical shape
    name, b, h, nq, lk, d, dv, causal = SHAPES[0]
    q = (torch.randn(b, h, nq, d, device=dev) * (d ** -0.5)).bfloat16()
    k = torch.randn(b, h, lk, d, device=dev).bfloat16()
    v = torch.randn(b, h, lk, dv, device=dev).bfloat16()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        s = torch.matmul(q, k.transpose(-2, -1))
        p = s.softmax(-1)
        o = torch.matmul(p, v)
    torch.cuda.synchronize()
    eager_ms = (time.perf_counter() - t0) / args.iters * 1e3
    print(f"eager mlm-sa fwd {eager_ms:.3f} ms (bmm+softmax+bmm)")


if __name__ == "__main__":
    main()
