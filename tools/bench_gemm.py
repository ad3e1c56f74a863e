"""Microbench: custom deep-pipeline GEMM vs hipBLASLt (F.linear) on the
flagship projection shapes. Run on a GPU box:
    python tools/bench_gemm.py
Prints TF/s for both paths per shape.
"""
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))

import torch
import torch.nn.functional as F

from perceiver_amd.ops import hip as hip_ops


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters  # ms


def main():
    ext = hip_ops.ext()
    dev = torch.device("cuda")
    shapes = [
        (16384, 1280, 1280),   # MLM self-attn q/k/v/o and MLP
        (16384, 256, 1280),    # encoder CA q
        (65536, 256, 768),     # encoder CA k/v
        (16384, 1280, 2816),   # wide-K variant (dgrad-transposed class)
        (16384, 2816, 1280),
    ]
    for M, N, K in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
        b = torch.randn(N, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        t_ref = bench(lambda: F.linear(x, w, b))
        tf_ref = flops / (t_ref * 1e9)
        if ext.gemm_bt_applicable(M, N, K):
            # parity spot-check before timing
            y = ext.gemm_bt(x, w, b)
            ref = (x.float() @ w.float().t() + b.float()).to(torch.bfloat16)
            match = (y == ref).float().mean().item()
            t_new = bench(lambda: ext.gemm_bt(x, w, b))
            tf_new = flops / (t_new * 1e9)
            print(
                f"M={M} N={N} K={K}: hipBLASLt {t_ref:.3f} ms ({tf_ref:.0f} TF/s) | "
                f"gemm_bt {t_new:.3f} ms ({tf_new:.0f} TF/s) | "
                f"speedup {t_ref / t_new:.2f}x | exact-match {match:.4f}"
            )
        else:
            print(f"M={M} N={N} K={K}: hipBLASLt {t_ref:.3f} ms ({tf_ref:.0f} TF/s) | gemm_bt n/a")

    # dgrad comparison: dx = dy @ w (NN for hipBLASLt; gemm_bt runs B^T on a
    # transposed weight copy, amortized here exactly as the autograd path does)
    print("-- dgrad (dx = dy @ w) --")
    for M, N, K in [(16384, 1280, 1280), (16384, 1792, 1280), (65536, 768, 768)]:
        dy = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
        flops = 2.0 * M * N * K
        t_nn = bench(lambda: dy.matmul(w))
        if ext.gemm_bt_applicable(M, K, N):
            t_bt = bench(lambda: ext.gemm_bt(dy, w.t().contiguous(), None))
            ref = (dy.float() @ w.float()).to(torch.bfloat16)
            got = ext.gemm_bt(dy, w.t().contiguous(), None)
            match = (got == ref).float().mean().item()
            print(f"M={M} N={N} K={K}: torch-NN {t_nn:.3f} ms ({flops/(t_nn*1e9):.0f} TF/s) | "
                  f"gemm_bt+T {t_bt:.3f} ms ({flops/(t_bt*1e9):.0f} TF/s) | "
                  f"speedup {t_nn/t_bt:.2f}x | exact-match {match:.4f}")


if __name__ == "__main__":
    main()
