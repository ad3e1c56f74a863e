"""Microbench for the reduction kernels: LN dw/db and colsum, MLM/img shapes."""
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))

import torch

from perceiver_amd.ops import hip as hip_ops


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


def main():
    ext = hip_ops.ext()
    dev = torch.device("cuda")
    for rows, C in [(16384, 1280), (65536, 768), (32768, 1024)]:
        x = torch.randn(rows, C, device=dev, dtype=torch.bfloat16)
        dy = torch.randn(rows, C, device=dev, dtype=torch.bfloat16)
        w = torch.randn(C, device=dev, dtype=torch.bfloat16)
        y, mean, rstd = ext.ln_fwd(x, w, None, 1e-5)
        gb = 2 * rows * C * 2 / 1e9
        t_dwdb = bench(lambda: ext.ln_bwd(dy, x, w, mean, rstd, True))
        t_dx = bench(lambda: ext.ln_bwd(dy, x, w, mean, rstd, False))
        t_cs = bench(lambda: ext.colsum_bf16(dy))
        print(f"rows={rows} C={C}: ln_bwd full {t_dwdb:.1f} us | dx-only {t_dx:.1f} us | "
              f"dwdb-part {(t_dwdb - t_dx):.1f} us ({gb/((t_dwdb-t_dx)/1e6):.2f} GB/s eff) | "
              f"colsum {t_cs:.1f} us ({gb/2/(t_cs/1e6):.2f} GB/s)")


if __name__ == "__main__":
    main()
