"""Single-shape gemm_bt + hipBLASLt loop for PMC counter collection."""
import os
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))

import torch
import torch.nn.functional as F

from perceiver_amd.ops import hip as hip_ops


def main():
    ext = hip_ops.ext()
    dev = torch.device("cuda")
    M = int(os.environ.get("GM", 16384))
    N = int(os.environ.get("GN", 1280))
    K = int(os.environ.get("GK", 1280))
    iters = int(os.environ.get("GI", 20))
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
    for _ in range(iters):
        y = ext.gemm_bt(x, w, None)
    for _ in range(iters):
        z = F.linear(x, w)
    torch.cuda.synchronize()
    print("done", y.shape, z.shape)


if __name__ == "__main__":
    main()
