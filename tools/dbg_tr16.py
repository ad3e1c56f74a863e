"""Debug the tr16 PV path: sweep tile counts / nq / dv and localize the error."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from perceiver_amd.ops import hip
from perceiver_amd.ops.attention import eager_attention

ext = hip.ext()
dev = "cuda"
torch.manual_seed(0)

for nq in (64, 128):
    for lk in (64, 128, 192, 256, 320):
        for dv in (96, 160):
            q = (torch.randn(2, 8, nq, 32, device=dev) * 0.1).float()
            k = torch.randn(2, 8, lk, 32, device=dev).float()
            v = torch.randn(2, 8, lk, dv, device=dev).float()
            ref = eager_attention(q, k, v)
            out, _ = ext.flash_fwd(q.bfloat16(), k.bfloat16(), v.bfloat16(), None, False, 0.0, 0)
            err = (out.float() - ref).abs()
            flat = err.flatten()
            mx = err.max().item()
            tag = "OK " if mx < 2e-2 else "BAD"
            loc = ""
            if mx >= 2e-2:
                idx = err.argmax()
                b, h, i, c = torch.unravel_index(idx, err.shape)
                bad_rows = (err.amax(-1) > 2e-2).sum().item()
                bad_cols = sorted(set((err > 2e-2).nonzero()[:, 3].tolist()))
                loc = f" argmax b{b} h{h} row{i} col{c} badrows={bad_rows} badcolrange=[{bad_cols[0]}..{bad_cols[-1]}] n={len(bad_cols)}"
            print(f"{tag} nq={nq:3d} lk={lk:3d} dv={dv:3d} maxerr={mx:.4f}{loc}")
