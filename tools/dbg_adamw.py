import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from perceiver_amd.ops import hip

ext = hip.ext()
torch.manual_seed(0)
n = 1000
master = torch.randn(n, device="cuda")
m = torch.zeros(n, device="cuda")
v = torch.zeros(n, device="cuda")
g = torch.randn(n, device="cuda")

m2, v2, p2 = m.clone(), v.clone(), master.clone()
lr, b1, b2, eps, wd = 1e-2, 0.9, 0.999, 1e-8, 0.01
for step in (1, 2):
    ext.adamw_step(master, m, v, g, lr, b1, b2, eps, wd, step)
    # reference
    p2.mul_(1 - lr * wd)
    m2.mul_(b1).add_(g, alpha=1 - b1)
    v2.mul_(b2).addcmul_(g, g, value=1 - b2)
    mh = m2 / (1 - b1 ** step)
    vh = v2 / (1 - b2 ** step)
    p2.add_(-lr * mh / (vh.sqrt() + eps))
    print(step, "master err", (master - p2).abs().max().item(),
          "m err", (m - m2).abs().max().item(), "v err", (v - v2).abs().max().item())
# also check tail oddsize
master = torch.randn(37, device="cuda"); m = torch.zeros(37, device="cuda")
v = torch.zeros(37, device="cuda"); g = torch.randn(37, device="cuda")
pm = master.clone()
ext.adamw_step(master, m, v, g, lr, b1, b2, eps, wd, 1)
print("tail changed all:", bool((master != pm).all()))
