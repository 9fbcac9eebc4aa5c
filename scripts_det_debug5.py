"""Replay-after-fresh-write ordering test: inputs computed on-stream right
before each graph replay, compared against eager."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np, torch
from dmosopt_amd.models import gp_core
from dmosopt_amd import ops
dev = torch.device("cuda", 0)
gg = torch.Generator().manual_seed(0)
X = torch.rand(300, 30, generator=gg).float().to(dev)
bl = torch.tensor([np.log(1e-4), np.log(1e-3), np.log(1e-9)]).float().to(dev)
bu = torch.tensor([np.log(1e3), np.log(100.0), np.log(1e-2)]).float().to(dev)
y = torch.randn(42, 300, generator=gg).float().to(dev)
tgen = torch.Generator(device=dev); tgen.manual_seed(7)

gp_core._nmll_graphs.clear()
bad = 0
for it in range(30):
    # fresh device-side RNG draw right before the call (like rand_points)
    u = torch.rand(42, 3, device=dev, generator=tgen)
    th = u * (bu - bl) + bl
    got = gp_core.batched_nmll(X, y, th, nu=2.5, anisotropic=False)
    want = ops.gp_nmll_fused(X, th, y, 2.5, False, 1e-10)
    if not torch.equal(got, want):
        nb = int((~torch.isclose(got, want, equal_nan=True)).sum())
        print(f"iter {it}: graph != eager at {nb} elems")
        bad += 1
print("mismatched iters:", bad, "/ 30")
