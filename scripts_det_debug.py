"""Bisect the graph-on bit-determinism failure: fit the same GP twice and
compare theta; then NMLL streams call-by-call."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np, torch
from bench import make_archive, D_IN, N_OBJ
from dmosopt_amd.models.gp import GPRMatern
from dmosopt_amd.models import gp_core

dev = torch.device("cuda", 0)
X, Y = make_archive(seed=9)

def fit():
    gp = GPRMatern(X, Y, D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN),
                   optimizer="sceua", seed=5, device=dev)
    return gp.theta.clone()

t1 = fit()
t2 = fit()
print("theta equal:", torch.equal(t1, t2))
if not torch.equal(t1, t2):
    print("diff:", (t1 - t2).abs().max().item())
    print(t1.cpu().numpy())
    print(t2.cpu().numpy())

# call-by-call: identical inputs, repeated graph runs
gp_core._nmll_graphs.clear()
g = torch.Generator().manual_seed(0)
Xt = torch.rand(300, 30, generator=g).float().to(dev)
yt = torch.randn(18, 300, generator=g).float().to(dev)
th = torch.cat([torch.randn(18,1,generator=g)*0.3,
                torch.randn(18,1,generator=g)*0.5,
                torch.full((18,1), -8.0)], 1).float().to(dev)
outs = [gp_core.batched_nmll(Xt, yt, th, nu=2.5, anisotropic=False) for _ in range(5)]
for i in range(1, 5):
    if not torch.equal(outs[0], outs[i]):
        print(f"replay {i} differs by", (outs[0]-outs[i]).abs().max().item())
        break
else:
    print("5 replays identical")

# interleave two keys (B=42 and B=18) like a real fit does
gp_core._nmll_graphs.clear()
y42 = torch.randn(42, 300, generator=g).float().to(dev)
th42 = torch.cat([torch.randn(42,1,generator=g)*0.3,
                  torch.randn(42,1,generator=g)*0.5,
                  torch.full((42,1), -8.0)], 1).float().to(dev)
a1 = gp_core.batched_nmll(Xt, y42, th42, nu=2.5, anisotropic=False)
b1 = gp_core.batched_nmll(Xt, yt, th, nu=2.5, anisotropic=False)
a2 = gp_core.batched_nmll(Xt, y42, th42, nu=2.5, anisotropic=False)
b2 = gp_core.batched_nmll(Xt, yt, th, nu=2.5, anisotropic=False)
print("interleaved A equal:", torch.equal(a1, a2), "B equal:", torch.equal(b1, b2))
