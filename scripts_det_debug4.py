"""Does capturing a second CUDA graph corrupt the first one's replays?"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np, torch
from dmosopt_amd.models import gp_core
dev = torch.device("cuda", 0)
g = torch.Generator().manual_seed(0)
X = torch.rand(300, 30, generator=g).float().to(dev)
bl = np.array([np.log(1e-4), np.log(1e-3), np.log(1e-9)])
bu = np.array([np.log(1e3), np.log(100.0), np.log(1e-2)])
th42 = torch.as_tensor(torch.rand(42,3,generator=g).double().numpy()*(bu-bl)+bl).float().to(dev)
y42 = torch.randn(42, 300, generator=g).float().to(dev)
th18 = torch.as_tensor(torch.rand(18,3,generator=g).double().numpy()*(bu-bl)+bl).float().to(dev)
y18 = torch.randn(18, 300, generator=g).float().to(dev)

gp_core._nmll_graphs.clear()
a1 = gp_core.batched_nmll(X, y42, th42, nu=2.5, anisotropic=False).clone()  # capture 42
a2 = gp_core.batched_nmll(X, y42, th42, nu=2.5, anisotropic=False).clone()  # replay 42
print("42 pre-18:", torch.equal(a1, a2))
b1 = gp_core.batched_nmll(X, y18, th18, nu=2.5, anisotropic=False).clone()  # capture 18
a3 = gp_core.batched_nmll(X, y42, th42, nu=2.5, anisotropic=False).clone()  # replay 42 after capture 18
print("42 post-18-capture:", torch.equal(a1, a3))
if not torch.equal(a1, a3):
    bad = torch.nonzero(~torch.isclose(a1, a3, equal_nan=True)).flatten()
    print("differs at", bad.numel(), "elems:", bad[:5].tolist())
    for j in bad[:3].tolist():
        print(f"  {a1[j].item():.9g} vs {a3[j].item():.9g}")
b2 = gp_core.batched_nmll(X, y18, th18, nu=2.5, anisotropic=False).clone()
print("18 replay consistent:", torch.equal(b1, b2))
