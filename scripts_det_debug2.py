"""Find the first divergent NMLL call between two same-seed fits."""
import os, sys, hashlib
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np, torch
from bench import make_archive, D_IN, N_OBJ
from dmosopt_amd.models import gp_core
from dmosopt_amd.models.gp import GPRMatern

dev = torch.device("cuda", 0)
X, Y = make_archive(seed=9)

orig = gp_core.batched_nmll
log = []

def spy(Xt, y, theta, **kw):
    out = orig(Xt, y, theta, **kw)
    def h(t):
        return hashlib.blake2b(t.detach().cpu().numpy().tobytes(), digest_size=8).hexdigest()
    log.append((tuple(theta.shape), h(theta), h(y), h(out)))
    return out

gp_core_batched = gp_core.batched_nmll
import dmosopt_amd.models.gp as gpmod
# gp.py closes over batched_nmll by name import; patch there
gpmod.batched_nmll = spy

def fit():
    log.clear()
    gp = GPRMatern(X, Y, D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN),
                   optimizer="sceua", seed=5, device=dev)
    return list(log), gp.theta.clone()

l1, t1 = fit()
l2, t2 = fit()
print("fit1 calls:", len(l1), "fit2 calls:", len(l2))
print("theta equal:", torch.equal(t1, t2))
n = min(len(l1), len(l2))
for i in range(n):
    if l1[i] != l2[i]:
        print(f"first divergence at call {i}:")
        print("  fit1:", l1[i])
        print("  fit2:", l2[i])
        same_in = l1[i][1] == l2[i][1] and l1[i][2] == l2[i][2]
        print("  inputs equal:", same_in, "-> output equal:", l1[i][3] == l2[i][3])
        break
else:
    print("all common calls identical")
