"""Is the captured NMLL graph's replay corrupted by interleaved GPU work?"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np, torch
from dmosopt_amd.models import gp_core
from dmosopt_amd import ops

dev = torch.device("cuda", 0)
g = torch.Generator().manual_seed(0)
X = torch.rand(300, 30, generator=g).float().to(dev)
# EXTREME thetas like an SCE-UA init population (wide log bounds)
bl = np.array([np.log(1e-4), np.log(1e-3), np.log(1e-9)])
bu = np.array([np.log(1e3), np.log(100.0), np.log(1e-2)])
u = torch.rand(42, 3, generator=g).double().numpy()
th = torch.as_tensor(u * (bu - bl) + bl).float().to(dev)
y = torch.randn(42, 300, generator=g).float().to(dev)

gp_core._nmll_graphs.clear()
outs = []
for rep in range(6):
    o = gp_core.batched_nmll(X, y, th, nu=2.5, anisotropic=False)
    outs.append(o.clone())
    # unrelated interleaved GPU work (like the MOEA/sceua kernels)
    w = torch.randn(2048, 2048, device=dev)
    (w @ w).sum().item()
    junk = torch.full((4096, 4096), float("nan"), device=dev)
    junk.mul_(2.0)
    del w, junk
    torch.cuda.empty_cache()  # force allocator churn

ref = outs[0]
for i, o in enumerate(outs[1:], 1):
    if not torch.equal(ref, o):
        d = (ref - o)
        bad = torch.nonzero(~torch.isclose(ref, o, equal_nan=True)).flatten()
        print(f"replay {i} differs at {bad.numel()} elems; idx {bad[:6].tolist()}")
        for j in bad[:4].tolist():
            print(f"  elem {j}: {ref[j].item():.9g} vs {o[j].item():.9g}  theta={th[j].tolist()}")
    else:
        print(f"replay {i} identical")
