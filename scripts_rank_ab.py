"""A/B pareto ranking paths at N=2048 (config-#3 regime): single-block
peel_from_y vs grid-wide matvec peel."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np, torch
from dmosopt_amd import ops, _hipops
assert ops.native_available()

dev = torch.device("cuda", 0)
w = torch.randn(512, 512, device=dev); (w @ w).sum().item()

def bench(fn, Y, reps=30):
    for _ in range(5): fn(Y)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps): out = fn(Y)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3, out

g = torch.Generator().manual_seed(0)
cases = {}
# random 2d (many fronts ~ O(log n)... measured below)
cases["rand2d"] = torch.rand(2048, 2, generator=g).float().to(dev)
# near-converged front: tight band around a curve -> very many fronts
t = torch.rand(2048, 1, generator=g)
cases["band2d"] = torch.cat([t, 1 - t.sqrt() + 0.01 * torch.rand(2048, 1, generator=g)], 1).float().to(dev)
cases["rand5d"] = torch.rand(2048, 5, generator=g).float().to(dev)

for name, Y in cases.items():
    t1, r1 = bench(lambda y: _hipops.pareto_rank(y.contiguous()), Y)
    t2, r2 = bench(lambda y: ops._pareto_rank_gpu(y), Y)
    same = torch.equal(r1.cpu(), r2.cpu())
    nf = int(r1.max().item()) + 1
    print(f"{name}: fronts={nf} single-block {t1:.3f} ms  matvec {t2:.3f} ms  agree={same}")

# cooperative-path timings at large N
print("--- large-N (cooperative route) ---")
for n, m in ((1024, 2), (1536, 2), (2048, 2), (3200, 2), (8192, 2), (4096, 5)):
    Yr = torch.rand(n, m, generator=g).float().to(dev)
    t3, r3 = bench(lambda y: ops.pareto_rank(y), Yr, reps=20)
    from dmosopt_amd.ops import torch_ref
    r_ref = torch_ref.pareto_rank(Yr.cpu().double())
    print(f"N={n} m={m}: dispatched {t3:.3f} ms  fronts={int(r3.max())+1}  correct={torch.equal(r3.cpu(), r_ref)}")
    tb, rb = bench(lambda y: ops._pareto_rank_gpu(y), Yr, reps=10)
    print(f"          matvec {tb:.3f} ms")
