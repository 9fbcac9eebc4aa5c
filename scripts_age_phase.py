import os, sys, time
sys.path.insert(0, "/root/repo")
import numpy as np, torch
from dmosopt_amd.moea import agemoea as ag
from dmosopt_amd import ops, _hipops
dev = torch.device("cuda", 0)
w = torch.randn(512, 512, device=dev); (w @ w).sum().item()

rng = np.random.default_rng(0)
n, d, m, pop = 2048, 30, 2, 1024
X = torch.rand(n, d, device=dev)
# converged-ish front: most points near rank 0
t = torch.rand(n, 1, device=dev)
Y = torch.cat([t, 1 - t + 0.02 * torch.rand(n, 1, device=dev)], 1)

def tt(fn, reps=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3

print("envsel total: %.2f ms" % tt(lambda: ag.environmental_selection(np.random.default_rng(1), X, Y, pop, d, m)))
print("  rank dev: %.2f" % tt(lambda: ops.pareto_rank(Y.double())))
ys = Y.double().cpu().numpy()
rank = ops.pareto_rank(Y.double()).cpu().numpy()
front1 = np.flatnonzero(rank == 0)
print("  front1 size:", len(front1))
ideal = ys[front1].min(axis=0)
import dmosopt_amd.moea.agemoea as A
print("  survival_score: %.2f" % tt(lambda: A.survival_score(ys, front1, ideal), reps=5))
yf = ys[front1] - ideal
ex = A.find_corner_solutions(yf)
norm = A.normalize_front(yf, ex)
yn = yf / norm
p = A.get_geometry(yn, ex)
At = torch.as_tensor(yn, dtype=torch.float32, device="cuda")
Dt = _hipops.minkowski_norm_matrix(At.contiguous(), float(p))
pre = torch.zeros(len(front1), dtype=torch.uint8, device="cuda"); pre[:len(ex)] = 1
print("  minkowski mat: %.2f" % tt(lambda: _hipops.minkowski_norm_matrix(At.contiguous(), float(p))))
print("  survival kernel: %.2f" % tt(lambda: _hipops.agemoea_survival(Dt, pre).cpu(), reps=5))
print("  host prep (corner+norm+geom): %.2f" % tt(lambda: (A.find_corner_solutions(yf), A.normalize_front(yf, ex), A.get_geometry(yn, ex)), reps=5))
print("  D2H ys+rank: %.2f" % tt(lambda: (Y.double().cpu().numpy(), X.double().cpu().numpy())))
