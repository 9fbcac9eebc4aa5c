"""Peel kernel timing on a converged-like population (many tiny fronts)."""
import os, sys, time
import torch
torch.set_num_threads(8)
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from dmosopt_amd import ops
dev = torch.device("cuda")
torch.manual_seed(0)
# ZDT1-like 2-obj population mid-optimization: strong correlation -> many fronts
N = 400
f1 = torch.rand(N, device=dev)
Y = torch.stack([f1, 1.0 - f1.sqrt() + 0.05 * torch.randn(N, device=dev)], 1)
r = ops.pareto_rank(Y)
print("fronts:", int(r.max().item()) + 1)
ref = ops.torch_ref.pareto_rank(Y.cpu())
assert torch.equal(r.cpu(), ref), "MISMATCH"
for _ in range(10):
    ops.pareto_rank(Y)
torch.cuda.synchronize()
ts = []
for _ in range(9):
    t0 = time.perf_counter()
    for _ in range(20):
        ops.pareto_rank(Y)
    torch.cuda.synchronize()
    ts.append((time.perf_counter() - t0) / 20)
ts.sort()
print(f"TPB={os.environ.get('DMOSOPT_PEEL_TPB','auto')}: {1e6*ts[4]:.1f} us (N={N}, incl. python dispatch)")
