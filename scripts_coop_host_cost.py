"""Does hipLaunchCooperativeKernel block the host? Time async submission.

python scripts_coop_host_cost.py  (GPU box)
"""
import sys, time
import torch
sys.path.insert(0, "/root/repo")
from dmosopt_amd import _hipops

dev = torch.device("cuda")
g = torch.Generator().manual_seed(3)
for N in (400, 3200):
    Y = torch.rand(N, 2, generator=g).float().to(dev)
    x_gen = torch.rand(N // 2, 30, generator=g).float().to(dev)
    y_gen = Y[: N // 2].contiguous()
    pp = torch.rand(N // 2, 30, generator=g).float().to(dev)
    po = Y[N // 2 :].contiguous()
    for _ in range(5):
        _hipops.nsga2_select(x_gen, y_gen, pp, po, N // 2)
    torch.cuda.synchronize()
    # submit 50 calls, measure host time WITHOUT sync, then with
    t0 = time.perf_counter()
    for _ in range(50):
        _hipops.nsga2_select(x_gen, y_gen, pp, po, N // 2)
    t1 = time.perf_counter()
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    print(f"N={N}: host submit {1e6*(t1-t0)/50:.1f} us/call, gpu drain adds {1e6*(t2-t1)/50:.1f} us/call")
    # pareto_rank alone
    t0 = time.perf_counter()
    for _ in range(50):
        _hipops.pareto_rank(Y, N // 2)
    t1 = time.perf_counter()
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    print(f"   pareto_rank: submit {1e6*(t1-t0)/50:.1f} us, drain {1e6*(t2-t1)/50:.1f} us")
