"""Same-box A/B: full cooperative peel vs early-stop (nsga2_select route).

Builds a REAL evolved population (front structure matters: random clouds
have ~10 fronts, converged NSGA2 merges have ~100+), then CUDA-event times
native.pareto_rank(Y) vs native.pareto_rank(Y, N//2).

python scripts_peel_stop_ab.py  (GPU box)
"""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from bench import D_IN, N_OBJ, make_archive
from dmosopt_amd.core import engine
from dmosopt_amd.models.model import Model
from dmosopt_amd.moea.nsga2 import NSGA2Optimizer
from dmosopt_amd.ops import _load_native, pareto_rank

native = _load_native()
dev = torch.device("cuda", 0)
X, Y = make_archive(seed=1)

for pop in (1600, 3200):
    gp = engine.train(
        D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN), X, Y, None,
        surrogate_method_name="gpr",
        surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua",
                                 "seed": 7},
        logger=None, device=dev)
    mdl = Model(objective=gp)
    opt = NSGA2Optimizer(popsize=pop, nInput=D_IN, nOutput=N_OBJ, model=mdl,
                         distance_metric="crowding", sampling_method="slh",
                         mutation_rate=None, nchildren=1)
    opt.set_device(dev)
    res = engine.optimize_loop(
        120, opt, mdl, D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN),
        popsize=pop, initial=(X.astype(np.float32), Y.astype(np.float32)),
        local_random=np.random.default_rng(3))
    # merged parent+child objective block, like gen-steady-state selection
    yb = torch.as_tensor(res.best_y, dtype=torch.float32, device=dev)
    merged = torch.cat([yb, yb + 0.01 * torch.randn_like(yb)], 0)
    N = merged.shape[0]
    nf = int(pareto_rank(merged).max()) + 1
    for tag, stop in (("full", -1), (f"stop={N//2}", N // 2)):
        for _ in range(3):
            native.pareto_rank(merged, stop)
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(True), torch.cuda.Event(True)
        s.record()
        for _ in range(20):
            native.pareto_rank(merged, stop)
        e.record()
        torch.cuda.synchronize()
        print(f"pop={pop} N={N} fronts={nf} {tag:10s}: "
              f"{s.elapsed_time(e) / 20 * 1000:8.1f} us")
