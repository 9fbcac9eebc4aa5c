"""Phase split of the headline epoch: GP fit vs generation loop vs resample.

python scripts_phase_split.py  (GPU box)
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch

from bench import ARCHIVE_N, D_IN, N_OBJ, make_archive
from dmosopt_amd.core import engine
from dmosopt_amd.models.model import Model

dev = torch.device("cuda", 0)
X, Y = make_archive(seed=1)


def fit(seed):
    return engine.train(
        D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN), X, Y, None,
        surrogate_method_name="gpr",
        surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua",
                                 "seed": seed},
        logger=None, device=dev)


def genloop(gp, seed, n_gen=200):
    from dmosopt_amd.moea.nsga2 import NSGA2Optimizer

    mdl = Model(objective=gp)
    optimizer = NSGA2Optimizer(
        popsize=200, nInput=D_IN, nOutput=N_OBJ, model=mdl,
        distance_metric="crowding", sampling_method="slh", mutation_rate=None,
        nchildren=1)
    optimizer.set_device(dev)
    res = engine.optimize_loop(
        n_gen, optimizer, mdl, D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN),
        popsize=200, initial=(X.astype(np.float32), Y.astype(np.float32)),
        local_random=np.random.default_rng(seed))
    return res


# warmup
gp = fit(99)
genloop(gp, 99, n_gen=10)
torch.cuda.synchronize()

for label, fn in (
    ("gp_fit", lambda s: fit(s)),
    ("gen200", lambda s: genloop(gp, s)),
):
    ts = []
    for rep in range(4):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn(200 + rep)
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) * 1e3)
    print(f"{label}: {min(ts):.1f} ms (reps {['%.1f' % t for t in ts]})")

# NMLL graph on/off for the fit
os.environ["DMOSOPT_NMLL_GRAPH"] = "0"
from dmosopt_amd.models import gp_core

gp_core._nmll_graphs.clear()
ts = []
for rep in range(4):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    fit(300 + rep)
    torch.cuda.synchronize()
    ts.append((time.perf_counter() - t0) * 1e3)
print(f"gp_fit nograph: {min(ts):.1f} ms")

# ---- fine-grained epoch split (appended round 2) ----
import time as _t

def fine_split(seed=31):
    from dmosopt_amd.moea.nsga2 import NSGA2Optimizer
    from dmosopt_amd import ops
    from dmosopt_amd.hv.exact import hv_2d
    from dmosopt_amd.benchmarks.problems import zdt1
    import torch as _torch

    t0 = _t.perf_counter()
    gp = fit(seed)
    _torch.cuda.synchronize(); t1 = _t.perf_counter()
    mdl = Model(objective=gp)
    opt = NSGA2Optimizer(popsize=200, nInput=D_IN, nOutput=N_OBJ, model=mdl,
                         distance_metric="crowding", sampling_method="slh",
                         mutation_rate=None, nchildren=1)
    opt.set_device(dev)
    local_random = np.random.default_rng(seed + 1)
    bounds = np.column_stack((np.zeros(D_IN), np.ones(D_IN)))
    x0 = opt.generate_initial(bounds, local_random)
    y0 = engine._surrogate_eval(mdl, x0, False)
    x0 = np.asarray(engine._to_np(x0), dtype=np.float32)
    y0 = np.asarray(engine._to_np(y0), dtype=np.float32)
    x0 = np.vstack((X.astype(np.float32), x0))
    y0 = np.vstack((Y.astype(np.float32), y0))
    opt.initialize_strategy(x0, y0, bounds, local_random)
    _torch.cuda.synchronize(); t2 = _t.perf_counter()
    for _ in range(200):
        xg, gs = opt.generate()
        yg = engine._surrogate_eval(mdl, xg, False)
        opt.update(xg, yg, gs)
    _torch.cuda.synchronize(); t3 = _t.perf_counter()
    bx, by = opt.population_objectives
    best_x = _torch.as_tensor(bx, dtype=_torch.float32, device=dev)
    best_y = _torch.as_tensor(by, dtype=_torch.float32, device=dev)
    Dc = ops.crowding_distance(best_y)
    idx = _torch.argsort(Dc, descending=True)[:50]
    y_res = zdt1(best_x[idx].double())
    hv = hv_2d(y_res, np.array([11.0, 11.0]))
    _torch.cuda.synchronize(); t4 = _t.perf_counter()
    print(f"fit {1e3*(t1-t0):6.1f}  init {1e3*(t2-t1):6.1f}  gen200 {1e3*(t3-t2):6.1f}  resample {1e3*(t4-t3):6.1f}  total {1e3*(t4-t0):6.1f} ms")

if os.environ.get("FINE", "0") == "1":
    for rep in range(4):
        fine_split(seed=40 + rep)
