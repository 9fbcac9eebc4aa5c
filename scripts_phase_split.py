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
