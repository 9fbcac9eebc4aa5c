"""Host-side per-phase timing of the NSGA2 generation loop (pop=200).

The gen loop is host-dispatch-bound (rocprof: GPU 33% busy in the loop
window), so host perf_counter deltas around each phase identify where the
wall time goes even though the GPU runs async.

python scripts_gen_host_split.py  (GPU box)
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from bench import D_IN, N_OBJ, make_archive
from dmosopt_amd.core import engine
from dmosopt_amd.models.model import Model
from dmosopt_amd.moea.nsga2 import NSGA2Optimizer

dev = torch.device("cuda", 0)
X, Y = make_archive(seed=1)
gp = engine.train(
    D_IN, N_OBJ, np.zeros(D_IN), np.ones(D_IN), X, Y, None,
    surrogate_method_name="gpr",
    surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua",
                             "seed": 7},
    logger=None, device=dev)
mdl = Model(objective=gp)

POP = int(os.environ.get("POP", "200"))
opt = NSGA2Optimizer(popsize=POP, nInput=D_IN, nOutput=N_OBJ, model=mdl,
                     distance_metric="crowding", sampling_method="slh",
                     mutation_rate=None, nchildren=1)
opt.set_device(dev)
rng = np.random.default_rng(3)
bounds = np.column_stack([np.zeros(D_IN), np.ones(D_IN)])
opt.initialize_strategy(X.astype(np.float32), Y.astype(np.float32), bounds, rng)

t = {"generate": 0.0, "evaluate": 0.0, "update": 0.0}
N_GEN = 300
# warmup
for _ in range(30):
    x_gen, gs = opt.generate()
    y_gen = engine._surrogate_eval(mdl, x_gen, False)
    opt.update(x_gen, y_gen, gs)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(N_GEN):
    a = time.perf_counter()
    x_gen, gs = opt.generate()
    b = time.perf_counter()
    y_gen = engine._surrogate_eval(mdl, x_gen, False)
    c = time.perf_counter()
    opt.update(x_gen, y_gen, gs)
    d = time.perf_counter()
    t["generate"] += b - a
    t["evaluate"] += c - b
    t["update"] += d - c
torch.cuda.synchronize()
wall = time.perf_counter() - t0
print(f"pop={POP} {N_GEN} gens: wall {1e6*wall/N_GEN:.0f} us/gen")
for k, v in t.items():
    print(f"  {k:9s}: {1e6*v/N_GEN:7.1f} us/gen (host)")
print(f"  sync tail : {1e6*(wall - sum(t.values()))/N_GEN:7.1f} us/gen")
