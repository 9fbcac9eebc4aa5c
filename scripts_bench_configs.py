"""Measure BASELINE configs #3 (ZDT3 AGEMOEA pop=1024) and #5 (TNK CMA-ES
pop=4096) epoch components on the current device. Results -> stdout
(recorded in profiles/CONFIGS.md)."""

import os
import sys
import time

import numpy as np
import torch

torch.set_num_threads(min(8, os.cpu_count() or 8))

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from dmosopt_amd.benchmarks.problems import tnk, zdt3

dev = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
print(f"device: {dev}")


def time_gens(opt, obj_fn, n_gens, warmup=2):
    for _ in range(warmup):
        xg, st = opt.generate()
        yg = obj_fn(xg)
        opt.update(xg, yg, st)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n_gens):
        xg, st = opt.generate()
        yg = obj_fn(xg)
        opt.update(xg, yg, st)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n_gens * 1e3


# ---------------- config 3: ZDT3 d=30 AGEMOEA pop=1024 --------------------
from dmosopt_amd.moea.agemoea import AGEMOEAOptimizer

rng = np.random.default_rng(0)
d, pop = 30, 1024
opt = AGEMOEAOptimizer(popsize=pop, nInput=d, nOutput=2, model=None)
if dev.type == "cuda":
    opt.set_device(dev)
bounds = np.stack([np.zeros(d), np.ones(d)], axis=1)
x0 = opt.generate_initial(bounds, rng)
y0 = zdt3(x0).numpy()
opt.initialize_strategy(x0, y0, bounds, rng)


def zdt3_dev(x):
    return zdt3(x.double().cpu()).to(x.device).to(x.dtype) if isinstance(x, torch.Tensor) else zdt3(x)


ms = time_gens(opt, zdt3_dev, 10)
print(f"config#3 AGEMOEA ZDT3 d=30 pop=1024: {ms:.1f} ms/gen")

# ---------------- config 5: TNK d=2 CMA-ES pop=4096 -----------------------
from dmosopt_amd.moea.cmaes import CMAESOptimizer

rng = np.random.default_rng(1)
d, pop = 2, 4096
opt = CMAESOptimizer(popsize=pop, nInput=d, nOutput=2, model=None)
if dev.type == "cuda":
    opt.set_device(dev)
bounds = np.stack([np.full(d, 1e-9), np.full(d, np.pi)], axis=1)
x0 = opt.generate_initial(bounds, rng)
f0, c0 = tnk(x0)


def tnk_f(x):
    f, c = tnk(x.double().cpu() if isinstance(x, torch.Tensor) else x)
    return f.to(x.device).to(x.dtype) if isinstance(x, torch.Tensor) else f


opt.initialize_strategy(x0, f0.numpy(), bounds, rng)
ms = time_gens(opt, tnk_f, 5)
print(f"config#5 CMA-ES TNK d=2 pop=4096: {ms:.1f} ms/gen (incl. batched rank-1 Cholesky updates)")
