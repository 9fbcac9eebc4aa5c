"""torch.profiler over a few pop=1600 generations: identify the stray
(3200,*)-shaped elementwise/copy ops by op name + input shape.

POP=1600 python scripts_torchprof_gen.py  (GPU box)
"""
import os, sys
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
    surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua", "seed": 7},
    logger=None, device=dev)
mdl = Model(objective=gp)
POP = int(os.environ.get("POP", "1600"))
opt = NSGA2Optimizer(popsize=POP, nInput=D_IN, nOutput=N_OBJ, model=mdl,
                     distance_metric="crowding", sampling_method="slh",
                     mutation_rate=None, nchildren=1)
opt.set_device(dev)
rng = np.random.default_rng(3)
bounds = np.column_stack([np.zeros(D_IN), np.ones(D_IN)])
opt.initialize_strategy(X.astype(np.float32), Y.astype(np.float32), bounds, rng)
for _ in range(20):
    xg, gs = opt.generate()
    yg = engine._surrogate_eval(mdl, xg, False)
    opt.update(xg, yg, gs)
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True, with_stack=True) as prof:
    for _ in range(20):
        xg, gs = opt.generate()
        yg = engine._surrogate_eval(mdl, xg, False)
        opt.update(xg, yg, gs)
    torch.cuda.synchronize()
ka = prof.key_averages(group_by_stack_n=6)
rows = [e for e in ka if e.key in ("aten::copy_", "aten::zeros", "aten::zero_", "aten::to", "aten::contiguous", "aten::fill_") or "copy_" in e.key]
rows.sort(key=lambda e: -e.device_time_total)
for e in rows[:8]:
    print(f"== {e.key}  cuda_total={e.device_time_total/1e3:.2f}ms calls={e.count}")
    for ln in (e.stack or [])[:6]:
        print("   ", ln.strip()[:140])
