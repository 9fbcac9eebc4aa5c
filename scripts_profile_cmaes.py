import cProfile, io, os, pstats, sys, time
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np
import torch
torch.set_num_threads(min(8, os.cpu_count() or 8))
from dmosopt_amd.moea.cmaes import CMAESOptimizer
from dmosopt_amd.benchmarks.problems import tnk

dev = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
rng = np.random.default_rng(1)
d, pop = 2, 4096
opt = CMAESOptimizer(popsize=pop, nInput=d, nOutput=2, model=None)
if dev.type == "cuda":
    opt.set_device(dev)
bounds = np.stack([np.full(d, 1e-9), np.full(d, np.pi)], axis=1)
x0 = opt.generate_initial(bounds, rng)
f0, c0 = tnk(x0)
opt.initialize_strategy(x0, f0.numpy(), bounds, rng)
def tnk_f(x):
    f, c = tnk(x.double().cpu()); return f.to(x.device).to(x.dtype)
for _ in range(2):
    xg, st = opt.generate(); opt.update(xg, tnk_f(xg), st)
pr = cProfile.Profile(); pr.enable()
for _ in range(3):
    xg, st = opt.generate(); opt.update(xg, tnk_f(xg), st)
pr.disable()
s = io.StringIO(); pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(18)
print(s.getvalue())
