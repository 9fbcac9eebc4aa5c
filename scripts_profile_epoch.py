"""Profile one bench epoch on GPU: cProfile (host-side) breakdown."""
import cProfile, pstats, io, sys, os, time
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from bench import make_archive, one_epoch

dev = torch.device("cuda", 0)
X, Y = make_archive(seed=1)
# warmup
one_epoch(X, Y, pop=200, rank=0, world=1, device=dev, seed=1, n_gen=5)
torch.cuda.synchronize()
pr = cProfile.Profile()
t0 = time.time()
pr.enable()
one_epoch(X, Y, pop=200, rank=0, world=1, device=dev, seed=2, n_gen=50)
pr.disable()
torch.cuda.synchronize()
print(f"epoch(50 gens) wall: {time.time()-t0:.2f}s")
s = io.StringIO()
ps = pstats.Stats(pr, stream=s).sort_stats("cumulative")
ps.print_stats(35)
ps.print_callers("as_tensor")
ps.print_callers("'cpu'")
print(s.getvalue())
