"""GPU integration soak: a realistic multi-epoch run() with surrogate,
constraints, HV termination, save + resume — end-to-end through the
public API on cuda, twice (fresh + resumed)."""
import os, sys, tempfile, time
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np
import dmosopt_amd
from dmosopt_amd.benchmarks import problems as bp

def obj(pp):
    x = np.array([pp[f"x{i+1}"] for i in range(8)])
    y = bp.zdt3(x).numpy()[0]
    c = np.array([x[0] - 0.02])
    return y, c

fp = os.path.join(tempfile.gettempdir(), "soak.h5")
if os.path.exists(fp):
    os.remove(fp)
base = {
    "opt_id": "soak", "obj_fun": obj, "problem_parameters": {},
    "space": {f"x{i+1}": [0.0, 1.0] for i in range(8)},
    "objective_names": ["y1", "y2"], "constraint_names": ["c1"],
    "population_size": 200, "num_generations": 100,
    "optimizer": ["nsga2", "age"], "feasibility_method_name": "logreg",
    "termination_conditions": {"strategy": "simple", "min_generations": 20},
    "n_initial": 3, "n_epochs": 3, "random_seed": 77,
    "save": True, "file_path": fp, "save_surrogate_evals": True,
}
t0 = time.time()
best = dmosopt_amd.run(dict(base), verbose=False)
t1 = time.time()
bestx, besty = best
y = np.column_stack([v for _, v in besty])
print(f"fresh run: {t1-t0:.1f}s, best front {y.shape}, finite={np.isfinite(y).all()}")
# resume for 2 more epochs
t0 = time.time()
best2 = dmosopt_amd.run(dict(base, n_epochs=5), verbose=False)
t1 = time.time()
y2 = np.column_stack([v for _, v in best2[1]])
print(f"resumed run: {t1-t0:.1f}s, best front {y2.shape}, finite={np.isfinite(y2).all()}")
assert y2.shape[0] >= 1
from dmosopt_amd.hv.exact import hv_2d
print("front HV(11,11):", round(hv_2d(y2, np.array([11.0, 11.0])), 3))
print("SOAK OK")
