"""TNK constrained problem with MO-CMA-ES + logistic feasibility model."""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import dmosopt_amd


def obj_fun(pp):
    x1, x2 = pp["x1"], pp["x2"]
    f = np.array([x1, x2])
    c1 = x1**2 + x2**2 - 1.0 - 0.1 * np.cos(16.0 * np.arctan2(x1, max(x2, 1e-30)))
    c2 = 0.5 - (x1 - 0.5) ** 2 - (x2 - 0.5) ** 2
    return f, np.array([c1, c2])


if __name__ == "__main__":
    params = {
        "opt_id": "example_tnk",
        "obj_fun_name": "example_tnk_constrained.obj_fun",
        "problem_parameters": {},
        "space": {"x1": [1e-9, np.pi], "x2": [1e-9, np.pi]},
        "objective_names": ["f1", "f2"],
        "constraint_names": ["c1", "c2"],
        "population_size": 100,
        "num_generations": 50,
        "optimizer": "cmaes",
        "feasibility_method_name": "logreg",
        "n_initial": 10,
        "n_epochs": 3,
    }
    best = dmosopt_amd.run(params, verbose=True)
    if best is not None:
        bestx, besty = best
        print(f"{len(besty[0][1])} feasible non-dominated solutions")
