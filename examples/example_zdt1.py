"""ZDT1 d=30 with NSGA-II + GP surrogate (the canonical README config)."""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import dmosopt_amd


def obj_fun(pp):
    x = np.array([pp[f"x{i + 1}"] for i in range(30)])
    f1 = x[0]
    g = 1 + 9 * x[1:].mean()
    return np.array([f1, g * (1 - np.sqrt(f1 / g))])


if __name__ == "__main__":
    params = {
        "opt_id": "example_zdt1",
        "obj_fun_name": "example_zdt1.obj_fun",
        "problem_parameters": {},
        "space": {f"x{i + 1}": [0.0, 1.0] for i in range(30)},
        "objective_names": ["y1", "y2"],
        "population_size": 200,
        "num_generations": 200,
        "initial_maxiter": 10,
        "optimizer": "nsga2",
        "termination_conditions": True,
        "n_initial": 3,
        "n_epochs": 2,
    }
    best = dmosopt_amd.run(params, verbose=True)
    if best is not None:
        bestx, besty = best
        y = np.column_stack([v for _, v in besty])
        print(f"{y.shape[0]} non-dominated solutions; f1 in "
              f"[{y[:, 0].min():.3f}, {y[:, 0].max():.3f}]")
