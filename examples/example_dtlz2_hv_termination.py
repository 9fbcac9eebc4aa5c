"""DTLZ2 with 5 objectives + hypervolume-progress termination.

The problem family of BASELINE config #4 (DTLZ2, many objectives). At five
objectives the exact hypervolume goes through the Lacour box decomposition;
termination uses the multi-fidelity HV-progress machinery (coarse/medium/
fine precision schedule + convergence detector).

Usage: python examples/example_dtlz2_hv_termination.py
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import dmosopt_amd
from dmosopt_amd.benchmarks.problems import dtlz2

D, M = 12, 5


def objective(pp):
    names = sorted(pp.keys())
    x = np.array([pp[k] for k in names])
    return dtlz2(x[None, :], n_obj=M).numpy()[0]


def main():
    params = {
        "opt_id": "dtlz2_hv",
        "obj_fun": objective,
        "problem_parameters": {},
        "space": {f"x{i:02d}": [0.0, 1.0] for i in range(D)},
        "objective_names": [f"f{j}" for j in range(M)],
        "population_size": 80,
        "num_generations": 60,
        "n_initial": 4,
        "initial_maxiter": 3,
        "n_epochs": 2,
        "surrogate_method_name": "gpr",
        "surrogate_method_kwargs": {"anisotropic": False, "optimizer": "sceua"},
        "optimizer": "nsga2",
        "termination_conditions": True,  # adaptive termination preset
        "random_seed": 17,
    }
    best = dmosopt_amd.run(params, verbose=True)
    bx, by = best
    y = np.column_stack([v for _, v in by])

    from dmosopt_amd.hv.adaptive import AdaptiveHyperVolume

    hv = AdaptiveHyperVolume(np.full(M, 2.0)).compute(y)
    print(f"{y.shape[0]} non-dominated solutions, hypervolume (ref=2^5): {hv:.3f}")
    # DTLZ2's front is the unit hypersphere octant: ||f|| = 1 on the front
    norms = np.linalg.norm(y, axis=1)
    print(f"front radius: median {np.median(norms):.3f} (ideal 1.0)")


if __name__ == "__main__":
    main()
