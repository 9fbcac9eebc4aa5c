"""Checkpoint/resume example: run 2 epochs, stop, resume for 2 more.

The HDF5 results file doubles as the checkpoint (reference parity:
init_from_h5 / save_to_h5): re-running with resume=True restores the
archive, random seed state, parameter space and epoch counter and
continues where the first run stopped.

Usage: python examples/example_resume.py
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import dmosopt_amd


def zdt1_objective(pp):
    names = sorted(pp.keys())
    x = np.array([pp[k] for k in names])
    f1 = x[0]
    g = 1.0 + 9.0 * np.mean(x[1:])
    return np.array([f1, g * (1.0 - np.sqrt(f1 / g))])


def make_params(file_path, resume):
    return {
        "opt_id": "zdt1_resume",
        "obj_fun": zdt1_objective,
        "problem_parameters": {},
        "space": {f"x{i:02d}": [0.0, 1.0] for i in range(10)},
        "objective_names": ["f1", "f2"],
        "population_size": 100,
        "num_generations": 40,
        "n_initial": 5,
        "initial_maxiter": 3,
        "n_epochs": 2,
        "surrogate_method_name": "gpr",
        "surrogate_method_kwargs": {"anisotropic": False, "optimizer": "sceua"},
        "optimizer": "nsga2",
        "random_seed": 37,
        "file_path": file_path,
        "save": True,
        "resume": resume,
    }


def main():
    fp = "/tmp/zdt1_resume.h5"
    if os.path.exists(fp):
        os.remove(fp)

    print("=== first run: 2 epochs ===")
    dmosopt_amd.run(make_params(fp, resume=False), verbose=True)
    dmosopt_amd.sopt_dict.clear()

    print("=== resumed run: 2 more epochs from the checkpoint ===")
    best = dmosopt_amd.run(make_params(fp, resume=True), verbose=True)
    bestx, besty = best
    y = np.column_stack([v for _, v in besty])
    print(f"non-dominated solutions after resume: {y.shape[0]}")
    print(f"best f1 range: [{y[:, 0].min():.4f}, {y[:, 0].max():.4f}]")


if __name__ == "__main__":
    main()
