"""dmosopt-onestep: one offline MO-ASMO resample step from a results file.

Implements the INTENT of the reference's dmosopt_onestep.py (which imports
a non-existent MOASMO.onestep — SURVEY.md section 2.8): load the archive,
run ONE epoch of surrogate fit + inner MOEA, and emit the resample batch
that would be evaluated next.
"""

from __future__ import annotations

import argparse
import sys

import numpy as np


def main(argv=None):
    ap = argparse.ArgumentParser(prog="dmosopt-onestep")
    ap.add_argument("--file-path", "-p", required=True)
    ap.add_argument("--opt-id", required=True)
    ap.add_argument("--population-size", type=int, default=100)
    ap.add_argument("--num-generations", type=int, default=100)
    ap.add_argument("--resample-fraction", type=float, default=0.25)
    ap.add_argument("--optimizer", default="nsga2")
    ap.add_argument("--surrogate-method", default="gpr")
    ap.add_argument("--seed", type=int, default=None)
    ap.add_argument("--output-file", default=None)
    args = ap.parse_args(argv)

    from dmosopt_amd.core import engine
    from dmosopt_amd.storage import h5 as h5store

    (
        seed, _max_epoch, old_evals, param_space, objective_names,
        _features, constraint_names, _pp, problem_ids,
    ) = h5store.init_from_h5(args.file_path, None, args.opt_id, None)
    if problem_ids is None:
        problem_ids = [0]

    rng = np.random.default_rng(args.seed if args.seed is not None else seed)
    for pid in sorted(problem_ids):
        evals = old_evals[pid]
        x = np.vstack([e.parameters for e in evals])
        y = np.vstack([e.objectives for e in evals])
        c = (
            np.vstack([e.constraints for e in evals])
            if constraint_names is not None
            else None
        )
        result = engine.run_epoch(
            args.num_generations,
            param_space.parameter_names,
            objective_names,
            param_space.bound1,
            param_space.bound2,
            args.resample_fraction,
            x, y, c,
            pop=args.population_size,
            optimizer_name=args.optimizer,
            surrogate_method_name=args.surrogate_method,
            local_random=rng,
        )
        x_res = result["x_resample"]
        y_pred = result["y_pred"]
        print(f"problem {pid}: {x_res.shape[0]} resample candidates")
        for i in range(x_res.shape[0]):
            print(f"  [{i}] predicted {dict(zip(objective_names, np.round(y_pred[i], 6)))}")
        if args.output_file:
            np.savez(args.output_file, x_resample=x_res, y_pred=y_pred)
            print(f"wrote {args.output_file}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
