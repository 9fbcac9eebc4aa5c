"""dmosopt-train: offline surrogate fit from a results HDF5 file.

Implements the INTENT of the reference's dmosopt_train.py (which is stale
against its own API — SURVEY.md section 2.8): load the evaluation archive,
fit the requested surrogate, report in-sample accuracy, optionally save
the surrogate predictions at the evaluated points.
"""

from __future__ import annotations

import argparse
import sys

import numpy as np


def main(argv=None):
    ap = argparse.ArgumentParser(prog="dmosopt-train")
    ap.add_argument("--file-path", "-p", required=True)
    ap.add_argument("--opt-id", required=True)
    ap.add_argument("--surrogate-method", default="gpr")
    ap.add_argument("--surrogate-options", default="{}",
                    help="JSON dict of surrogate kwargs")
    ap.add_argument("--output-file", default=None)
    ap.add_argument("--verbose", "-v", action="store_true")
    args = ap.parse_args(argv)

    import json

    from dmosopt_amd.core import engine
    from dmosopt_amd.storage import h5 as h5store

    (
        _seed, _max_epoch, old_evals, param_space, objective_names,
        _features, constraint_names, _pp, problem_ids,
    ) = h5store.init_from_h5(args.file_path, None, args.opt_id, None)
    if problem_ids is None:
        problem_ids = [0]

    kwargs = json.loads(args.surrogate_options)
    for pid in sorted(problem_ids):
        evals = old_evals[pid]
        x = np.vstack([e.parameters for e in evals])
        y = np.vstack([e.objectives for e in evals])
        c = (
            np.vstack([e.constraints for e in evals])
            if constraint_names is not None
            else None
        )
        sm = engine.train(
            x.shape[1], y.shape[1], param_space.bound1, param_space.bound2,
            x, y, c,
            surrogate_method_name=args.surrogate_method,
            surrogate_method_kwargs=kwargs,
        )
        mean, var = sm.predict(x)
        mae = np.mean(np.abs(mean - y), axis=0)
        print(f"problem {pid}: surrogate {args.surrogate_method} fit on "
              f"{x.shape[0]} evals; in-sample MAE per objective: {np.round(mae, 6)}")
        if args.output_file:
            np.savez(args.output_file, x=x, y=y, mean=mean, var=var)
            print(f"wrote {args.output_file}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
