"""dmosopt-analyze: inspect an optimization results HDF5 file.

Role parity with reference dmosopt_analyze.py:30-198: load the file,
compute the Pareto-best set per problem, optionally select the k nearest
neighbors to the objective-space origin (cKDTree), sort by the requested
keys, print and/or export.
"""

from __future__ import annotations

import argparse
import sys

import numpy as np


def main(argv=None):
    ap = argparse.ArgumentParser(prog="dmosopt-analyze")
    ap.add_argument("--file-path", "-p", required=True)
    ap.add_argument("--opt-id", required=True)
    ap.add_argument("--sort-key", action="append", default=[])
    ap.add_argument("--knn", type=int, default=0)
    ap.add_argument("--filter-objectives", type=str, default=None)
    ap.add_argument("--output-file", type=str, default=None)
    ap.add_argument("--no-constraints", action="store_true")
    ap.add_argument("--verbose", "-v", action="store_true")
    args = ap.parse_args(argv)

    from dmosopt_amd.core.engine import get_best
    from dmosopt_amd.storage import h5 as h5store

    (
        _seed, _max_epoch, old_evals, param_space, objective_names,
        feature_names, constraint_names, _pp, problem_ids,
    ) = h5store.init_from_h5(args.file_path, None, args.opt_id, None)

    if problem_ids is None:
        problem_ids = [0]

    obj_names = list(objective_names)
    if args.filter_objectives:
        keep = args.filter_objectives.split(",")
        keep_idx = [obj_names.index(k) for k in keep]
        obj_names = keep
    else:
        keep_idx = list(range(len(obj_names)))

    out = {}
    for pid in sorted(problem_ids):
        evals = old_evals[pid]
        x = np.vstack([e.parameters for e in evals])
        y = np.vstack([e.objectives for e in evals])[:, keep_idx]
        c = None
        if constraint_names is not None and not args.no_constraints:
            c = np.vstack([e.constraints for e in evals])
        epochs = np.asarray([e.epoch if e.epoch is not None else 0 for e in evals])

        best_x, best_y, _bf, best_c, best_epoch, _ = get_best(
            x, y, None, c, x.shape[1], y.shape[1], epochs=epochs, feasible=True
        )

        if args.knn > 0 and len(best_y) > args.knn:
            from scipy.spatial import cKDTree

            tree = cKDTree(best_y)
            _, nn = tree.query(np.zeros(best_y.shape[1]), k=args.knn)
            nn = np.atleast_1d(nn)
            best_x, best_y = best_x[nn], best_y[nn]
            if best_epoch is not None:
                best_epoch = best_epoch[nn]

        if args.sort_key:
            keys = []
            for k in reversed(args.sort_key):
                keys.append(best_y[:, obj_names.index(k)])
            order = np.lexsort(tuple(keys))
            best_x, best_y = best_x[order], best_y[order]
            if best_epoch is not None:
                best_epoch = best_epoch[order]

        out[pid] = (best_x, best_y, best_epoch)
        print(f"problem {pid}: {len(best_x)} non-dominated solutions")
        for i in range(len(best_x)):
            objs = dict(zip(obj_names, np.round(best_y[i], 6)))
            print(f"  [{i}] {objs}")
            if args.verbose:
                prms = dict(zip(param_space.parameter_names, np.round(best_x[i], 6)))
                print(f"      {prms}")

    if args.output_file:
        np.savez(
            args.output_file,
            **{
                f"best_x_{pid}": v[0] for pid, v in out.items()
            },
            **{f"best_y_{pid}": v[1] for pid, v in out.items()},
        )
        print(f"wrote {args.output_file}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
