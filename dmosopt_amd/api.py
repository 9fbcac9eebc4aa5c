"""Public API: dmosopt_amd.run(dopt_params) (reference dmosopt.py:2441-2596).

Multi-rank execution (one process per GPU over RCCL, or gloo on CPU) is
REPLICATED CONTROL FLOW: every rank runs the full driver/epoch loop with
identical seeds, while objective results, surrogate predictions and GP
hyperparameters move as tensor collectives (parallel/context.py). Rank 0
alone writes the H5 file, logs, and returns the best set; other ranks
return None (API parity with the reference's MPI workers). Single-process
runs use a local farm — no process group needed.
"""

from __future__ import annotations

import logging
from typing import Dict, Optional

import numpy as np

from dmosopt_amd.config import import_object_by_path
from dmosopt_amd.core.driver import DistOptimizer
from dmosopt_amd.parallel import comm

dopt_dict: Dict[str, DistOptimizer] = {}


def _first_result_reduce(xs):
    return xs[0]


def _resolve_obj_fun(dopt_params, worker=None):
    objfun = dopt_params.get("obj_fun", None)
    if objfun is not None and callable(objfun):
        return objfun
    objfun_name = dopt_params.get("obj_fun_name", None)
    if objfun_name is not None:
        return import_object_by_path(objfun_name)
    objfun_init_name = dopt_params.get("obj_fun_init_name", None)
    objfun_init_args = dopt_params.get("obj_fun_init_args", None) or {}
    if objfun_init_name is None:
        raise RuntimeError("dmosopt_amd.run: objective function not provided")
    objfun_init = import_object_by_path(objfun_init_name)
    return objfun_init(**objfun_init_args, worker=worker)


def run(
    dopt_params,
    time_limit=None,
    feasible=True,
    return_features=False,
    return_constraints=False,
    verbose=True,
    **_compat_kwargs,
):
    """Run a distributed optimization; returns best (params, objectives) on
    rank 0 and None on worker ranks (reference run(), dmosopt.py:2526)."""
    import dmosopt_amd

    dopt_params = dict(dopt_params)
    rank, world = comm.init_from_env()
    opt_id = dopt_params["opt_id"]
    logger = logging.getLogger(opt_id)
    if verbose:
        logging.basicConfig(level=logging.INFO)
        logger.setLevel(logging.INFO)

    objfun = _resolve_obj_fun(dopt_params, worker=rank if rank > 0 else None)
    params = dict(dopt_params)
    params["obj_fun"] = objfun
    if "optimizer" in params and "optimizer_name" not in params:
        params["optimizer_name"] = params.pop("optimizer")
    reducefun_name = params.pop("reduce_fun_name", None)
    if reducefun_name is not None:
        params["reduce_fun"] = import_object_by_path(reducefun_name)
    for k in ("obj_fun_name", "obj_fun_init_name", "obj_fun_init_args",
              "broker_fun_name", "broker_module_name"):
        params.pop(k, None)
    ctrl_init_fun_name = params.pop("controller_init_fun_name", None)
    ctrl_init_fun_args = params.pop("controller_init_fun_args", {})

    # Replicated control flow (parallel/context.py): EVERY rank builds the
    # full driver with identical seeds and runs the same epoch loop; the
    # data plane (objective results, surrogate predictions, GP
    # hyperparameters) moves as tensor collectives. Rank 0 alone touches
    # the H5 file for writing, logs, and returns the best set (API parity:
    # non-root ranks return None like the reference's workers).
    from dmosopt_amd.parallel.context import get_context

    ctx = get_context()
    if rank == 0 and ctrl_init_fun_name is not None:
        import_object_by_path(ctrl_init_fun_name)(**ctrl_init_fun_args)
    if rank != 0:
        params["save"] = False  # restore-only: read the H5, never write it
    if (
        world > 1
        and params.get("random_seed") is None
        and params.get("local_random") is None
    ):
        # replicated control flow requires a shared seed: an unseeded run
        # would silently diverge, so rank 0 draws one and broadcasts it
        seed = int(np.random.default_rng().integers(2**31)) if rank == 0 else 0
        params["random_seed"] = ctx.bcast_int(seed, src=0)
    dopt = DistOptimizer(**params, verbose=verbose and rank == 0)
    if world > 1:
        spec = comm.ResultSpec(
            problem_ids=tuple(dopt.problem_ids),
            n_objectives=len(dopt.objective_names),
            n_constraints=len(dopt.constraint_names) if dopt.constraint_names else 0,
            feature_dtype=(
                np.dtype(dopt.feature_dtypes) if dopt.feature_dtypes else None
            ),
        )
        farm = comm.make_farm({opt_id: dopt.eval_fun}, spec)
    else:
        farm = comm.make_farm({opt_id: dopt.eval_fun})
    dopt.farm = farm
    dopt.initialize_strategy()
    dopt_dict[opt_id] = dopt
    dmosopt_amd.sopt_dict[opt_id] = dopt
    if rank == 0:
        logger.info(f"Optimizing for {dopt.n_epochs} epochs...")
    import time as _time

    t_start = _time.time()
    try:
        if dopt.n_epochs <= 0:
            dopt.run_epoch(completed_epoch=True)
        else:
            while dopt.epoch_count < dopt.n_epochs:
                # rank 0 owns the wall-clock decision; replicated ranks
                # must agree, so the stop flag is broadcast
                over = time_limit is not None and (_time.time() - t_start) > time_limit
                if ctx is not None and ctx.world > 1:
                    over = ctx.bcast_flag(over, src=0)
                if over:
                    if rank == 0:
                        logger.info("Time limit reached; stopping.")
                    break
                dopt.run_epoch()
    finally:
        farm.shutdown()
    if rank != 0:
        return None
    dopt.print_best()
    return dopt.get_best(
        feasible=feasible,
        return_features=return_features,
        return_constraints=return_constraints,
    )
