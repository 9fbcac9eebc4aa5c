"""MO-ASMO epoch engine.

Replaces the reference's generator/StopIteration protocol (MOASMO.py:21-470)
with explicit functions: ``optimize_loop`` runs the inner MOEA generations
(against the surrogate, or via an ``evaluator`` callback for the
surrogate-free path), ``run_epoch`` performs one full MO-ASMO epoch
(feasibility fit -> surrogate train -> sensitivity -> inner optimization ->
resample selection). Behavior parity is with MOASMO.py; the call protocol is
not kept (the generator pumping is a CPython idiom, the
behavior to keep is the epoch inputs/outputs).
"""

from __future__ import annotations

import time
from typing import Callable, Dict, Optional, Tuple

import numpy as np
import torch
from scipy import stats as scipy_stats

from dmosopt_amd import config as cfg
from dmosopt_amd import ops, sampling
from dmosopt_amd.datatypes import EpochResults, OptHistory
from dmosopt_amd.models.model import Model


def _to_np(a):
    if isinstance(a, torch.Tensor):
        return a.detach().cpu().numpy()
    return np.asarray(a)


def xinit(
    nEval,
    param_names,
    xlb,
    xub,
    nPrevious=None,
    method="glp",
    maxiter=5,
    local_random=None,
    logger=None,
):
    """Initial design: nEval * nInput points (reference MOASMO.py:134-193)."""
    nInput = len(param_names)
    Ninit = nInput * nEval
    if local_random is None:
        local_random = np.random.default_rng()
    nPrevious = nPrevious or 0
    if Ninit <= 0 or Ninit <= nPrevious:
        return None

    if isinstance(method, dict):
        Xinit = np.column_stack([method[k] for k in param_names])
        for i in range(Xinit.shape[1]):
            in_bounds = np.all((Xinit[:, i] <= xub[i]) & (Xinit[:, i] >= xlb[i]))
            if not in_bounds and logger is not None:
                logger.error(f"xinit: out of bounds values for {param_names[i]}")
            assert in_bounds
        return Xinit

    if logger is not None:
        logger.info(f"xinit: generating {Ninit} initial parameters...")
    if callable(method):
        Xinit = method(Ninit, nInput, local_random)
    else:
        fn = cfg.resolve(cfg.sampler_registry, method)
        Xinit = fn(Ninit, nInput, local_random=local_random, maxiter=maxiter)
    return Xinit[nPrevious:, :] * (xub - xlb) + xlb


def _surrogate_eval(mdl: Model, x, optimize_mean_variance: bool):
    if optimize_mean_variance:
        y_mean, y_var = mdl.objective.evaluate(_to_np(x))
        return np.column_stack((y_mean, np.round(y_var, 6)))
    # device-resident fast path: no numpy round trip per generation
    if isinstance(x, torch.Tensor) and hasattr(mdl.objective, "evaluate_tensor"):
        return mdl.objective.evaluate_tensor(x)
    return mdl.objective.evaluate(_to_np(x))


def optimize_loop(
    num_generations,
    optimizer,
    mdl: Model,
    nInput,
    nOutput,
    xlb,
    xub,
    popsize=100,
    initial=None,
    termination=None,
    local_random=None,
    logger=None,
    optimize_mean_variance=False,
    evaluator: Optional[Callable] = None,
    **kwargs,
) -> EpochResults:
    """Inner generation loop (reference MOASMO.optimize, MOASMO.py:21-131).

    With a surrogate, y comes from mdl.objective; without one, ``evaluator``
    (the distributed eval farm) is called with each generation's batch.
    """
    if local_random is None:
        local_random = np.random.default_rng()
    bounds = np.column_stack((xlb, xub))

    x = optimizer.generate_initial(bounds, local_random)
    if mdl.objective is None:
        y = evaluator(x)
    else:
        y = _surrogate_eval(mdl, x, optimize_mean_variance)
    x = np.asarray(_to_np(x), dtype=np.float32)
    y = np.asarray(_to_np(y), dtype=np.float32)

    if initial is not None:
        x_initial, y_initial = initial
        if x_initial is not None:
            x = np.vstack((np.asarray(x_initial, dtype=np.float32), x))
        if y_initial is not None:
            y = np.vstack((np.asarray(y_initial, dtype=np.float32), y))

    optimizer.initialize_strategy(x, y, bounds, local_random, **kwargs)
    if logger is not None:
        logger.info(f"{optimizer.name}: optimizer parameters are {repr(optimizer.opt_params)}")

    gen_indexes = [np.zeros((x.shape[0],), dtype=np.uint32)]
    x_new, y_new = [], []
    n_eval = 0
    it = range(1, num_generations + 1)
    if termination is not None:
        import itertools

        it = itertools.count(1)
    for i in it:
        if termination is not None:
            pop_x, pop_y = optimizer.population_objectives
            opt_hist = OptHistory(i, n_eval, _to_np(pop_x), _to_np(pop_y), None)
            if termination.has_terminated(opt_hist):
                break
        x_gen, gen_state = optimizer.generate()
        if mdl.objective is None:
            y_gen = evaluator(_to_np(x_gen))
        else:
            y_gen = _surrogate_eval(mdl, x_gen, optimize_mean_variance)
        optimizer.update(x_gen, y_gen, gen_state)
        n_eval += x_gen.shape[0]
        # archive ON DEVICE; the host copy (for H5) happens once after the
        # loop — a per-generation .cpu() here serializes every generation
        # against the GPU queue
        x_new.append(x_gen)
        y_new.append(y_gen)
        gen_indexes.append(np.full((x_gen.shape[0],), i, dtype=np.uint32))

    def _stack_np(first, parts):
        if parts and all(isinstance(p, torch.Tensor) for p in parts):
            return np.vstack([first, _to_np(torch.cat(parts, dim=0)).astype(first.dtype)])
        return np.vstack([first] + [np.asarray(_to_np(p), dtype=first.dtype) for p in parts])

    gen_index = np.concatenate(gen_indexes)
    x_all = _stack_np(x, x_new)
    y_all = _stack_np(y, y_new)
    bestx, besty = optimizer.population_objectives
    return EpochResults(_to_np(bestx), _to_np(besty), gen_index, x_all, y_all, optimizer)


def train(
    nInput,
    nOutput,
    xlb,
    xub,
    Xinit,
    Yinit,
    C,
    surrogate_method_name="gpr",
    surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua"},
    surrogate_return_mean_variance=False,
    logger=None,
    file_path=None,
    device=None,
):
    """Fit the objective surrogate on the feasible, deduplicated archive
    (reference MOASMO.py:473-532)."""
    x = np.asarray(Xinit, dtype=np.float64).copy()
    y = np.asarray(Yinit, dtype=np.float64).copy()
    if C is not None:
        feasible = np.argwhere(np.all(C > 0.0, axis=1))
        if len(feasible) > 0:
            feasible = feasible.ravel()
            x, y = x[feasible, :], y[feasible, :]
            if logger is not None:
                logger.info(f"Found {len(feasible)} feasible solutions")
    xt, yt = ops.remove_duplicates(
        torch.as_tensor(x, dtype=torch.float64), torch.as_tensor(y, dtype=torch.float64)
    )
    x, y = xt.numpy(), yt.numpy()
    surrogate_cls = cfg.resolve(cfg.surrogate_registry, surrogate_method_name)
    kwargs = dict(surrogate_method_kwargs)
    if device is not None:
        kwargs.setdefault("device", device)
    # a user-provided return_mean_variance in the kwargs wins over the
    # engine-level flag (passing both is a reference API footgun:
    # "got multiple values for keyword argument")
    kwargs.setdefault("return_mean_variance", surrogate_return_mean_variance)

    # Multi-rank hyperparameter broadcast ("broadcast
    # of surrogate hyperparameters"): surrogate classes whose fitted state is
    # fully determined by (archive, theta) opt in via supports_theta_broadcast.
    # Rank 0 runs the hyperparameter search; everyone else receives theta in
    # ONE tensor broadcast and rebuilds the posterior locally — correct by
    # construction even if the search itself were nondeterministic.
    from dmosopt_amd.parallel.context import get_context

    ctx = get_context()
    if (
        ctx is not None
        and ctx.world > 1
        and getattr(surrogate_cls, "supports_theta_broadcast", False)
    ):
        if ctx.is_root:
            try:
                sm = surrogate_cls(
                    x, y, nInput, nOutput, xlb, xub, logger=logger, **kwargs
                )
            except BaseException:
                # non-root ranks are waiting in bcast_payload — poison the
                # header so they raise instead of hanging forever
                ctx.bcast_poison(src=0)
                raise
            ctx.bcast_payload(sm.theta.detach().to("cpu", torch.float64), src=0)
            return sm
        theta = ctx.bcast_payload(None, src=0).cpu().numpy()
        return surrogate_cls(
            x, y, nInput, nOutput, xlb, xub, logger=logger,
            theta_override=theta, **kwargs,
        )
    return surrogate_cls(x, y, nInput, nOutput, xlb, xub, logger=logger, **kwargs)


def analyze_sensitivity(
    sm,
    xlb,
    xub,
    param_names,
    objective_names,
    sensitivity_method_name=None,
    sensitivity_method_kwargs={},
    di_min=1.0,
    di_max=20.0,
    logger=None,
):
    """SA on the surrogate -> distribution indices (MOASMO.py:535-578)."""
    di_mutation = di_crossover = None
    if sensitivity_method_name is not None:
        sens_cls = cfg.resolve(cfg.sensitivity_registry, sensitivity_method_name)
        sens = sens_cls(xlb, xub, param_names, objective_names, **sensitivity_method_kwargs)
        res = sens.analyze(sm)
        S1s = np.vstack([res["S1"][name] for name in objective_names])
        S1s = np.nan_to_num(S1s, copy=False)
        S1max = np.max(S1s, axis=0)
        denom = np.max(S1max) if np.max(S1max) > 0 else 1.0
        S1nmax = S1max / denom
        di_mutation = np.clip(S1nmax * di_max, di_min, None)
        di_crossover = np.clip(S1nmax * di_max, di_min, None)
    if logger is not None:
        logger.info(f"analyze_sensitivity: di_mutation = {di_mutation}")
    return {"di_mutation": di_mutation, "di_crossover": di_crossover}


def run_epoch(
    num_generations,
    param_names,
    objective_names,
    xlb,
    xub,
    pct,
    Xinit,
    Yinit,
    C,
    pop=100,
    sampling_method_name=None,
    feasibility_method_name=None,
    feasibility_method_kwargs={},
    optimizer_name="nsga2",
    optimizer_kwargs={},
    surrogate_method_name="gpr",
    surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua"},
    surrogate_custom_training=None,
    surrogate_custom_training_kwargs=None,
    sensitivity_method_name=None,
    sensitivity_method_kwargs={},
    optimize_mean_variance=False,
    termination=None,
    local_random=None,
    logger=None,
    file_path=None,
    evaluator: Optional[Callable] = None,
    device=None,
) -> Dict:
    """One MO-ASMO epoch (reference MOASMO.epoch, MOASMO.py:196-470).

    Returns the reference's return_dict: with a surrogate, the resample
    batch {x_resample, y_pred, ...}; without, {best_x, best_y, ...}.
    """
    nInput, nOutput = len(param_names), len(objective_names)
    N_resample = int(pop * pct)

    x_0 = np.asarray(Xinit, dtype=np.float32).copy()
    y_0 = np.asarray(Yinit, dtype=np.float32).copy()
    if optimize_mean_variance:
        y_0 = np.column_stack((y_0, np.zeros_like(y_0)))

    optimizer_cls = cfg.resolve(cfg.optimizer_registry, optimizer_name)

    stats: Dict = {"model_init_start": time.time()}
    mdl = Model(return_mean_variance=optimize_mean_variance)

    if surrogate_custom_training is not None:
        custom_training = (
            surrogate_custom_training
            if callable(surrogate_custom_training)
            else cfg.import_object_by_path(surrogate_custom_training)
        )
        (optimizer_cls, mdl.objective, mdl.feasibility, mdl.sensitivity) = custom_training(
            optimizer_cls, Xinit, Yinit, C, xlb, xub, file_path,
            options={
                "optimizer_name": optimizer_name,
                "optimizer_kwargs": optimizer_kwargs,
                "surrogate_method_name": surrogate_method_name,
                "surrogate_method_kwargs": surrogate_method_kwargs,
                "feasibility_method_name": feasibility_method_name,
                "feasibility_method_kwargs": feasibility_method_kwargs,
                "sensitivity_method_name": sensitivity_method_name,
                "sensitivity_method_kwargs": sensitivity_method_kwargs,
                "return_mean_variance": optimize_mean_variance,
            },
            **(surrogate_custom_training_kwargs or {}),
        )

    if feasibility_method_name is not None and mdl.feasibility is None and C is not None:
        try:
            if logger is not None:
                logger.info("Constructing feasibility model...")
            feas_cls = cfg.resolve(cfg.feasibility_registry, feasibility_method_name)
            mdl.feasibility = feas_cls(Xinit, C, **(feasibility_method_kwargs or {}))
        except Exception as e:
            if logger is not None:
                logger.warning(f"Unable to fit feasibility model: {e}")

    if surrogate_method_name is not None and mdl.objective is None:
        smk = dict(surrogate_method_kwargs or {})
        if smk.get("seed") is None and local_random is not None:
            # the surrogate's OWN hyperparameter search (SCE-UA / Adam init)
            # must inherit determinism from the run seed: the reference
            # leaves it unseeded (model.py:1192 seed=None default -> OS
            # entropy), which silently breaks same-seed reproducibility —
            # two same-seed runs fit slightly different thetas and ~10% of
            # the time the resample selection flips. Drawing the seed from
            # local_random is replicated identically on every rank (all
            # ranks execute this line), so the theta-broadcast scheme and
            # the per-epoch archive hash guard stay consistent.
            smk["seed"] = int(local_random.integers(0, 2**31 - 1))
        mdl.objective = train(
            nInput, nOutput, xlb, xub, Xinit, Yinit, C,
            surrogate_method_name=surrogate_method_name,
            surrogate_method_kwargs=smk,
            surrogate_return_mean_variance=optimize_mean_variance,
            logger=logger, file_path=file_path, device=device,
        )

    # Multi-rank: shard every surrogate prediction (the per-generation hot
    # path AND the sensitivity sweep below) across ranks; results reassemble
    # via one all_gather so every replicated rank sees identical values.
    from dmosopt_amd.parallel.context import get_context as _get_ctx

    _ctx = _get_ctx()
    if _ctx is not None and _ctx.world > 1 and mdl.objective is not None:
        from dmosopt_amd.parallel.sharded import ShardedObjective

        mdl.objective = ShardedObjective(mdl.objective, _ctx)

    if sensitivity_method_name is not None and mdl.sensitivity is None:
        class _S:
            def __init__(s):
                s._di_dict = analyze_sensitivity(
                    mdl.objective, xlb, xub, param_names, objective_names,
                    sensitivity_method_name=sensitivity_method_name,
                    sensitivity_method_kwargs=sensitivity_method_kwargs,
                    logger=logger,
                )

            def di_dict(s):
                return dict(s._di_dict)

        mdl.sensitivity = _S()

    optimizer_kwargs_ = {"sampling_method": "slh", "mutation_rate": None, "nchildren": 1}
    optimizer_kwargs_.update(optimizer_kwargs)
    if mdl.sensitivity is not None:
        di_dict = mdl.sensitivity.di_dict()
        optimizer_kwargs_["di_mutation"] = di_dict["di_mutation"]
        optimizer_kwargs_["di_crossover"] = di_dict["di_crossover"]

    stats["model_init_end"] = time.time()
    stats.update(mdl.get_stats())

    optimizer_kwargs_.setdefault("distance_metric", None)
    optimizer = optimizer_cls(
        nInput=nInput, nOutput=nOutput, popsize=pop, model=mdl,
        optimize_mean_variance=optimize_mean_variance,
        **optimizer_kwargs_,
    )
    if device is not None:
        optimizer.set_device(device)

    if C is not None:
        feasible = np.argwhere(np.all(C > 0.0, axis=1))
        if len(feasible) > 0:
            feasible = feasible.ravel()
            x_0 = x_0[feasible, :]
            y_0 = y_0[feasible, :]

    res = optimize_loop(
        num_generations, optimizer, mdl, nInput, nOutput, xlb, xub,
        initial=(x_0, y_0), logger=logger, popsize=pop,
        local_random=local_random, termination=termination,
        optimize_mean_variance=optimize_mean_variance, evaluator=evaluator,
        **optimizer_kwargs_,
    )
    best_x, best_y = res.best_x, res.best_y

    if mdl.objective is not None:
        # dedupe best against the real-evaluated archive, rank by crowding
        # distance, keep top N_resample for real evaluation next epoch
        bx = torch.as_tensor(best_x, dtype=torch.float64)
        x0t = torch.as_tensor(np.asarray(Xinit, dtype=np.float64))
        dup = _cross_duplicates(bx, x0t)
        best_x, best_y = best_x[~dup], best_y[~dup]
        if best_x.shape[0] == 0:
            best_x, best_y = res.best_x, res.best_y
        D = _to_np(ops.crowding_distance(torch.as_tensor(best_y, dtype=torch.float64)))
        idxr = D.argsort()[::-1][:N_resample]
        return {
            "x_resample": best_x[idxr, :],
            "y_pred": best_y[idxr, :],
            "gen_index": res.gen_index,
            "x_sm": res.x,
            "y_sm": res.y,
            "optimizer": res.optimizer,
            "stats": stats,
        }
    return {
        "best_x": best_x,
        "best_y": best_y,
        "gen_index": res.gen_index,
        "x": res.x,
        "y": res.y,
        "optimizer": res.optimizer,
        "stats": stats,
    }


def _cross_duplicates(X: torch.Tensor, Y: torch.Tensor, eps: float = 1e-16) -> np.ndarray:
    """Rows of X within eps of any row of Y (intent of reference
    MOEA.get_duplicates(X, Y); the reference's triu masking artifact on
    cross-matrices is not reproduced)."""
    if X.numel() == 0 or Y.numel() == 0:
        return np.zeros(X.shape[0], dtype=bool)
    # direct-difference mode: exact-duplicate detection at eps=1e-16 must
    # not depend on GEMM ULP noise (see ops/torch_ref.get_duplicates)
    D = torch.cdist(X.double(), Y.double(),
                    compute_mode="donot_use_mm_for_euclid_dist")
    D = torch.nan_to_num(D, nan=float("inf"))
    return (D <= eps).any(dim=1).cpu().numpy()


# -------------------------------------------------------------- best / final
def get_best(
    x, y, f, c, nInput, nOutput, epochs=None, feasible=True,
    return_perm=False, return_feasible=False, delete_duplicates=True,
):
    """Feasible filter + dedupe + non-dominated front (MOASMO.py:581-639)."""
    xtmp, ytmp = x, y
    feas_idx = None
    if feasible and c is not None:
        feas_idx = np.argwhere(np.all(c > 0.0, axis=1)).ravel()
        if len(feas_idx) > 0:
            xtmp, ytmp = x[feas_idx, :], y[feas_idx, :]
            if f is not None:
                f = f[feas_idx]
            c = c[feas_idx, :]
            if epochs is not None:
                epochs = epochs[feas_idx]

    if delete_duplicates:
        dup = _to_np(ops.get_duplicates(torch.as_tensor(ytmp, dtype=torch.float64)))
        xtmp, ytmp = xtmp[~dup], ytmp[~dup]
        if f is not None:
            f = f[~dup]
        if c is not None:
            c = c[~dup]
        if epochs is not None:
            epochs = epochs[~dup]

    xt = torch.as_tensor(xtmp, dtype=torch.float64)
    yt = torch.as_tensor(ytmp, dtype=torch.float64)
    perm, rank, _ = ops.order_mo(xt, yt)
    perm = perm.cpu().numpy()
    rank = rank.cpu().numpy()
    xs, ys = xtmp[perm], ytmp[perm]
    idxp = rank == 0
    best_x, best_y = xs[idxp, :], ys[idxp, :]
    best_f = f[perm][idxp] if f is not None else None
    best_c = c[perm, :][idxp, :] if c is not None else None
    best_epoch = epochs[perm][idxp] if epochs is not None else None
    if not return_perm:
        perm = None
    if return_feasible:
        return best_x, best_y, best_f, best_c, best_epoch, perm, feas_idx
    return best_x, best_y, best_f, best_c, best_epoch, perm


def get_feasible(x, y, f, c, nInput, nOutput, epochs=None):
    """Feasible filter + full rank/epoch cross-indexing (MOASMO.py:642-700).

    Returns (perm_arrs, rnk_arrs, epc_arrs, rnk_epc_idx): the permuted
    arrays, unique-rank and unique-epoch index groupings, and their
    intersection matrix.
    """
    xtmp, ytmp = np.copy(x), np.copy(y)
    feas = None
    if c is not None:
        feas = np.argwhere(np.all(c > 0.0, axis=1))
        if len(feas) > 0:
            feas = feas.ravel()
            xtmp, ytmp = xtmp[feas, :], ytmp[feas, :]
            if f is not None:
                f = f[feas]
            c = c[feas, :]
            if epochs is not None:
                epochs = epochs[feas]

    xt = torch.as_tensor(xtmp, dtype=torch.float64)
    yt = torch.as_tensor(ytmp, dtype=torch.float64)
    perm_t, rank_t, _ = ops.order_mo(xt, yt)
    perm = perm_t.cpu().numpy()
    rank = rank_t.cpu().numpy()
    perm_x, perm_y = xtmp[perm], ytmp[perm]
    perm_f = f[perm] if f is not None else None
    perm_epoch = epochs[perm] if epochs is not None else None
    perm_c = c[perm] if c is not None else None

    uniq_rank, rnk_inv, rnk_cnt = np.unique(rank, return_inverse=True, return_counts=True)
    rank_idx = np.array(
        [np.flatnonzero(rnk_inv == i) for i in range(len(uniq_rank))], dtype=object
    )
    if perm_epoch is not None:
        uniq_epc, epc_inv, epc_cnt = np.unique(
            perm_epoch, return_inverse=True, return_counts=True
        )
        epc_idx = np.array(
            [np.flatnonzero(epc_inv == i) for i in range(len(uniq_epc))], dtype=object
        )
    else:
        uniq_epc = np.array([0])
        epc_cnt = np.array([len(perm)])
        epc_idx = np.array([np.arange(len(perm))], dtype=object)

    rnk_epc_idx = np.empty((len(uniq_rank), len(uniq_epc)), dtype=object)
    for i in range(len(uniq_rank)):
        for j in range(len(uniq_epc)):
            rnk_epc_idx[i, j] = np.intersect1d(
                rank_idx[i], epc_idx[j], assume_unique=True
            )

    perm_arrs = (perm_x, perm_y, perm_f, perm_epoch, perm, feas)
    rnk_arrs = (uniq_rank, rank_idx, rnk_cnt)
    epc_arrs = (uniq_epc, epc_idx, epc_cnt)
    return perm_arrs, rnk_arrs, epc_arrs, rnk_epc_idx


def epsilon_get_best(x, y, f, c, feasible=True, delete_duplicates=True, epsilons=None):
    """Epsilon-box Pareto archive selection (MOASMO.py:703-758)."""
    from dmosopt_amd.moea.epsilon import EpsilonSort

    if feasible and c is not None:
        feas = np.argwhere(np.all(c > 0.0, axis=1)).ravel()
        if len(feas) > 0:
            x, y = x[feas, :], y[feas, :]
            if f is not None:
                f = f[feas]
            c = c[feas, :]
    if delete_duplicates:
        dup = _to_np(ops.get_duplicates(torch.as_tensor(y, dtype=torch.float64)))
        x, y = x[~dup], y[~dup]
        if f is not None:
            f = f[~dup]
        if c is not None:
            c = c[~dup]

    if epsilons is None:
        epsilons = [1e-9] * y.shape[1]
    elif isinstance(epsilons, (int, float)):
        epsilons = [float(epsilons)] * y.shape[1]
    elif epsilons == "auto":
        epsilons = 0.05 * scipy_stats.iqr(y, axis=0)

    if y.shape[0] == 0:
        return x, y, f, c, epsilons

    sorter = EpsilonSort(epsilons)
    for i in range(y.shape[0]):
        sorter.sortinto(y[i], tagalong=i)
    m = np.array(sorter.tagalongs)
    best_f = f[m] if f is not None else None
    best_c = c[m] if c is not None else None
    return x[m], y[m], best_f, best_c, epsilons
