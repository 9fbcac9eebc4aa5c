"""Distributed optimization driver (reference DistOptimizer,
dmosopt.py:546-1470).

Orchestrates per-problem strategies, the evaluation farm, HDF5 persistence
and statistics. The distwq request pump becomes a synchronous-collective
loop: drain the strategy's request queue into ONE batched farm call
(broadcast + shard + gather over RCCL/gloo) instead of per-point MPI task
dispatch.
"""

from __future__ import annotations

import logging
import os
import time
from functools import partial
from typing import Dict, Optional, Sequence

import numpy as np
from numpy.random import default_rng

from dmosopt_amd.config import import_object_by_path
from dmosopt_amd.core import engine
from dmosopt_amd.core.strategy import DistOptStrategy
from dmosopt_amd.datatypes import (
    EvalRequest,
    OptProblem,
    ParameterSpace,
    StrategyState,
    update_nested_dict,
)

logger = logging.getLogger("dmosopt_amd")


def eval_obj_fun_sp(
    obj_fun, pp, param_space, nested_parameter_space, obj_fun_args, problem_id, space_vals
):
    """Single-problem objective evaluation (reference dmosopt.py:2352-2389):
    merge fixed problem parameters with sampled values, call, time."""
    this_space_vals = space_vals[problem_id]
    if nested_parameter_space:
        this_pp = update_nested_dict(pp.unflatten(), param_space.unflatten(this_space_vals))
    else:
        this_pp = {}
        if pp is not None:
            this_pp.update(
                (item.name, int(item.value) if item.is_integer else item.value)
                for item in pp.items
            )
        this_pp.update(
            (name, this_space_vals[i])
            for i, name in enumerate(param_space.parameter_names)
        )
    if obj_fun_args is None:
        obj_fun_args = ()
    t = time.time()
    result = obj_fun(this_pp, *obj_fun_args)
    return {problem_id: result, "time": time.time() - t}


def eval_obj_fun_mp(
    obj_fun, pp, param_space, nested_parameter_space, obj_fun_args, problem_ids, space_vals
):
    """Multi-problem objective evaluation (reference dmosopt.py:2391-2434)."""
    mpp = {}
    for problem_id in problem_ids:
        this_space_vals = space_vals[problem_id]
        if nested_parameter_space:
            this_pp = update_nested_dict(
                pp.unflatten(), param_space.unflatten(this_space_vals)
            )
        else:
            this_pp = {}
            if pp is not None:
                this_pp.update(
                    (item.name, int(item.value) if item.is_integer else item.value)
                    for item in pp.items
                )
            this_pp.update(
                (name, this_space_vals[i])
                for i, name in enumerate(param_space.parameter_names)
            )
        mpp[problem_id] = this_pp
    if obj_fun_args is None:
        obj_fun_args = ()
    t = time.time()
    result_dict = obj_fun(mpp, *obj_fun_args)
    result_dict["time"] = time.time() - t
    return result_dict


class DistOptimizer:
    def __init__(
        self,
        opt_id,
        obj_fun,
        obj_fun_args=None,
        objective_names=None,
        feature_dtypes=None,
        feature_class=None,
        constraint_names=None,
        n_initial=10,
        initial_maxiter=5,
        initial_method="slh",
        dynamic_initial_sampling=None,
        dynamic_initial_sampling_kwargs=None,
        verbose=False,
        reduce_fun=None,
        reduce_fun_args=None,
        problem_ids=None,
        problem_parameters=None,
        space=None,
        population_size=100,
        num_generations=200,
        resample_fraction=0.25,
        distance_metric=None,
        n_epochs=10,
        save_eval=10,
        file_path=None,
        save=False,
        save_surrogate_evals=False,
        save_optimizer_params=True,
        metadata=None,
        nested_parameter_space=False,
        surrogate_method_name="gpr",
        surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua"},
        surrogate_custom_training=None,
        surrogate_custom_training_kwargs=None,
        optimizer_name="nsga2",
        optimizer_kwargs={"mutation_prob": 0.1, "crossover_prob": 0.9},
        sensitivity_method_name=None,
        sensitivity_method_kwargs={},
        optimize_mean_variance=False,
        local_random=None,
        random_seed=None,
        feasibility_method_name=None,
        feasibility_method_kwargs=None,
        termination_conditions=None,
        farm=None,
        device=None,
        **kwargs,
    ) -> None:
        if device is None:
            import torch

            if torch.cuda.is_available():
                # a GPU box runs the whole stack on the GPU unless told
                # otherwise (surrogate fit AND the MOEA populations)
                device = torch.device("cuda")
        if (random_seed is not None) and (local_random is not None):
            raise RuntimeError(
                "Both random_seed and local_random are specified! Only one may be."
            )
        if random_seed is not None:
            local_random = default_rng(seed=random_seed)

        self.farm = farm
        self.opt_id = opt_id
        self.verbose = verbose
        self.device = device
        self.population_size = population_size
        self.num_generations = num_generations
        self.resample_fraction = min(resample_fraction, 1.0)
        self.distance_metric = distance_metric
        self.dynamic_initial_sampling = dynamic_initial_sampling
        self.dynamic_initial_sampling_kwargs = dynamic_initial_sampling_kwargs
        self.surrogate_method_name = surrogate_method_name
        self.surrogate_method_kwargs = surrogate_method_kwargs
        self.surrogate_custom_training = surrogate_custom_training
        self.surrogate_custom_training_kwargs = surrogate_custom_training_kwargs
        self.sensitivity_method_name = sensitivity_method_name
        self.sensitivity_method_kwargs = sensitivity_method_kwargs
        self.optimizer_name = (
            optimizer_name
            if isinstance(optimizer_name, Sequence) and not isinstance(optimizer_name, str)
            else (optimizer_name,)
        )
        self.optimizer_kwargs = (
            optimizer_kwargs if isinstance(optimizer_kwargs, Sequence) else (optimizer_kwargs,)
        )
        self.optimize_mean_variance = optimize_mean_variance
        self.feasibility_method_name = feasibility_method_name
        self.feasibility_method_kwargs = feasibility_method_kwargs
        self.termination_conditions = termination_conditions
        self.metadata = metadata
        self.local_random = local_random
        self.random_seed = random_seed

        self.logger = logging.getLogger(opt_id)
        if self.verbose:
            self.logger.setLevel(logging.INFO)

        if file_path is None:
            if problem_parameters is None or space is None:
                raise ValueError(
                    "You must specify at least file name `file_path` or problem "
                    "parameters `problem_parameters` along with a parameter space `space`."
                )
            if save:
                raise ValueError("If you want to save you must specify `file_path`.")
        else:
            if not os.path.isfile(file_path):
                if problem_parameters is None or space is None:
                    raise FileNotFoundError(file_path)

        param_space = ParameterSpace.from_dict(space) if space is not None else None
        if problem_parameters is not None:
            problem_parameters = ParameterSpace.from_dict(problem_parameters, is_value_only=True)

        old_evals = {}
        max_epoch = -1
        stored_random_seed = None
        if file_path is not None and os.path.isfile(file_path):
            from dmosopt_amd.storage import h5 as h5store

            (
                stored_random_seed,
                max_epoch,
                old_evals,
                param_space,
                objective_names,
                feature_dtypes,
                constraint_names,
                problem_parameters,
                problem_ids,
            ) = h5store.init_from_h5(file_path, param_space.parameter_names if param_space else None, opt_id, self.logger)
        if stored_random_seed is not None:
            if local_random is not None and self.logger is not None:
                self.logger.warning("Using saved random seed to create local RNG. ")
            self.local_random = default_rng(seed=stored_random_seed)
            self.random_seed = stored_random_seed
        if self.local_random is None:
            self.local_random = default_rng()

        if problem_parameters is not None and param_space is not None:
            assert set(param_space.parameter_names).isdisjoint(
                set(problem_parameters.parameter_names)
            )
        assert param_space is not None and param_space.n_parameters > 0
        self.param_space = param_space
        self.param_names = param_space.parameter_names
        assert objective_names is not None
        self.objective_names = objective_names

        # A restored file always carries problem_ids=[0] for single-problem
        # runs; treat {0} as the single-problem convention so the objective
        # keeps its flat-dict signature across resume.
        has_problem_ids = problem_ids is not None and set(problem_ids) != {0}
        if not has_problem_ids:
            problem_ids = set([0])

        self.n_initial = n_initial
        self.initial_maxiter = initial_maxiter
        self.initial_method = initial_method
        self.problem_parameters = problem_parameters
        self.file_path, self.save = file_path, save

        for okw in self.optimizer_kwargs:
            if okw is None:
                continue
            for key in ("di_crossover", "di_mutation"):
                val = okw.get(key, None)
                if isinstance(val, dict):
                    okw[key] = param_space.flatten(val)

        self.epoch_count = 0
        self.start_epoch = max_epoch if max_epoch > 0 else 0
        self.n_epochs = n_epochs
        self.save_eval = save_eval
        self.save_surrogate_evals_ = save_surrogate_evals
        self.save_optimizer_params_ = save_optimizer_params
        self.saved_eval_count = 0
        self.eval_count = 0

        self.obj_fun_args = obj_fun_args
        if has_problem_ids:
            self.eval_fun = partial(
                eval_obj_fun_mp, obj_fun, self.problem_parameters, self.param_space,
                nested_parameter_space, self.obj_fun_args, problem_ids,
            )
        else:
            self.eval_fun = partial(
                eval_obj_fun_sp, obj_fun, self.problem_parameters, self.param_space,
                nested_parameter_space, self.obj_fun_args, 0,
            )

        self.reduce_fun = reduce_fun
        self.reduce_fun_args = reduce_fun_args
        self.old_evals = old_evals
        self.has_problem_ids = has_problem_ids
        self.problem_ids = problem_ids
        self.optimizer_dict: Dict = {}
        self.storage_dict: Dict = {}

        self.feature_constructor = lambda x: x
        if feature_class is not None:
            self.feature_constructor = import_object_by_path(feature_class)
        self.feature_dtypes = feature_dtypes
        self.feature_names = [dt[0] for dt in feature_dtypes] if feature_dtypes else None
        self.constraint_names = constraint_names

        if self.save and file_path is not None and not os.path.isfile(file_path):
            from dmosopt_amd.storage import h5 as h5store

            h5store.init_h5(
                self.opt_id, self.problem_ids, self.has_problem_ids, self.param_space,
                self.objective_names, self.feature_dtypes, self.constraint_names,
                self.problem_parameters, self.metadata, self.random_seed, self.file_path,
                surrogate_mean_variance=self.optimize_mean_variance,
            )
        self.stats: Dict = {}

    # ------------------------------------------------------------- strategy
    def initialize_strategy(self):
        opt_prob = OptProblem(
            self.param_names, self.objective_names, self.feature_dtypes,
            self.feature_constructor, self.constraint_names, self.param_space,
            self.eval_fun, logger=self.logger,
        )
        dim = len(self.param_names)
        any_restored = False
        for problem_id in self.problem_ids:
            initial = None
            if problem_id in self.old_evals and len(self.old_evals[problem_id]) > 0:
                evals = self.old_evals[problem_id]
                epochs = None
                if evals[0].epoch is not None:
                    epochs = np.concatenate([np.atleast_1d(e.epoch) for e in evals], axis=None)
                x = np.vstack([e.parameters for e in evals])
                y = np.vstack([e.objectives for e in evals])
                f = None
                if self.feature_dtypes is not None:
                    f = self.feature_constructor(
                        np.concatenate([np.atleast_1d(e.features) for e in evals], axis=0)
                    )
                c = None
                if self.constraint_names is not None:
                    c = np.vstack([e.constraints for e in evals])
                initial = (epochs, x, y, f, c)
                if len(evals) >= self.n_initial * dim:
                    self.start_epoch += 1

            self.optimizer_dict[problem_id] = DistOptStrategy(
                opt_prob,
                self.n_initial,
                initial=initial,
                resample_fraction=self.resample_fraction,
                population_size=self.population_size,
                num_generations=self.num_generations,
                initial_maxiter=self.initial_maxiter,
                initial_method=self.initial_method,
                distance_metric=self.distance_metric,
                surrogate_method_name=self.surrogate_method_name,
                surrogate_method_kwargs=self.surrogate_method_kwargs,
                surrogate_custom_training=self.surrogate_custom_training,
                surrogate_custom_training_kwargs=self.surrogate_custom_training_kwargs,
                sensitivity_method_name=self.sensitivity_method_name,
                sensitivity_method_kwargs=self.sensitivity_method_kwargs,
                optimizer_name=self.optimizer_name,
                optimizer_kwargs=self.optimizer_kwargs,
                feasibility_method_name=self.feasibility_method_name,
                feasibility_method_kwargs=self.feasibility_method_kwargs,
                termination_conditions=self.termination_conditions,
                optimize_mean_variance=self.optimize_mean_variance,
                local_random=self.local_random,
                logger=self.logger,
                file_path=self.file_path,
                device=self.device,
            )
            self.storage_dict[problem_id] = []
            any_restored = any_restored or (initial is not None)
        if any_restored:
            self.print_best()

    # ------------------------------------------------------------- requests
    def _complete_one(self, problem_id, req, res):
        """Apply reduce_fun, slice this problem's entry out of the shared
        result dict, and complete the strategy request."""
        strategy = self.optimizer_dict[problem_id]
        if self.reduce_fun is not None:
            args = self.reduce_fun_args or ()
            res = self.reduce_fun([res], *args)
        entry = res[problem_id] if isinstance(res, dict) else res
        t_eval = res.get("time", -1.0) if isinstance(res, dict) else -1.0
        y, f, c = _split_result(
            entry, self.objective_names, self.feature_names, self.constraint_names
        )
        strategy.complete_request(
            req.parameters, y, epoch=req.epoch, f=f, c=c,
            pred=req.prediction, time=t_eval,
        )
        self.storage_dict[problem_id].append(strategy.completed[-1])
        self.eval_count += 1
        if self.verbose:
            self.logger.info(f"problem {problem_id}: eval {self.eval_count}: y = {y}")

    def _process_requests(self):
        """Drain all strategies' request queues through the farm as batched
        collective evaluations; complete the requests with the results.

        Multi-problem dispatch zips ONE request per problem id into a single
        farm point ({pid: its own parameters}) so the user objective runs
        once per zipped point, completing each problem's request from its
        slice of the shared result (reference dmosopt.py:1291-1313). Any
        ragged tail (a problem with more queued requests than its siblings)
        is dispatched per-problem with its own parameters in every slot —
        progress is guaranteed, at the cost of redundant sibling entries for
        those tail points only."""
        if not self.has_problem_ids:
            strategy = self.optimizer_dict[0]
            reqs = []
            while strategy.has_requests():
                reqs.append(strategy.get_next_request())
            if reqs:
                points = [{0: r.parameters} for r in reqs]
                results = self.farm.evaluate(self.opt_id, points)
                for req, res in zip(reqs, results):
                    self._complete_one(0, req, res)
        else:
            pids = list(self.problem_ids)
            strategies = {pid: self.optimizer_dict[pid] for pid in pids}
            zipped = []
            while all(s.has_requests() for s in strategies.values()):
                zipped.append({pid: strategies[pid].get_next_request() for pid in pids})
            if zipped:
                points = [{pid: rd[pid].parameters for pid in pids} for rd in zipped]
                results = self.farm.evaluate(self.opt_id, points)
                for rd, res in zip(zipped, results):
                    for pid in pids:
                        self._complete_one(pid, rd[pid], res)
            for pid in pids:
                strategy = strategies[pid]
                tail = []
                while strategy.has_requests():
                    tail.append(strategy.get_next_request())
                if tail:
                    points = [{q: r.parameters for q in pids} for r in tail]
                    results = self.farm.evaluate(self.opt_id, points)
                    for req, res in zip(tail, results):
                        self._complete_one(pid, req, res)
        if self.save and (self.eval_count - self.saved_eval_count) >= self.save_eval:
            self.save_evals()
            self.saved_eval_count = self.eval_count
        # flush the tail below the cadence (reference dmosopt.py:1329-1335:
        # nothing unsaved survives the end of a request pump)
        if self.save and 0 < self.saved_eval_count < self.eval_count or (
            self.save and self.saved_eval_count == 0 and self.eval_count > 0
        ):
            self.save_evals()
            self.saved_eval_count = self.eval_count
        return self.eval_count, self.saved_eval_count

    # ---------------------------------------------------------------- epoch
    def run_epoch(self, completed_epoch=False):
        epoch = self.epoch_count + self.start_epoch
        advance_epoch = self.epoch_count < self.n_epochs - 1

        self.stats["init_sampling_start"] = time.time()
        self._process_requests()

        for problem_id in self.problem_ids:
            strategy = self.optimizer_dict[problem_id]
            if self.dynamic_initial_sampling is not None and self.epoch_count == 0:
                sampler = (
                    self.dynamic_initial_sampling
                    if callable(self.dynamic_initial_sampling)
                    else import_object_by_path(self.dynamic_initial_sampling)
                )
                it = 0
                while True:
                    more = sampler(
                        file_path=self.file_path,
                        iteration=it,
                        evaluated_samples=strategy.completed,
                        next_samples=engine.xinit(
                            self.n_initial, strategy.prob.param_names,
                            strategy.prob.lb, strategy.prob.ub,
                            nPrevious=None, maxiter=self.initial_maxiter,
                            method=self.initial_method,
                            local_random=self.local_random, logger=self.logger,
                        ),
                        sampler={
                            "n_initial": self.n_initial,
                            "maxiter": self.initial_maxiter,
                            "method": self.initial_method,
                            "param_names": strategy.prob.param_names,
                            "xlb": strategy.prob.lb,
                            "xub": strategy.prob.ub,
                        },
                        **(self.dynamic_initial_sampling_kwargs or {}),
                    )
                    if more is None:
                        break
                    for i in range(more.shape[0]):
                        strategy.append_request(EvalRequest(more[i, :], None, 0))
                    self._process_requests()
                    it += 1

            strategy.initialize_epoch(epoch)
        self.stats["init_sampling_end"] = time.time()

        # per-problem completion tracking: with per-problem termination one
        # strategy can finish its epoch before its siblings; a completed
        # strategy must not see another update_epoch call (its runner is
        # already cleared by _finish_epoch)
        done = set(self.problem_ids) if completed_epoch else set()
        while len(done) < len(self.problem_ids):
            self._process_requests()
            for problem_id in self.problem_ids:
                if problem_id in done:
                    continue
                state, value, completed_evals = self.optimizer_dict[problem_id].update_epoch(
                    resample=advance_epoch
                )
                if state == StrategyState.CompletedEpoch:
                    done.add(problem_id)
                    res = value
                    if completed_evals is not None and epoch > 1:
                        self._report_surrogate_accuracy(problem_id, epoch, completed_evals)
                    if advance_epoch and epoch > 0 and self.save:
                        if self.save_surrogate_evals_:
                            self.save_surrogate_evals(
                                problem_id, epoch, res.gen_index, res.x, res.y
                            )
                        if self.save_optimizer_params_:
                            optimizer = res.optimizer
                            self.save_optimizer_params(
                                problem_id, epoch, optimizer.name, optimizer.opt_parameters
                            )
        if self.save:
            for problem_id in self.problem_ids:
                self.save_stats(problem_id, epoch)
        # replicated-mode divergence guard: the per-problem archives must be
        # bit-identical on every rank after each epoch — fail loudly if any
        # nondeterminism crept into the replicated control flow
        from dmosopt_amd.parallel.context import get_context

        ctx = get_context()
        if ctx is not None and ctx.world > 1:
            for problem_id in self.problem_ids:
                s = self.optimizer_dict[problem_id]
                ctx.assert_synchronized(
                    [s.x, s.y, s.c], tag=f"epoch{epoch}/problem{problem_id}"
                )
        self.epoch_count += 1
        return self.epoch_count

    def _report_surrogate_accuracy(self, problem_id, epoch, completed_evals):
        x_c, y_c, pred_c, _, c_c = completed_evals
        if c_c is not None:
            feasible = np.argwhere(np.all(c_c > 0.0, axis=1))
            if len(feasible) > 0:
                feasible = feasible.ravel()
                x_c, y_c, pred_c = x_c[feasible, :], y_c[feasible, :], pred_c[feasible, :]
        if x_c.shape[0] > 0:
            mae = []
            for i in range(y_c.shape[1]):
                y_i, p_i = y_c[:, i], pred_c[:, i]
                valid = ~np.isnan(y_i) & ~np.isnan(p_i)
                mae.append(float(np.mean(np.abs(y_i[valid] - p_i[valid]))) if valid.any() else np.nan)
            self.logger.info(
                f"surrogate accuracy at epoch {epoch - 1} for problem {problem_id} was {mae}"
            )

    # ----------------------------------------------------------------- save
    def save_evals(self):
        from dmosopt_amd.storage import h5 as h5store

        finished = {}
        n = len(self.objective_names)
        for problem_id in self.problem_ids:
            evals = self.storage_dict[problem_id]
            if len(evals) > 0:
                n_pred = 2 * n if self.optimize_mean_variance else n
                finished[problem_id] = (
                    [e.epoch for e in evals],
                    [e.parameters for e in evals],
                    [e.objectives for e in evals],
                    [e.features for e in evals] if self.feature_names else None,
                    [e.constraints for e in evals] if self.constraint_names else None,
                    [([np.nan] * n_pred if e.prediction is None else e.prediction) for e in evals],
                )
                self.storage_dict[problem_id] = []
        if finished:
            h5store.save_to_h5(
                self.opt_id, self.problem_ids, self.has_problem_ids,
                self.param_space.parameter_names, self.objective_names,
                self.feature_dtypes, self.constraint_names, self.optimize_mean_variance,
                finished, self.file_path, self.logger,
            )

    def save_surrogate_evals(self, problem_id, epoch, gen_index, x_sm, y_sm):
        from dmosopt_amd.storage import h5 as h5store

        h5store.save_surrogate_evals_to_h5(
            self.opt_id, problem_id, epoch, self.param_space.parameter_names,
            self.objective_names, gen_index, x_sm, y_sm, self.file_path, self.logger,
        )

    def save_optimizer_params(self, problem_id, epoch, optimizer_name, optimizer_params):
        from dmosopt_amd.storage import h5 as h5store

        h5store.save_optimizer_params_to_h5(
            self.opt_id, problem_id, epoch, optimizer_name, optimizer_params,
            self.file_path, self.logger,
        )

    def save_stats(self, problem_id, epoch):
        from dmosopt_amd.storage import h5 as h5store

        h5store.save_stats_to_h5(
            self.opt_id, problem_id, epoch, self.get_stats(), self.file_path, self.logger
        )

    # ---------------------------------------------------------------- stats
    def get_stats(self):
        for problem_id in self.problem_ids:
            if problem_id in self.optimizer_dict:
                self.stats.update(
                    {
                        f"{problem_id}_{k}" if problem_id > 0 else k: v
                        for k, v in self.optimizer_dict[problem_id].stats.items()
                    }
                )
        result = {}
        for key in self.stats:
            if not key.endswith("_start") and not key.endswith("_end"):
                result[key] = self.stats[key]
                continue
            name, period = key.rsplit("_", 1)
            if period == "start" and f"{name}_end" in self.stats:
                result[name] = self.stats[f"{name}_end"] - self.stats[key]
        if self.farm is not None and hasattr(self.farm, "stats"):
            result.update(self.farm.stats())
        return result

    # ----------------------------------------------------------------- best
    def get_best(self, feasible=True, return_features=False, return_constraints=False):
        best = {}
        for problem_id in self.problem_ids:
            strategy = self.optimizer_dict[problem_id]
            bestx, besty, bestf, bestc = strategy.get_best_evals(feasible=feasible)
            prms = list(zip(self.param_names, [bestx[:, i] for i in range(bestx.shape[1])]))
            lres = list(zip(self.objective_names, [besty[:, i] for i in range(besty.shape[1])]))
            vals = [prms, lres]
            if return_features:
                vals.append(bestf)
            if return_constraints:
                vals.append(bestc)
            best[problem_id] = tuple(vals)
        if not self.has_problem_ids:
            return best[0]
        return best

    def get_evals(self, return_features=False, return_constraints=False):
        out = {
            pid: self.optimizer_dict[pid].get_evals(
                return_features=return_features, return_constraints=return_constraints
            )
            for pid in self.problem_ids
        }
        return out if self.has_problem_ids else out[0]

    def print_best(self, feasible=True):
        for problem_id in self.problem_ids:
            strategy = self.optimizer_dict[problem_id]
            if strategy.x is None:
                continue
            bestx, besty, bestf, bestc = strategy.get_best_evals(feasible=feasible)
            if bestx is None:
                continue
            n = bestx.shape[0]
            for i in range(n):
                prms = dict(zip(self.param_names, bestx[i]))
                objs = dict(zip(self.objective_names, besty[i]))
                self.logger.info(
                    f"problem {problem_id}: best {i + 1}/{n}: {objs} @ {prms}"
                )


def _split_result(entry, objective_names, feature_names, constraint_names):
    """Unpack a user objective return value: y | (y, f) | (y, f, c) | (y, c)."""
    f = c = None
    if isinstance(entry, tuple):
        if len(entry) == 3:
            y, f, c = entry
        elif len(entry) == 2:
            if feature_names is not None:
                y, f = entry
            else:
                y, c = entry
        else:
            y = entry[0]
    else:
        y = entry
    y = np.asarray(y, dtype=np.float64).ravel()
    if c is not None:
        c = np.asarray(c, dtype=np.float64).ravel()
    if f is not None:
        f = np.asarray(f)
    return y, f, c
