"""Per-problem optimization strategy state machine.

Replaces the reference's generator-pumping DistOptStrategy
(dmosopt.py:43-543) with an explicit state machine (SURVEY.md section 7
hard-part 4): ``EpochRunner`` exposes start()/step() transitions returning
('request', x_gen) or ('done', result_dict); ``DistOptStrategy`` keeps the
reference's public protocol — initialize_epoch / update_epoch returning
StrategyState.{EnqueuedRequests, WaitingRequests, CompletedEpoch}, the
request queue, the eval archive with dedupe/truncation — so driver-level
behavior matches the reference epoch for epoch.
"""

from __future__ import annotations

import itertools
from typing import Dict, Optional, Sequence, Union

import numpy as np
import torch

from dmosopt_amd import ops
from dmosopt_amd.core import engine
from dmosopt_amd.datatypes import (
    EpochResults,
    EvalEntry,
    EvalRequest,
    OptProblem,
    StrategyState,
)


def _anyclose(x, X, rtol=1e-4, atol=1e-8):
    if X is None or len(X) == 0:
        return False
    return bool(np.isclose(X, x[None, :], rtol=rtol, atol=atol).all(axis=1).any())


class EpochRunner:
    """One MO-ASMO epoch as an explicit state machine.

    Surrogate path: everything runs inside start() (model fit + inner MOEA
    against the surrogate) -> ('done', result_dict).
    Surrogate-free path: start()/step() exchange generation batches with the
    caller for real evaluation, mirroring the reference's yield protocol
    (each request carries reduce_evals=True, MOASMO.py:421-423).
    """

    def __init__(self, strategy: "DistOptStrategy", optimizer_index: int, epoch_kwargs: Dict):
        self.strategy = strategy
        self.kw = epoch_kwargs
        self.result: Optional[Dict] = None
        self._optimizer = None
        self._gen_iter = None
        self._pending_initial = None
        self._gen_indexes = []
        self._x_new = []
        self._y_new = []
        self._n_eval = 0
        self._gen_i = 0
        self._x0 = None
        self._y0 = None

    @property
    def surrogate_free(self) -> bool:
        return self.kw.get("surrogate_method_name") is None and self.kw.get(
            "surrogate_custom_training"
        ) is None

    def start(self):
        if not self.surrogate_free:
            result = engine.run_epoch(**self.kw)
            self.result = result
            return ("done", result)
        # surrogate-free: build optimizer only, request initial batch
        kw = self.kw
        import dmosopt_amd.config as cfg

        optimizer_kwargs_ = {"sampling_method": "slh", "mutation_rate": None, "nchildren": 1}
        optimizer_kwargs_.update(kw.get("optimizer_kwargs") or {})
        optimizer_kwargs_.setdefault("distance_metric", None)
        optimizer_cls = cfg.resolve(cfg.optimizer_registry, kw.get("optimizer_name", "nsga2"))
        from dmosopt_amd.models.model import Model

        self._mdl = Model(return_mean_variance=kw.get("optimize_mean_variance", False))
        nInput = len(kw["param_names"])
        nOutput = len(kw["objective_names"])
        self._optimizer = optimizer_cls(
            nInput=nInput,
            nOutput=nOutput,
            popsize=kw["pop"],
            model=self._mdl,
            optimize_mean_variance=kw.get("optimize_mean_variance", False),
            **optimizer_kwargs_,
        )
        if kw.get("device") is not None:
            self._optimizer.set_device(kw["device"])
        self._bounds = np.column_stack((kw["xlb"], kw["xub"]))
        self._local_random = kw.get("local_random") or np.random.default_rng()
        x = self._optimizer.generate_initial(self._bounds, self._local_random)
        self._phase = "init"
        self._x_pending = np.asarray(x, dtype=np.float32)
        return ("request", self._x_pending)

    def step(self, x, y, c=None):
        """Feed evaluated (x, y, c) for the pending request; advance."""
        assert self.surrogate_free
        kw = self.kw
        x = np.asarray(x, dtype=np.float32)
        y = np.asarray(y, dtype=np.float32)
        if self._phase == "init":
            x_0 = np.asarray(kw["Xinit"], dtype=np.float32)
            y_0 = np.asarray(kw["Yinit"], dtype=np.float32)
            C = kw.get("C")
            if C is not None:
                feasible = np.argwhere(np.all(np.asarray(C) > 0.0, axis=1))
                if len(feasible) > 0:
                    feasible = feasible.ravel()
                    x_0, y_0 = x_0[feasible, :], y_0[feasible, :]
            x_all = np.vstack((x_0, x))
            y_all = np.vstack((y_0, y))
            self._optimizer.initialize_strategy(
                x_all, y_all, self._bounds, self._local_random
            )
            self._gen_indexes.append(np.zeros((x_all.shape[0],), dtype=np.uint32))
            self._x0, self._y0 = x_all, y_all
            self._phase = "loop"
        else:
            self._optimizer.update(x, y, self._gen_state)
            self._n_eval += x.shape[0]
            self._x_new.append(x)
            self._y_new.append(y)
            self._gen_indexes.append(
                np.full((x.shape[0],), self._gen_i, dtype=np.uint32)
            )

        # next generation or finish
        termination = kw.get("termination")
        num_generations = kw["num_generations"]
        self._gen_i += 1
        finished = termination is None and self._gen_i > num_generations
        if not finished and termination is not None:
            pop_x, pop_y = self._optimizer.population_objectives
            from dmosopt_amd.datatypes import OptHistory

            hist = OptHistory(
                self._gen_i,
                self._n_eval,
                pop_x.cpu().numpy() if isinstance(pop_x, torch.Tensor) else pop_x,
                pop_y.cpu().numpy() if isinstance(pop_y, torch.Tensor) else pop_y,
                None,
            )
            finished = termination.has_terminated(hist)
        if finished:
            bestx, besty = self._optimizer.population_objectives
            to_np = lambda t: t.cpu().numpy() if isinstance(t, torch.Tensor) else t
            gen_index = np.concatenate(self._gen_indexes)
            x_all = np.vstack([self._x0] + self._x_new)
            y_all = np.vstack([self._y0] + self._y_new)
            self.result = {
                "best_x": to_np(bestx),
                "best_y": to_np(besty),
                "gen_index": gen_index,
                "x": x_all,
                "y": y_all,
                "optimizer": self._optimizer,
                "stats": {},
            }
            return ("done", self.result)
        x_gen, self._gen_state = self._optimizer.generate()
        x_gen = x_gen.cpu().numpy() if isinstance(x_gen, torch.Tensor) else x_gen
        self._x_pending = np.asarray(x_gen, dtype=np.float32)
        return ("request", self._x_pending)


class DistOptStrategy:
    def __init__(
        self,
        prob: OptProblem,
        n_initial: int = 10,
        initial=None,
        initial_maxiter: int = 5,
        initial_method: str = "slh",
        population_size: int = 100,
        resample_fraction: float = 0.25,
        num_generations: int = 100,
        surrogate_method_name: Optional[str] = "gpr",
        surrogate_method_kwargs: Dict = {"anisotropic": False, "optimizer": "sceua"},
        surrogate_custom_training: Optional[str] = None,
        surrogate_custom_training_kwargs: Optional[Dict] = None,
        sensitivity_method_name: Optional[str] = None,
        sensitivity_method_kwargs={},
        distance_metric=None,
        optimizer_name: Union[str, Sequence[str]] = "nsga2",
        optimizer_kwargs: Union[Dict, Sequence[Dict]] = {
            "crossover_prob": 0.9,
            "mutation_prob": 0.1,
        },
        feasibility_method_name=None,
        feasibility_method_kwargs={},
        termination_conditions=None,
        optimize_mean_variance=False,
        local_random=None,
        logger=None,
        file_path=None,
        device=None,
    ):
        if local_random is None:
            local_random = np.random.default_rng()
        self.local_random = local_random
        self.logger = logger
        self.file_path = file_path
        self.device = device
        self.feasibility_method_name = feasibility_method_name
        self.feasibility_method_kwargs = feasibility_method_kwargs
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
        self.optimizer_iter = itertools.cycle(range(len(self.optimizer_name)))
        self.distance_metric = distance_metric
        self.prob = prob
        self.completed = []
        self.reqs = []
        self.t = None
        if initial is None:
            self.x = self.y = self.f = self.c = None
        else:
            epochs, self.x, self.y, self.f, self.c = initial
        self.resample_fraction = resample_fraction
        self.num_generations = num_generations
        self.population_size = population_size

        self.termination = None
        if callable(termination_conditions):
            self.termination = termination_conditions(prob)
        elif termination_conditions:
            from dmosopt_amd.termination.adaptive import create_adaptive_termination

            termination_kwargs = {"strategy": "comprehensive", "n_max_gen": num_generations}
            if isinstance(termination_conditions, dict):
                termination_kwargs.update(termination_conditions)
            self.termination = create_adaptive_termination(prob, **termination_kwargs)

        nPrevious = self.x.shape[0] if self.x is not None else None
        x_init = engine.xinit(
            n_initial,
            prob.param_names,
            prob.lb,
            prob.ub,
            nPrevious=nPrevious,
            maxiter=initial_maxiter,
            method=initial_method,
            local_random=self.local_random,
            logger=self.logger,
        )
        self.reqs = []
        if x_init is not None:
            assert x_init.shape[1] == prob.dim
            if initial is None:
                self.reqs = [EvalRequest(x_init[i, :], None, 0) for i in range(x_init.shape[0])]
            else:
                self.reqs = [
                    req
                    for req in (
                        EvalRequest(x_init[i, :], None, 0) for i in range(x_init.shape[0])
                    )
                    if not _anyclose(req.parameters, self.x)
                ]
        self.runner: Optional[EpochRunner] = None
        self._runner_done: Optional[Dict] = None
        self.epoch_index = -1
        self.stats = {}

    # ------------------------------------------------------------- requests
    def append_request(self, req):
        self.reqs.append(req)

    def has_requests(self):
        return len(self.reqs) > 0

    def get_next_request(self):
        return self.reqs.pop(0) if self.reqs else None

    def complete_request(self, x, y, epoch=None, f=None, c=None, pred=None, time=-1.0):
        if x.shape[0] != self.prob.dim:
            raise ValueError(
                f"evaluation returned {x.shape[0]} parameters; the space has "
                f"{self.prob.dim}"
            )
        if y.shape[0] != self.prob.n_objectives:
            raise ValueError(
                f"objective function returned {y.shape[0]} values but "
                f"{self.prob.n_objectives} objective_names are declared "
                f"({self.prob.objective_names})"
            )
        if self.optimize_mean_variance and pred is not None:
            if pred.shape[0] == self.prob.n_objectives:
                pred = np.column_stack((pred, np.zeros_like(pred)))
        if (f is not None) and (np.ndim(f) == 1):
            f = np.reshape(f, (1, -1))
        entry = EvalEntry(epoch, x, y, f, c, pred, time)
        self.completed.append(entry)
        return entry

    def has_completed(self):
        return len(self.completed) > 0

    # -------------------------------------------------------------- archive
    def _remove_duplicate_evals(self):
        dup = ops.get_duplicates(torch.as_tensor(self.x, dtype=torch.float64)).numpy()
        self.x = self.x[~dup]
        self.y = self.y[~dup]
        if self.f is not None:
            self.f = self.f[~dup]
        if self.c is not None:
            self.c = self.c[~dup]

    def _reduce_evals(self):
        self._remove_duplicate_evals()
        perm, _, _ = ops.order_mo(
            torch.as_tensor(self.x, dtype=torch.float64),
            torch.as_tensor(self.y, dtype=torch.float64),
        )
        perm = perm.cpu().numpy()[: self.population_size]
        self.x = self.x[perm, :]
        self.y = self.y[perm, :]
        if self.c is not None:
            self.c = self.c[perm, :]
        if self.f is not None:
            self.f = self.f[perm]

    def _update_evals(self):
        result = None
        if len(self.completed) > 0 and not self.has_requests():
            x_c = np.vstack([e.parameters for e in self.completed])
            y_c = np.vstack([e.objectives for e in self.completed])
            n_obj = y_c.shape[1]
            y_pred = np.vstack(
                [
                    [np.nan] * n_obj if e.prediction is None else e.prediction
                    for e in self.completed
                ]
            )
            f_c = None
            if self.prob.n_features is not None:
                f_c = np.concatenate([e.features for e in self.completed], axis=0)
            c_c = None
            if self.prob.n_constraints is not None:
                c_c = np.vstack([e.constraints for e in self.completed])

            assert x_c.shape[1] == self.prob.dim
            assert y_c.shape[1] == self.prob.n_objectives
            if self.x is None:
                self.x, self.y, self.f, self.c = x_c, y_c, f_c, c_c
            else:
                self.x = np.vstack((self.x, x_c))
                self.y = np.vstack((self.y, y_c))
                if self.prob.n_features is not None:
                    self.f = np.concatenate((self.f, f_c), axis=0)
                if self.prob.n_constraints is not None:
                    self.c = np.vstack((self.c, c_c))
            t_c = np.vstack([e.time for e in self.completed])
            self.t = t_c if self.t is None else np.vstack((self.t, t_c))
            ts = self.t[self.t > 0.0]
            if len(ts) > 0:
                self.stats.update(
                    {
                        "eval_min": np.min(ts),
                        "eval_max": np.max(ts),
                        "eval_mean": np.mean(ts),
                        "eval_std": np.std(ts),
                        "eval_sum": np.sum(ts),
                        "eval_median": np.median(ts),
                    }
                )
            else:
                self.stats.update(
                    {k: -1 for k in ("eval_min", "eval_max", "eval_mean", "eval_std", "eval_sum", "eval_median")}
                )
            self._remove_duplicate_evals()
            self.completed = []
            result = x_c, y_c, y_pred, f_c, c_c
        return result

    # --------------------------------------------------------------- epochs
    def _epoch_kwargs(self, optimizer_index: int) -> Dict:
        optimizer_kwargs = {}
        # a single kwargs dict broadcasts across a cycled optimizer list
        # (the reference indexes 1:1 and crashes on the mismatch,
        # dmosopt.py:96-100 + epoch call site)
        kw = self.optimizer_kwargs[optimizer_index % len(self.optimizer_kwargs)]
        if kw is not None:
            optimizer_kwargs.update(kw)
        if self.distance_metric is not None:
            optimizer_kwargs["distance_metric"] = self.distance_metric
        return dict(
            num_generations=self.num_generations,
            param_names=self.prob.param_names,
            objective_names=self.prob.objective_names,
            xlb=self.prob.lb,
            xub=self.prob.ub,
            pct=self.resample_fraction,
            Xinit=self.x,
            Yinit=self.y,
            C=self.c,
            pop=self.population_size,
            optimizer_name=self.optimizer_name[optimizer_index],
            optimizer_kwargs=optimizer_kwargs,
            surrogate_method_name=self.surrogate_method_name,
            surrogate_method_kwargs=self.surrogate_method_kwargs,
            surrogate_custom_training=self.surrogate_custom_training,
            surrogate_custom_training_kwargs=self.surrogate_custom_training_kwargs,
            sensitivity_method_name=self.sensitivity_method_name,
            sensitivity_method_kwargs=self.sensitivity_method_kwargs,
            feasibility_method_name=self.feasibility_method_name,
            feasibility_method_kwargs=self.feasibility_method_kwargs,
            optimize_mean_variance=self.optimize_mean_variance,
            termination=self.termination,
            local_random=self.local_random,
            logger=self.logger,
            file_path=self.file_path,
            device=self.device,
        )

    def initialize_epoch(self, epoch_index: int):
        assert self.runner is None, "Epoch already active in DistOptStrategy"
        optimizer_index = next(self.optimizer_iter)
        self._update_evals()
        assert epoch_index > self.epoch_index
        self.epoch_index = epoch_index
        self.runner = EpochRunner(self, optimizer_index, self._epoch_kwargs(optimizer_index))
        state, value = self.runner.start()
        if state == "done":
            self._runner_done = value
        else:
            self._reduce_evals()
            x_gen = value
            for i in range(x_gen.shape[0]):
                self.append_request(EvalRequest(x_gen[i, :], None, self.epoch_index))

    def _finish_epoch(self, result_dict: Dict, resample: bool):
        self.stats.update(result_dict.get("stats", {}))
        self.runner = None
        self._runner_done = None
        if "best_x" in result_dict:
            return (
                StrategyState.CompletedEpoch,
                EpochResults(
                    result_dict["best_x"],
                    result_dict["best_y"],
                    result_dict["gen_index"],
                    result_dict["x"],
                    result_dict["y"],
                    result_dict["optimizer"],
                ),
            )
        x_resample = result_dict["x_resample"]
        y_pred = result_dict["y_pred"]
        if resample and x_resample is not None:
            for i in range(x_resample.shape[0]):
                self.append_request(
                    EvalRequest(x_resample[i, :], y_pred[i], self.epoch_index + 1)
                )
        return (
            StrategyState.CompletedEpoch,
            EpochResults(
                x_resample,
                y_pred,
                result_dict["gen_index"],
                result_dict["x_sm"],
                result_dict["y_sm"],
                result_dict["optimizer"],
            ),
        )

    def update_epoch(self, resample: bool = False):
        assert self.runner is not None or self._runner_done is not None, "Epoch not initialized"
        completed_evals = self._update_evals()

        if self._runner_done is not None:
            state, value = self._finish_epoch(self._runner_done, resample)
            return state, value, completed_evals

        if completed_evals is None:
            if self.has_requests():
                return StrategyState.WaitingRequests, None, completed_evals
            # surrogate-free runner waiting for results that never arrived:
            # nothing to do (matches reference falling through to next(gen))
            return StrategyState.WaitingRequests, None, completed_evals

        x_gen, y_gen = completed_evals[0], completed_evals[1]
        c_gen = completed_evals[4]
        state, value = self.runner.step(x_gen, y_gen, c_gen)
        if state == "done":
            st, val = self._finish_epoch(value, resample)
            return st, val, completed_evals
        self._reduce_evals()
        for i in range(value.shape[0]):
            self.append_request(EvalRequest(value[i, :], None, self.epoch_index))
        return StrategyState.EnqueuedRequests, value, completed_evals

    # ----------------------------------------------------------------- best
    def get_best_evals(self, feasible: bool = True):
        if self.x is not None:
            bestx, besty, bestf, bestc, beste, perm = engine.get_best(
                self.x, self.y, self.f, self.c, self.prob.dim, self.prob.n_objectives,
                feasible=feasible,
            )
            return bestx, besty, self.prob.feature_constructor(bestf), bestc
        return None, None, None, None

    def get_evals(self, return_features=False, return_constraints=False):
        if return_features and return_constraints:
            return (self.x, self.y, self.f, self.c)
        if return_features:
            return (self.x, self.y, self.f)
        if return_constraints:
            return (self.x, self.y, self.c)
        return (self.x, self.y)

    def get_completed(self):
        if len(self.completed) > 0:
            x_c = [e.parameters for e in self.completed]
            y_c = [e.objectives for e in self.completed]
            f_c = [e.features for e in self.completed] if self.prob.n_features else None
            c_c = [e.constraints for e in self.completed] if self.prob.n_constraints else None
            return (x_c, y_c, f_c, c_c)
        return None
