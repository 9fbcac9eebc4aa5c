"""Core termination criteria (parity with reference termination.py:14-347)."""

from __future__ import annotations

from abc import abstractmethod

import numpy as np

from dmosopt_amd.hv.indicators import IGD, SlidingWindow
from dmosopt_amd.normalization import normalize


def _log(problem, msg):
    lg = getattr(problem, "logger", None)
    if lg is not None:
        lg.info(msg)


class Termination:
    def __init__(self, problem) -> None:
        self.problem = problem
        self.force_termination = False

    def do_continue(self, opt):
        if self.force_termination:
            return False
        return self._do_continue(opt)

    def _do_continue(self, opt, **kwargs):
        pass

    def has_terminated(self, opt):
        return not self.do_continue(opt)


class TerminationCollection(Termination):
    def __init__(self, problem, *args) -> None:
        super().__init__(problem)
        self.terminations = args

    def _do_continue(self, opt):
        for term in self.terminations:
            if not term.do_continue(opt):
                return False
        return True


class MaximumGenerationTermination(Termination):
    def __init__(self, problem, n_max_gen) -> None:
        super().__init__(problem)
        self.n_max_gen = n_max_gen if n_max_gen is not None else float("inf")

    def _do_continue(self, opt):
        if opt.n_gen > self.n_max_gen:
            _log(self.problem, f"Optimization terminated: maximum generations ({opt.n_gen}) reached")
        return opt.n_gen <= self.n_max_gen


class SlidingWindowTermination(TerminationCollection):
    """Store -> metric -> decide pipeline over sliding windows with an
    nth_gen cadence and a min_generations floor persistent across epochs
    (reference termination.py:90-207)."""

    def __init__(
        self,
        problem,
        metric_window_size=None,
        data_window_size=None,
        min_data_for_metric=1,
        nth_gen=1,
        n_max_gen=None,
        min_generations=0,
        truncate_metrics=True,
        truncate_data=True,
    ):
        super().__init__(problem, MaximumGenerationTermination(problem, n_max_gen=n_max_gen))
        self.data_window_size = data_window_size
        self.metric_window_size = metric_window_size
        self.truncate_data = truncate_data
        self.data = SlidingWindow(data_window_size) if truncate_data else []
        self.truncate_metrics = truncate_metrics
        self.metrics = SlidingWindow(metric_window_size) if truncate_metrics else []
        self.nth_gen = nth_gen
        self.min_data_for_metric = min_data_for_metric
        self.min_generations = min_generations
        self.n_total_gens = 0

    def reset(self):
        self.data = SlidingWindow(self.data_window_size) if self.truncate_data else []
        self.metrics = SlidingWindow(self.metric_window_size) if self.truncate_metrics else []

    def _do_continue(self, opt):
        if not super()._do_continue(opt):
            return False
        self.n_total_gens += 1
        obj = self._store(opt)
        if obj is not None:
            self.data.append(obj)
        if len(self.data) >= self.min_data_for_metric:
            metric = self._metric(self.data[-self.data_window_size :])
            if metric is not None:
                self.metrics.append(metric)
        if self.n_total_gens < self.min_generations:
            return True
        if opt.n_gen % self.nth_gen == 0 and len(self.metrics) >= self.metric_window_size:
            return self._decide(self.metrics[-self.metric_window_size :])
        return True

    def _store(self, opt):
        return opt

    @abstractmethod
    def _decide(self, metrics):
        ...

    @abstractmethod
    def _metric(self, data):
        ...

    def get_metric(self):
        return self.metrics[-1] if len(self.metrics) > 0 else None


class ParameterToleranceTermination(SlidingWindowTermination):
    """IGD between consecutive normalized X populations <= tol."""

    def __init__(self, problem, n_last=10, tol=1e-6, nth_gen=1, n_max_gen=None, **kwargs):
        super().__init__(
            problem,
            metric_window_size=n_last,
            data_window_size=2,
            min_data_for_metric=2,
            nth_gen=nth_gen,
            n_max_gen=n_max_gen,
            **kwargs,
        )
        self.tol = tol

    def _store(self, opt):
        X = opt.x
        if X.dtype != object:
            if self.problem.lb is not None and self.problem.ub is not None:
                X = normalize(X, xl=self.problem.lb, xu=self.problem.ub)
            return X

    def _metric(self, data):
        last, current = data[-2], data[-1]
        return IGD(current).do(last)

    def _decide(self, metrics):
        mean = np.asarray(metrics).mean()
        if mean <= self.tol:
            _log(self.problem, f"Optimization terminated: mean parameter distance {mean} below {self.tol}")
        return mean > self.tol


def calc_delta_norm(a, b, norm):
    return np.max(np.abs((a - b) / norm))


class MultiObjectiveToleranceTermination(SlidingWindowTermination):
    """Delta-ideal + IGD of normalized F windows <= tol (default 0.0025)."""

    def __init__(self, problem, tol=0.0025, n_last=10, nth_gen=1, n_max_gen=None, **kwargs):
        super().__init__(
            problem,
            metric_window_size=n_last,
            data_window_size=2,
            min_data_for_metric=2,
            nth_gen=nth_gen,
            n_max_gen=n_max_gen,
            **kwargs,
        )
        self.tol = tol

    def _store(self, opt):
        F = opt.y
        return {"ideal": F.min(axis=0), "nadir": F.max(axis=0), "F": F}

    def _metric(self, data):
        last, current = data[-2], data[-1]
        norm = current["nadir"] - current["ideal"]
        norm[norm < 1e-32] = 1
        delta_ideal = calc_delta_norm(current["ideal"], last["ideal"], norm)
        c_F, c_ideal, c_nadir = current["F"], current["ideal"], current["nadir"]
        c_N = normalize(c_F, c_ideal, c_nadir)
        l_N = normalize(last["F"], c_ideal, c_nadir)
        delta_f = IGD(c_N).do(l_N)
        return {"delta_ideal": delta_ideal, "delta_f": delta_f}

    def _decide(self, metrics):
        delta_ideal = [e["delta_ideal"] for e in metrics]
        delta_f = [e["delta_f"] for e in metrics]
        max_delta = max(np.mean(delta_ideal), np.mean(delta_f))
        if max_delta <= self.tol:
            _log(
                self.problem,
                f"Optimization terminated: objective mean delta "
                f"{(np.mean(delta_ideal), np.mean(delta_f))} below {self.tol}",
            )
        else:
            _log(self.problem, f"Objective mean delta: {(np.mean(delta_ideal), np.mean(delta_f))}")
        return max_delta > self.tol


class ConstraintViolationToleranceTermination(SlidingWindowTermination):
    def __init__(self, problem, n_last=10, tol=1e-6, nth_gen=1, n_max_gen=None, **kwargs):
        super().__init__(
            problem,
            metric_window_size=n_last,
            data_window_size=2,
            min_data_for_metric=2,
            nth_gen=nth_gen,
            n_max_gen=n_max_gen,
            **kwargs,
        )
        self.tol = tol

    def _store(self, opt):
        return opt.c

    def _metric(self, data):
        last, current = data[-2], data[-1]
        return {"cv": current, "delta_cv": abs(last - current)}

    def _decide(self, metrics):
        cv = np.asarray([e["cv"] for e in metrics])
        delta_cv = np.asarray([e["delta_cv"] for e in metrics])
        n_feasible = (cv > 0).sum()
        if n_feasible == len(metrics):
            return False
        if 0 < n_feasible < len(metrics):
            return True
        return delta_cv.max() > self.tol
