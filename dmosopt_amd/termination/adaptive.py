"""Adaptive stopping criteria.

Role parity with the reference's ``adaptive_termination.py`` (SURVEY.md
section 2.7) on the same decision semantics, built on the
capture/reduce/verdict window engine of :mod:`.basic` with vectorized
per-objective / per-timescale state (numpy arrays instead of per-objective
state objects).
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from dmosopt_amd.hv.indicators import crowding_distance_metric
from dmosopt_amd.termination.basic import (
    MaximumGenerationTermination,
    Termination,
    TerminationCollection,
    WindowedCriterion,
    Windows,
    _log,
)
from dmosopt_amd.termination.hv_progress import HypervolumeProgressTermination

#: consecutive sub-tolerance windows before an objective counts as settled
_SETTLE_STREAK = 3


def _norm_span(F: np.ndarray) -> np.ndarray:
    span = F.max(axis=0) - F.min(axis=0)
    span[span < 1e-32] = 1.0
    return span


class PerObjectiveConvergence(WindowedCriterion):
    """Track each objective's normalized ideal-point movement separately and
    stop once a fraction of the objectives have settled.

    An objective settles after ``_SETTLE_STREAK`` consecutive windows whose
    mean movement is below ``obj_tol``; any louder window unsettles it. All
    per-objective state is vectorized: a rolling (n_last, m) delta matrix
    plus integer streak counters.
    """

    def __init__(
        self,
        problem,
        obj_tol: float = 1e-4,
        min_converged_fraction: float = 0.8,
        n_last: int = 20,
        nth_gen: int = 5,
        n_max_gen: Optional[int] = None,
        min_generations: int = 0,
        **kw,
    ):
        super().__init__(
            problem,
            Windows(raw=2, signal=n_last, warmup=2, cadence=nth_gen,
                    ceiling=n_max_gen, floor=min_generations, **kw),
        )
        m = problem.n_objectives
        self.obj_tol = obj_tol
        self.min_converged_fraction = min_converged_fraction
        self._deltas = np.empty((0, m))  # rolling per-objective movement rows
        self._streaks = np.zeros(m, dtype=np.int64)
        self._settled = np.zeros(m, dtype=bool)
        self._last_rates = np.zeros(m)

    def capture(self, opt):
        return np.asarray(opt.y)

    def reduce(self, raws):
        prev_F, curr_F = raws[-2], raws[-1]
        m = self._settled.shape[0]
        move = np.abs(curr_F.min(axis=0) - prev_F.min(axis=0)) / _norm_span(curr_F)
        move = move[:m]
        n_last = self.windows.signal
        self._deltas = np.vstack([self._deltas, move[None, :]])[-n_last:]
        if self._deltas.shape[0] >= n_last:
            rates = self._deltas.mean(axis=0)
            self._last_rates = rates
            quiet = rates < self.obj_tol
            self._streaks = np.where(quiet, self._streaks + 1, 0)
            self._settled = np.where(
                quiet, self._streaks >= _SETTLE_STREAK, False
            )
        return move

    def verdict(self, signals):
        m = self._settled.shape[0]
        n_done = int(self._settled.sum())
        share = n_done / m
        if share >= self.min_converged_fraction:
            _log(self.problem,
                 f"stop: {n_done} of {m} objectives settled ({share:.0%})")
            return True
        _log(self.problem,
             f"settling: {n_done}/{m} objectives ({share:.0%}), "
             f"mean movement {float(self._last_rates.mean()):.2e}")
        return False


class MultiScaleStagnationTermination(WindowedCriterion):
    """Compare the current ideal point / diversity against lagged snapshots
    at several timescales at once; stop when enough scales are quiet."""

    def __init__(
        self,
        problem,
        timescales: List[int] = [5, 10, 20, 40],
        stagnation_tol: float = 1e-4,
        min_scales_stagnant: int = 3,
        n_max_gen: Optional[int] = None,
        nth_gen: int = 1,
        min_generations: int = 0,
        **kw,
    ):
        horizon = max(timescales)
        super().__init__(
            problem,
            Windows(raw=horizon + 1, signal=horizon, warmup=2, cadence=nth_gen,
                    ceiling=n_max_gen, floor=min_generations, **kw),
        )
        self.timescales = np.asarray(sorted(timescales))
        self.stagnation_tol = stagnation_tol
        self.min_scales_stagnant = min_scales_stagnant

    def capture(self, opt):
        F = np.asarray(opt.y)
        return {
            "ideal": F.min(axis=0),
            "span": _norm_span(F),
            "spread": float(np.mean(crowding_distance_metric(F))),
        }

    def reduce(self, raws):
        now = raws[-1]
        depth = len(raws) - 1
        lags = self.timescales[self.timescales <= depth]
        if lags.size == 0:
            return None
        # vectorized over available scales: lagged ideal points as a matrix
        past_ideals = np.stack([raws[-(int(s) + 1)]["ideal"] for s in lags])
        moves = np.mean(np.abs(now["ideal"][None, :] - past_ideals) / now["span"], axis=1)
        return {"scales": lags, "moves": moves}

    def verdict(self, signals):
        sig = signals[-1]
        if sig is None or sig["scales"].size < self.timescales.size:
            return False  # wait until every timescale has lag data
        quiet = sig["moves"] < self.stagnation_tol
        if int(quiet.sum()) >= self.min_scales_stagnant:
            _log(self.problem,
                 f"stop: {int(quiet.sum())} of {len(self.timescales)} timescales "
                 f"quiet (scales {sig['scales'][quiet].tolist()})")
            return True
        return False


class AdaptiveWindowTermination(WindowedCriterion):
    """Patience window that grows while progress is still loud: judged over
    ``current`` signals, and expanded by ``expansion_rate`` (up to
    ``max_window``) whenever the window mean exceeds 10x tolerance."""

    def __init__(
        self,
        problem,
        initial_window: int = 10,
        max_window: int = 50,
        expansion_rate: float = 1.2,
        tol: float = 1e-4,
        n_max_gen: Optional[int] = None,
        **kw,
    ):
        super().__init__(
            problem,
            Windows(raw=2, signal=initial_window, warmup=2, cadence=1,
                    ceiling=n_max_gen, floor=kw.pop("min_generations", 0), **kw),
        )
        self.max_window = max_window
        self.expansion_rate = expansion_rate
        self.tol = tol
        # the signal ring must be able to hold the fully-expanded window
        self._signal_ring.size = max_window

    @property
    def current_window_size(self) -> int:
        return self.windows.signal

    def reduce(self, raws):
        prev_F, curr_F = np.asarray(raws[-2].y), np.asarray(raws[-1].y)
        return float(
            np.mean(np.abs(curr_F.min(axis=0) - prev_F.min(axis=0)) / _norm_span(curr_F))
        )

    def verdict(self, signals):
        width = self.windows.signal
        if len(signals) < width:
            return False
        level = float(np.mean(signals[-width:]))
        if level > self.tol * 10:
            wider = min(int(width * self.expansion_rate), self.max_window)
            if wider > width:
                self.windows.signal = wider
                _log(self.problem, f"patience window widened to {wider}")
        if level < self.tol:
            _log(self.problem,
                 f"stop: movement {level:.2e} below {self.tol:.2e} "
                 f"over the last {width} generations")
            return True
        return False


class CompositeAdaptiveTermination(TerminationCollection):
    """Any-of bundle for high-dimensional runs: generation cap plus the
    per-objective, hypervolume-progress and multi-scale criteria, each
    individually switchable."""

    def __init__(
        self,
        problem,
        n_max_gen: int = 2000,
        obj_tol: float = 1e-4,
        min_converged_fraction: float = 0.8,
        hv_tol: float = 1e-5,
        ref_point: Optional[np.ndarray] = None,
        timescales: Optional[List[int]] = None,
        stagnation_tol: float = 1e-4,
        min_generations: int = 50,
        use_per_objective: bool = True,
        use_hypervolume: bool = True,
        use_multiscale: bool = True,
        **kw,
    ):
        members: List[Termination] = [
            MaximumGenerationTermination(problem, n_max_gen=n_max_gen)
        ]
        if use_per_objective:
            members.append(PerObjectiveConvergence(
                problem, obj_tol=obj_tol,
                min_converged_fraction=min_converged_fraction,
                n_last=20, nth_gen=5, min_generations=min_generations, **kw,
            ))
        if use_hypervolume:
            members.append(HypervolumeProgressTermination(
                problem, ref_point=ref_point, hv_tol=hv_tol,
                n_last=15, nth_gen=5, min_generations=min_generations, **kw,
            ))
        if use_multiscale:
            if timescales is None:
                base = max(5, problem.n_objectives // 5)
                timescales = [base << i for i in range(4)]
            members.append(MultiScaleStagnationTermination(
                problem, timescales=timescales, stagnation_tol=stagnation_tol,
                min_scales_stagnant=3, nth_gen=2,
                min_generations=min_generations, **kw,
            ))
        super().__init__(problem, *members)
        _log(problem, f"composite termination armed with {len(members)} criteria")


class ResourceAwareTermination(Termination):
    """Hard resource budget: wall-clock seconds, evaluation count, or a
    quality threshold — whichever trips first."""

    def __init__(
        self,
        problem,
        max_time_seconds: Optional[float] = None,
        max_function_evals: Optional[int] = None,
        target_quality_threshold: Optional[float] = None,
        **kw,
    ):
        super().__init__(problem)
        self.max_time_seconds = max_time_seconds
        self.max_function_evals = max_function_evals
        self.target_quality_threshold = target_quality_threshold
        self._armed_at: Optional[float] = None

    def _stop(self, opt) -> bool:
        if self._armed_at is None:
            self._armed_at = _time.time()
        if self.max_time_seconds is not None:
            used = _time.time() - self._armed_at
            over = used > self.max_time_seconds
            # multi-rank: local clocks diverge and a split decision would
            # hang the replicated control flow — rank 0's clock decides
            from dmosopt_amd.parallel.context import get_context

            ctx = get_context()
            if ctx is not None and ctx.world > 1:
                over = ctx.bcast_flag(over, src=0)
            if over:
                _log(self.problem, f"stop: wall-clock budget used ({used:.1f}s)")
                return True
        if self.max_function_evals is not None:
            spent = getattr(opt, "n_eval", None) or getattr(opt, "n_gen", 0)
            if spent and spent > self.max_function_evals:
                _log(self.problem, f"stop: evaluation budget used ({spent})")
                return True
        if self.target_quality_threshold is not None:
            q = getattr(opt, "quality_metric", None)
            if q is not None and q > self.target_quality_threshold:
                _log(self.problem, "stop: target quality reached")
                return True
        return False

    # compat: the reference exposes start_time
    @property
    def start_time(self):
        return self._armed_at


_PRESETS = {
    "comprehensive": dict(use_per_objective=True, use_hypervolume=True,
                          use_multiscale=True, hv_tol=1e-6),
    "fast": dict(use_per_objective=False, use_hypervolume=True,
                 use_multiscale=True),
    "conservative": dict(use_per_objective=True, use_hypervolume=False,
                         use_multiscale=True),
}


def create_adaptive_termination(
    problem, n_max_gen: int = 2000, strategy: str = "comprehensive", **kw
) -> Termination:
    """Factory for the preset strategies comprehensive|fast|conservative|
    simple (same preset names and compositions as the reference)."""
    if strategy == "simple":
        return HypervolumeProgressTermination(
            problem, n_last=20, nth_gen=5, n_max_gen=n_max_gen, **kw
        )
    if strategy not in _PRESETS:
        raise ValueError(
            f"Unknown strategy {strategy!r}; choose comprehensive|fast|conservative|simple"
        )
    return CompositeAdaptiveTermination(problem, n_max_gen=n_max_gen, **_PRESETS[strategy], **kw)
