"""Generation-loop stopping criteria.

Role parity with the reference's ``termination.py``:
the same stopping decisions over the same ``OptHistory`` inputs, but
restructured around ONE windowed-pipeline engine instead of an inheritance
chain. Every windowed criterion is three stages over ring buffers::

    capture(opt)      -> raw snapshot appended to the raw ring
    reduce(raw tail)  -> derived signal appended to the signal ring
    verdict(signals)  -> stop / keep going, judged on a cadence

configured declaratively by a :class:`Windows` dataclass (raw/signal ring
sizes, warmup, cadence, a cross-epoch generation floor and an absolute
generation ceiling). The concrete criteria below plug formulas into that
engine; the formulas themselves (IGD drift, normalized ideal-point delta,
constraint-violation transitions) match the reference semantics.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, List, Optional

import numpy as np

from dmosopt_amd.hv.indicators import IGD, SlidingWindow
from dmosopt_amd.normalization import normalize


def _log(problem, msg):
    lg = getattr(problem, "logger", None)
    if lg is not None:
        lg.info(msg)


class Termination:
    """Base protocol: ``has_terminated(opt) -> bool`` over an OptHistory.

    ``force_termination`` short-circuits to True (kept for API parity with
    the reference's base class).
    """

    def __init__(self, problem) -> None:
        self.problem = problem
        self.force_termination = False

    def has_terminated(self, opt) -> bool:
        if self.force_termination:
            return True
        return self._stop(opt)

    def _stop(self, opt) -> bool:
        return False

    # reference-compatible inverse form
    def do_continue(self, opt) -> bool:
        return not self.has_terminated(opt)


class TerminationCollection(Termination):
    """Any-of composition: stops as soon as one member criterion stops."""

    def __init__(self, problem, *criteria) -> None:
        super().__init__(problem)
        self.terminations = tuple(criteria)

    def _stop(self, opt) -> bool:
        return any(t.has_terminated(opt) for t in self.terminations)


class MaximumGenerationTermination(Termination):
    def __init__(self, problem, n_max_gen) -> None:
        super().__init__(problem)
        self.n_max_gen = float("inf") if n_max_gen is None else n_max_gen

    def _stop(self, opt) -> bool:
        if opt.n_gen > self.n_max_gen:
            _log(self.problem, f"stop: generation cap hit at gen {opt.n_gen}")
            return True
        return False


@dataclass
class Windows:
    """Ring-buffer and cadence configuration for a WindowedCriterion.

    raw:     how many captures the raw ring keeps (None = unbounded)
    signal:  signal-ring size; also how many signals the verdict needs/sees
    warmup:  captures required before reduce() runs
    cadence: judge only every cadence-th generation
    floor:   total captures (across epochs) before any stop verdict counts
    ceiling: absolute generation cap (None = unlimited)
    """

    raw: Optional[int] = None
    signal: Optional[int] = None
    warmup: int = 1
    cadence: int = 1
    floor: int = 0
    ceiling: Optional[int] = None


class WindowedCriterion(Termination):
    """The capture -> reduce -> verdict pipeline over two ring buffers.

    Subclasses implement :meth:`capture`, :meth:`reduce` and
    :meth:`verdict`. The engine keeps a cross-epoch capture counter so a
    ``floor`` survives epoch restarts (the rings themselves can be cleared
    with :meth:`reset` between epochs).
    """

    def __init__(self, problem, windows: Windows):
        super().__init__(problem)
        self.windows = windows
        self._lifetime_captures = 0
        self._alloc_rings()

    def _alloc_rings(self):
        w = self.windows
        self._raw_ring: List[Any] = SlidingWindow(w.raw) if w.raw is not None else []
        self._signal_ring: List[Any] = (
            SlidingWindow(w.signal) if w.signal is not None else []
        )

    def reset(self):
        """Clear the rings; the lifetime capture counter (floor) persists."""
        self._alloc_rings()

    # ---- stages ----------------------------------------------------------
    def capture(self, opt) -> Any:
        return opt

    def reduce(self, raws: List[Any]) -> Any:
        raise NotImplementedError

    def verdict(self, signals: List[Any]) -> bool:
        raise NotImplementedError

    # ---- engine ----------------------------------------------------------
    def _stop(self, opt) -> bool:
        w = self.windows
        if w.ceiling is not None and opt.n_gen > w.ceiling:
            _log(self.problem, f"stop: generation cap hit at gen {opt.n_gen}")
            return True
        self._lifetime_captures += 1
        snap = self.capture(opt)
        if snap is not None:
            self._raw_ring.append(snap)
        if len(self._raw_ring) >= w.warmup:
            tail = self._raw_ring if w.raw is None else self._raw_ring[-w.raw :]
            sig = self.reduce(tail)
            if sig is not None:
                self._signal_ring.append(sig)
        if self._lifetime_captures < w.floor:
            return False
        need = w.signal if w.signal is not None else 1
        if opt.n_gen % w.cadence == 0 and len(self._signal_ring) >= need:
            tail = (
                self._signal_ring
                if w.signal is None
                else self._signal_ring[-w.signal :]
            )
            return self.verdict(tail)
        return False

    def latest_signal(self):
        return self._signal_ring[-1] if self._signal_ring else None


# kept as the exported name the reference uses for this role
SlidingWindowTermination = WindowedCriterion


class ParameterToleranceTermination(WindowedCriterion):
    """Stop when consecutive normalized parameter populations stop moving:
    mean IGD(X_t -> X_{t-1}) over a window falls to/below ``tol``."""

    def __init__(self, problem, n_last=10, tol=1e-6, nth_gen=1, n_max_gen=None, **kw):
        super().__init__(
            problem,
            Windows(raw=2, signal=n_last, warmup=2, cadence=nth_gen,
                    ceiling=n_max_gen, floor=kw.pop("min_generations", 0), **kw),
        )
        self.tol = tol

    def capture(self, opt):
        X = opt.x
        if X.dtype == object:
            return None
        lb, ub = self.problem.lb, self.problem.ub
        return normalize(X, xl=lb, xu=ub) if lb is not None and ub is not None else X

    def reduce(self, raws):
        prev, curr = raws[-2], raws[-1]
        return IGD(curr).do(prev)

    def verdict(self, signals):
        drift = float(np.mean(signals))
        if drift <= self.tol:
            _log(self.problem,
                 f"stop: parameter drift {drift:.3e} within tolerance {self.tol:.3e}")
            return True
        return False


class MultiObjectiveToleranceTermination(WindowedCriterion):
    """Stop when both the normalized ideal-point shift and the IGD drift of
    consecutive objective populations average at/below ``tol``."""

    def __init__(self, problem, tol=0.0025, n_last=10, nth_gen=1, n_max_gen=None, **kw):
        super().__init__(
            problem,
            Windows(raw=2, signal=n_last, warmup=2, cadence=nth_gen,
                    ceiling=n_max_gen, floor=kw.pop("min_generations", 0), **kw),
        )
        self.tol = tol

    def capture(self, opt):
        return np.asarray(opt.y)

    def reduce(self, raws):
        prev_F, curr_F = raws[-2], raws[-1]
        ideal, nadir = curr_F.min(axis=0), curr_F.max(axis=0)
        span = nadir - ideal
        span[span < 1e-32] = 1.0
        ideal_shift = float(np.max(np.abs((curr_F.min(axis=0) - prev_F.min(axis=0)) / span)))
        front_drift = float(
            IGD(normalize(curr_F, ideal, nadir)).do(normalize(prev_F, ideal, nadir))
        )
        return np.array([ideal_shift, front_drift])

    def verdict(self, signals):
        means = np.mean(np.vstack(signals), axis=0)  # [ideal shift, front drift]
        worst = float(means.max())
        if worst <= self.tol:
            _log(self.problem,
                 f"stop: objective drift (ideal {means[0]:.3e}, front {means[1]:.3e}) "
                 f"within tolerance {self.tol:.3e}")
            return True
        _log(self.problem,
             f"objective drift: ideal {means[0]:.3e}, front {means[1]:.3e}")
        return False


class ConstraintViolationToleranceTermination(WindowedCriterion):
    """Constraint-violation window logic: stop while everything in the
    window is feasible, keep going through a feasibility transition, and
    otherwise stop only when the violation level has stopped changing."""

    def __init__(self, problem, n_last=10, tol=1e-6, nth_gen=1, n_max_gen=None, **kw):
        super().__init__(
            problem,
            Windows(raw=2, signal=n_last, warmup=2, cadence=nth_gen,
                    ceiling=n_max_gen, floor=kw.pop("min_generations", 0), **kw),
        )
        self.tol = tol

    def capture(self, opt):
        return opt.c

    def reduce(self, raws):
        prev, curr = raws[-2], raws[-1]
        return (curr, abs(curr - prev))

    def verdict(self, signals):
        cv = np.asarray([s[0] for s in signals])
        dcv = np.asarray([s[1] for s in signals])
        n_ok = int((cv > 0).sum())
        if n_ok == len(signals):
            return True  # fully feasible window
        if n_ok > 0:
            return False  # mid-transition: keep optimizing
        return bool(dcv.max() <= self.tol)
