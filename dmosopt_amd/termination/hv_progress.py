"""Hypervolume-progress termination with multi-fidelity tracking.

Compact re-implementation of the reference hv_termination.py machinery
(ProgressivePrecisionScheduler :90, HVAlgorithmRouter :225,
MultiFidelityHVTracker :446, ConvergenceDetector :684,
HypervolumeProgressTermination :960): progressive coarse->medium->fine
precision, dimension-based algorithm routing (exact box decomposition for
low d, Monte Carlo for high d), cadence per fidelity, and stagnation
detection over a sliding window of best-fidelity HV estimates.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import numpy as np

from dmosopt_amd.hv.adaptive import AdaptiveHyperVolume
from dmosopt_amd.termination.basic import WindowedCriterion, Windows, _log


@dataclass
class HVEstimate:
    value: float
    fidelity: str
    generation: int
    eps: Optional[float] = None


class ProgressivePrecisionScheduler:
    """Coarse/medium/fine epsilon and cadence by generation with progress
    adaptation (reference hv_termination.py:90-223)."""

    def __init__(self, coarse_until: int = 50, medium_until: int = 150):
        self.coarse_until = coarse_until
        self.medium_until = medium_until
        self.configs = {
            "coarse": {"eps": 0.05, "cadence": 2},
            "medium": {"eps": 0.02, "cadence": 5},
            "fine": {"eps": 0.005, "cadence": 10},
        }
        self._boost = 0

    def fidelity_for(self, generation: int) -> str:
        g = generation + self._boost
        if g < self.coarse_until:
            return "coarse"
        if g < self.medium_until:
            return "medium"
        return "fine"

    def get_precision_config(self, generation: int) -> Dict[str, Any]:
        fid = self.fidelity_for(generation)
        cfg = dict(self.configs[fid])
        cfg["fidelity"] = fid
        return cfg

    def adapt_to_progress(self, rel_improvement: float):
        # near-stagnant progress escalates precision sooner
        if rel_improvement < 1e-4:
            self._boost += 20
        elif rel_improvement > 1e-2:
            self._boost = max(0, self._boost - 10)


class HVAlgorithmRouter:
    """Pick exact vs MC by dimensionality (reference hv_termination.py:225)."""

    def __init__(self, box_dim_threshold: int = 10, reduced_dim_threshold: int = 20):
        self.box_dim_threshold = box_dim_threshold
        self.reduced_dim_threshold = reduced_dim_threshold

    def select_algorithm(self, n_objectives: int, n_points: int, eps: float) -> str:
        if n_objectives < self.box_dim_threshold:
            return "box"
        if n_objectives < self.reduced_dim_threshold:
            return "adaptive_mc"
        return "reduced_mc"

    def compute_hypervolume(self, front: np.ndarray, ref: np.ndarray, eps: float) -> float:
        d = front.shape[1]
        algo = self.select_algorithm(d, len(front), eps)
        if algo == "box":
            return AdaptiveHyperVolume(ref, mc_dim_threshold=10**9).compute(front)
        if algo == "adaptive_mc":
            return AdaptiveHyperVolume(ref, mc_dim_threshold=0, mc_eps=eps).compute(front)
        # reduced MC: project to the most-varying 8 objectives
        spread = front.max(axis=0) - front.min(axis=0)
        keep = np.argsort(-spread)[:8]
        return AdaptiveHyperVolume(ref[keep], mc_dim_threshold=0, mc_eps=eps).compute(
            front[:, keep]
        )


class MultiFidelityHVTracker:
    """Cadenced HV estimates per fidelity (reference hv_termination.py:446)."""

    def __init__(self, ref_point: np.ndarray, scheduler: ProgressivePrecisionScheduler,
                 router: HVAlgorithmRouter):
        self.ref_point = np.asarray(ref_point, dtype=np.float64)
        self.scheduler = scheduler
        self.router = router
        self.estimates: List[HVEstimate] = []
        self._last_gen_by_fidelity: Dict[str, int] = {}

    def should_compute(self, generation: int, fidelity: str) -> bool:
        cadence = self.scheduler.configs[fidelity]["cadence"]
        last = self._last_gen_by_fidelity.get(fidelity, -10**9)
        return generation - last >= cadence

    def compute_and_update(self, front: np.ndarray, generation: int) -> Optional[HVEstimate]:
        cfg = self.scheduler.get_precision_config(generation)
        fid = cfg["fidelity"]
        if not self.should_compute(generation, fid):
            return None
        val = self.router.compute_hypervolume(front, self.ref_point, cfg["eps"])
        est = HVEstimate(val, fid, generation, cfg["eps"])
        self.estimates.append(est)
        self._last_gen_by_fidelity[fid] = generation
        return est

    def get_best_estimate(self) -> Optional[HVEstimate]:
        return self.estimates[-1] if self.estimates else None


class ConvergenceDetector:
    """Stagnation + relative-threshold + confidence (hv_termination.py:684)."""

    def __init__(self, stagnation_threshold=1e-6, stagnation_window=5,
                 relative_threshold=1e-7, min_generations=20):
        self.stagnation_threshold = stagnation_threshold
        self.stagnation_window = stagnation_window
        self.relative_threshold = relative_threshold
        self.min_generations = min_generations

    def check_convergence(self, estimates: List[HVEstimate], generation: int) -> Dict:
        if generation < self.min_generations or len(estimates) < self.stagnation_window:
            return {"converged": False, "confidence": 0.0, "reason": "insufficient-data"}
        window = estimates[-self.stagnation_window :]
        vals = np.array([e.value for e in window])
        base = max(abs(vals[-1]), 1e-300)
        abs_improve = np.max(vals) - np.min(vals)
        rel_improve = abs_improve / base
        stagnant = abs_improve < self.stagnation_threshold or rel_improve < self.relative_threshold
        # confidence grows with estimate precision and window consistency
        fine = sum(1 for e in window if e.fidelity == "fine") / len(window)
        consistency = 1.0 - min(1.0, float(np.std(vals)) / base)
        confidence = 0.5 * fine + 0.5 * consistency if stagnant else 0.0
        return {
            "converged": bool(stagnant and confidence > 0.5),
            "confidence": float(confidence),
            "rel_improvement": float(rel_improve),
            "abs_improvement": float(abs_improve),
        }


class HypervolumeProgressTermination(WindowedCriterion):
    """Terminate when best-fidelity hypervolume stops improving
    (reference hv_termination.py:960-1162)."""

    def __init__(
        self,
        problem,
        ref_point: Optional[np.ndarray] = None,
        hv_tol: float = 1e-6,
        n_last: int = 15,
        nth_gen: int = 5,
        n_max_gen: Optional[int] = None,
        adaptive_ref_point: bool = True,
        min_generations: int = 20,
        verbose: bool = False,
        **kwargs,
    ):
        super().__init__(
            problem,
            Windows(raw=2, signal=n_last, warmup=2, cadence=nth_gen,
                    ceiling=n_max_gen, floor=min_generations, **kwargs),
        )
        self.ref_point = np.copy(ref_point) if ref_point is not None else None
        self.hv_tol = hv_tol
        self.adaptive_ref_point = adaptive_ref_point
        self.verbose = verbose
        self._scheduler = ProgressivePrecisionScheduler()
        self._router = HVAlgorithmRouter()
        self._tracker: Optional[MultiFidelityHVTracker] = None
        self._detector = ConvergenceDetector(
            stagnation_threshold=hv_tol,
            stagnation_window=min(n_last, 5),
            relative_threshold=hv_tol / 10,
            min_generations=min_generations,
        )
        self._gen = 0

    def _ensure_tracker(self, F: np.ndarray):
        if self._tracker is None:
            if self.ref_point is None:
                nadir = F.max(axis=0)
                span = np.maximum(F.max(axis=0) - F.min(axis=0), 1e-12)
                self.ref_point = nadir + 0.1 * span
            self._tracker = MultiFidelityHVTracker(self.ref_point, self._scheduler, self._router)
        elif self.adaptive_ref_point:
            nadir = F.max(axis=0)
            grown = nadir >= self._tracker.ref_point
            if grown.any():
                span = np.maximum(F.max(axis=0) - F.min(axis=0), 1e-12)
                self._tracker.ref_point = np.maximum(
                    self._tracker.ref_point, nadir + 0.1 * span
                )

    def capture(self, opt):
        F = np.asarray(opt.y)
        self._gen = opt.n_gen
        self._ensure_tracker(F)
        return self._tracker.compute_and_update(F, opt.n_gen)

    def reduce(self, raws):
        est = raws[-1]
        return {"hv": est.value, "fidelity": est.fidelity, "gen": est.generation}

    def verdict(self, signals):
        result = self._detector.check_convergence(self._tracker.estimates, self._gen)
        self._scheduler.adapt_to_progress(result.get("rel_improvement", 1.0))
        if result["converged"]:
            _log(
                self.problem,
                f"stop: hypervolume stagnant "
                f"(rel improvement {result['rel_improvement']:.2e}, "
                f"confidence {result['confidence']:.2f})",
            )
            return True
        if self.verbose:
            _log(self.problem, f"HV progress: {signals[-1]}")
        return False
