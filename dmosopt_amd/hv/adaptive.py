"""Adaptive hypervolume computation: exact for low d, MC for high d.

Role parity with reference hv.py:77-381 (AdaptiveHyperVolume router) and
hv_adaptive.py:576-973 (hybrid routing): exact 2D/3D sweeps and box
decomposition below ``mc_dim_threshold`` objectives, Monte Carlo (FPRAS or
MCM2RV) above, with with-confidence variants.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

from dmosopt_amd.hv.exact import HyperVolumeBoxDecomposition
from dmosopt_amd.hv.mc import hv_fpras, hv_mcm2rv


@dataclass
class HVResult:
    value: float
    method: str
    eps: Optional[float] = None
    delta: Optional[float] = None
    confidence: Optional[float] = None


class DominanceAnalysis:
    """cKDTree-accelerated dominance queries for MC estimators
    (reference hv_adaptive.py:40-180)."""

    def __init__(self, points: np.ndarray):
        from scipy.spatial import cKDTree

        self.points = np.asarray(points, dtype=np.float64)
        self.tree = cKDTree(self.points)

    def dominates_any(self, samples: np.ndarray, k: int = 16) -> np.ndarray:
        """For each sample, is it dominated by any point (all dims <=)?

        kNN pre-screen: only the k nearest points are checked exactly; falls
        back to the full check for samples whose neighbors are inconclusive
        and far ('far' = kth distance < distance to the sample's antiideal
        corner would not guarantee coverage, so we do the exact check).
        """
        samples = np.asarray(samples, dtype=np.float64)
        k = min(k, len(self.points))
        _, idx = self.tree.query(samples, k=k)
        idx = np.atleast_2d(idx)
        neigh = self.points[idx]  # (n, k, d)
        dominated = (neigh <= samples[:, None, :]).all(axis=2).any(axis=1)
        # exact fallback for the not-yet-dominated ones
        unresolved = np.flatnonzero(~dominated)
        if len(unresolved):
            sub = samples[unresolved]
            full = (self.points[None, :, :] <= sub[:, None, :]).all(axis=2).any(axis=1)
            dominated[unresolved] = full
        return dominated


def estimate_overlap(points: np.ndarray, ref_point: np.ndarray,
                     n_probe: int = 4096, seed: int = 0) -> float:
    """Probe the overlap of the per-point dominated hyperboxes: the ratio of
    the union volume to the sum of box volumes (reference
    hv_adaptive.py:469-...). High overlap favors the uniform (MCM2RV)
    estimator; low overlap favors FPRAS."""
    P = np.asarray(points, dtype=np.float64)
    r = np.asarray(ref_point, dtype=np.float64)
    mask = (P < r).all(axis=1)
    P = P[mask]
    if len(P) == 0:
        return 0.0
    rng = np.random.default_rng(seed)
    vols = np.prod(r - P, axis=1)
    probs = vols / vols.sum()
    box = rng.choice(len(P), size=n_probe, p=probs)
    u = rng.random((n_probe, P.shape[1]))
    samples = P[box] + u * (r - P[box])
    counts = (P[None, :, :] <= samples[:, None, :]).all(axis=2).sum(axis=1)
    # E[1/multiplicity] = union / sum(vols)
    return float(np.mean(1.0 / counts))


class AdaptiveHyperVolume:
    """Routes hypervolume computation by dimensionality and front size."""

    def __init__(
        self,
        ref_point,
        mc_dim_threshold: int = 10,
        mc_eps: float = 0.01,
        mc_delta: float = 0.01,
        mc_method: str = "fpras",
        seed: Optional[int] = None,
        device=None,
    ):
        self.ref_point = np.asarray(ref_point, dtype=np.float64)
        self.d = len(self.ref_point)
        self.mc_dim_threshold = mc_dim_threshold
        self.mc_eps = mc_eps
        self.mc_delta = mc_delta
        self.mc_method = mc_method
        self.seed = seed
        self.device = device
        self._box = HyperVolumeBoxDecomposition(self.ref_point)

    def compute(self, points, eps: Optional[float] = None) -> float:
        return self.compute_with_statistics(points, eps=eps).value

    # reference-compatible alias
    def do(self, points) -> float:
        return self.compute(points)

    def compute_with_statistics(self, points, eps: Optional[float] = None) -> HVResult:
        points = np.asarray(points, dtype=np.float64)
        if len(points) == 0:
            return HVResult(0.0, "empty")
        if self.d < self.mc_dim_threshold:
            return HVResult(self._box.compute_hypervolume(points), "box", confidence=1.0)
        eps = eps if eps is not None else self.mc_eps
        method = self.mc_method
        if method == "hybrid":
            # overlap probing routes between uniform-box MC and FPRAS
            # (reference hv_adaptive.py:576-860 hybrid scheme)
            overlap = estimate_overlap(points, self.ref_point, seed=self.seed or 0)
            method = "mcm2rv" if overlap < 0.25 else "fpras"
        if method == "mcm2rv":
            val = hv_mcm2rv(points, self.ref_point, seed=self.seed, device=self.device)
        else:
            val = hv_fpras(
                points, self.ref_point, eps=eps, delta=self.mc_delta,
                seed=self.seed, device=self.device,
            )
        return HVResult(val, method, eps=eps, delta=self.mc_delta,
                        confidence=1.0 - self.mc_delta)

    def compute_with_confidence(self, points, eps: Optional[float] = None):
        res = self.compute_with_statistics(points, eps=eps)
        return res.value, res.confidence
