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
            return HVResult(self._box.compute_hypervolume(points), "box")
        eps = eps if eps is not None else self.mc_eps
        if self.mc_method == "mcm2rv":
            val = hv_mcm2rv(points, self.ref_point, seed=self.seed, device=self.device)
        else:
            val = hv_fpras(
                points, self.ref_point, eps=eps, delta=self.mc_delta,
                seed=self.seed, device=self.device,
            )
        return HVResult(val, self.mc_method, eps=eps, delta=self.mc_delta)
