"""Adaptive early-stopping utilities for iterative surrogate training.

Role parity with reference model_gpytorch.py:579-989 (ModelType,
EarlyStoppingConfig.for_model_type, AdaptiveEarlyStopping.should_stop,
analyze_loss_trajectory, suggest_hyperparameters): multi-criteria stopping
over a loss trajectory — percent-change, absolute/relative convergence,
plateau detection, optional validation-loss divergence.
"""

from __future__ import annotations

from dataclasses import dataclass
from enum import Enum
from typing import Dict, List, Optional

import numpy as np


class ModelType(Enum):
    EXACT_GP = "exact_gp"
    MULTITASK_GP = "multitask_gp"
    DEEP_GP = "deep_gp"
    DSPP = "dspp"
    VARIATIONAL_GP = "variational_gp"


@dataclass
class EarlyStoppingConfig:
    pct_change_tol: float = 0.1
    abs_tol: float = 1e-6
    rel_tol: float = 1e-5
    plateau_window: int = 20
    plateau_tol: float = 1e-4
    min_iterations: int = 30
    patience: int = 10
    validation_divergence_window: int = 10

    @classmethod
    def for_model_type(cls, model_type: ModelType) -> "EarlyStoppingConfig":
        if model_type in (ModelType.DEEP_GP, ModelType.DSPP):
            return cls(min_iterations=80, plateau_window=30, patience=20)
        if model_type == ModelType.VARIATIONAL_GP:
            return cls(min_iterations=50, patience=15)
        return cls()


class AdaptiveEarlyStopping:
    def __init__(self, config: Optional[EarlyStoppingConfig] = None):
        self.config = config or EarlyStoppingConfig()
        self.losses: List[float] = []
        self.val_losses: List[float] = []
        self._plateau_count = 0

    def should_stop(self, loss: float, val_loss: Optional[float] = None) -> bool:
        c = self.config
        self.losses.append(float(loss))
        if val_loss is not None:
            self.val_losses.append(float(val_loss))
        n = len(self.losses)
        if n < c.min_iterations:
            return False
        prev, cur = self.losses[-2], self.losses[-1]
        # percent change
        if abs(prev) > 0 and abs(prev - cur) / abs(prev) * 100.0 < c.pct_change_tol:
            self._plateau_count += 1
        else:
            self._plateau_count = 0
        if self._plateau_count >= c.patience:
            return True
        # absolute / relative convergence
        if abs(prev - cur) < c.abs_tol:
            return True
        if abs(prev) > 0 and abs(prev - cur) / abs(prev) < c.rel_tol:
            return True
        # plateau over a window
        if n >= c.plateau_window:
            w = np.asarray(self.losses[-c.plateau_window :])
            if (w.max() - w.min()) / max(abs(w.mean()), 1e-12) < c.plateau_tol:
                return True
        # validation divergence
        if len(self.val_losses) >= c.validation_divergence_window:
            w = self.val_losses[-c.validation_divergence_window :]
            if all(w[i] <= w[i + 1] for i in range(len(w) - 1)):
                return True
        return False


def analyze_loss_trajectory(losses: List[float]) -> Dict:
    """Summary statistics of a training trajectory (slopes, noise level,
    convergence estimate)."""
    x = np.asarray(losses, dtype=float)
    if len(x) < 3:
        return {"n": len(x), "converged": False}
    diffs = np.diff(x)
    tail = x[-max(5, len(x) // 5) :]
    return {
        "n": len(x),
        "final": float(x[-1]),
        "best": float(x.min()),
        "mean_step": float(diffs.mean()),
        "noise": float(np.std(diffs)),
        "tail_slope": float(np.polyfit(np.arange(len(tail)), tail, 1)[0]),
        "converged": bool(abs(np.polyfit(np.arange(len(tail)), tail, 1)[0]) < 1e-4),
    }


def suggest_hyperparameters(losses: List[float], lr: float) -> Dict:
    """Heuristic LR suggestion from the trajectory shape."""
    stats = analyze_loss_trajectory(losses)
    if stats.get("n", 0) < 3:
        return {"lr": lr}
    if stats["noise"] > 10 * abs(stats["mean_step"]):
        return {"lr": lr * 0.5, "reason": "noisy trajectory"}
    if stats["mean_step"] > 0:
        return {"lr": lr * 0.25, "reason": "diverging"}
    if abs(stats["tail_slope"]) < 1e-6:
        return {"lr": lr * 2.0, "reason": "stalled"}
    return {"lr": lr}
