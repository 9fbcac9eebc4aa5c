"""Model container: objective / feasibility / sensitivity (reference
model.py:70-95)."""

from __future__ import annotations


class Model:
    def __init__(
        self,
        return_mean_variance=False,
        objective=None,
        feasibility=None,
        sensitivity=None,
        **kwargs,
    ):
        self.objective = objective
        self.feasibility = feasibility
        self.sensitivity = sensitivity
        self.stats = {}
        self.return_mean_variance = return_mean_variance

    def get_stats(self):
        for part in (self.objective, self.feasibility, self.sensitivity):
            if part is not None:
                self.stats.update(getattr(part, "stats", {}))
        return self.stats.copy()
