"""Trust Region Search (reference TRS.py:19-335).

Per-point trust boxes around the current population, Sobol perturbations
with a dimension mask (Regis & Shoemaker DYCORS-style), front + HVI-fill
survivor selection, and a success-window-driven expand/shrink/restart of
the trust-region length.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import numpy as np
import torch

from dmosopt_amd import ops, sampling
from dmosopt_amd.datatypes import Struct
from dmosopt_amd.hv.indicators import (
    HypervolumeImprovement,
    PopulationDiversity,
    SlidingWindow,
)
from dmosopt_amd.moea.base import MOEA


@dataclass
class TrState:
    dim: int
    is_constrained: bool = False
    length: float = 0.05
    length_init: float = 0.1
    length_min: float = 0.00001
    length_max: float = 1.0
    failure_tolerance: float = float("nan")
    success_tolerance: float = 0.51
    restart: bool = False
    success_counter: int = 0
    failure_counter: int = 0

    def __post_init__(self):
        self.failure_tolerance = min(1.0 / self.dim, self.success_tolerance / 2.0)


class TRSOptimizer(MOEA):
    def __init__(
        self,
        popsize: int,
        nInput: int,
        nOutput: int,
        model: Optional[Any] = None,
        optimize_mean_variance: bool = False,
        **kwargs,
    ):
        super().__init__(name="TRS", popsize=popsize, nInput=nInput, nOutput=nOutput, **kwargs)
        self.model = model
        self.x_distance_fns = None
        if model is not None and getattr(model, "feasibility", None) is not None:
            self.x_distance_fns = [model.feasibility.rank]
        self.indicator = HypervolumeImprovement
        self.diversity_indicator = PopulationDiversity()
        self.optimize_mean_variance = optimize_mean_variance

    @property
    def default_parameters(self) -> Dict[str, Any]:
        return {
            "nchildren": 1,
            "success_window_size": 64,
            "max_population_size": 600,
            "min_population_size": 100,
            "adaptive_population_size": False,
        }

    def _x_dists(self, x):
        if self.x_distance_fns is None:
            return None
        return [
            torch.as_tensor(np.asarray(fn(x.cpu().numpy())), dtype=x.dtype, device=x.device)
            for fn in self.x_distance_fns
        ]

    def initialize_state(self, x, y, bounds, local_random, **params):
        perm, rank, _ = ops.order_mo(x, y, x_dists=self._x_dists(x))
        pop = self.opt_params.popsize
        return Struct(
            bounds=bounds,
            population_parm=x[perm][:pop],
            population_obj=y[perm][:pop],
            rank=rank[:pop],
            tr=TrState(dim=self.nInput),
            success_window=SlidingWindow(self.opt_params.success_window_size),
        )

    def generate_strategy(self, **params):
        popsize = self.opt_params.popsize
        rng = self.local_random
        st = self.state
        xlb, xub = st.bounds[:, 0], st.bounds[:, 1]

        parm, obj = ops.remove_duplicates(st.population_parm, st.population_obj)
        x_centers = parm
        weights = (xub - xlb).clone()
        weights = weights / weights.mean()
        weights = weights / torch.prod(weights ** (1.0 / len(weights)))
        tr_lb = (x_centers - weights * st.tr.length / 2.0).clamp(xlb, xub)
        tr_ub = (x_centers + weights * st.tr.length / 2.0).clamp(xlb, xub)

        pert_np = sampling.sobol(x_centers.shape[0], self.nInput, rng)
        pert = self._as_tensor(pert_np)
        pert = tr_lb + (tr_ub - tr_lb) * pert

        prob_perturb = min(20.0 / st.tr.dim, 1.0)
        mask_np = rng.random(st.tr.dim) <= prob_perturb
        mask = torch.as_tensor(mask_np, device=self.device)
        X_cand = x_centers.clone()
        X_cand[:, mask] = pert[:, mask]

        if X_cand.shape[0] < popsize:
            extra = sampling.sobol(popsize - X_cand.shape[0], self.nInput, rng)
            extra_t = self._as_tensor(extra) * (xub - xlb) + xlb
            X_cand = torch.cat([X_cand, extra_t], dim=0)
        return X_cand, {}

    def select_candidates(self, candidates_x, candidates_y):
        popsize = self.opt_params.popsize
        n = candidates_x.shape[0]
        if n <= popsize:
            return (
                np.ones(n, dtype=bool),
                np.zeros(n, dtype=bool),
                ops.pareto_rank(candidates_y).cpu().numpy(),
            )
        perm, rank, _ = ops.order_mo(
            candidates_x, candidates_y, x_dists=self._x_dists(candidates_x)
        )
        rank_np = rank.cpu().numpy()
        order_inv = np.argsort(perm.cpu().numpy(), kind="stable")
        chosen = np.zeros(n, dtype=bool)
        not_chosen = np.zeros(n, dtype=bool)
        mid_front = None
        full = False
        chosen_count = 0
        for r in range(int(rank_np.max()) + 1):
            front_r = order_inv[np.flatnonzero(rank_np == r)]
            if chosen_count + len(front_r) <= popsize and not full:
                chosen[front_r] = True
                chosen_count += len(front_r)
            elif mid_front is None and chosen_count < popsize:
                mid_front = front_r.copy()
                full = True
            else:
                not_chosen[front_r] = True
        k = popsize - chosen_count
        if k > 0:
            y_np = candidates_y.cpu().numpy()
            ref = np.max(y_np, axis=0) + 1
            indicator = self.indicator(ref_point=ref, nds=True)
            assert mid_front is not None and len(mid_front) > 0
            if chosen_count > 0:
                selected = indicator.do(
                    y_np[chosen], y_np[mid_front], np.ones_like(y_np[mid_front, :]), k
                )
            else:
                selected = np.arange(k)
            selected = np.asarray(selected)[:k]
            chosen[mid_front[selected]] = True
            mask = np.ones(len(mid_front), bool)
            mask[selected] = False
            not_chosen[mid_front[mask]] = True
        # rank of chosen, aligned to candidate order
        full_rank = np.empty(n, dtype=np.int64)
        full_rank[perm.cpu().numpy()] = rank_np
        return chosen, not_chosen, full_rank[chosen]

    def update_state(self, X_next, Y_next, is_offspring):
        tr = self.state.tr
        if tr.restart:
            self.restart_state()
        chosen, not_chosen, chosen_rank = self.select_candidates(X_next, Y_next)
        success_counter = int(np.count_nonzero(is_offspring & chosen))
        self.state.success_window.append(success_counter)
        success_mean = float(np.mean(self.state.success_window[:]))
        success_frac = min(1.0, success_mean / self.opt_params.popsize)
        if success_frac > tr.success_tolerance:
            tr.length = min(
                (1.0 + (success_frac - tr.success_tolerance)) * tr.length, tr.length_max
            )
            tr.success_counter = 0
        elif success_frac <= tr.failure_tolerance:
            tr.length /= 2.0
            tr.success_counter = 0
        if tr.length < tr.length_min:
            tr.restart = True
        idx = torch.as_tensor(np.flatnonzero(chosen), dtype=torch.long, device=self.device)
        return (
            X_next[idx],
            Y_next[idx],
            torch.as_tensor(chosen_rank, dtype=torch.long, device=self.device),
        )

    def update_strategy(self, x_gen, y_gen, gen_state, **params):
        st = self.state
        C = x_gen.shape[0]
        P = st.population_parm.shape[0]
        candidates_x = torch.cat([x_gen, st.population_parm], dim=0)
        candidates_y = torch.cat([y_gen, st.population_obj], dim=0)
        is_offspring = np.concatenate([np.ones(C, bool), np.zeros(P, bool)])
        parm, obj, rank = self.update_state(candidates_x, candidates_y, is_offspring)
        st.population_parm = parm
        st.population_obj = obj
        st.rank = rank
        if self.opt_params.adaptive_population_size:
            self.update_population_size()

    def restart_state(self):
        tr = self.state.tr
        tr.failure_counter = 0
        tr.length = tr.length_init
        tr.restart = False
        self.state.success_window = SlidingWindow(self.opt_params.success_window_size)

    def get_population_strategy(self):
        return self.state.population_parm.clone(), self.state.population_obj.clone()

    def update_population_size(self):
        diversity, cd_spread = self.diversity_indicator.do(
            self.state.rank, self.state.population_obj
        )
        p = self.opt_params
        if diversity < 0.1 or cd_spread < 2.0:
            new_size = min(p.max_population_size, int(p.popsize * 1.1))
        elif diversity > 0.4 and cd_spread > 1.0:
            new_size = max(p.min_population_size, int(p.popsize * 0.9))
        else:
            new_size = p.popsize
        p.popsize = new_size
