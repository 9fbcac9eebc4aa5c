"""SMPSO: speed-constrained multi-objective PSO (reference SMPSO.py:19-348).

swarm_size independent subswarms of popsize particles; constriction-factor
velocity update toward crowding-picked archive leaders, position clip,
polynomial mutation, per-swarm elitist survivor selection. The reference's
per-element velocity double loop is ONE tensor expression here, and all
swarms' mutations run as a single fused batch.
"""

from __future__ import annotations

import math
from typing import Any, Dict, Optional

import numpy as np
import torch

from dmosopt_amd import ops
from dmosopt_amd.datatypes import Struct
from dmosopt_amd.hv.indicators import PopulationDiversity
from dmosopt_amd.moea.base import MOEA


def velocity_vector(rng, position, velocity, archive, crowding, xlb, xub):
    """Constriction-factor velocity update (reference SMPSO.py:316-348),
    vectorized: same scalar draws (r1, r2, w, c1, c2, leaders) per swarm."""
    r1 = float(rng.uniform(0.0, 1.0, size=1)[0])
    r2 = float(rng.uniform(0.0, 1.0, size=1)[0])
    w = float(rng.uniform(0.1, 0.5, size=1)[0])
    c1 = float(rng.uniform(1.5, 2.5, size=1)[0])
    c2 = float(rng.uniform(1.5, 2.5, size=1)[0])
    phi = c1 + c2 if c1 + c2 > 4 else 0.0
    chi = 2.0 / (2.0 - phi - math.sqrt(max(phi**2 - 4.0 * phi, 0.0)))
    if archive.shape[0] > 2:
        ind_1, ind_2 = rng.integers(low=0, high=archive.shape[0], size=2)
        if crowding[ind_1] < crowding[ind_2]:
            ind_1, ind_2 = ind_2, ind_1
    else:
        ind_1 = ind_2 = 0
    if (
        position.device.type == "cuda"
        and position.dtype == torch.float32
        and ops.native_available()
    ):
        # fused velocity + clamp in one launch (ops/hip/moea_ops.hip)
        from dmosopt_amd import _hipops

        return _hipops.smpso_velocity(
            position.contiguous(), velocity.contiguous(),
            archive[int(ind_1)].contiguous(), archive[int(ind_2)].contiguous(),
            xlb.float().contiguous(), xub.float().contiguous(),
            w, c1 * r1, c2 * r2, chi,
        )
    delta = (xub - xlb) / 2.0
    out = (
        w * velocity
        + c1 * r1 * (archive[int(ind_1)][None, :] - position)
        + c2 * r2 * (archive[int(ind_2)][None, :] - position)
    ) * chi
    return out.clamp(-delta, delta)


class SMPSOOptimizer(MOEA):
    def __init__(
        self,
        popsize: int,
        nInput: int,
        nOutput: int,
        model: Optional[Any] = None,
        distance_metric: Optional[Any] = None,
        optimize_mean_variance: bool = False,
        **kwargs,
    ):
        swarm_size = kwargs.get("swarm_size", self.default_parameters["swarm_size"])
        kwargs["initial_size"] = popsize * swarm_size
        super().__init__(name="SMPSO", popsize=popsize, nInput=nInput, nOutput=nOutput, **kwargs)
        self.model = model
        self.distance_metric = distance_metric
        self.y_distance_metrics = [distance_metric] if distance_metric is not None else None
        self.x_distance_fns = None
        if model is not None and getattr(model, "feasibility", None) is not None:
            self.x_distance_fns = [model.feasibility.rank]
        p = self.opt_params
        if np.isscalar(p.di_mutation):
            p.di_mutation = np.full(nInput, float(p.di_mutation))
        if p.mutation_rate is None:
            p.mutation_rate = 1.0 / float(nInput)
        self.optimize_mean_variance = optimize_mean_variance
        self.diversity_indicator = PopulationDiversity()
        self._set_slices()

    def _set_slices(self):
        p = self.opt_params
        self.pop_slices = [
            slice(s * p.popsize, (s + 1) * p.popsize) for s in range(p.swarm_size)
        ]

    @property
    def default_parameters(self) -> Dict[str, Any]:
        return {
            "mutation_rate": None,
            "nchildren": 1,
            "swarm_size": 5,
            "di_mutation": 20.0,
            "max_population_size": 2000,
            "min_population_size": 100,
            "min_success_rate": 0.2,
            "max_success_rate": 0.75,
            "adaptive_population_size": False,
            "adaptive_operator_rates": False,
        }

    def _x_dists(self, x):
        if self.x_distance_fns is None:
            return None
        return [
            torch.as_tensor(np.asarray(fn(x.cpu().numpy())), dtype=x.dtype, device=x.device)
            for fn in self.x_distance_fns
        ]

    def initialize_state(self, x, y, bounds, local_random, **params):
        p = self.opt_params
        popsize, swarm = p.popsize, p.swarm_size
        xlb, xub = bounds[:, 0], bounds[:, 1]
        n_total = swarm * popsize
        # initial x may be smaller than swarm*popsize; tile as needed
        if x.shape[0] < n_total:
            reps = (n_total + x.shape[0] - 1) // x.shape[0]
            x = x.repeat(reps, 1)[:n_total]
            y = y.repeat(reps, 1)[:n_total]
        parm = torch.zeros(n_total, self.nInput, dtype=self.dtype, device=self.device)
        obj = torch.zeros(n_total, self.nOutput, dtype=self.dtype, device=self.device)
        velocity = (
            self._as_tensor(local_random.uniform(size=(n_total, self.nInput)))
            * (xub - xlb)
            + xlb
        )
        ranks = []
        for s in range(swarm):
            sl = self.pop_slices[s]
            xs, ys = x[sl], y[sl]
            perm, rank, _ = ops.order_mo(
                xs, ys, x_dists=self._x_dists(xs), y_distance_metrics=self.y_distance_metrics
            )
            parm[sl] = xs[perm][:popsize]
            obj[sl] = ys[perm][:popsize]
            ranks.append(rank[:popsize])
        return Struct(
            bounds=bounds,
            population_parm=parm,
            population_obj=obj,
            ranks=ranks,
            velocity=velocity,
            successful_children=0,
        )

    def generate_strategy(self, **params):
        p = self.opt_params
        rng = self.local_random
        st = self.state
        xlb, xub = st.bounds[:, 0], st.bounds[:, 1]
        popsize, swarm = p.popsize, p.swarm_size
        di_m = torch.as_tensor(p.di_mutation, dtype=self.dtype, device=self.device)

        per_swarm = []
        for s in range(swarm):
            sl = self.pop_slices[s]
            moved = (st.population_parm[sl] + st.velocity[sl]).clamp(xlb, xub)
            per_swarm.append([moved])

        # popsize mutation rounds; parents drawn per swarm per round, then
        # ONE fused batched mutation launch for all rounds x swarms
        parent_rows = []
        for _ in range(popsize):
            pidx = rng.integers(low=0, high=popsize, size=(swarm, 1))
            for s in range(swarm):
                parent_rows.append(self.pop_slices[s].start + int(pidx[s, 0]))
        idx_t = torch.as_tensor(parent_rows, dtype=torch.long, device=self.device)
        mutants = ops.mutation_from_pool(
            st.population_parm, idx_t, di_m, xlb, xub, p.mutation_rate,
            seed=int(rng.integers(0, 2**62)), generator=self.torch_random,
        )  # (popsize*swarm, d) in round-major order
        mutants = mutants.reshape(popsize, swarm, self.nInput)
        for s in range(swarm):
            per_swarm[s].append(mutants[:, s, :])
        x_gen = torch.cat([torch.cat(chunks, dim=0) for chunks in per_swarm], dim=0)
        return x_gen, {}

    def update_strategy(self, x_gen, y_gen, gen_state, **params):
        p = self.opt_params
        st = self.state
        rng = self.local_random
        xlb, xub = st.bounds[:, 0], st.bounds[:, 1]
        popsize, swarm = p.popsize, p.swarm_size
        # x_gen has 2*popsize rows per swarm (moved + mutants)
        gen_per_swarm = x_gen.shape[0] // swarm

        for s in range(swarm):
            sl = self.pop_slices[s]
            gsl = slice(s * gen_per_swarm, (s + 1) * gen_per_swarm)
            D = ops.crowding_distance(y_gen[gsl])
            st.velocity[sl] = velocity_vector(
                rng, st.population_parm[sl], st.velocity[sl], x_gen[gsl], D, xlb, xub
            )

        total_children = x_gen.shape[0]
        for s in range(swarm):
            sl = self.pop_slices[s]
            gsl = slice(s * gen_per_swarm, (s + 1) * gen_per_swarm)
            parm_s = torch.cat([x_gen[gsl], st.population_parm[sl]], dim=0)
            obj_s = torch.cat([y_gen[gsl], st.population_obj[sl]], dim=0)
            xs, ys, rank, perm = ops.remove_worst(
                parm_s, obj_s, popsize,
                x_dists=self._x_dists(parm_s),
                y_distance_metrics=self.y_distance_metrics,
            )
            st.population_parm[sl] = xs
            st.population_obj[sl] = ys
            st.ranks[s] = rank
            survived = np.isin(
                np.arange(total_children), perm.cpu().numpy(), assume_unique=True
            )
            st.successful_children += int(np.count_nonzero(survived))

        if p.adaptive_population_size:
            self.update_population_size()
        if p.adaptive_operator_rates:
            self.update_operator_rates()

    def get_population_strategy(self):
        x, y = ops.remove_duplicates(
            self.state.population_parm.clone(), self.state.population_obj.clone()
        )
        return x, y

    def update_population_size(self):
        ranks = torch.cat([r for r in self.state.ranks])
        diversity, cd_spread = self.diversity_indicator.do(
            ranks, self.state.population_obj
        )
        p = self.opt_params
        if diversity < 0.5 and cd_spread < 2.0:
            new_size = min(p.max_population_size, int(p.popsize * 1.2))
        elif diversity > 0.9 or cd_spread > 1.0:
            new_size = max(p.min_population_size, int(p.popsize * 0.9))
        else:
            new_size = p.popsize
        p.popsize = new_size
        self._set_slices()

    def update_operator_rates(self):
        p = self.opt_params
        st = self.state
        success_rate = st.successful_children / (p.popsize * p.swarm_size)
        if success_rate < p.min_success_rate:
            p.di_mutation = np.maximum(1.0, p.di_mutation * 0.9)
            p.mutation_rate = min(0.95, p.mutation_rate * 1.1)
        elif success_rate > p.max_success_rate:
            p.di_mutation = np.minimum(100.0, p.di_mutation * 1.1)
            p.mutation_rate = max(0.05 / self.nInput, p.mutation_rate * 0.9)
        st.successful_children = 0
