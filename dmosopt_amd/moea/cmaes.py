"""MO-CMA-ES (Suttorp/Voss-Hansen-Igel) — batched covariance updates.

Parity with reference CMAES.py:22-537: per-individual step sizes, Cholesky
factor A / inverse Ainv, evolution path pc, success probability psucc;
selection = front fill + hypervolume-improvement on the mid front; success-
based step-size adaptation and rank-1 Cholesky update
(A <- a A + b pc w^T, Ainv <- Ainv/a - c w (w^T Ainv), w = Ainv pc).

Redesign: the reference's per-individual numpy update loop
(CMAES.py:345-410, 489-537) becomes BATCHED tensor ops over all chosen
offspring at once — (K, d, d) bmm + outer products, the 'large-pop MFMA
covariance update' workload of BASELINE config #5. Parent bookkeeping that
is sequential per-parent (multiple offspring of one parent) is processed in
multiplicity rounds (vectorized per round; round count = max offspring per
parent, usually 1-2).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from dmosopt_amd import ops
from dmosopt_amd.datatypes import Struct
from dmosopt_amd.hv.indicators import HypervolumeImprovement, PopulationDiversity
from dmosopt_amd.moea.base import MOEA


def _sort_mo(x: torch.Tensor, y: torch.Tensor, x_distance_fns=None):
    """(perm, rank) by pareto rank then per-front x-distance (CMAES.py:458)."""
    rank = ops.pareto_rank(y)
    keys = [rank.to(torch.float64)]
    if x_distance_fns:
        rank_np = rank.cpu().numpy()
        xd = np.zeros(len(rank_np))
        xnp = x.cpu().numpy()
        for front in range(int(rank_np.max()) + 1):
            mask = rank_np == front
            for fn in x_distance_fns:
                xd[mask] = fn(xnp[mask, :])
        keys = [torch.as_tensor(-xd, device=x.device)] + keys
    perm = ops.lexsort(keys)
    return perm, rank


def batched_cholesky_update(A, Ainv, pc, z, psucc, cc, ccov, pthresh):
    """Rank-1 Cholesky update for K individuals at once (CMAES.py:489-537).

    A, Ainv: (K, d, d); pc, z: (K, d); psucc: (K,). Returns updated copies.
    On GPU this is ONE fused gfx950 kernel launch (ops/hip/cmaes_update.hip);
    the torch path below is the CPU/oracle implementation.
    """
    K, d, _ = A.shape
    if A.is_cuda and A.dtype == torch.float32 and d <= 256:
        from dmosopt_amd import ops as _ops

        if _ops.native_available():
            from dmosopt_amd import _hipops

            A2 = A.contiguous().clone()
            Ainv2 = Ainv.contiguous().clone()
            pc2 = pc.contiguous().clone()
            _hipops.cmaes_update_(
                A2, Ainv2, pc2, z.contiguous().float(), psucc.contiguous().float(),
                float(cc), float(ccov), float(pthresh),
            )
            return A2, Ainv2, pc2
    below = (psucc < pthresh)[:, None]
    pc_new = torch.where(
        below,
        (1.0 - cc) * pc + float(np.sqrt(cc * (2.0 - cc))) * z,
        (1.0 - cc) * pc,
    )
    alpha = torch.where(
        below[:, 0], torch.full_like(psucc, 1.0 - ccov),
        torch.full_like(psucc, (1.0 - ccov) + ccov * cc * (2.0 - cc)),
    )
    beta = ccov
    w = torch.bmm(Ainv, pc_new[:, :, None])[:, :, 0]  # (K, d)
    do_update = w.max(dim=1).values > 1e-20

    wAinv = torch.bmm(w[:, None, :], Ainv)[:, 0, :]  # (K, d) = w^T Ainv
    a = torch.sqrt(alpha)
    norm_w2 = (w * w).sum(dim=1).clamp_min(1e-300)
    root = torch.sqrt(1.0 + beta / alpha * norm_w2)
    b = a / norm_w2 * (root - 1.0)
    A_new = a[:, None, None] * A + b[:, None, None] * (pc_new[:, :, None] @ w[:, None, :])
    c = 1.0 / (a * norm_w2) * (1.0 - 1.0 / root)
    Ainv_new = (1.0 / a)[:, None, None] * Ainv - c[:, None, None] * (
        w[:, :, None] @ wAinv[:, None, :]
    )
    upd = do_update[:, None, None]
    return (
        torch.where(upd, A_new, A),
        torch.where(upd, Ainv_new, Ainv),
        pc_new,
    )


class CMAESOptimizer(MOEA):
    def __init__(
        self,
        popsize: int,
        nInput: int,
        nOutput: int,
        model: Optional[Any] = None,
        optimize_mean_variance: bool = False,
        **kwargs,
    ):
        super().__init__(name="CMAES", popsize=popsize, nInput=nInput, nOutput=nOutput, **kwargs)
        self.model = model
        self.x_distance_fns = None
        if model is not None and getattr(model, "feasibility", None) is not None:
            self.x_distance_fns = [model.feasibility.rank]
        p = self.opt_params
        if np.isscalar(p.di_mutation):
            p.di_mutation = np.full(nInput, float(p.di_mutation))
        self.indicator = HypervolumeImprovement
        self.optimize_mean_variance = optimize_mean_variance
        self.diversity_indicator = PopulationDiversity()

    @property
    def default_parameters(self) -> Dict[str, Any]:
        nInput, nOutput = self.nInput, self.nOutput
        ptarg = 1.0 / (5.0 + 0.5)
        return {
            "sigma": 0.001,
            "mu": self.popsize // 2,
            "lambda_": 1,
            "d": 1.0 + nOutput / 2.0,
            "ptarg": ptarg,
            "cp": ptarg / (1.0 + ptarg),
            "cc": 2.0 / (nInput + 2.0),
            "ccov": 2.0 / (nInput**2 + 6.0),
            "pthresh": 0.44,
            "di_mutation": 30.0,
            "max_population_size": 600,
            "min_population_size": 100,
            "adaptive_population_size": False,
        }

    def initialize_state(self, x, y, bounds, local_random, **params):
        dim = self.nInput
        P = self.opt_params.popsize
        p = self.opt_params
        sigmas = self._as_tensor(
            np.tile(p.sigma * (1.0 / (np.asarray(p.di_mutation) + 1.0)), (P, 1))
        )
        eye = torch.eye(dim, dtype=self.dtype, device=self.device)
        A = eye[None].repeat(P, 1, 1)
        Ainv = eye[None].repeat(P, 1, 1)
        pc = torch.zeros(P, dim, dtype=self.dtype, device=self.device)
        psucc = torch.full((P,), p.ptarg, dtype=self.dtype, device=self.device)
        perm, rank = _sort_mo(x, y, self.x_distance_fns)
        sel = perm[:P]
        return Struct(
            bounds=bounds,
            parents_x=x[sel].clone(),
            parents_y=y[sel].clone(),
            sigmas=sigmas,
            A=A,
            Ainv=Ainv,
            pc=pc,
            psucc=psucc,
            rank=rank[sel].clone(),
        )

    # ------------------------------------------------------------- generate
    def generate_strategy(self, **params):
        p = self.opt_params
        dim = self.nInput
        mu, lambda_ = p.mu, p.lambda_
        rng = self.local_random
        st = self.state

        arz = torch.as_tensor(
            rng.normal(size=(lambda_ * mu, dim)), dtype=self.dtype, device=self.device
        )
        perm, rank = _sort_mo(st.parents_x, st.parents_y, self.x_distance_fns)
        rank_np = rank.cpu().numpy()
        parent_selection = []
        count = 0
        for r in range(int(rank_np.max()) + 1):
            front_r = np.flatnonzero(rank_np == r)
            parent_selection.append(front_r)
            count += len(front_r)
            if count >= mu:
                break
        parent_selection = np.concatenate(parent_selection)[:mu]
        js = rng.choice(len(parent_selection), size=lambda_ * mu)
        p_idx = torch.as_tensor(
            parent_selection[js], dtype=torch.long, device=self.device
        )
        steps = st.sigmas[p_idx] * torch.bmm(st.A[p_idx], arz[:, :, None])[:, :, 0]
        individuals = st.parents_x[p_idx] + steps
        xrng = self.bounds[:, 1] - self.bounds[:, 0]
        denom = individuals.abs().max().clamp_min(1e-30)
        x_new = (individuals / denom) * xrng + self.bounds[:, 0]
        return x_new, {"p_idx": p_idx.cpu().numpy()}

    # --------------------------------------------------------------- select
    def _select(self, candidates_x, candidates_y):
        popsize = self.opt_params.popsize
        n = candidates_x.shape[0]
        if n <= popsize:
            chosen = np.ones(n, dtype=bool)
            return chosen, np.zeros(n, dtype=bool), ops.pareto_rank(candidates_y).cpu().numpy()
        perm, rank = _sort_mo(candidates_x, candidates_y, self.x_distance_fns)
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
                    y_np[chosen], y_np[mid_front, :], np.ones_like(y_np[mid_front, :]), k
                )
            else:
                selected = np.arange(k)
            selected = np.asarray(selected)[:k]
            chosen[mid_front[selected]] = True
            mask = np.ones(len(mid_front), bool)
            mask[selected] = False
            not_chosen[mid_front[mask]] = True
        return chosen, not_chosen, rank_np

    # --------------------------------------------------------------- update
    def update_strategy(self, x_gen, y_gen, gen_state, **params):
        p = self.opt_params
        dim = self.nInput
        st = self.state
        dev = self.device
        p_idxs = np.asarray(gen_state["p_idx"])

        P = st.parents_x.shape[0]
        C = x_gen.shape[0]
        candidates_x = torch.cat([x_gen, st.parents_x], dim=0)
        candidates_y = torch.cat([y_gen, st.parents_y], dim=0)
        is_offspring = np.concatenate([np.ones(C, bool), np.zeros(P, bool)])
        cand_pidxs = np.concatenate([p_idxs, np.arange(P)])

        chosen, not_chosen, rank = self._select(candidates_x, candidates_y)

        cp, cc, ccov = p.cp, p.cc, p.ccov
        d_damp, ptarg, pthresh = p.d, p.ptarg, p.pthresh
        xlb, xub = self.bounds[:, 0], self.bounds[:, 1]

        # per-offspring parameter copies (chosen offspring only)
        chosen_off = np.flatnonzero(chosen & is_offspring)
        off_parent = cand_pidxs[chosen_off]
        off_parent_t = torch.as_tensor(off_parent, dtype=torch.long, device=dev)
        K = len(chosen_off)
        if K > 0:
            sig = st.sigmas[off_parent_t].clone()
            last_steps = sig.clone()
            Ainv = st.Ainv[off_parent_t].clone()
            A = st.A[off_parent_t].clone()
            pc = st.pc[off_parent_t].clone()
            ps = st.psucc[off_parent_t].clone()

            # offspring updates (batched): success => psucc up, sigma scaled,
            # rank-1 Cholesky update with normalized step z
            ps = (1.0 - cp) * ps + cp
            sig = sig * torch.exp((ps - ptarg) / (d_damp * (1.0 - ptarg)))[:, None]
            xp = candidates_x[torch.as_tensor(chosen_off, dtype=torch.long, device=dev)]
            xparent = st.parents_x[off_parent_t]
            z = ((xp - xparent) / (xub - xlb)[None, :]) / last_steps
            A, Ainv, pc = batched_cholesky_update(A, Ainv, pc, z, ps, cc, ccov, pthresh)

        # parent-side sequential psucc/sigma updates, in multiplicity rounds
        succ_parents = off_parent  # chosen offspring's parents (success hits)
        fail_parents = cand_pidxs[np.flatnonzero(not_chosen & is_offspring)]
        psucc_host = st.psucc.clone()
        sigmas_host = st.sigmas.clone()

        def _rounds(parents_arr, success: bool):
            nonlocal psucc_host, sigmas_host
            if len(parents_arr) == 0:
                return
            counts = np.bincount(parents_arr, minlength=P)
            max_mult = int(counts.max())
            for r in range(max_mult):
                active = np.flatnonzero(counts > r)
                idx = torch.as_tensor(active, dtype=torch.long, device=dev)
                psu = psucc_host[idx]
                psu = (1.0 - cp) * psu + (cp if success else 0.0)
                psucc_host[idx] = psu
                sigmas_host[idx] = sigmas_host[idx] * torch.exp(
                    (psu - ptarg) / (d_damp * (1.0 - ptarg))
                )[:, None]

        _rounds(succ_parents, success=True)
        _rounds(fail_parents, success=False)

        # rebuild parent set from chosen candidates
        chosen_idx = np.flatnonzero(chosen)
        chosen_t = torch.as_tensor(chosen_idx, dtype=torch.long, device=dev)
        new_x = candidates_x[chosen_t]
        new_y = candidates_y[chosen_t]
        new_rank = torch.as_tensor(rank[chosen_idx], dtype=torch.long, device=dev)

        chosen_is_off = is_offspring[chosen_idx]
        src_parent = torch.as_tensor(
            cand_pidxs[chosen_idx], dtype=torch.long, device=dev
        )
        n_new = len(chosen_idx)
        new_sig = sigmas_host[src_parent]
        new_A = st.A[src_parent]
        new_Ainv = st.Ainv[src_parent]
        new_pc = st.pc[src_parent]
        new_ps = psucc_host[src_parent]
        if K > 0:
            off_slots = np.flatnonzero(chosen_is_off)
            slot_t = torch.as_tensor(off_slots, dtype=torch.long, device=dev)
            new_sig[slot_t] = sig
            new_A[slot_t] = A
            new_Ainv[slot_t] = Ainv
            new_pc[slot_t] = pc
            new_ps[slot_t] = ps

        st.parents_x = new_x
        st.parents_y = new_y
        st.rank = new_rank
        st.sigmas = new_sig
        st.A = new_A
        st.Ainv = new_Ainv
        st.pc = new_pc
        st.psucc = new_ps

        if p.adaptive_population_size:
            self.update_population_size()

    def get_population_strategy(self):
        x, y = ops.remove_duplicates(self.state.parents_x, self.state.parents_y)
        if x.shape[0] > 0:
            x, y, _, _ = ops.remove_worst(x, y, self.popsize)
        return x, y

    def update_population_size(self):
        diversity, cd_spread = self.diversity_indicator.do(
            self.state.rank, self.state.parents_y
        )
        p = self.opt_params
        if diversity < 0.1 or cd_spread < 2.0:
            new_size = min(p.max_population_size, int(p.popsize * 1.1))
        elif diversity > 0.4 and cd_spread > 1.0:
            new_size = max(p.min_population_size, int(p.popsize * 0.9))
        else:
            new_size = p.popsize
        p.popsize = new_size
        p.mu = new_size // 2
