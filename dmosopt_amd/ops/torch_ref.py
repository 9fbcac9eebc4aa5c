"""Pure-PyTorch reference implementations of the population ops.

These are the numerics oracles for the HIP kernels in ``ops/hip`` and the
CPU execution path. Semantics follow the reference implementation
(``/root/reference/dmosopt/dda.py``, ``MOEA.py``, ``indicators.py``) but are
fully vectorized: no per-individual Python loops. All functions take/return
torch tensors; float64 on CPU, float32 on device by default.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch

Tensor = torch.Tensor


# ------------------------------------------------------------------ ranking
def dominance_degree_matrix(Y: Tensor) -> Tensor:
    """D[i,j] = #objectives where Y[i,k] <= Y[j,k]  (N x N int32).

    Reference: dda.py:25-31.
    """
    n, d = Y.shape
    D = torch.zeros((n, n), dtype=torch.int32, device=Y.device)
    for k in range(d):
        yk = Y[:, k]
        D += (yk[:, None] <= yk[None, :]).to(torch.int32)
    return D


def pareto_rank(Y: Tensor) -> Tensor:
    """0-based Pareto front index per row of Y (minimization).

    DDA ranking (dda.py:34-76 semantics: D[i][j] == m, after zeroing
    mutual-domination entries of identical rows, means i dominates j) with
    a dominator-COUNT front peel: n_dom[j] = #alive dominators; a front is
    n_dom == 0; peeling subtracts only the peeled rows' contributions —
    O(N^2) total instead of a full-matrix max per front.
    """
    n, d = Y.shape
    if n == 0:
        return torch.zeros(0, dtype=torch.long, device=Y.device)
    D = dominance_degree_matrix(Y)
    # zero out mutual-domination entries for identical objective rows
    identical = (D == d) & (D.T == d)
    Dom = (D == d) & ~identical  # Dom[i][j]: i dominates j
    n_dom = Dom.sum(dim=0)  # (n,) alive dominator counts
    rank = torch.zeros(n, dtype=torch.long, device=Y.device)
    alive = torch.ones(n, dtype=torch.bool, device=Y.device)
    k = 0
    remaining = n
    while remaining > 0:
        front = alive & (n_dom == 0)
        if not bool(front.any()):
            front = alive  # numerical safety: avoid infinite loop
        rank[front] = k
        alive &= ~front
        idx = front.nonzero(as_tuple=True)[0]
        remaining -= int(idx.numel())
        if remaining > 0:
            n_dom = n_dom - Dom[idx].sum(dim=0)
        k += 1
    return rank


def crowding_distance(Y: Tensor) -> Tensor:
    """Crowding distance metric (reference indicators.py:12-51), vectorized.

    Normalizes Y per-dimension to [0,1]; boundary points get 1.0 per dim,
    interior points get the gap US[i+1]-US[i-1]; scatter-added back.
    """
    n, d = Y.shape
    if n == 1:
        return torch.ones(1, dtype=Y.dtype, device=Y.device)
    lb = Y.min(dim=0, keepdim=True).values
    ub = Y.max(dim=0, keepdim=True).values
    span = (ub - lb).clamp_min_(0)
    span = torch.where(span == 0, torch.ones_like(span), span)
    U = (Y - lb) / span

    idx = U.argsort(dim=0)  # (n, d) indices of sorted order per dim
    US = torch.gather(U, 0, idx)
    DS = torch.empty_like(US)
    DS[0, :] = 1.0
    DS[-1, :] = 1.0
    if n > 2:
        DS[1:-1, :] = US[2:, :] - US[:-2, :]
    # scatter-add per dimension: D[idx[i,j]] += DS[i,j]
    D = torch.zeros(n, dtype=Y.dtype, device=Y.device)
    D.scatter_add_(0, idx.T.reshape(-1), DS.T.reshape(-1))
    D = torch.nan_to_num(D, nan=0.0)
    return D


def euclidean_distance_metric(Y: Tensor) -> Tensor:
    """Row norms of per-dimension normalized Y (indicators.py:54-65)."""
    lb = Y.min(dim=0).values
    ub = Y.max(dim=0).values
    span = ub - lb
    span = torch.where(span == 0, torch.ones_like(span), span)
    U = (Y - lb) / span
    return torch.sqrt((U * U).sum(dim=1))


def lexsort(keys: Sequence[Tensor]) -> Tensor:
    """np.lexsort equivalent: last key is primary. Stable sorts in sequence."""
    n = keys[0].shape[0]
    perm = torch.arange(n, device=keys[0].device)
    for k in keys:  # least-significant first
        kk = k[perm]
        order = torch.argsort(kk, stable=True)
        perm = perm[order]
    return perm


def order_mo(
    x: Tensor,
    y: Tensor,
    x_dists: Optional[List[Tensor]] = None,
    y_distance_metrics: Optional[List] = None,
) -> Tuple[Tensor, Tensor, Tuple[Tensor, ...]]:
    """Permutation of a non-dominated sort: by (rank, -y_dists..., -x_dists...).

    Mirrors reference MOEA.sortMO/orderMO (MOEA.py:242-347): primary key is
    pareto rank, then negated objective-space distances, then negated
    x-space distances. Returns (perm, rank[perm], y_dists_sorted).
    """
    rank = pareto_rank(y)
    y_dist_vals: List[Tensor] = []
    if y_distance_metrics:
        for metric in y_distance_metrics:
            if callable(metric):
                y_dist_vals.append(metric(y))
            elif metric == "crowding":
                y_dist_vals.append(crowding_distance(y))
            elif metric == "euclidean":
                y_dist_vals.append(euclidean_distance_metric(y))
            else:
                raise RuntimeError(f"order_mo: unknown distance metric {metric}")
    x_dist_vals: List[Tensor] = list(x_dists) if x_dists else []
    # np.lexsort((−x0, ..., −y0, ..., rank)): rank primary, then −y, then −x
    keys = [-d for d in x_dist_vals] + [-d for d in y_dist_vals] + [rank.to(y.dtype)]
    perm = lexsort(keys)
    y_sorted_dists = tuple(d[perm] for d in y_dist_vals)
    return perm, rank[perm], y_sorted_dists


# ----------------------------------------------------------------- variation
def sbx_crossover_batch(
    parent1: Tensor,
    parent2: Tensor,
    di_crossover: Tensor,
    xlb: Tensor,
    xub: Tensor,
    u: Optional[Tensor] = None,
    generator: Optional[torch.Generator] = None,
) -> Tuple[Tensor, Tensor]:
    """Batched SBX crossover (reference MOEA.py:215-239).

    parent1/parent2: (B, d). Returns two (B, d) children clipped to bounds.
    """
    if u is None:
        u = torch.rand(parent1.shape, dtype=parent1.dtype, device=parent1.device, generator=generator)
    di = di_crossover.to(parent1.dtype)
    beta = torch.where(
        u <= 0.5,
        (2.0 * u) ** (1.0 / (di + 1.0)),
        (1.0 / (2.0 * (1.0 - u))) ** (1.0 / (di + 1.0)),
    )
    c1 = 0.5 * ((1.0 - beta) * parent1 + (1.0 + beta) * parent2)
    c2 = 0.5 * ((1.0 + beta) * parent1 + (1.0 - beta) * parent2)
    return c1.clamp(xlb, xub), c2.clamp(xlb, xub)


def polynomial_mutation_batch(
    parent: Tensor,
    di_mutation: Tensor,
    xlb: Tensor,
    xub: Tensor,
    mutation_rate: float = 0.5,
    u: Optional[Tensor] = None,
    generator: Optional[torch.Generator] = None,
) -> Tensor:
    """Batched polynomial mutation (reference MOEA.py:191-212).

    A gene mutates 'low' when u < mutation_rate (delta in [-1,0]) else 'high'
    (delta in [0,1]); the child is parent + (xub-xlb)*delta, clipped.
    """
    if u is None:
        u = torch.rand(parent.shape, dtype=parent.dtype, device=parent.device, generator=generator)
    di = di_mutation.to(parent.dtype)
    delta_lo = (2.0 * u) ** (1.0 / (di + 1.0)) - 1.0
    delta_hi = 1.0 - (2.0 * (1.0 - u)) ** (1.0 / (di + 1.0))
    delta = torch.where(u < mutation_rate, delta_lo, delta_hi)
    child = parent + (xub - xlb) * delta
    return child.clamp(xlb, xub)


_TOURNAMENT_PROB_CACHE = {}


def tournament_prob_vector(n: int, p: float = 0.5) -> torch.Tensor:
    """Geometric selection probabilities p(1-p)^i over sorted candidates
    (cached per (n, p): rebuilt thousands of times per epoch otherwise)."""
    key = (n, p)
    out = _TOURNAMENT_PROB_CACHE.get(key)
    if out is None:
        i = torch.arange(n, dtype=torch.float64)
        prob = p * (1.0 - p) ** i
        out = prob / prob.sum()
        _TOURNAMENT_PROB_CACHE[key] = out
    return out


def tournament_selection(
    pop: int,
    poolsize: int,
    metrics: Sequence[Tensor],
    np_random,
    generator=None,
) -> Tensor:
    """Tournament selection into the mating pool (reference MOEA.py:385-395).

    Sorts candidates by lexsort(metrics) and samples `poolsize` without
    replacement with geometric probability over sorted position. With a
    device ``generator`` and device metrics the weighted draw runs on the
    GPU via the Gumbel top-k trick (exact weighted sampling without
    replacement: argmax_k of log(p_i) + Gumbel noise); otherwise the host
    numpy Generator draws, matching the reference's control stream.
    """
    # adaptive population sizing can briefly set poolsize above the CURRENT
    # population (the reference's update_population_size jumps straight to
    # min_population_size, NSGA2.py:268-270, and would crash its own
    # replace=False draw) — clamp to what exists
    poolsize = min(poolsize, pop)
    dev_metrics = [m if isinstance(m, torch.Tensor) else torch.as_tensor(m) for m in metrics]
    sorted_candidates = lexsort(dev_metrics)
    dev = sorted_candidates.device
    if generator is not None and dev.type == "cuda":
        logp = _tournament_logp(pop, dev)
        u = torch.rand(pop, dtype=torch.float32, device=dev, generator=generator)
        keys = logp - torch.log(-torch.log(u.clamp_min(1e-12)).clamp_min(1e-12))
        pool_pos = torch.topk(keys, poolsize).indices
        return sorted_candidates[pool_pos]
    prob = tournament_prob_vector(pop).numpy()  # cached
    pool_pos = np_random.choice(pop, size=poolsize, p=prob, replace=False)
    pool_pos = torch.as_tensor(pool_pos, dtype=torch.long, device=dev)
    return sorted_candidates[pool_pos]


_TOURNAMENT_LOGP_CACHE = {}


def _tournament_logp(pop: int, device) -> Tensor:
    key = (pop, str(device))
    out = _TOURNAMENT_LOGP_CACHE.get(key)
    if out is None:
        out = tournament_prob_vector(pop).log().to(device=device, dtype=torch.float32)
        _TOURNAMENT_LOGP_CACHE[key] = out
    return out


# ----------------------------------------------------------------- distance
def get_duplicates(X: Tensor, eps: float = 1e-16) -> Tensor:
    """Boolean mask of rows that duplicate an earlier row.

    Reference MOEA.get_duplicates (MOEA.py:426-436): pairwise euclidean
    distances, upper triangle (incl. diagonal) set to inf, then a row is a
    duplicate if any distance in its row is <= eps. With Y=X this marks, for
    each pair within eps, the row with the *smaller* index.
    """
    n = X.shape[0]
    if n == 0:
        return torch.zeros(0, dtype=torch.bool, device=X.device)
    # donot_use_mm: the GEMM (x^2-2xy+y^2) formulation leaves exact
    # duplicates at ~eps-sized residuals whose ULPs depend on the BLAS
    # threading/partitioning of the moment — the eps=1e-16 threshold then
    # flips run to run (observed: same-seed archives diverging). The
    # direct-difference path is deterministic and exact at zero distance.
    D = torch.cdist(X.double(), X.double(),
                    compute_mode="donot_use_mm_for_euclid_dist")
    iu = torch.triu_indices(n, n, offset=0, device=X.device)
    D[iu[0], iu[1]] = float("inf")
    D = torch.nan_to_num(D, nan=float("inf"))
    return (D <= eps).any(dim=1)


def anyclose(x: Tensor, X: Tensor, rtol: float = 1e-4, atol: float = 1e-8) -> bool:
    """True if any row of X is allclose to x."""
    if X.numel() == 0:
        return False
    return bool(torch.isclose(X, x[None, :], rtol=rtol, atol=atol).all(dim=1).any())


# ------------------------------------------------------------------- misc
def remove_worst(
    population_parm: Tensor,
    population_obj: Tensor,
    pop: int,
    x_dists: Optional[List[Tensor]] = None,
    y_distance_metrics: Optional[List] = None,
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Elitist survivor selection: non-dominated sort, keep top `pop`.

    Returns (x, y, rank, perm) truncated to pop rows (MOEA.py:398-423).
    """
    perm, rank, _ = order_mo(
        population_parm, population_obj, x_dists=x_dists, y_distance_metrics=y_distance_metrics
    )
    perm = perm[:pop]
    return population_parm[perm], population_obj[perm], rank[:pop], perm


def filter_samples(y: Tensor, *companions, nan: str = "remove", outliers: str = "ignore"):
    """NaN / outlier filtering of objective rows (MOEA.py:445-467)."""
    mask = torch.ones(y.shape[0], dtype=torch.bool, device=y.device)
    if nan == "max":
        m = torch.nan_to_num(y).max(dim=0).values
        fill = torch.maximum(1e3 * m, torch.full_like(m, 1e5))
        y = torch.where(torch.isnan(y), fill[None, :], y)
    elif nan == "remove":
        mask = ~torch.isnan(y).any(dim=1)
    else:
        y = torch.nan_to_num(y, nan=float(nan))

    if outliers == "zscore":
        ylog = torch.log(y + 1.0)
        z = (ylog - ylog.mean(dim=0)) / ylog.std(dim=0, unbiased=True)
        mask = ~(z.abs() > 2).any(dim=1)

    out = [y[mask]]
    for c in companions:
        out.append(c[mask] if c is not None else None)
    return tuple(out)
