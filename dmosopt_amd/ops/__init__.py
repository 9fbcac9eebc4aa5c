"""Op dispatch layer.

Every op has a pure-PyTorch reference implementation (``torch_ref``) used on
CPU and as the numerics oracle. On CUDA (= HIP/ROCm) tensors the hand-written
gfx950 kernels from the in-tree extension ``dmosopt_amd._hipops`` are used.
Per the framework contract, running on a GPU without the native extension is
an ERROR (no silent eager fallback): build it with
``python setup.py build_ext --inplace`` or ``__graft_entry__.build()``.

Set ``DMOSOPT_AMD_FORCE_TORCH=1`` to force the torch path (debugging only).
"""

from __future__ import annotations

import os
from typing import List, Optional, Sequence, Tuple

import torch

from . import torch_ref
from .torch_ref import (  # re-exported torch-native helpers
    anyclose,
    filter_samples,
    lexsort,
    sbx_crossover_batch,
    polynomial_mutation_batch,
    tournament_selection,
    tournament_prob_vector,
)

_FORCE_TORCH = os.environ.get("DMOSOPT_AMD_FORCE_TORCH", "0") == "1"

_native = None
_native_err: Optional[str] = None


def _load_native():
    global _native, _native_err
    if _native is not None or _native_err is not None:
        return _native
    try:
        from dmosopt_amd import _hipops  # built in-tree by setup.py

        _native = _hipops
    except ImportError as e:  # remember why, for the loud failure path
        _native_err = str(e)
    return _native


def native_available() -> bool:
    return _load_native() is not None


def _use_native(*tensors: torch.Tensor) -> bool:
    if _FORCE_TORCH:
        return False
    if not any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor)):
        return False
    native = _load_native()
    if native is None:
        raise RuntimeError(
            "dmosopt_amd: tensors are on GPU but the native HIP extension "
            "dmosopt_amd._hipops is not built (import error: "
            f"{_native_err}). Build it with `python setup.py build_ext "
            "--inplace` (PYTORCH_ROCM_ARCH=gfx950); refusing to fall back "
            "to eager PyTorch on the GPU."
        )
    return True


# ------------------------------------------------------------------ ranking
def pareto_rank(Y: torch.Tensor) -> torch.Tensor:
    if _use_native(Y):
        # the native binding routes internally: N <= 2048 one-workgroup
        # bit/LDS peels; N > 2048 the grid-wide COOPERATIVE peel (sync-free
        # from the host, all CUs; falls back to the chased matvec peel only
        # if cooperative launch is unavailable). Host syncs in ranking stall
        # pipelined generation loops far beyond their kernel-time cost
        # (round-2 lesson, NOTES.md).
        return _native.pareto_rank(Y.contiguous().float())
    return torch_ref.pareto_rank(Y)


def _pareto_rank_gpu(Y: torch.Tensor) -> torch.Tensor:
    """GPU ranking: native fused dominance-degree matrix, then a
    dominator-count peel driven by one float matvec per front (O(N^2) GEMV
    work per front is microseconds on-device; the former per-front full
    column-max scan was O(N^2) int traffic x #fronts — 208 ms at N=8192 on
    many-front problems). Syncs amortized over 16-front chases."""
    n, m = Y.shape
    if n == 0:
        return torch.zeros(0, dtype=torch.long, device=Y.device)
    D = _native.dominance_degree_matrix(Y.contiguous().float())
    # identical rows already zeroed by the native kernel (incl. diagonal)
    Dom_f = (D == m).to(torch.float32)  # Dom[i][j]: i dominates j
    n_dom = Dom_f.sum(dim=0)  # float counts (exact for n < 2^24)
    rank = torch.zeros(n, dtype=torch.long, device=Y.device)
    alive = torch.ones(n, dtype=torch.bool, device=Y.device)
    k = 0
    remaining = n
    CHASE = 16
    while remaining > 0 and k < n + CHASE:
        for c in range(CHASE):
            front = alive & (n_dom == 0)
            rank = torch.where(front, torch.full_like(rank, k + c), rank)
            alive &= ~front
            # subtract peeled dominators' contributions with one matvec
            n_dom = n_dom - front.to(torch.float32) @ Dom_f
            n_dom = torch.where(alive, n_dom, torch.ones_like(n_dom))
        k += CHASE
        remaining = int(alive.sum().item())
    return rank


def dominance_degree_matrix(Y: torch.Tensor) -> torch.Tensor:
    if _use_native(Y):
        return _native.dominance_degree_matrix(Y.contiguous().float())
    return torch_ref.dominance_degree_matrix(Y)


def crowding_distance(Y: torch.Tensor) -> torch.Tensor:
    if _use_native(Y):
        return _native.crowding_distance(Y.contiguous().float()).to(Y.dtype)
    return torch_ref.crowding_distance(Y)


def euclidean_distance_metric(Y: torch.Tensor) -> torch.Tensor:
    return torch_ref.euclidean_distance_metric(Y)


def fused_rank_metric_perm(rank: torch.Tensor, metric: torch.Tensor) -> torch.Tensor:
    """Fused sort key for the common (rank, -metric) case: one stable argsort
    on (rank << 32 | descending_monotone_bits(metric)) instead of two radix
    sorts + gathers per generation. The IEEE sign-flip mapping
    (b | 0x80000000 for b >= 0, ~b for b < 0) is a total-order bijection
    float32 -> uint32 that handles NEGATIVE metric values too (user-supplied
    callables may return them); ties fall back to index order exactly like
    np.lexsort."""
    d = torch.nan_to_num(
        metric.float(),
        nan=0.0,
        posinf=float(torch.finfo(torch.float32).max),
        neginf=float(torch.finfo(torch.float32).min),
    )
    b = d.view(torch.int32).to(torch.int64) & 0xFFFFFFFF  # raw bits as u32
    asc = torch.where(b >= 0x80000000, (~b) & 0xFFFFFFFF, b | 0x80000000)
    key = (rank.to(torch.int64) << 32) | ((0xFFFFFFFF - asc) & 0xFFFFFFFF)
    return torch.argsort(key, stable=True)


def order_mo(
    x: torch.Tensor,
    y: torch.Tensor,
    x_dists: Optional[List[torch.Tensor]] = None,
    y_distance_metrics: Optional[List] = None,
):
    # rank+crowding use dispatched kernels internally via the metric calls
    rank = pareto_rank(y)
    y_dist_vals: List[torch.Tensor] = []
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
    x_dist_vals = list(x_dists) if x_dists else []
    if (
        not x_dist_vals
        and len(y_dist_vals) == 1
        and y.device.type == "cuda"
    ):
        perm = fused_rank_metric_perm(rank, y_dist_vals[0])
        return perm, rank[perm], (y_dist_vals[0][perm],)
    keys = [-d for d in x_dist_vals] + [-d for d in y_dist_vals] + [rank.to(y.dtype)]
    perm = lexsort(keys)
    return perm, rank[perm], tuple(d[perm] for d in y_dist_vals)


def remove_worst(
    population_parm: torch.Tensor,
    population_obj: torch.Tensor,
    pop: int,
    x_dists: Optional[List[torch.Tensor]] = None,
    y_distance_metrics: Optional[List] = None,
):
    perm, rank, _ = order_mo(
        population_parm, population_obj, x_dists=x_dists, y_distance_metrics=y_distance_metrics
    )
    perm = perm[:pop]
    return population_parm[perm], population_obj[perm], rank[:pop], perm


def top_k_mo(x: torch.Tensor, y: torch.Tensor, top_k=None):
    """Top-k rows by non-dominated sort (reference MOEA.top_k_MO,
    MOEA.py:350-372); returns (x, y) unchanged when top_k is not an int or
    the population is already small enough."""
    if not isinstance(top_k, int) or x.shape[0] <= top_k:
        return x, y
    perm, _, _ = order_mo(x, y)
    perm = perm[:top_k]
    return x[perm], y[perm]


def get_duplicates(X: torch.Tensor, eps: float = 1e-16) -> torch.Tensor:
    if _use_native(X):
        return _native.get_duplicates(X.contiguous().float(), eps)
    return torch_ref.get_duplicates(X, eps)


def remove_duplicates(x: torch.Tensor, y: torch.Tensor, eps: float = 1e-16):
    dup = get_duplicates(x, eps)
    return x[~dup], y[~dup]


# ------------------------------------------------------------- variation ops
def sbx_from_pool(pool, i1, i2, di, lo, hi, seed: int, generator=None):
    """SBX on pool rows (i1, i2) -> (child1, child2), fused RNG on GPU."""
    if _use_native(pool) and pool.dtype == torch.float32:
        return _native.sbx_batch(
            pool.contiguous(),
            i1.to(torch.int32).contiguous(),
            i2.to(torch.int32).contiguous(),
            di.float().contiguous(),
            lo.float().contiguous(),
            hi.float().contiguous(),
            int(seed),
        )
    return sbx_crossover_batch(pool[i1], pool[i2], di, lo, hi, generator=generator)


def mutation_from_pool(pool, idx, di, lo, hi, mutation_rate: float, seed: int, generator=None):
    """Polynomial mutation on pool rows idx, fused RNG on GPU."""
    if _use_native(pool) and pool.dtype == torch.float32:
        return _native.mutation_batch(
            pool.contiguous(),
            idx.to(torch.int32).contiguous(),
            di.float().contiguous(),
            lo.float().contiguous(),
            hi.float().contiguous(),
            float(mutation_rate),
            int(seed),
        )
    return polynomial_mutation_batch(
        pool[idx], di, lo, hi, mutation_rate=mutation_rate, generator=generator
    )


# -------------------------------------------------------------------- GP ops
def gp_predict_mean_fused(Xq, X, theta, alpha, y_mean, y_std, nu, anisotropic):
    """Fused normalized-query posterior mean; None when native doesn't apply."""
    if _use_native(Xq) and X.dtype == torch.float32:
        nu_arg = 0.0 if (nu is None or nu == float("inf")) else float(nu)
        return _native.gp_predict_mean(
            Xq.to(torch.float32).contiguous(), X.contiguous(), theta.contiguous().float(),
            alpha.contiguous().float(), y_mean.float(), y_std.float(),
            nu_arg, bool(anisotropic),
        )
    return None


def gp_nmll_fused(X, theta, y, nu, anisotropic, jitter):
    """Fused batched GP NMLL (assemble + Cholesky + solve + reduce) in one
    extension call; None if the native path does not apply."""
    if _use_native(X) and X.dtype == torch.float32:
        nu_arg = 0.0 if (nu is None or nu == float("inf")) else float(nu)
        return _native.gp_nmll(
            X.contiguous(), theta.contiguous().float(), y.contiguous().float(),
            nu_arg, bool(anisotropic), float(jitter),
        )
    return None


def matern_train_kernel(X, theta, nu, anisotropic, jitter):
    """Batched symmetric kernel matrices K (B,N,N) with noise+jitter diag."""
    if _use_native(X) and X.dtype == torch.float32:
        nu_arg = 0.0 if (nu is None or nu == float("inf")) else float(nu)
        return _native.matern_train(
            X.contiguous(), theta.contiguous().float(), nu_arg, bool(anisotropic), float(jitter)
        )
    from dmosopt_amd.models import gp_core

    return gp_core.build_kernel_torch(X, None, theta, nu=nu, anisotropic=anisotropic, jitter=jitter)


def matern_cross_kernel(Xq, X, theta, nu, anisotropic):
    """Batched cross kernel K* (B,P,N), no diagonal noise."""
    if _use_native(X) and X.dtype == torch.float32:
        nu_arg = 0.0 if (nu is None or nu == float("inf")) else float(nu)
        return _native.matern_cross(
            Xq.contiguous(), X.contiguous(), theta.contiguous().float(), nu_arg, bool(anisotropic)
        )
    from dmosopt_amd.models import gp_core

    return gp_core.build_kernel_torch(Xq, X, theta, nu=nu, anisotropic=anisotropic)


def matern_cross_bf16_kernel(Xq, X, theta, nu, anisotropic, q_lb=None, q_invrg=None):
    """bf16-MFMA cross kernel (config #2 precision path): inputs rounded to
    bf16 in LDS, dot products on v_mfma_f32_16x16x32_bf16 (fp32 accumulate),
    Matern transform and output in fp32. GPU-only — no host fallback (the
    bf16 route is explicitly requested, silence would hide a missing
    extension)."""
    if not (_use_native(X) and X.dtype == torch.float32):
        raise RuntimeError(
            "matern_cross_bf16 requires the gfx950 native extension and "
            "float32 CUDA tensors"
        )
    nu_arg = 0.0 if (nu is None or nu == float("inf")) else float(nu)
    return _native.matern_cross_bf16(
        Xq.contiguous(), X.contiguous(), theta.contiguous().float(), nu_arg,
        bool(anisotropic), q_lb, q_invrg,
    )


def chol_factor_batched_bf16(K):
    """Batched Cholesky with the trailing SYRK updates on the bf16 matrix
    units (panels stay exact fp32). Falls back to the fp32 factorization
    when the native path is unavailable (CPU tests)."""
    if _use_native(K) and K.dtype == torch.float32 and K.shape[1] > 32:
        logdet, info = _native.cholesky_batched_bf16_(K)
        return K, logdet, info
    return chol_factor_batched(K)


def chol_factor_batched(K):
    """Factor K (B,N,N) in place -> (L, logdet (B,), info (B,)).

    Routing: the native launcher picks between the multi-launch
    right-looking path (small batches; panel + MFMA SYRK tile launches)
    and the one-workgroup-per-matrix kernel (B >= ~48) — either way the
    ONLY working path at N ~ 300, where ROCm 7.2's batched rocSOLVER f32
    cholesky raises launch failures. Single large factorizations
    (N >= 1200, where rocSOLVER's multi-CU decomposition wins ~10x and
    was verified working) dispatch to torch/rocSOLVER."""
    if _use_native(K) and K.dtype == torch.float32:
        # crossover re-measured round 2 (profiles/README.md large-N table):
        # native wins to N=2048, parity at 4096
        if K.shape[1] < 2048:
            logdet, info = _native.cholesky_batched_(K)
            return K, logdet, info
    L, info = torch.linalg.cholesky_ex(K)
    logdet = torch.log(torch.diagonal(L, dim1=-2, dim2=-1)).sum(dim=-1)
    return L, logdet, info


def chol_solve_batched(L, Y):
    """Solve K X = Y given the factor L (B,N,N); Y (B,N,R) -> X (B,N,R)."""
    if _use_native(L) and L.dtype == torch.float32:
        Z = Y.contiguous().clone()
        _native.forward_solve_(L.contiguous(), Z)
        _native.backward_solve_(L.contiguous(), Z)
        return Z
    return torch.cholesky_solve(Y, L)


def tri_solve_forward(L, Y):
    """Solve L Z = Y (B,N,R)."""
    if _use_native(L) and L.dtype == torch.float32:
        Z = Y.contiguous().clone()
        _native.forward_solve_(L.contiguous(), Z)
        return Z
    return torch.linalg.solve_triangular(L, Y, upper=False)


__all__ = [
    "pareto_rank",
    "dominance_degree_matrix",
    "crowding_distance",
    "euclidean_distance_metric",
    "order_mo",
    "remove_worst",
    "top_k_mo",
    "get_duplicates",
    "remove_duplicates",
    "lexsort",
    "sbx_crossover_batch",
    "polynomial_mutation_batch",
    "tournament_selection",
    "tournament_prob_vector",
    "anyclose",
    "filter_samples",
    "native_available",
]
