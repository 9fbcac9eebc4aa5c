"""Exact-GP math core — batched over hyperparameter candidates and objectives.

All heavy math flows through a small internal API (kernel build, Cholesky,
solves) so the gfx950 HIP kernels can replace the torch ops 1:1:

    K = signal_var * Matern_nu(||x-x'|| / ell) + noise_var * I

theta layout (log-space, sklearn-compatible ordering; reference
model.py:1227-1229): [log signal_var, log ell (1 or d values), log noise_var].

Reference semantics: per-objective sklearn GaussianProcessRegressor with
ConstantKernel*Matern(nu=2.5)+WhiteKernel, inputs scaled to [0,1], y
standardized (normalize_y=True), hyperparameters from SCE-UA on the negative
log marginal likelihood (reference model.py:1182-1275, 1419-1753).
"""

from __future__ import annotations

import contextlib
import math
from typing import Optional, Tuple

import torch


def _strict_cpu_det() -> bool:
    import os

    return os.environ.get("DMOSOPT_CPU_STRICT_DET", "0") == "1"


@contextlib.contextmanager
def _single_threaded_cpu_math(enabled: bool):
    """OPT-IN (DMOSOPT_CPU_STRICT_DET=1): pin torch to ONE intra-op thread
    for a CPU numerics region.

    Defense against thread-team-dependent BLAS rounding. The same-seed
    reproducibility flake this was built for turned out to be the
    UNSEEDED surrogate search (profiles/README.md, determinism hunt) —
    with that fixed, 40/40 same-seed process pairs reproduce bitwise and
    direct MKL repeatability probes (200x nmll under load) never showed
    drift, so the ~8x CPU fit cost of sequential math is not paid by
    default. Enable for belt-and-braces CPU-rank determinism auditing."""
    if not (enabled and _strict_cpu_det()):
        yield
        return
    n0 = torch.get_num_threads()
    torch.set_num_threads(1)
    try:
        yield
    finally:
        torch.set_num_threads(n0)

SQRT5 = math.sqrt(5.0)
SQRT3 = math.sqrt(3.0)
LOG2PI = math.log(2.0 * math.pi)


def pairwise_sq_dists(X1: torch.Tensor, X2: torch.Tensor) -> torch.Tensor:
    """Squared euclidean distances, (..., N1, N2). Uses the |a|^2+|b|^2-2ab
    expansion so the inner product maps onto MFMA GEMM in the HIP kernel."""
    n1 = (X1 * X1).sum(dim=-1, keepdim=True)  # (..., N1, 1)
    n2 = (X2 * X2).sum(dim=-1, keepdim=True).transpose(-1, -2)  # (..., 1, N2)
    d2 = n1 + n2 - 2.0 * (X1 @ X2.transpose(-1, -2))
    return d2.clamp_min_(0.0)


def matern_from_d2(d2: torch.Tensor, nu: float) -> torch.Tensor:
    """Matern correlation from squared scaled distance r^2 = d2.

    sqrt is taken of d2 + tiny so autograd through the kernel diagonal
    (d2 == 0) yields finite gradients instead of the inf * 0 = NaN that
    d(sqrt)/d(d2)|_0 produces; the value perturbation is ~1e-12 in r.
    """
    if nu == 2.5:
        r = torch.sqrt(d2 + 1e-24)
        s = SQRT5 * r
        return (1.0 + s + (5.0 / 3.0) * d2) * torch.exp(-s)
    if nu == 1.5:
        r = torch.sqrt(d2 + 1e-24)
        s = SQRT3 * r
        return (1.0 + s) * torch.exp(-s)
    if nu == 0.5:
        return torch.exp(-torch.sqrt(d2 + 1e-24))
    if nu == float("inf") or nu is None:  # RBF
        return torch.exp(-0.5 * d2)
    raise ValueError(f"Unsupported Matern nu={nu}")


def build_kernel_torch(
    X1: torch.Tensor,
    X2: Optional[torch.Tensor],
    theta: torch.Tensor,
    nu: float = 2.5,
    anisotropic: bool = False,
    jitter: float = 0.0,
) -> torch.Tensor:
    """Kernel matrices for a BATCH of hyperparameters.

    X1: (N1, d); X2: (N2, d) or None (=X1, adds noise_var+jitter on diag).
    theta: (B, p) log-hyperparameters. Returns (B, N1, N2).
    """
    B = theta.shape[0]
    d = X1.shape[-1]
    sf2 = torch.exp(theta[:, 0])  # (B,)
    noise = torch.exp(theta[:, -1])  # (B,)
    sym = X2 is None
    if X2 is None:
        X2 = X1
    if anisotropic:
        ell = torch.exp(theta[:, 1 : 1 + d])  # (B, d)
        x1s = X1[None, :, :] / ell[:, None, :]
        x2s = X2[None, :, :] / ell[:, None, :]
        d2 = pairwise_sq_dists(x1s, x2s)  # (B, N1, N2)
    else:
        ell = torch.exp(theta[:, 1])  # (B,)
        d2 = pairwise_sq_dists(X1, X2)[None, :, :] / (ell * ell)[:, None, None]
    K = sf2[:, None, None] * matern_from_d2(d2, nu)
    if sym:
        n = X1.shape[0]
        idx = torch.arange(n, device=X1.device)
        K[:, idx, idx] += (noise + jitter)[:, None]
    return K


def build_kernel(
    X1: torch.Tensor,
    X2: Optional[torch.Tensor],
    theta: torch.Tensor,
    nu: float = 2.5,
    anisotropic: bool = False,
    jitter: float = 0.0,
) -> torch.Tensor:
    """Dispatching kernel build: gfx950 fused-assembly kernel on GPU,
    torch on CPU."""
    from dmosopt_amd import ops

    if X2 is None:
        return ops.matern_train_kernel(X1, theta, nu, anisotropic, jitter)
    return ops.matern_cross_kernel(X1, X2, theta, nu, anisotropic)


class _NmllGraph:
    """hipGraph-captured fused NMLL pipeline (gfx950).

    One SCE-UA fit issues ~170 fused-NMLL calls with IDENTICAL shapes (the
    stage batch is always 3*S*G candidates against the same X and the same
    per-candidate y rows) — each call a fixed ~27-launch pipeline (kernel
    assembly + 20 Cholesky panel/SYRK launches + solve + reduce) that is
    launch-gap bound at B ~ 18 workgroups. Capturing the pipeline once and
    replaying it turns those ~27 dispatches into 3 small D2D input copies +
    one graph launch. Inputs are copied into graph-owned buffers before
    replay, so allocator address reuse cannot alias stale data."""

    def __init__(self, X, y, theta, nu, anisotropic, jitter):
        from dmosopt_amd import ops

        self.Xb = X.contiguous().clone()
        self.yb = y.contiguous().clone()
        self.tb = theta.contiguous().clone()
        self.args = (nu, anisotropic, jitter)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):  # warmup outside capture
                ops.gp_nmll_fused(self.Xb, self.tb, self.yb, nu, anisotropic, jitter)
        torch.cuda.current_stream().wait_stream(side)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = ops.gp_nmll_fused(
                self.Xb, self.tb, self.yb, nu, anisotropic, jitter
            )
        # NOTE: a hipMemsetAsync enqueued inside the captured region raced
        # with the following kernel's accumulation ON REPLAY (ROCm 7.2;
        # bit-nondeterministic outputs with bit-identical inputs,
        # scripts_det_debug6.py) — the multik Cholesky therefore zeroes
        # logdet with an explicit kernel (cholesky.hip zero_f32_kernel)
        # instead of a memset. Keep memsets out of captured pipelines.

    def run(self, X, y, theta):
        self.Xb.copy_(X)
        self.yb.copy_(y)
        self.tb.copy_(theta)
        self.graph.replay()
        return self.out.clone()


_nmll_graphs: dict = {}


def _nmll_graph_enabled() -> bool:
    import os

    return (
        os.environ.get("DMOSOPT_NMLL_GRAPH", "1") == "1"
        and os.environ.get("DMOSOPT_CHOL_OVERLAP", "0") != "1"  # multi-stream
    )


def batched_nmll(
    X: torch.Tensor,
    y: torch.Tensor,
    theta: torch.Tensor,
    nu: float = 2.5,
    anisotropic: bool = False,
    jitter: float = 1e-10,
    differentiable: bool = False,
) -> torch.Tensor:
    """Negative log marginal likelihood for a batch of theta.

    X: (N, d), y standardized targets: (N,) shared by all candidates, or
    (B, N) per-candidate rows (mixed-objective batches). theta: (B, p).
    Returns (B,). Failed factorizations get +inf (SCE-UA treats them as bad
    points). ``differentiable=True`` forces the autograd-capable torch path
    (Adam optimizer); the default dispatches to the fused gfx950 kernels.
    """
    with _single_threaded_cpu_math(not X.is_cuda):
        N = X.shape[0]
        B = theta.shape[0]

        def _yb():
            return (
                y[None, :, None].expand(B, N, 1) if y.dim() == 1 else y[:, :, None]
            ).contiguous()

        if differentiable:
            yb = _yb()
            K = build_kernel_torch(X, None, theta, nu=nu, anisotropic=anisotropic, jitter=jitter)
            L, info = torch.linalg.cholesky_ex(K)
            alpha = torch.cholesky_solve(yb, L)
            quad = (yb * alpha).sum(dim=(1, 2))
            logdet = 2.0 * torch.log(torch.diagonal(L, dim1=-2, dim2=-1)).sum(dim=-1)
        else:
            from dmosopt_amd import ops

            if (
                X.is_cuda
                and X.dtype == torch.float32
                and ops.native_available()
                and _nmll_graph_enabled()
            ):
                key = (
                    B, N, X.shape[1], theta.shape[1], y.dim(), nu, anisotropic,
                    float(jitter),
                )
                g = _nmll_graphs.get(key)
                if g is None:
                    if len(_nmll_graphs) >= 8:  # archives grow: drop stale shapes
                        _nmll_graphs.clear()
                    g = _nmll_graphs[key] = _NmllGraph(
                        X.float(), y.float(), theta.float(), nu, anisotropic, jitter
                    )
                return g.run(X.float(), y.float(), theta.float())
            fused = ops.gp_nmll_fused(
                X, theta, y if y.dim() == 2 else y, nu, anisotropic, jitter
            )
            if fused is not None:
                return fused
            K = build_kernel(X, None, theta, nu=nu, anisotropic=anisotropic, jitter=jitter)
            L, half_logdet, info = ops.chol_factor_batched(K)
            # both paths return sum(log diag L)
            logdet = 2.0 * half_logdet
            z = ops.tri_solve_forward(L, _yb())  # L z = y
            quad = (z * z).sum(dim=(1, 2))
        nmll = 0.5 * quad + 0.5 * logdet + 0.5 * N * LOG2PI
        nmll = torch.where(info != 0, torch.full_like(nmll, float("inf")), nmll)
        nmll = torch.where(torch.isfinite(nmll), nmll, torch.full_like(nmll, float("inf")))
        return nmll


class FittedGP:
    """Posterior state for a batch of independent per-objective GPs sharing X.

    theta: (m, p) per-objective log-hyperparameters.
    """

    def __init__(
        self,
        X: torch.Tensor,
        Y: torch.Tensor,
        theta: torch.Tensor,
        y_mean: torch.Tensor,
        y_std: torch.Tensor,
        nu: float = 2.5,
        anisotropic: bool = False,
        jitter: float = 1e-10,
        compute: str = "fp32",
    ):
        self.X = X
        self.theta = theta
        self.nu = nu
        self.anisotropic = anisotropic
        self.y_mean = y_mean  # (m,)
        self.y_std = y_std  # (m,)
        # "bf16": posterior Cholesky uses bf16-MFMA trailing updates and
        # predictions use the bf16 cross kernel (BASELINE config #2); the
        # SCE-UA fit that produced theta is always fp32 (bit-stability)
        self.compute = compute if X.is_cuda else "fp32"
        from dmosopt_amd import ops

        m, N = theta.shape[0], X.shape[0]
        with _single_threaded_cpu_math(not X.is_cuda):
            self._build_posterior(X, Y, theta, y_mean, y_std, nu, anisotropic,
                                  jitter, ops, m, N)

    def _build_posterior(self, X, Y, theta, y_mean, y_std, nu, anisotropic,
                         jitter, ops, m, N):
        K = build_kernel(X, None, theta, nu=nu, anisotropic=anisotropic, jitter=jitter)
        if self.compute == "bf16":
            self.L, _, info = ops.chol_factor_batched_bf16(K)
        else:
            self.L, _, info = ops.chol_factor_batched(K)  # (m, N, N)
        if int(info.sum()) != 0:
            # escalate jitter for failed objectives
            for _ in range(5):
                bad = info != 0
                if not bool(bad.any()):
                    break
                jitter *= 100.0
                K2 = build_kernel(
                    X, None, theta, nu=nu, anisotropic=anisotropic, jitter=jitter
                )
                L2, _, info = ops.chol_factor_batched(K2)
                self.L = torch.where(bad[:, None, None], L2, self.L)
        Yn = (Y - y_mean[None, :]) / y_std[None, :]  # (N, m) standardized
        yb = Yn.T[:, :, None].contiguous()  # (m, N, 1)
        self.alpha = ops.chol_solve_batched(self.L, yb)  # (m, N, 1)
        # On GPU, precompute K^-1 so per-generation posterior variance is a
        # pair of batched GEMMs (rocBLAS/MFMA) instead of P triangular solves
        self.Kinv: Optional[torch.Tensor] = None
        if X.is_cuda:
            eye = torch.eye(N, dtype=X.dtype, device=X.device)[None].expand(m, N, N).contiguous()
            self.Kinv = ops.chol_solve_batched(self.L, eye)

    def predict(self, Xq: torch.Tensor, return_var: bool = True):
        """Posterior mean and variance at Xq (P, d) -> ((P, m), (P, m)).

        Variance matches sklearn's return_std**2: diag K(x*,x*) [incl. noise
        from the White term] minus k*^T K^-1 k*, scaled by y_std^2. With
        ``return_var=False`` (the per-generation surrogate-evaluate path)
        the quadratic term is skipped and (mean, None) returned.
        """
        if not return_var and self.compute != "bf16":
            from dmosopt_amd import ops

            fused = ops.gp_predict_mean_fused(
                Xq, self.X, self.theta, self.alpha, self.y_mean, self.y_std,
                self.nu, self.anisotropic,
            )
            if fused is not None:
                return fused, None
        if self.compute == "bf16":
            from dmosopt_amd import ops

            Ks = ops.matern_cross_bf16_kernel(
                Xq.float(), self.X, self.theta, self.nu, self.anisotropic
            )
        elif not Xq.is_cuda:
            # deterministic CPU math (thread-team-independent): the
            # replicated-rank scheme compares these values across processes
            with _single_threaded_cpu_math(True):
                return self._predict_torch(Xq, return_var)
        else:
            Ks = build_kernel(Xq, self.X, self.theta, nu=self.nu, anisotropic=self.anisotropic)
        # (m, P, N)
        mean_n = torch.bmm(Ks, self.alpha)[:, :, 0]  # (m, P)
        if not return_var:
            return self.y_mean[None, :] + self.y_std[None, :] * mean_n.T, None
        return self._finish_var(Xq, Ks, mean_n)

    def _predict_torch(self, Xq, return_var):
        Ks = build_kernel(Xq, self.X, self.theta, nu=self.nu,
                          anisotropic=self.anisotropic)
        mean_n = torch.bmm(Ks, self.alpha)[:, :, 0]  # (m, P)
        if not return_var:
            return self.y_mean[None, :] + self.y_std[None, :] * mean_n.T, None
        return self._finish_var(Xq, Ks, mean_n)

    def _finish_var(self, Xq, Ks, mean_n):
        sf2 = torch.exp(self.theta[:, 0])
        noise = torch.exp(self.theta[:, -1])
        kss = (sf2 + noise)[:, None]  # (m, 1): k(x,x) = sf2*1 + noise
        if self.Kinv is not None:
            W = torch.bmm(Ks, self.Kinv)  # (m, P, N)
            quad = (Ks * W).sum(dim=2)  # (m, P)
        else:
            v = torch.linalg.solve_triangular(self.L, Ks.transpose(-1, -2), upper=False)
            quad = (v * v).sum(dim=1)
        var_n = (kss - quad).clamp_min(0.0)  # (m, P)
        mean = self.y_mean[None, :] + self.y_std[None, :] * mean_n.T
        var = (self.y_std[None, :] ** 2) * var_n.T
        return mean, var
