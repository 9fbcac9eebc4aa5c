"""Multitask exact GP with ICM (intrinsic coregionalization) task covariance.

Role parity with the reference's GPyTorch MEGP_Matern
(model_gpytorch.py:1623-1928): K((x,t),(x',t')) = B[t,t'] * k_x(x,x') +
noise_t — the Kronecker MultitaskKernel structure — with B = W W^T +
diag(v) learned jointly with ARD Matern-5/2 hyperparameters by Adam on the
exact marginal likelihood. The (N*m) x (N*m) system factorizes through the
framework's batched Cholesky path.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch

from dmosopt_amd.models.gp_core import LOG2PI, matern_from_d2, pairwise_sq_dists


class MEGPMaternICM:
    """Surrogate duck-type: __init__(xin, yin, nInput, nOutput, xlb, xub),
    predict(x) -> (mean, var), evaluate(x)."""

    def __init__(
        self, xin, yin, nInput, nOutput, xlb, xub,
        n_iter=150, lr=0.08, rank: Optional[int] = None, seed=None,
        return_mean_variance=False, logger=None, device=None, **kwargs,
    ):
        self.nInput, self.nOutput = nInput, nOutput
        self.xlb = np.asarray(xlb, dtype=np.float64)
        self.xub = np.asarray(xub, dtype=np.float64)
        self.xrg = np.where(self.xub - self.xlb == 0, 1.0, self.xub - self.xlb)
        self.return_mean_variance = return_mean_variance
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        dtype = torch.float64 if self.device.type == "cpu" else torch.float32
        self._dtype = dtype
        if seed is not None:
            torch.manual_seed(int(seed))

        x = (np.asarray(xin, dtype=np.float64) - self.xlb) / self.xrg
        y = np.asarray(yin, dtype=np.float64)
        if y.ndim == 1:
            y = y[:, None]
        self.y_mean = y.mean(axis=0)
        self.y_std = np.where(y.std(axis=0) < 1e-12, 1.0, y.std(axis=0))
        yn = (y - self.y_mean) / self.y_std

        m = nOutput
        r = rank or m
        X = torch.as_tensor(x, dtype=dtype, device=self.device)
        Yv = torch.as_tensor(yn, dtype=dtype, device=self.device).T.reshape(-1)
        # task-major vectorization: rows [t*N + i]
        N = X.shape[0]

        log_ell = torch.full((nInput,), math.log(0.5), dtype=dtype, device=self.device,
                             requires_grad=True)
        W = (0.5 * torch.randn(m, r, dtype=dtype, device=self.device)).requires_grad_(True)
        log_v = torch.full((m,), math.log(0.5), dtype=dtype, device=self.device,
                           requires_grad=True)
        log_noise = torch.full((m,), math.log(1e-2), dtype=dtype, device=self.device,
                               requires_grad=True)
        params = [log_ell, W, log_v, log_noise]
        opt = torch.optim.Adam(params, lr=lr)

        eyeN = torch.eye(N, dtype=dtype, device=self.device)

        def full_K():
            Xs = X / torch.exp(log_ell)[None, :]
            Kx = matern_from_d2(pairwise_sq_dists(Xs, Xs), 2.5)  # (N, N)
            B = W @ W.T + torch.diag(torch.exp(log_v))  # (m, m)
            K = torch.kron(B, Kx)  # (mN, mN), task-major blocks
            noise = torch.exp(log_noise).repeat_interleave(N)
            return K + torch.diag(noise + 1e-6)

        for _ in range(n_iter):
            opt.zero_grad(set_to_none=True)
            K = full_K()
            L, info = torch.linalg.cholesky_ex(K)
            if int(info.item() if info.dim() == 0 else info.sum()) != 0:
                break
            alpha = torch.cholesky_solve(Yv[:, None], L)
            nmll = 0.5 * (Yv[:, None] * alpha).sum() + torch.log(
                torch.diagonal(L)
            ).sum() + 0.5 * len(Yv) * LOG2PI
            nmll.backward()
            opt.step()
            with torch.no_grad():
                log_ell.clamp_(math.log(1e-3), math.log(1e2))
                log_v.clamp_(math.log(1e-4), math.log(1e3))
                log_noise.clamp_(math.log(1e-6), math.log(1e1))

        with torch.no_grad():
            self.X = X
            self.log_ell = log_ell.detach()
            self.B = (W @ W.T + torch.diag(torch.exp(log_v))).detach()
            self.log_noise = log_noise.detach()
            K = full_K().detach()
            self.L = torch.linalg.cholesky(
                K + 1e-5 * torch.eye(len(Yv), dtype=dtype, device=self.device)
            )
            self.alpha = torch.cholesky_solve(Yv[:, None], self.L)  # (mN, 1)

    def predict(self, xin):
        xin = np.asarray(xin, dtype=np.float64)
        if xin.ndim == 1:
            xin = xin[None, :]
        xq = (xin - self.xlb) / self.xrg
        Xq = torch.as_tensor(xq, dtype=self._dtype, device=self.device)
        with torch.no_grad():
            ell = torch.exp(self.log_ell)
            Kxq = matern_from_d2(
                pairwise_sq_dists(Xq / ell[None, :], self.X / ell[None, :]), 2.5
            )  # (P, N)
            Ks = torch.kron(self.B, Kxq)  # (mP, mN)
            mean_v = (Ks @ self.alpha)[:, 0]  # (mP,) task-major
            v = torch.linalg.solve_triangular(self.L, Ks.T, upper=False)  # (mN, mP)
            kss = torch.diagonal(self.B).repeat_interleave(Xq.shape[0]) + torch.exp(
                self.log_noise
            ).repeat_interleave(Xq.shape[0])
            var_v = (kss - (v**2).sum(dim=0)).clamp_min(0.0)
        P = Xq.shape[0]
        mean = mean_v.reshape(self.nOutput, P).T.cpu().numpy() * self.y_std + self.y_mean
        var = var_v.reshape(self.nOutput, P).T.cpu().numpy() * (self.y_std**2)
        return mean, var

    def evaluate(self, x):
        mean, var = self.predict(x)
        if self.return_mean_variance:
            return mean, var
        return mean
