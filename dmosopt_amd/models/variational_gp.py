"""Variational GP surrogates (registry: vgp, svgp, spv, siv, crv).

Role parity with the reference's GPflow models (model.py:98-1179):
per-objective (sparse) variational GPs with Matern-5/2 kernels and ELBO
training, plus the three multi-output SVGP structures — SeparateIndependent
('spv'), SharedIndependent ('siv') and LinearCoregionalization ('crv', W
mixing). Implementation is torch-native and BATCHED over latent GPs: all
latents' inducing kernels, Cholesky factors and ELBO terms evaluate as one
(L, M, M) batch per step, device-capable.

Whitened parameterization: q(u) = N(m, S), S = T T^T, f = K_zx^T K_zz^{-1/2} u.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch

from dmosopt_amd.models.gp_core import matern_from_d2, pairwise_sq_dists

LOG2PI = math.log(2 * math.pi)


class _BatchedSVGP(torch.nn.Module):
    """L independent latent SVGPs with shared-structure batched math."""

    def __init__(self, L, d, M, Z_init, ard=True, shared_kernel=False,
                 nu=2.5, n_noise=None, dtype=torch.float64, device="cpu"):
        super().__init__()
        self.L, self.d, self.M = L, d, M
        self.nu = nu
        self.shared_kernel = shared_kernel
        kshape = (1,) if shared_kernel else (L,)
        ell_shape = kshape + ((d,) if ard else (1,))
        self.log_ell = torch.nn.Parameter(
            torch.full(ell_shape, math.log(0.5), dtype=dtype, device=device)
        )
        self.log_sf2 = torch.nn.Parameter(torch.zeros(kshape, dtype=dtype, device=device))
        self.log_noise = torch.nn.Parameter(
            torch.full((n_noise or L,), math.log(1e-2), dtype=dtype, device=device)
        )
        Zt = torch.as_tensor(Z_init, dtype=dtype, device=device)
        if Zt.dim() == 2:
            Zt = Zt[None].repeat(1 if shared_kernel else L, 1, 1)
        self.Z = torch.nn.Parameter(Zt.clone())
        self.q_mu = torch.nn.Parameter(torch.zeros(L, M, dtype=dtype, device=device))
        eye = torch.eye(M, dtype=dtype, device=device)
        self.q_sqrt = torch.nn.Parameter(eye[None].repeat(L, 1, 1).clone())
        self.dtype, self.device_ = dtype, device

    def _k(self, X1, X2):
        """Batched kernel (B, n1, n2); X inputs (B, n, d) or (n, d)."""
        ell = torch.exp(self.log_ell)
        sf2 = torch.exp(self.log_sf2)
        if X1.dim() == 2:
            X1 = X1[None]
        if X2.dim() == 2:
            X2 = X2[None]
        x1 = X1 / ell[:, None, :]
        x2 = X2 / ell[:, None, :]
        d2 = pairwise_sq_dists(x1, x2)
        return sf2[:, None, None] * matern_from_d2(d2, self.nu)

    def _post(self, X):
        """q(f) mean/var at X (n, d) -> ((L, n), (L, n))."""
        M = self.M
        Kzz = self._k(self.Z, self.Z)
        eye = torch.eye(M, dtype=Kzz.dtype, device=Kzz.device)[None]
        sf2max = torch.exp(self.log_sf2).max().detach()
        jitter = 1e-6 * float(sf2max) + 1e-8
        Lz = None
        for _ in range(6):
            Lz, info = torch.linalg.cholesky_ex(Kzz + jitter * eye)
            if int(info.sum()) == 0:
                break
            jitter *= 10.0
        if Lz is None or int(info.sum()) != 0:
            Lz, _ = torch.linalg.cholesky_ex(Kzz + (1e-1 * float(sf2max) + 1e-3) * eye)
        Kzx = self._k(self.Z, X)  # (1|L, M, n)
        A = torch.linalg.solve_triangular(Lz, Kzx, upper=False)  # (1|L, M, n)
        if A.shape[0] == 1 and self.L > 1:
            A = A.expand(self.L, *A.shape[1:])
        mean = (A * self.q_mu[:, :, None]).sum(dim=1)  # (L, n)
        sf2 = torch.exp(self.log_sf2)
        kxx = sf2.expand(self.L)[:, None].expand(self.L, X.shape[0])
        q_sqrt = torch.tril(self.q_sqrt)
        SA = torch.bmm(q_sqrt.transpose(1, 2), A.contiguous())
        var = kxx - (A**2).sum(dim=1) + (SA**2).sum(dim=1)
        return mean, var.clamp_min(1e-12)

    def elbo(self, X, Y):
        """Y: (n, L) targets. Returns scalar ELBO (to maximize)."""
        mean, var = self._post(X)
        noise = torch.exp(self.log_noise)[:, None]
        ll = -0.5 * (
            LOG2PI + torch.log(noise) + ((Y.T - mean) ** 2 + var) / noise
        ).sum()
        q_sqrt = torch.tril(self.q_sqrt)
        diag = torch.diagonal(q_sqrt, dim1=1, dim2=2).abs().clamp_min(1e-300)
        kl = 0.5 * (
            (self.q_mu**2).sum()
            + (q_sqrt**2).sum()
            - self.L * self.M
            - 2.0 * torch.log(diag).sum()
        )
        return ll - kl


class _VGPBase:
    """Surrogate duck-type wrapper around _BatchedSVGP."""

    n_latent_mode = "per_objective"  # or "shared", "coregional"
    inducing_fraction: Optional[float] = None  # None -> all points (VGP)
    min_inducing = 100

    def __init__(
        self, xin, yin, nInput, nOutput, xlb, xub,
        n_iter=300, lr=0.05, num_latent=None, seed=None,
        batch_size=None, return_mean_variance=False, logger=None, device=None,
        **kwargs,
    ):
        self.nInput, self.nOutput = nInput, nOutput
        self.xlb = np.asarray(xlb, dtype=np.float64)
        self.xub = np.asarray(xub, dtype=np.float64)
        self.xrg = np.where(self.xub - self.xlb == 0, 1.0, self.xub - self.xlb)
        self.return_mean_variance = return_mean_variance
        self.logger = logger
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        dtype = torch.float64 if self.device.type == "cpu" else torch.float32

        x = (np.asarray(xin, dtype=np.float64) - self.xlb) / self.xrg
        y = np.asarray(yin, dtype=np.float64)
        if y.ndim == 1:
            y = y[:, None]
        self.y_mean = y.mean(axis=0)
        self.y_std = np.where(y.std(axis=0) < 1e-12, 1.0, y.std(axis=0))
        yn = (y - self.y_mean) / self.y_std

        rng = np.random.default_rng(seed)
        N = x.shape[0]
        if self.inducing_fraction is None:
            M = N
            Z = x.copy()
        else:
            M = min(N, max(self.min_inducing, int(self.inducing_fraction * N)))
            Z = x[rng.permutation(N)[:M]]

        if self.n_latent_mode == "coregional":
            L = num_latent or nOutput
        elif self.n_latent_mode == "shared":
            L = nOutput
        else:
            L = nOutput
        shared = self.n_latent_mode == "shared"
        self.L = L
        self.model = _BatchedSVGP(
            L, nInput, M, Z, shared_kernel=shared,
            n_noise=max(L, nOutput), dtype=dtype, device=self.device,
        )
        self.W = None
        if self.n_latent_mode == "coregional":
            self.W = torch.nn.Parameter(
                torch.randn(nOutput, L, dtype=dtype, device=self.device) * 0.5
            )

        X_t = torch.as_tensor(x, dtype=dtype, device=self.device)
        Y_t = torch.as_tensor(yn, dtype=dtype, device=self.device)
        params = list(self.model.parameters()) + ([self.W] if self.W is not None else [])
        opt = torch.optim.Adam(params, lr=lr)
        prev = None
        for it in range(n_iter):
            opt.zero_grad(set_to_none=True)
            if self.W is None:
                loss = -self.model.elbo(X_t, Y_t)
            else:
                mean, var = self.model._post(X_t)  # (L, n)
                fm = (self.W @ mean)  # (m, n)
                fv = (self.W**2) @ var
                noise = torch.exp(self.model.log_noise[: self.nOutput])[:, None]
                ll = -0.5 * (
                    LOG2PI + torch.log(noise) + ((Y_t.T - fm) ** 2 + fv) / noise
                ).sum()
                q_sqrt = torch.tril(self.model.q_sqrt)
                diag = torch.diagonal(q_sqrt, dim1=1, dim2=2).abs().clamp_min(1e-300)
                kl = 0.5 * (
                    (self.model.q_mu**2).sum() + (q_sqrt**2).sum()
                    - self.model.L * self.model.M - 2.0 * torch.log(diag).sum()
                )
                loss = -(ll - kl)
            loss.backward()
            opt.step()
            with torch.no_grad():  # keep kernel hyper-parameters in sane ranges
                self.model.log_ell.clamp_(math.log(1e-3), math.log(1e2))
                self.model.log_sf2.clamp_(math.log(1e-4), math.log(1e3))
                self.model.log_noise.clamp_(math.log(1e-6), math.log(1e1))
            # ELBO %-change early stop (reference model.py ELBO loop)
            cur = float(loss.detach())
            if prev is not None and it > 50 and abs(prev - cur) / max(abs(prev), 1e-12) < 1e-5:
                break
            prev = cur
        self._dtype = dtype

    def predict(self, xin):
        xin = np.asarray(xin, dtype=np.float64)
        if xin.ndim == 1:
            xin = xin[None, :]
        x = (xin - self.xlb) / self.xrg
        X_t = torch.as_tensor(x, dtype=self._dtype, device=self.device)
        with torch.no_grad():
            mean, var = self.model._post(X_t)  # (L, n)
            if self.W is not None:
                mean = self.W @ mean
                var = (self.W**2) @ var
            noise = torch.exp(self.model.log_noise[: self.nOutput])[:, None]
            var = var[: self.nOutput] + noise
        m = mean[: self.nOutput].T.cpu().numpy() * self.y_std + self.y_mean
        v = var.T.cpu().numpy() * (self.y_std**2)
        return m, v

    def evaluate(self, x):
        mean, var = self.predict(x)
        if self.return_mean_variance:
            return mean, var
        return mean


class VGPMatern(_VGPBase):
    """'vgp': variational GP with inducing points at all training inputs
    (reference VGP_Matern, model.py:991-1179)."""

    inducing_fraction = None


class SVGPMatern(_VGPBase):
    """'svgp': sparse variational GP, inducing fraction 0.2 / min 100
    (reference SVGP_Matern, model.py:769-988)."""

    inducing_fraction = 0.2


class SPVMatern(_VGPBase):
    """'spv': SeparateIndependent multi-output SVGP (model.py:547-766)."""

    inducing_fraction = 0.2
    n_latent_mode = "per_objective"


class SIVMatern(_VGPBase):
    """'siv': SharedIndependent multi-output SVGP (model.py:328-544)."""

    inducing_fraction = 0.2
    n_latent_mode = "shared"


class CRVMatern(_VGPBase):
    """'crv': LinearCoregionalization multi-output SVGP with mixing W
    (model.py:98-325)."""

    inducing_fraction = 0.2
    n_latent_mode = "coregional"
