"""Deep GP surrogates (registry: mdgp, mdspp).

Role parity with the reference's GPyTorch deep models
(model_gpytorch.py:991-1622): MDGPMatern is a 2-layer doubly-stochastic
variational deep GP (hidden SVGP layer -> head SVGP layer, Monte Carlo
propagation); MDSPPMatern is the deep sigma-point-process variant — the
hidden-layer distribution is propagated through DETERMINISTIC quadrature
(sigma) points with learned mixture weights instead of MC samples.
Built on the batched SVGP machinery of models/variational_gp.py; inducing
inputs initialize from k-means-style subsampling.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch

from dmosopt_amd.models.variational_gp import _BatchedSVGP

LOG2PI = math.log(2 * math.pi)


class _DeepGPBase:
    n_quadrature: Optional[int] = None  # None -> MC samples (deep GP)
    n_samples: int = 3

    def __init__(
        self, xin, yin, nInput, nOutput, xlb, xub,
        hidden_dim=None, n_inducing=64, n_iter=300, lr=0.03, seed=None,
        return_mean_variance=False, logger=None, device=None, **kwargs,
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
        self._dtype = dtype
        if seed is not None:
            torch.manual_seed(int(seed))
        rng = np.random.default_rng(seed)

        x = (np.asarray(xin, dtype=np.float64) - self.xlb) / self.xrg
        y = np.asarray(yin, dtype=np.float64)
        if y.ndim == 1:
            y = y[:, None]
        self.y_mean = y.mean(axis=0)
        self.y_std = np.where(y.std(axis=0) < 1e-12, 1.0, y.std(axis=0))
        yn = (y - self.y_mean) / self.y_std

        H = hidden_dim or min(nInput, 5)
        self.H = H
        N = x.shape[0]
        M = min(N, n_inducing)
        Z0 = x[rng.permutation(N)[:M]]
        self.layer1 = _BatchedSVGP(H, nInput, M, Z0, dtype=dtype, device=self.device)
        Z1 = rng.standard_normal((M, H)) * 0.5
        self.layer2 = _BatchedSVGP(nOutput, H, M, Z1, dtype=dtype, device=self.device)

        if self.n_quadrature is not None:
            # DSPP: learned mixture weights over fixed sigma points
            Q = self.n_quadrature
            self.quad_z = torch.linspace(-2.0, 2.0, Q, dtype=dtype, device=self.device)
            self.quad_logits = torch.nn.Parameter(
                torch.zeros(Q, dtype=dtype, device=self.device)
            )

        X_t = torch.as_tensor(x, dtype=dtype, device=self.device)
        Y_t = torch.as_tensor(yn, dtype=dtype, device=self.device)
        params = list(self.layer1.parameters()) + list(self.layer2.parameters())
        if self.n_quadrature is not None:
            params.append(self.quad_logits)
        opt = torch.optim.Adam(params, lr=lr)
        for it in range(n_iter):
            opt.zero_grad(set_to_none=True)
            loss = -self._elbo(X_t, Y_t)
            if not torch.isfinite(loss):
                break
            loss.backward()
            opt.step()
            with torch.no_grad():
                for lyr in (self.layer1, self.layer2):
                    lyr.log_ell.clamp_(math.log(1e-3), math.log(1e2))
                    lyr.log_sf2.clamp_(math.log(1e-4), math.log(1e3))
                    lyr.log_noise.clamp_(math.log(1e-6), math.log(1e1))

    def _kl(self, layer):
        q_sqrt = torch.tril(layer.q_sqrt)
        diag = torch.diagonal(q_sqrt, dim1=1, dim2=2).abs().clamp_min(1e-300)
        return 0.5 * (
            (layer.q_mu**2).sum() + (q_sqrt**2).sum()
            - layer.L * layer.M - 2.0 * torch.log(diag).sum()
        )

    def _hidden_draws(self, X):
        """Propagate X through layer 1: list of (H-dim hidden inputs, weight)."""
        m1, v1 = self.layer1._post(X)  # (H, n)
        std1 = torch.sqrt(v1)
        draws = []
        if self.n_quadrature is None:
            for _ in range(self.n_samples):
                eps = torch.randn_like(m1)
                draws.append(((m1 + eps * std1).T, 1.0 / self.n_samples))
        else:
            w = torch.softmax(self.quad_logits, dim=0)
            for q in range(len(self.quad_z)):
                draws.append(((m1 + self.quad_z[q] * std1).T, w[q]))
        return draws

    def _elbo(self, X, Y):
        n = X.shape[0]
        lik = torch.zeros((), dtype=X.dtype, device=X.device)
        noise = torch.exp(self.layer2.log_noise)[:, None]
        for h, w in self._hidden_draws(X):
            m2, v2 = self.layer2._post(h)  # (m, n)
            ll = -0.5 * (LOG2PI + torch.log(noise) + ((Y.T - m2) ** 2 + v2) / noise).sum()
            lik = lik + (w if isinstance(w, torch.Tensor) else torch.tensor(w, dtype=X.dtype, device=X.device)) * ll
        return lik - self._kl(self.layer1) - self._kl(self.layer2)

    # ----------------------------------------------------------------- API
    def predict(self, xin):
        xin = np.asarray(xin, dtype=np.float64)
        if xin.ndim == 1:
            xin = xin[None, :]
        x = (xin - self.xlb) / self.xrg
        X_t = torch.as_tensor(x, dtype=self._dtype, device=self.device)
        with torch.no_grad():
            means, variances, weights = [], [], []
            for h, w in self._hidden_draws(X_t):
                m2, v2 = self.layer2._post(h)
                means.append(m2)
                variances.append(v2)
                weights.append(float(w))
            W = np.asarray(weights)
            W = W / W.sum()
            Ms = torch.stack(means)  # (S, m, n)
            Vs = torch.stack(variances)
            Wt = torch.as_tensor(W, dtype=Ms.dtype, device=Ms.device)[:, None, None]
            mean = (Wt * Ms).sum(dim=0)
            # mixture variance: E[v] + E[m^2] - E[m]^2 + noise
            var = (Wt * (Vs + Ms**2)).sum(dim=0) - mean**2
            var = var + torch.exp(self.layer2.log_noise)[:, None]
        m = mean.T.cpu().numpy() * self.y_std + self.y_mean
        v = var.T.cpu().numpy() * (self.y_std**2)
        return m, np.maximum(v, 0.0)

    def evaluate(self, x):
        mean, var = self.predict(x)
        if self.return_mean_variance:
            return mean, var
        return mean


class MDGPMatern(_DeepGPBase):
    """'mdgp': 2-layer MC deep GP (reference model_gpytorch.py:1308-1622)."""

    n_quadrature = None


class MDSPPMatern(_DeepGPBase):
    """'mdspp': deep sigma-point process (reference model_gpytorch.py:991-1307)."""

    n_quadrature = 8
