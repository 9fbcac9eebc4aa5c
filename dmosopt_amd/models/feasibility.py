"""Feasibility models (registry name 'logreg').

Interface parity with the reference LogisticFeasibilityModel
(feasibility.py:14-67): per-constraint classifier of P(c_i > 0) with
``predict``, ``predict_proba`` (stacked [P(infeasible), P(feasible)]) and
``rank`` (mean feasible probability — used as an x-distance metric by the
optimizers).

Implementation is torch-native instead of sklearn GridSearchCV pipelines:
standardize -> PCA (SVD) -> L1-regularized logistic regression fit with
batched full-gradient Adam + soft-threshold proximal step, with the
regularization strength chosen by a small validation grid — one batched fit
per constraint, device-capable.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch


class _TorchLogregBatch:
    """G independent L1-logistic fits as ONE batched Adam run: W is (G, d),
    the loss is the sum of per-column BCEs (gradients stay independent), the
    proximal soft-threshold applies per-column lambda. Replaces the former
    per-candidate sequential loop (the whole regularization grid now costs
    one optimization, with a plateau early stop)."""

    def __init__(self, X: torch.Tensor, c: torch.Tensor, l1s, iters: int = 300):
        n, d = X.shape
        l1s = torch.as_tensor(np.asarray(l1s, dtype=np.float64), dtype=X.dtype,
                              device=X.device)
        G = l1s.shape[0]
        W = torch.zeros(G, d, dtype=X.dtype, device=X.device, requires_grad=True)
        b = torch.zeros(G, dtype=X.dtype, device=X.device, requires_grad=True)
        opt = torch.optim.Adam([W, b], lr=0.1)
        lam = (l1s / n)[:, None]
        ct = c[:, None].expand(n, G)
        prev = None
        for it in range(iters):
            opt.zero_grad(set_to_none=True)
            logits = X @ W.T + b[None, :]
            loss = torch.nn.functional.binary_cross_entropy_with_logits(
                logits, ct, reduction="mean"
            )
            loss.backward()
            opt.step()
            with torch.no_grad():  # proximal soft-threshold for L1
                step = 0.1 * lam
                W.copy_(torch.sign(W) * (W.abs() - step).clamp_min(0.0))
            if it % 25 == 24:
                cur = float(loss.detach())
                if prev is not None and abs(prev - cur) < 1e-7:
                    break
                prev = cur
        self.W = W.detach()
        self.b = b.detach()

    def proba_all(self, X: torch.Tensor) -> torch.Tensor:
        """(n, G) feasibility probabilities per grid candidate."""
        return torch.sigmoid(X @ self.W.T + self.b[None, :])

    def select(self, g: int) -> "_TorchLogreg":
        return _TorchLogreg(self.W[g], self.b[g])


class _TorchLogreg:
    def __init__(self, w: torch.Tensor, b: torch.Tensor):
        self.w = w
        self.b = b

    def proba(self, X: torch.Tensor) -> torch.Tensor:
        return torch.sigmoid(X @ self.w + self.b)


class LogisticFeasibilityModel:
    def __init__(self, X, C, device=None, n_components: Optional[int] = None, seed=None):
        X = np.asarray(X, dtype=np.float64)
        C = np.asarray(C, dtype=np.float64)
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        self.dtype = torch.float64 if self.device.type == "cpu" else torch.float32
        Xt = torch.as_tensor(X, dtype=self.dtype, device=self.device)
        # standardize then PCA basis via SVD
        self.mu = Xt.mean(dim=0)
        self.sigma = Xt.std(dim=0, unbiased=False).clamp_min(1e-12)
        Xs = (Xt - self.mu) / self.sigma
        q = min(Xt.shape) if n_components is None else n_components
        U, S, Vh = torch.linalg.svd(Xs, full_matrices=False)
        self.components = Vh[:q]  # (q, d)
        Z = Xs @ self.components.T
        self.X = X
        self.n_constraints = C.shape[1]
        self.clfs = []
        rng = np.random.default_rng(seed)
        for i in range(self.n_constraints):
            c_i = torch.as_tensor(
                (C[:, i] > 0.0).astype(np.float64), dtype=self.dtype, device=self.device
            )
            if len(torch.unique(c_i)) > 1:
                clf = self._fit_with_grid(Z, c_i, rng)
            else:
                clf = None
            self.clfs.append(clf)

    def _fit_with_grid(self, Z, c, rng):
        n = Z.shape[0]
        if n >= 20:
            idx = rng.permutation(n)
            n_val = max(1, n // 5)
            val, tr = idx[:n_val], idx[n_val:]
            l1s = 1.0 / np.logspace(-4, 4, 4)
            batch = _TorchLogregBatch(Z[tr], c[tr], l1s)
            p = batch.proba_all(Z[val]).clamp(1e-7, 1 - 1e-7)  # (n_val, G)
            losses = torch.nn.functional.binary_cross_entropy(
                p, c[val][:, None].expand_as(p), reduction="none"
            ).mean(dim=0)
            return batch.select(int(losses.argmin()))
        batch = _TorchLogregBatch(Z, c, [1.0])
        return batch.select(0)

    # ------------------------------------------------------------- interface
    def _transform(self, x) -> torch.Tensor:
        xt = torch.as_tensor(np.asarray(x, dtype=np.float64), dtype=self.dtype, device=self.device)
        if xt.ndim == 1:
            xt = xt[None, :]
        return ((xt - self.mu) / self.sigma) @ self.components.T

    def predict(self, x):
        Z = self._transform(x)
        ps = []
        for clf in self.clfs:
            if clf is not None:
                ps.append((clf.proba(Z) > 0.5).to(torch.int64).cpu().numpy())
            else:
                ps.append(np.ones(Z.shape[0], dtype=np.int64))
        return np.column_stack(ps)

    def predict_proba(self, x):
        Z = self._transform(x)
        probs = []
        for clf in self.clfs:
            if clf is not None:
                p1 = clf.proba(Z).cpu().numpy()
                probs.append(np.stack([1.0 - p1, p1], axis=1))
            else:
                probs.append(np.tile([0.0, 1.0], (Z.shape[0], 1)))
        return np.stack(probs)

    def rank(self, x):
        pr = self.predict_proba(x)
        return np.mean(pr[:, :, 1], axis=0)
