"""Sensitivity analysis on the surrogate (registry 'dgsm', 'fast').

Interface parity with reference sa.py (SA_DGSM / SA_FAST: __init__(lo, hi,
param_names, output_names), analyze(model, num_samples) -> {'S1': {output:
per-param values}, ...}). SALib is not a dependency here — both methods are
implemented natively and batched:

* DGSM: derivative-based global sensitivity measure — mean squared partial
  derivative over Sobol'-sampled points, central finite differences
  evaluated as ONE batched surrogate call of (num_samples x 2d) points.
* FAST: classic Fourier Amplitude Sensitivity Test on the search curve
  x_i = 0.5 + arcsin(sin(w_i s + phi_i))/pi, first-order indices from the
  Fourier power at each parameter frequency's harmonics.
"""

from __future__ import annotations

import math
from typing import List

import numpy as np


class SA_DGSM:
    def __init__(self, lo_bounds, hi_bounds, param_names, output_names, logger=None):
        self.lo = np.asarray(lo_bounds, dtype=np.float64)
        self.hi = np.asarray(hi_bounds, dtype=np.float64)
        self.param_names = list(param_names)
        self.output_names = list(output_names)
        self.logger = logger

    def sample(self, num_samples=10000, seed=0):
        from scipy.stats import qmc

        d = len(self.lo)
        n = min(num_samples, 4096)
        s = qmc.Sobol(d=d, scramble=True, seed=seed).random(n)
        return self.lo + s * (self.hi - self.lo)

    def analyze(self, model, num_samples=10000):
        d = len(self.lo)
        base = self.sample(num_samples)
        n = base.shape[0]
        h = 1e-4 * (self.hi - self.lo)
        # build (n * 2d, d) perturbation block; one batched surrogate call
        plus = np.repeat(base[:, None, :], d, axis=1)
        minus = plus.copy()
        idx = np.arange(d)
        plus[:, idx, idx] = np.clip(plus[:, idx, idx] + h[idx], self.lo[idx], self.hi[idx])
        minus[:, idx, idx] = np.clip(minus[:, idx, idx] - h[idx], self.lo[idx], self.hi[idx])
        queries = np.concatenate([plus.reshape(n * d, d), minus.reshape(n * d, d)], axis=0)
        Y = model.evaluate(queries)
        if isinstance(Y, tuple):
            Y = Y[0]
        Y = np.asarray(Y)
        m = Y.shape[1]
        Yp = Y[: n * d].reshape(n, d, m)
        Ym = Y[n * d :].reshape(n, d, m)
        step = (plus[:, idx, idx] - minus[:, idx, idx])[:, :, None]  # actual step
        grad = (Yp - Ym) / np.where(step == 0, 1.0, step)
        v = (grad**2).mean(axis=0)  # (d, m) DGSM measure
        S1s = [v[:, i] for i in range(m)]
        return {"S1": dict(zip(self.output_names, S1s))}


class SA_FAST:
    def __init__(self, lo_bounds, hi_bounds, param_names, output_names, logger=None, M: int = 4):
        self.lo = np.asarray(lo_bounds, dtype=np.float64)
        self.hi = np.asarray(hi_bounds, dtype=np.float64)
        self.param_names = list(param_names)
        self.output_names = list(output_names)
        self.logger = logger
        self.M = M  # interference factor / number of harmonics

    def _frequencies(self, d: int, n: int) -> np.ndarray:
        # classic FAST frequency assignment (Cukier): w_1 large, rest spread
        M = self.M
        wmax = (n - 1) // (2 * M)
        w = np.ones(d, dtype=np.int64)
        w[0] = wmax
        if d > 1:
            step = max(1, (wmax // (2 * M)) // max(1, d - 1))
            w[1:] = 1 + step * np.arange(d - 1)
        return w

    def sample(self, num_samples=10000, seed=0):
        d = len(self.lo)
        n = num_samples if num_samples % 2 == 1 else num_samples + 1
        self._n = n
        w = self._frequencies(d, n)
        self._w = w
        s = (2.0 * math.pi / n) * np.arange(n)
        rng = np.random.default_rng(seed)
        phi = rng.uniform(0, 2 * math.pi, size=d)
        self._phi = phi
        x01 = 0.5 + np.arcsin(np.sin(w[None, :] * s[:, None] + phi[None, :])) / math.pi
        return self.lo + x01 * (self.hi - self.lo)

    def analyze(self, model, num_samples=10000):
        X = self.sample(num_samples)
        Y = model.evaluate(X)
        if isinstance(Y, tuple):
            Y = Y[0]
        Y = np.asarray(Y)
        n, m = Y.shape[0], Y.shape[1]
        w = self._w
        F = np.fft.rfft(Y - Y.mean(axis=0, keepdims=True), axis=0)
        P = (np.abs(F) ** 2) / n  # power spectrum per output
        total_var = P[1:].sum(axis=0)
        S1s: List[np.ndarray] = []
        for i in range(m):
            s1 = np.zeros(len(w))
            for j, wj in enumerate(w):
                harm = [wj * k for k in range(1, self.M + 1) if wj * k < len(P)]
                s1[j] = P[harm, i].sum() / max(total_var[i], 1e-300)
            S1s.append(s1)
        STs = [np.clip(1.0 - s, 0.0, 1.0) for s in S1s]  # rough complement
        return {
            "S1": dict(zip(self.output_names, S1s)),
            "ST": dict(zip(self.output_names, STs)),
        }
