"""Normalization helpers (parity with reference normalization.py)."""

from __future__ import annotations

import numpy as np


class NoNormalization:
    def forward(self, X):
        return X

    def backward(self, X):
        return X


class ZeroToOneNormalization:
    """Normalize to [0,1] given (possibly partially known) bounds."""

    def __init__(self, xl=None, xu=None):
        if xl is None and xu is None:
            self.xl = self.xu = None
            return
        if xl is None:
            xl = np.full_like(np.asarray(xu, dtype=float), np.nan)
        if xu is None:
            xu = np.full_like(np.asarray(xl, dtype=float), np.nan)
        xl = np.array(xl, dtype=float)
        xu = np.array(xu, dtype=float)
        xu[xl == xu] = np.nan
        self.xl, self.xu = xl, xu
        xl_nan, xu_nan = np.isnan(xl), np.isnan(xu)
        self.xl_only = ~xl_nan & xu_nan
        self.xu_only = xl_nan & ~xu_nan
        self.both_nan = xl_nan & xu_nan
        self.neither_nan = ~self.both_nan & ~self.xl_only & ~self.xu_only
        any_nan = xl_nan | xu_nan
        assert np.all((xu >= xl) | any_nan), "xl must be <= xu"

    def forward(self, X):
        if X is None or (self.xl is None and self.xu is None):
            return X
        N = np.array(X, dtype=float)
        nn = self.neither_nan
        N[..., nn] = (np.asarray(X)[..., nn] - self.xl[nn]) / (self.xu[nn] - self.xl[nn])
        N[..., self.xl_only] = np.asarray(X)[..., self.xl_only] - self.xl[self.xl_only]
        N[..., self.xu_only] = 1.0 - (self.xu[self.xu_only] - np.asarray(X)[..., self.xu_only])
        return N

    def backward(self, N):
        if N is None or (self.xl is None and self.xu is None):
            return N
        X = np.array(N, dtype=float)
        nn = self.neither_nan
        X[..., nn] = self.xl[nn] + np.asarray(N)[..., nn] * (self.xu[nn] - self.xl[nn])
        X[..., self.xl_only] = np.asarray(N)[..., self.xl_only] + self.xl[self.xl_only]
        X[..., self.xu_only] = self.xu[self.xu_only] - (1.0 - np.asarray(N)[..., self.xu_only])
        return X


class PreNormalization:
    def __init__(self, zero_to_one=False, ideal=None, nadir=None, **kwargs):
        self.ideal, self.nadir = ideal, nadir
        if zero_to_one:
            assert ideal is not None and nadir is not None
            n_dim = len(ideal)
            self.normalization = ZeroToOneNormalization(ideal, nadir)
            self.ideal, self.nadir = np.zeros(n_dim), np.ones(n_dim)
        else:
            self.normalization = NoNormalization()

    def do(self, *args, **kwargs):
        pass


def normalize(X, xl=None, xu=None, return_bounds=False, estimate_bounds_if_none=True):
    if estimate_bounds_if_none:
        if xl is None:
            xl = np.min(X, axis=0)
        if xu is None:
            xu = np.max(X, axis=0)
    if isinstance(xl, (int, float)):
        xl = np.full(X.shape[-1], xl)
    if isinstance(xu, (int, float)):
        xu = np.full(X.shape[-1], xu)
    norm = ZeroToOneNormalization(xl, xu)
    Xn = norm.forward(X)
    if return_bounds:
        return Xn, norm.xl, norm.xu
    return Xn
