"""Design discrepancy metrics (reference discrepancy.py:38-151).

MD2 / CD2 / SD2 / WD2 (L2-discrepancy families), MinDist and corrscore —
all vectorized over the (n, n) pair matrix instead of the reference's
double Python loops.
"""

from __future__ import annotations

import numpy as np


def _pair_terms(x: np.ndarray):
    xi = x[:, None, :]
    xj = x[None, :, :]
    return xi, xj


def MD2(x: np.ndarray) -> float:
    """Modified L2-discrepancy (the reference's MD2, discrepancy.py:38-59),
    vectorized: sqrt((4/3)^s - 2^(1-s)/n * sum_k prod_j (3 - x_kj^2)
    + 1/n^2 * sum_kl prod_j (2 - max(x_kj, x_lj)))."""
    n, s = x.shape
    t1 = (4.0 / 3.0) ** s
    t2 = (2.0 ** (1 - s) / n) * np.prod(3.0 - x**2, axis=1).sum()
    xi, xj = _pair_terms(x)
    t3 = np.prod(2.0 - np.maximum(xi, xj), axis=2).sum() / (n * n)
    return float(np.sqrt(max(t1 - t2 + t3, 0.0)))


def mixture_discrepancy(x: np.ndarray) -> float:
    """Mixture discrepancy (Zhou et al.)."""
    n, s = x.shape
    d1 = np.abs(x - 0.5)
    t1 = (19.0 / 12.0) ** s
    t2 = (2.0 / n) * np.prod(
        5.0 / 3.0 - 0.25 * d1 - 0.25 * d1**2, axis=1
    ).sum()
    xi, xj = _pair_terms(x)
    di = np.abs(xi - 0.5)
    dj = np.abs(xj - 0.5)
    dij = np.abs(xi - xj)
    t3 = np.prod(15.0 / 8.0 - 0.25 * di - 0.25 * dj - 0.75 * dij + 0.5 * dij**2, axis=2).sum() / (n * n)
    return float(np.sqrt(max(t1 - t2 + t3, 0.0)))


def CD2(x: np.ndarray) -> float:
    """Centered L2 discrepancy."""
    n, s = x.shape
    d = np.abs(x - 0.5)
    t1 = (13.0 / 12.0) ** s
    t2 = (2.0 / n) * np.prod(1.0 + 0.5 * d - 0.5 * d * d, axis=1).sum()
    xi, xj = _pair_terms(x)
    a = 1.0 + 0.5 * (np.abs(xi - 0.5) + np.abs(xj - 0.5)) - 0.5 * np.abs(xi - xj)
    t3 = np.prod(a, axis=2).sum() / (n * n)
    return float(np.sqrt(max(t1 - t2 + t3, 0.0)))


def SD2(x: np.ndarray) -> float:
    """Symmetric L2 discrepancy."""
    n, s = x.shape
    t1 = (4.0 / 3.0) ** s
    t2 = (2.0 / n) * np.prod(1.0 + 2.0 * x - 2.0 * x * x, axis=1).sum()
    xi, xj = _pair_terms(x)
    t3 = (2.0 ** s / (n * n)) * np.prod(1.0 - np.abs(xi - xj), axis=2).sum()
    return float(np.sqrt(max(t1 - t2 + t3, 0.0)))


def WD2(x: np.ndarray) -> float:
    """Wrap-around L2 discrepancy."""
    n, s = x.shape
    xi, xj = _pair_terms(x)
    dij = np.abs(xi - xj)
    t = np.prod(1.5 - dij * (1.0 - dij), axis=2).sum() / (n * n)
    return float(np.sqrt(max(-((4.0 / 3.0) ** s) + t, 0.0)))


def MinDist(x: np.ndarray) -> float:
    """Minimum pairwise euclidean distance (maximin criterion)."""
    n = x.shape[0]
    diff = x[:, None, :] - x[None, :, :]
    D = np.sqrt((diff**2).sum(axis=2))
    D[np.arange(n), np.arange(n)] = np.inf
    return float(D.min())


def corrscore(x: np.ndarray) -> float:
    """Max absolute off-diagonal column correlation."""
    c = np.corrcoef(x.T)
    np.fill_diagonal(c, 0.0)
    return float(np.abs(c).max())
