"""Benchmark problem suite — batched torch implementations.

Covers the reference suites (benchmarks/moo_benchmarks.py: DTLZ1-5,7; WFG1,4;
MaF1,2,4; plus the ZDT and constrained problems used in its tests). All
functions here are BATCHED: input (N, d) tensor -> (N, m) objectives (and
optionally (N, c) constraints), so whole populations evaluate as single
device launches. ``scalar(fn)`` adapts any of them to the single-point numpy
convention used by user objective callables.
"""

from __future__ import annotations

import math
from typing import Callable, Optional, Tuple

import numpy as np
import torch

PI = math.pi


def _as2d(x) -> torch.Tensor:
    t = x if isinstance(x, torch.Tensor) else torch.as_tensor(np.asarray(x, dtype=np.float64))
    if t.ndim == 1:
        t = t[None, :]
    return t


# ------------------------------------------------------------------- ZDT
def zdt1(x) -> torch.Tensor:
    x = _as2d(x)
    f1 = x[:, 0]
    g = 1.0 + 9.0 * x[:, 1:].mean(dim=1)
    f2 = g * (1.0 - torch.sqrt(f1 / g))
    return torch.stack([f1, f2], dim=1)


def zdt2(x) -> torch.Tensor:
    x = _as2d(x)
    f1 = x[:, 0]
    g = 1.0 + 9.0 * x[:, 1:].mean(dim=1)
    f2 = g * (1.0 - (f1 / g) ** 2)
    return torch.stack([f1, f2], dim=1)


def zdt3(x) -> torch.Tensor:
    x = _as2d(x)
    f1 = x[:, 0]
    g = 1.0 + 9.0 * x[:, 1:].mean(dim=1)
    r = f1 / g
    f2 = g * (1.0 - torch.sqrt(r) - r * torch.sin(10.0 * PI * f1))
    return torch.stack([f1, f2], dim=1)


def zdt4(x) -> torch.Tensor:
    x = _as2d(x)
    f1 = x[:, 0]
    xi = x[:, 1:]
    g = 1.0 + 10.0 * xi.shape[1] + (xi**2 - 10.0 * torch.cos(4.0 * PI * xi)).sum(dim=1)
    f2 = g * (1.0 - torch.sqrt(f1 / g))
    return torch.stack([f1, f2], dim=1)


def zdt6(x) -> torch.Tensor:
    x = _as2d(x)
    f1 = 1.0 - torch.exp(-4.0 * x[:, 0]) * torch.sin(6.0 * PI * x[:, 0]) ** 6
    g = 1.0 + 9.0 * (x[:, 1:].mean(dim=1)) ** 0.25
    f2 = g * (1.0 - (f1 / g) ** 2)
    return torch.stack([f1, f2], dim=1)


def zdt1_pareto(n_points: int = 100) -> np.ndarray:
    f = np.zeros((n_points, 2))
    f[:, 0] = np.linspace(0, 1, n_points)
    f[:, 1] = 1.0 - np.sqrt(f[:, 0])
    return f


def zdt3_pareto(n_points: int = 500) -> np.ndarray:
    """Sampled true front of ZDT3 (disconnected)."""
    x = np.linspace(0, 1, 10 * n_points)
    f2 = 1.0 - np.sqrt(x) - x * np.sin(10 * np.pi * x)
    pts = np.stack([x, f2], axis=1)
    # keep non-dominated
    keep = np.ones(len(pts), dtype=bool)
    best = np.inf
    order = np.argsort(pts[:, 0])
    for i in order:
        if pts[i, 1] < best:
            best = pts[i, 1]
        else:
            keep[i] = False
    return pts[keep][:n_points]


# ------------------------------------------------------------------ DTLZ
def _dtlz_g2(xm: torch.Tensor) -> torch.Tensor:
    return ((xm - 0.5) ** 2).sum(dim=1)


def _dtlz_g1(xm: torch.Tensor) -> torch.Tensor:
    k = xm.shape[1]
    return 100.0 * (
        k + ((xm - 0.5) ** 2 - torch.cos(20.0 * PI * (xm - 0.5))).sum(dim=1)
    )


def _dtlz_linear(xfront: torch.Tensor, g: torch.Tensor, n_obj: int) -> torch.Tensor:
    n = xfront.shape[0]
    f = torch.empty(n, n_obj, dtype=xfront.dtype, device=xfront.device)
    for i in range(n_obj):
        fi = 0.5 * (1.0 + g)
        for j in range(n_obj - i - 1):
            fi = fi * xfront[:, j]
        if i > 0:
            fi = fi * (1.0 - xfront[:, n_obj - i - 1])
        f[:, i] = fi
    return f


def _dtlz_concave(theta: torch.Tensor, g: torch.Tensor, n_obj: int) -> torch.Tensor:
    n = theta.shape[0]
    f = torch.empty(n, n_obj, dtype=theta.dtype, device=theta.device)
    for i in range(n_obj):
        fi = 1.0 + g
        for j in range(n_obj - i - 1):
            fi = fi * torch.cos(theta[:, j] * PI / 2.0)
        if i > 0:
            fi = fi * torch.sin(theta[:, n_obj - i - 1] * PI / 2.0)
        f[:, i] = fi
    return f


def dtlz1(x, n_obj: int = 3) -> torch.Tensor:
    x = _as2d(x)
    k = x.shape[1] - n_obj + 1
    return _dtlz_linear(x[:, : n_obj - 1], _dtlz_g1(x[:, -k:]), n_obj)


def dtlz2(x, n_obj: int = 3) -> torch.Tensor:
    x = _as2d(x)
    k = x.shape[1] - n_obj + 1
    return _dtlz_concave(x[:, : n_obj - 1], _dtlz_g2(x[:, -k:]), n_obj)


def dtlz3(x, n_obj: int = 3) -> torch.Tensor:
    x = _as2d(x)
    k = x.shape[1] - n_obj + 1
    return _dtlz_concave(x[:, : n_obj - 1], _dtlz_g1(x[:, -k:]), n_obj)


def dtlz4(x, n_obj: int = 3, alpha: float = 100.0) -> torch.Tensor:
    x = _as2d(x)
    k = x.shape[1] - n_obj + 1
    theta = x[:, : n_obj - 1] ** alpha
    return _dtlz_concave(theta, _dtlz_g2(x[:, -k:]), n_obj)


def dtlz5(x, n_obj: int = 3) -> torch.Tensor:
    x = _as2d(x)
    k = x.shape[1] - n_obj + 1
    g = _dtlz_g2(x[:, -k:])
    theta = x[:, : n_obj - 1].clone()
    if n_obj > 2:
        denom = 2.0 * (1.0 + g[:, None])
        theta[:, 1:] = (1.0 + 2.0 * g[:, None] * x[:, 1 : n_obj - 1]) / denom
    return _dtlz_concave(theta, g, n_obj)


def dtlz7(x, n_obj: int = 3) -> torch.Tensor:
    x = _as2d(x)
    k = x.shape[1] - n_obj + 1
    g = 1.0 + 9.0 * x[:, -k:].mean(dim=1)
    f_front = x[:, : n_obj - 1]
    h = n_obj - (
        (f_front / (1.0 + g[:, None])) * (1.0 + torch.sin(3.0 * PI * f_front))
    ).sum(dim=1)
    f_last = (1.0 + g) * h
    return torch.cat([f_front, f_last[:, None]], dim=1)


# ------------------------------------------------------------------- WFG
def _wfg_shape_linear(t: torch.Tensor, m: int) -> torch.Tensor:
    n = t.shape[0]
    f = torch.empty(n, m, dtype=t.dtype, device=t.device)
    for i in range(m):
        fi = torch.ones(n, dtype=t.dtype, device=t.device)
        for j in range(m - i - 1):
            fi = fi * t[:, j]
        if i > 0:
            fi = fi * (1.0 - t[:, m - i - 1])
        f[:, i] = fi
    return f


def _wfg_shape_convex(t: torch.Tensor, m: int) -> torch.Tensor:
    n = t.shape[0]
    f = torch.empty(n, m, dtype=t.dtype, device=t.device)
    for i in range(m):
        fi = torch.ones(n, dtype=t.dtype, device=t.device)
        for j in range(m - i - 1):
            fi = fi * (1.0 - torch.cos(t[:, j] * PI / 2.0))
        if i > 0:
            fi = fi * (1.0 - torch.sin(t[:, m - i - 1] * PI / 2.0))
        f[:, i] = fi
    return f


def wfg1(x, n_obj: int = 3, k: Optional[int] = None) -> torch.Tensor:
    """WFG1 (value parity with reference moo_benchmarks.py wfg1, batched):
    [0, 2i] normalization, ^0.02 bias on the tail, 0.35 + 0.65 scaling,
    chunk-max shape vector, convex shape scaled by (1 + i)."""
    x = _as2d(x)
    n_var = x.shape[1]
    if k is None:
        k = n_obj - 1
    ll = n_var - k
    idx = torch.arange(1, n_var + 1, dtype=x.dtype, device=x.device)
    y = x / (2.0 * idx)
    t1 = y.clone()
    t1[:, k:] = y[:, k:].clamp_min(0.0) ** 0.02
    t2 = t1.clone()
    t2[:, k:] = 0.35 + 0.65 * t1[:, k:]
    xv = torch.empty(x.shape[0], n_obj, dtype=x.dtype, device=x.device)
    for i in range(n_obj - 1):
        xv[:, i] = t2[:, i * ll : (i + 1) * ll].max(dim=1).values
    xv[:, -1] = t2[:, -ll:].mean(dim=1)
    scale = torch.arange(2, n_obj + 2, dtype=x.dtype, device=x.device)
    return _wfg_shape_convex(xv, n_obj) * scale[None, :]


def wfg4(x, n_obj: int = 3, k: Optional[int] = None) -> torch.Tensor:
    """WFG4 (value parity with the reference, batched): multi-modal
    transform y + 0.35 - 0.15 cos(10 pi y - 5), chunk-mean shape vector,
    convex shape scaled by (1 + i)."""
    x = _as2d(x)
    n_var = x.shape[1]
    if k is None:
        k = n_obj - 1
    ll = n_var - k
    idx = torch.arange(1, n_var + 1, dtype=x.dtype, device=x.device)
    y = x / (2.0 * idx)
    t1 = y + 0.35 - 0.15 * torch.cos(10.0 * PI * y - 5.0)
    xv = torch.empty(x.shape[0], n_obj, dtype=x.dtype, device=x.device)
    for i in range(n_obj - 1):
        xv[:, i] = t1[:, i * ll : (i + 1) * ll].mean(dim=1)
    xv[:, -1] = t1[:, -ll:].mean(dim=1)
    scale = torch.arange(2, n_obj + 2, dtype=x.dtype, device=x.device)
    return _wfg_shape_convex(xv, n_obj) * scale[None, :]


# ------------------------------------------------------------------- MaF
def _maf_products(x: torch.Tensor, g: torch.Tensor, n_obj: int,
                  trig: bool) -> torch.Tensor:
    """Common MaF objective products: f[i] = (1+g) * prod_{j<m-i-1} p_j *
    (q_{m-i-1} if i > 0), with (p, q) = (cos(x pi/2), sin(x pi/2)) for the
    concave family or (x, 1-x) for the linear family."""
    n, m = x.shape[0], n_obj
    if trig:
        p = torch.cos(x[:, :m] * PI / 2.0)
        q = torch.sin(x[:, :m] * PI / 2.0)
    else:
        p = x[:, :m]
        q = 1.0 - x[:, :m]
    f = torch.empty(n, m, dtype=x.dtype, device=x.device)
    base = 1.0 + g
    for i in range(m):
        fi = base.clone()
        for j in range(m - i - 1):
            fi = fi * p[:, j]
        if i > 0:
            fi = fi * q[:, m - i - 1]
        f[:, i] = fi
    return f


def maf1(x, n_obj: int = 5) -> torch.Tensor:
    """MaF1 (value parity with the reference): linear products with a
    Rastrigin-style distance term."""
    x = _as2d(x)
    nd = x.shape[1] - n_obj + 1
    xm = x[:, -nd:]
    g = ((xm - 0.5) ** 2 - torch.cos(20.0 * PI * (xm - 0.5))).sum(dim=1)
    return _maf_products(x, g, n_obj, trig=False)


def maf2(x, n_obj: int = 5) -> torch.Tensor:
    """MaF2 (value parity with the reference): concave products with a
    spherical distance term."""
    x = _as2d(x)
    nd = x.shape[1] - n_obj + 1
    g = ((x[:, -nd:] - 0.5) ** 2).sum(dim=1)
    return _maf_products(x, g, n_obj, trig=True)


def maf4(x, n_obj: int = 5) -> torch.Tensor:
    """MaF4 (value parity with the reference): MaF2 objectives scaled by
    10^(2i) — badly-scaled objective ranges."""
    x = _as2d(x)
    nd = x.shape[1] - n_obj + 1
    g = ((x[:, -nd:] - 0.5) ** 2).sum(dim=1)
    f = _maf_products(x, g, n_obj, trig=True)
    scale = torch.pow(
        torch.tensor(10.0, dtype=x.dtype, device=x.device),
        2.0 * torch.arange(n_obj, dtype=x.dtype, device=x.device),
    )
    return f * scale[None, :]


# ----------------------------------------------------- constrained problems
def tnk(x) -> Tuple[torch.Tensor, torch.Tensor]:
    """TNK: 2 vars in [0, pi], 2 objectives, 2 constraints (c >= 0 feasible)."""
    x = _as2d(x)
    f = x.clone()
    x1, x2 = x[:, 0], x[:, 1]
    atan = torch.atan2(x1, x2.clamp_min(1e-30))
    c1 = x1**2 + x2**2 - 1.0 - 0.1 * torch.cos(16.0 * atan)
    c2 = 0.5 - (x1 - 0.5) ** 2 - (x2 - 0.5) ** 2
    return f, torch.stack([c1, c2], dim=1)


def constr(x) -> Tuple[torch.Tensor, torch.Tensor]:
    """CONSTR: x1 in [0.1,1], x2 in [0,5]."""
    x = _as2d(x)
    x1, x2 = x[:, 0], x[:, 1]
    f1 = x1
    f2 = (1.0 + x2) / x1
    c1 = x2 + 9.0 * x1 - 6.0
    c2 = -x2 + 9.0 * x1 - 1.0
    return torch.stack([f1, f2], dim=1), torch.stack([c1, c2], dim=1)


def srn(x) -> Tuple[torch.Tensor, torch.Tensor]:
    """SRN: x in [-20,20]^2."""
    x = _as2d(x)
    x1, x2 = x[:, 0], x[:, 1]
    f1 = 2.0 + (x1 - 2.0) ** 2 + (x2 - 1.0) ** 2
    f2 = 9.0 * x1 - (x2 - 1.0) ** 2
    c1 = 225.0 - x1**2 - x2**2
    c2 = -(x1 - 3.0 * x2 + 10.0)
    return torch.stack([f1, f2], dim=1), torch.stack([c1, c2], dim=1)


def osy(x) -> Tuple[torch.Tensor, torch.Tensor]:
    """OSY: 6 vars, 2 objectives, 6 constraints."""
    x = _as2d(x)
    x1, x2, x3, x4, x5, x6 = (x[:, i] for i in range(6))
    f1 = -(
        25.0 * (x1 - 2.0) ** 2
        + (x2 - 2.0) ** 2
        + (x3 - 1.0) ** 2
        + (x4 - 4.0) ** 2
        + (x5 - 1.0) ** 2
    )
    f2 = (x**2).sum(dim=1)
    c1 = x1 + x2 - 2.0
    c2 = 6.0 - x1 - x2
    c3 = 2.0 - x2 + x1
    c4 = 2.0 - x1 + 3.0 * x2
    c5 = 4.0 - (x3 - 3.0) ** 2 - x4
    c6 = (x5 - 3.0) ** 2 + x6 - 4.0
    return torch.stack([f1, f2], dim=1), torch.stack([c1, c2, c3, c4, c5, c6], dim=1)


def sphere(x, n_obj: int = 2) -> torch.Tensor:
    """Multi-objective sphere: f_i = sum (x - center_i)^2, centers at unit axes."""
    x = _as2d(x)
    n, d = x.shape
    f = torch.empty(n, n_obj, dtype=x.dtype, device=x.device)
    for i in range(n_obj):
        center = torch.zeros(d, dtype=x.dtype, device=x.device)
        center[i % d] = 1.0
        f[:, i] = ((x - center) ** 2).sum(dim=1)
    return f


# ------------------------------------------------------------------ helpers
_PROBLEMS = {
    "zdt1": (zdt1, 2, (0.0, 1.0)),
    "zdt2": (zdt2, 2, (0.0, 1.0)),
    "zdt3": (zdt3, 2, (0.0, 1.0)),
    "zdt4": (zdt4, 2, (0.0, 1.0)),
    "zdt6": (zdt6, 2, (0.0, 1.0)),
    "dtlz1": (dtlz1, None, (0.0, 1.0)),
    "dtlz2": (dtlz2, None, (0.0, 1.0)),
    "dtlz3": (dtlz3, None, (0.0, 1.0)),
    "dtlz4": (dtlz4, None, (0.0, 1.0)),
    "dtlz5": (dtlz5, None, (0.0, 1.0)),
    "dtlz7": (dtlz7, None, (0.0, 1.0)),
    "wfg1": (wfg1, None, None),
    "wfg4": (wfg4, None, None),
    "maf1": (maf1, None, (0.0, 1.0)),
    "maf2": (maf2, None, (0.0, 1.0)),
    "maf4": (maf4, None, (0.0, 1.0)),
    "sphere": (sphere, None, (-5.0, 5.0)),
}


def get_problem(name: str) -> Callable:
    return _PROBLEMS[name][0]


def generate_problem_space(problem_name: str, n_var: int) -> dict:
    """Parameter-space dict {x_i: [lo, hi]} for a named problem."""
    entry = _PROBLEMS[problem_name.lower()]
    if entry[2] is None:  # WFG domain [0, 2i]
        return {f"x{i + 1}": [0.0, 2.0 * (i + 1)] for i in range(n_var)}
    lo, hi = entry[2]
    return {f"x{i + 1}": [lo, hi] for i in range(n_var)}


def get_problem_metadata(problem_name: str, n_obj: int) -> dict:
    name = problem_name.lower()
    meta = {
        "name": name,
        "n_obj": n_obj,
        "pareto_front_type": "unknown",
        "challenges": [],
    }
    fronts = {
        "zdt1": "convex",
        "zdt2": "concave",
        "zdt3": "disconnected",
        "dtlz1": "linear",
        "dtlz2": "concave",
        "dtlz3": "concave-multimodal",
        "dtlz4": "concave-biased",
        "dtlz5": "degenerate-curve",
        "dtlz7": "disconnected",
        "maf1": "inverted-linear",
        "maf2": "concave",
        "maf4": "badly-scaled",
        "wfg1": "mixed-biased",
        "wfg4": "concave-multimodal",
    }
    meta["pareto_front_type"] = fronts.get(name, "unknown")
    return meta


def scalar(fn: Callable, **kwargs) -> Callable:
    """Adapt a batched problem to single-point numpy convention."""

    def wrapped(x: np.ndarray):
        out = fn(np.asarray(x, dtype=np.float64), **kwargs)
        if isinstance(out, tuple):
            return tuple(o.numpy()[0] for o in out)
        return out.numpy()[0]

    return wrapped
