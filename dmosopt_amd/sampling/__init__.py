"""Initial-design sampling in the unit hypercube [0,1]^{n x s}.

API parity with the reference sampler registry (reference sampling.py,
GLP.py, discrepancy.py): ``mc/lh/slh/glp/sobol(n, s, local_random, maxiter)``
return an (n, s) float64 design. ``maxiter > 0`` applies ranked Gram-Schmidt
de-correlation iterations.

All constructions here are vectorized numpy (the reference uses per-element
Python loops); the GLP generator search scores whole candidate designs with a
vectorized centered-L2 discrepancy.
"""

from __future__ import annotations

import numpy as np
from scipy.stats import qmc


# ---------------------------------------------------------------- designs
def sobol_design(n: int, s: int, local_random) -> np.ndarray:
    sampler = qmc.Sobol(d=s, scramble=True, seed=local_random)
    m = 10
    while (1 << m) < n:
        m += 1
    return sampler.random_base2(m)[:n]


def mc_design(n: int, s: int, local_random) -> np.ndarray:
    return local_random.random(size=(n, s))


def lh_design(n: int, s: int, local_random) -> np.ndarray:
    return qmc.LatinHypercube(d=s, seed=local_random).random(n=n)


def slh_design(n: int, s: int, local_random) -> np.ndarray:
    """Symmetric Latin hypercube: per-column permutations built so that row i
    and row n-1-i are reflections of each other (reference sampling.py:43-77,
    vectorized)."""
    centers = (2.0 * np.arange(1, n + 1) - 1.0) / (2.0 * n)  # n cell midpoints
    k = n // 2
    p = np.zeros((n, s), dtype=np.int64)
    p[:, 0] = np.arange(n)
    if n % 2 == 1:
        p[k, :] = k  # middle row is self-symmetric
    for j in range(1, s):
        top = local_random.permutation(np.arange(k))
        flip = local_random.random(k) < 0.5
        lo = np.where(flip, top, n - 1 - top)
        hi = np.where(flip, n - 1 - top, top)
        p[:k, j] = lo
        p[n - 1 : n - 1 - k : -1, j] = hi
    return centers[p]


# -------------------------------------------------- centered-L2 discrepancy
def cd2(x: np.ndarray) -> float:
    """Centered L2 discrepancy of a design in [0,1]^{n x s} (vectorized)."""
    n, s = x.shape
    d = np.abs(x - 0.5)
    term1 = (13.0 / 12.0) ** s
    term2 = (2.0 / n) * np.prod(1.0 + 0.5 * d - 0.5 * d * d, axis=1).sum()
    # pairwise product term
    a = 1.0 + 0.5 * (d[:, None, :] + d[None, :, :]) - 0.5 * np.abs(
        x[:, None, :] - x[None, :, :]
    )
    term3 = np.prod(a, axis=2).sum() / (n * n)
    return float(np.sqrt(max(term1 - term2 + term3, 0.0)))


def _euler_totatives(n: int) -> np.ndarray:
    v = np.arange(1, n)
    return v[np.gcd(v, n) == 1]


def glp_design(n: int, s: int, local_random, max_candidates: int = 64) -> np.ndarray:
    """Good lattice points with CD2-minimizing generating vector.

    Builds candidate generating vectors from (a) subsets of totatives of
    n+1 (the classical GLP construction uses n+1 points with the last row
    dropped) and (b) power generators a^j mod (n+1); scores each candidate
    design's CD2 and keeps the best. Reference: GLP.py:14-139 semantics.
    """
    m = n + 1
    tot = _euler_totatives(m)
    designs = []
    # power-generator candidates: g_j = a^j mod m
    rng_idx = local_random.permutation(len(tot))[:max_candidates]
    for a in tot[rng_idx]:
        g = np.empty(s, dtype=np.int64)
        val = 1
        ok = True
        seen = set()
        for j in range(s):
            val = (val * int(a)) % m
            if val in seen or val == 0:
                ok = False
                break
            seen.add(val)
            g[j] = val
        if ok:
            designs.append(g)
    if not designs:
        # fall back: random distinct totatives
        for _ in range(max_candidates):
            if len(tot) >= s:
                designs.append(local_random.choice(tot, size=s, replace=False))
    if not designs:
        return lh_design(n, s, local_random)

    best, best_score = None, np.inf
    i = np.arange(1, m)[:, None]  # (n, 1) — drop the final all-ones row later
    for g in designs:
        u = (i * g[None, :]) % m
        # two lattice-to-unit-cube maps compete on CD2: the plain u/m and
        # the reference's CENTERED (2u-1)/(2m) (GLP.py glpmod + centering),
        # which usually scores lower discrepancy
        for x in (u / float(m), (2.0 * u - 1.0) / (2.0 * m)):
            x = x[:n]
            score = cd2(x) if n <= 512 else _cd2_cheap(x)
            if score < best_score:
                best, best_score = x, score
    return best


def _cd2_cheap(x: np.ndarray) -> float:
    """O(n s) surrogate for CD2 used to rank large candidate designs."""
    d = np.abs(x - 0.5)
    return float(-np.prod(1.0 + 0.5 * d - 0.5 * d * d, axis=1).sum())


# ----------------------------------------------------------- decorrelation
def _rmtrend(x: np.ndarray, y: np.ndarray) -> np.ndarray:
    xm = x - x.mean()
    ym = y - y.mean()
    b = (xm * ym).sum() / (xm**2).sum()
    return y - b * xm


def _rand2rank(r: np.ndarray) -> np.ndarray:
    out = np.empty(len(r))
    out[r.argsort()] = np.arange(len(r))
    return out


def decorr(x: np.ndarray, n: int, s: int) -> np.ndarray:
    """One ranked Gram-Schmidt de-correlation sweep (forward + backward)."""
    for j in range(1, s):
        for k in range(j):
            z = _rmtrend(x[:, j], x[:, k])
            x[:, k] = (_rand2rank(z) + 0.5) / n
    for j in range(s - 2, -1, -1):
        for k in range(s - 1, j, -1):
            z = _rmtrend(x[:, j], x[:, k])
            x[:, k] = (_rand2rank(z) + 0.5) / n
    return x


def _with_decorr(design_fun, n, s, local_random, maxiter):
    x = design_fun(n, s, local_random)
    for _ in range(maxiter):
        x = decorr(x, n, s)
    return x


# ------------------------------------------------------------- public API
def mc(n, s, local_random, maxiter=0):
    return mc_design(n, s, local_random)


def lh(n, s, local_random, maxiter=0):
    return _with_decorr(lh_design, n, s, local_random, maxiter)


def slh(n, s, local_random, maxiter=0):
    return _with_decorr(slh_design, n, s, local_random, maxiter)


def glp(n, s, local_random, maxiter=0):
    return _with_decorr(glp_design, n, s, local_random, maxiter)


def sobol(n, s, local_random, maxiter=0):
    return sobol_design(n, s, local_random)
