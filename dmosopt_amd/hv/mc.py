"""Monte Carlo hypervolume estimation (FPRAS and MCM2RV).

Semantics parity with reference hv_adaptive.py:188-461; implemented as
batched torch so the dominance tests (millions of samples x N points) run as
a handful of device launches — the embarrassingly-parallel HIP target of
SURVEY.md section 2.9.

* FPRAS (Karp-Luby union-of-boxes estimator): sample box i with probability
  V_i / sum(V), a uniform point inside it, count the sample iff i is the
  first (lowest-index) box containing it; HV = sum(V) * mean(count).
  Budget M1 = 8 * (1 + eps) * n * ln(2/delta) / eps^2.
* MCM2RV: uniform samples in the [ideal, ref] bounding box; HV =
  box_volume * fraction dominated by any point.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch


def _prep(points, ref_point, device):
    P = torch.as_tensor(np.asarray(points, dtype=np.float64), device=device)
    r = torch.as_tensor(np.asarray(ref_point, dtype=np.float64), device=device)
    mask = (P < r).all(dim=1)
    return P[mask], r


def hv_fpras(
    points,
    ref_point,
    eps: float = 0.01,
    delta: float = 0.01,
    max_samples: int = 2_000_000,
    seed: Optional[int] = None,
    device=None,
    chunk: int = 262_144,
) -> float:
    """FPRAS estimate of the hypervolume (minimization, ref dominated)."""
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    P, r = _prep(points, ref_point, device)
    n = P.shape[0]
    if n == 0:
        return 0.0
    gen = torch.Generator(device=device)
    if seed is not None:
        gen.manual_seed(int(seed))

    vols = torch.prod(r[None, :] - P, dim=1)  # (n,)
    total = float(vols.sum())
    if total <= 0:
        return 0.0
    probs = vols / vols.sum()

    M = int(min(max_samples, math.ceil(8 * (1 + eps) * n * math.log(2.0 / delta) / eps**2)))
    hits = 0
    done = 0
    while done < M:
        b = min(chunk, M - done)
        # choose boxes ~ volume
        box_idx = torch.multinomial(probs, b, replacement=True, generator=gen)
        lo = P[box_idx]  # (b, d)
        u = torch.rand((b, P.shape[1]), dtype=P.dtype, device=device, generator=gen)
        samples = lo + u * (r[None, :] - lo)
        # first containing box: smallest index j with P[j] <= sample (all dims)
        # (b, n) containment matrix via chunked broadcasting
        contains = (P[None, :, :] <= samples[:, None, :]).all(dim=2)  # (b, n)
        first = torch.argmax(contains.to(torch.int8), dim=1)  # first True index
        hits += int((first == box_idx).sum())
        done += b
    return total * hits / M


def hv_mcm2rv(
    points,
    ref_point,
    n_samples: int = 1_000_000,
    seed: Optional[int] = None,
    device=None,
    chunk: int = 262_144,
) -> float:
    """Uniform-sampling MC estimate over the [ideal, ref] bounding box."""
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    P, r = _prep(points, ref_point, device)
    if P.shape[0] == 0:
        return 0.0
    gen = torch.Generator(device=device)
    if seed is not None:
        gen.manual_seed(int(seed))
    ideal = P.min(dim=0).values
    box_vol = float(torch.prod(r - ideal))
    if box_vol <= 0:
        return 0.0
    dominated = 0
    done = 0
    while done < n_samples:
        b = min(chunk, n_samples - done)
        u = torch.rand((b, P.shape[1]), dtype=P.dtype, device=device, generator=gen)
        samples = ideal + u * (r - ideal)[None, :]
        dom = (P[None, :, :] <= samples[:, None, :]).all(dim=2).any(dim=1)
        dominated += int(dom.sum())
        done += b
    return box_vol * dominated / n_samples
