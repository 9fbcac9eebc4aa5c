"""Property-based invariants for the core ops (hypothesis).

Complements the oracle tests: instead of fixed seeds, these assert
structural properties that must hold for EVERY input — the style of
guarantee a fuzzer can falsify but an example test cannot.
"""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st
from hypothesis.extra.numpy import arrays

from dmosopt_amd import ops
from dmosopt_amd.hv.exact import hv_2d

_finite = st.floats(min_value=0.0, max_value=1.0, allow_nan=False, width=64)


def _front_arrays(max_n=40, m=2):
    return arrays(np.float64, st.tuples(st.integers(2, max_n), st.just(m)),
                  elements=_finite)


@settings(max_examples=60, deadline=None)
@given(_front_arrays(m=3))
def test_front0_is_nondominated_and_exists(Y):
    rank = ops.pareto_rank(torch.as_tensor(Y)).numpy()
    f0 = Y[rank == 0]
    assert len(f0) >= 1
    # no strict dominator of any front-0 member exists anywhere
    for p in f0:
        le = (Y <= p).all(axis=1)
        lt = (Y < p).any(axis=1)
        assert not np.any(le & lt)


@settings(max_examples=60, deadline=None)
@given(_front_arrays(m=3))
def test_rank_monotone_under_domination(Y):
    """Appending a point that strictly dominates everything puts it alone
    in front 0 and shifts every other rank by exactly one."""
    r0 = ops.pareto_rank(torch.as_tensor(Y)).numpy()
    dom = Y.min(axis=0) - 0.25
    Y2 = np.vstack([Y, dom[None, :]])
    r1 = ops.pareto_rank(torch.as_tensor(Y2)).numpy()
    assert r1[-1] == 0
    np.testing.assert_array_equal(r1[:-1], r0 + 1)


@settings(max_examples=60, deadline=None)
@given(_front_arrays(m=2))
def test_hv2d_monotone_and_bounded(Y):
    ref = np.array([1.5, 1.5])
    hv = hv_2d(Y, ref)
    assert 0.0 <= hv <= ref[0] * ref[1] + 1e-12
    # adding a point never decreases the hypervolume
    extra = Y.mean(axis=0) * 0.5
    hv2 = hv_2d(np.vstack([Y, extra[None, :]]), ref)
    assert hv2 >= hv - 1e-12


@settings(max_examples=40, deadline=None)
@given(_front_arrays(m=3), st.integers(1, 20))
def test_remove_worst_keeps_best_ranks(Y, k):
    X = np.ascontiguousarray(Y[:, :2].copy())
    pop = min(k, Y.shape[0])
    _, obj, rank, perm = ops.remove_worst(
        torch.as_tensor(X), torch.as_tensor(Y), pop,
        y_distance_metrics=["crowding"],
    )
    full_rank = ops.pareto_rank(torch.as_tensor(Y)).numpy()
    kept = np.sort(rank.numpy())
    best_possible = np.sort(full_rank)[:pop]
    np.testing.assert_array_equal(kept, best_possible)


@settings(max_examples=40, deadline=None)
@given(_front_arrays(max_n=30, m=4))
def test_crowding_bounded_by_dimensions(Y):
    """Reference semantics (indicators.py:12-51): per-dim normalized
    contributions are <= 1 (boundary points contribute exactly 1), so the
    total crowding is within [0, m]."""
    d = ops.crowding_distance(torch.as_tensor(Y)).numpy()
    assert np.all(d >= -1e-12)
    assert np.all(d <= Y.shape[1] + 1e-9)
