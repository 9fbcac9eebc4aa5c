"""Hypervolume tests: known values, cross-method agreement, MC accuracy."""

import itertools

import numpy as np
import pytest

from dmosopt_amd.hv.exact import (
    HyperVolumeBoxDecomposition,
    compute_hypervolume_box_decomposition,
    hv_2d,
    hv_3d,
)
from dmosopt_amd.hv.mc import hv_fpras, hv_mcm2rv


def grid_hv_oracle(points, ref, n_grid=64):
    """Brute-force grid oracle: fraction of dominated cells x box volume."""
    points = np.asarray(points)
    ref = np.asarray(ref)
    d = points.shape[1]
    ideal = points.min(axis=0)
    axes = [np.linspace(ideal[j], ref[j], n_grid, endpoint=False) + (ref[j] - ideal[j]) / (2 * n_grid) for j in range(d)]
    mesh = np.stack(np.meshgrid(*axes, indexing="ij"), axis=-1).reshape(-1, d)
    dominated = (points[None, :, :] <= mesh[:, None, :]).all(axis=2).any(axis=1)
    cell = np.prod((ref - ideal) / n_grid)
    return dominated.sum() * cell


def test_hv2d_single_point():
    assert hv_2d(np.array([[1.0, 1.0]]), np.array([3.0, 3.0])) == pytest.approx(4.0)


def test_hv2d_staircase():
    pts = np.array([[1.0, 3.0], [2.0, 2.0], [3.0, 1.0]])
    ref = np.array([4.0, 4.0])
    # manual: areas (1)(1) strips: x in [1,2): h=1 ... compute directly
    expected = (2 - 1) * (4 - 3) + (3 - 2) * (4 - 2) + (4 - 3) * (4 - 1)
    assert hv_2d(pts, ref) == pytest.approx(expected)


def test_hv2d_dominated_points_ignored():
    pts = np.array([[1.0, 1.0], [2.0, 2.0]])
    assert hv_2d(pts, np.array([3.0, 3.0])) == pytest.approx(4.0)


def test_hv3d_cube():
    pts = np.array([[1.0, 1.0, 1.0]])
    assert hv_3d(pts, np.array([2.0, 2.0, 2.0])) == pytest.approx(1.0)


def test_hv3d_two_points():
    pts = np.array([[0.0, 1.0, 0.0], [1.0, 0.0, 0.0]])
    ref = np.array([2.0, 2.0, 1.0])
    # z slice [0,1): union of two rectangles: 2x1 + 1x2 - 1x1 = 3
    assert hv_3d(pts, ref) == pytest.approx(3.0)


@pytest.mark.parametrize("d", [4, 5])
def test_box_decomposition_vs_grid(d, rng):
    pts = rng.random((12, d))
    ref = np.ones(d) * 1.1
    hv_box = HyperVolumeBoxDecomposition(ref).compute_hypervolume(pts)
    oracle = grid_hv_oracle(pts, ref, n_grid=24 if d == 4 else 14)
    assert hv_box == pytest.approx(oracle, rel=0.08)


def test_box_decomposition_consistent_with_sweeps(rng):
    # pad 2D points into 4D with constant dims: volumes scale by pad volume
    pts2 = rng.random((20, 2))
    ref2 = np.array([1.2, 1.2])
    hv2 = hv_2d(pts2, ref2)
    pad = np.full((20, 2), 0.5)
    pts4 = np.hstack([pts2, pad])
    ref4 = np.array([1.2, 1.2, 1.0, 1.0])
    hv4 = HyperVolumeBoxDecomposition(ref4).compute_hypervolume(pts4)
    assert hv4 == pytest.approx(hv2 * 0.5 * 0.5, rel=1e-9)


def test_functional_interface_routes():
    pts = np.array([[0.5, 0.5]])
    assert compute_hypervolume_box_decomposition(pts, np.array([1.0, 1.0])) == pytest.approx(0.25)


def test_mc_estimators_agree_with_exact(rng):
    pts = rng.random((15, 3))
    ref = np.ones(3) * 1.1
    exact = hv_3d(pts, ref)
    est_f = hv_fpras(pts, ref, eps=0.03, seed=1, device="cpu")
    est_m = hv_mcm2rv(pts, ref, n_samples=400_000, seed=2, device="cpu")
    assert est_f == pytest.approx(exact, rel=0.05)
    assert est_m == pytest.approx(exact, rel=0.05)


def test_ehvi_matches_reference_formula(rng):
    """Oracle re-derivation of the reference's per-box EHVI expression
    (hv_box_decomposition.py:391-440): partial = std*(phi(l)-phi(u)) +
    mean*(Phi(u)-Phi(l)) per dim, product over dims, sum over boxes."""
    from scipy.stats import norm

    front = rng.random((6, 2))
    ref = np.array([1.5, 1.5])
    hv = HyperVolumeBoxDecomposition(ref)
    means = rng.random((5, 2))
    variances = np.full((5, 2), 0.02)
    sel, vals = hv.select_candidates(front, means, variances, n_select=5)

    lowers, uppers = hv._decompose_dominated_space(front)
    expected = np.zeros(5)
    for i in range(5):
        std = np.sqrt(variances[i])
        total = 0.0
        for L, U in zip(lowers, uppers):
            prod = 1.0
            for j in range(2):
                Phi_l = 0.0 if np.isinf(L[j]) else norm.cdf((L[j] - means[i, j]) / std[j])
                Phi_u = 1.0 if np.isinf(U[j]) else norm.cdf((U[j] - means[i, j]) / std[j])
                phi_l = 0.0 if np.isinf(L[j]) else norm.pdf((L[j] - means[i, j]) / std[j])
                phi_u = 0.0 if np.isinf(U[j]) else norm.pdf((U[j] - means[i, j]) / std[j])
                prod *= std[j] * (phi_l - phi_u) + means[i, j] * (Phi_u - Phi_l)
            total += prod
        expected[i] = total
    order = np.argsort(-expected, kind="stable")
    assert np.array_equal(sel, order)
    assert np.allclose(vals, expected[order], rtol=1e-10)


def test_adaptive_router_and_hybrid(rng):
    from dmosopt_amd.hv.adaptive import AdaptiveHyperVolume, DominanceAnalysis, estimate_overlap

    pts = rng.random((15, 3))
    ref = np.ones(3) * 1.1
    exact = hv_3d(pts, ref)
    # exact routing below threshold
    ahv = AdaptiveHyperVolume(ref, mc_dim_threshold=10)
    assert ahv.compute(pts) == pytest.approx(exact, rel=1e-12)
    # MC routing above threshold agrees within tolerance
    ahv_mc = AdaptiveHyperVolume(ref, mc_dim_threshold=0, mc_method="hybrid", seed=3)
    val, conf = ahv_mc.compute_with_confidence(pts)
    assert val == pytest.approx(exact, rel=0.06)
    assert 0 < conf <= 1

    # dominance analysis agrees with brute force
    da = DominanceAnalysis(pts)
    samples = rng.random((500, 3)) * 1.1
    brute = (pts[None, :, :] <= samples[:, None, :]).all(axis=2).any(axis=1)
    assert np.array_equal(da.dominates_any(samples), brute)

    ov = estimate_overlap(pts, ref)
    assert 0 < ov <= 1


def test_engine_get_feasible(rng):
    from dmosopt_amd.core.engine import get_feasible

    x = rng.random((40, 4))
    y = rng.random((40, 2))
    c = rng.random((40, 1)) - 0.3  # some infeasible
    epochs = rng.integers(0, 3, 40)
    perm_arrs, rnk_arrs, epc_arrs, rnk_epc = get_feasible(x, y, None, c, 4, 2, epochs)
    perm_x, perm_y, _, perm_epoch, perm, feas = perm_arrs
    assert perm_x.shape[0] == (c > 0).all(axis=1).sum()
    uniq_rank, rank_idx, rnk_cnt = rnk_arrs
    assert sum(rnk_cnt) == perm_x.shape[0]
    assert rnk_epc.shape == (len(uniq_rank), len(epc_arrs[0]))


def test_box_decomposition_d6_vs_mc(rng):
    """Six-objective exact HV (Lacour box decomposition) cross-checked
    against a tight FPRAS Monte Carlo estimate."""
    pts = rng.random((10, 6)) * 0.8
    ref = np.ones(6) * 1.05
    hv_box = HyperVolumeBoxDecomposition(ref).compute_hypervolume(pts)
    est = hv_fpras(pts, ref, eps=0.02, seed=3, device="cpu")
    assert hv_box > 0
    assert est == pytest.approx(hv_box, rel=0.05)


def test_igd_and_hv_indicator_values():
    """IGD = mean min-distance front->pf (0 on the pf itself, sqrt(2)*shift
    under a diagonal shift); Hypervolume indicator matches the exact value."""
    from dmosopt_amd.hv.indicators import IGD, Hypervolume

    pf = np.array([[0.0, 1.0], [0.5, 0.5], [1.0, 0.0]])
    ind = IGD(pf)
    assert ind.do(pf) == pytest.approx(0.0)
    assert ind.do(pf + 0.1) == pytest.approx(0.1 * np.sqrt(2))
    assert ind.do(pf + 0.3) == pytest.approx(0.3 * np.sqrt(2))

    hv = Hypervolume(ref_point=np.array([2.0, 2.0]))
    # staircase: 1x2 + 1.5x1.5 strip + 2x1 union = 3.25
    assert hv.do(pf) == pytest.approx(3.25)
