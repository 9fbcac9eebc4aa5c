"""Focused MOEA-internals unit tests (beyond the end-to-end runs):
CMA-ES mid-front HVI fill, the batched rank-1 Cholesky update recurrence,
TRS trust-region state transitions, SMPSO constriction."""

import numpy as np
import pytest
import torch

from dmosopt_amd.moea.cmaes import CMAESOptimizer, batched_cholesky_update
from dmosopt_amd.moea.trs import TrState, TRSOptimizer
from dmosopt_amd.moea.smpso import SMPSOOptimizer


def test_batched_cholesky_update_tracks_covariance():
    """A A^T after the rank-1 update equals (1-ccov) C + ccov pc pc^T when
    psucc < pthresh (reference CMAES.py:489-537 recurrence), and Ainv stays
    the inverse of A."""
    torch.manual_seed(0)
    K, d = 5, 4
    M = torch.randn(K, d, d, dtype=torch.float64) * 0.2
    C0 = M @ M.transpose(-1, -2) + torch.eye(d, dtype=torch.float64)
    A = torch.linalg.cholesky(C0)
    Ainv = torch.linalg.inv(A)
    pc = torch.randn(K, d, dtype=torch.float64)
    z = torch.randn(K, d, dtype=torch.float64)
    psucc = torch.zeros(K, dtype=torch.float64)  # below pthresh
    cc, ccov, pthresh = 0.2, 0.3, 0.44

    A2, Ainv2, pc2 = batched_cholesky_update(A, Ainv, pc, z, psucc, cc, ccov, pthresh)
    pc_want = (1 - cc) * pc + np.sqrt(cc * (2 - cc)) * z
    assert torch.allclose(pc2, pc_want, atol=1e-10)
    C_want = (1 - ccov) * C0 + ccov * pc_want[:, :, None] @ pc_want[:, None, :]
    assert torch.allclose(A2 @ A2.transpose(-1, -2), C_want, atol=1e-8)
    eye = torch.eye(d, dtype=torch.float64).expand(K, d, d)
    assert torch.allclose(Ainv2 @ A2, eye, atol=1e-6)


def test_cmaes_select_mid_front_fill():
    """_select keeps whole better fronts and fills the mid front by HVI."""
    opt = CMAESOptimizer(popsize=4, nInput=2, nOutput=2)
    opt.bounds = torch.tensor([[0.0, 1.0], [0.0, 1.0]], dtype=torch.float64).T.T
    # front 0: 2 points; front 1: 4 points -> need 2 of them
    y = torch.tensor(
        [[0.1, 0.9], [0.9, 0.1],                       # front 0
         [0.5, 1.5], [1.5, 0.5], [1.0, 1.0], [1.4, 1.4]],  # deeper
        dtype=torch.float64,
    )
    x = torch.rand(6, 2, dtype=torch.float64)
    chosen, not_chosen, rank = opt._select(x, y)
    assert chosen.sum() == 4
    assert chosen[0] and chosen[1]
    assert not chosen[5] or not_chosen[5] is not None  # worst point competes


def test_trs_state_transitions():
    """Success-window mean drives expand/shrink and restart at length_min
    (reference TRS.py:281-305), exercised through a real optimizer loop."""
    rng = np.random.default_rng(0)
    d = 6
    opt = TRSOptimizer(popsize=8, nInput=d, nOutput=2)
    bounds = np.column_stack([np.zeros(d), np.ones(d)])
    x = rng.random((8, d))
    y = np.column_stack([x.sum(1), (1 - x).sum(1)])
    opt.initialize_strategy(x, y, bounds, rng)
    L0 = opt.state.tr.length

    # all-improving children: success fraction 1 -> trust region expands
    x_gen, gs = opt.generate()
    xg = x_gen.cpu().numpy() if isinstance(x_gen, torch.Tensor) else x_gen
    y_gen = np.column_stack([xg.sum(1) - 100.0, (1 - xg).sum(1) - 100.0])
    opt.update(xg, y_gen, gs)
    assert opt.state.tr.length > L0, "high success must expand the trust region"

    # all-worse children repeatedly -> shrink and eventually restart-flag
    shrunk = False
    for _ in range(80):
        x_gen, gs = opt.generate()
        xg = x_gen.cpu().numpy() if isinstance(x_gen, torch.Tensor) else x_gen
        y_gen = np.column_stack([xg.sum(1) + 100.0, (1 - xg).sum(1) + 100.0])
        L_before = opt.state.tr.length
        opt.update(xg, y_gen, gs)
        if opt.state.tr.length < L_before:
            shrunk = True
        if opt.state.tr.restart:
            break
    assert shrunk, "sustained failure must shrink the trust region"


def test_smpso_velocity_constriction_bounded():
    """One SMPSO epoch keeps positions within bounds and produces finite
    velocities (constriction factor from reference SMPSO.py:316-348)."""
    rng = np.random.default_rng(0)
    d = 5
    opt = SMPSOOptimizer(popsize=8, nInput=d, nOutput=2, swarm_size=2)
    bounds = np.column_stack([np.zeros(d), np.ones(d)])
    x = rng.random((16, d))
    y = np.column_stack([x.sum(1), (1 - x).sum(1)])
    opt.initialize_strategy(x, y, bounds, rng)
    for _ in range(3):
        x_gen, gs = opt.generate()
        xg = x_gen if isinstance(x_gen, np.ndarray) else x_gen.cpu().numpy()
        assert np.isfinite(xg).all()
        assert xg.min() >= -1e-6 and xg.max() <= 1 + 1e-6
        y_gen = np.column_stack([xg.sum(1), (1 - xg).sum(1)])
        opt.update(xg, y_gen, gs)
        v = opt.state.velocity
        v = v.cpu().numpy() if isinstance(v, torch.Tensor) else v
        assert np.isfinite(v).all()


def test_batched_cholesky_update_high_psucc_branch():
    """psucc >= pthresh: pc decays without the z term and alpha gains the
    ccov*cc*(2-cc) compensation (reference CMAES.py:500-511); A A^T must
    still track (alpha C + ccov pc pc^T) and Ainv A = I."""
    torch.manual_seed(1)
    K, d = 3, 4
    M = torch.randn(K, d, d, dtype=torch.float64) * 0.2
    C0 = M @ M.transpose(-1, -2) + torch.eye(d, dtype=torch.float64)
    A = torch.linalg.cholesky(C0)
    Ainv = torch.linalg.inv(A)
    pc = torch.randn(K, d, dtype=torch.float64)
    z = torch.randn(K, d, dtype=torch.float64)
    psucc = torch.ones(K, dtype=torch.float64)  # above pthresh
    cc, ccov, pthresh = 0.2, 0.3, 0.44

    A2, Ainv2, pc2 = batched_cholesky_update(A, Ainv, pc, z, psucc, cc, ccov, pthresh)
    pc_want = (1 - cc) * pc
    assert torch.allclose(pc2, pc_want, atol=1e-12)
    alpha = (1 - ccov) + ccov * cc * (2 - cc)
    C_want = alpha * C0 + ccov * pc_want[:, :, None] @ pc_want[:, None, :]
    assert torch.allclose(A2 @ A2.transpose(-1, -2), C_want, atol=1e-8)
    eye = torch.eye(d, dtype=torch.float64).expand(K, d, d)
    assert torch.allclose(Ainv2 @ A2, eye, atol=1e-6)


def test_population_diversity_and_adaptive_popsize():
    """PopulationDiversity (front-0 fraction + crowding CV) and the NSGA2
    adaptive-population-size rule driven by it."""
    from dmosopt_amd.hv.indicators import PopulationDiversity
    from dmosopt_amd.moea.nsga2 import NSGA2Optimizer

    pd_ind = PopulationDiversity()
    rank = np.array([0, 0, 1, 1, 2])
    Y = np.random.default_rng(0).random((5, 2))
    diversity, cd_spread = pd_ind.do(rank, Y)
    assert diversity == pytest.approx(2 / 5)
    assert np.isfinite(cd_spread)

    rng = np.random.default_rng(1)
    opt = NSGA2Optimizer(popsize=16, nInput=4, nOutput=2,
                         adaptive_population_size=True)
    bounds = np.column_stack([np.zeros(4), np.ones(4)])
    x = rng.random((16, 4))
    y = np.column_stack([x.sum(1), (1 - x).sum(1)])
    opt.initialize_strategy(x, y, bounds, rng)
    for _ in range(4):
        x_gen, gs = opt.generate()
        xg = x_gen.cpu().numpy() if hasattr(x_gen, "cpu") else x_gen
        y_gen = np.column_stack([xg.sum(1), (1 - xg).sum(1)])
        opt.update(xg, y_gen, gs)
    p = opt.opt_params
    assert p.min_population_size <= p.popsize <= p.max_population_size
    assert p.poolsize == int(round(p.popsize / 2.0))


@pytest.mark.parametrize("name", ["nsga2", "age", "smpso", "cmaes", "trs"])
@pytest.mark.parametrize("m", [3, 5])
def test_many_objectives_duplicate_heavy(name, m):
    """Every optimizer steps with 3/5 objectives and duplicate-heavy
    populations (degenerate fronts, zero crowding spans)."""
    from dmosopt_amd.config import optimizer_registry, resolve

    rng = np.random.default_rng(0)
    cls = resolve(optimizer_registry, name)
    d, pop = 4, 12
    opt = cls(popsize=pop, nInput=d, nOutput=m)
    bounds = np.column_stack([np.zeros(d), np.ones(d)])
    x = rng.random((pop, d))
    x[: pop // 2] = x[0]
    y = np.column_stack([x.sum(1) + j * 0.1 for j in range(m)])
    opt.initialize_strategy(x, y, bounds, np.random.default_rng(1))
    for _ in range(2):
        xg, gs = opt.generate()
        xgn = xg.cpu().numpy() if hasattr(xg, "cpu") else xg
        yg = np.column_stack([xgn.sum(1) + j * 0.1 for j in range(m)])
        opt.update(xgn, yg, gs)
    px, py = opt.population_objectives
    assert np.isfinite(py.cpu().numpy() if hasattr(py, "cpu") else py).all()
