"""GP surrogate tests against the sklearn oracle and analytic identities."""

import math

import numpy as np
import pytest
import torch

from dmosopt_amd.models.gp import GPRMatern
from dmosopt_amd.models.gp_core import batched_nmll, build_kernel, matern_from_d2
from dmosopt_amd.models.sceua import sceua_batched


def _sklearn_gpr(X, y, theta):
    from sklearn.gaussian_process import GaussianProcessRegressor
    from sklearn.gaussian_process.kernels import ConstantKernel, Matern, WhiteKernel

    sf2, ell, noise = np.exp(theta)
    k = ConstantKernel(sf2, "fixed") * Matern(ell, "fixed", nu=2.5) + WhiteKernel(
        noise, "fixed"
    )
    return GaussianProcessRegressor(kernel=k, optimizer=None, normalize_y=True).fit(X, y)


def test_kernel_matches_sklearn():
    from sklearn.gaussian_process.kernels import Matern

    rng = np.random.default_rng(0)
    X = rng.random((40, 6))
    ell = 0.37
    K_skl = Matern(ell, nu=2.5)(X)
    theta = torch.tensor([[0.0, math.log(ell), math.log(1e-9)]], dtype=torch.float64)
    K_ours = build_kernel(torch.as_tensor(X), torch.as_tensor(X), theta, nu=2.5)
    assert np.allclose(K_ours[0].numpy(), K_skl, atol=1e-10)


def test_posterior_matches_sklearn_fixed_theta():
    rng = np.random.default_rng(1)
    d, N = 5, 60
    X = rng.random((N, d))
    y = np.sin(3 * X[:, 0]) + X[:, 1] ** 2
    theta = np.array([math.log(1.5), math.log(0.4), math.log(1e-6)])

    skl = _sklearn_gpr(X, y, theta)
    Xq = rng.random((30, d))
    mean_skl, std_skl = skl.predict(Xq, return_std=True)

    from dmosopt_amd.models.gp_core import FittedGP

    Xt = torch.as_tensor(X)
    Yt = torch.as_tensor(y[:, None])
    fitted = FittedGP(
        Xt,
        Yt,
        torch.as_tensor(theta[None, :]),
        Yt.mean(dim=0),
        Yt.std(dim=0, unbiased=False),
        nu=2.5,
        jitter=0.0,
    )
    mean_ours, var_ours = fitted.predict(torch.as_tensor(Xq))
    assert np.allclose(mean_ours[:, 0].numpy(), mean_skl, atol=1e-8)
    assert np.allclose(np.sqrt(var_ours[:, 0].numpy()), std_skl, atol=1e-7)


def test_nmll_matches_sklearn_lml():
    rng = np.random.default_rng(2)
    X = rng.random((50, 4))
    y = X.sum(axis=1) + 0.01 * rng.standard_normal(50)
    theta = np.array([math.log(0.8), math.log(0.6), math.log(1e-4)])
    skl = _sklearn_gpr(X, y, theta)
    lml = skl.log_marginal_likelihood()
    yn = (y - y.mean()) / y.std()
    ours = batched_nmll(
        torch.as_tensor(X), torch.as_tensor(yn), torch.as_tensor(theta[None, :]), jitter=0.0
    )
    assert ours[0].item() == pytest.approx(-lml, rel=1e-8)


def test_gpr_matern_end_to_end_accuracy():
    rng = np.random.default_rng(3)
    d, N = 8, 100
    X = rng.random((N, d))
    Y = np.column_stack([X[:, 0] ** 2, np.sin(2 * X[:, 1])])
    gp = GPRMatern(X, Y, d, 2, np.zeros(d), np.ones(d), optimizer="sceua", seed=7)
    Xq = rng.random((50, d))
    mean, var = gp.predict(Xq)
    Ytrue = np.column_stack([Xq[:, 0] ** 2, np.sin(2 * Xq[:, 1])])
    rmse = np.sqrt(((mean - Ytrue) ** 2).mean(axis=0))
    assert (rmse < 0.05).all()
    assert (var >= 0).all()


def test_gpr_adam_optimizer():
    rng = np.random.default_rng(4)
    d, N = 4, 80
    X = rng.random((N, d))
    Y = (X[:, :1] * 2.0) ** 2
    gp = GPRMatern(X, Y, d, 1, np.zeros(d), np.ones(d), optimizer="adam", seed=5)
    Xq = rng.random((40, d))
    mean, _ = gp.predict(Xq)
    rmse = float(np.sqrt(((mean[:, 0] - (Xq[:, 0] * 2.0) ** 2) ** 2).mean()))
    assert rmse < 0.2


def test_sceua_batched_minimizes_quadratic():
    def f(x, stream):
        centers = torch.tensor([0.3, 0.7], dtype=x.dtype)
        c = centers[stream]
        return ((x - c[:, None]) ** 2).sum(dim=1)

    bestx, bestf, icall = sceua_batched(
        f, np.zeros(3), np.ones(3), nopt=3, n_streams=2, seed=0, maxn=2000
    )
    assert np.allclose(bestx[0], 0.3, atol=1e-3)
    assert np.allclose(bestx[1], 0.7, atol=1e-3)
    assert (bestf < 1e-5).all()
