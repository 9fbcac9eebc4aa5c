"""Variational / deep GP surrogate variants: fit quality + registry resolution."""

import numpy as np
import pytest

from dmosopt_amd import config as cfg


def _data(seed=0, n=150, d=5):
    rng = np.random.default_rng(seed)
    X = rng.random((n, d))
    Y = np.column_stack([np.sin(3 * X[:, 0]), X[:, 1] ** 2])
    return X, Y


@pytest.mark.parametrize("name", ["vgp", "svgp", "spv", "siv", "crv"])
def test_variational_registry_fit(name):
    X, Y = _data()
    cls = cfg.resolve(cfg.surrogate_registry, name)
    m = cls(X, Y, 5, 2, np.zeros(5), np.ones(5), seed=1, n_iter=150)
    Xq, Yq = _data(seed=3, n=40)
    mean, var = m.predict(Xq)
    assert mean.shape == (40, 2) and var.shape == (40, 2)
    assert (var >= 0).all()
    rmse = np.sqrt(((mean - Yq) ** 2).mean())
    assert rmse < 0.5  # loose: short training budget


@pytest.mark.parametrize("name", ["mdgp", "mdspp"])
def test_deep_gp_registry_fit(name):
    X, Y = _data(seed=2)
    cls = cfg.resolve(cfg.surrogate_registry, name)
    m = cls(X, Y, 5, 2, np.zeros(5), np.ones(5), seed=1, n_iter=150)
    Xq, Yq = _data(seed=4, n=30)
    mean, var = m.predict(Xq)
    assert mean.shape == (30, 2) and (var >= 0).all()
    assert np.isfinite(mean).all()


def test_egp_and_megp_registry():
    X, Y = _data(seed=5, n=80)
    for name in ["egp", "megp"]:
        cls = cfg.resolve(cfg.surrogate_registry, name)
        m = cls(X, Y, 5, 2, np.zeros(5), np.ones(5), seed=1, adam_iters=60)
        mean, var = m.predict(X[:10])
        assert mean.shape == (10, 2) and (var >= 0).all()


def test_mean_variance_mode():
    X, Y = _data(seed=6, n=60)
    cls = cfg.resolve(cfg.surrogate_registry, "gpr")
    m = cls(X, Y, 5, 2, np.zeros(5), np.ones(5), seed=1, return_mean_variance=True)
    out = m.evaluate(X[:5])
    assert isinstance(out, tuple) and len(out) == 2


def test_feasibility_model_discriminates():
    """logreg feasibility: P(feasible) separates clearly on a linearly
    separable constraint (reference feasibility.py:14-67 pipeline role)."""
    from dmosopt_amd.models.feasibility import LogisticFeasibilityModel

    rng = np.random.default_rng(0)
    X = rng.random((120, 4))
    C = np.column_stack([X[:, 0] - 0.5])
    fsbm = LogisticFeasibilityModel(X, C)
    Xhi = rng.random((20, 4)); Xhi[:, 0] = 0.9
    Xlo = rng.random((20, 4)); Xlo[:, 0] = 0.1
    assert float(np.mean(fsbm.rank(Xhi))) > float(np.mean(fsbm.rank(Xlo))) + 0.5


def test_sensitivity_methods_identify_influential_dims():
    """DGSM and FAST attribute sensitivity to the right inputs on a
    separable model (reference sa.py role: S1 per objective)."""
    from dmosopt_amd.models.sa import SA_DGSM, SA_FAST

    class M:
        def evaluate(self, x):
            x = np.atleast_2d(x)
            return np.column_stack([3.0 * x[:, 0], 2.0 * x[:, 1]])

    for cls in (SA_DGSM, SA_FAST):
        sa = cls(np.zeros(4), np.ones(4), [f"x{i}" for i in range(4)], ["f1", "f2"])
        S1 = sa.analyze(M(), 2048)["S1"]
        assert np.argmax(S1["f1"]) == 0 and np.argmax(S1["f2"]) == 1, (cls, S1)
