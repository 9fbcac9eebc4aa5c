"""Configuration-matrix shakeout: run() across optimizer x surrogate x
constraint combinations (miniature budgets). Guards the glue paths the
focused tests don't cross (reference exercises these through its
integration scripts, tests/test_zdt1_*.py etc.)."""

import numpy as np
import pytest

import dmosopt_amd


def _sphere2(pp):
    names = sorted(pp.keys())
    x = np.array([pp[k] for k in names])
    return np.array([np.sum(x**2), np.sum((x - 1.0) ** 2)])


def _constrained(pp):
    names = sorted(pp.keys())
    x = np.array([pp[k] for k in names])
    y = np.array([np.sum(x**2), np.sum((x - 1.0) ** 2)])
    c = np.array([x[0] + x[1] - 0.2])  # feasible when sum > 0.2
    return y, c


def _params(opt_id, **over):
    p = {
        "opt_id": opt_id,
        "obj_fun": _sphere2,
        "problem_parameters": {},
        "space": {f"x{i}": [0.0, 1.0] for i in range(4)},
        "objective_names": ["f1", "f2"],
        "population_size": 16,
        "num_generations": 4,
        "initial_maxiter": 2,
        "n_initial": 2,
        "n_epochs": 2,
        "surrogate_method_name": "gpr",
        "surrogate_method_kwargs": {"anisotropic": False, "optimizer": "sceua"},
        "random_seed": 5,
    }
    p.update(over)
    return p


@pytest.mark.parametrize("optimizer", ["nsga2", "age", "smpso", "cmaes", "trs"])
def test_every_optimizer_with_gpr_surrogate(optimizer):
    best = dmosopt_amd.run(_params(f"m_{optimizer}", optimizer=optimizer),
                           verbose=False)
    assert best is not None
    _, by = best
    y = np.column_stack([v for _, v in by])
    assert np.isfinite(y).all()


@pytest.mark.parametrize(
    "surrogate",
    ["egp", "vgp", "megp", "svgp", "spv", "siv", "crv", "mdgp", "mdspp", None],
)
def test_surrogate_variants_end_to_end(surrogate):
    over = {"surrogate_method_name": surrogate}
    if surrogate == "vgp":
        over["surrogate_method_kwargs"] = {"n_iter": 25, "num_inducing": 16}
    elif surrogate is not None:
        over["surrogate_method_kwargs"] = {"n_iter": 25}
    best = dmosopt_amd.run(_params(f"m_s_{surrogate}", **over), verbose=False)
    assert best is not None


def test_constrained_with_feasibility_model():
    p = _params(
        "m_constr",
        obj_fun=_constrained,
        constraint_names=["c1"],
        feasibility_method_name="logreg",
        num_generations=5,
    )
    best = dmosopt_amd.run(p, verbose=False)
    assert best is not None
    _, by = best
    y = np.column_stack([v for _, v in by])
    assert np.isfinite(y).all()


@pytest.mark.parametrize("optimizer", ["age", "cmaes", "smpso", "trs"])
def test_feasibility_model_with_every_optimizer(optimizer):
    """The logreg feasibility model's rank() feeds each optimizer's
    x-distance path without error."""
    p = _params(
        f"m_feas_{optimizer}",
        obj_fun=_constrained,
        constraint_names=["c1"],
        feasibility_method_name="logreg",
        optimizer=optimizer,
        num_generations=3,
    )
    assert dmosopt_amd.run(p, verbose=False) is not None


@pytest.mark.parametrize("method", ["glp", "slh", "lh", "mc", "sobol"])
def test_every_initial_method_through_run(method):
    p = _params(f"m_init_{method}", surrogate_method_name=None,
                initial_method=method, num_generations=2)
    assert dmosopt_amd.run(p, verbose=False) is not None


@pytest.mark.parametrize("sa", ["dgsm", "fast"])
def test_sensitivity_methods_through_run(sa):
    """SA on the surrogate sets per-dimension di vectors for the inner
    optimizer (reference MOASMO.py:329-361 wiring)."""
    p = _params(f"m_sa_{sa}", sensitivity_method_name=sa, num_generations=3)
    assert dmosopt_amd.run(p, verbose=False) is not None


def test_joint_transformer_custom_training_through_run():
    """surrogate_custom_training='...transformer.joint' wires the
    FT-Transformer in as objective + feasibility + sensitivity provider
    (reference model_transformer.py:1112-1233 / MOASMO.py:267-295)."""
    p = _params(
        "m_joint",
        obj_fun=_constrained,
        constraint_names=["c1"],
        surrogate_method_name=None,
        surrogate_custom_training="dmosopt_amd.models.transformer.joint",
        surrogate_custom_training_kwargs={"epochs": 30},
        num_generations=3,
    )
    assert dmosopt_amd.run(p, verbose=False) is not None
