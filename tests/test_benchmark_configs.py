"""Scaled-down versions of the BASELINE target configurations (CPU):
config #3 (ZDT3 + AGEMOEA + HV termination), #4 (DTLZ2 multi-objective),
#5 (TNK constrained + CMA-ES + feasibility model). Mirrors the reference's
integration-test strategy (tests/test_constrained_benchmarks.py,
test_moo_benchmarks.py) at pytest-friendly budgets.
"""

import numpy as np
import pytest

import dmosopt_amd
from dmosopt_amd.benchmarks import problems as bp


def test_zdt3_agemoea_hv_termination():
    def obj_fun(pp):
        x = np.array([pp[f"x{i + 1}"] for i in range(8)])
        return bp.zdt3(x).numpy()[0]

    params = {
        "opt_id": "t_cfg3",
        "obj_fun": obj_fun,
        "problem_parameters": {},
        "space": {f"x{i + 1}": [0.0, 1.0] for i in range(8)},
        "objective_names": ["y1", "y2"],
        "population_size": 40,
        "num_generations": 30,
        "optimizer": "age",
        "termination_conditions": {"strategy": "simple", "min_generations": 5},
        "n_initial": 3,
        "n_epochs": 2,
        "random_seed": 17,
    }
    best = dmosopt_amd.run(params, verbose=False)
    bestx, besty = best
    y = np.column_stack([v for _, v in besty])
    assert y.shape[0] > 0 and np.isfinite(y).all()


def test_dtlz2_five_objectives():
    n_var, n_obj = 12, 5

    def obj_fun(pp):
        x = np.array([pp[f"x{i + 1}"] for i in range(n_var)])
        return bp.dtlz2(x, n_obj=n_obj).numpy()[0]

    params = {
        "opt_id": "t_cfg4",
        "obj_fun": obj_fun,
        "problem_parameters": {},
        "space": {f"x{i + 1}": [0.0, 1.0] for i in range(n_var)},
        "objective_names": [f"f{j}" for j in range(n_obj)],
        "population_size": 30,
        "num_generations": 10,
        "optimizer": "nsga2",
        "n_initial": 2,
        "n_epochs": 2,
        "random_seed": 19,
    }
    best = dmosopt_amd.run(params, verbose=False)
    bestx, besty = best
    y = np.column_stack([v for _, v in besty])
    assert y.shape == (y.shape[0], n_obj)
    # DTLZ2 front: sum f_i^2 ~ 1 for converged points; loosely check range
    assert np.isfinite(y).all() and (y >= 0).all()


def test_tnk_constrained_cmaes_feasibility():
    def obj_fun(pp):
        x = np.array([[pp["x1"], pp["x2"]]])
        f, c = bp.tnk(x)
        return f.numpy()[0], c.numpy()[0]

    params = {
        "opt_id": "t_cfg5",
        "obj_fun": obj_fun,
        "problem_parameters": {},
        "space": {"x1": [1e-9, np.pi], "x2": [1e-9, np.pi]},
        "objective_names": ["f1", "f2"],
        "constraint_names": ["c1", "c2"],
        "population_size": 24,
        "num_generations": 8,
        "optimizer": "cmaes",
        "feasibility_method_name": "logreg",
        "n_initial": 10,
        "n_epochs": 2,
        "random_seed": 23,
    }
    best = dmosopt_amd.run(params, verbose=False)
    bestx, besty = best
    y = np.column_stack([v for _, v in besty])
    assert y.shape[0] > 0
    # best set is feasible-filtered: verify against the true constraints
    x = np.column_stack([v for _, v in bestx])
    _, c = bp.tnk(x)
    assert (c.numpy() > -1e-6).all()


def test_problem_space_generation():
    space = bp.generate_problem_space("dtlz2", 12)
    assert len(space) == 12 and space["x1"] == [0.0, 1.0]
    meta = bp.get_problem_metadata("zdt3", 2)
    assert meta["pareto_front_type"] == "disconnected"


@pytest.mark.parametrize(
    "name,n_obj",
    [("zdt1", 2), ("zdt2", 2), ("zdt3", 2), ("zdt4", 2), ("zdt6", 2),
     ("dtlz1", 3), ("dtlz2", 3), ("dtlz3", 3), ("dtlz4", 3), ("dtlz5", 3),
     ("dtlz7", 3), ("wfg1", 3), ("wfg4", 3), ("maf1", 5), ("maf2", 5),
     ("maf4", 5), ("sphere", 2)],
)
def test_problem_zoo_shapes_and_finiteness(name, n_obj):
    """Every benchmark problem evaluates batched, with the declared number
    of objectives, finite values, and known anchor behavior where cheap."""
    from dmosopt_amd.benchmarks import problems as P

    fn = P.get_problem(name)
    rng = np.random.default_rng(0)
    d = 12
    x = rng.random((17, d))
    kwargs = {} if name.startswith("zdt") or name == "sphere" else {"n_obj": n_obj}
    y = fn(x, **kwargs) if kwargs else fn(x)
    y = y.numpy()
    assert y.shape == (17, n_obj)
    assert np.isfinite(y).all()


def test_zdt1_known_values():
    from dmosopt_amd.benchmarks.problems import zdt1

    # on the Pareto front (tail dims zero): f2 = 1 - sqrt(f1)
    x = np.zeros((3, 30))
    x[:, 0] = [0.0, 0.25, 1.0]
    y = zdt1(x).numpy()
    assert np.allclose(y[:, 0], [0.0, 0.25, 1.0])
    assert np.allclose(y[:, 1], 1.0 - np.sqrt([0.0, 0.25, 1.0]))


def test_constrained_problems_shapes():
    from dmosopt_amd.benchmarks.problems import constr, osy, srn, tnk

    rng = np.random.default_rng(1)
    for fn, d, n_c in [(tnk, 2, 2), (constr, 2, 2), (srn, 2, 2), (osy, 6, 6)]:
        x = rng.random((9, d))
        y, c = fn(x)
        assert y.shape[0] == 9 and c.shape == (9, n_c)
        assert np.isfinite(y.numpy()).all()
