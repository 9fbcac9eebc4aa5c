"""Direct unit tests for the termination criteria (reference semantics:
termination.py / adaptive_termination.py)."""

import numpy as np
import pytest

from dmosopt_amd.datatypes import OptHistory
from dmosopt_amd.termination import (
    AdaptiveWindowTermination,
    ConstraintViolationToleranceTermination,
    MaximumGenerationTermination,
    MultiObjectiveToleranceTermination,
    MultiScaleStagnationTermination,
    ParameterToleranceTermination,
    PerObjectiveConvergence,
    ResourceAwareTermination,
    TerminationCollection,
    create_adaptive_termination,
)


class _Prob:
    def __init__(self, n_objectives=2, dim=3):
        self.n_objectives = n_objectives
        self.lb = np.zeros(dim)
        self.ub = np.ones(dim)
        self.logger = None


def _hist(gen, X, F, c=None):
    return OptHistory(gen, gen * 10, X, F, c)


def test_maximum_generation():
    t = MaximumGenerationTermination(_Prob(), 5)
    X = np.random.default_rng(0).random((4, 3))
    F = np.random.default_rng(1).random((4, 2))
    assert not t.has_terminated(_hist(5, X, F))
    assert t.has_terminated(_hist(6, X, F))
    # None cap never stops
    t2 = MaximumGenerationTermination(_Prob(), None)
    assert not t2.has_terminated(_hist(10**6, X, F))


def test_force_termination_short_circuits():
    t = MaximumGenerationTermination(_Prob(), 100)
    t.force_termination = True
    assert t.has_terminated(_hist(1, np.zeros((2, 3)), np.zeros((2, 2))))


def test_parameter_tolerance_stops_on_frozen_population():
    prob = _Prob()
    t = ParameterToleranceTermination(prob, n_last=3, tol=1e-6)
    rng = np.random.default_rng(2)
    X = rng.random((8, 3))
    F = rng.random((8, 2))
    stopped_at = None
    for g in range(1, 20):
        if t.has_terminated(_hist(g, X, F)):  # identical X every generation
            stopped_at = g
            break
    # needs 2 captures to produce signal 1, then 3 signals -> gen 4
    assert stopped_at == 4


def test_parameter_tolerance_keeps_going_while_moving():
    prob = _Prob()
    t = ParameterToleranceTermination(prob, n_last=3, tol=1e-6)
    rng = np.random.default_rng(3)
    for g in range(1, 15):
        X = rng.random((8, 3))  # fresh population every generation
        assert not t.has_terminated(_hist(g, X, rng.random((8, 2))))


def test_multiobjective_tolerance_stop_and_go():
    prob = _Prob()
    rng = np.random.default_rng(4)
    F = rng.random((16, 2))
    X = rng.random((16, 3))
    t = MultiObjectiveToleranceTermination(prob, tol=0.01, n_last=4)
    hit = [g for g in range(1, 20) if t.has_terminated(_hist(g, X, F))]
    assert hit and hit[0] == 5  # 2 captures + 4 signals
    # a front that keeps improving does not stop
    t2 = MultiObjectiveToleranceTermination(prob, tol=1e-8, n_last=4)
    for g in range(1, 20):
        F = F * 0.9  # ideal point keeps moving
        assert not t2.has_terminated(_hist(g, X, F))


def test_constraint_violation_window():
    prob = _Prob()
    X = np.zeros((4, 3))
    F = np.zeros((4, 2))
    # all-feasible window stops
    t = ConstraintViolationToleranceTermination(prob, n_last=3)
    hit = [g for g in range(1, 10) if t.has_terminated(_hist(g, X, F, c=1.0))]
    assert hit and hit[0] == 4
    # stuck infeasible (violation not changing) stops too
    t2 = ConstraintViolationToleranceTermination(prob, n_last=3, tol=1e-6)
    hit2 = [g for g in range(1, 10) if t2.has_terminated(_hist(g, X, F, c=-5.0))]
    assert hit2 and hit2[0] == 4
    # violation still changing -> keep optimizing
    t3 = ConstraintViolationToleranceTermination(prob, n_last=3, tol=1e-6)
    for g in range(1, 10):
        assert not t3.has_terminated(_hist(g, X, F, c=-5.0 + 0.1 * g))


def test_per_objective_convergence_vectorized_state():
    prob = _Prob(n_objectives=2)
    t = PerObjectiveConvergence(prob, obj_tol=1e-3, n_last=4, nth_gen=1,
                                min_converged_fraction=0.8)
    rng = np.random.default_rng(5)
    X = rng.random((8, 3))
    F = rng.random((8, 2)) + 1.0
    stopped = None
    for g in range(1, 40):
        if t.has_terminated(_hist(g, X, F)):  # frozen front
            stopped = g
            break
    assert stopped is not None
    assert t._settled.all()
    # a moving front never settles
    t2 = PerObjectiveConvergence(prob, obj_tol=1e-3, n_last=4, nth_gen=1)
    F2 = rng.random((8, 2)) + 1.0
    for g in range(1, 30):
        F2 = F2 * 0.8
        assert not t2.has_terminated(_hist(g, X, F2))
    assert not t2._settled.any()


def test_multiscale_stagnation_requires_all_scales():
    prob = _Prob()
    t = MultiScaleStagnationTermination(
        prob, timescales=[2, 4, 8], stagnation_tol=1e-4, min_scales_stagnant=2,
        nth_gen=1,
    )
    rng = np.random.default_rng(6)
    X = rng.random((8, 3))
    F = rng.random((8, 2))
    hits = [g for g in range(1, 30) if t.has_terminated(_hist(g, X, F))]
    # cannot stop before the largest scale (8) has lag data
    assert hits and hits[0] >= 9


def test_adaptive_window_expands_then_stops():
    prob = _Prob()
    t = AdaptiveWindowTermination(prob, initial_window=3, max_window=10,
                                  expansion_rate=1.5, tol=1e-3)
    rng = np.random.default_rng(7)
    X = rng.random((8, 3))
    F = rng.random((8, 2)) + 1.0
    w0 = t.current_window_size
    # loud phase: keep improving -> window widens, no stop
    for g in range(1, 8):
        F = F * 0.5
        assert not t.has_terminated(_hist(g, X, F))
    assert t.current_window_size > w0
    # quiet phase: frozen front eventually stops
    stopped = False
    for g in range(8, 40):
        if t.has_terminated(_hist(g, X, F)):
            stopped = True
            break
    assert stopped


def test_resource_aware_eval_budget():
    prob = _Prob()
    t = ResourceAwareTermination(prob, max_function_evals=35)
    X, F = np.zeros((2, 3)), np.zeros((2, 2))
    assert not t.has_terminated(_hist(3, X, F))  # n_eval = 30
    assert t.has_terminated(_hist(4, X, F))  # n_eval = 40 > 35


def test_collection_any_of():
    prob = _Prob()
    t = TerminationCollection(
        prob,
        MaximumGenerationTermination(prob, 100),
        MaximumGenerationTermination(prob, 3),
    )
    X, F = np.zeros((2, 3)), np.zeros((2, 2))
    assert not t.has_terminated(_hist(3, X, F))
    assert t.has_terminated(_hist(4, X, F))


@pytest.mark.parametrize("strategy", ["comprehensive", "fast", "conservative", "simple"])
def test_presets_construct_and_run(strategy):
    prob = _Prob()
    t = create_adaptive_termination(prob, n_max_gen=50, strategy=strategy,
                                    min_generations=2)
    rng = np.random.default_rng(8)
    X = rng.random((8, 3))
    F = rng.random((8, 2))
    stopped = False
    for g in range(1, 80):
        if t.has_terminated(_hist(g, X, F)):
            stopped = True
            break
    assert stopped  # frozen run must stop at or before the generation cap


def test_preset_unknown_strategy_raises():
    with pytest.raises(ValueError):
        create_adaptive_termination(_Prob(), strategy="nope")
