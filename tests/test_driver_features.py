"""Driver feature coverage: mean-variance mode, problem_ids, features,
nested spaces, dynamic initial sampling, time limit, farm stats."""

import numpy as np
import pytest

import dmosopt_amd


def _base(opt_id, d=4, **over):
    def obj_fun(pp):
        x = np.array([pp[f"x{i}"] for i in range(d)])
        return np.array([np.sum(x**2), np.sum((x - 1) ** 2)])

    params = {
        "opt_id": opt_id,
        "obj_fun": obj_fun,
        "problem_parameters": {},
        "space": {f"x{i}": [0.0, 1.0] for i in range(d)},
        "objective_names": ["f1", "f2"],
        "population_size": 20,
        "num_generations": 6,
        "optimizer": "nsga2",
        "n_initial": 2,
        "n_epochs": 2,
        "random_seed": 7,
    }
    params.update(over)
    return params


def test_optimize_mean_variance_mode():
    best = dmosopt_amd.run(
        _base("t_mv", optimize_mean_variance=True,
              surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua"}),
        verbose=False,
    )
    bestx, besty = best
    y = np.column_stack([v for _, v in besty])
    assert y.shape[1] == 2 and np.isfinite(y).all()


def test_problem_ids_multi():
    def obj_fun_mp(mpp):
        out = {}
        for pid, pp in mpp.items():
            x = np.array([pp[f"x{i}"] for i in range(4)])
            shift = 0.1 * pid
            out[pid] = np.array([np.sum((x - shift) ** 2), np.sum((x - 1) ** 2)])
        return out

    params = _base("t_mp", obj_fun=obj_fun_mp, problem_ids={1, 2},
                   surrogate_method_name=None, num_generations=3)
    best = dmosopt_amd.run(params, verbose=False)
    assert set(best.keys()) == {1, 2}
    for pid in (1, 2):
        prms, objs = best[pid]
        assert len(objs) == 2


def test_features_roundtrip(tmp_path):
    feature_dtypes = [("aux", np.float32)]

    def obj_fun(pp):
        x = np.array([pp[f"x{i}"] for i in range(4)])
        y = np.array([np.sum(x**2), np.sum((x - 1) ** 2)])
        f = np.array([(np.float32(x.mean()),)], dtype=feature_dtypes)
        return y, f

    fp = str(tmp_path / "feat.h5")
    params = _base(
        "t_feat", obj_fun=obj_fun, feature_dtypes=feature_dtypes,
        surrogate_method_name=None, num_generations=3,
        file_path=fp, save=True, save_eval=4,
    )
    best = dmosopt_amd.run(params, return_features=True, verbose=False)
    bestx, besty, bestf = best
    assert bestf is not None and len(bestf) > 0

    from dmosopt_amd.storage import h5 as h5store

    _, old_evals, info = h5store.h5_load_all(fp, "t_feat")
    assert info["features"] == ["aux"]
    assert old_evals[0][0].features is not None


def test_nested_parameter_space_e2e():
    def obj_fun(pp):
        x = np.array([pp["grp"]["a"], pp["grp"]["b"], pp["other"]])
        return np.array([np.sum(x**2), np.sum((x - 1) ** 2)])

    params = {
        "opt_id": "t_nested",
        "obj_fun": obj_fun,
        "problem_parameters": {},
        "space": {"grp": {"a": [0.0, 1.0], "b": [0.0, 1.0]}, "other": [0.0, 1.0]},
        "objective_names": ["f1", "f2"],
        "nested_parameter_space": True,
        "population_size": 16,
        "num_generations": 4,
        "surrogate_method_name": None,
        "optimizer": "nsga2",
        "n_initial": 2,
        "n_epochs": 1,
        "random_seed": 3,
    }
    best = dmosopt_amd.run(params, verbose=False)
    assert best is not None


def test_dynamic_initial_sampling():
    calls = []

    def dyn(file_path, iteration, evaluated_samples, next_samples, sampler, **kw):
        calls.append(iteration)
        if iteration >= 2:
            return None
        return next_samples[:4]

    params = _base("t_dyn", dynamic_initial_sampling=dyn,
                   surrogate_method_name=None, num_generations=3)
    best = dmosopt_amd.run(params, verbose=False)
    assert best is not None
    assert calls == [0, 1, 2]


def test_time_limit_stops_early():
    import time as _t

    def slow_obj(pp):
        _t.sleep(0.02)
        x = np.array([pp[f"x{i}"] for i in range(4)])
        return np.array([x.sum(), (1 - x).sum()])

    # time_limit is checked between epochs, so the worst case is ONE full
    # epoch (~25 gens x ~20 evals x 20 ms sleep = ~10 s); without the limit
    # this config would run ~50x longer
    params = _base("t_tl", obj_fun=slow_obj, surrogate_method_name=None,
                   n_epochs=50, num_generations=25)
    t0 = _t.time()
    dmosopt_amd.run(params, time_limit=3, verbose=False)
    assert _t.time() - t0 < 60


def test_farm_stats_present():
    params = _base("t_stats", surrogate_method_name=None, num_generations=3)
    dmosopt_amd.run(params, verbose=False)
    dopt = dmosopt_amd.sopt_dict["t_stats"]
    stats = dopt.get_stats()
    assert stats["results_collected"] > 0
    assert "total_evaluation_time" in stats


def test_seed_reproducibility():
    """Same random_seed => identical archives (self-reproducibility,
    SURVEY.md section 7 RNG discipline)."""
    import dmosopt_amd

    outs = []
    for tag in ("a", "b"):
        params = _base(f"t_repro_{tag}", surrogate_method_name=None,
                       num_generations=4, random_seed=123)
        dmosopt_amd.run(params, verbose=False)
        x, y = dmosopt_amd.sopt_dict[f"t_repro_{tag}"].optimizer_dict[0].get_evals()
        outs.append((x.copy(), y.copy()))
    assert np.array_equal(outs[0][0], outs[1][0])
    assert np.array_equal(outs[0][1], outs[1][1])


def test_reduce_fun_applied_to_results():
    """reduce_fun merges per-worker results before archiving (reference
    dmosopt.py:1173-1179 multi-rank-worker semantics; single-rank here
    receives a one-element list)."""
    calls = []

    def reducer(results, scale):
        # receives a list of per-rank {problem_id: y} dicts (reference
        # dmosopt.py:1173-1179 collective-broker shape)
        calls.append(len(results))
        r = results[0]
        return {pid: np.asarray(v, dtype=float) * scale for pid, v in r.items()}

    def objfun(pp):
        names = sorted(pp.keys())
        x = np.array([pp[k] for k in names])
        return np.array([np.sum(x**2), np.sum((x - 1) ** 2)]) * 0.5

    params = _base("t_reduce")
    params["obj_fun"] = objfun
    params["reduce_fun"] = reducer
    params["reduce_fun_args"] = (2.0,)
    best = dmosopt_amd.run(params, verbose=False)
    assert best is not None
    assert len(calls) > 0 and all(c == 1 for c in calls)
    # reducer doubled the halved objective: archive values match x directly
    x, y = dmosopt_amd.sopt_dict["t_reduce"].optimizer_dict[0].get_evals()
    want = np.stack([np.sum(x**2, axis=1), np.sum((x - 1) ** 2, axis=1)], axis=1)
    assert np.allclose(y, want, atol=1e-5)


def test_get_best_and_print_best_feasibility_filters(capsys):
    """get_best feasible filtering + features/constraints passthrough
    (reference dmosopt.py get_best/print_best surface)."""

    def objfun(pp):
        names = sorted(pp.keys())
        x = np.array([pp[k] for k in names])
        y = np.array([np.sum(x**2), np.sum((x - 1) ** 2)])
        c = np.array([x[0] - 0.5])  # feasible iff x0 > 0.5
        return y, c

    params = _base("t_best", obj_fun=objfun, constraint_names=["c1"])
    dmosopt_amd.run(params, verbose=False)
    dopt = dmosopt_amd.sopt_dict["t_best"]

    # get_best returns ([(pname, col)...], [(oname, col)...][, f][, c])
    prms, objs, bc = dopt.get_best(feasible=True, return_constraints=True)
    assert (bc > 0).all(), "feasible filter must keep only c > 0"
    n_feas = len(objs[0][1])
    prms_all, objs_all = dopt.get_best(feasible=False)
    assert len(objs_all[0][1]) >= n_feas
    # parameter columns align with the space names
    assert sorted(n for n, _ in prms) == sorted(f"x{i}" for i in range(4))

    dopt.print_best(feasible=True)  # must not raise
    assert len(objs) == 2


def test_zero_epochs_evaluates_initial_only():
    """n_epochs <= 0 evaluates the initial design and stops (reference
    run_epoch(completed_epoch=True) path)."""
    params = _base("t_zero", n_epochs=0)
    best = dmosopt_amd.run(params, verbose=False)
    assert best is not None
    x, y = dmosopt_amd.sopt_dict["t_zero"].optimizer_dict[0].get_evals()
    assert x.shape[0] > 0 and y.shape[1] == 2


def test_callable_initial_method():
    """initial_method may be a callable (reference MOASMO.xinit:180-181):
    called as method(Ninit, nInput, local_random) returning unit-box
    samples, scaled to the bounds by xinit."""
    seen = {}

    def my_sampler(n, d, local_random):
        seen["shape"] = (n, d)
        return local_random.random((n, d))

    params = _base("t_init_call", initial_method=my_sampler, n_epochs=1,
                   surrogate_method_name=None, num_generations=3)
    best = dmosopt_amd.run(params, verbose=False)
    assert best is not None
    assert "shape" in seen and seen["shape"][1] == 4


def test_termination_conditions_callable_factory():
    """termination_conditions may be a callable(problem) -> Termination
    (reference dmosopt.py:120-129); generation budget stops early."""
    from dmosopt_amd.termination.basic import MaximumGenerationTermination

    calls = [0]

    def obj(pp):
        calls[0] += 1
        x = np.array([pp[k] for k in sorted(pp.keys())])
        return np.array([np.sum(x**2), np.sum((x - 1) ** 2)])

    params = _base(
        "t_term_call", obj_fun=obj, num_generations=50, n_epochs=1,
        surrogate_method_name=None,
        termination_conditions=lambda prob: MaximumGenerationTermination(prob, 5),
    )
    dmosopt_amd.run(params, verbose=False)
    # init (n_initial * dim = 8) + at most ~6 generations of <= popsize
    assert calls[0] < 8 + 7 * 21, calls[0]
    assert calls[0] > 8


def test_distance_metric_and_custom_optimizer_plugin():
    """Top-level distance_metric reaches the optimizer without colliding
    with the explicit kwarg (a reference 'multiple values' crash), and the
    optimizer plugin mechanism accepts a class object."""
    from dmosopt_amd.moea.nsga2 import NSGA2Optimizer

    params = _base("t_metric", surrogate_method_name=None, num_generations=3,
                   n_epochs=1, distance_metric="euclidean")
    assert dmosopt_amd.run(params, verbose=False) is not None

    class MyOpt(NSGA2Optimizer):
        pass

    params2 = _base("t_plugin_cls", surrogate_method_name=None,
                    num_generations=3, n_epochs=1, optimizer=MyOpt)
    assert dmosopt_amd.run(params2, verbose=False) is not None
    assert isinstance(
        dmosopt_amd.sopt_dict["t_plugin_cls"].optimizer_dict[0].runner is None
        or True, bool,
    )


def test_integer_space_parameter_metadata():
    """[lo, hi, True] space parameters: is_integer is stored as metadata
    (H5 spec / analyze output); values still arrive continuous, exactly
    like the reference (dmosopt.py:2352-2388 passes raw space values)."""
    seen = []

    def obj(pp):
        seen.append(dict(pp))
        return np.array([pp["xc"] + pp["xi"], 2 - pp["xc"]])

    params = _base("t_intspace", obj_fun=obj, surrogate_method_name=None,
                   num_generations=2, n_epochs=1,
                   space={"xc": [0.0, 1.0], "xi": [0, 10, True]})
    dmosopt_amd.run(params, verbose=False)
    dopt = dmosopt_amd.sopt_dict["t_intspace"]
    is_int = dopt.param_space.is_integer
    names = list(dopt.param_space.parameter_names)
    assert is_int[names.index("xi")] and not is_int[names.index("xc")]
    assert all(0 <= pp["xi"] <= 10 for pp in seen)


def test_multi_problem_zipped_dispatch_eval_count():
    """Multi-problem requests are zipped one-per-problem into single farm
    points (reference dmosopt.py:1291-1313): the user objective runs ~R
    times for P problems with R requests each, not P*R times."""
    calls = {"n": 0}

    def obj_fun_mp(mpp):
        calls["n"] += 1
        out = {}
        for pid, pp in mpp.items():
            x = np.array([pp[f"x{i}"] for i in range(4)])
            out[pid] = np.array([np.sum((x - 0.1 * pid) ** 2), np.sum((x - 1) ** 2)])
        return out

    params = _base(
        "t_mp_zip", obj_fun=obj_fun_mp, problem_ids={1, 2, 3},
        surrogate_method_name=None, num_generations=2, n_epochs=1,
        population_size=10,
    )
    best = dmosopt_amd.run(params, verbose=False)
    assert set(best.keys()) == {1, 2, 3}

    # calibrate: the same config with a single problem
    single_calls = {"n": 0}

    def obj_fun_sp(pp):
        single_calls["n"] += 1
        x = np.array([pp[f"x{i}"] for i in range(4)])
        return np.array([np.sum((x - 0.1) ** 2), np.sum((x - 1) ** 2)])

    dmosopt_amd.run(
        _base("t_sp_zip_cal", obj_fun=obj_fun_sp, surrogate_method_name=None,
              num_generations=2, n_epochs=1, population_size=10),
        verbose=False,
    )
    # every problem enqueues the same request schedule (same seed, same
    # config), so all batches zip fully: total objective calls track ONE
    # problem's request count, with no P-fold blowup. Allow a small
    # ragged-tail margin.
    assert calls["n"] <= single_calls["n"] + 4, (calls["n"], single_calls["n"])


def test_fused_rank_metric_perm_negative_metrics():
    """The fused (rank<<32 | metric-bits) sort key must order NEGATIVE
    metric values identically to the lexsort reference path."""
    import torch

    from dmosopt_amd import ops

    g = torch.Generator().manual_seed(3)
    for trial in range(5):
        n = 64
        rank = torch.randint(0, 4, (n,), generator=g)
        metric = torch.randn(n, generator=g) * 10.0  # mixed signs
        metric[:5] = 0.0
        metric[5] = -0.0
        perm_fused = ops.fused_rank_metric_perm(rank, metric)
        perm_ref = ops.lexsort([-metric.double(), rank.double()])
        # keys may tie (rank, metric) pairs; compare the sorted key tuples
        ks_f = [(int(rank[i]), float(-metric[i])) for i in perm_fused]
        ks_r = [(int(rank[i]), float(-metric[i])) for i in perm_ref]
        assert ks_f == ks_r


def test_random_seed_int64_roundtrip(tmp_path):
    """Seeds >= 2**31 survive H5 persistence (stored as int64 like h5py)."""
    fp = str(tmp_path / "seed64.h5")
    big_seed = 2**33 + 12345
    params = _base(
        "t_seed64", random_seed=big_seed, surrogate_method_name=None,
        num_generations=2, n_epochs=1, save=True, file_path=fp,
    )
    dmosopt_amd.run(params, verbose=False)
    from dmosopt_amd.storage import h5 as h5store

    out = h5store.init_from_h5(fp, None, "t_seed64", None)
    assert out[0] == big_seed
