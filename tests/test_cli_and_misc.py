"""CLI, constrained sampling, and discrepancy tests."""

import os

import numpy as np
import pytest

import dmosopt_amd
from dmosopt_amd.sampling import discrepancy
from dmosopt_amd.sampling.constrained import ParamSpacePoints, safe_eval_arith


def _make_run_file(tmp_path, opt_id="t_cli"):
    fp = str(tmp_path / "cli.h5")

    def obj_fun(pp):
        x = np.array([pp[f"x{i}"] for i in range(3)])
        return np.array([np.sum(x**2), np.sum((x - 1) ** 2)])

    params = {
        "opt_id": opt_id,
        "obj_fun": obj_fun,
        "problem_parameters": {},
        "space": {f"x{i}": [0.0, 1.0] for i in range(3)},
        "objective_names": ["f1", "f2"],
        "population_size": 16,
        "num_generations": 4,
        "surrogate_method_name": None,
        "optimizer": "nsga2",
        "n_initial": 2,
        "n_epochs": 1,
        "random_seed": 5,
        "file_path": fp,
        "save": True,
        "save_eval": 5,
    }
    dmosopt_amd.run(params, verbose=False)
    return fp


def test_analyze_cli(tmp_path, capsys):
    from dmosopt_amd.cli.analyze import main

    fp = _make_run_file(tmp_path)
    rc = main(["-p", fp, "--opt-id", "t_cli", "--sort-key", "f1", "--knn", "5"])
    assert rc == 0
    out = capsys.readouterr().out
    assert "non-dominated solutions" in out


def test_train_cli(tmp_path, capsys):
    from dmosopt_amd.cli.train import main

    fp = _make_run_file(tmp_path, "t_cli2")
    rc = main(["-p", fp, "--opt-id", "t_cli2", "--surrogate-method", "gpr"])
    assert rc == 0
    assert "MAE" in capsys.readouterr().out


def test_onestep_cli(tmp_path, capsys):
    from dmosopt_amd.cli.onestep import main

    fp = _make_run_file(tmp_path, "t_cli3")
    rc = main(
        ["-p", fp, "--opt-id", "t_cli3", "--population-size", "16",
         "--num-generations", "4", "--seed", "3"]
    )
    assert rc == 0
    assert "resample candidates" in capsys.readouterr().out


# ----------------------------------------------------- constrained sampling
def test_safe_eval_arith():
    assert safe_eval_arith("0.5 * 2 + 1") == pytest.approx(2.0)
    with pytest.raises(Exception):
        safe_eval_arith("__import__('os')")


def test_constrained_sampling_dag():
    space = {
        "a": [0.0, 1.0],
        "b": {"abs": (0.0, 5.0), "lb": [("a", "* 2")], "ub": [("a", "* 2 + 1")],
              "method": ("uniform",)},
        "c": {"abs": (0.0, 10.0), "lb": [("b", "+ 0.5")], "method": ("uniform",)},
    }
    psp = ParamSpacePoints(50, space, seed=1)
    vals = psp.as_dict()
    a, b, c = vals["a"], vals["b"], vals["c"]
    assert ((b >= 2 * a - 1e-9) & (b <= 2 * a + 1 + 1e-9)).all()
    assert (c >= b + 0.5 - 1e-9).all() and (c <= 10.0 + 1e-9).all()


def test_constrained_circular_raises():
    space = {
        "a": {"abs": (0, 1), "lb": [("b", "+ 0")], "method": ("uniform",)},
        "b": {"abs": (0, 1), "lb": [("a", "+ 0")], "method": ("uniform",)},
    }
    with pytest.raises(ValueError):
        ParamSpacePoints(10, space, seed=1)


# ------------------------------------------------------------- discrepancy
def test_discrepancy_metrics_prefer_lh(rng):
    from dmosopt_amd import sampling as S

    mc = S.mc(64, 4, np.random.default_rng(1))
    lh = S.lh(64, 4, np.random.default_rng(1))
    assert discrepancy.CD2(lh) < discrepancy.CD2(mc)
    assert discrepancy.MD2(lh) < discrepancy.MD2(mc)
    assert discrepancy.MinDist(lh) > 0
    assert 0 <= discrepancy.corrscore(lh) <= 1
    for fn in (discrepancy.SD2, discrepancy.WD2):
        v = fn(lh)
        assert np.isfinite(v) and v >= 0


def test_early_stopping_utils():
    from dmosopt_amd.models.early_stopping import (
        AdaptiveEarlyStopping, EarlyStoppingConfig, ModelType,
        analyze_loss_trajectory, suggest_hyperparameters,
    )

    cfg = EarlyStoppingConfig.for_model_type(ModelType.DEEP_GP)
    assert cfg.min_iterations > EarlyStoppingConfig().min_iterations
    es = AdaptiveEarlyStopping(EarlyStoppingConfig(min_iterations=5, patience=3))
    losses = [1.0 / (i + 1) for i in range(4)] + [0.2] * 30
    stopped_at = None
    for i, l in enumerate(losses):
        if es.should_stop(l):
            stopped_at = i
            break
    assert stopped_at is not None and stopped_at < len(losses) - 1
    stats = analyze_loss_trajectory(losses[: stopped_at + 1])
    assert stats["n"] == stopped_at + 1
    assert "lr" in suggest_hyperparameters(losses, 0.1)


def test_normalization_roundtrip():
    """ZeroToOneNormalization forward/backward inverse; degenerate bounds
    handled (reference normalization.py semantics)."""
    from dmosopt_amd.normalization import (
        NoNormalization,
        PreNormalization,
        ZeroToOneNormalization,
    )

    rng = np.random.default_rng(0)
    xl = np.array([0.0, -2.0, 5.0])
    xu = np.array([1.0, 2.0, 5.0])  # last dim degenerate
    z = ZeroToOneNormalization(xl, xu)
    X = rng.random((20, 3)) * (xu - xl) + xl
    N = z.forward(X)
    assert N.min() >= -1e-9 and N[:, :2].max() <= 1 + 1e-9
    assert np.allclose(z.backward(N), X, atol=1e-9)

    n = NoNormalization()
    assert np.allclose(n.forward(X), X) and np.allclose(n.backward(X), X)

    p = PreNormalization(zero_to_one=True, ideal=xl, nadir=xu)
    assert p.ideal is not None


def test_bench_json_contract():
    """bench.py must emit the driver-contract JSON line: required fields,
    whole-job value, weak scaling declaration, BASELINE metric name."""
    import json
    import subprocess
    import sys as _sys

    out = subprocess.run(
        [_sys.executable, "bench.py", "--steps", "1", "--warmup", "0", "--gens", "2"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"].startswith("MO-ASMO epoch time")
    assert d["n_gpus"] == 1 and d["steps"] == 1
    assert d["higher_is_better"] is False and d["scaling"] == "weak"
    assert d["value"] == d["ms_per_step"] > 0
    cfg = d["config"]
    for key in ("model", "global_batch", "parallelism", "population_size",
                "num_generations", "archive_size", "final_hypervolume_ref11"):
        assert key in cfg, key
    assert "synthetic" in d["data"]


@pytest.mark.parametrize("mod", ["analyze", "train", "onestep"])
def test_cli_help_runs(mod):
    import subprocess
    import sys as _sys

    out = subprocess.run(
        [_sys.executable, "-m", f"dmosopt_amd.cli.{mod}", "--help"],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert out.returncode == 0
    assert "--file-path" in out.stdout or "-p" in out.stdout
