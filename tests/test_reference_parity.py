"""Behavioral parity against the ACTUAL reference implementation.

Runs the reference's MOASMO.epoch (pure Python, no MPI needed) and our
engine.run_epoch on the SAME archive and seed, and compares the
resample-batch quality (hypervolume of the real-evaluated resamples).
Statistical parity, not bitwise (different RNG consumption orders by
design; SURVEY.md section 7 hard-part 3). Skipped when the reference
checkout is not present.
"""

import os
import sys
import warnings

import numpy as np
import pytest

REF = "/root/reference"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF, "dmosopt")),
    reason="reference checkout not available",
)


def _zdt1(x):
    f1 = x[:, 0]
    g = 1 + 9 * x[:, 1:].mean(axis=1)
    return np.column_stack([f1, g * (1 - np.sqrt(f1 / g))])


def test_epoch_resample_quality_matches_reference():
    from dmosopt_amd.core import engine
    from dmosopt_amd.hv.exact import hv_2d

    rng = np.random.default_rng(42)
    D = 30
    X = rng.random((120, D))
    Y = _zdt1(X)
    ref_pt = np.array([11.0, 11.0])

    res = engine.run_epoch(
        100, [f"x{i}" for i in range(D)], ["f1", "f2"],
        np.zeros(D), np.ones(D), 0.25, X, Y, None, pop=100,
        optimizer_name="nsga2", surrogate_method_name="gpr",
        surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua",
                                 "seed": 3},
        local_random=np.random.default_rng(4),
    )
    ours_hv = hv_2d(_zdt1(res["x_resample"]), ref_pt)

    if REF not in sys.path:
        sys.path.insert(0, REF)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        from dmosopt.MOASMO import epoch as ref_epoch

        gen = ref_epoch(
            100, [f"x{i}" for i in range(D)], ["f1", "f2"],
            np.zeros(D), np.ones(D), 0.25, X, Y, None, pop=100,
            optimizer_name="nsga2", surrogate_method_name="gpr",
            surrogate_method_kwargs={"anisotropic": False, "optimizer": "sceua"},
            local_random=np.random.default_rng(4),
        )
        try:
            next(gen)
            while True:
                gen.send(None)
        except StopIteration as si:
            ref_out = si.args[0] if si.args else None
    ref_hv = hv_2d(_zdt1(ref_out["x_resample"]), ref_pt)

    assert res["x_resample"].shape == ref_out["x_resample"].shape
    # same-seed archives, independent RNG streams: the resample fronts'
    # quality must agree closely (measured 120.02 vs 119.98; the analytic
    # ideal at this reference point is ~120.66)
    assert ours_hv == pytest.approx(ref_hv, abs=0.5)
    assert ours_hv > 118.0
