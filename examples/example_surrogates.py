"""Surrogate-variant tour: fit every registered surrogate family on the
same archive and compare hold-out RMSE.

Families (registry names, reference backend in parentheses):
  gpr   exact GP, Matern-5/2, SCE-UA MLL search   (sklearn)
  egp   exact GP, ARD, Adam MLL                   (GPyTorch)
  megp  multitask ICM GP, kron(B, Kx)             (GPyTorch multitask)
  mdgp  2-layer deep GP, DSVI                     (GPyTorch deep GP)
  mdspp sigma-point deep GP                       (GPyTorch DSPP)
  vgp/svgp/spv/siv/crv  variational GPs           (GPflow)

Usage: python examples/example_surrogates.py
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dmosopt_amd.benchmarks.problems import zdt1
from dmosopt_amd.config import surrogate_registry, resolve

D, N_TRAIN, N_TEST = 6, 120, 64


def main():
    rng = np.random.default_rng(0)
    X = rng.random((N_TRAIN, D))
    Y = zdt1(X).numpy()
    Xq = rng.random((N_TEST, D))
    Yq = zdt1(Xq).numpy()

    for name in ["gpr", "egp", "megp", "vgp", "svgp", "crv", "mdgp"]:
        cls = resolve(surrogate_registry, name)
        model = cls(X, Y, D, 2, np.zeros(D), np.ones(D), seed=1)
        pred = model.evaluate(Xq)
        pred = pred[0] if isinstance(pred, tuple) else pred
        rmse = float(np.sqrt(np.mean((pred - Yq) ** 2)))
        print(f"{name:6s} holdout rmse = {rmse:.4f}")


if __name__ == "__main__":
    main()
