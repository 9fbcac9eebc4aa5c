"""Transformer surrogate tests (role of reference tests/test_model_transformer.py)."""

import numpy as np
import pytest

from dmosopt_amd.config import import_object_by_path
from dmosopt_amd.models.transformer import JointFTTransformer, joint


def test_import_by_path():
    fn = import_object_by_path("dmosopt_amd.models.transformer.joint")
    assert fn is joint


def test_fit_and_predict_objectives():
    rng = np.random.default_rng(0)
    X = rng.random((150, 5)).astype(np.float32)
    Y = np.column_stack([X[:, 0] * 2, X[:, 1] + X[:, 2]]).astype(np.float32)
    m = JointFTTransformer(5, 0, 2, mode="o", xlb=np.zeros(5), xub=np.ones(5),
                           n_blocks=2, seed=0, device="cpu")
    m.fit(X, Y, epochs=150, verbose=0)
    pred = m.predict_objectives(X)
    mae = np.mean(np.abs(pred - Y))
    assert mae < 0.25


def test_joint_end_to_end():
    rng = np.random.default_rng(1)
    X = rng.random((80, 4))
    Y = np.column_stack([X.sum(axis=1), (X**2).sum(axis=1)])
    C = np.column_stack([X[:, 0] - 0.3])  # feasible when x0 > 0.3

    class FakeOpt:
        pass

    opt_cls, obj_model, feas_model, sens_model = joint(
        FakeOpt, X, Y, C, np.zeros(4), np.ones(4), None, {},
        constraints=True, epochs=60,
    )
    assert opt_cls is FakeOpt
    mean = obj_model.evaluate(X[:10])
    assert mean.shape == (10, 2)
    ranks = feas_model.rank(X[:10])
    assert ranks.shape == (10,)
    assert (ranks >= 0).all() and (ranks <= 1).all()
    di = sens_model.di_dict()
    assert di["di_mutation"].shape == (4,)
    assert (di["di_mutation"] >= 1).all()


def test_sensitivity_identifies_active_dims():
    rng = np.random.default_rng(2)
    X = rng.random((200, 6)).astype(np.float32)
    Y = (5.0 * X[:, :1]).astype(np.float32)  # only dim 0 matters
    m = JointFTTransformer(6, 0, 1, mode="o", xlb=np.zeros(6), xub=np.ones(6),
                           n_blocks=2, seed=1, device="cpu")
    m.fit(X, Y, epochs=200)
    pts = rng.random((256, 6))
    sens = m.sensitivity(pts)["objectives"]
    assert np.argmax(sens) == 0
