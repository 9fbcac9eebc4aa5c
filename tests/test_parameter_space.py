import numpy as np

from dmosopt_amd.datatypes import ParameterSpace, update_nested_dict


def test_flat_space_roundtrip():
    space = {"a": [0.0, 1.0], "b": [2.0, 5.0, True], "c": [-1.0, 1.0]}
    ps = ParameterSpace.from_dict(space)
    assert ps.parameter_names == ["a", "b", "c"]
    assert np.allclose(ps.bound1, [0.0, 2.0, -1.0])
    assert np.allclose(ps.bound2, [1.0, 5.0, 1.0])
    assert list(ps.is_integer) == [False, True, False]
    vals = np.array([0.3, 3.0, 0.5])
    d = ps.unflatten(vals)
    assert d == {"a": 0.3, "b": 3.0, "c": 0.5}
    assert np.allclose(ps.flatten(d), vals)


def test_nested_space_roundtrip():
    space = {
        "soma": {"gk": [0.001, 0.1], "gna": [0.01, 0.2]},
        "axon": {"gx": [0.5, 1.5]},
    }
    ps = ParameterSpace.from_dict(space)
    assert ps.parameter_names == ["axon.gx", "soma.gk", "soma.gna"]
    vals = np.array([1.0, 0.05, 0.1])
    nested = ps.unflatten(vals)
    assert nested["axon"]["gx"] == 1.0
    assert nested["soma"]["gk"] == 0.05
    assert np.allclose(ps.flatten(nested), vals)


def test_inverted_bounds_swap():
    ps = ParameterSpace.from_dict({"a": [5.0, 1.0]})
    assert ps.bound1[0] == 1.0 and ps.bound2[0] == 5.0


def test_value_only_space():
    ps = ParameterSpace.from_dict({"beta": 0.44, "n": 3}, is_value_only=True)
    assert ps.is_value_space
    vals = ps.parameter_values
    assert np.allclose(sorted(vals), [0.44, 3.0])


def test_update_nested_dict():
    base = {"a": {"b": 1, "c": 2}, "d": 3}
    upd = {"a": {"c": 5}, "e": 6}
    out = update_nested_dict(base, upd)
    assert out == {"a": {"b": 1, "c": 5}, "d": 3, "e": 6}
