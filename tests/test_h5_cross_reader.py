"""Cross-reader proof for the own-C++ HDF5 writer (storage/h5cpp).

The reference stack reads/writes these files with h5py; h5py is not
installed in this image, so the independent reader here is the OFFICIAL
HDF5 CLI tooling shipped with libhdf5 (/opt/conda/bin/h5dump, h5ls): a
completely separate code path from our _h5core writer/reader. The tests
assert (a) the committed datatype layout matches the reference's schema
byte-for-byte (enum basetype, compound field names/sizes/offsets,
parameter_paths fixed-depth strings — reference dmosopt.py:1499-1789) and
(b) an independent tool extracts the same VALUES our own reader returns.
"""

import json
import os
import re
import subprocess

import numpy as np
import pytest

import dmosopt_amd

H5DUMP = "/opt/conda/bin/h5dump"
H5LS = "/opt/conda/bin/h5ls"

pytestmark = pytest.mark.skipif(
    not os.path.exists(H5DUMP), reason="HDF5 CLI tools not available"
)


def _make_file(tmp_path, opt_id="xh5", nested=False, constraints=False, seed=321):
    fp = str(tmp_path / f"{opt_id}.h5")

    if constraints:
        def obj(pp):
            x = np.array([pp["a"], pp["b"]])
            return (
                np.array([float(np.sum(x**2)), float(np.sum((x - 1) ** 2))]),
                np.array([x[0] - 0.1]),
            )

        space = {"a": [0.0, 1.0], "b": [0.0, 1.0]}
    elif nested:
        def obj(pp):
            x = np.array([pp["grp"]["a"], pp["grp"]["deep"]["b"]])
            return np.array([float(np.sum(x**2)), float(np.sum((x - 1) ** 2))])

        space = {"grp": {"a": [0.0, 1.0], "deep": {"b": [0.0, 1.0]}}}
    else:
        def obj(pp):
            x = np.array([pp["a"], pp["b"]])
            return np.array([float(np.sum(x**2)), float(np.sum((x - 1) ** 2))])

        space = {"a": [0.0, 1.0], "b": [0.0, 1.0]}

    params = {
        "opt_id": opt_id,
        "obj_fun": obj,
        "problem_parameters": {},
        "space": space,
        "objective_names": ["f1", "f2"],
        "population_size": 8,
        "num_generations": 2,
        "surrogate_method_name": None,
        "optimizer": "nsga2",
        "n_initial": 2,
        "n_epochs": 1,
        "random_seed": seed,
        "save": True,
        "file_path": fp,
        "nested_parameter_space": nested,
    }
    if constraints:
        params["constraint_names"] = ["c1"]
    best = dmosopt_amd.run(params, verbose=False)
    return fp, best


def _dump(fp, *args):
    out = subprocess.run(
        [H5DUMP, *args, fp], capture_output=True, text=True, timeout=120
    )
    assert out.returncode == 0, out.stderr
    return out.stdout


def test_h5dump_opens_and_lists_schema(tmp_path):
    fp, _ = _make_file(tmp_path)
    listing = subprocess.run(
        [H5LS, "-r", fp], capture_output=True, text=True, timeout=120
    )
    assert listing.returncode == 0, listing.stderr
    for node in (
        "/xh5/objective_enum", "/xh5/objective_spec", "/xh5/parameter_enum",
        "/xh5/parameter_spec", "/xh5/random_seed", "/xh5/problem_ids",
        "/xh5/0/epochs", "/xh5/0/parameters", "/xh5/0/objectives",
        "/xh5/0/predictions",
    ):
        assert node in listing.stdout, (node, listing.stdout)


def test_committed_types_match_reference_layout(tmp_path):
    """Byte-level dtype assertions against the reference's h5py-produced
    layout (dmosopt.py:1613-1616: objectives/parameters are float32 compound
    with named fields; enums are uint16-based)."""
    fp, _ = _make_file(tmp_path)
    hdr = _dump(fp, "-H")
    # enum with uint16 base and the objective names as members
    assert re.search(r'DATATYPE "objective_enum"\s+H5T_ENUM\s*{\s*H5T_STD_U16LE', hdr)
    assert '"f1"' in hdr and '"f2"' in hdr
    # objectives are stored through a COMMITTED compound type of IEEE
    # float32 fields named f1, f2 (the reference commits the same named
    # types: opt_grp["objective_type"] = dt)
    m = re.search(
        r'DATATYPE "objective_type"\s+H5T_COMPOUND\s*{(.*?)}', hdr, re.S
    )
    assert m, hdr
    block = m.group(1)
    assert 'H5T_IEEE_F32LE "f1"' in block and 'H5T_IEEE_F32LE "f2"' in block
    assert re.search(r'DATASET "objectives" {\s*DATATYPE\s+"/xh5/objective_type"', hdr)
    # parameters: committed float32 compound with the parameter names
    m = re.search(
        r'DATATYPE "parameter_space_type"\s+H5T_COMPOUND\s*{(.*?)}', hdr, re.S
    )
    assert m and 'H5T_IEEE_F32LE "a"' in m.group(1)
    assert re.search(
        r'DATASET "parameters" {\s*DATATYPE\s+"/xh5/parameter_space_type"', hdr
    )
    # random_seed persisted as int64 (h5py stores the full python int)
    assert re.search(r'DATASET "random_seed" {\s*DATATYPE\s+H5T_STD_I64LE', hdr)
    # epochs as unsigned 32-bit like the reference's uint32 epochs column
    assert re.search(r'DATASET "epochs" {\s*DATATYPE\s+H5T_STD_U32LE', hdr)


def test_parameter_paths_fixed_depth_strings(tmp_path):
    """Nested spaces write the reference's parameter_paths table:
    (parameter enum, path_length i32, components S128 x 10)
    (dmosopt.py:1499-1523)."""
    fp, _ = _make_file(tmp_path, opt_id="xh5n", nested=True)
    hdr = _dump(fp, "-H")
    m = re.search(
        r'DATATYPE "parameter_path_type"\s+H5T_COMPOUND\s*{(.*?)\n      }',
        hdr, re.S,
    )
    assert m, hdr
    block = m.group(1)
    assert re.search(
        r'DATASET "parameter_paths" {\s*DATATYPE\s+"/xh5n/parameter_path_type"', hdr
    )
    assert '"path_length"' in block and "H5T_STD_I32LE" in block
    assert '"components"' in block
    assert re.search(r"H5T_ARRAY\s*{\s*\[10\]", block)
    assert re.search(r"STRSIZE\s+128", block)
    # and the path values themselves round-trip through h5dump
    data = _dump(fp, "-d", "/xh5n/parameter_paths")
    # component strings are NULLPAD S128: "grp\000..." etc.
    assert "grp" in data and "deep" in data


def test_values_via_independent_reader_match(tmp_path):
    """h5dump-extracted objective values equal what our own reader
    returns: the file is not merely structurally valid, an independent
    implementation decodes identical numbers."""
    fp, _ = _make_file(tmp_path, opt_id="xh5v", seed=777)
    from dmosopt_amd.storage import h5 as h5store

    out = h5store.init_from_h5(fp, None, "xh5v", None)
    old_evals = out[2]
    ours = np.array(
        [e.objectives for e in old_evals[0]], dtype=np.float32
    )  # (n, 2)
    # %.9g: 9 significant digits round-trip any float32 exactly
    dump = _dump(fp, "-d", "/xh5v/0/objectives", "-O", "-m", "%.9g")
    # h5dump compound rows print as "(i): {\n v1,\n v2\n }"
    rows = re.findall(r"\(\d+\):\s*{\s*([-\d.e+]+),\s*([-\d.e+]+)\s*}", dump, re.S)
    theirs = np.array([[float(a), float(b)] for a, b in rows], dtype=np.float32)
    assert theirs.shape == ours.shape
    np.testing.assert_array_equal(theirs, ours)


def test_constraints_schema(tmp_path):
    fp, _ = _make_file(tmp_path, opt_id="xh5c", constraints=True)
    hdr = _dump(fp, "-H")
    assert re.search(r'DATATYPE "constraint_enum"\s+H5T_ENUM\s*{\s*H5T_STD_U16LE', hdr)
    m = re.search(
        r'DATATYPE "constraint_type"\s+H5T_COMPOUND\s*{(.*?)}', hdr, re.S
    )
    assert m and 'H5T_IEEE_F32LE "c1"' in m.group(1)
    assert re.search(r'DATASET "constraints" {\s*DATATYPE\s+"/xh5c/constraint_type"', hdr)
