"""HDF5 results/checkpoint schema (reference dmosopt.py:1473-2349).

Layout under /{opt_id}: committed enum + spec datasets (objective_enum/
objective_spec, parameter_enum/parameter_spec, constraint_*, feature_*,
problem_parameters, parameter_paths), metadata, problem_ids, random_seed;
per-problem growing 1-D compound datasets epochs/parameters/objectives/
predictions[/features][/constraints]; surrogate_evals/{epochs, generations,
parameters, objectives}; optimizer_params/{epoch}; optimizer_stats/{epoch}/
stats. Objectives/parameters are float32 compound rows with named fields;
mean-variance mode doubles the prediction fields to '{name} mean' /
'{name} variance'.

Backed by the native libhdf5 extension dmosopt_amd._h5core (h5py is not a
dependency of this framework); numpy structured arrays are written with
memory layout == file layout.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from dmosopt_amd.datatypes import EvalEntry, ParameterSpace

PATH_MAX_DEPTH = 10
PATH_MAX_NAME = 128


def _h5core():
    import torch  # noqa: F401  — loads shared deps first

    from dmosopt_amd import _h5core as core

    return core


def _np_members(dtype: np.dtype, enum_fields: Optional[Dict[str, str]] = None):
    """numpy structured dtype -> commit_compound member list."""
    enum_fields = enum_fields or {}
    members = []
    for name in dtype.names:
        fdt, offset = dtype.fields[name][0], dtype.fields[name][1]
        nel = 1
        base = fdt
        if fdt.subdtype is not None:
            base, shape = fdt.subdtype
            nel = int(np.prod(shape))
        if name in enum_fields:
            code = enum_fields[name]
        elif base.kind == "S":
            code = f"S{base.itemsize}"
        elif base.kind == "b":
            code = "b1"
        else:
            code = f"{base.kind}{base.itemsize}"
        members.append((name, int(offset), code, nel))
    return members


def _desc_to_dtype(desc) -> np.dtype:
    """Reconstruct a numpy dtype from _h5core.dataset_type output."""
    kind = desc[0] if isinstance(desc, tuple) else None
    if isinstance(desc, str):
        return np.dtype(desc)
    if kind == "compound":
        _, size, members = desc
        names, formats, offsets = [], [], []
        for name, off, sub, nel in members:
            sub_dt = _desc_to_dtype(sub)
            if nel > 1:
                sub_dt = np.dtype((sub_dt, (nel,)))
            names.append(name)
            formats.append(sub_dt)
            offsets.append(off)
        return np.dtype({"names": names, "formats": formats, "offsets": offsets, "itemsize": size})
    if kind == "enum":
        _, ssize, names, values = desc
        if ssize == 1 and set(names) == {"FALSE", "TRUE"}:
            return np.dtype(np.bool_)
        return np.dtype(f"u{ssize}") if ssize in (1, 2, 4, 8) else np.dtype(np.uint16)
    if kind == "string":
        return np.dtype(f"S{desc[1]}")
    return np.dtype(np.float64)


def _enum_names(desc) -> Dict[int, str]:
    assert desc[0] == "enum"
    _, _, names, values = desc
    return {int(v): n for n, v in zip(names, values)}


def _read_structured(f, path) -> np.ndarray:
    desc = f.dataset_type(path)
    dt = _desc_to_dtype(desc)
    raw = f.read_rows(path)
    return np.frombuffer(raw, dtype=dt).copy()


def _append_structured(f, path, arr: np.ndarray, type_path: str):
    if not f.has(path):
        f.create_dataset(path, type_path, 0, -1)
    f.append_rows(path, np.ascontiguousarray(arr).tobytes(), len(arr))


# ------------------------------------------------------------------ schema
def h5_init_types(
    f,
    opt_id,
    objective_names,
    feature_dtypes,
    constraint_names,
    problem_parameters: ParameterSpace,
    parameter_space: ParameterSpace,
    surrogate_mean_variance=False,
):
    g = f"/{opt_id}"
    f.create_group(g)

    # objectives
    obj_names = list(objective_names)
    f.commit_enum(f"{g}/objective_enum", obj_names, list(range(len(obj_names))))
    spec_dt = np.dtype({"names": ["objective"], "formats": [np.uint16]})
    f.commit_compound(
        f"{g}/objective_spec_type", spec_dt.itemsize,
        [("objective", 0, f"{g}/objective_enum", 1)],
    )
    obj_dt = np.dtype({"names": obj_names, "formats": [np.float32] * len(obj_names)})
    f.commit_compound(f"{g}/objective_type", obj_dt.itemsize, _np_members(obj_dt))

    if surrogate_mean_variance:
        sm_names = [f"{n} mean" for n in obj_names] + [f"{n} variance" for n in obj_names]
    else:
        sm_names = obj_names
    sm_dt = np.dtype({"names": sm_names, "formats": [np.float32] * len(sm_names)})
    f.commit_compound(f"{g}/surrogate_objective_type", sm_dt.itemsize, _np_members(sm_dt))

    spec = np.zeros(len(obj_names), dtype=spec_dt)
    spec["objective"] = np.arange(len(obj_names))
    f.create_dataset(f"{g}/objective_spec", f"{g}/objective_spec_type", 0, len(obj_names))
    f.write_rows(f"{g}/objective_spec", spec.tobytes(), len(obj_names))

    # features
    if feature_dtypes is not None:
        feat_names = [fd[0] for fd in feature_dtypes]
        f.commit_enum(f"{g}/feature_enum", feat_names, list(range(len(feat_names))))
        fspec_dt = np.dtype({"names": ["feature"], "formats": [np.uint16]})
        f.commit_compound(
            f"{g}/feature_spec_type", fspec_dt.itemsize,
            [("feature", 0, f"{g}/feature_enum", 1)],
        )
        feat_dt = np.dtype(feature_dtypes)
        f.commit_compound(f"{g}/feature_type", feat_dt.itemsize, _np_members(feat_dt))
        spec = np.zeros(len(feat_names), dtype=fspec_dt)
        spec["feature"] = np.arange(len(feat_names))
        f.create_dataset(f"{g}/feature_spec", f"{g}/feature_spec_type", 0, len(feat_names))
        f.write_rows(f"{g}/feature_spec", spec.tobytes(), len(feat_names))

    # constraints
    if constraint_names is not None:
        cons = list(constraint_names)
        f.commit_enum(f"{g}/constraint_enum", cons, list(range(len(cons))))
        cspec_dt = np.dtype({"names": ["constraint"], "formats": [np.uint16]})
        f.commit_compound(
            f"{g}/constraint_spec_type", cspec_dt.itemsize,
            [("constraint", 0, f"{g}/constraint_enum", 1)],
        )
        cons_dt = np.dtype({"names": cons, "formats": [np.float32] * len(cons)})
        f.commit_compound(f"{g}/constraint_type", cons_dt.itemsize, _np_members(cons_dt))
        spec = np.zeros(len(cons), dtype=cspec_dt)
        spec["constraint"] = np.arange(len(cons))
        f.create_dataset(f"{g}/constraint_spec", f"{g}/constraint_spec_type", 0, len(cons))
        f.write_rows(f"{g}/constraint_spec", spec.tobytes(), len(cons))

    # parameters: problem params first, then space params (reference order)
    param_keys: List[str] = []
    for name in problem_parameters.parameter_names:
        if name not in param_keys:
            param_keys.append(name)
    for name in parameter_space.parameter_names:
        if name not in param_keys:
            param_keys.append(name)
    param_mapping = {name: i for i, name in enumerate(param_keys)}
    f.commit_enum(f"{g}/parameter_enum", param_keys, list(range(len(param_keys))))

    ps_names = parameter_space.parameter_names
    ps_dt = np.dtype({"names": ps_names, "formats": [np.float32] * len(ps_names)})
    f.commit_compound(f"{g}/parameter_space_type", ps_dt.itemsize, _np_members(ps_dt))

    pp_dt = np.dtype(
        [("parameter", np.uint16), ("is_integer", np.bool_), ("value", np.float32)]
    )
    f.commit_compound(
        f"{g}/problem_parameters_type", pp_dt.itemsize,
        [
            ("parameter", 0, f"{g}/parameter_enum", 1),
            ("is_integer", pp_dt.fields["is_integer"][1], "b1", 1),
            ("value", pp_dt.fields["value"][1], "f4", 1),
        ],
    )
    n_pp = problem_parameters.n_parameters
    a = np.zeros(n_pp, dtype=pp_dt)
    for i, parm in enumerate(problem_parameters.items):
        a[i]["parameter"] = param_mapping[parm.name]
        a[i]["value"] = parm.value
        a[i]["is_integer"] = parm.is_integer
    f.create_dataset(f"{g}/problem_parameters", f"{g}/problem_parameters_type", 0, max(n_pp, 1))
    if n_pp:
        f.write_rows(f"{g}/problem_parameters", a.tobytes(), n_pp)

    spec_dt2 = np.dtype(
        [
            ("parameter", np.uint16),
            ("is_integer", np.bool_),
            ("lower", np.float32),
            ("upper", np.float32),
        ]
    )
    f.commit_compound(
        f"{g}/parameter_spec_type", spec_dt2.itemsize,
        [
            ("parameter", 0, f"{g}/parameter_enum", 1),
            ("is_integer", spec_dt2.fields["is_integer"][1], "b1", 1),
            ("lower", spec_dt2.fields["lower"][1], "f4", 1),
            ("upper", spec_dt2.fields["upper"][1], "f4", 1),
        ],
    )
    n_sp = parameter_space.n_parameters
    a = np.zeros(n_sp, dtype=spec_dt2)
    for i, parm in enumerate(parameter_space.items):
        a[i]["parameter"] = param_mapping[parm.name]
        a[i]["is_integer"] = parm.is_integer
        a[i]["lower"] = parm.lower
        a[i]["upper"] = parm.upper
    f.create_dataset(f"{g}/parameter_spec", f"{g}/parameter_spec_type", 0, n_sp)
    f.write_rows(f"{g}/parameter_spec", a.tobytes(), n_sp)

    # parameter paths
    path_dt = np.dtype(
        [
            ("parameter", np.uint16),
            ("path_length", np.int32),
            ("components", f"S{PATH_MAX_NAME}", (PATH_MAX_DEPTH,)),
        ]
    )
    f.commit_compound(
        f"{g}/parameter_path_type", path_dt.itemsize,
        [
            ("parameter", 0, f"{g}/parameter_enum", 1),
            ("path_length", path_dt.fields["path_length"][1], "i4", 1),
            ("components", path_dt.fields["components"][1], f"S{PATH_MAX_NAME}", PATH_MAX_DEPTH),
        ],
    )
    all_paths = parameter_space.parameter_paths
    all_paths.update(problem_parameters.parameter_paths)
    arr = np.zeros(len(all_paths), dtype=path_dt)
    for i, (name, path) in enumerate(all_paths.items()):
        arr[i]["parameter"] = param_mapping[name]
        arr[i]["path_length"] = len(path)
        for j, comp in enumerate(path):
            arr[i]["components"][j] = comp.encode("ascii")
    f.create_dataset(f"{g}/parameter_paths", f"{g}/parameter_path_type", 0, max(len(all_paths), 1))
    if len(all_paths):
        f.write_rows(f"{g}/parameter_paths", arr.tobytes(), len(arr))


def h5_init_opt_group(
    f, opt_id, objective_names, feature_dtypes, constraint_names,
    problem_parameters, parameter_space, problem_ids, has_problem_ids,
    metadata, random_seed, surrogate_mean_variance=False,
):
    if f.has(f"/{opt_id}"):
        return
    h5_init_types(
        f, opt_id, objective_names, feature_dtypes, constraint_names,
        problem_parameters, parameter_space,
        surrogate_mean_variance=surrogate_mean_variance,
    )
    g = f"/{opt_id}"
    if metadata is not None:
        if isinstance(metadata, str):
            f.write_string(f"{g}/metadata", metadata)
        else:
            md = np.asarray(metadata)
            f.write_simple(f"{g}/metadata", f"{md.dtype.kind}{md.dtype.itemsize}",
                           md.tobytes(), list(md.shape))
    pids = np.asarray(sorted(problem_ids) if has_problem_ids else [0], dtype=np.int32)
    f.write_simple(f"{g}/problem_ids", "i4", pids.tobytes(), [len(pids)])
    if random_seed is not None:
        # int64: h5py stores the full Python int; an i4 would overflow for
        # seeds >= 2**31 and break resume reproducibility
        rs = np.asarray([random_seed], dtype=np.int64)
        f.write_simple(f"{g}/random_seed", "i8", rs.tobytes(), [1])


def init_h5(
    opt_id, problem_ids, has_problem_ids, parameter_space, objective_names,
    feature_dtypes, constraint_names, problem_parameters, metadata,
    random_seed, fpath, surrogate_mean_variance=False,
):
    core = _h5core()
    f = core.H5File(fpath, "a")
    try:
        h5_init_opt_group(
            f, opt_id, objective_names, feature_dtypes, constraint_names,
            problem_parameters, parameter_space, problem_ids, has_problem_ids,
            metadata, random_seed, surrogate_mean_variance=surrogate_mean_variance,
        )
    finally:
        f.close()


# -------------------------------------------------------------------- save
def save_to_h5(
    opt_id, problem_ids, has_problem_ids, param_names, objective_names,
    feature_dtypes, constraint_names, optimize_mean_variance, evals, fpath,
    logger=None,
):
    core = _h5core()
    f = core.H5File(fpath, "a")
    try:
        g = f"/{opt_id}"
        obj_names = list(objective_names)
        obj_dt = np.dtype({"names": obj_names, "formats": [np.float32] * len(obj_names)})
        if optimize_mean_variance:
            sm_names = [f"{n} mean" for n in obj_names] + [f"{n} variance" for n in obj_names]
        else:
            sm_names = obj_names
        sm_dt = np.dtype({"names": sm_names, "formats": [np.float32] * len(sm_names)})
        ps_dt = np.dtype({"names": param_names, "formats": [np.float32] * len(param_names)})
        cons_dt = None
        if constraint_names is not None:
            cons_dt = np.dtype(
                {"names": list(constraint_names), "formats": [np.float32] * len(constraint_names)}
            )
        feat_dt = np.dtype(feature_dtypes) if feature_dtypes is not None else None

        for problem_id, ev in evals.items():
            epochs, xs, ys, fs, cs, preds = ev
            base = f"{g}/{problem_id}"
            n = len(ys)
            if logger is not None:
                logger.info(f"Saving {n} evaluations for problem id {problem_id} to {fpath}.")
            ep = np.asarray(
                [0 if e is None else e for e in epochs], dtype=np.uint32
            )
            if not f.has(f"{base}/epochs"):
                f.create_dataset(f"{base}/epochs", "u4", 0, -1)
            f.append_rows(f"{base}/epochs", ep.tobytes(), n)

            ya = np.array([tuple(np.asarray(y).ravel()[: len(obj_names)]) for y in ys], dtype=obj_dt)
            _append_structured(f, f"{base}/objectives", ya, f"{g}/objective_type")

            xa = np.array([tuple(np.asarray(x).ravel()) for x in xs], dtype=ps_dt)
            _append_structured(f, f"{base}/parameters", xa, f"{g}/parameter_space_type")

            if fs is not None and feat_dt is not None:
                fa = np.concatenate([np.atleast_1d(np.asarray(fe, dtype=feat_dt)) for fe in fs])
                _append_structured(f, f"{base}/features", fa, f"{g}/feature_type")

            if cs is not None and cons_dt is not None:
                ca = np.array([tuple(np.asarray(c).ravel()) for c in cs], dtype=cons_dt)
                _append_structured(f, f"{base}/constraints", ca, f"{g}/constraint_type")

            pa = np.array(
                [tuple(np.asarray(p).ravel()[: len(sm_names)]) for p in preds], dtype=sm_dt
            )
            _append_structured(f, f"{base}/predictions", pa, f"{g}/surrogate_objective_type")
    finally:
        f.close()


def save_surrogate_evals_to_h5(
    opt_id, problem_id, epoch, param_names, objective_names, gen_index,
    x_sm, y_sm, fpath, logger=None,
):
    core = _h5core()
    f = core.H5File(fpath, "a")
    try:
        g = f"/{opt_id}"
        sm = f"{g}/surrogate_evals"
        n = x_sm.shape[0]
        if logger is not None:
            logger.info(f"Saving {n} surrogate evaluations for problem id {problem_id}.")
        if not f.has(f"{sm}/epochs"):
            f.create_dataset(f"{sm}/epochs", "u4", 0, -1)
        f.append_rows(f"{sm}/epochs", np.full(n, epoch, dtype=np.uint32).tobytes(), n)
        if not f.has(f"{sm}/generations"):
            f.create_dataset(f"{sm}/generations", "u4", 0, -1)
        f.append_rows(
            f"{sm}/generations", np.asarray(gen_index, dtype=np.uint32).tobytes(), n
        )
        sm_desc = f.committed_type(f"{g}/surrogate_objective_type")
        sm_dt = _desc_to_dtype(sm_desc)
        ya = np.zeros(n, dtype=sm_dt)
        y_arr = np.asarray(y_sm, dtype=np.float32)
        for j, name in enumerate(sm_dt.names):
            if j < y_arr.shape[1]:
                ya[name] = y_arr[:, j]
        _append_structured(f, f"{sm}/objectives", ya, f"{g}/surrogate_objective_type")
        ps_dt = _desc_to_dtype(f.committed_type(f"{g}/parameter_space_type"))
        xa = np.zeros(n, dtype=ps_dt)
        x_arr = np.asarray(x_sm, dtype=np.float32)
        for j, name in enumerate(ps_dt.names):
            xa[name] = x_arr[:, j]
        _append_structured(f, f"{sm}/parameters", xa, f"{g}/parameter_space_type")
    finally:
        f.close()


def save_optimizer_params_to_h5(
    opt_id, problem_id, epoch, optimizer_name, optimizer_params, fpath, logger=None
):
    core = _h5core()
    f = core.H5File(fpath, "a")
    try:
        base = f"/{opt_id}/optimizer_params/{epoch}"
        f.create_group(base)
        if not f.has(f"{base}/optimizer_name"):
            f.write_string(f"{base}/optimizer_name", str(optimizer_name))
        for k, v in optimizer_params.items():
            if v is None or f.has(f"{base}/{k}"):
                continue
            if isinstance(v, str):
                f.write_string(f"{base}/{k}", v)
            elif isinstance(v, (bool, np.bool_)):
                f.write_simple(f"{base}/{k}", "u1", np.asarray([v], np.uint8).tobytes(), [])
            elif isinstance(v, (int, np.integer)):
                f.write_simple(f"{base}/{k}", "i8", np.asarray(v, np.int64).tobytes(), [])
            elif isinstance(v, (float, np.floating)):
                f.write_simple(f"{base}/{k}", "f8", np.asarray(v, np.float64).tobytes(), [])
            elif isinstance(v, np.ndarray) and v.dtype.kind in "fiu":
                arr = np.ascontiguousarray(v)
                f.write_simple(
                    f"{base}/{k}", f"{arr.dtype.kind}{arr.dtype.itemsize}",
                    arr.tobytes(), list(arr.shape),
                )
            # other types (callables, dicts) are skipped
    finally:
        f.close()


def save_stats_to_h5(opt_id, problem_id, epoch, stats, fpath, logger=None):
    core = _h5core()
    f = core.H5File(fpath, "a")
    try:
        g = f"/{opt_id}/optimizer_stats/{epoch}"
        numeric = {
            k: float(v) for k, v in stats.items() if isinstance(v, (int, float, np.number))
        }
        if not numeric:
            return
        names = sorted(numeric)
        dt = np.dtype({"names": names, "formats": [np.float64] * len(names)})
        tpath = f"{g}/stats_type"
        f.commit_compound(tpath, dt.itemsize, _np_members(dt))
        arr = np.array([tuple(numeric[k] for k in names)], dtype=dt)
        _append_structured(f, f"{g}/stats", arr, tpath)
    finally:
        f.close()


# -------------------------------------------------------------------- load
def h5_load_raw(input_file, opt_id):
    core = _h5core()
    f = core.H5File(input_file, "r")
    try:
        g = f"/{opt_id}"
        obj_enum = _enum_names(f.committed_type(f"{g}/objective_enum"))
        obj_spec = _read_structured(f, f"{g}/objective_spec")
        objective_names = [obj_enum[int(s[0])] for s in obj_spec]

        constraint_names = None
        if f.has(f"{g}/constraint_enum"):
            c_enum = _enum_names(f.committed_type(f"{g}/constraint_enum"))
            c_spec = _read_structured(f, f"{g}/constraint_spec")
            constraint_names = [c_enum[int(s[0])] for s in c_spec]

        feature_names = None
        if f.has(f"{g}/feature_enum"):
            f_enum = _enum_names(f.committed_type(f"{g}/feature_enum"))
            f_spec = _read_structured(f, f"{g}/feature_spec")
            feature_names = [f_enum[int(s[0])] for s in f_spec]

        parameter_paths = None
        if f.has(f"{g}/parameter_paths"):
            arr = _read_structured(f, f"{g}/parameter_paths")
            parameter_paths = {}
            for row in arr:
                comps = [
                    c.decode("ascii").rstrip("\x00")
                    for c in row["components"][: row["path_length"]]
                ]
                parameter_paths[".".join(comps)] = comps

        p_enum = _enum_names(f.committed_type(f"{g}/parameter_enum"))

        problem_parameters = {}
        pp = _read_structured(f, f"{g}/problem_parameters")
        for entry in pp:
            name = p_enum[int(entry["parameter"])]
            value = float(entry["value"])
            d = problem_parameters
            if parameter_paths is not None and name in parameter_paths:
                for comp in parameter_paths[name][:-1]:
                    d = d.setdefault(comp, {})
                d[parameter_paths[name][-1]] = value
            else:
                d[name] = value

        spec_arr = _read_structured(f, f"{g}/parameter_spec")
        raw_spec = {}
        param_names = []
        for entry in spec_arr:
            name = p_enum[int(entry["parameter"])]
            param_names.append(name)
            spec = [float(entry["lower"]), float(entry["upper"]), bool(entry["is_integer"])]
            d = raw_spec
            if parameter_paths is not None and name in parameter_paths:
                for comp in parameter_paths[name][:-1]:
                    d = d.setdefault(comp, {})
                d[parameter_paths[name][-1]] = spec
            else:
                d[name] = spec

        problem_ids = None
        if f.has(f"{g}/problem_ids"):
            raw = f.read_rows(f"{g}/problem_ids")
            problem_ids = set(np.frombuffer(raw, dtype=np.int32).tolist())

        raw_results = {}
        for pid in problem_ids if problem_ids is not None else [0]:
            base = f"{g}/{pid}"
            if f.has(f"{base}/objectives"):
                entry = {
                    "objectives": _read_structured(f, f"{base}/objectives"),
                    "parameters": _read_structured(f, f"{base}/parameters"),
                }
                for opt_name in ("features", "constraints", "predictions"):
                    if f.has(f"{base}/{opt_name}"):
                        entry[opt_name] = _read_structured(f, f"{base}/{opt_name}")
                if f.has(f"{base}/epochs"):
                    entry["epochs"] = np.frombuffer(
                        f.read_rows(f"{base}/epochs"), dtype=np.uint32
                    ).copy()
                raw_results[pid] = entry

        random_seed = None
        if f.has(f"{g}/random_seed"):
            raw = f.read_rows(f"{g}/random_seed")
            dt = np.int64 if len(raw) == 8 else np.int32  # i4 = pre-0.2 files
            random_seed = int(np.frombuffer(raw, dtype=dt)[0])

        info = {
            "random_seed": random_seed,
            "objectives": objective_names,
            "features": feature_names,
            "constraints": constraint_names,
            "params": param_names,
            "problem_parameters": problem_parameters,
            "problem_ids": problem_ids,
        }
        return raw_spec, raw_results, info
    finally:
        f.close()


def h5_load_all(file_path, opt_id):
    raw_spec, raw_problem_results, info = h5_load_raw(file_path, opt_id)
    evals = {}
    for pid, raw in raw_problem_results.items():
        out = []
        ys, xs = raw["objectives"], raw["parameters"]
        epochs = raw.get("epochs")
        fs = raw.get("features")
        cs = raw.get("constraints")
        preds = raw.get("predictions")
        for i in range(len(ys)):
            out.append(
                EvalEntry(
                    int(epochs[i]) if epochs is not None else None,
                    list(xs[i]),
                    list(ys[i]),
                    fs[i] if fs is not None else None,
                    list(cs[i]) if cs is not None else None,
                    list(preds[i]) if preds is not None else None,
                )
            )
        evals[pid] = out
    return raw_spec, evals, info


def init_from_h5(file_path, param_names, opt_id, logger=None):
    raw_spec, old_evals, info = h5_load_all(file_path, opt_id)
    param_space = ParameterSpace.from_dict(raw_spec)
    saved_params = info["params"]
    max_epoch = -1
    for pid in old_evals:
        if logger is not None:
            logger.info(f"Restored {len(old_evals[pid])} trials for problem {pid}")
        for ev in old_evals[pid]:
            if ev.epoch is not None:
                max_epoch = max(max_epoch, ev.epoch)
            else:
                break
    if param_names is not None and list(param_names) != list(saved_params):
        raise RuntimeError(
            f"Saved parameters {saved_params} differ from currently specified {param_names}."
        )
    problem_parameters = ParameterSpace.from_dict(
        info["problem_parameters"], is_value_only=True
    )
    return (
        info["random_seed"],
        max_epoch,
        old_evals,
        param_space,
        info["objectives"],
        info["features"],
        info["constraints"],
        problem_parameters,
        info["problem_ids"],
    )
