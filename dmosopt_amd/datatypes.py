"""Core datatypes for dmosopt_amd.

Semantics-compatible with the reference implementation's parameter-space
handling (``/root/reference/dmosopt/datatypes.py``): nested parameter
dictionaries flatten to dotted-path names in sorted order, bounds auto-swap
when inverted, and value-only spaces carry fixed problem parameters.

The implementation here is written fresh for the MI355X-native framework:
bounds are materialized once as contiguous float64 numpy arrays (and cached
torch tensors on demand) so that device-resident optimizers can consume them
without per-call Python traversal.
"""

from __future__ import annotations

import copy
from collections import namedtuple
from dataclasses import dataclass, field
from enum import IntEnum
from typing import Any, Dict, List, Optional, Sequence, Union

import numpy as np


class Struct:
    """Attribute bag used for optimizer hyper-parameter bundles."""

    def __init__(self, **items):
        self.__dict__.update(items)

    def update(self, items):
        self.__dict__.update(items)

    def __call__(self):
        return self.__dict__

    def __getitem__(self, key):
        return self.__dict__[key]

    def __setitem__(self, key, val):
        self.__dict__[key] = val

    def items(self):
        return self.__dict__.items()

    def __contains__(self, key):
        return key in self.__dict__

    def __repr__(self):
        return f"Struct({self.__dict__})"

    def __str__(self):
        return "<Struct>"


@dataclass
class ParameterValue:
    """A fixed (non-optimized) parameter value."""

    value: float
    is_integer: bool = False
    name: Optional[str] = None


@dataclass
class ParameterDefn:
    """An optimizable parameter range. Inverted bounds are swapped."""

    lower: float
    upper: float
    is_integer: bool = False
    name: Optional[str] = None

    def __post_init__(self):
        if self.lower > self.upper:
            self.lower, self.upper = self.upper, self.lower


_Leaf = Union[ParameterDefn, ParameterValue]


@dataclass
class ParameterSpace:
    """Nested -> flat parameter space with dotted-path naming.

    ``from_dict`` accepts either flat dicts ``{name: [lo, hi, is_int?]}`` or
    arbitrarily nested dicts whose leaves are such lists (or scalars when
    ``is_value_only``). Flattening visits keys in sorted order at every level,
    matching the reference semantics so optimization results line up
    column-for-column.
    """

    ranges: Dict[str, Union[_Leaf, "ParameterSpace"]] = field(default_factory=dict)
    _flat: List[_Leaf] = field(default_factory=list, init=False)
    _paths: Dict[str, List[str]] = field(default_factory=dict, init=False)

    def __post_init__(self):
        self._rebuild()

    # -- construction --------------------------------------------------
    @classmethod
    def from_dict(cls, config: Dict, is_value_only: bool = False) -> "ParameterSpace":
        def parse(node: Any) -> Union[_Leaf, "ParameterSpace"]:
            if isinstance(node, (list, tuple)):
                lo, hi = float(node[0]), float(node[1])
                is_int = bool(node[2]) if len(node) > 2 else False
                return ParameterDefn(lower=lo, upper=hi, is_integer=is_int)
            if isinstance(node, (int, float, np.integer, np.floating)) and is_value_only:
                return ParameterValue(
                    value=float(node), is_integer=isinstance(node, (int, np.integer))
                )
            if isinstance(node, dict):
                return cls(ranges={k: parse(v) for k, v in node.items()})
            raise ValueError(f"Unexpected parameter spec type: {type(node)}")

        out = parse(config)
        if not isinstance(out, ParameterSpace):
            raise ValueError("Top-level parameter spec must be a dict")
        return out

    def _rebuild(self, prefix: str = "") -> None:
        self._flat = []
        self._paths = {}
        for name in sorted(self.ranges):
            item = self.ranges[name]
            path = f"{prefix}.{name}" if prefix else name
            if isinstance(item, (ParameterDefn, ParameterValue)):
                item.name = path
                self._flat.append(item)
                self._paths[path] = path.split(".")
            elif isinstance(item, ParameterSpace):
                item._rebuild(path)
                self._flat.extend(item._flat)
                self._paths.update(item._paths)
            else:
                raise ValueError(f"Unexpected item in ParameterSpace: {type(item)}")

    # -- properties ----------------------------------------------------
    @property
    def is_value_space(self) -> bool:
        return all(isinstance(r, ParameterValue) for r in self._flat)

    @property
    def parameter_values(self) -> np.ndarray:
        if not self.is_value_space:
            raise ValueError("Not a value-only parameter space")
        return np.asarray([p.value for p in self._flat], dtype=np.float64)

    @property
    def parameter_names(self) -> List[str]:
        return [p.name for p in self._flat]

    @property
    def parameter_paths(self) -> Dict[str, List[str]]:
        return dict(self._paths)

    @property
    def items(self) -> List[_Leaf]:
        return list(self._flat)

    @property
    def n_parameters(self) -> int:
        return len(self._flat)

    @property
    def bound1(self) -> np.ndarray:
        if self.is_value_space:
            raise ValueError("Cannot get bounds from a value-only parameter space")
        return np.asarray([p.lower for p in self._flat], dtype=np.float64)

    @property
    def bound2(self) -> np.ndarray:
        if self.is_value_space:
            raise ValueError("Cannot get bounds from a value-only parameter space")
        return np.asarray([p.upper for p in self._flat], dtype=np.float64)

    @property
    def is_integer(self) -> np.ndarray:
        return np.asarray([p.is_integer for p in self._flat], dtype=bool)

    # -- conversions ---------------------------------------------------
    def flatten(self, params: Dict) -> np.ndarray:
        out = np.zeros(self.n_parameters, dtype=np.float64)
        for i, leaf in enumerate(self._flat):
            node = params
            path = self._paths[leaf.name]
            for key in path[:-1]:
                node = node[key]
            out[i] = node[path[-1]]
        return out

    def unflatten(self, flat_params: Optional[np.ndarray] = None) -> Dict:
        if flat_params is None:
            return self.unflatten(self.parameter_values)
        params: Dict = {}
        for i, leaf in enumerate(self._flat):
            node = params
            path = self._paths[leaf.name]
            for key in path[:-1]:
                node = node.setdefault(key, {})
            node[path[-1]] = flat_params[i]
        return params


class StrategyState(IntEnum):
    EnqueuedRequests = 1
    WaitingRequests = 2
    CompletedEpoch = 3
    CompletedGeneration = 4


EvalEntry = namedtuple(
    "EvalEntry",
    ["epoch", "parameters", "objectives", "features", "constraints", "prediction", "time"],
    defaults=[None, None, None, None, None, None, -1.0],
)

EvalRequest = namedtuple("EvalRequest", ["parameters", "prediction", "epoch"])

OptHistory = namedtuple("OptHistory", ["n_gen", "n_eval", "x", "y", "c"])

EpochResults = namedtuple(
    "EpochResults", ["best_x", "best_y", "gen_index", "x", "y", "optimizer"]
)

GenerationResults = namedtuple(
    "GenerationResults", ["best_x", "best_y", "gen_index", "x", "y", "optimizer_params"]
)


class OptProblem:
    """Problem definition: dimensions, bounds, eval callable, names."""

    __slots__ = (
        "dim",
        "lb",
        "ub",
        "int_var",
        "eval_fun",
        "param_names",
        "objective_names",
        "feature_dtypes",
        "feature_constructor",
        "constraint_names",
        "n_objectives",
        "n_features",
        "n_constraints",
        "logger",
    )

    def __init__(
        self,
        param_names: Sequence[str],
        objective_names: Sequence[str],
        feature_dtypes,
        feature_constructor,
        constraint_names,
        spec: ParameterSpace,
        eval_fun,
        logger=None,
    ):
        self.lb = spec.bound1
        self.ub = spec.bound2
        self.dim = len(self.lb)
        assert self.dim > 0
        self.int_var = spec.is_integer
        self.eval_fun = eval_fun
        self.param_names = list(param_names)
        self.objective_names = list(objective_names)
        self.feature_dtypes = feature_dtypes
        self.feature_constructor = feature_constructor
        self.constraint_names = constraint_names
        self.n_objectives = len(objective_names)
        self.n_features = len(feature_dtypes) if feature_dtypes is not None else None
        self.n_constraints = len(constraint_names) if constraint_names is not None else None
        self.logger = logger


def update_nested_dict(base: Dict, update: Dict) -> Dict:
    """Recursive dict merge; `update` wins on conflicts."""
    result = dict(base)
    for key, value in update.items():
        if key in result and isinstance(result[key], dict) and isinstance(value, dict):
            result[key] = update_nested_dict(result[key], value)
        else:
            result[key] = value
    return result


def deepcopy_params(d: Dict) -> Dict:
    return copy.deepcopy(d)
