"""Constrained parameter-space sampling (reference constrained_sampling.py).

``ParamSpacePoints(N, Space, Method, seed, parents)`` samples a space whose
entries are either unconstrained ranges ``[lo, hi]`` or constrained specs:

    {"abs": (lo, hi),                       # absolute fallback bounds
     "lb": [("a", "* 2 + 1"), ...],         # lower bounds: expr of params
     "ub": [("b", "- 0.5")],                # upper bounds
     "method": ("uniform",) | ("normal", mu, kappa) | ("percentile", q)}

Dependencies form a DAG solved in rank order; the reference's sly
Lexer/Parser for the bound expressions is replaced by a safe Python-ast
arithmetic evaluator (numbers, + - * / ** and parentheses only).
"""

from __future__ import annotations

import ast
import operator
from typing import Dict, Optional

import numpy as np
from numpy.random import default_rng

from dmosopt_amd import sampling

_OPS = {
    ast.Add: operator.add,
    ast.Sub: operator.sub,
    ast.Mult: operator.mul,
    ast.Div: operator.truediv,
    ast.Pow: operator.pow,
    ast.USub: operator.neg,
    ast.UAdd: operator.pos,
}


def safe_eval_arith(expr: str) -> float:
    """Evaluate a pure-arithmetic expression string safely."""

    def ev(node):
        if isinstance(node, ast.Expression):
            return ev(node.body)
        if isinstance(node, ast.Constant) and isinstance(node.value, (int, float)):
            return float(node.value)
        if isinstance(node, ast.BinOp) and type(node.op) in _OPS:
            return _OPS[type(node.op)](ev(node.left), ev(node.right))
        if isinstance(node, ast.UnaryOp) and type(node.op) in _OPS:
            return _OPS[type(node.op)](ev(node.operand))
        raise ValueError(f"Disallowed expression element: {ast.dump(node)}")

    return ev(ast.parse(expr, mode="eval"))


class ParamSpacePoints:
    def __init__(self, N, Space: Dict, Method=None, seed=None, parents=None):
        self.seed = seed
        self.rng = default_rng(seed)
        self.N_params = N
        self.Space = Space
        self.parents_dict = parents
        self.MethodUnc = Method or "slh"

        self.param_keys = np.sort(list(Space.keys()))
        self.prm_idx_unc = np.array(
            [i for i, k in enumerate(self.param_keys) if isinstance(Space[k], list)],
            dtype=int,
        )
        self.prm_idx_con = np.array(
            [i for i, k in enumerate(self.param_keys) if isinstance(Space[k], dict)],
            dtype=int,
        )
        self.param_dim = len(self.param_keys)
        self.param_arr = np.full((N, self.param_dim), np.nan)

        self._generate_unconstrained()
        if len(self.prm_idx_con):
            self._generate_constrained()

    # ------------------------------------------------------------- parts
    def _generate_unconstrained(self):
        d = len(self.prm_idx_unc)
        if d == 0:
            return
        intervals = np.array(
            [self.Space[self.param_keys[i]] for i in self.prm_idx_unc], dtype=float
        )
        sampler = getattr(sampling, self.MethodUnc, sampling.slh)
        u = sampler(self.N_params, d, self.rng)
        vals = intervals[:, 0] + u * (intervals[:, 1] - intervals[:, 0])
        self.param_arr[:, self.prm_idx_unc] = vals

    def _dependency_order(self):
        """Rank constrained params by unresolved dependencies (DAG order)."""
        keys = [self.param_keys[i] for i in self.prm_idx_con]
        resolved = set(self.param_keys[i] for i in self.prm_idx_unc)
        order = []
        remaining = dict()
        for k in keys:
            spec = self.Space[k]
            deps = set()
            for bkey in ("lb", "ub"):
                for prm, _rel in spec.get(bkey, []):
                    deps.add(prm)
            remaining[k] = deps
        guard = 0
        while remaining and guard <= len(keys) + 1:
            ready = [k for k, deps in remaining.items() if deps <= resolved]
            if not ready:
                raise ValueError(
                    f"Circular or unresolved constraint dependencies: {remaining}"
                )
            for k in sorted(ready):
                order.append(k)
                resolved.add(k)
                del remaining[k]
            guard += 1
        return order

    def _values_of(self, name: str) -> np.ndarray:
        idx = int(np.where(self.param_keys == name)[0][0])
        return self.param_arr[:, idx]

    def _bound_values(self, cons_list, lower: bool) -> np.ndarray:
        """Per-sample bound: max over lower constraints / min over upper."""
        per_con = []
        for prm, rel in cons_list:
            vals = self._values_of(prm)
            out = np.array([safe_eval_arith(f"{v} {rel}") for v in vals])
            per_con.append(out)
        stack = np.stack(per_con, axis=1)
        return stack.max(axis=1) if lower else stack.min(axis=1)

    def _generate_constrained(self):
        for key in self._dependency_order():
            spec = self.Space[key]
            absbnds = spec.get("abs")
            lb = ub = None
            if "lb" in spec:
                lb = self._bound_values(spec["lb"], lower=True)
            if "ub" in spec:
                ub = self._bound_values(spec["ub"], lower=False)
            if absbnds is None and (lb is None or ub is None):
                raise KeyError(
                    "Constrained parameter requires both bounds when no "
                    "absolute bounds are specified."
                )
            if lb is None:
                lb = np.full(self.N_params, absbnds[0])
            if ub is None:
                ub = np.full(self.N_params, absbnds[1])
            if absbnds is not None:
                bad = ~(lb < ub)
                if bad.any():
                    lb[bad] = absbnds[0]
                    ub[bad] = absbnds[1]
            method = spec.get("method", ("uniform",))
            vals = self._sample(lb, ub, method)
            idx = int(np.where(self.param_keys == key)[0][0])
            self.param_arr[:, idx] = vals

    def _sample(self, lb, ub, method):
        kind = method[0]
        if kind == "uniform":
            return self.rng.uniform(lb, ub)
        if kind == "normal":
            mu = method[1] if len(method) > 1 else 0.0
            kappa = method[2] if len(method) > 2 else 4.0
            off = 0.5 * self.rng.vonmises(mu, kappa, size=self.N_params) / np.pi
            return 0.5 * (lb + ub) + off * (ub - lb)
        if kind == "percentile":
            q = method[1] if len(method) > 1 else 0.5
            return lb + q * (ub - lb)
        raise ValueError(f"Unknown constrained sampling method {kind}")

    # ------------------------------------------------------------- output
    @property
    def values(self) -> np.ndarray:
        return self.param_arr

    def as_dict(self) -> Dict[str, np.ndarray]:
        return {k: self.param_arr[:, i] for i, k in enumerate(self.param_keys)}
