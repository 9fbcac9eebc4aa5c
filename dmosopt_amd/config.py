"""Plugin resolution and shorthand registries.

Parity with the reference registries (``/root/reference/dmosopt/config.py``):
samplers, optimizers, surrogates, sensitivity methods, feasibility methods are
resolvable either by shorthand name or by dotted import path.
"""

from __future__ import annotations

import importlib
import sys
from typing import Any


def import_object_by_path(path: str) -> Any:
    """Resolve ``"module.sub.object"`` to the Python object.

    ``__main__.name`` resolves against the running script, matching the
    reference plugin mechanism (reference config.py:5-11).
    """
    if "." not in path:
        raise ValueError(f"Cannot import object from path without module: {path!r}")
    module_name, obj_name = path.rsplit(".", 1)
    if module_name == "__main__":
        module = sys.modules["__main__"]
    else:
        module = importlib.import_module(module_name)
    return getattr(module, obj_name)


# Shorthand -> dotted path. Kept as paths (not objects) so importing this
# module stays cheap and optional heavy deps load lazily.
sampler_registry = {
    "glp": "dmosopt_amd.sampling.glp",
    "slh": "dmosopt_amd.sampling.slh",
    "lh": "dmosopt_amd.sampling.lh",
    "mc": "dmosopt_amd.sampling.mc",
    "sobol": "dmosopt_amd.sampling.sobol",
}

optimizer_registry = {
    "nsga2": "dmosopt_amd.moea.nsga2.NSGA2Optimizer",
    "age": "dmosopt_amd.moea.agemoea.AGEMOEAOptimizer",
    "smpso": "dmosopt_amd.moea.smpso.SMPSOOptimizer",
    "cmaes": "dmosopt_amd.moea.cmaes.CMAESOptimizer",
    "trs": "dmosopt_amd.moea.trs.TRSOptimizer",
}

surrogate_registry = {
    "gpr": "dmosopt_amd.models.gp.GPRMatern",
    "egp": "dmosopt_amd.models.gp.EGPMatern",
    "megp": "dmosopt_amd.models.multitask_gp.MEGPMaternICM",
    "mdgp": "dmosopt_amd.models.deep_gp.MDGPMatern",
    "mdspp": "dmosopt_amd.models.deep_gp.MDSPPMatern",
    "vgp": "dmosopt_amd.models.variational_gp.VGPMatern",
    "svgp": "dmosopt_amd.models.variational_gp.SVGPMatern",
    "spv": "dmosopt_amd.models.variational_gp.SPVMatern",
    "siv": "dmosopt_amd.models.variational_gp.SIVMatern",
    "crv": "dmosopt_amd.models.variational_gp.CRVMatern",
}

sensitivity_registry = {
    "dgsm": "dmosopt_amd.models.sa.SA_DGSM",
    "fast": "dmosopt_amd.models.sa.SA_FAST",
}

feasibility_registry = {
    "logreg": "dmosopt_amd.models.feasibility.LogisticFeasibilityModel",
}


def resolve(registry: dict, name_or_path):
    """Resolve a shorthand name, dotted path, or callable to an object."""
    if callable(name_or_path):
        return name_or_path
    path = registry.get(name_or_path, name_or_path)
    try:
        return import_object_by_path(path)
    except (ValueError, ImportError, AttributeError) as e:
        raise ValueError(
            f"Cannot resolve {name_or_path!r}: not a registered shorthand "
            f"({sorted(registry)}) and not an importable dotted path ({e})"
        ) from e
