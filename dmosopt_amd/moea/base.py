"""MOEA base class — device-resident populations.

Semantics follow the reference optimizer protocol (MOEA.py:55-188):
``generate_initial(bounds, local_random)``, ``initialize_strategy(x, y,
bounds, local_random)``, ``generate() -> (x, gen_state)``, ``update(x, y,
gen_state)``, ``population_objectives``. Populations are torch tensors on
the engine's device; the host numpy Generator drives control-flow draws
while batched gene-level RNG uses a torch Generator seeded from it.
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import numpy as np
import torch

from dmosopt_amd import sampling
from dmosopt_amd.datatypes import Struct


class MOEA:
    def __init__(self, name: str, popsize: int, nInput: int, nOutput: int, **kwargs):
        self.name = name
        self.popsize = popsize
        self.nInput = nInput
        self.nOutput = nOutput
        self.opt_params = Struct(**self.default_parameters)
        self.opt_params.update(
            {
                "popsize": popsize,
                "nInput": nInput,
                "nOutput": nOutput,
                "initial_size": popsize,
                "initial_sampling_method": None,
                "initial_sampling_method_params": None,
            }
        )
        for k, v in kwargs.items():
            if k not in self.opt_params:
                self.opt_params[k] = v
            elif v is not None:
                self.opt_params[k] = v
        self.local_random: Optional[np.random.Generator] = None
        self.torch_random: Optional[torch.Generator] = None
        self.state = None
        self.bounds: Optional[torch.Tensor] = None
        self.device = torch.device("cpu")
        self.dtype = torch.float64
        self.logger = kwargs.get("logger", None)

    # ---------------------------------------------------------------- misc
    @property
    def default_parameters(self) -> Dict[str, Any]:
        return {}

    @property
    def opt_parameters(self) -> Dict[str, Any]:
        params = self.opt_params()
        return params

    @property
    def population_objectives(self) -> Tuple[torch.Tensor, torch.Tensor]:
        return self.get_population_strategy()

    def get_population_strategy(self):
        raise NotImplementedError

    def _as_tensor(self, a) -> torch.Tensor:
        return torch.as_tensor(a, dtype=self.dtype, device=self.device)

    def _make_torch_generator(self, local_random: Optional[np.random.Generator]):
        g = torch.Generator(device=self.device)
        if local_random is not None:
            g.manual_seed(int(local_random.integers(0, 2**63 - 1)))
        return g

    # ------------------------------------------------------------ protocol
    def set_device(self, device, dtype=None):
        self.device = torch.device(device)
        if dtype is not None:
            self.dtype = dtype
        elif self.device.type == "cuda":
            self.dtype = torch.float32
        return self

    def generate_initial(self, bounds, local_random) -> np.ndarray:
        """Initial design in the bounded space (host-side numpy, scaled)."""
        bounds = np.asarray(bounds, dtype=np.float64)
        xlb, xub = bounds[:, 0], bounds[:, 1]
        n = self.opt_params.initial_size
        method = self.opt_params.initial_sampling_method
        method_params = self.opt_params.initial_sampling_method_params
        if method is None:
            x = sampling.lh(n, self.nInput, local_random)
            x = x * (xub - xlb) + xlb
        elif method == "sobol":
            x = sampling.sobol(n, self.nInput, local_random)
            x = x * (xub - xlb) + xlb
        elif callable(method):
            if method_params is None:
                x = method(local_random, n, self.nInput, xlb, xub)
            else:
                x = method(local_random, **method_params)
        else:
            raise RuntimeError(f"Unknown sampling method {method}")
        return x

    def initialize_strategy(self, x, y, bounds, local_random=None, **params):
        self.local_random = local_random if local_random is not None else np.random.default_rng()
        self.torch_random = self._make_torch_generator(self.local_random)
        bounds_t = self._as_tensor(np.asarray(bounds, dtype=np.float64))
        self.bounds = bounds_t
        x_t = self._as_tensor(np.asarray(x))
        y_t = self._as_tensor(np.asarray(y))
        self.state = self.initialize_state(x_t, y_t, bounds_t, self.local_random, **params)
        return self.state

    def generate(self, **params):
        x, gen_state = self.generate_strategy(**params)
        # strategies whose kernels already clamp every gene to the bounds
        # mark the gen_state: the redundant per-generation clamp (a kernel
        # plus ~8 us of host dispatch) is skipped on the hot path
        if not (isinstance(gen_state, dict) and gen_state.get("clamped")):
            x = x.clamp(self.bounds[:, 0], self.bounds[:, 1])
        return x, gen_state

    def update(self, x, y, gen_state, **params):
        x_t = x if isinstance(x, torch.Tensor) else self._as_tensor(x)
        y_t = y if isinstance(y, torch.Tensor) else self._as_tensor(y)
        self.update_strategy(x_t, y_t, gen_state, **params)
        return self.state

    # --------------------------------------------------- strategy-specific
    def initialize_state(self, x, y, bounds, local_random, **params):
        raise NotImplementedError

    def generate_strategy(self, **params):
        raise NotImplementedError

    def update_strategy(self, x, y, gen_state, **params):
        raise NotImplementedError
