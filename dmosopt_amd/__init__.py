"""dmosopt_amd — MI355X-native multi-objective adaptive surrogate optimization.

A from-scratch rebuild of the capabilities of dmosopt (MO-ASMO) designed for
AMD Instinct MI355X: PyTorch-ROCm tensors, hand-written gfx950 HIP kernels
for the hot population/GP math, RCCL collectives over xGMI for distribution.
"""

__version__ = "0.1.0"

import os as _os

# Run-to-run bit determinism on the CPU path: torch's CPU backend is MKL,
# and MKL_DYNAMIC (default TRUE) lets MKL SHRINK its thread team when the
# machine is loaded — a different team size changes GEMM/reduction
# partitioning, the CPU GP fit's floats move by ULPs, SCE-UA accept
# decisions flip, and two same-seed runs diverge (observed ~8% of
# processes under load). The replicated-control-flow multi-rank scheme
# and the seeded-reproducibility contract both require fixed teams.
# setdefault: a user who explicitly set these keeps their choice.
_os.environ.setdefault("MKL_DYNAMIC", "FALSE")
_os.environ.setdefault("OMP_DYNAMIC", "FALSE")

from dmosopt_amd.datatypes import ParameterSpace  # noqa: F401

sopt_dict = {}


def run(dopt_params, time_limit=None, feasible=True, return_features=False,
        return_constraints=False, spawn_workers=False, sequential_spawn=False,
        spawn_startup_wait=None, spawn_executable=None, spawn_args=[],
        nprocs_per_worker=1, collective_mode="gather", verbose=True,
        worker_debug=False):
    """Run a distributed optimization. See dmosopt_amd.api.run.

    The MPI-spawn arguments of the reference surface (spawn_workers,
    sequential_spawn, spawn_startup_wait, spawn_executable, spawn_args,
    nprocs_per_worker, collective_mode, worker_debug) are accepted for API
    compatibility; distribution here is torch.distributed (one process per
    GPU, launched externally via torchrun), so they have no effect.
    """
    from dmosopt_amd.api import run as _run

    if spawn_workers or nprocs_per_worker != 1:
        import logging

        logging.getLogger("dmosopt_amd").warning(
            "spawn_workers/nprocs_per_worker are MPI-spawn options of the "
            "reference API; launch ranks with torchrun instead (ignored)"
        )

    return _run(
        dopt_params,
        time_limit=time_limit,
        feasible=feasible,
        return_features=return_features,
        return_constraints=return_constraints,
        verbose=verbose,
    )
