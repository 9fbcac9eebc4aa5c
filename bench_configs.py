"""Reproducible benchmarks for EVERY BASELINE.json config (#1..#5).

One command per config, same JSON contract as bench.py (the driver-parsed
headline stays in bench.py; this harness covers the rest of the table):

  python bench_configs.py --config 1            # README CPU config
  python bench_configs.py --config 2            # ZDT1 GP bf16, 1 GPU
  python bench_configs.py --config 3            # ZDT3 AGEMOEA pop=1024 + HV term
  torchrun --nnodes=1 --nproc-per-node 8 bench_configs.py --config 4
  torchrun --nnodes=1 --nproc-per-node 8 bench_configs.py --config 5

A "step" is one full MO-ASMO epoch of the named config: surrogate fit
(+ feasibility fit where the config names one) + the full inner MOEA run +
resample selection, on synthetic random-init archives of the named shape.
Where BASELINE.json leaves a size unstated (#4's population, #3/#5 epoch
generations) the value is fixed here and recorded in the JSON config block.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

_local_world = int(os.environ.get("LOCAL_WORLD_SIZE", os.environ.get("WORLD_SIZE", "1")))
torch.set_num_threads(max(1, min(8, (os.cpu_count() or 8) // max(1, _local_world))))

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from dmosopt_amd.benchmarks import problems as bp
from dmosopt_amd.core import engine
from dmosopt_amd.parallel import comm


class _Prob:
    """Termination-facing problem shim."""

    def __init__(self, n_objectives, d):
        self.n_objectives = n_objectives
        self.lb = np.zeros(d)
        self.ub = np.ones(d)
        self.logger = None


def _epoch_common(step_seed, *, d, n_obj, pop, gens, archive_n, problem_fn,
                  optimizer_name, optimizer_kwargs=None, constraints=False,
                  feasibility=None, termination_factory=None, device=None,
                  surrogate_kwargs=None, pct=0.25):
    rng = np.random.default_rng(1234)
    X = rng.random((archive_n, d))
    out = problem_fn(X)
    C = None
    if constraints:
        Y, C = out
        Y, C = Y.numpy(), C.numpy()
    else:
        Y = out.numpy()
    term = termination_factory(_Prob(n_obj, d)) if termination_factory else None
    res = engine.run_epoch(
        gens,
        [f"x{i}" for i in range(d)],
        [f"f{j}" for j in range(n_obj)],
        np.zeros(d), np.ones(d), pct, X, Y, C,
        pop=pop,
        optimizer_name=optimizer_name,
        optimizer_kwargs=optimizer_kwargs or {},
        surrogate_method_name="gpr",
        surrogate_method_kwargs=dict(
            {"anisotropic": False, "optimizer": "sceua", "seed": step_seed},
            **(surrogate_kwargs or {}),
        ),
        feasibility_method_name=feasibility,
        termination=term,
        local_random=np.random.default_rng(step_seed + 1),
        device=device,
    )
    return res


def config_1(step_seed, device, world):
    """README canonical config on CPU: ZDT1 d=30, nsga2+gpr, pop=200,
    gens=200 (BASELINE.md 'plumbing, no GPU')."""
    return _epoch_common(
        step_seed, d=30, n_obj=2, pop=200, gens=200, archive_n=300,
        problem_fn=bp.zdt1, optimizer_name="nsga2",
        device=torch.device("cpu"),
    )


def config_2(step_seed, device, world):
    """ZDT1 d=30 GP surrogate, pop=200, bf16 posterior path, 1 MI355X."""
    return _epoch_common(
        step_seed, d=30, n_obj=2, pop=200, gens=200, archive_n=300,
        problem_fn=bp.zdt1, optimizer_name="nsga2", device=device,
        surrogate_kwargs={"compute": "bf16" if device.type == "cuda" else "fp32"},
    )


def config_3(step_seed, device, world):
    """ZDT3 d=30 AGEMOEA pop=1024 gen<=500, HV-box-decomposition progress
    termination (device hv2d kernel on GPU), 1 MI355X."""
    from dmosopt_amd.termination.hv_progress import HypervolumeProgressTermination

    def term(prob):
        return HypervolumeProgressTermination(
            prob, n_last=10, nth_gen=5, n_max_gen=500, min_generations=50
        )

    return _epoch_common(
        step_seed, d=30, n_obj=2, pop=1024, gens=500, archive_n=512,
        problem_fn=bp.zdt3, optimizer_name="age", device=device,
        termination_factory=term,
    )


def config_4(step_seed, device, world):
    """DTLZ2 d=128, 5 objectives, GP surrogate; replicated MOEA +
    rank-sharded GP prediction over RCCL/xGMI at world ranks."""
    return _epoch_common(
        step_seed, d=128, n_obj=5, pop=512, gens=100, archive_n=1024,
        problem_fn=lambda x: bp.dtlz2(x, n_obj=5),
        optimizer_name="nsga2", device=device,
    )


def config_5(step_seed, device, world):
    """TNK constrained + CMA-ES pop=4096 + logreg feasibility model
    (large-pop batched rank-1 Cholesky updates; EHVI selection)."""
    return _epoch_common(
        step_seed, d=2, n_obj=2, pop=4096, gens=30, archive_n=512,
        problem_fn=bp.tnk, optimizer_name="cmaes", constraints=True,
        feasibility="logreg", device=device,
    )


CONFIGS = {
    1: (config_1, "ZDT1 d=30 NSGA2+GPR pop=200 gen=200 (CPU)",
        dict(model="GPR-Matern52 + NSGA2", global_batch=200, seq_len=None,
             parallelism="single-cpu", population_size=200,
             num_generations=200, archive_size=300), "fp32"),
    2: (config_2, "ZDT1 d=30 GP surrogate pop=200 bf16 (1 GPU)",
        dict(model="GPR-Matern52 + NSGA2", global_batch=200, seq_len=None,
             parallelism="1gpu", population_size=200, num_generations=200,
             archive_size=300), "bf16"),
    3: (config_3, "ZDT3 d=30 AGEMOEA pop=1024 HV-termination (1 GPU)",
        dict(model="GPR-Matern52 + AGEMOEA", global_batch=1024, seq_len=None,
             parallelism="1gpu", population_size=1024, num_generations=500,
             archive_size=512, termination="hv-progress(box)"), "fp32"),
    4: (config_4, "DTLZ2 d=128 5-obj GP surrogate (RCCL sharded)",
        dict(model="GPR-Matern52 + NSGA2", global_batch=512, seq_len=None,
             population_size=512, num_generations=100, archive_size=1024),
        "fp32"),
    5: (config_5, "TNK CMA-ES pop=4096 + feasibility (RCCL)",
        dict(model="GPR-Matern52 + MO-CMA-ES + logreg", global_batch=4096,
             seq_len=None, population_size=4096, num_generations=30,
             archive_size=512), "fp32"),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=int, required=True, choices=sorted(CONFIGS))
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--gpus", type=int, default=1)
    args = ap.parse_args()

    rank, world = comm.init_from_env()
    if torch.cuda.is_available():
        device = torch.device(
            "cuda", int(os.environ.get("LOCAL_RANK", 0)) % torch.cuda.device_count()
        )
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    fn, label, cfg_block, dtype = CONFIGS[args.config]

    import torch.distributed as dist

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for w in range(args.warmup):
        fn(100 + w, device, world)
    barrier_sync()
    t0 = time.perf_counter()
    for s in range(args.steps):
        fn(200 + s, device, world)
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if device.type == "cuda":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms = 1000.0 * elapsed / args.steps
    if rank == 0:
        cfg = dict(cfg_block)
        cfg["parallelism"] = (
            f"replicated-moea+sharded-gp dp{world}" if world > 1
            else cfg.get("parallelism", "1gpu")
        )
        print(json.dumps({
            "metric": f"MO-ASMO epoch time, BASELINE config #{args.config}: {label}",
            "value": ms, "unit": "ms_per_epoch", "n_gpus": world,
            "steps": args.steps, "warmup": args.warmup, "ms_per_step": ms,
            "higher_is_better": False, "scaling": "weak", "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic random-init archive",
            "config": cfg,
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
