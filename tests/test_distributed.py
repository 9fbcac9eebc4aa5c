"""Multi-process farm tests on gloo (CPU, world_size 2).

Mirrors the reference's cluster-free distributed test
(tests/mpi/test_mpi_distributed.py: oversubscribed ranks on one box) using
torch.distributed gloo instead of MPI.
"""

import os
import sys

import numpy as np
import pytest
import torch.multiprocessing as mp


def _sphere_objfun(pp):
    names = sorted(pp.keys())
    x = np.array([pp[k] for k in names])
    return np.array([np.sum(x**2), np.sum((x - 1.0) ** 2)])


def _worker(rank, world_size, port, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import dmosopt_amd

    params = {
        "opt_id": "t_dist",
        "obj_fun": _sphere_objfun,
        "problem_parameters": {},
        "space": {f"x{i}": [0.0, 1.0] for i in range(4)},
        "objective_names": ["f1", "f2"],
        "population_size": 8,
        "num_generations": 2,
        "surrogate_method_name": None,
        "optimizer": "nsga2",
        "n_initial": 2,
        "n_epochs": 1,
        "random_seed": 9,
    }
    best = dmosopt_amd.run(params, verbose=False)
    out_q.put((rank, best is not None))


def test_two_rank_farm():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29631
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, has_best = q.get(timeout=300)
        results[rank] = has_best
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # controller gets a best set; workers return None
    assert results[0] is True
    assert results[1] is False
