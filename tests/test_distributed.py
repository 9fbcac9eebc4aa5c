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


def _shard_worker(rank, world_size, port, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    from bench import ShardedGPObjective, make_archive
    from dmosopt_amd.models.gp import GPRMatern

    X, Y = make_archive(seed=5)
    dev = torch.device("cpu")
    gp = GPRMatern(X, Y, 30, 2, np.zeros(30), np.ones(30),
                   optimizer="sceua", seed=7, device=dev)
    obj = ShardedGPObjective(gp, rank, world_size, dev)
    rng = np.random.default_rng(3)
    xq = torch.as_tensor(rng.random((13, 30)))  # odd count exercises padding
    got = obj.evaluate_tensor(xq)
    want = gp.evaluate_tensor(xq)  # single-model reference on every rank
    err = float((got.double() - want.double()).abs().max())
    dist.destroy_process_group()
    out_q.put((rank, err))


def test_sharded_gp_objective_matches_single_rank():
    """The bench's rank-sharded surrogate prediction (the path the driver
    scales to 8 GPUs) must reassemble to the single-rank result exactly
    (identical GP on every rank; all_gather interleave + padding)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_shard_worker, args=(r, 2, 29737, q)) for r in range(2)]
    for p in procs:
        p.start()
    errs = []
    for _ in range(2):
        _, err = q.get(timeout=300)
        errs.append(err)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert max(errs) < 1e-5, errs


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


def test_four_rank_farm_oversubscribed():
    """Four oversubscribed ranks on one box (the reference's mpi test runs
    `mpirun --oversubscribe -n 4`): controller + 3 workers."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 4, 29651, q)) for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, has_best = q.get(timeout=600)
        results[rank] = has_best
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert results[0] is True
    assert all(results[r] is False for r in (1, 2, 3))
