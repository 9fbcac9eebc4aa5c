"""Distributed evaluation farm over torch.distributed (RCCL on ROCm).

Replaces the reference's distwq MPI task farm: one
process per GPU, rank 0 is the controller/driver. Instead of point-to-point
task dispatch, candidate batches are BROADCAST to all ranks, each rank
evaluates an even shard (rank strided), and results return via a single
gather — the collective pattern sized for xGMI (few large messages over the
7 p2p links, not thousands of small sends).

Backend: "cpu:gloo,cuda:nccl" when CUDA is present (nccl IS RCCL on ROCm),
plain gloo otherwise. Control-plane objects ride the gloo lane; bulk tensors
ride RCCL.
"""

from __future__ import annotations

import datetime
import os
import time
from typing import Callable, Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist


def dist_is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def init_from_env(timeout_s: int = 1800) -> Tuple[int, int]:
    """Initialize the process group from torchrun env vars if present.

    Returns (rank, world_size); (0, 1) when not launched distributed.
    """
    if dist_is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 1
    world_size = int(os.environ["WORLD_SIZE"])
    if world_size <= 1:
        return 0, 1
    if torch.cuda.is_available():
        local_rank = int(os.environ.get("LOCAL_RANK", os.environ["RANK"]))
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        backend = "cpu:gloo,cuda:nccl"
    else:
        backend = "gloo"
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    return dist.get_rank(), dist.get_world_size()


def is_controller() -> bool:
    return (not dist_is_initialized()) or dist.get_rank() == 0


class _FarmStats:
    """Per-rank evaluation accounting (role of the reference's distwq
    controller stats, dmosopt.py:855-882)."""

    def __init__(self, world: int):
        self.world = world
        self.n_processed = np.zeros(world, dtype=np.int64)
        self.total_time = np.zeros(world, dtype=np.float64)
        self.call_times: List[float] = []

    def record(self, rank: int, n: int, t: float):
        self.n_processed[rank] += n
        self.total_time[rank] += t

    def summary(self) -> Dict:
        out = {
            "results_collected": int(self.n_processed.sum()),
            "total_evaluation_time": float(self.total_time.sum()),
        }
        if self.call_times:
            ct = np.asarray(self.call_times)
            out.update(
                mean_time_per_call=float(ct.mean()),
                stdev_time_per_call=float(ct.std()),
            )
        if self.world > 1:
            out.update(
                mean_calls_per_worker=float(self.n_processed.mean()),
                stdev_calls_per_worker=float(self.n_processed.std()),
                min_calls_per_worker=int(self.n_processed.min()),
                max_calls_per_worker=int(self.n_processed.max()),
                mean_time_per_worker=float(self.total_time.mean()),
                stdev_time_per_worker=float(self.total_time.std()),
            )
        return out


class LocalFarm:
    """Single-process evaluation (controller evaluates everything)."""

    def __init__(self, eval_funs: Dict[str, Callable]):
        self.eval_funs = eval_funs
        self._stats = _FarmStats(1)

    def evaluate(self, opt_id: str, points: List) -> List:
        """points: list of per-request eval_fun arguments (space-vals dicts)."""
        fn = self.eval_funs[opt_id]
        t0 = time.time()
        out = [fn(p) for p in points]
        self._stats.record(0, len(points), time.time() - t0)
        self._stats.call_times.append(time.time() - t0)
        return out

    def stats(self) -> Dict:
        return self._stats.summary()

    def shutdown(self):
        pass


class TorchDistFarm:
    """Collective evaluation farm: rank 0 drives, every rank evaluates.

    Protocol per batch (all ranks participate):
      1. broadcast_object_list([("eval", opt_id, x_batch)])  (gloo lane)
      2. each rank r evaluates rows r, r+W, r+2W, ...
      3. gather_object(shard_results) to rank 0
    Rank 0 calls evaluate(); other ranks sit in worker_loop() until the
    controller broadcasts ("stop",).
    """

    def __init__(self, eval_funs: Dict[str, Callable]):
        assert dist_is_initialized()
        self.eval_funs = eval_funs
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self._stats = _FarmStats(self.world)

    # ------------------------------------------------------------ controller
    def evaluate(self, opt_id: str, points: List) -> List:
        assert self.rank == 0
        t_call = time.time()
        cmd = [("eval", opt_id, points)]
        dist.broadcast_object_list(cmd, src=0)
        my_results = self._eval_shard(opt_id, points)
        gathered: List = [None] * self.world
        dist.gather_object(my_results, gathered, dst=0)
        # interleave shards back into original order
        out: List = [None] * len(points)
        for r, shard in enumerate(gathered):
            shard_t = 0.0
            for j, res in enumerate(shard):
                out[r + j * self.world] = res
                if isinstance(res, dict) and "time" in res:
                    shard_t += res["time"]
            self._stats.record(r, len(shard), shard_t)
        self._stats.call_times.append(time.time() - t_call)
        return out

    def stats(self) -> Dict:
        return self._stats.summary()

    def shutdown(self):
        if self.rank == 0:
            dist.broadcast_object_list([("stop",)], src=0)

    # --------------------------------------------------------------- worker
    def worker_loop(self):
        assert self.rank != 0
        while True:
            cmd = [None]
            dist.broadcast_object_list(cmd, src=0)
            tag = cmd[0][0]
            if tag == "stop":
                break
            if tag == "eval":
                _, opt_id, points = cmd[0]
                my_results = self._eval_shard(opt_id, points)
                dist.gather_object(my_results, None, dst=0)

    # --------------------------------------------------------------- shared
    def _eval_shard(self, opt_id: str, points: List) -> List:
        fn = self.eval_funs[opt_id]
        return [fn(points[i]) for i in range(self.rank, len(points), self.world)]


def make_farm(eval_funs: Dict[str, Callable]):
    if dist_is_initialized() and dist.get_world_size() > 1:
        return TorchDistFarm(eval_funs)
    return LocalFarm(eval_funs)
