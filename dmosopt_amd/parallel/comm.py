"""Distributed evaluation farm over torch.distributed (RCCL on ROCm).

Replaces the reference's distwq MPI task farm.
Execution model: REPLICATED control flow — every rank runs the same driver
code with identical seeds and derives the same candidate batches — so task
dispatch needs NO control messages at all. Each rank evaluates the
rank-strided shard of every batch and the results return as ONE tensor
all_gather of packed (y[,c], t) rows (+ one uint8 all_gather when feature
records ride along): a single large collective over the xGMI links instead
of per-point pickle traffic.

Backend: "cpu:gloo,cuda:nccl" when CUDA is present (nccl IS RCCL on ROCm),
plain gloo otherwise.
"""

from __future__ import annotations

import datetime
import os
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from dmosopt_amd.parallel.context import ParallelContext, get_context, set_context


def dist_is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def init_from_env(timeout_s: int = 1800) -> Tuple[int, int]:
    """Initialize the process group from torchrun env vars if present and
    install the global ParallelContext.

    Returns (rank, world_size); (0, 1) when not launched distributed.
    """
    if dist_is_initialized():
        if get_context() is None:
            _install_context()
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
    _install_context()
    return dist.get_rank(), dist.get_world_size()


def _install_context():
    device = (
        torch.device("cuda", torch.cuda.current_device())
        if torch.cuda.is_available()
        else torch.device("cpu")
    )
    set_context(ParallelContext(dist.get_rank(), dist.get_world_size(), device))


def is_controller() -> bool:
    return (not dist_is_initialized()) or dist.get_rank() == 0


@dataclass
class ResultSpec:
    """Fixed per-point result layout the tensor farm packs/unpacks.

    Derived from the driver's configuration (objective/constraint names,
    feature dtypes) so every rank agrees on the wire format without any
    negotiation round."""

    problem_ids: Tuple[int, ...]
    n_objectives: int
    n_constraints: int = 0
    feature_dtype: Optional[np.dtype] = None

    @property
    def floats_per_problem(self) -> int:
        return self.n_objectives + self.n_constraints

    @property
    def float_width(self) -> int:
        # per point: per-problem (y | c) blocks + one shared eval time
        return len(self.problem_ids) * self.floats_per_problem + 1

    @property
    def feature_itemsize(self) -> int:
        return 0 if self.feature_dtype is None else np.dtype(self.feature_dtype).itemsize

    @property
    def byte_width(self) -> int:
        return len(self.problem_ids) * self.feature_itemsize


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
    """Single-process evaluation (the world-size-1 path)."""

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


class CollectiveFarm:
    """Symmetric tensor-collective evaluation farm.

    Every rank calls :meth:`evaluate` with the IDENTICAL points list (the
    replicated-control-flow invariant). Each rank runs the user objective on
    rows [rank::world], packs each result into a fixed-layout float64 row
    (per-problem objectives, constraints, then the eval time) plus an
    optional uint8 feature-record row, and the full batch is reassembled
    from ONE float all_gather (+ one byte all_gather when features are
    configured). The returned result dicts — {pid: y | (y,f) | (y,f,c) |
    (y,c), "time": t} — are bit-identical on every rank.
    """

    def __init__(self, eval_funs: Dict[str, Callable], spec: ResultSpec):
        ctx = get_context()
        assert ctx is not None and ctx.world > 1
        self.ctx = ctx
        self.eval_funs = eval_funs
        self.spec = spec
        self._stats = _FarmStats(ctx.world)

    # ---------------------------------------------------------------- pack
    def _pack_one(self, res: Dict, frow: np.ndarray, brow: Optional[np.ndarray]):
        spec = self.spec
        col = 0
        bcol = 0
        for pid in spec.problem_ids:
            entry = res[pid]
            f = c = None
            if isinstance(entry, tuple):
                if len(entry) == 3:
                    y, f, c = entry
                elif len(entry) == 2:
                    if spec.feature_dtype is not None:
                        y, f = entry
                    else:
                        y, c = entry
                else:
                    y = entry[0]
            else:
                y = entry
            y = np.asarray(y, dtype=np.float64).ravel()
            if y.shape[0] != spec.n_objectives:
                raise ValueError(
                    f"objective returned {y.shape[0]} values for problem {pid}; "
                    f"expected {spec.n_objectives}"
                )
            frow[col : col + spec.n_objectives] = y
            col += spec.n_objectives
            if spec.n_constraints:
                cv = np.asarray(c, dtype=np.float64).ravel()
                if cv.shape[0] != spec.n_constraints:
                    raise ValueError(
                        f"objective returned {cv.shape[0]} constraints for "
                        f"problem {pid}; expected {spec.n_constraints}"
                    )
                frow[col : col + spec.n_constraints] = cv
                col += spec.n_constraints
            if spec.feature_dtype is not None:
                fa = np.asarray(f)
                raw = fa.tobytes()
                if len(raw) != spec.feature_itemsize:
                    raise ValueError(
                        "multi-rank feature transport requires one fixed-size "
                        f"record per evaluation: got {len(raw)} bytes, dtype "
                        f"itemsize is {spec.feature_itemsize} (problem {pid})"
                    )
                brow[bcol : bcol + spec.feature_itemsize] = np.frombuffer(raw, np.uint8)
                bcol += spec.feature_itemsize
        frow[col] = float(res.get("time", -1.0))

    def _unpack_one(self, frow: np.ndarray, brow: Optional[np.ndarray]) -> Dict:
        spec = self.spec
        out: Dict = {}
        col = 0
        bcol = 0
        for pid in spec.problem_ids:
            y = frow[col : col + spec.n_objectives].copy()
            col += spec.n_objectives
            c = None
            if spec.n_constraints:
                c = frow[col : col + spec.n_constraints].copy()
                col += spec.n_constraints
            f = None
            if spec.feature_dtype is not None:
                f = np.frombuffer(
                    brow[bcol : bcol + spec.feature_itemsize].tobytes(),
                    dtype=spec.feature_dtype,
                )
                bcol += spec.feature_itemsize
            if f is not None and c is not None:
                out[pid] = (y, f, c)
            elif f is not None:
                out[pid] = (y, f)
            elif c is not None:
                out[pid] = (y, c)
            else:
                out[pid] = y
        out["time"] = float(frow[-1])
        return out

    # ------------------------------------------------------------ evaluate
    def evaluate(self, opt_id: str, points: List) -> List:
        ctx, spec = self.ctx, self.spec
        fn = self.eval_funs[opt_id]
        P = len(points)
        t_call = time.time()
        # replicated-flow guard: a rank with a different batch size would
        # otherwise hang in the mismatched all_gather below — make it a
        # loud error instead (one tiny collective per BATCH, not per point)
        sizes = ctx.all_gather_stack(
            torch.tensor([P], dtype=torch.int64, device=ctx.device)
        ).flatten().tolist()
        if len(set(sizes)) != 1:
            raise RuntimeError(
                f"CollectiveFarm: request-batch sizes diverged across ranks "
                f"({sizes}); the replicated control flow is broken "
                "(unseeded RNG or rank-dependent strategy state?)"
            )
        my_idx = list(ctx.shard_indices(P))
        max_shard = ctx.max_shard_size(P)
        fbuf = np.zeros((max_shard, spec.float_width), dtype=np.float64)
        bbuf = (
            np.zeros((max_shard, spec.byte_width), dtype=np.uint8)
            if spec.byte_width
            else None
        )
        for j, i in enumerate(my_idx):
            res = fn(points[i])
            self._pack_one(res, fbuf[j], bbuf[j] if bbuf is not None else None)

        full_f = ctx.all_gather_interleaved(torch.from_numpy(fbuf), P).cpu().numpy()
        full_b = None
        if bbuf is not None:
            full_b = ctx.all_gather_interleaved(torch.from_numpy(bbuf), P).cpu().numpy()

        out: List = []
        for i in range(P):
            out.append(
                self._unpack_one(full_f[i], full_b[i] if full_b is not None else None)
            )
        # per-rank accounting from the gathered time column (identical on
        # every rank, so the stats written to H5 by rank 0 cover everyone)
        for r in range(ctx.world):
            rows = range(r, P, ctx.world)
            self._stats.record(
                r, len(rows), float(sum(max(full_f[i][-1], 0.0) for i in rows))
            )
        self._stats.call_times.append(time.time() - t_call)
        return out

    def stats(self) -> Dict:
        return self._stats.summary()

    def shutdown(self):
        pass


def make_farm(eval_funs: Dict[str, Callable], spec: Optional[ResultSpec] = None):
    if dist_is_initialized() and dist.get_world_size() > 1:
        assert spec is not None, "multi-rank farm needs a ResultSpec"
        return CollectiveFarm(eval_funs, spec)
    return LocalFarm(eval_funs)
