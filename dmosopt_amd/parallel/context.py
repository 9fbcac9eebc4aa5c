"""Parallel execution context: one process per GPU, RCCL over xGMI.

The multi-rank execution model (replacing the reference's distwq MPI task
farm) is REPLICATED CONTROL FLOW + SHARDED DATA
PLANE: every rank runs the same driver/strategy/engine code with identical
seeds, so every rank independently derives the same candidate batches; the
expensive data movement — objective evaluation results, surrogate
predictions, GP hyperparameters — travels as tensor collectives:

  * evaluated (x, y[,c,f], t) blocks: packed rows, ONE all_gather per batch
  * surrogate posterior queries: rank-strided shards, ONE all_gather per
    generation (parallel/sharded.py)
  * GP hyperparameters: fit on rank 0, ONE broadcast per epoch (theta)

On a GPU box the collectives ride RCCL (torch.distributed "nccl" backend IS
RCCL on ROCm) over the 7 xGMI p2p links; on CPU they ride gloo, which is
what the world>1 CPU tests exercise.

Determinism: replicated control flow requires every rank to make identical
decisions. All randomness flows from the seeded per-problem RNG; the
collectives themselves are deterministic (rank-ordered reassembly). The
context's ``assert_synchronized`` hash guard turns any silent divergence
into a loud error (used by the driver each epoch, and per-generation in
tests).
"""

from __future__ import annotations

import hashlib
from typing import List, Optional

import numpy as np
import torch
import torch.distributed as dist


class ParallelContext:
    """Rank/world handle plus the small collective vocabulary the framework
    uses. Constructed by ``comm.init_from_env`` when WORLD_SIZE > 1."""

    def __init__(self, rank: int, world: int, device: torch.device):
        self.rank = rank
        self.world = world
        self.device = device

    # ------------------------------------------------------------ helpers
    @property
    def is_root(self) -> bool:
        return self.rank == 0

    def barrier(self):
        dist.barrier()

    def shard_indices(self, n: int) -> range:
        """Rank-strided shard of n items (row i belongs to rank i % world)."""
        return range(self.rank, n, self.world)

    def shard_size(self, n: int) -> int:
        return len(self.shard_indices(n))

    def max_shard_size(self, n: int) -> int:
        return (n + self.world - 1) // self.world

    # --------------------------------------------------------- collectives
    def bcast_tensor(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        """Broadcast a tensor (shape/dtype must already agree on all ranks)."""
        t = t.to(self.device)
        dist.broadcast(t, src=src)
        return t

    def bcast_payload(self, t: Optional[torch.Tensor], src: int = 0) -> torch.Tensor:
        """Broadcast a tensor whose shape only ``src`` knows.

        Two collectives: an i64 shape header (rank 8 max), then the payload
        as float64. Non-src ranks pass None. A header with nd == -2 is the
        POISON value sent by :meth:`bcast_poison` when the source failed to
        produce the payload — receivers raise instead of hanging.
        """
        header = torch.full((9,), -1, dtype=torch.int64, device=self.device)
        if self.rank == src:
            assert t is not None and t.dim() <= 8
            header[0] = t.dim()
            for i, s in enumerate(t.shape):
                header[1 + i] = s
        dist.broadcast(header, src=src)
        nd = int(header[0])
        if nd == -2:
            raise RuntimeError(
                f"bcast_payload: rank {src} signalled failure while producing "
                "the broadcast payload (see its traceback)"
            )
        shape = [int(header[1 + i]) for i in range(nd)]
        if self.rank == src:
            payload = t.to(self.device, torch.float64).contiguous()
        else:
            payload = torch.empty(shape, dtype=torch.float64, device=self.device)
        dist.broadcast(payload, src=src)
        return payload

    def bcast_poison(self, src: int = 0):
        """Tell bcast_payload receivers the payload will never come (the
        producing rank hit an exception): they raise instead of hanging."""
        header = torch.full((9,), -2, dtype=torch.int64, device=self.device)
        dist.broadcast(header, src=src)

    def bcast_flag(self, value: bool, src: int = 0) -> bool:
        return bool(self.bcast_int(1 if value else 0, src=src))

    def bcast_int(self, value: int, src: int = 0) -> int:
        t = torch.tensor([int(value)], dtype=torch.int64, device=self.device)
        dist.broadcast(t, src=src)
        return int(t.item())

    def all_gather_stack(self, t: torch.Tensor) -> torch.Tensor:
        """All-gather equal-shaped per-rank tensors -> (world, *shape)."""
        t = t.to(self.device).contiguous()
        out = [torch.empty_like(t) for _ in range(self.world)]
        dist.all_gather(out, t)
        return torch.stack(out, dim=0)

    def all_gather_interleaved(self, shard: torch.Tensor, n_total: int) -> torch.Tensor:
        """Reassemble rank-strided shards into original row order.

        Every rank passes its (padded) shard of shape (max_shard, ...); rows
        were taken as [rank::world]. Returns the (n_total, ...) tensor,
        identical on every rank.
        """
        g = self.all_gather_stack(shard)  # (world, max_shard, ...)
        # interleave: row j of rank r is original row r + j*world
        full = g.transpose(0, 1).reshape(-1, *shard.shape[1:])
        return full[:n_total]

    # ------------------------------------------------------ divergence guard
    def assert_synchronized(self, arrays, tag: str = ""):
        """Raise if the given arrays are not bit-identical on all ranks.

        Cheap guard for the replicated-control-flow invariant: hashes the
        byte content host-side, all-gathers the 64-bit digests over the
        gloo/RCCL lane and compares. Call once per epoch (driver) or per
        generation (tests / debugging).
        """
        h = hashlib.blake2b(digest_size=8)
        if not isinstance(arrays, (list, tuple)):
            arrays = [arrays]
        for a in arrays:
            if a is None:
                h.update(b"\x00none")
                continue
            if isinstance(a, torch.Tensor):
                a = a.detach().cpu().numpy()
            a = np.ascontiguousarray(a)
            h.update(str(a.dtype).encode())
            h.update(str(a.shape).encode())
            h.update(a.tobytes())
        digest = int.from_bytes(h.digest(), "little", signed=True)
        t = torch.tensor([digest], dtype=torch.int64, device=self.device)
        g = self.all_gather_stack(t).flatten().tolist()
        if len(set(g)) != 1:
            raise RuntimeError(
                f"replicated state diverged across ranks (tag={tag!r}): "
                f"digests {g}. A nondeterministic operation has broken the "
                "replicated-control-flow invariant."
            )


_context: Optional[ParallelContext] = None


def set_context(ctx: Optional[ParallelContext]):
    global _context
    _context = ctx


def get_context() -> Optional[ParallelContext]:
    return _context
