"""Rank-sharded surrogate prediction (the per-generation data plane).

Inner-loop surrogate queries arrive replicated on every rank (replicated
MOEA control flow); each rank predicts only its strided shard on its own
GPU and the full result is reassembled with ONE all_gather per generation —
a single large collective over xGMI instead of redundant full-batch
predicts on every rank (replaces the reference's
controller-local surrogate evaluation at MOASMO.py:61-64,114).

The reassembled tensor is identical on every rank by construction
(rank-ordered interleave), which preserves the replicated-control-flow
invariant regardless of floating-point reduction order differences between
a sharded and a full-batch predict.
"""

from __future__ import annotations

import numpy as np
import torch

from dmosopt_amd.parallel.context import ParallelContext


class ShardedObjective:
    """Wrap a fitted surrogate's evaluate()/evaluate_tensor() so each rank
    predicts rows [rank::world] and the result is all-gathered.

    Works for any surrogate exposing ``evaluate`` (numpy in/out, mean or
    (mean, var)); surrogates with a device-resident ``evaluate_tensor``
    (the GP fast path) keep it on the sharded route too.
    """

    def __init__(self, inner, ctx: ParallelContext):
        self._inner = inner
        self._ctx = ctx

    def __getattr__(self, name):
        # delegate everything else (return_mean_variance, predict, ...)
        return getattr(self._inner, name)

    # ---------------------------------------------------------------- torch
    def evaluate_tensor(self, x: torch.Tensor) -> torch.Tensor:
        ctx = self._ctx
        P = int(x.shape[0])
        pad = (ctx.world - P % ctx.world) % ctx.world
        xp = torch.cat([x, x[-1:].expand(pad, -1)], dim=0) if pad else x
        shard = xp[ctx.rank :: ctx.world]
        inner = self._inner
        if hasattr(inner, "evaluate_tensor"):
            m = inner.evaluate_tensor(shard)
        else:
            m = torch.as_tensor(
                np.asarray(inner.evaluate(shard.detach().cpu().numpy())),
                dtype=torch.float32,
            )
        full = ctx.all_gather_interleaved(m.to(torch.float32).contiguous(), P)
        return full.to(dtype=x.dtype, device=x.device)

    # ---------------------------------------------------------------- numpy
    def evaluate(self, x):
        ctx = self._ctx
        x = np.asarray(x, dtype=np.float64)
        if x.ndim == 1:
            x = x.reshape(1, -1)
        P = x.shape[0]
        pad = (ctx.world - P % ctx.world) % ctx.world
        xp = np.vstack([x, np.repeat(x[-1:], pad, axis=0)]) if pad else x
        shard = xp[ctx.rank :: ctx.world]
        out = self._inner.evaluate(shard)
        if isinstance(out, tuple):  # (mean, var) in mean-variance mode
            mean, var = out
            payload = np.hstack([np.asarray(mean), np.asarray(var)])
            t = torch.as_tensor(payload, dtype=torch.float64)
            full = ctx.all_gather_interleaved(t, P).cpu().numpy()
            m = full.shape[1] // 2
            return full[:, :m], full[:, m:]
        t = torch.as_tensor(np.asarray(out, dtype=np.float64))
        return ctx.all_gather_interleaved(t, P).cpu().numpy()
