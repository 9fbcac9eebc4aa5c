"""Shared event-stream variation for genetic MOEAs.

The reference NSGA2/AGEMOEA generate loop (NSGA2.py:141-177,
AGEMOEA.py:146-180): a Bernoulli event stream of crossovers (2 children)
and mutations (1 child) until popsize-1 children exist. Here the stream is
drawn on the host and ALL variation executes as two fused device launches
(Philox SBX + polynomial mutation kernels on gfx950), assembled into event
order with a single gather.
"""

from __future__ import annotations

from typing import Tuple

import numpy as np
import torch

from dmosopt_amd import ops


def event_stream_variation(
    pool: torch.Tensor,
    rng: np.random.Generator,
    popsize: int,
    poolsize: int,
    crossover_prob: float,
    mutation_prob: float,
    mutation_rate: float,
    di_crossover: torch.Tensor,
    di_mutation: torch.Tensor,
    xlb: torch.Tensor,
    xub: torch.Tensor,
    torch_random=None,
) -> Tuple[torch.Tensor, np.ndarray, np.ndarray]:
    """Returns (x_gen, crossover_slot_indices, mutation_slot_indices)."""
    cross_pairs = []
    mut_parents = []
    order = []
    count = 0
    while count < popsize - 1:
        if rng.random() < crossover_prob:
            pidx = rng.choice(poolsize, 2, replace=False)
            cross_pairs.append((int(pidx[0]), int(pidx[1])))
            order.append("c")
            count += 2
        if rng.random() < mutation_prob:
            mut_parents.append(int(rng.integers(low=0, high=poolsize)))
            order.append("m")
            count += 1

    children_c1 = children_c2 = children_m = None
    if cross_pairs:
        i1 = torch.tensor([a for a, _ in cross_pairs], dtype=torch.long, device=pool.device)
        i2 = torch.tensor([b for _, b in cross_pairs], dtype=torch.long, device=pool.device)
        children_c1, children_c2 = ops.sbx_from_pool(
            pool, i1, i2, di_crossover, xlb, xub,
            seed=int(rng.integers(0, 2**62)), generator=torch_random,
        )
    if mut_parents:
        im = torch.tensor(mut_parents, dtype=torch.long, device=pool.device)
        children_m = ops.mutation_from_pool(
            pool, im, di_mutation, xlb, xub, mutation_rate,
            seed=int(rng.integers(0, 2**62)), generator=torch_random,
        )

    C = len(cross_pairs)
    src_rows = []
    crossover_indices = []
    mutation_indices = []
    ci = mi = 0
    slot = 0
    for ev in order:
        if ev == "c":
            src_rows.extend([ci, C + ci])
            crossover_indices.extend([slot, slot + 1])
            ci += 1
            slot += 2
        else:
            src_rows.append(2 * C + mi)
            mutation_indices.append(slot)
            mi += 1
            slot += 1
    parts = [t for t in (children_c1, children_c2, children_m) if t is not None]
    src = torch.cat(parts, dim=0)
    gather_idx = torch.tensor(src_rows, dtype=torch.long, device=src.device)
    return (
        src[gather_idx],
        np.asarray(crossover_indices, dtype=int),
        np.asarray(mutation_indices, dtype=int),
    )
