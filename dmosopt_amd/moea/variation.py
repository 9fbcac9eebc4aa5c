"""Shared event-stream variation for genetic MOEAs.

The reference NSGA2/AGEMOEA generate loop (NSGA2.py:141-177,
AGEMOEA.py:146-180): a Bernoulli event stream of crossovers (2 children)
and mutations (1 child) until popsize-1 children exist. The stream here is
drawn VECTORIZED on the host (one `rng.random((n, 2))` draw instead of a
per-event Python loop), all index bookkeeping is assembled with numpy, a
single pinned H2D transfer carries every index array, and on GPU ALL
variation (SBX + mutation + event-order placement) executes as ONE
event-decoded gfx950 launch — each event scatters its own child rows, so
the host never builds an inverse slot map (CPU fallback: split Philox
SBX/mutation kernels + row scatter, bitwise-identical output). The
generation loop is HOST-dispatch-bound (rocprof: GPU 33% busy inside the
loop window), so host numpy work here is sized in single-digit
microseconds: one uniform draw covers all parent indices. Pair selection is
uniform over ordered distinct pairs (statistically equivalent to the
reference's `choice(poolsize, 2, replace=False)`).
"""

from __future__ import annotations

from typing import Tuple

import numpy as np
import torch

from dmosopt_amd import ops

def _to_device_pinned(arr: np.ndarray, device: torch.device) -> torch.Tensor:
    if device.type != "cuda":
        return torch.as_tensor(arr, dtype=torch.long, device=device)
    # torch's caching host allocator recycles the pinned block only after
    # the async copy's stream event completes, so a fresh pin_memory tensor
    # per call is both cheap and safe even when the host runs generations
    # ahead of the GPU (a manual reused buffer would race)
    buf = torch.empty(arr.shape[0], dtype=torch.long, pin_memory=True)
    buf.copy_(torch.from_numpy(arr))
    out = torch.empty(arr.shape[0], dtype=torch.long, device=device)
    out.copy_(buf, non_blocking=True)
    return out


def _draw_event_stream(rng, popsize: int, crossover_prob: float, mutation_prob: float):
    """Vectorized replica of the reference's while-loop event stream.

    Each iteration draws (u_c, u_m); a crossover event adds 2 children, a
    mutation event adds 1; the loop body runs while count < popsize - 1 at
    iteration start. Returns boolean arrays (c_ev, m_ev) over the executed
    iterations.
    """
    target = popsize - 1
    c_parts, m_parts = [], []
    cum = 0
    while True:
        chunk = max(64, popsize)
        u = rng.random((chunk, 2))
        c = u[:, 0] < crossover_prob
        m = u[:, 1] < mutation_prob
        inc = 2 * c.astype(np.int64) + m.astype(np.int64)
        cs = cum + np.cumsum(inc)
        j0 = int(np.searchsorted(cs, target, side="left"))
        if j0 < chunk:  # iteration j0 is the first to START with count >= target
            c_parts.append(c[: j0 + 1])
            m_parts.append(m[: j0 + 1])
            break
        c_parts.append(c)
        m_parts.append(m)
        cum = int(cs[-1])
    return np.concatenate(c_parts), np.concatenate(m_parts)


def _draw_event_stream_tail(rng, remaining_target, crossover_prob, mutation_prob):
    """Continue an under-shot event stream until `remaining_target` more
    children exist (rare batched-prefetch overflow path)."""
    if remaining_target <= 0:
        return (np.zeros(0, dtype=bool), np.zeros(0, dtype=bool))
    c_parts, m_parts = [], []
    cum = 0
    while True:
        u = rng.random((64, 2))
        c = u[:, 0] < crossover_prob
        m = u[:, 1] < mutation_prob
        cs = cum + np.cumsum(2 * c.astype(np.int64) + m.astype(np.int64))
        j0 = int(np.searchsorted(cs, remaining_target, side="left"))
        if j0 < 64:
            c_parts.append(c[: j0 + 1])
            m_parts.append(m[: j0 + 1])
            break
        c_parts.append(c)
        m_parts.append(m)
        cum = int(cs[-1])
    return np.concatenate(c_parts), np.concatenate(m_parts)


def _assemble_event_indices(rng, popsize, poolsize, crossover_prob, mutation_prob):
    """Draw + assemble the generation's event indices (host numpy side).

    Returns (combined int64 array [i1|i2|im|ci|mi], C, M, seed_sbx, seed_mut).
    """
    c_ev, m_ev = _draw_event_stream(rng, popsize, crossover_prob, mutation_prob)
    C = int(c_ev.sum())
    M = int(m_ev.sum())
    u = rng.random(2 * C + M)
    if C:
        i1 = (u[:C] * poolsize).astype(np.int64)
        i2 = (u[C : 2 * C] * (poolsize - 1)).astype(np.int64)
        i2 = i2 + (i2 >= i1)
    else:
        i1 = i2 = np.empty(0, dtype=np.int64)
    im = (u[2 * C :] * poolsize).astype(np.int64) if M else np.empty(0, dtype=np.int64)
    seed_sbx, seed_mut = (int(s) for s in rng.integers(0, 2**62, 2))
    it_sizes = 2 * c_ev + m_ev
    it_starts = np.cumsum(it_sizes) - it_sizes
    crossover_indices = np.repeat(it_starts[c_ev], 2)
    crossover_indices[1::2] += 1
    mutation_indices = (it_starts + 2 * c_ev)[m_ev]
    combined = np.concatenate([i1, i2, im, crossover_indices, mutation_indices])
    return combined, C, M, seed_sbx, seed_mut


class _SpawnPrefetch:
    """Chunked prefetch of the per-generation host randomness.

    The event stream, parent indices and slot layout depend ONLY on the
    rng (never on results), so drawing CHUNK generations ahead — in the
    SAME rng call order the sequential loop would use — yields a
    bit-identical stream while amortizing the numpy assembly and the slow
    pinned H2D (~1 GB/s on this platform, profiles/README.md) over one
    transfer instead of CHUNK."""

    CHUNK = 16

    def __init__(self):
        self.key = None
        self.items = []

    def refill(self, rng, popsize, poolsize, pc, pm, device):
        """Batched assembly: each numpy Generator call costs ~6 us of host
        time and each numpy op a few more, so the CHUNK generations' draws
        are made with ONE call per component (event uniforms, parent
        uniforms, seeds) and the per-generation work is reduced to the
        ragged slot-layout math."""
        G = self.CHUNK
        target = popsize - 1
        chunk = max(64, popsize)
        U = rng.random((G, chunk, 2))
        c_all = U[:, :, 0] < pc
        m_all = U[:, :, 1] < pm
        inc = 2 * c_all + m_all
        cs = np.cumsum(inc, axis=1)
        reached = cs >= target
        ok = reached[:, -1]
        j0 = np.argmax(reached, axis=1)  # first iteration reaching target
        events = []
        for g in range(G):
            if ok[g]:
                events.append((c_all[g, : j0[g] + 1], m_all[g, : j0[g] + 1]))
            else:
                # rare overflow (the G*chunk draw under-shot): finish this
                # generation's stream with the sequential path
                extra_c, extra_m = _draw_event_stream_tail(
                    rng, target - int(cs[g, -1]), pc, pm
                )
                events.append((np.concatenate([c_all[g], extra_c]),
                               np.concatenate([m_all[g], extra_m])))
        # pad the ragged event streams to (G, Lmax) and do ALL remaining
        # index math with flat masks — zero per-generation numpy ops
        # (row-major flattening preserves generation order)
        Lmax = max(len(c) for c, _ in events)
        c_pad = np.zeros((G, Lmax), dtype=bool)
        m_pad = np.zeros((G, Lmax), dtype=bool)
        for g, (c_ev, m_ev) in enumerate(events):
            c_pad[g, : len(c_ev)] = c_ev
            m_pad[g, : len(m_ev)] = m_ev
        Cs = c_pad.sum(axis=1)
        Ms = m_pad.sum(axis=1)
        nC, nM = int(Cs.sum()), int(Ms.sum())
        u_par = rng.random(2 * nC + nM)
        seeds = rng.integers(0, 2**62, (G, 3))
        i1 = (u_par[:nC] * poolsize).astype(np.int64)
        i2 = (u_par[nC : 2 * nC] * (poolsize - 1)).astype(np.int64)
        i2 = i2 + (i2 >= i1)
        im = (u_par[2 * nC :] * poolsize).astype(np.int64)
        it_sizes = 2 * c_pad + m_pad
        it_starts = np.cumsum(it_sizes, axis=1) - it_sizes
        cstart = it_starts[c_pad]              # all gens' crossover starts
        ci = np.repeat(cstart, 2)
        ci[1::2] += 1
        mi = (it_starts + 2 * c_pad)[m_pad]    # all gens' mutation slots
        dev = _to_device_pinned(
            np.concatenate([i1, i2, im, ci, mi]), device
        )
        # component group offsets within dev
        o_i2 = nC
        o_im = 2 * nC
        o_ci = 2 * nC + nM
        o_mi = o_ci + 2 * nC
        coff = np.concatenate([[0], np.cumsum(Cs)])
        moff = np.concatenate([[0], np.cumsum(Ms)])
        items = []
        for g in range(G):
            C0, C1 = int(coff[g]), int(coff[g + 1])
            M0, M1 = int(moff[g]), int(moff[g + 1])
            items.append((
                int(seeds[g, 0]), C1 - C0, M1 - M0, int(seeds[g, 1]),
                int(seeds[g, 2]),
                (dev[C0:C1], dev[o_i2 + C0 : o_i2 + C1],
                 dev[o_im + M0 : o_im + M1],
                 dev[o_ci + 2 * C0 : o_ci + 2 * C1],
                 dev[o_mi + M0 : o_mi + M1]),
            ))
        items.reverse()  # pop() yields generation order
        self.items = items

    def next(self, rng, popsize, poolsize, pc, pm, device):
        key = (popsize, poolsize, pc, pm, str(device))
        if key != self.key or not self.items:
            self.key = key
            self.refill(rng, popsize, poolsize, pc, pm, device)
        return self.items.pop()


def spawn_generation_native(
    population: torch.Tensor,
    rank: torch.Tensor,
    poolsize: int,
    p_sel: float,
    rng: np.random.Generator,
    popsize: int,
    crossover_prob: float,
    mutation_prob: float,
    mutation_rate: float,
    di_crossover: torch.Tensor,
    di_mutation: torch.Tensor,
    xlb: torch.Tensor,
    xub: torch.Tensor,
    prefetch: "_SpawnPrefetch" = None,
):
    """Tournament selection + whole-generation variation in ONE extension
    call (generation_spawn binding): the host-dispatch-bound loop pays one
    binding round trip instead of two, and the pool tensor stays in C++.
    Returns (x_gen, crossover_slot_idx, mutation_slot_idx) or None when the
    tournament kernel refuses the shape (caller uses the split path)."""
    from dmosopt_amd import _hipops

    if prefetch is not None:
        seed_t, C, M, seed_sbx, seed_mut, views = prefetch.next(
            rng, popsize, poolsize, crossover_prob, mutation_prob,
            population.device,
        )
        i1_t, i2_t, im_t, c_idx_t, m_idx_t = views
    else:
        seed_t = int(rng.integers(0, 2**62))
        combined, C, M, seed_sbx, seed_mut = _assemble_event_indices(
            rng, popsize, poolsize, crossover_prob, mutation_prob
        )
        dev = _to_device_pinned(combined, population.device)
        o = 0
        i1_t = dev[o : o + C]; o += C
        i2_t = dev[o : o + C]; o += C
        im_t = dev[o : o + M]; o += M
        c_idx_t = dev[o : o + 2 * C]; o += 2 * C
        m_idx_t = dev[o : o + M]
    x_gen = _hipops.generation_spawn(
        population.float().contiguous(), rank.long().contiguous(), poolsize,
        float(p_sel), seed_t, c_idx_t, m_idx_t, i1_t, i2_t, im_t,
        di_crossover.float().contiguous(), di_mutation.float().contiguous(),
        xlb.float().contiguous(), xub.float().contiguous(),
        float(mutation_rate), seed_sbx, seed_mut,
        rank_sorted=True,  # NSGA2 state.rank comes out of nsga2_select sorted
    )
    if x_gen is None:
        return None
    return x_gen.to(population.dtype), c_idx_t, m_idx_t


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
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (x_gen, crossover_slot_indices, mutation_slot_indices).

    The slot-index outputs are int64 tensors on ``pool.device`` so callers
    can track operator success without a device->host sync.
    """
    # ONE uniform draw covers every parent index; slot layout in iteration
    # order (a crossover's two children precede the same iteration's
    # mutation child, matching the reference loop body). Single H2D
    # transfer for every index array, staged through a PINNED buffer with a
    # non-blocking copy — a pageable torch.as_tensor(...) H2D blocks the
    # host until the stream drains.
    combined, C, M, seed_sbx, seed_mut = _assemble_event_indices(
        rng, popsize, poolsize, crossover_prob, mutation_prob
    )
    total = 2 * C + M
    dev = _to_device_pinned(combined, pool.device)
    o = 0
    i1_t = dev[o : o + C]; o += C
    i2_t = dev[o : o + C]; o += C
    im_t = dev[o : o + M]; o += M
    c_idx_t = dev[o : o + 2 * C]; o += 2 * C
    m_idx_t = dev[o : o + M]

    if pool.device.type == "cuda" and ops.native_available():
        # whole-generation variation in ONE event-decoded launch: each event
        # scatters its own child rows, so no host-built src_rows inverse map
        # (bitwise identical values to the split sbx/mutation kernels)
        from dmosopt_amd import _hipops

        x_gen = _hipops.variation_events(
            pool.float().contiguous(), c_idx_t, m_idx_t, i1_t, i2_t, im_t,
            di_crossover.float().contiguous(), di_mutation.float().contiguous(),
            xlb.float().contiguous(), xub.float().contiguous(),
            float(mutation_rate), seed_sbx, seed_mut,
        )
        return x_gen.to(pool.dtype), c_idx_t, m_idx_t

    out = torch.empty((total, pool.shape[1]), dtype=pool.dtype, device=pool.device)
    if C:
        c1, c2 = ops.sbx_from_pool(
            pool, i1_t, i2_t, di_crossover, xlb, xub,
            seed=seed_sbx, generator=torch_random,
        )
        out[c_idx_t[0::2]] = c1
        out[c_idx_t[1::2]] = c2
    if M:
        out[m_idx_t] = ops.mutation_from_pool(
            pool, im_t, di_mutation, xlb, xub, mutation_rate,
            seed=seed_mut, generator=torch_random,
        )
    return out, c_idx_t, m_idx_t
