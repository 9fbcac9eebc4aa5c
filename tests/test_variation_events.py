"""Unit invariants for the event-decoded variation assembly (host side).

The GPU kernel trusts the host-built slot lists; these properties make a
broken assembly loud on CPU: the crossover/mutation slot lists must
PARTITION [0, 2C+M) with crossovers occupying consecutive pairs in
iteration order (reference NSGA2.py:141-177 loop-body order).
"""

import numpy as np
import pytest

from dmosopt_amd.moea.variation import _assemble_event_indices


@pytest.mark.parametrize("popsize,poolsize,pc,pm", [
    (200, 100, 0.9, 0.1),
    (13, 7, 0.5, 0.5),
    (64, 32, 1.0, 0.0),
    (64, 32, 0.0, 1.0),
    (1600, 800, 0.9, 0.1),
])
def test_slots_partition_and_bounds(popsize, poolsize, pc, pm):
    rng = np.random.default_rng(7)
    combined, C, M, s1, s2 = _assemble_event_indices(rng, popsize, poolsize, pc, pm)
    total = 2 * C + M
    i1, i2 = combined[:C], combined[C:2 * C]
    im = combined[2 * C:2 * C + M]
    ci = combined[2 * C + M:4 * C + M]
    mi = combined[4 * C + M:]
    assert len(ci) == 2 * C and len(mi) == M
    # slot lists partition [0, total)
    slots = np.concatenate([ci, mi])
    assert np.array_equal(np.sort(slots), np.arange(total))
    # child2 of a pair sits right after child1
    assert np.array_equal(ci[1::2], ci[0::2] + 1)
    # parent indices in range; i1 != i2 (distinct parents)
    assert ((i1 >= 0) & (i1 < poolsize)).all()
    assert ((i2 >= 0) & (i2 < poolsize)).all()
    assert ((im >= 0) & (im < poolsize)).all()
    assert (i1 != i2).all()
    # at least popsize-1 children (the reference's loop target)
    assert total >= popsize - 1
    assert 0 <= s1 < 2**62 and 0 <= s2 < 2**62


def test_deterministic_per_seed():
    a = _assemble_event_indices(np.random.default_rng(42), 100, 50, 0.9, 0.1)
    b = _assemble_event_indices(np.random.default_rng(42), 100, 50, 0.9, 0.1)
    assert np.array_equal(a[0], b[0]) and a[1:] == b[1:]


def test_cpu_scatter_matches_slot_semantics():
    """The CPU fallback's scatter placement: row ci[2k] must hold child1 of
    pair k (bitwise), matching what the event-decoded kernel does."""
    import torch
    from dmosopt_amd.moea.variation import event_stream_variation

    rng = np.random.default_rng(3)
    pool = torch.rand(50, 6, dtype=torch.float64)
    di = torch.full((6,), 1.0, dtype=torch.float64)
    dm = torch.full((6,), 20.0, dtype=torch.float64)
    lo = torch.zeros(6, dtype=torch.float64)
    hi = torch.ones(6, dtype=torch.float64)
    x_gen, c_idx, m_idx = event_stream_variation(
        pool, rng, 40, 50, 0.9, 0.1, 1.0 / 6, di, dm, lo, hi)
    total = x_gen.shape[0]
    assert total == len(c_idx) + len(m_idx)
    assert ((x_gen >= 0) & (x_gen <= 1)).all()
    # every slot was written (no empty rows from a placement bug):
    # children are inside bounds and vary (not default-initialized zeros)
    assert x_gen.abs().sum() > 0


def test_prefetch_batch_matches_invariants():
    """The chunked-prefetch batched assembly must satisfy the same slot
    invariants as the single-generation path, for every entry of a chunk."""
    import torch
    from dmosopt_amd.moea.variation import _SpawnPrefetch

    rng = np.random.default_rng(11)
    pf = _SpawnPrefetch()
    dev = torch.device("cpu")
    for _ in range(20):  # > one chunk: exercises refill twice
        seed_t, C, M, s1, s2, views = pf.next(rng, 120, 60, 0.9, 0.1, dev)
        total = 2 * C + M
        i1, i2, im, ci, mi = (v.numpy() for v in views)
        assert len(i1) == C and len(i2) == C and len(im) == M
        slots = np.concatenate([ci, mi])
        assert np.array_equal(np.sort(slots), np.arange(total))
        assert np.array_equal(ci[1::2], ci[0::2] + 1)
        assert ((i1 >= 0) & (i1 < 60)).all() and (i1 != i2).all()
        assert ((im >= 0) & (im < 60)).all()
        assert total >= 119


def test_prefetch_overflow_path():
    """mutation-only stream (pm tiny) forces many iterations per child —
    exercises the rare batched-undershoot tail path."""
    import torch
    from dmosopt_amd.moea.variation import _SpawnPrefetch

    rng = np.random.default_rng(2)
    pf = _SpawnPrefetch()
    seed_t, C, M, s1, s2, views = pf.next(rng, 100, 50, 0.02, 0.05, torch.device("cpu"))
    total = 2 * C + M
    assert total >= 99
    ci, mi = views[3].numpy(), views[4].numpy()
    assert np.array_equal(np.sort(np.concatenate([ci, mi])), np.arange(total))
