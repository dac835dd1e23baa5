"""Hypothesis property tests for geometry and philox invariants."""
from hypothesis import given, settings, strategies as st

import numpy as np

from oracle import geometry as og, philox
from distributedarrays_jl_amd import geometry as pg


@given(st.integers(1, 10**6), st.integers(1, 16))
@settings(max_examples=200, deadline=None)
def test_cuts_partition(sz, nc):
    cuts = pg.cuts1d(sz, nc)
    assert cuts == og.defaultdist_1d(sz, nc)
    rng = pg.ranges1d(cuts)
    assert len(rng) == nc
    total = 0
    prev_hi = 0
    for lo, hi in rng:
        assert hi >= lo
        if hi > lo:
            assert lo == prev_hi
            prev_hi = hi
        total += hi - lo
    assert total == sz
    if sz >= nc:
        lens = [hi - lo for lo, hi in rng]
        assert max(lens) - min(lens) <= 1


@given(st.lists(st.integers(1, 4096), min_size=1, max_size=3),
       st.integers(1, 16))
@settings(max_examples=200, deadline=None)
def test_defaultdist_product_bound(dims, nr):
    dist = pg.defaultdist(dims, nr)
    assert dist == og.defaultdist_dims(dims, nr)
    p = 1
    for c in dist:
        p *= c
    assert 1 <= p <= nr


@given(st.integers(0, 2**40), st.integers(1, 512),
       st.integers(0, 2**62))
@settings(max_examples=50, deadline=None)
def test_philox_offset_consistency(seed, n, off):
    a = philox.fill_uniform_f64(n, seed, offset=off)
    b = np.concatenate([philox.fill_uniform_f64(n // 2, seed, offset=off),
                        philox.fill_uniform_f64(n - n // 2, seed,
                                                offset=off + n // 2)])
    assert np.array_equal(a, b)
    assert a.min() >= 0.0 and a.max() < 1.0


@given(st.integers(0, 3000), st.integers(1, 8), st.integers(0, 2**31 - 1))
@settings(max_examples=60, deadline=None)
def test_samplesort_sim_property(n, P, seed):
    from tests.test_sort import simulate_samplesort
    x = philox.fill_uniform_f64(n, seed) if n else np.empty(0)
    got, sizes = simulate_samplesort(x, P)
    assert np.array_equal(got, np.sort(x))
    assert sum(sizes) == n and len(sizes) == P


@given(st.integers(1, 60), st.integers(1, 60),
       st.integers(1, 8), st.integers(0, 2**31 - 1))
@settings(max_examples=60, deadline=None)
def test_halo_plan_property(m, n, nr, seed):
    from distributedarrays_jl_amd import plan
    rng = np.random.default_rng(seed)
    dist = pg.defaultdist((m, n), nr)
    idxs, _ = pg.chunk_indices((m, n), dist)
    nchunks = len(idxs)
    ranks = list(range(nchunks))
    r0 = int(rng.integers(0, m)); r1 = int(rng.integers(r0, m)) + 1
    c0 = int(rng.integers(0, n)); c1 = int(rng.integers(c0, n)) + 1
    boxes = [None] * nchunks
    boxes[int(rng.integers(0, nchunks))] = ((r0, r1), (c0, c1))
    pieces = plan.halo_plan(idxs, ranks, boxes)
    cover = np.zeros((m, n), dtype=int)
    for (src, dst, b) in pieces:
        sl = tuple(slice(lo, hi) for lo, hi in b)
        cover[sl] += 1
        # every piece must lie inside its source chunk
        for (lo, hi), (clo, chi) in zip(b, idxs[src]):
            assert clo <= lo and hi <= chi
    assert (cover[r0:r1, c0:c1] == 1).all()
    assert cover.sum() == (r1 - r0) * (c1 - c0)
