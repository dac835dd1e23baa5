"""Product geometry vs oracle restatement (independent implementations
of darray.jl:251-307) over many shapes, plus reference pins."""
import numpy as np

from distributedarrays_jl_amd import geometry as pg
from oracle import geometry as og


def test_defaultdist_matches_oracle():
    shapes = [(50,), (100,), (7,), (1,), (2**28,), (100, 100),
              (16384, 16384), (32768, 8192), (50, 60, 70), (3, 1000)]
    for dims in shapes:
        for nr in (1, 2, 3, 4, 5, 6, 7, 8, 12, 16):
            assert pg.defaultdist(dims, nr) == og.defaultdist_dims(dims, nr), \
                (dims, nr)


def test_cuts_match_oracle():
    for sz in (1, 2, 3, 7, 50, 100, 1000, 2**20):
        for nc in (1, 2, 3, 4, 7, 8):
            assert pg.cuts1d(sz, nc) == og.defaultdist_1d(sz, nc)


def test_reference_pin():
    assert pg.cuts1d(50, 4) == [1, 14, 27, 39, 51]  # test/darray.jl:66


def test_chunk_indices_match_oracle():
    for dims in [(50,), (100, 64), (33, 17), (8, 8, 8)]:
        for nr in (1, 2, 4, 8):
            dist = pg.defaultdist(dims, nr)
            pi, pc = pg.chunk_indices(dims, dist)
            oi, oc = og.chunk_idxs(dims, dist)
            assert pi == oi and pc == oc


def test_grid_pos_roundtrip():
    dist = (2, 4)
    for r in range(8):
        assert pg.grid_rank(pg.grid_pos(r, dist), dist) == r


def test_locate():
    # darray.jl:448-456 semantics, 0-based
    cuts = [pg.cuts1d(50, 4)]
    assert pg.locate(cuts, (0,)) == (0,)
    assert pg.locate(cuts, (12,)) == (0,)
    assert pg.locate(cuts, (13,)) == (1,)
    assert pg.locate(cuts, (49,)) == (3,)


def test_coverage_partition():
    rng = np.random.default_rng(0)
    for _ in range(20):
        dims = tuple(int(rng.integers(1, 40)) for _ in range(2))
        nr = int(rng.integers(1, 9))
        dist = pg.defaultdist(dims, nr)
        idxs, _ = pg.chunk_indices(dims, dist)
        grid = np.zeros(dims, dtype=int)
        for idx in idxs:
            sl = tuple(slice(lo, hi) for lo, hi in idx)
            grid[sl] += 1
        assert (grid == 1).all(), (dims, nr, dist)
