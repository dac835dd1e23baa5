"""Distributed samplesort (src/sort.jl:103-170; reference behaviour
pinned at test/darray.jl:1015-1025 over sizes and eltypes): CPU
simulation of the splitter/partition/exchange logic vs np.sort, and GPU
parity of the rocPRIM local-sort path."""
import numpy as np
import pytest

from distributedarrays_jl_amd import geometry as pg
from oracle import philox


def simulate_samplesort(x, P, s=64):
    """The exact host logic of ops.dsort in numpy."""
    idxs, _ = pg.chunk_indices((x.size,), (P,))
    chunks = [np.sort(x[lo:hi]) for (lo, hi), in idxs]
    samples = []
    for ch in chunks:
        k = min(s, ch.size)
        if k:
            sel = ((np.arange(k) + 0.5) * ch.size / k).astype(np.int64)
            samples.append(ch[sel])
    allsamp = np.sort(np.concatenate(samples)) if samples else \
        np.empty(0, x.dtype)
    if allsamp.size < P:
        splitters = allsamp[:max(P - 1, 0)]
    else:
        splitters = allsamp[[(i + 1) * allsamp.size // P
                             for i in range(P - 1)]]
    if splitters.size < P - 1:   # tiny inputs: trailing buckets empty
        pad = (np.inf if x.dtype.kind == "f"
               else np.iinfo(x.dtype).max)
        splitters = np.concatenate(
            [splitters, np.full(P - 1 - splitters.size, pad, x.dtype)])
    buckets = [[] for _ in range(P)]
    for ch in chunks:
        edges = [0] + [int(np.searchsorted(ch, sp, side="left"))
                       for sp in splitters] + [ch.size]
        for j in range(P):
            buckets[j].append(ch[edges[j]:edges[j + 1]])
    out = [np.sort(np.concatenate(b)) if b else np.empty(0, x.dtype)
           for b in buckets]
    return np.concatenate(out), [o.size for o in out]


@pytest.mark.parametrize("n", [0, 1, 2, 10, 1000, 10 ** 5])
@pytest.mark.parametrize("P", [1, 2, 4, 8])
def test_samplesort_sim_f64(n, P):
    x = philox.fill_uniform_f64(n, seed=5) if n else np.empty(0)
    got, sizes = simulate_samplesort(x, P)
    assert np.array_equal(got, np.sort(x))
    assert sum(sizes) == n


def test_samplesort_sim_i64_duplicates():
    with np.errstate(over="ignore"):
        x = np.abs(philox.fill_int64(10000, seed=6)) % 50  # heavy dups
        got, _ = simulate_samplesort(x, 4)
        assert np.array_equal(got, np.sort(x))


@pytest.mark.gpu
@pytest.mark.parametrize("n", [0, 1, 2, 1000, (1 << 20) + 7])
def test_gpu_dsort_f64(n):
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = philox.fill_uniform_f64(n, seed=9) if n else np.empty(0)
    d = dja.distribute(x) if n else dja.dzeros((0,))
    r = dja.dsort(d)
    assert np.array_equal(r.localpart(), np.sort(x))
    r.close(); d.close()


@pytest.mark.gpu
def test_gpu_dsort_i64():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    with np.errstate(over="ignore"):
        x = philox.fill_int64(100001, seed=10)
        d = dja.distribute(x)
        r = dja.dsort(d)
        assert np.array_equal(r.localpart(), np.sort(x))
        r.close(); d.close()


@pytest.mark.gpu
def test_gpu_dsort_f32():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = philox.fill_uniform_f32(200003, seed=11)
    d = dja.distribute(x)
    r = dja.dsort(d)
    assert np.array_equal(r.localpart(), np.sort(x, kind="stable"))
    r.close(); d.close()
