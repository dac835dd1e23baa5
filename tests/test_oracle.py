"""Pin the oracle itself: published Philox vectors + the reference's own
known-answer tests for chunk geometry and reduction semantics."""
import numpy as np
import pytest

from oracle import philox, geometry, ops


# ---------------------------------------------------------------- Philox KAT
def test_philox_kat():
    """Random123 1.09 published known-answer vectors for philox4x32-10."""
    out = philox.philox4x32([0], [0], [0], [0], 0, 0)
    assert [hex(int(o[0])) for o in out] == [
        "0x6627e8d5", "0xe169c58d", "0xbc57ac4c", "0x9b00dbd8"]
    ff = 0xFFFFFFFF
    out = philox.philox4x32([ff], [ff], [ff], [ff], ff, ff)
    assert [hex(int(o[0])) for o in out] == [
        "0x408f276d", "0x41c83b0e", "0xa20bc7c6", "0x6d5451fd"]
    out = philox.philox4x32([0x243f6a88], [0x85a308d3], [0x13198a2e],
                            [0x03707344], 0xa4093822, 0x299f31d0)
    assert [hex(int(o[0])) for o in out] == [
        "0xd16cfe09", "0x94fdcceb", "0x5001e420", "0x24126ea1"]


def test_uniform_ranges_and_determinism():
    for fill in (philox.fill_uniform_f64, philox.fill_uniform_f32):
        x = fill(10000, seed=1234)
        assert x.min() >= 0.0 and x.max() < 1.0
        assert abs(float(x.mean()) - 0.5) < 0.02
        y = fill(10000, seed=1234)
        assert np.array_equal(x, y)
        z = fill(10000, seed=1235)
        assert not np.array_equal(x, z)


def test_uniform_offset_slicing():
    """Counter-based: generating [offset, offset+n) matches a slice of the
    full stream — the property GPU spot-checks rely on."""
    full = philox.fill_uniform_f64(1000, seed=7)
    part = philox.fill_uniform_f64(100, seed=7, offset=137)
    assert np.array_equal(full[137:237], part)
    f32 = philox.fill_uniform_f32(1000, seed=7)
    p32 = philox.fill_uniform_f32(99, seed=7, offset=401)
    assert np.array_equal(f32[401:500], p32)


def test_normal_moments():
    x = philox.fill_normal_f64(200000, seed=42)
    assert abs(float(x.mean())) < 0.02
    assert abs(float(x.std()) - 1.0) < 0.02


# -------------------------------------------------------------- geometry pin
def test_defaultdist_1d_reference_pin():
    # /root/reference/test/darray.jl:66
    assert geometry.defaultdist_1d(50, 4) == [1, 14, 27, 39, 51]


def test_defaultdist_1d_properties():
    for sz in (1, 2, 7, 50, 100, 2**18):
        for nc in (1, 2, 3, 4, 7, 8):
            cuts = geometry.defaultdist_1d(sz, nc)
            rng = geometry.chunk_ranges_1d(cuts)
            covered = []
            for lo, hi in rng:
                covered.extend(range(lo, hi))
            if sz >= nc:
                assert covered == list(range(sz))
                lens = [hi - lo for lo, hi in rng]
                assert max(lens) - min(lens) <= 1


def test_defaultdist_1d_undersized():
    # sz < nc: [1..sz+1] zero-padded (darray.jl:294)
    assert geometry.defaultdist_1d(2, 4) == [1, 2, 3, 0, 0]
    rng = geometry.chunk_ranges_1d(geometry.defaultdist_1d(2, 4))
    assert rng == [(0, 1), (1, 2), (0, 0), (0, 0)]


def test_defaultdist_dims():
    # 8 procs on a square matrix -> 2x4 grid (factor 2 allocated 3 times,
    # ties resolve to highest dim: darray.jl:268)
    assert geometry.defaultdist_dims([16384, 16384], 8) == [2, 4]
    assert geometry.defaultdist_dims([32768, 8192], 4) == [4, 1]
    assert geometry.defaultdist_dims([100], 4) == [4]
    # issue #166 consistency (test/darray.jl:61-66): uneven 1-D split
    assert geometry.defaultdist_1d(3, 2) == [1, 3, 4]


def test_chunk_idxs_column_major_order():
    idxs, cuts = geometry.chunk_idxs([4, 6], [2, 3])
    assert len(idxs) == 6
    # rank 0 gets (rows 0:2, cols 0:2); rank 1 (rows 2:4, cols 0:2) —
    # first dim fastest (column-major reshape, darray.jl:161)
    assert idxs[0] == ((0, 2), (0, 2))
    assert idxs[1] == ((2, 4), (0, 2))
    assert idxs[2] == ((0, 2), (2, 4))


# ------------------------------------------------------- reduction semantics
def test_reduce_int_exact():
    rng = np.random.default_rng(0)
    a = rng.integers(-2**62, 2**62, size=10000, dtype=np.int64)
    idxs, _ = geometry.chunk_idxs([10000], [4])
    chunks = ops.make_chunks(a, idxs)
    # bit-exact vs flat sum mod 2^64 (test/darray.jl:286-294 exactness)
    assert ops.oracle_reduce("identity", "add", chunks) == a.sum()
    assert ops.oracle_reduce("identity", "max", chunks) == a.max()
    assert ops.oracle_reduce("identity", "min", chunks) == a.min()


def test_reduce_float_tolerance():
    x = philox.fill_uniform_f64(100001, seed=3)
    idxs, _ = geometry.chunk_idxs([100001], [3])
    chunks = ops.make_chunks(x, idxs)
    s = ops.oracle_reduce("identity", "add", chunks)
    assert abs(s - x.sum()) / abs(x.sum()) < 1e-12
    s2 = ops.oracle_reduce("abs2", "add", chunks)
    assert abs(s2 - (x * x).sum()) / (x * x).sum() < 1e-12


def test_reduce_empty_chunks():
    # undersized distribution has empty chunks; sum skips nothing (init=0)
    x = np.arange(3, dtype=np.float64)
    idxs, _ = geometry.chunk_idxs([3], [5])
    chunks = ops.make_chunks(x, idxs)
    assert ops.oracle_reduce("identity", "add", chunks) == 3.0


# ------------------------------------------------------------ matmul oracle
def test_matmul_blocked_vs_flat():
    m, k, n = 60, 50, 40
    A = np.asfortranarray(philox.fill_uniform_f64(m * k, 1).reshape(m, k, order="F"))
    B = np.asfortranarray(philox.fill_uniform_f64(k * n, 2).reshape(k, n, order="F"))
    rc = geometry.defaultdist_1d(m, 2)
    ic = geometry.defaultdist_1d(k, 4)
    cc = geometry.defaultdist_1d(n, 4)
    C = ops.oracle_matmul_blocked(A, B, rc, ic, cc)
    ref = A @ B
    assert np.allclose(C, ref, rtol=1e-12)
    # alpha/beta path (linalg.jl:232-251)
    C0 = np.asfortranarray(philox.fill_uniform_f64(m * n, 3).reshape(m, n, order="F"))
    C2 = ops.oracle_matmul_blocked(A, B, rc, ic, cc, alpha=2.0, beta=0.5, C0=C0)
    assert np.allclose(C2, 2.0 * ref + 0.5 * C0, rtol=1e-12)


def test_map_ops_sane():
    x = philox.fill_uniform_f64(1000, 5) * 0.8 + 0.1
    for op in ("sin", "abs2", "sqrt", "exp", "log", "neg"):
        y = ops.oracle_map(op, x)
        assert y.shape == x.shape
    assert np.array_equal(ops.oracle_map("identity", x), x)
    assert np.allclose(ops.oracle_bcast_fma(x, x, 0.5), x * x + 0.5)
