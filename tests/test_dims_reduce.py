"""dims-reductions (SURVEY §8f row 1 / mapreduce.jl:42-94): oracle
self-consistency on CPU, and GPU parity of the DArray-level op."""
import numpy as np
import pytest

from oracle import philox, ops as oops, geometry as ogeo


def test_oracle_reduce_dims_vs_numpy():
    x = np.asfortranarray(philox.fill_uniform_f64(60 * 40, 1)
                          .reshape(60, 40, order="F"))
    for nr in (1, 2, 4, 8):
        dist = ogeo.defaultdist_dims([60, 40], nr)
        idxs, _ = ogeo.chunk_idxs([60, 40], dist)
        chunks = oops.make_chunks(x, idxs)
        for axes in [(0,), (1,), (0, 1)]:
            got = oops.oracle_reduce_dims("identity", "add", chunks, idxs,
                                          (60, 40), axes)
            ref = x.sum(axis=axes, keepdims=True)
            assert np.allclose(got, ref, rtol=1e-12), (nr, axes)
            gmax = oops.oracle_reduce_dims("identity", "max", chunks, idxs,
                                           (60, 40), axes)
            assert np.array_equal(gmax, x.max(axis=axes, keepdims=True))


def test_oracle_reduce_dims_int_exact():
    with np.errstate(over="ignore"):
        x = np.asfortranarray(philox.fill_int64(32 * 24, 2)
                              .reshape(32, 24, order="F"))
        dist = [2, 2]
        idxs, _ = ogeo.chunk_idxs([32, 24], dist)
        chunks = oops.make_chunks(x, idxs)
        got = oops.oracle_reduce_dims("identity", "add", chunks, idxs,
                                      (32, 24), (0,))
        assert np.array_equal(got, x.sum(axis=0, keepdims=True))


@pytest.mark.gpu
@pytest.mark.parametrize("shape,axes", [
    ((128, 96), (0,)), ((128, 96), (1,)), ((128, 96), (0, 1)),
    ((1000,), (0,)), ((17, 33), (0,)), ((64, 32, 16), (1,)),
    ((64, 32, 16), (0, 2)),
])
def test_gpu_dims_reduce(shape, axes):
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    n = int(np.prod(shape))
    x = np.asfortranarray(philox.fill_uniform_f64(n, 7)
                          .reshape(shape, order="F"))
    d = dja.distribute(x)
    R = dja.dsum_dims(d, axes)
    ref = x.sum(axis=axes, keepdims=True)
    assert R.dims == ref.shape
    got = R.collect()
    assert np.allclose(got, ref, rtol=1e-12), (shape, axes)
    R.close()
    M = dja.dmaximum_dims(d, axes)
    assert np.array_equal(M.collect(), x.max(axis=axes, keepdims=True))
    M.close()
    A = dja.dmean_dims(d, axes)
    assert np.allclose(A.collect(), x.mean(axis=axes, keepdims=True),
                       rtol=1e-12)
    A.close()
    d.close()


@pytest.mark.gpu
def test_gpu_dims_reduce_abs2():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = np.asfortranarray(philox.fill_uniform_f64(96 * 64, 9)
                          .reshape(96, 64, order="F"))
    d = dja.distribute(x)
    R = dja.dreduce_dims("abs2", "add", d, (0,))
    assert np.allclose(R.collect(), (x * x).sum(axis=0, keepdims=True),
                       rtol=1e-12)
    R.close(); d.close()


@pytest.mark.gpu
def test_gpu_matvec():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    m, k = 200, 150
    A = np.asfortranarray(philox.fill_uniform_f64(m * k, 11)
                          .reshape(m, k, order="F"))
    x = philox.fill_uniform_f64(k, 12)
    dA = dja.distribute(A)
    y = dja.dmatvec(dA, x)
    assert y.dims == (m,)
    assert np.allclose(y.collect(), A @ x, rtol=1e-12)
    y.close()
    y2 = dja.dmatvec(dA, x, alpha=2.0)
    assert np.allclose(y2.collect(), 2.0 * (A @ x), rtol=1e-12)
    y2.close()
    # adjoint form (linalg.jl:124-167): y = alpha * A' * x
    xr = philox.fill_uniform_f64(m, 13)
    ya = dja.dmatvec_adj(dA, xr, alpha=1.5)
    assert ya.dims == (k,)
    assert np.allclose(ya.collect(), 1.5 * (A.T @ xr), rtol=1e-12)
    ya.close(); dA.close()


@pytest.mark.gpu
def test_gpu_dims_reduce_i64_exact():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    with np.errstate(over="ignore"):
        x = np.asfortranarray(philox.fill_int64(128 * 64, 13)
                              .reshape(128, 64, order="F"))
        d = dja.distribute(x)
        for axes in ((0,), (1,)):
            R = dja.dsum_dims(d, axes)
            assert np.array_equal(R.collect(),
                                  x.sum(axis=axes, keepdims=True))
            R.close()
        d.close()


@pytest.mark.gpu
def test_gpu_dims_reduce_f32():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = np.asfortranarray(philox.fill_uniform_f32(256 * 128, 14)
                          .reshape(256, 128, order="F"))
    d = dja.distribute(x)
    R = dja.dsum_dims(d, (1,))
    ref = x.astype(np.float64).sum(axis=1, keepdims=True)
    assert np.allclose(R.collect().astype(np.float64), ref, rtol=1e-4)
    R.close(); d.close()


@pytest.mark.gpu
def test_gpu_dims_reduce_large_axis_variant():
    """exercise the column-slice variant (few outputs, long axis)"""
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = np.asfortranarray(philox.fill_uniform_f64(2048 * 512, 15)
                          .reshape(2048, 512, order="F"))
    d = dja.distribute(x)
    R = dja.dsum_dims(d, (1,))
    assert np.allclose(R.collect(), x.sum(axis=1, keepdims=True),
                       rtol=1e-12)
    R.close(); d.close()


@pytest.mark.gpu
def test_ops_on_dims_reduction_result():
    """similar()/map on a non-identity-ranks DArray (regression: similar
    used to drop the owner mapping)."""
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = np.asfortranarray(philox.fill_uniform_f64(64 * 32, 17)
                          .reshape(64, 32, order="F"))
    d = dja.distribute(x)
    R = dja.dsum_dims(d, (0,))
    out = dja.dmap("abs2", R)
    ref = x.sum(axis=0, keepdims=True) ** 2
    assert np.allclose(out.collect(), ref, rtol=1e-12)
    assert abs(dja.dsum(R) - x.sum()) < 1e-9
    out.close(); R.close(); d.close()


@pytest.mark.gpu
def test_gpu_matvec_f32_and_norm_p():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    m, k = 96, 64
    A = np.asfortranarray(philox.fill_uniform_f32(m * k, 21)
                          .reshape(m, k, order="F"))
    x = philox.fill_uniform_f32(k, 22)
    dA = dja.distribute(A)
    y = dja.dmatvec(dA, x)
    assert np.allclose(y.collect(), A @ x, rtol=1e-5)
    xr = philox.fill_uniform_f32(m, 23)
    ya = dja.dmatvec_adj(dA, xr)
    assert np.allclose(ya.collect(), A.T @ xr, rtol=1e-5)
    y.close(); ya.close(); dA.close()
    # general-p norm (linalg.jl:47-52)
    v = philox.fill_uniform_f64(5001, 24) - 0.5
    dv = dja.distribute(v)
    ref = float(np.sum(np.abs(v) ** 3.5) ** (1 / 3.5))
    assert abs(dja.dnorm(dv, 3.5) - ref) <= 1e-12 * ref
    assert dja.dnorm(dv, 0) == float(np.count_nonzero(v))
    dv.close()


@pytest.mark.gpu
def test_gpu_mapslices_ppeval_redistribute():
    """mapreduce.jl:191-323 host-boundary veneers on the GPU path."""
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    g = np.asfortranarray(philox.fill_uniform_f64(24 * 10, 31)
                          .reshape(24, 10, order="F"))
    D = dja.distribute(g)
    R = dja.redistribute(D, (1, 1))
    assert np.array_equal(R.collect(), g)
    R.close()
    M = dja.dmapslices(lambda col: np.cumsum(col), D, (0,))
    assert np.allclose(M.collect(), np.cumsum(g, axis=0), rtol=0)
    M.close()
    M2 = dja.dmapslices(lambda col: np.array([col.sum()]), D, (0,))
    assert np.allclose(M2.collect(), g.sum(axis=0, keepdims=True),
                       rtol=1e-12)
    M2.close(); D.close()
    g3 = np.asfortranarray(philox.fill_uniform_f64(4 * 4 * 6, 32)
                           .reshape(4, 4, 6, order="F"))
    A3 = dja.distribute(g3)
    w = np.asfortranarray(philox.fill_uniform_f64(16, 33)
                          .reshape(4, 4, order="F"))
    P = dja.dppeval(lambda s, b: s @ b, A3, w)
    ref = np.stack([g3[:, :, i] @ w for i in range(6)], axis=-1)
    assert np.allclose(P.collect(), ref, rtol=1e-12)
    P.close(); A3.close()
