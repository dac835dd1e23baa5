"""Pin the C/OpenMP baseline (bench.py's cpu_baseline leg) against the
numpy oracle on identical seeded inputs."""
import ctypes

import numpy as np
import pytest

from oracle import cpu_baseline as cb, philox


@pytest.fixture(scope="module")
def lib():
    return cb.load()


def test_philox_bitexact(lib):
    a = cb.fill_uniform_f64(lib, 50000, 1234)
    assert np.array_equal(a, philox.fill_uniform_f64(50000, 1234))
    f = np.empty(50001, dtype=np.float32)
    lib.cb_fill_uniform_f32(cb.ptr(f), 50001, 77)
    assert np.array_equal(f, philox.fill_uniform_f32(50001, 77))


def test_sum_and_map(lib):
    a = cb.fill_uniform_f64(lib, 100000, 5)
    s = lib.cb_sum_f64(cb.ptr(a), 100000)
    assert abs(s - a.sum()) / a.sum() < 1e-12
    d = np.empty_like(a)
    lib.cb_map_sin_f64(cb.ptr(d), cb.ptr(a), a.size)
    assert np.allclose(d, np.sin(a), rtol=1e-15, atol=1e-15)
    o = np.empty_like(a)
    lib.cb_bcast_fma_f64(cb.ptr(o), cb.ptr(a), cb.ptr(d), 0.5, a.size)
    assert np.array_equal(o, a * d + 0.5)


def test_abs2_sum_f32(lib):
    f = np.empty(65536, dtype=np.float32)
    lib.cb_fill_uniform_f32(cb.ptr(f), f.size, 3)
    s = lib.cb_abs2_sum_f32(cb.ptr(f), f.size)
    ref = (f.astype(np.float64) ** 2).sum()
    assert abs(s - ref) / ref < 1e-4


def test_gemm(lib):
    m, k, n = 96, 80, 64
    A = np.asfortranarray(philox.fill_uniform_f64(m * k, 1)
                          .reshape(m, k, order="F"))
    B = np.asfortranarray(philox.fill_uniform_f64(k * n, 2)
                          .reshape(k, n, order="F"))
    C = np.zeros((m, n), order="F")
    lib.cb_gemm_f64(cb.ptr(C), cb.ptr(A), cb.ptr(B), m, n, k)
    assert np.allclose(C, A @ B, rtol=1e-12)
