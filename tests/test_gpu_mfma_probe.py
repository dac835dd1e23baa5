"""MFMA f64 16x16x4 lane-map probe: validates the fragment mapping the
GEMM kernel assumes; on mismatch, reverse-engineers the true mapping
from the raw per-lane accumulator dump and reports it."""
import ctypes

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_mfma_f64_lane_map():
    import distributedarrays_jl_amd as dja
    from distributedarrays_jl_amd._ffi import lib
    dja.comm.init()

    rng = np.random.default_rng(0)
    # asymmetric, all-distinct products (guide: always probe with
    # asymmetric B to catch transposes)
    A = np.asfortranarray(rng.uniform(1, 2, (16, 4)))
    B = np.asfortranarray(rng.uniform(1, 2, (4, 16)))
    ref = A @ B
    outc = np.zeros((16, 16), order="F")
    outraw = np.zeros(64 * 4)

    fn = lib.dbg_mfma_probe_f64
    fn.argtypes = [ctypes.c_void_p] * 4
    fn.restype = ctypes.c_int

    def hp(a):
        return a.ctypes.data_as(ctypes.c_void_p)

    def dev(nbytes):
        pp = ctypes.c_void_p()
        assert lib.da_alloc(nbytes, 0, ctypes.byref(pp)) == 0
        return pp

    dA, dB = dev(16 * 4 * 8), dev(4 * 16 * 8)
    dC, dR = dev(16 * 16 * 8), dev(64 * 4 * 8)
    assert lib.da_h2d(dA, hp(A), 16 * 4 * 8) == 0
    assert lib.da_h2d(dB, hp(B), 4 * 16 * 8) == 0
    rc = fn(dA, dB, dC, dR)
    assert rc == 0, rc
    assert lib.da_d2h(dC, hp(outc), 16 * 16 * 8) == 0
    assert lib.da_d2h(dR, hp(outraw), 64 * 4 * 8) == 0
    for d in (dA, dB, dC, dR):
        lib.da_free(d)

    if np.allclose(outc, ref, rtol=1e-13):
        return  # assumed mapping is correct

    # diagnose: find where each (lane, q) value sits in ref
    mapping = {}
    for l in range(64):
        for q in range(4):
            v = outraw[l * 4 + q]
            hits = np.argwhere(np.isclose(ref, v, rtol=1e-12))
            mapping[(l, q)] = [tuple(h) for h in hits]
    lines = ["lane-map mismatch; observed (lane,q)->(row,col):"]
    for l in (0, 1, 15, 16, 17, 31, 32, 48, 63):
        for q in range(4):
            lines.append("  (%d,%d)->%r" % (l, q, mapping[(l, q)][:2]))
    # also test transpose hypothesis
    if np.allclose(outc, ref.T, rtol=1e-13):
        lines.append("  outc == ref.T (C map transposed)")
    raise AssertionError("\n".join(lines))
