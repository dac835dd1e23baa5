"""Stateful differential fuzz: random op programs on a pool of DArrays,
mirrored in numpy, with BIT-EXACT comparison along every chain of
correctly-rounded ops (fills, +,-,*,min,max, sqrt, inv, abs, neg, abs2,
floor, sign, scalar broadcast, axpy) — which is most of the op surface.
Transcendental ops (exp/sin/...) make a chain 'unverified': its ops keep
running (kernel/pool/aliasing coverage) but values are no longer
compared, because error growth through long chains (cancellation,
discontinuities, chaotic trig) is unbounded and directed tests already
pin those ops at few-ulp tolerance on controlled inputs.

Catches: wrong buffers, aliasing, pooled-allocator reuse corruption,
opcode mixups, chunk-size bookkeeping.  Seeded, deterministic; FUZZ_OPS
scales the run (default 200)."""
import os

import numpy as np
import pytest

from oracle import ops as oops, philox

pytestmark = pytest.mark.gpu

EXACT_UNARY = ["neg", "abs", "abs2", "sqrt", "inv", "floor", "sign"]
TRANS_UNARY = ["exp", "log1p", "sin", "cos", "tanh"]
BINARY = ["add", "sub", "mul", "min2", "max2"]


def _mk(dja, rng, pool, mirror, exact):
    n = int(rng.integers(1, 20000))
    if rng.integers(0, 10) == 0:       # occasional multi-MB array
        n = int(rng.integers(1 << 20, 1 << 22))
    seed = int(rng.integers(0, 2 ** 31))
    d = dja.DArray((n,), "f64")
    d.rand_(seed_base=seed)
    pool.append(d)
    mirror.append(philox.fill_uniform_f64(n, seed))
    exact.append(True)


@pytest.mark.parametrize("seed", [20260915, 1, 2])
def test_differential_fuzz(seed):
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    nops = int(os.environ.get("FUZZ_OPS", "200"))
    rng = np.random.default_rng(int(os.environ.get("FUZZ_SEED", seed)))
    pool, mirror, exact = [], [], []
    for _ in range(4):
        _mk(dja, rng, pool, mirror, exact)
    checked = [0]

    def checkeq(i, tag):
        if not exact[i]:
            return
        ref = mirror[i]
        if ref.size > (1 << 20):
            # big arrays: bit-compare a 256 KiB head window (keeps long
            # soaks cheap; heads catch wholesale buffer/opcode bugs)
            import ctypes
            k = 1 << 15
            got = np.empty(k, dtype=np.float64)
            from distributedarrays_jl_amd._ffi import lib as _l, check as _c
            _c(_l.da_d2h(pool[i]._ptr(),
                         got.ctypes.data_as(ctypes.c_void_p), k * 8))
            assert np.array_equal(got, ref[:k], equal_nan=True), (tag, i)
        else:
            got = pool[i].localpart()
            assert np.array_equal(got, ref, equal_nan=True), (tag, i)
        checked[0] += 1

    for step in range(nops):
        action = rng.integers(0, 8)
        i = int(rng.integers(0, len(pool)))
        if action == 0:
            if len(pool) < 10:
                _mk(dja, rng, pool, mirror, exact)
            elif not exact[i]:
                # re-randomize an unverified array: restores a bit-exact
                # chain (keeps verification density high)
                seed = int(rng.integers(0, 2 ** 31))
                pool[i].rand_(seed_base=seed)
                mirror[i] = philox.fill_uniform_f64(mirror[i].size, seed)
                exact[i] = True
                checkeq(i, "rerand")
        elif action == 1:  # unary map (maybe in-place)
            if rng.integers(0, 4) == 0:
                op = TRANS_UNARY[int(rng.integers(0, len(TRANS_UNARY)))]
                is_exact = False
            else:
                op = EXACT_UNARY[int(rng.integers(0, len(EXACT_UNARY)))]
                is_exact = True
            x = mirror[i]
            if op in ("sqrt", "inv", "log1p"):
                dja.map_("abs", pool[i], pool[i])
                mirror[i] = np.abs(x)
                x = mirror[i]
            if rng.integers(0, 2):
                dja.map_(op, pool[i], pool[i])
                mirror[i] = oops.MAP_OPS[op](x)
                exact[i] = exact[i] and is_exact
                checkeq(i, "map_" + op)
            else:
                out = dja.dmap(op, pool[i])
                pool.append(out)
                mirror.append(oops.MAP_OPS[op](x))
                exact.append(exact[i] and is_exact)
                checkeq(len(pool) - 1, "dmap_" + op)
        elif action == 2:  # binary with a same-shape partner
            op = BINARY[int(rng.integers(0, len(BINARY)))]
            j = next((jj for jj in range(len(pool))
                      if jj != i and mirror[jj].shape == mirror[i].shape),
                     None)
            if j is None:
                continue
            out = dja.elementwise(op, pool[i], pool[j])
            pool.append(out)
            mirror.append(oops.MAP2_OPS[op](mirror[i], mirror[j]))
            exact.append(exact[i] and exact[j])
            checkeq(len(pool) - 1, "bin_" + op)
        elif action == 3:  # scalar broadcast (correctly rounded)
            c = float(rng.uniform(-2, 2))
            out = dja.elementwise_scalar("add", pool[i], c)
            pool.append(out)
            mirror.append(mirror[i] + c)
            exact.append(exact[i])
            checkeq(len(pool) - 1, "scalar_add")
        elif action == 4:  # reductions (verified on exact finite chains)
            x = mirror[i]
            s = dja.dsum(pool[i])
            if exact[i] and x.size and np.isfinite(x).all():
                ref = oops.oracle_reduce("identity", "add", [x])
                tol = 1e-11 * max(1.0, float(np.abs(x).sum()))
                assert abs(s - ref) <= tol, ("sum", i, s, ref)
                assert dja.dmaximum(pool[i]) == x.max()
                checked[0] += 1
        elif action == 5:  # axpy in place (two correctly-rounded ops,
            # evaluated identically on both sides)
            j = next((jj for jj in range(len(pool))
                      if jj != i and mirror[jj].shape == mirror[i].shape),
                     None)
            if j is None:
                continue
            a = float(rng.uniform(-1.5, 1.5))
            dja.axpy_(a, pool[j], pool[i])
            mirror[i] = mirror[i] + np.float64(a) * mirror[j]
            exact[i] = exact[i] and exact[j]
            checkeq(i, "axpy")
        elif action == 6 and len(pool) > 4:  # close + drop (pool churn)
            d = pool.pop(i)
            mirror.pop(i)
            exact.pop(i)
            d.close()
        elif action == 7:  # sort round trip
            if (mirror[i].size < 50000 and exact[i]
                    and not np.isnan(mirror[i]).any()):
                r = dja.dsort(pool[i])
                assert np.array_equal(r.localpart(), np.sort(mirror[i]))
                r.close()
                checked[0] += 1
    # the run must have actually verified a healthy number of ops
    assert checked[0] >= max(10, nops // 50), checked[0]
    for d in pool:
        d.close()
    assert dja.bytes_in_use() == 0


@pytest.mark.parametrize("seed", [5, 6])
def test_differential_fuzz_2d(seed):
    """2-D structural ops: transpose, gather_box, diagonal scaling and
    dims-reductions are data-movement/plumbing paths — fuzzed bit-exact
    (mul is correctly rounded; transpose/gather move bits)."""
    import ctypes
    import distributedarrays_jl_amd as dja
    from distributedarrays_jl_amd._ffi import lib, check
    dja.comm.init()
    nops = int(os.environ.get("FUZZ_OPS", "120"))
    rng = np.random.default_rng(seed)
    pool, mirror = [], []

    def mk():
        m = int(rng.integers(1, 200))
        n = int(rng.integers(1, 200))
        s = int(rng.integers(0, 2 ** 31))
        d = dja.DArray((m, n), "f64")
        d.rand_(seed_base=s)
        pool.append(d)
        mirror.append(np.asfortranarray(
            philox.fill_uniform_f64(m * n, s).reshape(m, n, order="F")))

    for _ in range(3):
        mk()
    for step in range(nops):
        act = rng.integers(0, 5)
        i = int(rng.integers(0, len(pool)))
        x = mirror[i]
        if act == 0 and len(pool) < 8:
            mk()
        elif act == 1:  # transpose (bit-exact data movement)
            tx = dja.dtranspose(pool[i])
            assert np.array_equal(tx.localpart(),
                                  np.asfortranarray(x.T)), ("T", i)
            tx.close()
        elif act == 2:  # random sub-box gather (copy2d indexing)
            m, n = x.shape
            r0 = int(rng.integers(0, m)); r1 = int(rng.integers(r0, m)) + 1
            c0 = int(rng.integers(0, n)); c1 = int(rng.integers(c0, n)) + 1
            buf, shape = dja.gather_box(pool[i], [((r0, r1), (c0, c1))])
            out = np.empty(shape, dtype=np.float64, order="F")
            check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                             out.size * 8))
            buf.free()
            assert np.array_equal(out, x[r0:r1, c0:c1]), ("box", i)
        elif act == 3:  # diagonal scaling in place (one rounded mul)
            if rng.integers(0, 2):
                dv = philox.fill_uniform_f64(x.shape[0], step + 1)
                dja.ddiag_lmul(dv, pool[i])
                mirror[i] = np.asfortranarray(dv[:, None] * x)
            else:
                dv = philox.fill_uniform_f64(x.shape[1], step + 1)
                dja.ddiag_rmul(pool[i], dv)
                mirror[i] = np.asfortranarray(x * dv[None, :])
            assert np.array_equal(pool[i].localpart(), mirror[i]), ("dg", i)
        elif act == 4:  # dims-reduction (tree order: tolerance)
            axes = [(0,), (1,), (0, 1)][int(rng.integers(0, 3))]
            R = dja.dsum_dims(pool[i], axes)
            ref = x.sum(axis=axes, keepdims=True)
            assert np.allclose(R.collect(), ref, rtol=1e-12,
                               atol=1e-12), ("dims", i, axes)
            R.close()
    for d in pool:
        d.close()
    assert dja.bytes_in_use() == 0
