"""Stateful differential fuzz: random sequences of hot-path ops applied
to a pool of DArrays, mirrored step-by-step in numpy, compared after
every op.  Exercises op interactions, the pooled allocator under churn,
in-place aliasing, and mixed dtypes — the reference's differential
pattern (DA vs Array on the same data) taken to random programs.

Seeded and deterministic; FUZZ_OPS env scales the run (default 200 for
the CI suite; tools/fuzz_long.py runs thousands)."""
import os

import numpy as np
import pytest

from oracle import ops as oops, philox

pytestmark = pytest.mark.gpu

UNARY = ["neg", "abs", "abs2", "sqrt", "exp", "log1p", "sin", "cos",
         "tanh", "floor", "sign", "inv"]
BINARY = ["add", "sub", "mul", "min2", "max2"]
EXACT_UNARY = {"neg", "abs", "abs2", "floor", "sign", "sqrt", "inv"}


def _mk(dja, rng, pool, mirror):
    n = int(rng.integers(1, 20000))
    if rng.integers(0, 10) == 0:       # occasional multi-MB array
        n = int(rng.integers(1 << 20, 1 << 22))
    seed = int(rng.integers(0, 2 ** 31))
    d = dja.DArray((n,), "f64")
    d.rand_(seed_base=seed)
    pool.append(d)
    mirror.append(philox.fill_uniform_f64(n, seed))


def test_differential_fuzz():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    nops = int(os.environ.get("FUZZ_OPS", "200"))
    rng = np.random.default_rng(20260915)
    pool, mirror = [], []
    for _ in range(4):
        _mk(dja, rng, pool, mirror)

    def checkeq(i, tolv, tag):
        got = pool[i].localpart()
        ref = mirror[i]
        # long chains legitimately reach inf-inf = NaN on BOTH sides;
        # compare NaNs as equal
        if tolv is None:
            assert np.array_equal(got, ref, equal_nan=True), (tag, i)
        elif tolv < 1e-6:
            assert np.allclose(got, ref, rtol=tolv, atol=tolv,
                               equal_nan=True), (tag, i, tolv)

    # per-array relative error budget: None = bit-exact so far; a float
    # is the tracked rtol bound, grown by each op's rough condition
    # number; arrays past 1e-6 stay in the pool (path coverage) but are
    # no longer value-compared.
    exactness = [None] * len(pool)

    def grow(tolv, x, factor=None):
        if factor is None:
            factor = 1.0 + float(np.nanmax(np.abs(x))) if x.size else 1.0
        base = 1e-14 if tolv is None else tolv
        return min(base * max(factor, 2.0) + 1e-14, 1.0)

    def cancel_factor(a, b, r):
        """amplification bound for a +/- b: (|a|+|b|) / |r|."""
        if r.size == 0:
            return 2.0
        with np.errstate(invalid="ignore"):
            num = float(np.nanmax(np.abs(a))) + float(np.nanmax(np.abs(b)))
            rm = np.abs(r[np.isfinite(r) & (r != 0)])
            rmin = float(rm.min()) if rm.size else 0.0
        if rmin == 0.0 or not np.isfinite(num):
            return 1e9          # full cancellation somewhere: untrack
        return min(max(num / rmin, 2.0), 1e9)
    for step in range(nops):
        action = rng.integers(0, 8)
        i = int(rng.integers(0, len(pool)))
        if action == 0 and len(pool) < 10:
            _mk(dja, rng, pool, mirror)
            exactness.append(None)
        elif action == 1:  # unary map (maybe in-place)
            op = UNARY[int(rng.integers(0, len(UNARY)))]
            x = mirror[i]
            if op in ("sqrt", "log1p", "inv"):
                # keep domain positive: abs first
                dja.map_("abs", pool[i], pool[i])
                mirror[i] = np.abs(x)
                x = mirror[i]
            ntol = (exactness[i] if op in EXACT_UNARY
                    else grow(exactness[i], x))
            if op in EXACT_UNARY and exactness[i] is not None:
                ntol = grow(exactness[i], x, 2.0)   # exact op, inexact input
            if rng.integers(0, 2):
                dja.map_(op, pool[i], pool[i])
                mirror[i] = oops.MAP_OPS[op](x)
                exactness[i] = ntol
                checkeq(i, exactness[i], "map_" + op)
            else:
                out = dja.dmap(op, pool[i])
                pool.append(out)
                mirror.append(oops.MAP_OPS[op](x))
                exactness.append(ntol)
                checkeq(len(pool) - 1, exactness[-1], "dmap_" + op)
        elif action == 2:  # binary with a same-shape partner (make one)
            op = BINARY[int(rng.integers(0, len(BINARY)))]
            j = next((jj for jj in range(len(pool))
                      if jj != i and mirror[jj].shape == mirror[i].shape),
                     None)
            if j is None:
                continue
            out = dja.elementwise(op, pool[i], pool[j])
            pool.append(out)
            mirror.append(oops.MAP2_OPS[op](mirror[i], mirror[j]))
            if exactness[i] is None and exactness[j] is None:
                exactness.append(None)
            else:
                fac = (cancel_factor(mirror[i], mirror[j], mirror[-1])
                       if op in ("add", "sub") else 4.0)
                exactness.append(grow(max(exactness[i] or 1e-14,
                                          exactness[j] or 1e-14),
                                      mirror[-1], fac))
            checkeq(len(pool) - 1, exactness[-1], "bin_" + op)
        elif action == 3:  # scalar broadcast
            c = float(rng.uniform(-2, 2))
            out = dja.elementwise_scalar("add", pool[i], c)
            pool.append(out)
            mirror.append(mirror[i] + c)
            exactness.append(None if exactness[i] is None
                             else grow(exactness[i], mirror[-1], 4.0))
            checkeq(len(pool) - 1, exactness[-1], "scalar_add")
        elif action == 4:  # reductions
            x = mirror[i]
            tracked = exactness[i] is None or exactness[i] < 1e-8
            if np.isnan(x).any() or np.isinf(x).any():
                dja.dsum(pool[i])   # path coverage only
            elif tracked:
                s = dja.dsum(pool[i])
                ref = oops.oracle_reduce("identity", "add", [x])
                tol = max(1e-11, (exactness[i] or 0) * 10) * \
                    max(1.0, abs(float(ref)), float(np.abs(x).sum()))
                assert abs(s - ref) <= tol, ("sum", i, s, ref)
                if x.size and exactness[i] is None:
                    assert dja.dmaximum(pool[i]) == x.max()
        elif action == 5:  # axpy / add / scale in place
            j = next((jj for jj in range(len(pool))
                      if jj != i and mirror[jj].shape == mirror[i].shape),
                     None)
            if j is None:
                continue
            a = float(rng.uniform(-1.5, 1.5))
            dja.axpy_(a, pool[j], pool[i])
            old_mirror_i = mirror[i]
            mirror[i] = mirror[i] + np.float64(a) * mirror[j]
            if exactness[i] is None and exactness[j] is None:
                exactness[i] = None
            else:
                fac = cancel_factor(old_mirror_i, mirror[j], mirror[i])
                exactness[i] = grow(max(exactness[i] or 1e-14,
                                        exactness[j] or 1e-14),
                                    mirror[i], fac)
            checkeq(i, exactness[i], "axpy")
        elif action == 6 and len(pool) > 4:  # close + drop (pool churn)
            d = pool.pop(i)
            mirror.pop(i)
            exactness.pop(i)
            d.close()
        elif action == 7:  # sort round trip (radix NaN-bit order
            # differs from np.sort's NaN-last; skip NaN inputs)
            if mirror[i].size < 50000 and not np.isnan(mirror[i]).any():
                r = dja.dsort(pool[i])
                if exactness[i] is None:
                    assert np.array_equal(r.localpart(),
                                          np.sort(mirror[i]))
                r.close()
    for d in pool:
        d.close()
    assert dja.bytes_in_use() == 0
