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

    def checkeq(i, exact, tag):
        got = pool[i].localpart()
        ref = mirror[i]
        # long chains legitimately reach inf-inf = NaN on BOTH sides;
        # compare NaNs as equal
        if exact:
            assert np.array_equal(got, ref, equal_nan=True), (tag, i)
        else:
            assert np.allclose(got, ref, rtol=1e-12, atol=1e-13,
                               equal_nan=True), (tag, i)

    exactness = [True] * len(pool)
    for step in range(nops):
        action = rng.integers(0, 8)
        i = int(rng.integers(0, len(pool)))
        if action == 0 and len(pool) < 10:
            _mk(dja, rng, pool, mirror)
            exactness.append(True)
        elif action == 1:  # unary map (maybe in-place)
            op = UNARY[int(rng.integers(0, len(UNARY)))]
            x = mirror[i]
            if op in ("sqrt", "log1p", "inv"):
                # keep domain positive: abs first
                dja.map_("abs", pool[i], pool[i])
                mirror[i] = np.abs(x)
                x = mirror[i]
            if rng.integers(0, 2):
                dja.map_(op, pool[i], pool[i])
                mirror[i] = oops.MAP_OPS[op](x)
                exactness[i] = exactness[i] and op in EXACT_UNARY
                checkeq(i, exactness[i], "map_" + op)
            else:
                out = dja.dmap(op, pool[i])
                pool.append(out)
                mirror.append(oops.MAP_OPS[op](x))
                exactness.append(exactness[i] and op in EXACT_UNARY)
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
            exactness.append(exactness[i] and exactness[j])
            checkeq(len(pool) - 1, exactness[-1], "bin_" + op)
        elif action == 3:  # scalar broadcast
            c = float(rng.uniform(-2, 2))
            out = dja.elementwise_scalar("add", pool[i], c)
            pool.append(out)
            mirror.append(mirror[i] + c)
            exactness.append(exactness[i])
            checkeq(len(pool) - 1, exactness[-1], "scalar_add")
        elif action == 4:  # reductions
            x = mirror[i]
            if np.isnan(x).any() or np.isinf(x).any():
                assert not np.isfinite(dja.dsum(pool[i])) \
                    or not exactness[i]
            else:
                s = dja.dsum(pool[i])
                ref = oops.oracle_reduce("identity", "add", [x])
                tol = 1e-11 * max(1.0, abs(float(ref)))
                assert abs(s - ref) <= tol or exactness[i] is False, "sum"
                if x.size:
                    assert (dja.dmaximum(pool[i]) == x.max()
                            or not exactness[i])
        elif action == 5:  # axpy / add / scale in place
            j = next((jj for jj in range(len(pool))
                      if jj != i and mirror[jj].shape == mirror[i].shape),
                     None)
            if j is None:
                continue
            a = float(rng.uniform(-1.5, 1.5))
            dja.axpy_(a, pool[j], pool[i])
            mirror[i] = mirror[i] + np.float64(a) * mirror[j]
            exactness[i] = exactness[i] and exactness[j]
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
                assert np.array_equal(r.localpart(), np.sort(mirror[i])) \
                    or not exactness[i]
                r.close()
    for d in pool:
        d.close()
    assert dja.bytes_in_use() == 0
