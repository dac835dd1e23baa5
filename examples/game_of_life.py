"""Conway's game of life on a distributed grid — the reference manual's
halo-exchange showcase (docs/src/index.md:160-181), built entirely from
this framework's primitives: gather_box (the makelocal halo fetch over
xGMI), strided device copies for the 8 neighbour shifts, and i64
elementwise ops for the rule (eq/and/or expressed arithmetically on 0/1
grids: a==k  ->  1 - min(1, |a-k|); a&b -> min; a|b -> max).

Run on an MI355X box:  python examples/game_of_life.py [n] [steps]
"""
import ctypes
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja
from distributedarrays_jl_amd._ffi import lib, check
from distributedarrays_jl_amd.ops import _Buf, _copy2d


def life_step(state):
    """One generation, in place.  state: 2-D i64 DArray of 0/1."""
    m, n = state.dims
    lm, ln = state.lshape
    (rlo, rhi), (clo, chi) = state.lidx
    esz = 8

    # halo box: local box grown by 1, clamped to the global grid —
    # derived from shared metadata on every rank (collective gather)
    boxes = [None] * state.nranks
    for c, r in enumerate(state.ranks):
        (a, b), (cc, d) = state.idxs[c]
        boxes[r] = ((max(a - 1, 0), min(b + 1, m)),
                    (max(cc - 1, 0), min(d + 1, n)))
    buf, bshape = dja.gather_box(state, boxes)

    # paste into a zero (lm+2) x (ln+2) pad (global edges stay dead)
    pm, pn = lm + 2, ln + 2
    pad = _Buf(pm * pn * esz)
    check(lib.da_fill(pad.p, 0.0, pm * pn, 2))
    (blo, _), (bco, _) = boxes[state.rank]
    off_r = blo - (rlo - 1)          # 0 at interior, 1 at the top edge
    off_c = bco - (clo - 1)
    _copy2d(pad.at((off_r + off_c * pm) * esz), pm * esz,
            buf.p, bshape[0] * esz, bshape[0] * esz, bshape[1])
    buf.free()

    # neighbour count: sum the 8 shifted (lm x ln) windows of the pad
    cnt = _Buf(lm * ln * esz)
    tmp = _Buf(lm * ln * esz)
    check(lib.da_fill(cnt.p, 0.0, lm * ln, 2))
    for di in (0, 1, 2):
        for dj in (0, 1, 2):
            if di == 1 and dj == 1:
                continue
            _copy2d(tmp.p, lm * esz,
                    pad.at((di + dj * pm) * esz), pm * esz,
                    lm * esz, ln)
            check(lib.da_add(cnt.p, tmp.p, 1.0, lm * ln, 2))
    pad.free()

    # rule: next = (cnt==3) | (alive & (cnt==2)), all on 0/1 i64
    def eq_k(src, k, out):
        # out = 1 - min(1, |src - k|)
        check(lib.da_map2_scalar(1, out.p, src, float(k), 0, lm * ln, 2))
        check(lib.da_map(2, out.p, out.p, lm * ln, 2))            # abs
        check(lib.da_map2_scalar(4, out.p, out.p, 1.0, 0, lm * ln, 2))  # min2
        check(lib.da_map2_scalar(1, out.p, out.p, 1.0, 1, lm * ln, 2))  # 1-x

    e3 = _Buf(lm * ln * esz)
    e2 = _Buf(lm * ln * esz)
    if lm * ln:
        eq_k(cnt.p, 3, e3)
        eq_k(cnt.p, 2, e2)
        check(lib.da_map2(4, e2.p, e2.p, state._ptr(), lm * ln, 2))  # & alive
        check(lib.da_map2(5, e3.p, e3.p, e2.p, lm * ln, 2))          # |
        check(lib.da_d2d(state._ptr(), e3.p, lm * ln * esz))
    check(lib.da_synchronize())
    for b in (cnt, tmp, e3, e2):
        b.free()
    return state


def numpy_life_step(a):
    p = np.zeros((a.shape[0] + 2, a.shape[1] + 2), dtype=a.dtype)
    p[1:-1, 1:-1] = a
    cnt = sum(p[di:di + a.shape[0], dj:dj + a.shape[1]]
              for di in range(3) for dj in range(3)
              if not (di == 1 and dj == 1))
    return ((cnt == 3) | ((a == 1) & (cnt == 2))).astype(a.dtype)


def run(n=256, steps=10, seed=12345):
    rng = np.random.default_rng(seed)
    init = (rng.random((n, n)) < 0.35).astype(np.int64)
    ref = np.asfortranarray(init.copy())
    d = dja.distribute(np.asfortranarray(init))
    for _ in range(steps):
        life_step(d)
        ref = np.asfortranarray(numpy_life_step(ref))
    got = d.collect()
    ok = np.array_equal(got, ref)
    pop = int(got.sum())
    d.close()
    return ok, pop


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 10
    ok, pop = run(n, steps)
    print("life %dx%d after %d steps: population %d, matches numpy: %s"
          % (n, n, steps, pop, ok))
