#!/usr/bin/env python3
"""Round-2 kernel probes on 1 GPU: fastmath map timings, expr kernel
throughput (flat + strided), and the DA_NT store A/B.  Emits one JSON
line per leg to stdout; run under gpurun.

Usage: python tools/r2_probe.py [--legs sin,expr,nt] [--elems N]
The NT leg re-execs itself with DA_NT=1 (the flag is latched per
process at first kernel launch)."""
import argparse
import ctypes
import json
import os
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench(fn, steps=10, warmup=3):
    from distributedarrays_jl_amd._ffi import lib, check
    for _ in range(warmup):
        fn()
    check(lib.da_synchronize())
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    check(lib.da_synchronize())
    return (time.perf_counter() - t0) / steps


def out(leg, ms, gbs, extra=None):
    rec = {"leg": leg, "ms": round(ms, 4), "gbs": round(gbs, 1)}
    if extra:
        rec.update(extra)
    print(json.dumps(rec), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--legs", default="sin,expr,nt")
    ap.add_argument("--elems", type=int, default=1 << 28)
    ap.add_argument("--nt-child", action="store_true")
    args = ap.parse_args()
    legs = args.legs.split(",")
    n = args.elems

    import distributedarrays_jl_amd as dja
    from distributedarrays_jl_amd import expr as E
    from distributedarrays_jl_amd._ffi import lib, check
    dja.comm.init()

    if args.nt_child:
        # NT=1 process: only the store-heavy legs
        D = dja.drand((n,), "f64")
        A = dja.drand((n,), "f64")
        B = dja.drand((n,), "f64")
        t = bench(lambda: D.fill_(1.0))
        out("fill_nt1", t * 1e3, n * 8 / t / 1e9)
        t = bench(lambda: dja.broadcast_fma(D, A, B, 0.5))
        out("bcast_fma_nt1", t * 1e3, n * 24 / t / 1e9)
        t = bench(lambda: dja.map_("sin", D, A))
        out("map_sin_nt1", t * 1e3, n * 16 / t / 1e9)
        return

    if "width" in legs:
        # structural gap probe: map identity vs raw D2D copy, W=2 vs 4
        D = dja.drand((n,), "f64")
        S = dja.DArray((n,), "f64")
        t = bench(lambda: dja.map_("identity", S, D))
        out("map_identity", t * 1e3, n * 16 / t / 1e9)
        t = bench(lambda: dja.map_("sin", S, D))
        out("map_sin_w", t * 1e3, n * 16 / t / 1e9)
        t = bench(lambda: check(lib.da_d2d(S._ptr(), D._ptr(), n * 8)))
        out("d2d_copy", t * 1e3, n * 16 / t / 1e9,
            {"note": "hipMemcpyAsync DtoD; 16 B/elem bus (R+W)"})
        A = dja.drand((n,), "f64")
        T2 = dja.DArray((n,), "f64")
        t = bench(lambda: dja.broadcast_fma(T2, D, A, 0.5))
        out("bcast_fma_w", t * 1e3, n * 24 / t / 1e9)
        for d in (D, S, A, T2):
            d.close()

    if "sin" in legs:
        D = dja.drand((n,), "f64")
        S = dja.DArray((n,), "f64")
        for op in ("sin", "cos", "exp", "sqrt", "tan", "log"):
            t = bench(lambda: dja.map_(op, S, D))
            out("map_" + op, t * 1e3, n * 16 / t / 1e9)
        # f32
        D32 = dja.drand((n,), "f32")
        S32 = dja.DArray((n,), "f32")
        t = bench(lambda: dja.map_("sin", S32, D32))
        out("map_sin_f32", t * 1e3, n * 8 / t / 1e9)
        D32.close(); S32.close()
        D.close(); S.close()

    if "expr" in legs:
        A = dja.drand((n,), "f64")
        B = dja.drand((n,), "f64")
        D = dja.DArray((n,), "f64")
        e1 = E.ref(A) * E.ref(B) + 0.5
        t = bench(lambda: E.materialize_(D, e1))
        out("expr_fma", t * 1e3, n * 24 / t / 1e9,
            {"note": "same op as bcast_fma (interpreter overhead probe)"})
        t = bench(lambda: dja.broadcast_fma(D, A, B, 0.5))
        out("bcast_fma_ref", t * 1e3, n * 24 / t / 1e9)
        e2 = E.sin(E.ref(A)) + E.ref(B) * 0.5
        t = bench(lambda: E.materialize_(D, e2))
        out("expr_sin_fused", t * 1e3, n * 24 / t / 1e9)
        # unfused equivalent: map into temp + fma
        T_ = dja.DArray((n,), "f64")
        def unfused():
            dja.map_("sin", T_, A)
            dja.broadcast_fma(D, T_, B, 0.0)   # placeholder compose
        t = bench(unfused)
        out("expr_sin_unfused2k", t * 1e3, n * 24 / t / 1e9,
            {"note": "2 kernels + temp: 56 B/elem actual traffic"})
        # longer chain
        e3 = (E.ref(A) + E.ref(B)) / (E.abs(E.ref(A)) + 1.0) - \
            E.ref(B) * 0.25
        t = bench(lambda: E.materialize_(D, e3))
        out("expr_chain5", t * 1e3, n * 24 / t / 1e9)
        A.close(); B.close(); D.close(); T_.close()
        print(json.dumps({"leg": "expr_jit_state",
                          "state": int(lib.da_expr_jit_state()),
                          "err": (lib.da_expr_jit_errstr() or b"")
                          .decode()[:200]}), flush=True)
        # strided: x - mean(x, dims=1) on 16384^2 (2 GiB)
        m = 16384
        X = dja.drand((m, m), "f64")
        M = dja.dmean_dims(X, (0,))
        D2 = dja.DArray((m, m), "f64")
        em = E.ref(X) - E.ref(M)
        t = bench(lambda: E.materialize_(D2, em), steps=5)
        out("expr_strided_demean", t * 1e3, m * m * 16 / t / 1e9,
            {"note": "16 B/elem algorithmic (R+W; mean row cached)"})
        X.close(); M.close(); D2.close()

    if "dims" in legs:
        m = 16384
        X = dja.drand((m, m), "f64")
        for red, name in [((0,), "dims0"), ((1,), "dims1")]:
            def run(red=red):
                R = dja.dsum_dims(X, red)
                R.close()
            t = bench(run, steps=8)
            out("reduce_" + name, t * 1e3, m * m * 8 / t / 1e9,
                {"note": "8 B/elem read, 16384^2 f64, whole-op"})
        X.close()

    if "nt" in legs:
        # NT=0 baselines in this process
        D = dja.drand((n,), "f64")
        A = dja.drand((n,), "f64")
        B = dja.drand((n,), "f64")
        t = bench(lambda: D.fill_(1.0))
        out("fill_nt0", t * 1e3, n * 8 / t / 1e9)
        t = bench(lambda: dja.broadcast_fma(D, A, B, 0.5))
        out("bcast_fma_nt0", t * 1e3, n * 24 / t / 1e9)
        t = bench(lambda: dja.map_("sin", D, A))
        out("map_sin_nt0", t * 1e3, n * 16 / t / 1e9)
        D.close(); A.close(); B.close()
        dja.d_closeall()
        env = dict(os.environ, DA_NT="1")
        subprocess.run([sys.executable, __file__, "--nt-child",
                        "--elems", str(n)], env=env, check=True)


if __name__ == "__main__":
    main()
