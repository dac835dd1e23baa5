"""Monte-Carlo pi on DArrays — the reference README-style usage demo:
philox-filled DVectors, fused elementwise ops, predicate reductions.

Run on an MI355X box:  python examples/monte_carlo_pi.py [n]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja


def estimate_pi(n=1 << 26):
    x = dja.drand((n,), "f64", seed_base=1)
    y = dja.drand((n,), "f64", seed_base=2)
    # r2 = x*x + y*y  (fused: r2 = x .* x .+ 0, then axpy-style add of y*y)
    r2 = dja.dzeros((n,))
    dja.broadcast_fma(r2, x, x, 0.0)
    yy = dja.elementwise("mul", y, y)
    dja.add_(r2, yy, 1.0)
    # inside = count(r2 .< 1) == n - count(floor(r2) != 0) for r2 in [0,2)
    fl = dja.dmap("floor", r2)
    outside = dja.dcount("nonzero", fl)
    pi = 4.0 * (n - outside) / n
    for d in (x, y, r2, yy, fl):
        d.close()
    return pi


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 1 << 26
    print("pi ~= %.6f (n=%d)" % (estimate_pi(n), n))
