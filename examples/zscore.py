"""Column z-score standardization of a distributed matrix — the
canonical nested-broadcast + dims-reduction composition:

    Z .= (X .- mean(X, dims=1)) ./ std(X, dims=1)

Built entirely from framework primitives: dims-reductions produce the
(1, n) mean and variance rows, and ONE fused expression kernel
(da_expr, hipRTC-compiled) materializes the whole tree with the two
row operands expanded via stride-0 — the `a .- mean(a, dims=1)` form
the reference pins at /root/reference/test/darray.jl:885-897,
generalized to two expanded operands.

Run on an MI355X box:  python examples/zscore.py [m] [n]
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja
from distributedarrays_jl_amd import expr as E


def zscore(m=8192, n=4096):
    X = dja.drand((m, n), "f64")
    mu = dja.dmean_dims(X, (0,))                      # (1, n)
    # var = E[X^2] - mu^2 (population variance, one extra reduction)
    X2 = dja.dmap("abs2", X)
    ex2 = dja.dmean_dims(X2, (0,))
    X2.close()
    # sigma row: sqrt(ex2 - mu^2), computed as a (1, n) expression
    sig = dja.DArray(mu.dims, "f64", mu.dist, ranks=list(mu.ranks))
    E.materialize_(sig, E.sqrt(E.ref(ex2) - E.ref(mu) * E.ref(mu)))
    ex2.close()
    # the fused standardization: one kernel over the full matrix,
    # both row operands stride-0-expanded
    Z = dja.DArray((m, n), "f64")
    E.materialize_(Z, (E.ref(X) - E.ref(mu)) / E.ref(sig))
    return X, mu, sig, Z


def main():
    m = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
    X, mu, sig, Z = zscore(m, n)
    # verify on the host
    hX = X.collect()
    hZ = Z.collect()
    ref = (hX - hX.mean(axis=0, keepdims=True))
    ref = ref / np.sqrt((hX * hX).mean(axis=0, keepdims=True)
                        - hX.mean(axis=0, keepdims=True) ** 2)
    err = np.max(np.abs(hZ - ref))
    print("zscore %dx%d: max |err| vs numpy = %.3g" % (m, n, err))
    assert err < 1e-11
    # standardized columns: mean ~ 0, var ~ 1
    zm = dja.dmean_dims(Z, (0,))
    assert np.max(np.abs(zm.collect())) < 1e-12 * m
    for d in (X, mu, sig, Z, zm):
        d.close()
    print("ok")


if __name__ == "__main__":
    main()
