"""Column statistics + median of a distributed matrix — exercises
dims-reductions, predicate counts and the distributed samplesort.

Run on an MI355X box:  python examples/column_stats.py [m] [n]
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja


def column_stats(m=8192, n=4096):
    A = dja.drand((m, n), "f64")
    means = dja.dmean_dims(A, (0,))           # 1 x n column means
    mx = dja.dmaximum_dims(A, (0,))
    finite = dja.dall("isfinite", A)
    v = dja.drand((m,), "f64")
    s = dja.dsort(v)                           # distributed samplesort
    med = s.getindex(m // 2)
    out = (means.collect().ravel(), mx.collect().ravel(), finite, med)
    for d in (A, means, mx, v, s):
        d.close()
    return out


if __name__ == "__main__":
    m = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
    means, mx, finite, med = column_stats(m, n)
    print("col means ~0.5: %.4f..%.4f; col max ~1: %.4f; finite=%s; "
          "median ~0.5: %.4f" % (means.min(), means.max(), mx.min(),
                                 finite, med))
