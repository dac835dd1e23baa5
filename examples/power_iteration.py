"""Power iteration for the dominant eigenvalue of a random symmetric-ish
matrix — exercises dmatvec, dnorm and scale_ across ranks.

Run on an MI355X box:  python examples/power_iteration.py [n]
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja


def dominant_eig(n=2048, iters=30):
    A = dja.drand((n, n), "f64")
    x = np.ones(n) / np.sqrt(n)
    lam = 0.0
    for _ in range(iters):
        y = dja.dmatvec(A, x)
        yv = y.collect()
        y.close()
        lam = float(np.linalg.norm(yv))
        x = yv / lam
    A.close()
    return lam


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 2048
    lam = dominant_eig(n)
    # uniform [0,1) matrix: dominant eigenvalue ~= n/2
    print("lambda_max ~= %.3f (n=%d, expect ~n/2=%d)" % (lam, n, n // 2))
