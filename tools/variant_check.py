"""Parity check of every env-gated kernel variant (documented
experiments must not rot): spawns one subprocess per variant on the GPU
box, each running a small gemm + sum parity check."""
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CHECK = r'''
import sys, os
sys.path.insert(0, %r)
import numpy as np
import distributedarrays_jl_amd as dja
from oracle import philox
dja.comm.init()
m = 256
A = np.asfortranarray(philox.fill_uniform_f64(m*m, 1).reshape(m, m, order="F"))
B = np.asfortranarray(philox.fill_uniform_f64(m*m, 2).reshape(m, m, order="F"))
dA, dB = dja.distribute(A), dja.distribute(B)
C = dja.dmatmul(dA, dB)
ref = A @ B
assert np.abs(C.localpart() - ref).max() / np.abs(ref).max() < 1e-12
x = philox.fill_uniform_f64(100001, 3)
d = dja.distribute(x)
assert abs(dja.dsum(d) - x.sum()) < 1e-6
from distributedarrays_jl_amd import expr as E
o = E.materialize(E.ref(d) * 2.0 + 1.0)
assert np.array_equal(o.localpart(), x * 2.0 + 1.0)
print("variant ok:", {k: v for k, v in os.environ.items() if k.startswith("DA_")})
'''

VARIANTS = [
    {"DA_GEMM_V": "1"}, {"DA_GEMM_V": "2"}, {"DA_GEMM_V": "3"},
    {"DA_GEMM_V": "5"}, {"DA_GEMM_V": "2", "DA_GEMM_BK": "32"},
    {"DA_RED_FUSED": "1"}, {"DA_NT": "1"}, {"DA_RV4": "1"},
    {"DA_RBLOCKS": "4096"}, {"DA_MM_OVERLAP": "0"},
    {"DA_EXPR_JIT": "0"},
]


def main():
    fails = 0
    for env in VARIANTS:
        e = dict(os.environ)
        e.update(env)
        r = subprocess.run([sys.executable, "-c", CHECK % ROOT], env=e,
                           capture_output=True, text=True, timeout=180)
        tag = ",".join("%s=%s" % kv for kv in env.items())
        if r.returncode != 0:
            print("FAIL", tag, r.stderr[-300:])
            fails += 1
        else:
            print("ok  ", tag)
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
