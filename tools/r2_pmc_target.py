#!/usr/bin/env python3
"""Minimal kernel sequence for PMC counter passes (rocprofv3 --pmc):
3 launches each of the round-2 kernels whose traffic we attribute —
ejit sin-fused (24 B/elem algorithmic), map sin (16 B/elem), ejit
strided de-mean (16 B/elem) — on 2^28 f64.  Few launches keep the
counter-replay pass short."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja
from distributedarrays_jl_amd import expr as E
from distributedarrays_jl_amd._ffi import lib, check

dja.comm.init()
n = 1 << 28
A = dja.drand((n,), "f64")
B = dja.drand((n,), "f64")
D = dja.dzeros((n,), "f64")
e = E.sin(E.ref(A)) + E.ref(B) * 0.5
E.materialize_(D, e)          # warm (JIT compile outside the counted 3)
for _ in range(3):
    E.materialize_(D, e)
for _ in range(3):
    dja.map_("sin", D, A)
m = 16384
X = dja.drand((m, m), "f64")
M = dja.dmean_dims(X, (0,))
D2 = dja.DArray((m, m), "f64")
em = E.ref(X) - E.ref(M)
E.materialize_(D2, em)
for _ in range(3):
    E.materialize_(D2, em)
check(lib.da_synchronize())
dja.d_closeall()
print("pmc target done; jit state", int(lib.da_expr_jit_state()))
