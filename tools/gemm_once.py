import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import distributedarrays_jl_amd as dja
from distributedarrays_jl_amd._ffi import lib, check
dja.comm.init()
n = 8192
A = dja.DArray((n, n), "f64"); A.rand_()
B = dja.DArray((n, n), "f64"); B.rand_()
C = dja.DArray((n, n), "f64"); C.fill_(0.0)
for _ in range(3):
    check(lib.da_gemm_f64(C._ptr(), A._ptr(), B._ptr(), n, n, n,
                          n, n, n, 1.0, 0.0))
check(lib.da_synchronize())
print("gemm done")
