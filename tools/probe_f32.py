import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import ctypes
import numpy as np
import distributedarrays_jl_amd as dja
from distributedarrays_jl_amd._ffi import lib
dja.comm.init()
rng = np.random.default_rng(0)
A = np.asfortranarray(rng.uniform(1, 2, (16, 4)).astype(np.float64)).astype(np.float32, order="F")
B = np.asfortranarray(rng.uniform(1, 2, (4, 16)).astype(np.float64)).astype(np.float32, order="F")
A = np.asfortranarray(A); B = np.asfortranarray(B)
ref = (A.astype(np.float64) @ B.astype(np.float64))
raw = np.zeros(256, dtype=np.float32)
def dev(nb):
    p = ctypes.c_void_p(); assert lib.da_alloc(nb, 1, ctypes.byref(p)) == 0; return p
dA, dB, dR = dev(64*4), dev(64*4), dev(256*4)
hp = lambda a: a.ctypes.data_as(ctypes.c_void_p)
assert lib.da_h2d(dA, hp(A), 64*4) == 0
assert lib.da_h2d(dB, hp(B), 64*4) == 0
fn = lib.dbg_mfma_probe_f32; fn.argtypes = [ctypes.c_void_p]*3; fn.restype = ctypes.c_int
assert fn(dA, dB, dR) == 0
assert lib.da_d2h(dR, hp(raw), 256*4) == 0
# hypotheses
h1 = np.zeros((16,16)); h2 = np.zeros((16,16))
for l in range(64):
    for q in range(4):
        v = raw[l*4+q]
        h1[4*q + (l>>4), l & 15] = v      # f64-style map
        h2[(l>>4)*4 + q, l & 15] = v      # documented f32 map
print("h1 (row=4q+l4) match:", np.allclose(h1, ref, rtol=1e-5))
print("h2 (row=4*l4+q) match:", np.allclose(h2, ref, rtol=1e-5))
