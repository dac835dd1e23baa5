"""ctypes wrapper over libcpu_baseline.so (ORACLE NOTICE: test/bench
infrastructure only — the timed `cpu_baseline` leg of bench.py and its
pinning tests; never the product path)."""
import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "libcpu_baseline.so")


def load():
    if not os.path.exists(_SO):
        import subprocess
        subprocess.run(["make"], cwd=_HERE, check=True,
                       capture_output=True)
    lib = ctypes.CDLL(_SO)
    u64, f64, f32 = ctypes.c_uint64, ctypes.c_double, ctypes.c_float
    p = ctypes.c_void_p
    lib.cb_fill_uniform_f64.argtypes = [p, u64, u64]
    lib.cb_fill_uniform_f32.argtypes = [p, u64, u64]
    lib.cb_sum_f64.argtypes = [p, u64]
    lib.cb_sum_f64.restype = f64
    lib.cb_abs2_sum_f32.argtypes = [p, u64]
    lib.cb_abs2_sum_f32.restype = f32
    lib.cb_map_sin_f64.argtypes = [p, p, u64]
    lib.cb_bcast_fma_f64.argtypes = [p, p, p, f64, u64]
    lib.cb_gemm_f64.argtypes = [p, p, p, ctypes.c_int64, ctypes.c_int64,
                                ctypes.c_int64]
    lib.cb_num_threads.restype = ctypes.c_int
    return lib


def ptr(a):
    return a.ctypes.data_as(ctypes.c_void_p)


def fill_uniform_f64(lib, n, seed):
    a = np.empty(n, dtype=np.float64)
    lib.cb_fill_uniform_f64(ptr(a), n, seed)
    return a
