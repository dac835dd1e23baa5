"""ctypes binding to libdarray_hip.so — the C-ABI boundary of
include/darray_hip.h.

The extension is the product's ONLY compute path: if the shared library
is missing the import fails loudly (no CPU fallback exists anywhere in
this package — see BASELINE.json north_star / SURVEY.md §8b)."""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "libdarray_hip.so")

if not os.path.exists(_SO):
    raise ImportError(
        "libdarray_hip.so is missing (%s). Build it first: "
        "`python -c \"import __graft_entry__; __graft_entry__.build()\"` "
        "or `make -C distributedarrays_jl_amd/csrc`. There is no CPU "
        "fallback: the HIP extension IS the compute path." % _SO)

# Load the system ROCm HIP runtime FIRST: torch wheels bundle their own
# libamdhip64.so.7 (rocm7.0) and whichever object owns that soname first
# wins the process; binding our gfx950 library against torch's bundled
# runtime makes device enumeration fail.  Importing this package before
# torch pins the /opt/rocm runtime (bench.py and tests do so).
try:
    ctypes.CDLL("/opt/rocm/lib/libamdhip64.so", mode=ctypes.RTLD_GLOBAL)
except OSError:
    pass

lib = ctypes.CDLL(_SO)

u64 = ctypes.c_uint64
i64 = ctypes.c_int64
i32 = ctypes.c_int
f64 = ctypes.c_double
ptr = ctypes.c_void_p

_sigs = {
    "da_init": ([i32, i32, i32, ctypes.c_char_p], i32),
    "da_shutdown": ([], i32),
    "da_rank": ([], i32),
    "da_nranks": ([], i32),
    "da_alloc": ([u64, i32, ctypes.POINTER(ptr)], i32),
    "da_free": ([ptr], i32),
    "da_pool_trim": ([], i32),
    "da_pool_bytes": ([], u64),
    "da_h2d": ([ptr, ptr, u64], i32),
    "da_d2h": ([ptr, ptr, u64], i32),
    "da_d2d": ([ptr, ptr, u64], i32),
    "da_copy2d": ([ptr, u64, ptr, u64, u64, u64], i32),
    "da_fill": ([ptr, f64, u64, i32], i32),
    "da_rand": ([ptr, u64, i32, u64, i32, u64], i32),
    "da_map": ([i32, ptr, ptr, u64, i32], i32),
    "da_map2": ([i32, ptr, ptr, ptr, u64, i32], i32),
    "da_bcast_fma": ([ptr, ptr, ptr, f64, u64, i32], i32),
    "da_expr": ([ptr, i32, ptr, ptr, i32, ptr, ptr, i32, ptr, i32,
                 u64, i32], i32),
    "da_expr_jit_state": ([], i32),
    "da_expr_jit_errstr": ([], ctypes.c_char_p),
    "da_map2_scalar": ([i32, ptr, ptr, f64, i32, u64, i32], i32),
    "da_axpby": ([ptr, ptr, f64, f64, u64, i32], i32),
    "da_add": ([ptr, ptr, f64, u64, i32], i32),
    "da_scale": ([ptr, f64, u64, i32], i32),
    "da_cast": ([ptr, i32, ptr, i32, u64], i32),
    "da_reduce": ([i32, i32, ptr, u64, i32, ptr], i32),
    "da_reduce_dims": ([i32, i32, ptr, u64, u64, u64, i32, ptr], i32),
    "da_transpose": ([ptr, ptr, u64, u64, i32], i32),
    "da_diag_scale": ([ptr, u64, u64, ptr, i32, i32], i32),
    "da_sort": ([ptr, u64, i32], i32),
    "da_sort_out": ([ptr, ptr, u64, i32], i32),
    "da_lower_bound": ([ptr, u64, i32, ptr, i32,
                        ctypes.POINTER(ctypes.c_uint64)], i32),
    "da_allreduce": ([ptr, i32, i32, i32], i32),
    "da_gemm_f64": ([ptr, ptr, ptr, i64, i64, i64, i64, i64, i64, f64, f64],
                    i32),
    "da_gemm_f32": ([ptr, ptr, ptr, i64, i64, i64, i64, i64, i64, f64, f64],
                    i32),
    "da_gemm_i64": ([ptr, ptr, ptr, i64, i64, i64, i64, i64, i64, i64, i64],
                    i32),
    "da_group_start": ([], i32),
    "da_group_end": ([], i32),
    "da_send": ([ptr, u64, i32], i32),
    "da_recv": ([ptr, u64, i32], i32),
    "da_sendrecv": ([ptr, i32, ptr, i32, u64], i32),
    "da_p2p_stream": ([i32], i32),
    "da_comm_after_compute": ([], i32),
    "da_main_after_comm": ([], i32),
    "da_comm_sync": ([], i32),
    "da_bcast": ([ptr, u64, i32], i32),
    "da_barrier": ([], i32),
    "da_synchronize": ([], i32),
    "da_event_create": ([ctypes.POINTER(ptr)], i32),
    "da_event_record": ([ptr], i32),
    "da_event_elapsed": ([ptr, ptr, ctypes.POINTER(ctypes.c_float)], i32),
    "da_event_destroy": ([ptr], i32),
    "da_errstr": ([i32], ctypes.c_char_p),
    "da_device_props": ([ctypes.c_char_p, i32, ctypes.POINTER(u64)], i32),
    "da_bytes_in_use": ([], u64),
}

for name, (args, res) in _sigs.items():
    fn = getattr(lib, name)
    fn.argtypes = args
    fn.restype = res


class DArrayError(RuntimeError):
    pass


def check(code):
    if code != 0:
        msg = lib.da_errstr(code)
        raise DArrayError("libdarray_hip error %d: %s"
                          % (code, (msg or b"?").decode()))
    return code
