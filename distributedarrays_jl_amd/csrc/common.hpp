// common.hpp — shared state & error plumbing for libdarray_hip.so.
// MI355X-native (gfx950) implementation; no CUDA compatibility paths.
#pragma once
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <stdint.h>
#include <stdio.h>
#include <string.h>
#include <mutex>
#include <unordered_map>

#include "darray_hip.h"

namespace da {

// Global per-process state: one device, one stream, one communicator
// (stream order == remotecall_wait order).
struct State {
    bool inited = false;
    int device = 0;
    int rank = 0;
    int nranks = 1;
    hipStream_t stream = nullptr;
    // second stream for point-to-point overlap (matmul partial exchange
    // rides xGMI while the next local GEMM runs); p2p ops route here
    // while p2p_comm is set (da_p2p_stream)
    hipStream_t comm_stream = nullptr;
    bool p2p_comm = false;
    hipEvent_t ev_main = nullptr, ev_comm = nullptr;
    ncclComm_t comm = nullptr;
    // small persistent device scratch for scalar allreduce / reduce outputs
    void* scratch = nullptr;
    size_t scratch_bytes = 0;
    // reduction partials buffer
    void* partials = nullptr;
    size_t partials_bytes = 0;
    // fused-reduce fan-in ticket (dedicated 4 bytes, zeroed at init;
    // must NOT live in `partials`, which dims-reduce reuses as slabs)
    unsigned int* red_ticket = nullptr;
    std::mutex mem_mtx;
    std::unordered_map<void*, uint64_t> allocs;
    uint64_t bytes_in_use = 0;
    // caching allocator: freed chunks kept per exact (256 B-rounded) size
    // — hipMalloc/hipFree of multi-GiB staging buffers costs ~100 ms,
    // which dominated the matmul leg (profiles/r01_kernel_stats.md)
    std::unordered_map<uint64_t, std::vector<void*>> pool;
    uint64_t pool_bytes = 0;
};

State& st();

// error reporting: negative codes; message captured in a buffer
extern char g_errbuf[1024];
int set_err(int code, const char* fmt, ...);

#define DA_CHECK_HIP(expr) do {                                          \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess)                                                \
        return da::set_err(-(1000 + (int)_e), "%s:%d hip error: %s",     \
                           __FILE__, __LINE__, hipGetErrorString(_e));   \
} while (0)

#define DA_CHECK_NCCL(expr) do {                                         \
    ncclResult_t _e = (expr);                                            \
    if (_e != ncclSuccess)                                               \
        return da::set_err(-(2000 + (int)_e), "%s:%d rccl error: %s",    \
                           __FILE__, __LINE__, ncclGetErrorString(_e));  \
} while (0)

#define DA_REQUIRE_INIT() do {                                           \
    if (!da::st().inited)                                                \
        return da::set_err(-1, "libdarray_hip: da_init() not called");   \
} while (0)

inline size_t dtype_size(int dtype) {
    switch (dtype) {
        case DA_F64: return 8;
        case DA_F32: return 4;
        case DA_I64: return 8;
    }
    return 0;
}

int ensure_scratch(size_t bytes);
int ensure_partials(size_t bytes);

// kernel-side entry points (implemented in kernels_*.hip)
int launch_fill(void* chunk, double v, uint64_t n, int dtype, hipStream_t s);
int launch_rand(void* chunk, uint64_t n, int dtype, uint64_t seed, int kind,
                uint64_t offset, hipStream_t s);
int launch_map(int opcode, void* dst, const void* src, uint64_t n, int dtype,
               hipStream_t s);
int launch_map2(int opcode, void* dst, const void* a, const void* b,
                uint64_t n, int dtype, hipStream_t s);
int launch_bcast_fma(void* d, const void* a, const void* b, double c,
                     uint64_t n, int dtype, hipStream_t s);
int launch_map2_scalar(int opcode, void* dst, const void* src, double c,
                       int rev, uint64_t n, int dtype, hipStream_t s);
int launch_expr(const int32_t* prog, int plen, void* dst,
                const uint64_t* dst_dims, int nd,
                void* const* srcs, const uint64_t* src_strides, int nsrcs,
                const double* consts, int nconsts,
                uint64_t n, int dtype, hipStream_t s);
// hipRTC path (expr_jit.hip): 0 = launched, 1 = unavailable (use the
// interpreter), <0 = error
int launch_expr_jit(const int32_t* prog, int plen, void* dst,
                    const uint64_t* dst_dims, int nd,
                    void* const* srcs, const uint64_t* src_strides,
                    int nsrcs, const double* consts, int nconsts,
                    uint64_t n, int dtype, hipStream_t s);
int expr_jit_state();
const char* expr_jit_err();
int launch_transpose(void* dst, const void* src, uint64_t m, uint64_t n,
                     int dtype, hipStream_t s);
int launch_diag_scale(void* a, uint64_t m, uint64_t n, const void* diag,
                      int side, int dtype, hipStream_t s);
int launch_axpby(void* y, const void* x, double alpha, double beta,
                 uint64_t n, int dtype, hipStream_t s);
int launch_add(void* dest, const void* src, double scale, uint64_t n,
               int dtype, hipStream_t s);
int launch_scale(void* a, double s_, uint64_t n, int dtype, hipStream_t s);
int launch_cast(void* dst, int dst_dtype, const void* src, int src_dtype,
                uint64_t n, hipStream_t s);
int launch_reduce(int mapop, int redop, const void* src, uint64_t n,
                  int dtype, void* out_host, hipStream_t s);
int launch_reduce_dims(int mapop, int redop, const void* src,
                       uint64_t inner, uint64_t axis, uint64_t outer,
                       int dtype, void* dst, hipStream_t s);
int launch_gemm_f64(void* C, const void* A, const void* B,
                    int64_t m, int64_t n, int64_t k,
                    int64_t lda, int64_t ldb, int64_t ldc,
                    double alpha, double beta, hipStream_t s);
int launch_gemm_i64(void* C, const void* A, const void* B,
                    int64_t m, int64_t n, int64_t k,
                    int64_t lda, int64_t ldb, int64_t ldc,
                    int64_t alpha, int64_t beta, hipStream_t s);
int launch_gemm_f32(void* C, const void* A, const void* B,
                    int64_t m, int64_t n, int64_t k,
                    int64_t lda, int64_t ldb, int64_t ldc,
                    double alpha, double beta, hipStream_t s);
int dbg_mfma_probe_impl(const void* A, const void* B, void* out_c,
                        void* out_raw, hipStream_t s);
int dbg_mfma_probe_f32_impl(const void* A, const void* B, void* out_raw,
                            hipStream_t s);

} // namespace da
