// abi.hip — C-ABI entry points of libdarray_hip.so (see include/darray_hip.h
// for the reference-seam citations per function).
#include "common.hpp"
#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <unistd.h>
#include <errno.h>
#include <time.h>

namespace da {

static State g_state;
State& st() { return g_state; }

char g_errbuf[1024] = "ok";

int set_err(int code, const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(g_errbuf, sizeof(g_errbuf), fmt, ap);
    va_end(ap);
    return code;
}

int ensure_scratch(size_t bytes) {
    if (st().scratch_bytes >= bytes) return 0;
    if (st().scratch) hipFree(st().scratch);
    st().scratch = nullptr;
    st().scratch_bytes = 0;
    DA_CHECK_HIP(hipMalloc(&st().scratch, bytes));
    st().scratch_bytes = bytes;
    return 0;
}

int ensure_partials(size_t bytes) {
    if (st().partials_bytes >= bytes) return 0;
    if (st().partials) hipFree(st().partials);
    st().partials = nullptr;
    st().partials_bytes = 0;
    DA_CHECK_HIP(hipMalloc(&st().partials, bytes));
    st().partials_bytes = bytes;
    return 0;
}

static ncclDataType_t nccl_dtype(int dtype) {
    switch (dtype) {
    case DA_F64: return ncclDouble;
    case DA_F32: return ncclFloat;
    default: return ncclInt64;
    }
}

static ncclRedOp_t nccl_redop(int redop) {
    switch (redop) {
    case DA_RED_ADD: return ncclSum;
    case DA_RED_MUL: return ncclProd;
    case DA_RED_MIN: return ncclMin;
    default: return ncclMax;
    }
}

} // namespace da

using namespace da;

extern "C" {

static uint64_t round_sz(uint64_t nbytes);
static int pool_trim_locked();

/* ---- lifecycle ------------------------------------------------------- */

int da_init(int device, int rank, int nranks, const char* rccl_uid_path) {
    if (st().inited) return 0;
    if (nranks < 1 || rank < 0 || rank >= nranks)
        return set_err(-2, "da_init: bad rank %d/%d", rank, nranks);
    DA_CHECK_HIP(hipSetDevice(device));
    DA_CHECK_HIP(hipStreamCreateWithFlags(&st().stream, hipStreamNonBlocking));
    DA_CHECK_HIP(hipStreamCreateWithFlags(&st().comm_stream,
                                          hipStreamNonBlocking));
    DA_CHECK_HIP(hipEventCreateWithFlags(&st().ev_main,
                                         hipEventDisableTiming));
    DA_CHECK_HIP(hipEventCreateWithFlags(&st().ev_comm,
                                         hipEventDisableTiming));
    DA_CHECK_HIP(hipMalloc(&st().red_ticket, sizeof(unsigned int)));
    DA_CHECK_HIP(hipMemset(st().red_ticket, 0, sizeof(unsigned int)));
    st().device = device;
    st().rank = rank;
    st().nranks = nranks;
    if (nranks > 1) {
        if (!rccl_uid_path)
            return set_err(-2, "da_init: nranks>1 needs a uid path");
        ncclUniqueId uid;
        if (rank == 0) {
            DA_CHECK_NCCL(ncclGetUniqueId(&uid));
            char tmp[1024];
            snprintf(tmp, sizeof(tmp), "%s.tmp", rccl_uid_path);
            FILE* f = fopen(tmp, "wb");
            if (!f) return set_err(-4, "da_init: cannot write %s", tmp);
            fwrite(&uid, sizeof(uid), 1, f);
            fclose(f);
            if (rename(tmp, rccl_uid_path) != 0)
                return set_err(-4, "da_init: rename failed: %s",
                               strerror(errno));
        } else {
            // poll for the rendezvous file (shared filesystem, one node)
            FILE* f = nullptr;
            for (int i = 0; i < 1200 && !f; ++i) {   // up to 120 s
                f = fopen(rccl_uid_path, "rb");
                if (!f) usleep(100000);
            }
            if (!f) return set_err(-4, "da_init: uid file never appeared");
            size_t got = fread(&uid, 1, sizeof(uid), f);
            fclose(f);
            if (got != sizeof(uid))
                return set_err(-4, "da_init: short uid file");
        }
        DA_CHECK_NCCL(ncclCommInitRank(&st().comm, nranks, uid, rank));
    }
    st().inited = true;
    return 0;
}

int da_shutdown(void) {
    if (!st().inited) return 0;
    if (st().comm) { ncclCommDestroy(st().comm); st().comm = nullptr; }
    if (st().scratch) { hipFree(st().scratch); st().scratch = nullptr; }
    if (st().partials) { hipFree(st().partials); st().partials = nullptr; }
    st().scratch_bytes = st().partials_bytes = 0;
    {
        std::lock_guard<std::mutex> g(st().mem_mtx);
        for (auto& kv : st().allocs) (void)hipFree(kv.first);
        st().allocs.clear();
        st().bytes_in_use = 0;
        (void)pool_trim_locked();
    }
    if (st().red_ticket) { (void)hipFree(st().red_ticket);
                           st().red_ticket = nullptr; }
    if (st().ev_main) { (void)hipEventDestroy(st().ev_main); st().ev_main = nullptr; }
    if (st().ev_comm) { (void)hipEventDestroy(st().ev_comm); st().ev_comm = nullptr; }
    if (st().comm_stream) { (void)hipStreamDestroy(st().comm_stream);
                            st().comm_stream = nullptr; }
    if (st().stream) { (void)hipStreamDestroy(st().stream); st().stream = nullptr; }
    st().inited = false;
    return 0;
}

int da_rank(void) { return st().rank; }
int da_nranks(void) { return st().nranks; }

/* ---- memory ----------------------------------------------------------- */

static uint64_t round_sz(uint64_t nbytes) {
    if (nbytes == 0) nbytes = 1;   // empty chunks keep a real handle
    return (nbytes + 255) & ~(uint64_t)255;
}

static int pool_trim_locked() {
    for (auto& kv : st().pool)
        for (void* p : kv.second)
            if (hipFree(p) != hipSuccess)
                return set_err(-5, "da_pool_trim: hipFree failed");
    st().pool.clear();
    st().pool_bytes = 0;
    return 0;
}

int da_alloc(uint64_t nbytes, int dtype, void** chunk) {
    DA_REQUIRE_INIT();
    if (!chunk) return set_err(-3, "da_alloc: null out");
    (void)dtype;
    uint64_t sz = round_sz(nbytes);
    void* p = nullptr;
    {
        std::lock_guard<std::mutex> g(st().mem_mtx);
        auto it = st().pool.find(sz);
        if (it != st().pool.end() && !it->second.empty()) {
            p = it->second.back();
            it->second.pop_back();
            st().pool_bytes -= sz;
        }
    }
    if (!p) {
        hipError_t e = hipMalloc(&p, sz);
        if (e != hipSuccess) {   // trim the cache and retry once
            std::lock_guard<std::mutex> g(st().mem_mtx);
            int rc = pool_trim_locked();
            if (rc) return rc;
            e = hipMalloc(&p, sz);
            if (e != hipSuccess)
                return set_err(-(1000 + (int)e), "da_alloc(%llu): %s",
                               (unsigned long long)sz,
                               hipGetErrorString(e));
        }
    }
    {
        std::lock_guard<std::mutex> g(st().mem_mtx);
        st().allocs[p] = sz;
        st().bytes_in_use += sz;
    }
    *chunk = p;
    return 0;
}

int da_free(void* chunk) {
    DA_REQUIRE_INIT();
    if (!chunk) return 0;
    std::lock_guard<std::mutex> g(st().mem_mtx);
    auto it = st().allocs.find(chunk);
    if (it == st().allocs.end())
        return set_err(-3, "da_free: unknown chunk %p", chunk);
    uint64_t sz = it->second;
    st().bytes_in_use -= sz;
    st().allocs.erase(it);
    st().pool[sz].push_back(chunk);   // cache; release via da_pool_trim
    st().pool_bytes += sz;
    return 0;
}

int da_pool_trim(void) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipStreamSynchronize(st().stream));
    std::lock_guard<std::mutex> g(st().mem_mtx);
    return pool_trim_locked();
}

uint64_t da_pool_bytes(void) {
    std::lock_guard<std::mutex> g(st().mem_mtx);
    return st().pool_bytes;
}

uint64_t da_bytes_in_use(void) {
    std::lock_guard<std::mutex> g(st().mem_mtx);
    return st().bytes_in_use;
}

int da_h2d(void* chunk, const void* host, uint64_t nbytes) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipMemcpyAsync(chunk, host, nbytes, hipMemcpyHostToDevice,
                                st().stream));
    DA_CHECK_HIP(hipStreamSynchronize(st().stream));
    return 0;
}

int da_d2h(const void* chunk, void* host, uint64_t nbytes) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipMemcpyAsync(host, chunk, nbytes, hipMemcpyDeviceToHost,
                                st().stream));
    DA_CHECK_HIP(hipStreamSynchronize(st().stream));
    return 0;
}

int da_d2d(void* dst, const void* src, uint64_t nbytes) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipMemcpyAsync(dst, src, nbytes, hipMemcpyDeviceToDevice,
                                st().stream));
    return 0;
}

int da_copy2d(void* dst, uint64_t dpitch, const void* src, uint64_t spitch,
              uint64_t width, uint64_t height) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipMemcpy2DAsync(dst, dpitch, src, spitch, width, height,
                                  hipMemcpyDeviceToDevice, st().stream));
    return 0;
}

/* ---- constructors / elementwise / reductions ------------------------- */

int da_fill(void* chunk, double v, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_fill(chunk, v, n, dtype, st().stream);
}

int da_rand(void* chunk, uint64_t n, int dtype, uint64_t seed, int kind,
            uint64_t offset) {
    DA_REQUIRE_INIT();
    return launch_rand(chunk, n, dtype, seed, kind, offset, st().stream);
}

int da_map(int opcode, void* dst, const void* src, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_map(opcode, dst, src, n, dtype, st().stream);
}

int da_map2(int opcode, void* dst, const void* a, const void* b,
            uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_map2(opcode, dst, a, b, n, dtype, st().stream);
}

int da_map2_scalar(int opcode, void* dst, const void* src, double c,
                   int reverse, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_map2_scalar(opcode, dst, src, c, reverse, n, dtype,
                              st().stream);
}

int da_bcast_fma(void* d, const void* a, const void* b, double c,
                 uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_bcast_fma(d, a, b, c, n, dtype, st().stream);
}

int da_expr(const int32_t* prog, int prog_len, void* dst,
            const uint64_t* dst_dims, int nd,
            void* const* srcs, const uint64_t* src_strides, int nsrcs,
            const double* consts, int nconsts, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_expr(prog, prog_len, dst, dst_dims, nd, srcs,
                       src_strides, nsrcs, consts, nconsts, n, dtype,
                       st().stream);
}

/* 1 = JIT ready (unused yet), 2 = active, -1 = failed (interpreter);
 * DA_EXPR_JIT=0 disables per call.  da_expr_jit_errstr() holds the
 * last hipRTC diagnostic. */
int da_expr_jit_state(void) { return expr_jit_state(); }
const char* da_expr_jit_errstr(void) { return expr_jit_err(); }

int da_axpby(void* y, const void* x, double alpha, double beta,
             uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_axpby(y, x, alpha, beta, n, dtype, st().stream);
}

int da_add(void* dest, const void* src, double scale, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_add(dest, src, scale, n, dtype, st().stream);
}

int da_scale(void* a, double s, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    return launch_scale(a, s, n, dtype, st().stream);
}

int da_cast(void* dst, int dst_dtype, const void* src, int src_dtype,
            uint64_t n) {
    DA_REQUIRE_INIT();
    return launch_cast(dst, dst_dtype, src, src_dtype, n, st().stream);
}

int da_reduce(int mapop, int redop, const void* src, uint64_t n, int dtype,
              void* out) {
    DA_REQUIRE_INIT();
    return launch_reduce(mapop, redop, src, n, dtype, out, st().stream);
}

int da_reduce_dims(int mapop, int redop, const void* src, uint64_t inner,
                   uint64_t axis, uint64_t outer, int dtype, void* dst) {
    DA_REQUIRE_INIT();
    return launch_reduce_dims(mapop, redop, src, inner, axis, outer,
                              dtype, dst, st().stream);
}

int da_allreduce(void* inout, int count, int dtype, int redop) {
    DA_REQUIRE_INIT();
    if (count <= 0) return set_err(-3, "da_allreduce: bad count");
    size_t bytes = (size_t)count * dtype_size(dtype);
    if (st().nranks == 1) return 0;   // single-rank fold is the local value
    int rc = ensure_scratch(bytes);
    if (rc) return rc;
    DA_CHECK_HIP(hipMemcpyAsync(st().scratch, inout, bytes,
                                hipMemcpyHostToDevice, st().stream));
    DA_CHECK_NCCL(ncclAllReduce(st().scratch, st().scratch, count,
                                nccl_dtype(dtype), nccl_redop(redop),
                                st().comm, st().stream));
    DA_CHECK_HIP(hipMemcpyAsync(inout, st().scratch, bytes,
                                hipMemcpyDeviceToHost, st().stream));
    DA_CHECK_HIP(hipStreamSynchronize(st().stream));
    return 0;
}

/* ---- linear algebra --------------------------------------------------- */

int da_gemm_f64(void* C, const void* A, const void* B,
                int64_t m, int64_t n, int64_t k,
                int64_t lda, int64_t ldb, int64_t ldc,
                double alpha, double beta) {
    DA_REQUIRE_INIT();
    return launch_gemm_f64(C, A, B, m, n, k, lda, ldb, ldc, alpha, beta,
                           st().stream);
}

int da_gemm_f32(void* C, const void* A, const void* B,
                int64_t m, int64_t n, int64_t k,
                int64_t lda, int64_t ldb, int64_t ldc,
                double alpha, double beta) {
    DA_REQUIRE_INIT();
    return launch_gemm_f32(C, A, B, m, n, k, lda, ldb, ldc, alpha, beta,
                           st().stream);
}

int da_gemm_i64(void* C, const void* A, const void* B,
                int64_t m, int64_t n, int64_t k,
                int64_t lda, int64_t ldb, int64_t ldc,
                int64_t alpha, int64_t beta) {
    DA_REQUIRE_INIT();
    return launch_gemm_i64(C, A, B, m, n, k, lda, ldb, ldc, alpha, beta,
                           st().stream);
}

int da_transpose(void* dst, const void* src, uint64_t m, uint64_t n,
                 int dtype) {
    DA_REQUIRE_INIT();
    return launch_transpose(dst, src, m, n, dtype, st().stream);
}

int da_diag_scale(void* a, uint64_t m, uint64_t n, const void* diag,
                  int side, int dtype) {
    DA_REQUIRE_INIT();
    return launch_diag_scale(a, m, n, diag, side, dtype, st().stream);
}

/* debug-only: MFMA lane-map probe (not part of the public ABI) */
int dbg_mfma_probe_f64(const void* A, const void* B, void* out_c,
                       void* out_raw) {
    DA_REQUIRE_INIT();
    return dbg_mfma_probe_impl(A, B, out_c, out_raw, st().stream);
}

int dbg_mfma_probe_f32(const void* A, const void* B, void* out_raw) {
    DA_REQUIRE_INIT();
    return dbg_mfma_probe_f32_impl(A, B, out_raw, st().stream);
}

/* ---- point-to-point --------------------------------------------------- */

int da_group_start(void) {
    DA_REQUIRE_INIT();
    DA_CHECK_NCCL(ncclGroupStart());
    return 0;
}

int da_group_end(void) {
    DA_REQUIRE_INIT();
    DA_CHECK_NCCL(ncclGroupEnd());
    return 0;
}

static hipStream_t p2p_stream() {
    return st().p2p_comm ? st().comm_stream : st().stream;
}

int da_send(const void* buf, uint64_t nbytes, int peer) {
    DA_REQUIRE_INIT();
    if (!st().comm) return set_err(-2, "da_send: no communicator");
    DA_CHECK_NCCL(ncclSend(buf, nbytes, ncclChar, peer, st().comm,
                           p2p_stream()));
    return 0;
}

int da_recv(void* buf, uint64_t nbytes, int peer) {
    DA_REQUIRE_INIT();
    if (!st().comm) return set_err(-2, "da_recv: no communicator");
    DA_CHECK_NCCL(ncclRecv(buf, nbytes, ncclChar, peer, st().comm,
                           p2p_stream()));
    return 0;
}

int da_p2p_stream(int use_comm) {
    DA_REQUIRE_INIT();
    st().p2p_comm = (use_comm != 0);
    return 0;
}

int da_comm_after_compute(void) {
    /* comm stream waits for everything queued so far on the main stream */
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipEventRecord(st().ev_main, st().stream));
    DA_CHECK_HIP(hipStreamWaitEvent(st().comm_stream, st().ev_main, 0));
    return 0;
}

int da_main_after_comm(void) {
    /* main stream waits for everything queued so far on the comm stream */
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipEventRecord(st().ev_comm, st().comm_stream));
    DA_CHECK_HIP(hipStreamWaitEvent(st().stream, st().ev_comm, 0));
    return 0;
}

int da_comm_sync(void) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipStreamSynchronize(st().comm_stream));
    return 0;
}

int da_sendrecv(const void* sbuf, int peer_s, void* rbuf, int peer_r,
                uint64_t nbytes) {
    DA_REQUIRE_INIT();
    if (!st().comm) return set_err(-2, "da_sendrecv: no communicator");
    DA_CHECK_NCCL(ncclGroupStart());
    DA_CHECK_NCCL(ncclSend(sbuf, nbytes, ncclChar, peer_s, st().comm,
                           p2p_stream()));
    DA_CHECK_NCCL(ncclRecv(rbuf, nbytes, ncclChar, peer_r, st().comm,
                           p2p_stream()));
    DA_CHECK_NCCL(ncclGroupEnd());
    return 0;
}

int da_bcast(void* buf, uint64_t nbytes, int root) {
    DA_REQUIRE_INIT();
    if (st().nranks == 1) return 0;
    DA_CHECK_NCCL(ncclBroadcast(buf, buf, nbytes, ncclChar, root, st().comm,
                                st().stream));
    return 0;
}

int da_barrier(void) {
    DA_REQUIRE_INIT();
    if (st().nranks == 1) return 0;
    int rc = ensure_scratch(8);
    if (rc) return rc;
    DA_CHECK_NCCL(ncclAllReduce(st().scratch, st().scratch, 1, ncclDouble,
                                ncclSum, st().comm, st().stream));
    DA_CHECK_HIP(hipStreamSynchronize(st().stream));
    return 0;
}

/* ---- stream & timing --------------------------------------------------*/

int da_synchronize(void) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipStreamSynchronize(st().stream));
    return 0;
}

int da_event_create(void** ev) {
    DA_REQUIRE_INIT();
    hipEvent_t e;
    DA_CHECK_HIP(hipEventCreate(&e));
    *ev = (void*)e;
    return 0;
}

int da_event_record(void* ev) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipEventRecord((hipEvent_t)ev, st().stream));
    return 0;
}

int da_event_elapsed(void* ev_start, void* ev_stop, float* ms) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipEventSynchronize((hipEvent_t)ev_stop));
    DA_CHECK_HIP(hipEventElapsedTime(ms, (hipEvent_t)ev_start,
                                     (hipEvent_t)ev_stop));
    return 0;
}

int da_event_destroy(void* ev) {
    DA_REQUIRE_INIT();
    DA_CHECK_HIP(hipEventDestroy((hipEvent_t)ev));
    return 0;
}

/* ---- errors / props ---------------------------------------------------- */

const char* da_errstr(int code) {
    (void)code;
    return g_errbuf;
}

int da_device_props(char* name, int name_len, uint64_t* hbm_bytes) {
    DA_REQUIRE_INIT();
    hipDeviceProp_t p;
    DA_CHECK_HIP(hipGetDeviceProperties(&p, st().device));
    if (name && name_len > 0) {
        strncpy(name, p.name, name_len - 1);
        name[name_len - 1] = 0;
    }
    if (hbm_bytes) *hbm_bytes = (uint64_t)p.totalGlobalMem;
    return 0;
}

} // extern "C"
