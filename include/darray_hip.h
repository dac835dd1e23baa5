/* darray_hip.h — C ABI of libdarray_hip.so: the MI355X-native local-compute
 * path for DistributedArrays.jl-style distributed arrays.
 *
 * This is the drop-in boundary of SURVEY.md §8(b): each entry point replaces
 * one worker-side local operation of the reference (citations below are
 * file:line into /root/reference).  The reference touches localparts only
 * through the init-constructor / localpart / fan-out seam
 * (src/darray.jl:76-118, :330-337; src/broadcast.jl:80; src/mapreduce.jl:31;
 * src/linalg.jl:224), so a Julia caller binds these with plain `ccall`
 *  and keeps every other line of the reference.
 *
 * Conventions:
 *  - one OS process per rank per GPU (1 Julia worker <-> 1 GPU);
 *  - all device work is serialised onto ONE HIP stream per process
 *    (stream order == remotecall_wait order);
 *  - chunks are opaque device pointers owned by the creating rank, freed
 *    exactly once via da_free (hook = release_localpart, src/core.jl:67);
 *  - every function returns int: 0 = ok, <0 = error (da_errstr(code));
 *  - no torch/HIP types cross this ABI: void*, integers, doubles only.
 */
#ifndef DARRAY_HIP_H
#define DARRAY_HIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* dtypes */
enum da_dtype { DA_F64 = 0, DA_F32 = 1, DA_I64 = 2 };

/* unary map opcodes — the C-math subset of the reference's scalar-math
 * test list (/root/reference/test/darray.jl:775-800); names match
 * oracle/ops.py MAP_OPS and distributedarrays_jl_amd/_opcodes.py. */
enum da_mapop {
    DA_OP_IDENTITY = 0, DA_OP_NEG, DA_OP_ABS, DA_OP_ABS2, DA_OP_INV,
    DA_OP_SQRT, DA_OP_CBRT, DA_OP_EXP, DA_OP_EXP2, DA_OP_EXP10,
    DA_OP_EXPM1, DA_OP_LOG, DA_OP_LOG2, DA_OP_LOG10, DA_OP_LOG1P,
    DA_OP_SIN, DA_OP_COS, DA_OP_TAN, DA_OP_ASIN, DA_OP_ACOS, DA_OP_ATAN,
    DA_OP_SINH, DA_OP_COSH, DA_OP_TANH, DA_OP_ASINH, DA_OP_ACOSH,
    DA_OP_ATANH, DA_OP_SINPI, DA_OP_COSPI, DA_OP_FLOOR, DA_OP_CEIL,
    DA_OP_ROUND, DA_OP_TRUNC, DA_OP_SIGN, DA_OP_DEG2RAD, DA_OP_RAD2DEG,
    DA_OP_SEC, DA_OP_CSC, DA_OP_COT,
    /* the remainder of the reference's scalar-math list
     * (test/darray.jl:775-800) that C math / OCML covers */
    DA_OP_ERF, DA_OP_ERFC, DA_OP_ERFINV, DA_OP_ERFCINV, DA_OP_ERFCX,
    DA_OP_GAMMA, DA_OP_LGAMMA, DA_OP_SINC, DA_OP_COSC,
    DA_OP_SIND, DA_OP_COSD, DA_OP_TAND, DA_OP_ASIND, DA_OP_ACOSD,
    DA_OP_ATAND, DA_OP_ACOT, DA_OP_ACOTD, DA_OP_ASEC, DA_OP_ACSC,
    DA_OP_ASECH, DA_OP_ACSCH, DA_OP_ACOTH,
    DA_OP_ISNAN, DA_OP_ISINF, DA_OP_ISFINITE,   /* 0/1-valued (Bool-array analog) */
    DA_OP__N
};

/* binary elementwise opcodes (src/mapreduce.jl:180-189 + broadcast) */
enum da_map2op {
    DA_OP2_ADD = 0, DA_OP2_SUB, DA_OP2_MUL, DA_OP2_DIV, DA_OP2_MIN2,
    DA_OP2_MAX2, DA_OP2_IDIV, DA_OP2_MOD, DA_OP2_REM, DA_OP2_AND,
    DA_OP2_OR, DA_OP2_XOR, DA_OP2_POW, DA_OP2_ATAN2,
    DA_OP2__N
};

/* reduction: redop x mapop (src/mapreduce.jl:17-39, :97-131) */
enum da_redop  { DA_RED_ADD = 0, DA_RED_MUL, DA_RED_MIN, DA_RED_MAX };
enum da_redf   { DA_REDF_IDENTITY = 0, DA_REDF_ABS, DA_REDF_ABS2,
                 DA_REDF_ISNAN, DA_REDF_ISFINITE, DA_REDF_NONZERO };
/* predicate mapops feed the all/any/count specials of
 * src/mapreduce.jl:97-131: count = (pred, ADD); any = (pred, MAX) == 1;
 * all = (pred, MIN) == 1 */

/* rand kinds (drand/drandn, src/darray.jl:502-532) */
enum da_randkind { DA_RAND_UNIFORM = 0, DA_RAND_NORMAL = 1 };

/* ---- lifecycle -------------------------------------------------------- */
/* HIP context + (nranks>1) RCCL communicator bootstrap over a shared-
 * filesystem uniqueId rendezvous.  Replaces the Distributed-stdlib worker
 * pool connection (1 worker <-> 1 GPU rank). */
int da_init(int device, int rank, int nranks, const char* rccl_uid_path);
int da_shutdown(void);
int da_rank(void);
int da_nranks(void);

/* ---- chunk memory (localpart storage; src/darray.jl:76-118 init /
 *      src/core.jl:67 release_localpart) ------------------------------- */
/* Pooled: da_free caches the chunk for exact-size reuse (hipMalloc of
 * multi-GiB staging buffers costs ~100 ms); da_pool_trim releases the
 * cache to the driver.  bytes_in_use counts only live user chunks. */
int da_alloc(uint64_t nbytes, int dtype, void** chunk);
int da_free(void* chunk);
int da_pool_trim(void);
uint64_t da_pool_bytes(void);
int da_h2d(void* chunk, const void* host, uint64_t nbytes);  /* distribute, darray.jl:544-555 */
int da_d2h(const void* chunk, void* host, uint64_t nbytes);  /* collect / makelocal */
int da_d2d(void* dst, const void* src, uint64_t nbytes);
/* strided 2-D device copy (column-major pack/unpack of sub-panels; used by
 * the matmul panel exchange replacing the B[...] DArray slice gather of
 * src/linalg.jl:215) */
int da_copy2d(void* dst, uint64_t dpitch, const void* src, uint64_t spitch,
              uint64_t width, uint64_t height);

/* ---- constructors on device ------------------------------------------ */
int da_fill(void* chunk, double v, uint64_t n, int dtype);   /* dzeros/dones/dfill/fill!, darray.jl:468-494,822-827 */
/* Philox4x32-10 fill; element mapping documented in oracle/philox.py.
 * seed = 1234 + rank per BASELINE.md; offset = counter origin. */
int da_rand(void* chunk, uint64_t n, int dtype, uint64_t seed, int kind,
            uint64_t offset);                                /* drand/drandn, darray.jl:502-532 */

/* ---- elementwise hot path -------------------------------------------- */
int da_map(int opcode, void* dst, const void* src, uint64_t n, int dtype);   /* map!, mapreduce.jl:5-12 */
int da_map2(int opcode, void* dst, const void* a, const void* b,
            uint64_t n, int dtype);                          /* elementwise +,-,.. mapreduce.jl:180-189 */
int da_bcast_fma(void* d, const void* a, const void* b, double c,
                 uint64_t n, int dtype);                     /* D .= A .* B .+ c, broadcast.jl:65-85 */

/* Fused broadcast composition — materializes an arbitrary Broadcasted
 * tree in one pass (copyto!(localpart, bclocal(bc)), broadcast.jl:65-98;
 * nested broadcast pinned at test/darray.jl:880-912).
 *
 * prog: postfix int32 program, kind = ins>>8, idx = ins&0xff:
 *   kind 0 = unary da_mapop on stack top; kind 1 = push argument idx;
 *   kind 2 = push constant idx; kind 3 = binary da_map2op (pops rhs
 *   then lhs).  Stack depth <= DA_EXPR_MAXSTACK (validated).
 * dst_dims/nd: destination LOCAL chunk shape (column-major), nd <= 4.
 * src_strides: nsrcs x nd element strides (row-major per arg); a 0
 *   stride expands a Julia-broadcast singleton dim.  NULL = every arg
 *   dense over the destination chunk (fast flat path).
 * Numerics are bit-identical to da_map/da_map2 chains of the same ops
 * (shared scalar functor tables, -ffp-contract=off). */
#define DA_EXPR_MAXLEN   40
#define DA_EXPR_MAXARGS  6
#define DA_EXPR_MAXCONSTS 6
#define DA_EXPR_MAXND    4
#define DA_EXPR_MAXSTACK 8
int da_expr(const int32_t* prog, int prog_len, void* dst,
            const uint64_t* dst_dims, int nd,
            void* const* srcs, const uint64_t* src_strides, int nsrcs,
            const double* consts, int nconsts, uint64_t n, int dtype);
/* da_expr compiles each distinct program to a dedicated kernel via
 * hipRTC (cached per process; DA_EXPR_JIT=0 forces the interpreter).
 * state: 1 ready, 2 active, -1 hipRTC failed (interpreter fallback). */
int da_expr_jit_state(void);
const char* da_expr_jit_errstr(void);
int da_map2_scalar(int opcode, void* dst, const void* src, double c,
                   int reverse, uint64_t n, int dtype);      /* D .+ 1 etc (scalar broadcast arg, broadcast.jl:124-133) */
int da_axpby(void* y, const void* x, double alpha, double beta,
             uint64_t n, int dtype);                         /* y = alpha*x + beta*y: axpy! linalg.jl:24-34 */
int da_add(void* dest, const void* src, double scale,
           uint64_t n, int dtype);                           /* add!, linalg.jl:62-76 */
int da_scale(void* a, double s, uint64_t n, int dtype);      /* rmul!, linalg.jl:54-59 */
/* dtype conversion (DArray{T2}(D) family); float->i64 rounds half-even */
int da_cast(void* dst, int dst_dtype, const void* src, int src_dtype,
            uint64_t n);

/* ---- reductions -------------------------------------------------------*/
/* Local (per-chunk) mapreduce stage: LDS + wavefront-shuffle tree.
 * Result written to *out as the dtype's own scalar type (f64/f32/i64) —
 * the per-worker mapreduce(f, op, localpart(d)) of mapreduce.jl:31.
 * n == 0 returns the fold identity. */
int da_reduce(int mapop, int redop, const void* src, uint64_t n, int dtype,
              void* out);
/* Per-chunk dims-reduction (src/mapreduce.jl:42-94, mapreducedim_within):
 * chunk viewed column-major as (inner, axis, outer); dst gets
 * inner*outer elements reduced over axis (identity when axis==0). */
int da_reduce_dims(int mapop, int redop, const void* src, uint64_t inner,
                   uint64_t axis, uint64_t outer, int dtype, void* dst);
/* Cross-rank fold of scalar partials (the caller-side reduce(op, results)
 * of mapreduce.jl:34, re-expressed as an RCCL allreduce over xGMI).
 * inout is HOST memory of `count` dtype elements; nranks==1 is a no-op. */
int da_allreduce(void* inout, int count, int dtype, int redop);

/* ---- dense linear algebra -------------------------------------------- */
/* Local C = alpha*A*B + beta*C, COLUMN-major f64 (Julia layout), MFMA-
 * tiled for gfx950 — the localpart(A)*B_jk of src/linalg.jl:224.
 * lda/ldb/ldc are element leading dimensions. */
int da_gemm_f64(void* C, const void* A, const void* B,
                int64_t m, int64_t n, int64_t k,
                int64_t lda, int64_t ldb, int64_t ldc,
                double alpha, double beta);
/* i64 local GEMM (exact, wrap mod 2^64; Julia Int matmul parity) */
int da_gemm_i64(void* C, const void* A, const void* B,
                int64_t m, int64_t n, int64_t k,
                int64_t lda, int64_t ldb, int64_t ldc,
                int64_t alpha, int64_t beta);
/* f32 local GEMM on the exact f32-input MFMA (v_mfma_f32_16x16x4_f32;
 * bitwise an fmaf chain — cdna_hip_programming.md §3) */
int da_gemm_f32(void* C, const void* A, const void* B,
                int64_t m, int64_t n, int64_t k,
                int64_t lda, int64_t ldb, int64_t ldc,
                double alpha, double beta);

/* ---- transpose / Diagonal scaling (linalg.jl:1-17, :169-187) --------- */
int da_transpose(void* dst, const void* src, uint64_t m, uint64_t n,
                 int dtype);                 /* dst = src^T, LDS-tiled */
int da_diag_scale(void* a, uint64_t m, uint64_t n, const void* diag,
                  int side, int dtype);      /* side 0: rows (lmul!), 1: cols (rmul!) */

/* ---- distributed samplesort building blocks (src/sort.jl:103-170) --- */
int da_sort(void* chunk, uint64_t n, int dtype);       /* per-chunk radix sort */
int da_sort_out(const void* src, void* dst, uint64_t n, int dtype);
int da_lower_bound(const void* sorted, uint64_t n, int dtype,
                   const void* splitters, int k, uint64_t* out);

/* ---- point-to-point (panel / halo exchange; replaces the remotecall
 *      shipping of B panels and partial products, linalg.jl:211-251,
 *      and makelocal's remote fetch, darray.jl:351-368) ----------------- */
int da_group_start(void);
int da_group_end(void);
int da_send(const void* buf, uint64_t nbytes, int peer);
int da_recv(void* buf, uint64_t nbytes, int peer);
int da_sendrecv(const void* sbuf, int peer_s, void* rbuf, int peer_r,
                uint64_t nbytes);
/* comm/compute overlap: route p2p ops onto a second stream and fence the
 * two streams with events (the matmul partial exchange rides xGMI while
 * the next local GEMM runs) */
int da_p2p_stream(int use_comm);
int da_comm_after_compute(void);
int da_main_after_comm(void);
int da_comm_sync(void);
int da_bcast(void* buf, uint64_t nbytes, int root);
int da_barrier(void);      /* device barrier over the communicator */

/* ---- stream & timing --------------------------------------------------*/
int da_synchronize(void);
int da_event_create(void** ev);
int da_event_record(void* ev);
int da_event_elapsed(void* ev_start, void* ev_stop, float* ms);
int da_event_destroy(void* ev);

/* ---- introspection / errors ------------------------------------------ */
const char* da_errstr(int code);
int da_device_props(char* name, int name_len, uint64_t* hbm_bytes);
uint64_t da_bytes_in_use(void);   /* leak check hook (test/runtests.jl:28-37 analog) */

#ifdef __cplusplus
}
#endif
#endif /* DARRAY_HIP_H */
