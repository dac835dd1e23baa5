// kernels_expr.hip — fused broadcast-composition interpreter for gfx950.
//
// The reference materializes an ARBITRARY Broadcasted tree in one local
// pass per worker (copyto!(localpart, bclocal(bc)),
// /root/reference/src/broadcast.jl:65-98; nested broadcast pinned at
// test/darray.jl:880-912: `a .- m .* sin.(c)` with a dims-expanded `m`).
// Round 1 had only the fixed opcode table plus one fused ternary; this
// kernel closes the gap: a postfix program over the SAME scalar functor
// tables (mapops.hpp — bit-identical numerics to da_map/da_map2)
// evaluates any unary/binary composition in ONE pass, algorithmic
// traffic only (one read per distinct operand element, one write).
//
// Program encoding (include/darray_hip.h): int32 instructions,
//   kind = ins >> 8, idx = ins & 0xff
//   kind 0: unary da_mapop idx applied to the stack top
//   kind 1: push element of argument idx
//   kind 2: push constant idx
//   kind 3: binary da_map2op idx (pops rhs, then lhs; pushes f(l, r))
// Stack depth <= DA_EXPR_MAXSTACK, validated host-side by simulation.
//
// Two variants:
//   flat    — every argument is dense over the destination's local chunk
//             (index i); 2-wide vectorized like map_kernel;
//   strided — per-argument element strides (0 on Julia-broadcast
//             singleton dims, e.g. the mean row of `a .- mean(a,dims=1)`);
//             the destination multi-index is decoded once per element
//             (u32 div chain) and shared by all arguments.
// Both HBM-bound; compiled -ffp-contract=off (Julia Base numerics).
#include "common.hpp"
#include "mapops.hpp"
#include <math.h>

namespace da {

namespace {
constexpr int TPB = 256;
constexpr int MAXBLOCKS = 8192;   // 1024 workgroups/XCD

static inline int nblocks(uint64_t work) {
    uint64_t b = (work + TPB - 1) / TPB;
    if (b > (uint64_t)MAXBLOCKS) b = MAXBLOCKS;
    if (b == 0) b = 1;
    return (int)b;
}
}  // namespace

// Program shipped BY VALUE in the kernel arguments (<= ~500 B: no
// device staging buffers, no extra H2D latency per launch).
struct ExprProg {
    int32_t ins[DA_EXPR_MAXLEN];
    int len;
    const void* srcs[DA_EXPR_MAXARGS];
    double consts[DA_EXPR_MAXCONSTS];
};

struct ExprStrides {
    uint32_t dims[DA_EXPR_MAXND];                      // dest local shape
    uint32_t str[DA_EXPR_MAXARGS][DA_EXPR_MAXND];      // elem strides; 0=expand
    int nd;
};

template <typename T>
__device__ __forceinline__ T expr_eval(const ExprProg& p, uint64_t i) {
    T stack[DA_EXPR_MAXSTACK];
    int sp = 0;
    for (int pc = 0; pc < p.len; ++pc) {
        int ins = p.ins[pc];
        int kind = ins >> 8, idx = ins & 0xff;
        switch (kind) {
        case 1: stack[sp++] = ((const T*)p.srcs[idx])[i]; break;
        case 2: stack[sp++] = (T)p.consts[idx]; break;
        case 0: stack[sp - 1] = apply_map<T>(idx, stack[sp - 1]); break;
        default: {
            T b = stack[--sp];
            stack[sp - 1] = apply_map2<T>(idx, stack[sp - 1], b);
        }
        }
    }
    return stack[0];
}

__device__ __forceinline__ int64_t expr_eval_i64(const ExprProg& p,
                                                 uint64_t i) {
    int64_t stack[DA_EXPR_MAXSTACK];
    int sp = 0;
    for (int pc = 0; pc < p.len; ++pc) {
        int ins = p.ins[pc];
        int kind = ins >> 8, idx = ins & 0xff;
        switch (kind) {
        case 1: stack[sp++] = ((const int64_t*)p.srcs[idx])[i]; break;
        case 2: stack[sp++] = (int64_t)p.consts[idx]; break;
        case 0: stack[sp - 1] = apply_map_i64(idx, stack[sp - 1]); break;
        default: {
            int64_t b = stack[--sp];
            stack[sp - 1] = apply_map2_i64(idx, stack[sp - 1], b);
        }
        }
    }
    return stack[0];
}

template <typename T>
__global__ void expr_flat_kernel(ExprProg p, T* __restrict__ dst,
                                 uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    // 2 adjacent elements per lane: the compiler merges the two
    // interleaved evaluations' loads into 16-B accesses on dense args
    uint64_t nv = n / 2;
    for (uint64_t jp = i; jp < nv; jp += stride) {
        uint64_t j = 2 * jp;
        T r0 = expr_eval<T>(p, j);
        T r1 = expr_eval<T>(p, j + 1);
        dst[j] = r0;
        dst[j + 1] = r1;
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride)
        dst[j] = expr_eval<T>(p, j);
}

__global__ void expr_flat_kernel_i64(ExprProg p, int64_t* __restrict__ dst,
                                     uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = expr_eval_i64(p, j);
}

// Strided variant: decode the destination's column-major multi-index
// once per element, then each argument reads at sum(idx_d * str[a][d])
// (stride 0 expands a singleton dim, the bclocal localisation of
// broadcast.jl:140-152 after gather).
template <typename T>
__device__ __forceinline__ void expr_strided_body(const ExprProg& p,
                                                  const ExprStrides& s,
                                                  T* dst, uint64_t n) {
    uint64_t i0 = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t gstride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = i0; i < n; i += gstride) {
        uint32_t idx[DA_EXPR_MAXND];
        uint32_t rem = (uint32_t)i;   // local chunks are < 2^32 elements
        for (int d = 0; d < s.nd; ++d) {
            idx[d] = rem % s.dims[d];
            rem /= s.dims[d];
        }
        uint64_t off[DA_EXPR_MAXARGS];
        for (int a = 0; a < DA_EXPR_MAXARGS; ++a) off[a] = 0;
        for (int d = 0; d < s.nd; ++d)
            for (int a = 0; a < DA_EXPR_MAXARGS; ++a)
                off[a] += (uint64_t)idx[d] * s.str[a][d];

        T stack[DA_EXPR_MAXSTACK];
        int sp = 0;
        for (int pc = 0; pc < p.len; ++pc) {
            int ins = p.ins[pc];
            int kind = ins >> 8, opi = ins & 0xff;
            switch (kind) {
            case 1: stack[sp++] = ((const T*)p.srcs[opi])[off[opi]]; break;
            case 2: stack[sp++] = (T)p.consts[opi]; break;
            case 0: stack[sp - 1] = apply_map<T>(opi, stack[sp - 1]); break;
            default: {
                T b = stack[--sp];
                stack[sp - 1] = apply_map2<T>(opi, stack[sp - 1], b);
            }
            }
        }
        dst[i] = stack[0];
    }
}

template <typename T>
__global__ void expr_strided_kernel(ExprProg p, ExprStrides s,
                                    T* __restrict__ dst, uint64_t n) {
    expr_strided_body<T>(p, s, dst, n);
}

int launch_expr(const int32_t* prog, int plen, void* dst,
                const uint64_t* dst_dims, int nd,
                void* const* srcs, const uint64_t* src_strides, int nsrcs,
                const double* consts, int nconsts,
                uint64_t n, int dtype, hipStream_t s) {
    if (plen <= 0 || plen > DA_EXPR_MAXLEN)
        return set_err(-3, "da_expr: bad program length %d", plen);
    if (nsrcs < 0 || nsrcs > DA_EXPR_MAXARGS || nconsts < 0 ||
        nconsts > DA_EXPR_MAXCONSTS)
        return set_err(-3, "da_expr: too many args/consts");
    // host-side validation by stack simulation
    int sp = 0;
    for (int pc = 0; pc < plen; ++pc) {
        int kind = prog[pc] >> 8, idx = prog[pc] & 0xff;
        switch (kind) {
        case 1:
            if (idx >= nsrcs) return set_err(-3, "da_expr: arg %d oob", idx);
            ++sp; break;
        case 2:
            if (idx >= nconsts)
                return set_err(-3, "da_expr: const %d oob", idx);
            ++sp; break;
        case 0:
            if (sp < 1 || idx >= DA_OP__N)
                return set_err(-3, "da_expr: bad unary at %d", pc);
            break;
        case 3:
            if (sp < 2 || idx >= DA_OP2__N)
                return set_err(-3, "da_expr: bad binary at %d", pc);
            --sp; break;
        default:
            return set_err(-3, "da_expr: bad kind at %d", pc);
        }
        if (sp > DA_EXPR_MAXSTACK)
            return set_err(-3, "da_expr: stack overflow at %d", pc);
    }
    if (sp != 1) return set_err(-3, "da_expr: program leaves %d values", sp);
    if (n == 0) return 0;

    // preferred path: hipRTC-compiled kernel for this exact program
    // (register pressure of the expression only; numerics identical —
    // generated code calls the same functor tables).  Falls back to
    // the interpreter below when disabled or compilation fails.
    {
        int rc = launch_expr_jit(prog, plen, dst, dst_dims, nd, srcs,
                                 src_strides, nsrcs, consts, nconsts, n,
                                 dtype, s);
        if (rc <= 0) return rc;
    }

    ExprProg P;
    memset(&P, 0, sizeof(P));
    for (int i = 0; i < plen; ++i) P.ins[i] = prog[i];
    P.len = plen;
    for (int i = 0; i < nsrcs; ++i) P.srcs[i] = srcs[i];
    for (int i = 0; i < nconsts; ++i) P.consts[i] = consts[i];

    if (src_strides == nullptr) {
        int g = nblocks(n / 2 + 1);
        switch (dtype) {
        case DA_F64:
            hipLaunchKernelGGL((expr_flat_kernel<double>), dim3(g),
                               dim3(TPB), 0, s, P, (double*)dst, n);
            break;
        case DA_F32:
            hipLaunchKernelGGL((expr_flat_kernel<float>), dim3(g),
                               dim3(TPB), 0, s, P, (float*)dst, n);
            break;
        case DA_I64:
            hipLaunchKernelGGL(expr_flat_kernel_i64, dim3(nblocks(n)),
                               dim3(TPB), 0, s, P, (int64_t*)dst, n);
            break;
        default: return set_err(-3, "da_expr: bad dtype %d", dtype);
        }
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }

    if (nd < 1 || nd > DA_EXPR_MAXND)
        return set_err(-3, "da_expr: nd %d unsupported (max %d)", nd,
                       DA_EXPR_MAXND);
    if (n >> 32)
        return set_err(-3, "da_expr: strided variant needs n < 2^32");
    ExprStrides S;
    memset(&S, 0, sizeof(S));
    S.nd = nd;
    uint64_t total = 1;
    for (int d = 0; d < nd; ++d) {
        S.dims[d] = (uint32_t)dst_dims[d];
        total *= dst_dims[d];
    }
    if (total != n)
        return set_err(-3, "da_expr: dims/numel mismatch");
    for (int a = 0; a < nsrcs; ++a)
        for (int d = 0; d < nd; ++d)
            S.str[a][d] = (uint32_t)src_strides[(size_t)a * nd + d];
    int g = nblocks(n);
    switch (dtype) {
    case DA_F64:
        hipLaunchKernelGGL((expr_strided_kernel<double>), dim3(g),
                           dim3(TPB), 0, s, P, S, (double*)dst, n);
        break;
    case DA_F32:
        hipLaunchKernelGGL((expr_strided_kernel<float>), dim3(g),
                           dim3(TPB), 0, s, P, S, (float*)dst, n);
        break;
    default:
        return set_err(-3, "da_expr: strided variant is float-only");
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

} // namespace da
