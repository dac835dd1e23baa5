// kernels_elementwise.hip — fill / philox-rand / map / map2 / fused
// broadcast / axpby / add / scale for gfx950 (CDNA4).
//
// These replace the reference's per-worker Base loops on localparts:
//   map!            /root/reference/src/mapreduce.jl:5-12
//   elementwise ops /root/reference/src/mapreduce.jl:180-189
//   broadcast       /root/reference/src/broadcast.jl:65-85
//   fill!/rand!     /root/reference/src/darray.jl:468-532,822-834
//   add!/axpy!/rmul!/root/reference/src/linalg.jl:24-76
//
// All HBM-bound: grid-stride loops, 16 B/lane vectorized access
// (double2 / float4 / longlong2), no LDS needed.  This file is compiled
// with -ffp-contract=off so that a*b+c keeps Julia-Base/numpy bit
// semantics (separate mul, add); transcendentals use OCML and are
// parity-tested at 2-3 ulp tolerance.
#include "common.hpp"
#include "philox_device.hpp"
#include "mapops.hpp"
#include <math.h>
#include <stdlib.h>

namespace da {

constexpr int TPB = 256;
constexpr int MAXBLOCKS = 8192;   // 1024 workgroups per XCD — fills the chip

static inline bool use_nt() {
    static int nt = -1;
    if (nt < 0) {
        const char* e = getenv("DA_NT");
        nt = e ? atoi(e) : 0;
    }
    return nt == 1;
}

static inline int nblocks(uint64_t work) {
    uint64_t b = (work + TPB - 1) / TPB;
    if (b > (uint64_t)MAXBLOCKS) b = MAXBLOCKS;
    if (b == 0) b = 1;
    return (int)b;
}

// ------------------------------------------------------------------- fill
// NT = nontemporal stores on the write-only stream (gfx950 `nt` flag) —
// A/B-gated via DA_NT; profiles/ record the measured winner.
template <typename T, bool NT>
__global__ void fill_kernel(T* __restrict__ p, T v, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    // 2 elements per thread via vector store where possible
    uint64_t nv = n / 2;
    using V = T __attribute__((ext_vector_type(2)));
    V vv = {v, v};
    V* pv = reinterpret_cast<V*>(p);
    for (uint64_t j = i; j < nv; j += stride) {
        if (NT) __builtin_nontemporal_store(vv, pv + j);
        else pv[j] = vv;
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride) p[j] = v;
}

int launch_fill(void* chunk, double v, uint64_t n, int dtype, hipStream_t s) {
    if (n == 0) return 0;
    int g = nblocks(n / 2 + 1);
    if (use_nt()) {
        switch (dtype) {
        case DA_F64:
            hipLaunchKernelGGL((fill_kernel<double, true>), dim3(g),
                               dim3(TPB), 0, s, (double*)chunk, v, n); break;
        case DA_F32:
            hipLaunchKernelGGL((fill_kernel<float, true>), dim3(g),
                               dim3(TPB), 0, s, (float*)chunk, (float)v, n);
            break;
        case DA_I64:
            hipLaunchKernelGGL((fill_kernel<int64_t, true>), dim3(g),
                               dim3(TPB), 0, s, (int64_t*)chunk,
                               (int64_t)v, n); break;
        default: return set_err(-3, "da_fill: bad dtype %d", dtype);
        }
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }
    switch (dtype) {
    case DA_F64:
        hipLaunchKernelGGL((fill_kernel<double, false>), dim3(g), dim3(TPB),
                           0, s, (double*)chunk, v, n); break;
    case DA_F32:
        hipLaunchKernelGGL((fill_kernel<float, false>), dim3(g), dim3(TPB),
                           0, s, (float*)chunk, (float)v, n); break;
    case DA_I64:
        hipLaunchKernelGGL((fill_kernel<int64_t, false>), dim3(g), dim3(TPB),
                           0, s, (int64_t*)chunk, (int64_t)v, n); break;
    default: return set_err(-3, "da_fill: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// ------------------------------------------------------------------- rand
// Element mapping documented in oracle/philox.py (bit-identical contract).
__global__ void rand_f64_uniform(double* __restrict__ p, uint64_t n,
                                 uint64_t seed, uint64_t offset) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t first_blk = offset >> 1, last_blk = (offset + n - 1) >> 1;
    for (uint64_t b = first_blk + t; b <= last_blk; b += stride) {
        u32x4 o = philox4x32_10(b, seed);
        double v0 = u01_f64(o.v[0], o.v[1]);
        double v1 = u01_f64(o.v[2], o.v[3]);
        uint64_t e0 = 2 * b, e1 = 2 * b + 1;
        if (e0 >= offset && e0 < offset + n) p[e0 - offset] = v0;
        if (e1 >= offset && e1 < offset + n) p[e1 - offset] = v1;
    }
}

__global__ void rand_f32_uniform(float* __restrict__ p, uint64_t n,
                                 uint64_t seed, uint64_t offset) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t first_blk = offset >> 2, last_blk = (offset + n - 1) >> 2;
    for (uint64_t b = first_blk + t; b <= last_blk; b += stride) {
        u32x4 o = philox4x32_10(b, seed);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            uint64_t e = 4 * b + j;
            if (e >= offset && e < offset + n) p[e - offset] = u01_f32(o.v[j]);
        }
    }
}

__global__ void rand_i64(int64_t* __restrict__ p, uint64_t n,
                         uint64_t seed, uint64_t offset) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t first_blk = offset >> 1, last_blk = (offset + n - 1) >> 1;
    for (uint64_t b = first_blk + t; b <= last_blk; b += stride) {
        u32x4 o = philox4x32_10(b, seed);
        int64_t v0 = (int64_t)(((uint64_t)o.v[1] << 32) | o.v[0]);
        int64_t v1 = (int64_t)(((uint64_t)o.v[3] << 32) | o.v[2]);
        uint64_t e0 = 2 * b, e1 = 2 * b + 1;
        if (e0 >= offset && e0 < offset + n) p[e0 - offset] = v0;
        if (e1 >= offset && e1 < offset + n) p[e1 - offset] = v1;
    }
}

__global__ void rand_f64_normal(double* __restrict__ p, uint64_t n,
                                uint64_t seed, uint64_t offset) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t first_blk = offset >> 1, last_blk = (offset + n - 1) >> 1;
    for (uint64_t b = first_blk + t; b <= last_blk; b += stride) {
        u32x4 o = philox4x32_10(b, seed);
        double u1 = u01_f64(o.v[0], o.v[1]);
        double u2 = u01_f64(o.v[2], o.v[3]);
        double r = sqrt(-2.0 * log1p(-u1));
        double th = 2.0 * M_PI * u2;
        uint64_t e0 = 2 * b, e1 = 2 * b + 1;
        if (e0 >= offset && e0 < offset + n) p[e0 - offset] = r * cos(th);
        if (e1 >= offset && e1 < offset + n) p[e1 - offset] = r * sin(th);
    }
}

__global__ void rand_f32_normal(float* __restrict__ p, uint64_t n,
                                uint64_t seed, uint64_t offset) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t first_blk = offset >> 1, last_blk = (offset + n - 1) >> 1;
    for (uint64_t b = first_blk + t; b <= last_blk; b += stride) {
        u32x4 o = philox4x32_10(b, seed);
        float u1 = u01_f32(o.v[0]);
        float u2 = u01_f32(o.v[1]);
        float r = sqrtf(-2.0f * log1pf(-u1));
        float th = 2.0f * (float)M_PI * u2;
        uint64_t e0 = 2 * b, e1 = 2 * b + 1;
        if (e0 >= offset && e0 < offset + n) p[e0 - offset] = r * cosf(th);
        if (e1 >= offset && e1 < offset + n) p[e1 - offset] = r * sinf(th);
    }
}

int launch_rand(void* chunk, uint64_t n, int dtype, uint64_t seed, int kind,
                uint64_t offset, hipStream_t s) {
    if (n == 0) return 0;
    int g = nblocks(n / 2 + 1);
    if (kind == DA_RAND_UNIFORM) {
        switch (dtype) {
        case DA_F64: hipLaunchKernelGGL(rand_f64_uniform, dim3(g), dim3(TPB),
                        0, s, (double*)chunk, n, seed, offset); break;
        case DA_F32: hipLaunchKernelGGL(rand_f32_uniform, dim3(nblocks(n/4+1)),
                        dim3(TPB), 0, s, (float*)chunk, n, seed, offset); break;
        case DA_I64: hipLaunchKernelGGL(rand_i64, dim3(g), dim3(TPB), 0, s,
                        (int64_t*)chunk, n, seed, offset); break;
        default: return set_err(-3, "da_rand: bad dtype %d", dtype);
        }
    } else if (kind == DA_RAND_NORMAL) {
        switch (dtype) {
        case DA_F64: hipLaunchKernelGGL(rand_f64_normal, dim3(g), dim3(TPB),
                        0, s, (double*)chunk, n, seed, offset); break;
        case DA_F32: hipLaunchKernelGGL(rand_f32_normal, dim3(g), dim3(TPB),
                        0, s, (float*)chunk, n, seed, offset); break;
        default: return set_err(-3, "da_rand: normal needs float dtype");
        }
    } else {
        return set_err(-3, "da_rand: bad kind %d", kind);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

template <typename T>
__global__ void map_kernel(int op, T* __restrict__ dst,
                           const T* __restrict__ src, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(2)));
    uint64_t nv = n / 2;
    const V* sv = reinterpret_cast<const V*>(src);
    V* dv = reinterpret_cast<V*>(dst);
    for (uint64_t j = i; j < nv; j += stride) {
        V v = sv[j];
        V r;
        r.x = apply_map<T>(op, v.x);
        r.y = apply_map<T>(op, v.y);
        dv[j] = r;
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride)
        dst[j] = apply_map<T>(op, src[j]);
}

__global__ void map_kernel_i64(int op, int64_t* __restrict__ dst,
                               const int64_t* __restrict__ src, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = apply_map_i64(op, src[j]);
}

// Hot ops get compile-time instantiations: the 63-case runtime switch
// inflates register pressure (map!(sin) measured 1.08 -> 1.99 ms when
// the extended op set landed); with OP a template constant the switch
// folds to one case.
template <typename T, int OP>
__global__ void map_fixed_kernel(T* __restrict__ dst,
                                 const T* __restrict__ src, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(2)));
    uint64_t nv = n / 2;
    const V* sv = reinterpret_cast<const V*>(src);
    V* dv = reinterpret_cast<V*>(dst);
    for (uint64_t j = i; j < nv; j += stride) {
        V v = sv[j];
        V r;
        r.x = apply_map<T>(OP, v.x);
        r.y = apply_map<T>(OP, v.y);
        dv[j] = r;
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride)
        dst[j] = apply_map<T>(OP, src[j]);
}

// 32 B/lane variant (double4): more memory-level parallelism per
// wave iteration — A/B-gated via DA_MAP_W=4 against the 16 B default.
template <typename T, int OP>
__global__ void map_fixed_kernel_w4(T* __restrict__ dst,
                                    const T* __restrict__ src, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(4)));
    uint64_t nv = n / 4;
    const V* sv = reinterpret_cast<const V*>(src);
    V* dv = reinterpret_cast<V*>(dst);
    for (uint64_t j = i; j < nv; j += stride) {
        V v = sv[j];
        V r;
        r.x = apply_map<T>(OP, v.x);
        r.y = apply_map<T>(OP, v.y);
        r.z = apply_map<T>(OP, v.z);
        r.w = apply_map<T>(OP, v.w);
        dv[j] = r;
    }
    for (uint64_t j = 4 * nv + i; j < n; j += stride)
        dst[j] = apply_map<T>(OP, src[j]);
}

static inline int map_width() {
    static int w = -1;
    if (w < 0) {
        const char* e = getenv("DA_MAP_W");
        w = e ? atoi(e) : 4;   // measured: W4 sin 0.878 ms vs W2 0.946
                               // (and vs same-box hipMemcpy 0.881)
    }
    return w;
}

#define DA_HOT_OPS(X)                                                   \
    X(DA_OP_IDENTITY) X(DA_OP_NEG) X(DA_OP_ABS) X(DA_OP_ABS2)           \
    X(DA_OP_INV) X(DA_OP_SQRT) X(DA_OP_EXP) X(DA_OP_LOG)                \
    X(DA_OP_SIN) X(DA_OP_COS) X(DA_OP_TAN) X(DA_OP_TANH)                \
    X(DA_OP_FLOOR) X(DA_OP_SIGN) X(DA_OP_LOG1P) X(DA_OP_EXPM1)

template <typename T>
static bool launch_map_hot(int opcode, T* dst, const T* src, uint64_t n,
                           int g, hipStream_t s) {
    switch (opcode) {
#define DA_CASE(OPC)                                                    \
    case OPC:                                                           \
        if (map_width() == 4)                                           \
            hipLaunchKernelGGL((map_fixed_kernel_w4<T, OPC>), dim3(g),  \
                               dim3(TPB), 0, s, dst, src, n);           \
        else                                                            \
            hipLaunchKernelGGL((map_fixed_kernel<T, OPC>), dim3(g),     \
                               dim3(TPB), 0, s, dst, src, n);           \
        return true;
    DA_HOT_OPS(DA_CASE)
#undef DA_CASE
    }
    return false;
}

int launch_map(int opcode, void* dst, const void* src, uint64_t n, int dtype,
               hipStream_t s) {
    if (n == 0) return 0;
    if (opcode < 0 || opcode >= DA_OP__N)
        return set_err(-3, "da_map: bad opcode %d", opcode);
    int g = nblocks(n / 2 + 1);
    switch (dtype) {
    case DA_F64:
        if (launch_map_hot<double>(opcode, (double*)dst,
                                   (const double*)src, n, g, s))
            break;
        hipLaunchKernelGGL(map_kernel<double>, dim3(g), dim3(TPB),
                    0, s, opcode, (double*)dst, (const double*)src, n); break;
    case DA_F32:
        if (launch_map_hot<float>(opcode, (float*)dst,
                                  (const float*)src, n, g, s))
            break;
        hipLaunchKernelGGL(map_kernel<float>, dim3(g), dim3(TPB),
                    0, s, opcode, (float*)dst, (const float*)src, n); break;
    case DA_I64:
        if (!(opcode == DA_OP_IDENTITY || opcode == DA_OP_NEG ||
              opcode == DA_OP_ABS || opcode == DA_OP_ABS2 ||
              opcode == DA_OP_SIGN))
            return set_err(-3, "da_map: opcode %d invalid for i64", opcode);
        hipLaunchKernelGGL(map_kernel_i64, dim3(g), dim3(TPB), 0, s,
                           opcode, (int64_t*)dst, (const int64_t*)src, n);
        break;
    default: return set_err(-3, "da_map: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// ------------------------------------------------------------------- map2
template <typename T>
__global__ void map2_kernel(int op, T* __restrict__ dst,
                            const T* __restrict__ a,
                            const T* __restrict__ b, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(2)));
    uint64_t nv = n / 2;
    const V* av = reinterpret_cast<const V*>(a);
    const V* bv = reinterpret_cast<const V*>(b);
    V* dv = reinterpret_cast<V*>(dst);
    for (uint64_t j = i; j < nv; j += stride) {
        V x = av[j], y = bv[j], r;
        r.x = apply_map2<T>(op, x.x, y.x);
        r.y = apply_map2<T>(op, x.y, y.y);
        dv[j] = r;
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride)
        dst[j] = apply_map2<T>(op, a[j], b[j]);
}

__global__ void map2_kernel_i64(int op, int64_t* __restrict__ dst,
                                const int64_t* __restrict__ a,
                                const int64_t* __restrict__ b, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = apply_map2_i64(op, a[j], b[j]);
}

int launch_map2(int opcode, void* dst, const void* a, const void* b,
                uint64_t n, int dtype, hipStream_t s) {
    if (n == 0) return 0;
    if (opcode < 0 || opcode >= DA_OP2__N)
        return set_err(-3, "da_map2: bad opcode %d", opcode);
    int g = nblocks(n / 2 + 1);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(map2_kernel<double>, dim3(g), dim3(TPB),
                    0, s, opcode, (double*)dst, (const double*)a,
                    (const double*)b, n); break;
    case DA_F32: hipLaunchKernelGGL(map2_kernel<float>, dim3(g), dim3(TPB),
                    0, s, opcode, (float*)dst, (const float*)a,
                    (const float*)b, n); break;
    case DA_I64: hipLaunchKernelGGL(map2_kernel_i64, dim3(g), dim3(TPB), 0, s,
                    opcode, (int64_t*)dst, (const int64_t*)a,
                    (const int64_t*)b, n); break;
    default: return set_err(-3, "da_map2: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// scalar-broadcast forms: D .= f.(src, c) / f.(c, src) (e.g. D .+ 1,
// the cfg-1 plumbing op; Base broadcast with a scalar argument —
// broadcast.jl:124-133 treats singletons as local, no distribution)
template <typename T>
__global__ void map2s_kernel(int op, T* __restrict__ dst,
                             const T* __restrict__ a, T c, int rev,
                             uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = rev ? apply_map2<T>(op, c, a[j])
                     : apply_map2<T>(op, a[j], c);
}

__global__ void map2s_kernel_i64(int op, int64_t* __restrict__ dst,
                                 const int64_t* __restrict__ a, int64_t c,
                                 int rev, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = rev ? apply_map2_i64(op, c, a[j])
                     : apply_map2_i64(op, a[j], c);
}

int launch_map2_scalar(int opcode, void* dst, const void* src, double c,
                       int rev, uint64_t n, int dtype, hipStream_t s) {
    if (n == 0) return 0;
    if (opcode < 0 || opcode >= DA_OP2__N)
        return set_err(-3, "da_map2_scalar: bad opcode %d", opcode);
    int g = nblocks(n);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(map2s_kernel<double>, dim3(g),
                    dim3(TPB), 0, s, opcode, (double*)dst,
                    (const double*)src, c, rev, n); break;
    case DA_F32: hipLaunchKernelGGL(map2s_kernel<float>, dim3(g),
                    dim3(TPB), 0, s, opcode, (float*)dst,
                    (const float*)src, (float)c, rev, n); break;
    case DA_I64: hipLaunchKernelGGL(map2s_kernel_i64, dim3(g), dim3(TPB),
                    0, s, opcode, (int64_t*)dst, (const int64_t*)src,
                    (int64_t)c, rev, n); break;
    default: return set_err(-3, "da_map2_scalar: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}


// ---------------------------------------------------- transpose / diag
// Local 2-D transpose (dst = src^T), LDS-tiled so both sides stay
// coalesced — the per-chunk piece of copy(::Transpose{<:DArray})
// (linalg.jl:10-17).
template <typename T>
__global__ void transpose_kernel(T* __restrict__ dst,
                                 const T* __restrict__ src,
                                 uint64_t m, uint64_t n) {
    __shared__ T tile[32][33];
    uint64_t bi = (uint64_t)blockIdx.x * 32;
    uint64_t bj = (uint64_t)blockIdx.y * 32;
    int tx = threadIdx.x, ty = threadIdx.y;   // 32 x 8
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        uint64_t i = bi + tx, j = bj + ty + r * 8;
        if (i < m && j < n) tile[ty + r * 8][tx] = src[i + j * m];
    }
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        uint64_t j = bj + tx, i = bi + ty + r * 8;   // dst is n x m
        if (j < n && i < m) dst[j + i * n] = tile[tx][ty + r * 8];
    }
}

int launch_transpose(void* dst, const void* src, uint64_t m, uint64_t n,
                     int dtype, hipStream_t s) {
    if (m == 0 || n == 0) return 0;
    dim3 t(32, 8), g((m + 31) / 32, (n + 31) / 32);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(transpose_kernel<double>, g, t, 0, s,
                    (double*)dst, (const double*)src, m, n); break;
    case DA_F32: hipLaunchKernelGGL(transpose_kernel<float>, g, t, 0, s,
                    (float*)dst, (const float*)src, m, n); break;
    case DA_I64: hipLaunchKernelGGL(transpose_kernel<int64_t>, g, t, 0, s,
                    (int64_t*)dst, (const int64_t*)src, m, n); break;
    default: return set_err(-3, "da_transpose: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// Diagonal scaling: side=0 rows (lmul!(Diagonal(d), A), linalg.jl:169-177),
// side=1 cols (rmul!(A, Diagonal(d)), linalg.jl:179-187).
template <typename T>
__global__ void diag_scale_kernel(T* __restrict__ a, uint64_t m, uint64_t n,
                                  const T* __restrict__ diag, int side) {
    uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t total = m * n;
    for (uint64_t t = e; t < total; t += stride) {
        uint64_t i = t % m, j = t / m;
        a[t] = a[t] * diag[side == 0 ? i : j];
    }
}

int launch_diag_scale(void* a, uint64_t m, uint64_t n, const void* diag,
                      int side, int dtype, hipStream_t s) {
    if (m == 0 || n == 0) return 0;
    int g = nblocks(m * n);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(diag_scale_kernel<double>, dim3(g),
                    dim3(TPB), 0, s, (double*)a, m, n,
                    (const double*)diag, side); break;
    case DA_F32: hipLaunchKernelGGL(diag_scale_kernel<float>, dim3(g),
                    dim3(TPB), 0, s, (float*)a, m, n,
                    (const float*)diag, side); break;
    default: return set_err(-3, "da_diag_scale: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// ------------------------------------------- fused broadcast & BLAS-1 like
template <typename T>
__global__ void bcast_fma_kernel_w4(T* __restrict__ d,
                                    const T* __restrict__ a,
                                    const T* __restrict__ b, T c,
                                    uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(4)));
    uint64_t nv = n / 4;
    const V* av = reinterpret_cast<const V*>(a);
    const V* bv = reinterpret_cast<const V*>(b);
    V* dv = reinterpret_cast<V*>(d);
    for (uint64_t j = i; j < nv; j += stride) {
        V x = av[j], y = bv[j], r;
        r.x = x.x * y.x + c;   // -ffp-contract=off (Julia Base)
        r.y = x.y * y.y + c;
        r.z = x.z * y.z + c;
        r.w = x.w * y.w + c;
        dv[j] = r;
    }
    for (uint64_t j = 4 * nv + i; j < n; j += stride) d[j] = a[j] * b[j] + c;
}

static inline int bc_width() {
    static int w = -1;
    if (w < 0) {
        const char* e = getenv("DA_BC_W");
        w = e ? atoi(e) : 4;   // measured 1.21 vs 1.37 ms on 2^28
    }
    return w;
}

template <typename T, bool NT>
__global__ void bcast_fma_kernel(T* __restrict__ d, const T* __restrict__ a,
                                 const T* __restrict__ b, T c, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(2)));
    uint64_t nv = n / 2;
    const V* av = reinterpret_cast<const V*>(a);
    const V* bv = reinterpret_cast<const V*>(b);
    V* dv = reinterpret_cast<V*>(d);
    for (uint64_t j = i; j < nv; j += stride) {
        V x = av[j], y = bv[j], r;
        r.x = x.x * y.x + c;   // -ffp-contract=off: mul then add (Julia Base)
        r.y = x.y * y.y + c;
        if (NT) __builtin_nontemporal_store(r, dv + j);
        else dv[j] = r;
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride) d[j] = a[j] * b[j] + c;
}

int launch_bcast_fma(void* d, const void* a, const void* b, double c,
                     uint64_t n, int dtype, hipStream_t s) {
    if (n == 0) return 0;
    int g = nblocks(n / 2 + 1);
    if (bc_width() == 4 && !use_nt()) {
        switch (dtype) {
        case DA_F64: hipLaunchKernelGGL((bcast_fma_kernel_w4<double>),
                        dim3(g), dim3(TPB), 0, s, (double*)d,
                        (const double*)a, (const double*)b, c, n); break;
        case DA_F32: hipLaunchKernelGGL((bcast_fma_kernel_w4<float>),
                        dim3(g), dim3(TPB), 0, s, (float*)d,
                        (const float*)a, (const float*)b, (float)c, n);
            break;
        default: return set_err(-3, "da_bcast_fma: bad dtype %d", dtype);
        }
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }
    if (use_nt()) {
        switch (dtype) {
        case DA_F64: hipLaunchKernelGGL((bcast_fma_kernel<double, true>),
                        dim3(g), dim3(TPB), 0, s, (double*)d,
                        (const double*)a, (const double*)b, c, n); break;
        case DA_F32: hipLaunchKernelGGL((bcast_fma_kernel<float, true>),
                        dim3(g), dim3(TPB), 0, s, (float*)d,
                        (const float*)a, (const float*)b, (float)c, n);
            break;
        default: return set_err(-3, "da_bcast_fma: bad dtype %d", dtype);
        }
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL((bcast_fma_kernel<double, false>),
                    dim3(g), dim3(TPB), 0, s, (double*)d, (const double*)a,
                    (const double*)b, c, n); break;
    case DA_F32: hipLaunchKernelGGL((bcast_fma_kernel<float, false>),
                    dim3(g), dim3(TPB), 0, s, (float*)d, (const float*)a,
                    (const float*)b, (float)c, n); break;
    default: return set_err(-3, "da_bcast_fma: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

template <typename T>
__global__ void axpby_kernel(T* __restrict__ y, const T* __restrict__ x,
                             T alpha, T beta, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(4)));
    uint64_t nv = n / 4;
    const V* xv = reinterpret_cast<const V*>(x);
    V* yv = reinterpret_cast<V*>(y);
    for (uint64_t j = i; j < nv; j += stride) {
        V a = xv[j], b = yv[j], r;
        r.x = alpha * a.x + beta * b.x;
        r.y = alpha * a.y + beta * b.y;
        r.z = alpha * a.z + beta * b.z;
        r.w = alpha * a.w + beta * b.w;
        yv[j] = r;
    }
    for (uint64_t j = 4 * nv + i; j < n; j += stride)
        y[j] = alpha * x[j] + beta * y[j];
}

int launch_axpby(void* y, const void* x, double alpha, double beta,
                 uint64_t n, int dtype, hipStream_t s) {
    if (n == 0) return 0;
    int g = nblocks(n);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(axpby_kernel<double>, dim3(g), dim3(TPB),
                    0, s, (double*)y, (const double*)x, alpha, beta, n); break;
    case DA_F32: hipLaunchKernelGGL(axpby_kernel<float>, dim3(g), dim3(TPB),
                    0, s, (float*)y, (const float*)x, (float)alpha,
                    (float)beta, n); break;
    default: return set_err(-3, "da_axpby: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// add! (linalg.jl:62-76): scale==1 adds without multiplying, matching the
// reference's fast path numerics exactly.
template <typename T>
__global__ void add_kernel(T* __restrict__ d, const T* __restrict__ s_,
                           T scale, int unit, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(4)));
    uint64_t nv = n / 4;
    const V* sv = reinterpret_cast<const V*>(s_);
    V* dv = reinterpret_cast<V*>(d);
    if (unit) {
        for (uint64_t j = i; j < nv; j += stride) {
            V a = dv[j], b = sv[j], r;
            r.x = a.x + b.x; r.y = a.y + b.y;
            r.z = a.z + b.z; r.w = a.w + b.w;
            dv[j] = r;
        }
        for (uint64_t j = 4 * nv + i; j < n; j += stride)
            d[j] = d[j] + s_[j];
    } else {
        for (uint64_t j = i; j < nv; j += stride) {
            V a = dv[j], b = sv[j], r;
            r.x = a.x + scale * b.x; r.y = a.y + scale * b.y;
            r.z = a.z + scale * b.z; r.w = a.w + scale * b.w;
            dv[j] = r;
        }
        for (uint64_t j = 4 * nv + i; j < n; j += stride)
            d[j] = d[j] + scale * s_[j];
    }
}

int launch_add(void* dest, const void* src, double scale, uint64_t n,
               int dtype, hipStream_t s) {
    if (n == 0) return 0;
    int g = nblocks(n);
    int unit = (scale == 1.0);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(add_kernel<double>, dim3(g), dim3(TPB),
                    0, s, (double*)dest, (const double*)src, scale, unit, n);
                 break;
    case DA_F32: hipLaunchKernelGGL(add_kernel<float>, dim3(g), dim3(TPB),
                    0, s, (float*)dest, (const float*)src, (float)scale,
                    unit, n); break;
    case DA_I64: hipLaunchKernelGGL(add_kernel<int64_t>, dim3(g), dim3(TPB),
                    0, s, (int64_t*)dest, (const int64_t*)src,
                    (int64_t)scale, unit, n); break;
    default: return set_err(-3, "da_add: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

template <typename T>
__global__ void scale_kernel(T* __restrict__ a, T v, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    using V = T __attribute__((ext_vector_type(4)));
    uint64_t nv = n / 4;
    V* av = reinterpret_cast<V*>(a);
    for (uint64_t j = i; j < nv; j += stride) {
        V x = av[j];
        x.x *= v; x.y *= v; x.z *= v; x.w *= v;
        av[j] = x;
    }
    for (uint64_t j = 4 * nv + i; j < n; j += stride) a[j] = a[j] * v;
}

int launch_scale(void* a, double s_, uint64_t n, int dtype, hipStream_t s) {
    if (n == 0) return 0;
    int g = nblocks(n);
    switch (dtype) {
    case DA_F64: hipLaunchKernelGGL(scale_kernel<double>, dim3(g), dim3(TPB),
                    0, s, (double*)a, s_, n); break;
    case DA_F32: hipLaunchKernelGGL(scale_kernel<float>, dim3(g), dim3(TPB),
                    0, s, (float*)a, (float)s_, n); break;
    case DA_I64: hipLaunchKernelGGL(scale_kernel<int64_t>, dim3(g), dim3(TPB),
                    0, s, (int64_t*)a, (int64_t)s_, n); break;
    default: return set_err(-3, "da_scale: bad dtype %d", dtype);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}


// ------------------------------------------------------------------ cast
// dtype conversion (the DArray{T2}(D) / convert family): dst[i] =
// (TD)src[i].  float->i64 rounds half-even (Julia round(Int, x)); the
// strict InexactError convert is a host-side concern.
template <typename TD, typename TS>
__global__ void cast_kernel(TD* __restrict__ dst,
                            const TS* __restrict__ src, uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride) dst[j] = (TD)src[j];
}

template <>
__global__ void cast_kernel<int64_t, double>(int64_t* __restrict__ dst,
                                             const double* __restrict__ src,
                                             uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = (int64_t)rint(src[j]);
}

template <>
__global__ void cast_kernel<int64_t, float>(int64_t* __restrict__ dst,
                                            const float* __restrict__ src,
                                            uint64_t n) {
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = i; j < n; j += stride)
        dst[j] = (int64_t)rintf(src[j]);
}

int launch_cast(void* dst, int dst_dtype, const void* src, int src_dtype,
                uint64_t n, hipStream_t s) {
    if (n == 0) return 0;
    if (dst_dtype == src_dtype)
        return set_err(-3, "da_cast: same dtype (use da_d2d)");
    int g = nblocks(n);
#define DA_CAST(TD, TS) \
    hipLaunchKernelGGL((cast_kernel<TD, TS>), dim3(g), dim3(TPB), 0, s, \
                       (TD*)dst, (const TS*)src, n)
    switch (dst_dtype * 4 + src_dtype) {
    case DA_F64 * 4 + DA_F32: DA_CAST(double, float); break;
    case DA_F64 * 4 + DA_I64: DA_CAST(double, int64_t); break;
    case DA_F32 * 4 + DA_F64: DA_CAST(float, double); break;
    case DA_F32 * 4 + DA_I64: DA_CAST(float, int64_t); break;
    case DA_I64 * 4 + DA_F64: DA_CAST(int64_t, double); break;
    case DA_I64 * 4 + DA_F32: DA_CAST(int64_t, float); break;
    default: return set_err(-3, "da_cast: bad dtypes %d<-%d",
                            dst_dtype, src_dtype);
    }
#undef DA_CAST
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

} // namespace da
