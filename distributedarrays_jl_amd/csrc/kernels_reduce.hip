// kernels_reduce.hip — per-chunk mapreduce stage for gfx950.
//
// Replaces the worker-side hot loop `mapreduce(f, op, localpart(d))` of
// /root/reference/src/mapreduce.jl:29-35 (and the specials :97-131).
// Two-stage tree: grid-stride vectorized loads -> per-thread partial ->
// 64-wide wavefront __shfl_down reduction -> LDS across the block's 4
// waves -> one partial per block; a single-block second kernel folds the
// block partials.  Integer add/mul wrap mod 2^64, so any order is
// bit-exact (test/darray.jl:286-294 exactness contract); float order is
// a tree, within the 1e-6 contract of BASELINE.json (the reference's own
// fold order is likewise unspecified: docs/src/index.md:208-236).
#include "common.hpp"
#include <stdlib.h>

namespace da {

constexpr int RTPB = 256;          // 4 waves per block
constexpr int RMAXB = 8192;        // block partials (fits scratch)

static inline int reduce_grid(uint64_t want) {
    static int cap = -1;
    if (cap < 0) {
        const char* e = getenv("DA_RBLOCKS");
        cap = e ? atoi(e) : 2048;
        if (cap < 1 || cap > RMAXB) cap = 2048;
    }
    if (want < 1) want = 1;
    return (int)(want > (uint64_t)cap ? (uint64_t)cap : want);
}

static inline bool reduce_v4() {
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("DA_RV4");
        v = e ? atoi(e) : 0;
    }
    return v == 1;
}

template <typename T> struct RedIdent {
    static __device__ __host__ T get(int redop) {
        switch (redop) {
        case DA_RED_ADD: return (T)0;
        case DA_RED_MUL: return (T)1;
        case DA_RED_MIN: return (T)INFINITY;
        case DA_RED_MAX: return (T)(-INFINITY);
        }
        return (T)0;
    }
};
template <> struct RedIdent<int64_t> {
    static __device__ __host__ int64_t get(int redop) {
        switch (redop) {
        case DA_RED_ADD: return 0;
        case DA_RED_MUL: return 1;
        case DA_RED_MIN: return INT64_MAX;
        case DA_RED_MAX: return INT64_MIN;
        }
        return 0;
    }
};

template <typename T>
__device__ __forceinline__ T red_comb(int redop, T a, T b) {
    switch (redop) {
    case DA_RED_ADD: return a + b;
    case DA_RED_MUL: return a * b;
    case DA_RED_MIN:  // NaN-propagating (Julia min)
        return a != a ? a : (b != b ? b : (a < b ? a : b));
    case DA_RED_MAX:
        return a != a ? a : (b != b ? b : (a > b ? a : b));
    }
    return a;
}
__device__ __forceinline__ int64_t red_comb(int redop, int64_t a, int64_t b) {
    switch (redop) {
    case DA_RED_ADD: return (int64_t)((uint64_t)a + (uint64_t)b);
    case DA_RED_MUL: return (int64_t)((uint64_t)a * (uint64_t)b);
    case DA_RED_MIN: return a < b ? a : b;
    case DA_RED_MAX: return a > b ? a : b;
    }
    return a;
}

template <typename T>
__device__ __forceinline__ T mapf(int mapop, T x) {
    switch (mapop) {
    case DA_REDF_IDENTITY: return x;
    case DA_REDF_ABS: return fabs(x);   // abs(-0.0) = +0.0
    case DA_REDF_ABS2: return x * x;
    case DA_REDF_ISNAN: return (T)(x != x ? 1 : 0);
    case DA_REDF_ISFINITE: return (T)(isfinite((double)x) ? 1 : 0);
    case DA_REDF_NONZERO: return (T)(x != (T)0 ? 1 : 0);
    }
    return x;
}
__device__ __forceinline__ int64_t mapf(int mapop, int64_t x) {
    switch (mapop) {
    case DA_REDF_IDENTITY: return x;
    case DA_REDF_ABS: return x < 0 ? (int64_t)(0ull - (uint64_t)x) : x;
    case DA_REDF_ABS2: return (int64_t)((uint64_t)x * (uint64_t)x);
    case DA_REDF_ISNAN: return 0;
    case DA_REDF_ISFINITE: return 1;
    case DA_REDF_NONZERO: return x != 0 ? 1 : 0;
    }
    return x;
}

template <typename T>
__device__ __forceinline__ T block_reduce(int redop, T v) {
    __shared__ T lds[RTPB / 64];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v = red_comb(redop, v, (T)__shfl_down(v, off, 64));
    int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
    if (lane == 0) lds[wave] = v;
    __syncthreads();
    if (wave == 0) {
        T w = lane < (RTPB / 64) ? lds[lane]
                                 : RedIdent<T>::get(redop);
#pragma unroll
        for (int off = 2; off > 0; off >>= 1)
            w = red_comb(redop, w, (T)__shfl_down(w, off, 64));
        return w;
    }
    return v;
}

// MOP/ROP >= 0 fold the op switches at compile time for the hot
// combos (same trick as map_fixed/dims-reduce); -1 = runtime args.
template <typename T, int MOP = -1, int ROP = -1>
__global__ void reduce_stage1(int mapop, int redop, const T* __restrict__ src,
                              uint64_t n, T* __restrict__ partials) {
    if (MOP >= 0) mapop = MOP;
    if (ROP >= 0) redop = ROP;
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    T acc = RedIdent<T>::get(redop);
    using V = T __attribute__((ext_vector_type(2)));
    uint64_t nv = n / 2;
    const V* sv = reinterpret_cast<const V*>(src);
    for (uint64_t j = i; j < nv; j += stride) {
        V v = sv[j];
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.x));
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.y));
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride)
        acc = red_comb(redop, acc, mapf<T>(mapop, src[j]));
    acc = block_reduce(redop, acc);
    if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

// 32 B/thread-iteration variant (two dwordx4 per lane): more memory-
// level parallelism per wave — A/B-gated via DA_RV4.
template <typename T, int MOP = -1, int ROP = -1>
__global__ void reduce_stage1_v4(int mapop, int redop,
                                 const T* __restrict__ src, uint64_t n,
                                 T* __restrict__ partials) {
    if (MOP >= 0) mapop = MOP;
    if (ROP >= 0) redop = ROP;
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    T acc = RedIdent<T>::get(redop);
    using V = T __attribute__((ext_vector_type(4)));
    uint64_t nv = n / 4;
    const V* sv = reinterpret_cast<const V*>(src);
    for (uint64_t j = i; j < nv; j += stride) {
        V v = sv[j];
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.x));
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.y));
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.z));
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.w));
    }
    for (uint64_t j = 4 * nv + i; j < n; j += stride)
        acc = red_comb(redop, acc, mapf<T>(mapop, src[j]));
    acc = block_reduce(redop, acc);
    if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

// Fused single-kernel reduce: stage-1 blocks publish their partial with
// an agent-scope release and take a ticket; the LAST arriver acquires,
// folds all partials and writes the result — the fan-in form of the
// guide's split-K seam recipe (cdna_hip_programming.md §5, "In-launch
// split-K reduction"; no grid-wide wait, so residency is irrelevant).
// The ticket counter lives in the partials buffer, is zeroed at
// allocation, and is reset by the last arriver; same-stream ordering
// makes the reset visible to the next launch.
template <typename T>
__global__ void reduce_fused(int mapop, int redop, const T* __restrict__ src,
                             uint64_t n, T* __restrict__ partials,
                             unsigned int* __restrict__ ticket,
                             T* __restrict__ out) {
    __shared__ int is_last;
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    T acc = RedIdent<T>::get(redop);
    using V = T __attribute__((ext_vector_type(2)));
    uint64_t nv = n / 2;
    const V* sv = reinterpret_cast<const V*>(src);
    for (uint64_t j = i; j < nv; j += stride) {
        V v = sv[j];
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.x));
        acc = red_comb(redop, acc, mapf<T>(mapop, (T)v.y));
    }
    for (uint64_t j = 2 * nv + i; j < n; j += stride)
        acc = red_comb(redop, acc, mapf<T>(mapop, src[j]));
    acc = block_reduce(redop, acc);
    if (threadIdx.x == 0) partials[blockIdx.x] = acc;
    // publish: wait the partial store, release, ticket (guide order:
    // fence THEN fetch_add, with the asm vmcnt wait restated after the
    // fence — ROCm 7.2 drops the post-wbl2 wait otherwise)
    __asm__ volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        __asm__ volatile("s_waitcnt vmcnt(0)" ::: "memory");
        unsigned int t = __hip_atomic_fetch_add(
            ticket, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        is_last = (t == gridDim.x - 1);
    }
    __syncthreads();
    if (!is_last) return;
    if (threadIdx.x == 0)
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    __syncthreads();
    T facc = RedIdent<T>::get(redop);
    for (unsigned int j = threadIdx.x; j < gridDim.x; j += blockDim.x)
        facc = red_comb(redop, facc, partials[j]);
    __syncthreads();   // block_reduce LDS reused after the first pass
    facc = block_reduce(redop, facc);
    if (threadIdx.x == 0) {
        out[0] = facc;
        *ticket = 0;   // next launch on this stream sees 0
    }
}

template <typename T>
__global__ void reduce_stage2(int redop, const T* __restrict__ partials,
                              int np, T* __restrict__ out) {
    T acc = RedIdent<T>::get(redop);
    for (int j = threadIdx.x; j < np; j += blockDim.x)
        acc = red_comb(redop, acc, partials[j]);
    acc = block_reduce(redop, acc);
    if (threadIdx.x == 0) out[0] = acc;
}

template <typename T>
static int do_reduce(int mapop, int redop, const T* src, uint64_t n,
                     void* out_host, hipStream_t s) {
    if (n == 0) {  // fold identity (empty-chunk semantics)
        *(T*)out_host = RedIdent<T>::get(redop);
        return 0;
    }
    bool v4 = reduce_v4();
    uint64_t want = (n / (v4 ? 4 : 2) + RTPB - 1) / RTPB;
    int g = reduce_grid(want);
    int rc = ensure_partials((RMAXB + 1) * sizeof(double));
    if (rc) return rc;
    T* parts = (T*)st().partials;
    T* dout = (T*)((double*)st().partials + RMAXB);
    unsigned int* ticket = st().red_ticket;
    static int fused = -1;
    if (fused < 0) {
        const char* e = getenv("DA_RED_FUSED");
        fused = e ? atoi(e) : 0;   // measured SLOWER at 2048 blocks (profiles)
    }
    if (fused && !v4) {
        hipLaunchKernelGGL(reduce_fused<T>, dim3(g), dim3(RTPB), 0, s,
                           mapop, redop, src, n, parts, ticket, dout);
        DA_CHECK_HIP(hipGetLastError());
    } else {
        if (v4)
            hipLaunchKernelGGL((reduce_stage1_v4<T>), dim3(g), dim3(RTPB),
                               0, s, mapop, redop, src, n, parts);
        else if (mapop == DA_REDF_IDENTITY && redop == DA_RED_ADD)
            hipLaunchKernelGGL((reduce_stage1<T, DA_REDF_IDENTITY,
                                DA_RED_ADD>), dim3(g), dim3(RTPB), 0, s,
                               mapop, redop, src, n, parts);
        else if (mapop == DA_REDF_ABS2 && redop == DA_RED_ADD)
            hipLaunchKernelGGL((reduce_stage1<T, DA_REDF_ABS2,
                                DA_RED_ADD>), dim3(g), dim3(RTPB), 0, s,
                               mapop, redop, src, n, parts);
        else if (mapop == DA_REDF_ABS && redop == DA_RED_ADD)
            hipLaunchKernelGGL((reduce_stage1<T, DA_REDF_ABS,
                                DA_RED_ADD>), dim3(g), dim3(RTPB), 0, s,
                               mapop, redop, src, n, parts);
        else if (mapop == DA_REDF_IDENTITY && redop == DA_RED_MAX)
            hipLaunchKernelGGL((reduce_stage1<T, DA_REDF_IDENTITY,
                                DA_RED_MAX>), dim3(g), dim3(RTPB), 0, s,
                               mapop, redop, src, n, parts);
        else if (mapop == DA_REDF_IDENTITY && redop == DA_RED_MIN)
            hipLaunchKernelGGL((reduce_stage1<T, DA_REDF_IDENTITY,
                                DA_RED_MIN>), dim3(g), dim3(RTPB), 0, s,
                               mapop, redop, src, n, parts);
        else
            hipLaunchKernelGGL((reduce_stage1<T>), dim3(g), dim3(RTPB),
                               0, s, mapop, redop, src, n, parts);
        DA_CHECK_HIP(hipGetLastError());
        hipLaunchKernelGGL(reduce_stage2<T>, dim3(1), dim3(RTPB), 0, s,
                           redop, parts, g, dout);
        DA_CHECK_HIP(hipGetLastError());
    }
    DA_CHECK_HIP(hipMemcpyAsync(out_host, dout, sizeof(T),
                                hipMemcpyDeviceToHost, s));
    DA_CHECK_HIP(hipStreamSynchronize(s));
    return 0;
}

// ---- dims-reduction (mapreduce.jl:42-94: per-chunk mapreduce(dims=..))
// The chunk is viewed column-major as (inner, axis, outer):
// src[i + a*inner + o*inner*axis] -> dst[i + o*inner], reduced over a.
// Variant A (inner >= 64): one thread per (i,o), serial over axis —
// coalesced across inner.  Variant B (inner < 64): one 64-lane wave per
// (i,o), lanes stride the axis (coalesced along the axis), then a
// wavefront shuffle tree.
// MOP/ROP >= 0 are compile-time op constants (the switches fold, the
// same 2x-recovery trick as map_fixed_kernel); -1 defers to the runtime
// arguments (long-tail combos).
template <typename T, int MOP, int ROP>
__global__ void reduce_dims_threads(int mapop, int redop,
                                    const T* __restrict__ src,
                                    uint64_t inner, uint64_t axis,
                                    uint64_t outer, T* __restrict__ dst) {
    const int mop_ = MOP >= 0 ? MOP : mapop;
    const int rop_ = ROP >= 0 ? ROP : redop;
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t total = inner * outer;
    for (uint64_t e = t; e < total; e += stride) {
        uint64_t i = e % inner, o = e / inner;
        const T* p = src + i + o * inner * axis;
        T acc = RedIdent<T>::get(rop_);
        for (uint64_t a = 0; a < axis; ++a)
            acc = red_comb(rop_, acc, mapf(mop_, p[a * inner]));
        dst[i + o * inner] = acc;
    }
}

template <typename T, int MOP, int ROP>
__global__ void reduce_dims_waves(int mapop, int redop,
                                  const T* __restrict__ src,
                                  uint64_t inner, uint64_t axis,
                                  uint64_t outer, T* __restrict__ dst) {
    const int mop_ = MOP >= 0 ? MOP : mapop;
    const int rop_ = ROP >= 0 ? ROP : redop;
    uint64_t wid = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) / 64;
    uint64_t nw = ((uint64_t)gridDim.x * blockDim.x) / 64;
    int lane = threadIdx.x & 63;
    uint64_t total = inner * outer;
    for (uint64_t e = wid; e < total; e += nw) {
        uint64_t i = e % inner, o = e / inner;
        const T* p = src + i + o * inner * axis;
        T acc = RedIdent<T>::get(rop_);
        for (uint64_t a = lane; a < axis; a += 64)
            acc = red_comb(rop_, acc, mapf(mop_, p[a * inner]));
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
            acc = red_comb(rop_, acc, (T)__shfl_down(acc, off, 64));
        if (lane == 0) dst[i + o * inner] = acc;
    }
}

// Variant C: one 256-thread block per output element — for few outputs
// over a long axis (e.g. sum(D, dims=2) of a square matrix), where the
// thread/wave variants leave the chip underfilled (measured 137 GB/s vs
// 3+ TB/s; profiles/r01_kernel_stats.md).
template <typename T, int MOP, int ROP>
__global__ void reduce_dims_blocks(int mapop, int redop,
                                   const T* __restrict__ src,
                                   uint64_t inner, uint64_t axis,
                                   uint64_t outer, T* __restrict__ dst) {
    const int mop_ = MOP >= 0 ? MOP : mapop;
    const int rop_ = ROP >= 0 ? ROP : redop;
    uint64_t total = inner * outer;
    for (uint64_t e = blockIdx.x; e < total; e += gridDim.x) {
        uint64_t i = e % inner, o = e / inner;
        const T* p = src + i + o * inner * axis;
        T acc = RedIdent<T>::get(rop_);
        for (uint64_t a = threadIdx.x; a < axis; a += blockDim.x)
            acc = red_comb(rop_, acc, mapf(mop_, p[a * inner]));
        acc = block_reduce(rop_, acc);
        if (threadIdx.x == 0) dst[i + o * inner] = acc;
        __syncthreads();   // LDS in block_reduce reused next iteration
    }
}

// Variant D: few outputs, LARGE inner (e.g. sum(8192x8192, dims=2)):
// split the axis into NB column slices; each block walks a row tile of
// its slice with fully coalesced reads (consecutive threads =
// consecutive rows), producing NB partial rows; a second kernel folds
// the NB partials per row.
template <typename T, int MOP, int ROP>
__global__ void reduce_dims_slices1(int mapop, int redop,
                                    const T* __restrict__ src,
                                    uint64_t inner, uint64_t axis,
                                    int nb, T* __restrict__ partials) {
    const int mop_ = MOP >= 0 ? MOP : mapop;
    const int rop_ = ROP >= 0 ? ROP : redop;
    uint64_t o = blockIdx.z;
    const T* base = src + o * inner * axis;
    uint64_t row = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (row >= inner) return;
    uint64_t a0 = axis * blockIdx.y / nb;
    uint64_t a1 = axis * (blockIdx.y + 1) / nb;
    T acc = RedIdent<T>::get(rop_);
    for (uint64_t a = a0; a < a1; ++a)
        acc = red_comb(rop_, acc, mapf(mop_, base[row + a * inner]));
    partials[(o * nb + blockIdx.y) * inner + row] = acc;
}

template <typename T>
__global__ void reduce_dims_slices2(int redop, const T* __restrict__ partials,
                                    uint64_t inner, int nb,
                                    T* __restrict__ dst, uint64_t outer) {
    uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t t = e; t < inner * outer; t += stride) {
        uint64_t row = t % inner, o = t / inner;
        T acc = RedIdent<T>::get(redop);
        for (int b = 0; b < nb; ++b)
            acc = red_comb(redop, acc,
                           partials[(o * nb + b) * inner + row]);
        dst[row + o * inner] = acc;
    }
}

template <typename T, int MOP, int ROP>
static int do_reduce_dims_t(int mapop, int redop, const T* src,
                            uint64_t inner, uint64_t axis, uint64_t outer,
                            T* dst, hipStream_t s) {
    uint64_t total = inner * outer;
    if (total == 0) return 0;
    // axis == 0 still runs: the loop body never executes and dst gets
    // the fold identity.
    if (total < 262144 && inner >= 64 && axis >= 64 && outer <= 64) {
        // fill ~2048 workgroups: (inner/256 row tiles) x nb slices
        uint64_t row_tiles = (inner + RTPB - 1) / RTPB;
        int nb = (int)(2048 / row_tiles);
        if (nb < 1) nb = 1;
        if (nb > 64) nb = 64;
        if ((uint64_t)nb > axis) nb = (int)axis;
        int rc = ensure_partials((uint64_t)nb * inner * outer * sizeof(T));
        if (rc) return rc;
        T* parts = (T*)st().partials;
        dim3 g((inner + RTPB - 1) / RTPB, nb, outer);
        hipLaunchKernelGGL((reduce_dims_slices1<T, MOP, ROP>), g,
                           dim3(RTPB), 0, s,
                           mapop, redop, src, inner, axis, nb, parts);
        DA_CHECK_HIP(hipGetLastError());
        uint64_t want = (total + RTPB - 1) / RTPB;
        int g2 = (int)(want < 1 ? 1 : (want > 2048 ? 2048 : want));
        hipLaunchKernelGGL(reduce_dims_slices2<T>, dim3(g2), dim3(RTPB), 0,
                           s, redop, parts, inner, nb, dst, outer);
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }
    if (total >= 262144 && inner >= 64) {
        // plenty of outputs, coalesced across inner: thread-per-output
        uint64_t want = (total + RTPB - 1) / RTPB;
        int g = (int)(want < 1 ? 1 : (want > 8192 ? 8192 : want));
        hipLaunchKernelGGL((reduce_dims_threads<T, MOP, ROP>), dim3(g),
                           dim3(RTPB), 0,
                           s, mapop, redop, src, inner, axis, outer, dst);
    } else if (inner < 64 && axis >= 64 && total >= 16384) {
        // inner tiny (reduce over leading dim): wave-per-output, lanes
        // stride the axis (coalesced along the axis)
        uint64_t want = (total * 64 + RTPB - 1) / RTPB;
        int g = (int)(want < 1 ? 1 : (want > 8192 ? 8192 : want));
        hipLaunchKernelGGL((reduce_dims_waves<T, MOP, ROP>), dim3(g),
                           dim3(RTPB), 0,
                           s, mapop, redop, src, inner, axis, outer, dst);
    } else {
        // few outputs / long axis: block-per-output
        int g = (int)(total < 1 ? 1 : (total > 8192 ? 8192 : total));
        hipLaunchKernelGGL((reduce_dims_blocks<T, MOP, ROP>), dim3(g),
                           dim3(RTPB), 0,
                           s, mapop, redop, src, inner, axis, outer, dst);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

// hot (mapop, redop) combos get constant-folded instantiations — the
// dims-reduce analog of the map_fixed 2x recovery; everything else
// rides the runtime-switch fallback.
template <typename T>
static int do_reduce_dims(int mapop, int redop, const T* src,
                          uint64_t inner, uint64_t axis, uint64_t outer,
                          T* dst, hipStream_t s) {
    if (redop == DA_RED_ADD) {
        if (mapop == DA_REDF_IDENTITY)
            return do_reduce_dims_t<T, DA_REDF_IDENTITY, DA_RED_ADD>(
                mapop, redop, src, inner, axis, outer, dst, s);
        if (mapop == DA_REDF_ABS2)
            return do_reduce_dims_t<T, DA_REDF_ABS2, DA_RED_ADD>(
                mapop, redop, src, inner, axis, outer, dst, s);
    }
    if (redop == DA_RED_MAX && mapop == DA_REDF_IDENTITY)
        return do_reduce_dims_t<T, DA_REDF_IDENTITY, DA_RED_MAX>(
            mapop, redop, src, inner, axis, outer, dst, s);
    if (redop == DA_RED_MIN && mapop == DA_REDF_IDENTITY)
        return do_reduce_dims_t<T, DA_REDF_IDENTITY, DA_RED_MIN>(
            mapop, redop, src, inner, axis, outer, dst, s);
    return do_reduce_dims_t<T, -1, -1>(mapop, redop, src, inner, axis,
                                       outer, dst, s);
}

int launch_reduce_dims(int mapop, int redop, const void* src,
                       uint64_t inner, uint64_t axis, uint64_t outer,
                       int dtype, void* dst, hipStream_t s) {
    if (mapop < 0 || mapop > DA_REDF_NONZERO || redop < 0 || redop > DA_RED_MAX)
        return set_err(-3, "da_reduce_dims: bad op (%d,%d)", mapop, redop);
    switch (dtype) {
    case DA_F64: return do_reduce_dims<double>(mapop, redop,
        (const double*)src, inner, axis, outer, (double*)dst, s);
    case DA_F32: return do_reduce_dims<float>(mapop, redop,
        (const float*)src, inner, axis, outer, (float*)dst, s);
    case DA_I64: return do_reduce_dims<int64_t>(mapop, redop,
        (const int64_t*)src, inner, axis, outer, (int64_t*)dst, s);
    }
    return set_err(-3, "da_reduce_dims: bad dtype %d", dtype);
}

int launch_reduce(int mapop, int redop, const void* src, uint64_t n,
                  int dtype, void* out_host, hipStream_t s) {
    if (mapop < 0 || mapop > DA_REDF_NONZERO || redop < 0 || redop > DA_RED_MAX)
        return set_err(-3, "da_reduce: bad op (%d,%d)", mapop, redop);
    switch (dtype) {
    case DA_F64: return do_reduce<double>(mapop, redop, (const double*)src,
                                          n, out_host, s);
    case DA_F32: return do_reduce<float>(mapop, redop, (const float*)src,
                                         n, out_host, s);
    case DA_I64: return do_reduce<int64_t>(mapop, redop, (const int64_t*)src,
                                           n, out_host, s);
    }
    return set_err(-3, "da_reduce: bad dtype %d", dtype);
}

} // namespace da
