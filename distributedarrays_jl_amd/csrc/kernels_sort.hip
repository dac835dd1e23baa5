// kernels_sort.hip — local sort + splitter search for the distributed
// samplesort (/root/reference/src/sort.jl:103-170: per-chunk sort,
// sampled splitters, scatter by boundary, local re-sort).  The local
// sorts run on rocPRIM's device radix sort (the MI355X-native analog of
// the reference's Base.sort! on the worker); boundaries are found with
// a tiny device binary-search kernel on the sorted chunk.
#include "common.hpp"
#include <string.h>
#include <rocprim/device/device_radix_sort.hpp>

namespace da {

template <typename T>
static int do_sort_out(const T* src, T* dst, uint64_t n, hipStream_t s) {
    if (n == 0) return 0;
    if (n == 1) {
        DA_CHECK_HIP(hipMemcpyAsync(dst, src, sizeof(T),
                                    hipMemcpyDeviceToDevice, s));
        return 0;
    }
    size_t tmp_bytes = 0;
    hipError_t e = rocprim::radix_sort_keys(nullptr, tmp_bytes, src,
                                            (T*)nullptr, n, 0,
                                            sizeof(T) * 8, s);
    if (e != hipSuccess)
        return set_err(-(1000 + (int)e), "radix_sort query: %s",
                       hipGetErrorString(e));
    void* tmp = nullptr;
    int rc = da_alloc(tmp_bytes, 0, &tmp);
    if (rc) return rc;
    e = rocprim::radix_sort_keys(tmp, tmp_bytes, src, dst, n, 0,
                                 sizeof(T) * 8, s);
    if (e != hipSuccess) {
        da_free(tmp);
        return set_err(-(1000 + (int)e), "radix_sort: %s",
                       hipGetErrorString(e));
    }
    DA_CHECK_HIP(hipStreamSynchronize(s));
    da_free(tmp);
    return 0;
}

template <typename T>
static int do_sort(T* chunk, uint64_t n, hipStream_t s) {
    if (n <= 1) return 0;
    void* out = nullptr;
    int rc = da_alloc(n * sizeof(T), 0, &out);
    if (rc) return rc;
    rc = do_sort_out<T>((const T*)chunk, (T*)out, n, s);
    if (rc) { da_free(out); return rc; }
    DA_CHECK_HIP(hipMemcpyAsync(chunk, out, n * sizeof(T),
                                hipMemcpyDeviceToDevice, s));
    DA_CHECK_HIP(hipStreamSynchronize(s));
    da_free(out);
    return 0;
}

template <typename T>
__global__ void lower_bound_kernel(const T* __restrict__ a, uint64_t n,
                                   const T* __restrict__ splits, int k,
                                   uint64_t* __restrict__ out) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= k) return;
    T s = splits[i];
    uint64_t lo = 0, hi = n;
    while (lo < hi) {
        uint64_t mid = (lo + hi) / 2;
        if (a[mid] < s) lo = mid + 1;
        else hi = mid;
    }
    out[i] = lo;
}

template <typename T>
static int do_lower_bound(const T* sorted, uint64_t n, const T* spl_host,
                          int k, uint64_t* out_host, hipStream_t s) {
    int rc = ensure_scratch(k * (sizeof(T) + sizeof(uint64_t)));
    if (rc) return rc;
    T* dspl = (T*)st().scratch;
    uint64_t* dout = (uint64_t*)(dspl + k);
    DA_CHECK_HIP(hipMemcpyAsync(dspl, spl_host, k * sizeof(T),
                                hipMemcpyHostToDevice, s));
    hipLaunchKernelGGL(lower_bound_kernel<T>, dim3((k + 63) / 64),
                       dim3(64), 0, s, sorted, n, dspl, k, dout);
    DA_CHECK_HIP(hipGetLastError());
    DA_CHECK_HIP(hipMemcpyAsync(out_host, dout, k * sizeof(uint64_t),
                                hipMemcpyDeviceToHost, s));
    DA_CHECK_HIP(hipStreamSynchronize(s));
    return 0;
}

} // namespace da

using namespace da;

extern "C" {

/* In-place ascending sort of a device chunk (per-chunk stage of the
 * samplesort, sort.jl:109-116). */
int da_sort(void* chunk, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    switch (dtype) {
    case DA_F64: return do_sort<double>((double*)chunk, n, st().stream);
    case DA_F32: return do_sort<float>((float*)chunk, n, st().stream);
    case DA_I64: return do_sort<int64_t>((int64_t*)chunk, n, st().stream);
    }
    return set_err(-3, "da_sort: bad dtype %d", dtype);
}

/* Out-of-place ascending sort: src -> dst (saves the in-place variant's
 * copy-back; the samplesort local stage sorts straight into the result
 * chunk). */
int da_sort_out(const void* src, void* dst, uint64_t n, int dtype) {
    DA_REQUIRE_INIT();
    switch (dtype) {
    case DA_F64: return do_sort_out<double>((const double*)src,
        (double*)dst, n, st().stream);
    case DA_F32: return do_sort_out<float>((const float*)src,
        (float*)dst, n, st().stream);
    case DA_I64: return do_sort_out<int64_t>((const int64_t*)src,
        (int64_t*)dst, n, st().stream);
    }
    return set_err(-3, "da_sort_out: bad dtype %d", dtype);
}

/* k lower-bound indices of host splitter values in a sorted device
 * chunk (the scatter-boundary search of sort.jl:118-140). */
int da_lower_bound(const void* sorted, uint64_t n, int dtype,
                   const void* splitters, int k, uint64_t* out) {
    DA_REQUIRE_INIT();
    if (k <= 0) return 0;
    switch (dtype) {
    case DA_F64: return do_lower_bound<double>((const double*)sorted, n,
        (const double*)splitters, k, out, st().stream);
    case DA_F32: return do_lower_bound<float>((const float*)sorted, n,
        (const float*)splitters, k, out, st().stream);
    case DA_I64: return do_lower_bound<int64_t>((const int64_t*)sorted, n,
        (const int64_t*)splitters, k, out, st().stream);
    }
    return set_err(-3, "da_lower_bound: bad dtype %d", dtype);
}

} // extern "C"
