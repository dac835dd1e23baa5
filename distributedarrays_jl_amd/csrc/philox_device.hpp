// philox_device.hpp — device-side Philox4x32-10, bit-identical to
// oracle/philox.py (the synthetic-input protocol of BASELINE.md).
// Pinned by the Random123 known-answer vectors (tests/test_gpu_parity.py
// compares device output against the numpy oracle bit-exactly).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

namespace da {

struct u32x4 { uint32_t v[4]; };

__device__ __forceinline__ u32x4 philox4x32_10(uint64_t block, uint64_t seed) {
    const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
    uint32_t c0 = (uint32_t)(block & 0xFFFFFFFFu);
    uint32_t c1 = (uint32_t)(block >> 32);
    uint32_t c2 = 0, c3 = 0;
    uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFu);
    uint32_t k1 = (uint32_t)(seed >> 32);
#pragma unroll
    for (int r = 0; r < 10; ++r) {
        uint64_t p0 = (uint64_t)M0 * c0;
        uint64_t p1 = (uint64_t)M1 * c2;
        uint32_t hi0 = (uint32_t)(p0 >> 32), lo0 = (uint32_t)p0;
        uint32_t hi1 = (uint32_t)(p1 >> 32), lo1 = (uint32_t)p1;
        uint32_t n0 = hi1 ^ c1 ^ k0;
        uint32_t n1 = lo1;
        uint32_t n2 = hi0 ^ c3 ^ k1;
        uint32_t n3 = lo0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += W0; k1 += W1;
    }
    return {c0, c1, c2, c3};
}

__device__ __forceinline__ double u01_f64(uint32_t lo, uint32_t hi) {
    uint64_t u = ((uint64_t)hi << 32) | lo;
    return (double)(u >> 11) * 0x1.0p-53;
}

__device__ __forceinline__ float u01_f32(uint32_t w) {
    return (float)(w >> 8) * 0x1.0p-24f;
}

} // namespace da
