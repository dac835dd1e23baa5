// kernels_gemm.hip — local dense C = alpha*A*B + beta*C, Float64,
#include <stdlib.h>
// COLUMN-major (Julia layout), MFMA-tiled for gfx950 (CDNA4).
//
// Replaces the worker-side `localpart(A) * Bjk` of
// /root/reference/src/linalg.jl:224 (the per-tile GEMM inside
// _matmatmul!, linalg.jl:190-253).  The reference runs OpenBLAS dgemm on
// the host; here the chunk lives in HBM and the GEMM runs on the MFMA
// f64 pipe (v_mfma_f64_16x16x4_f64).
//
// Structure (v1): 128x128 block tile, BK=16, 4 waves (2x2), each wave a
// 64x64 sub-tile = 4x4 fragments of 16x16, LDS-staged operands with an
// 18-double row stride (conflict-spread for ds_read_b64), accumulate in
// AGPR f64x4.  Sizes not multiples of the tile fall back to a naive
// per-thread kernel (parity path for small chunks).
#include "common.hpp"

namespace da {

typedef double f64x4 __attribute__((ext_vector_type(4)));
typedef double f64x2 __attribute__((ext_vector_type(2)));

#define BM 128
#define BN 128
#define BK 16
#define LSTR 17   // LDS row stride (odd: staging writes 2-way not 4-way conflicted; frag reads stay conflict-free — profiles/r01 PMC)

__global__ __launch_bounds__(256, 2)
void gemm_f64_mfma(const double* __restrict__ A, const double* __restrict__ B,
                   double* __restrict__ C, int64_t m, int64_t n, int64_t k,
                   int64_t lda, int64_t ldb, int64_t ldc,
                   double alpha, double beta) {
    __shared__ double As[BM * LSTR];   // As[mm][kk] at mm*LSTR + kk
    __shared__ double Bs[BN * LSTR];   // Bs[nn][kk] at nn*LSTR + kk

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int l4 = lane >> 4;
    const int wr = (wave >> 1) * 64;   // wave row origin in tile
    const int wc = (wave & 1) * 64;    // wave col origin

    const int64_t bm = (int64_t)blockIdx.x * BM;
    const int64_t bn = (int64_t)blockIdx.y * BN;

    f64x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.0, 0.0, 0.0, 0.0};

    const int64_t ktiles = k / BK;
    for (int64_t kt = 0; kt < ktiles; ++kt) {
        const int64_t k0 = kt * BK;
        // Stage A tile (BM x BK): column c of the tile is contiguous in
        // global memory (column-major).  1024 double2 pieces, 4/thread.
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int idx = tid + r * 256;          // 0..1023
            int c = idx >> 6;                 // k-column 0..15
            int row2 = (idx & 63) * 2;        // row pair
            f64x2 v = *reinterpret_cast<const f64x2*>(
                A + (k0 + c) * lda + bm + row2);
            As[(row2 + 0) * LSTR + c] = v.x;
            As[(row2 + 1) * LSTR + c] = v.y;
        }
        // Stage B tile (BK x BN): column n' has BK contiguous doubles.
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int idx = tid + r * 256;
            int nn = idx >> 3;                // tile column 0..127
            int k2 = (idx & 7) * 2;           // k pair
            f64x2 v = *reinterpret_cast<const f64x2*>(
                B + (bn + nn) * ldb + k0 + k2);
            Bs[nn * LSTR + k2] = v.x;
        Bs[nn * LSTR + k2 + 1] = v.y;
        }
        __syncthreads();
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            double a[4], b[4];
            const int kof = kk * 4 + l4;
#pragma unroll
            for (int i = 0; i < 4; ++i)
                a[i] = As[(wr + i * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int j = 0; j < 4; ++j)
                b[j] = Bs[(wc + j * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }

    // Epilogue: v_mfma_f64_16x16x4_f64 C/D fragment map (probed on
    // hardware, tests/test_gpu_mfma_probe.py):
    //   col = lane & 15, row = 4*q + (lane >> 4)   (q = acc register)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            int64_t col = bn + wc + j * 16 + l16;
            double* cp = C + col * ldc + bm + wr + i * 16 + l4;
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                double v = alpha * acc[i][j][q];
                cp[4 * q] = (beta == 0.0) ? v : v + beta * cp[4 * q];
            }
        }
    }
}

// v2: same 128x128x16 tiling with (a) a register-prefetch pipeline —
// tile t+1's global loads issue before tile t's MFMA phase and land in
// LDS after the barrier (write-after-barrier form of guideline T14) —
// and (b) a bijective XCD-aware block remap for per-XCD L2 locality
// (cdna_hip_programming.md §5: XCD swizzle, +10% when HBM-bound).
template <int TBK>
__global__ __launch_bounds__(256, 2)
void gemm_f64_mfma_v2(const double* __restrict__ A,
                      const double* __restrict__ B,
                      double* __restrict__ C, int64_t m, int64_t n,
                      int64_t k, int64_t lda, int64_t ldb, int64_t ldc,
                      double alpha, double beta) {
    constexpr int TSTR = TBK + 2;      // LDS row stride (bank-spread)
    constexpr int NP = TBK / 4;        // staging pieces per thread/side
    __shared__ double As[BM * TSTR];
    __shared__ double Bs[BN * TSTR];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int l4 = lane >> 4;
    const int wr = (wave >> 1) * 64;
    const int wc = (wave & 1) * 64;

    // bijective XCD remap (dispatcher places block b on XCD b%8)
    const int gx = gridDim.x, nwg = gridDim.x * gridDim.y;
    int w = blockIdx.y * gx + blockIdx.x;
    int q = nwg >> 3, rmd = nwg & 7, xcd = w & 7, idx = w >> 3;
    int sw = (xcd < rmd ? xcd * (q + 1) : rmd * (q + 1) + (xcd - rmd) * q)
             + idx;
    const int64_t bm = (int64_t)(sw % gx) * BM;
    const int64_t bn = (int64_t)(sw / gx) * BN;

    f64x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.0, 0.0, 0.0, 0.0};

    // per-thread staging coordinates (NP pieces of 2 doubles each side)
    int a_c[NP], a_r2[NP], b_n[NP], b_k2[NP];
#pragma unroll
    for (int r = 0; r < NP; ++r) {
        int idx2 = tid + r * 256;
        a_c[r] = idx2 >> 6;
        a_r2[r] = (idx2 & 63) * 2;
        b_n[r] = idx2 / (TBK / 2);
        b_k2[r] = (idx2 % (TBK / 2)) * 2;
    }

    f64x2 pa[NP], pb[NP];
    const int64_t ktiles = k / TBK;
    // prologue: tile 0 -> regs -> LDS
#pragma unroll
    for (int r = 0; r < NP; ++r) {
        pa[r] = *reinterpret_cast<const f64x2*>(A + (int64_t)a_c[r] * lda
                                                + bm + a_r2[r]);
        pb[r] = *reinterpret_cast<const f64x2*>(B + (bn + b_n[r]) * ldb
                                                + b_k2[r]);
    }
#pragma unroll
    for (int r = 0; r < NP; ++r) {
        As[(a_r2[r] + 0) * TSTR + a_c[r]] = pa[r].x;
        As[(a_r2[r] + 1) * TSTR + a_c[r]] = pa[r].y;
        Bs[b_n[r] * TSTR + b_k2[r]] = pb[r].x;
        Bs[b_n[r] * TSTR + b_k2[r] + 1] = pb[r].y;
    }

    for (int64_t kt = 0; kt < ktiles; ++kt) {
        __syncthreads();   // LDS tile kt visible to all
        if (kt + 1 < ktiles) {
            const int64_t k0 = (kt + 1) * TBK;
#pragma unroll
            for (int r = 0; r < NP; ++r) {
                pa[r] = *reinterpret_cast<const f64x2*>(
                    A + (k0 + a_c[r]) * lda + bm + a_r2[r]);
                pb[r] = *reinterpret_cast<const f64x2*>(
                    B + (bn + b_n[r]) * ldb + k0 + b_k2[r]);
            }
        }
#pragma unroll
        for (int kk = 0; kk < TBK / 4; ++kk) {
            double a[4], b[4];
            const int kof = kk * 4 + l4;
#pragma unroll
            for (int i = 0; i < 4; ++i)
                a[i] = As[(wr + i * 16 + l16) * TSTR + kof];
#pragma unroll
            for (int j = 0; j < 4; ++j)
                b[j] = Bs[(wc + j * 16 + l16) * TSTR + kof];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();   // MFMA phase done; LDS reusable
        if (kt + 1 < ktiles) {
#pragma unroll
            for (int r = 0; r < NP; ++r) {
                As[(a_r2[r] + 0) * TSTR + a_c[r]] = pa[r].x;
                As[(a_r2[r] + 1) * TSTR + a_c[r]] = pa[r].y;
                Bs[b_n[r] * TSTR + b_k2[r]] = pb[r].x;
                Bs[b_n[r] * TSTR + b_k2[r] + 1] = pb[r].y;
            }
        }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            int64_t col = bn + wc + j * 16 + l16;
            double* cp = C + col * ldc + bm + wr + i * 16 + l4;
#pragma unroll
            for (int qq = 0; qq < 4; ++qq) {
                double v = alpha * acc[i][j][qq];
                cp[4 * qq] = (beta == 0.0) ? v : v + beta * cp[4 * qq];
            }
        }
    }
}

// v3: 128x128x16 tile, 512 threads = 8 waves of 64x32 sub-tiles
// (acc = 8 fragments = 64 VGPRs vs v2's 128): 4 waves/SIMD co-residency
// so one block's stage phase hides under another's MFMA phase.
__global__ __launch_bounds__(512, 4)
void gemm_f64_mfma_v3(const double* __restrict__ A,
                      const double* __restrict__ B,
                      double* __restrict__ C, int64_t m, int64_t n,
                      int64_t k, int64_t lda, int64_t ldb, int64_t ldc,
                      double alpha, double beta) {
    __shared__ double As[BM * LSTR];
    __shared__ double Bs[BN * LSTR];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int l4 = lane >> 4;
    const int wr = (wave >> 2) * 64;   // 2 wave-rows of 64
    const int wc = (wave & 3) * 32;    // 4 wave-cols of 32

    const int gx = gridDim.x, nwg = gridDim.x * gridDim.y;
    int w = blockIdx.y * gx + blockIdx.x;
    int q = nwg >> 3, rmd = nwg & 7, xcd = w & 7, idx = w >> 3;
    int sw = (xcd < rmd ? xcd * (q + 1) : rmd * (q + 1) + (xcd - rmd) * q)
             + idx;
    const int64_t bm = (int64_t)(sw % gx) * BM;
    const int64_t bn = (int64_t)(sw / gx) * BN;

    f64x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.0, 0.0, 0.0, 0.0};

    // staging: A/B tiles are 2048 doubles each; 512 threads x 2 f64x2
    // (coordinates recomputed inline — keeping them in arrays costs
    // VGPRs against the __launch_bounds__(512,4) budget)
#define A_C(r)  ((tid + (r) * 512) >> 6)
#define A_R2(r) (((tid + (r) * 512) & 63) * 2)
#define B_N(r)  ((tid + (r) * 512) >> 3)
#define B_K2(r) (((tid + (r) * 512) & 7) * 2)

    f64x2 pa[2], pb[2];
    const int64_t ktiles = k / BK;
#pragma unroll
    for (int r = 0; r < 2; ++r) {
        pa[r] = *reinterpret_cast<const f64x2*>(A + (int64_t)A_C(r) * lda
                                                + bm + A_R2(r));
        pb[r] = *reinterpret_cast<const f64x2*>(B + (bn + B_N(r)) * ldb
                                                + B_K2(r));
    }
#pragma unroll
    for (int r = 0; r < 2; ++r) {
        As[(A_R2(r) + 0) * LSTR + A_C(r)] = pa[r].x;
        As[(A_R2(r) + 1) * LSTR + A_C(r)] = pa[r].y;
        Bs[B_N(r) * LSTR + B_K2(r)] = pb[r].x;
        Bs[B_N(r) * LSTR + B_K2(r) + 1] = pb[r].y;
    }

    for (int64_t kt = 0; kt < ktiles; ++kt) {
        __syncthreads();
        if (kt + 1 < ktiles) {
            const int64_t k0 = (kt + 1) * BK;
#pragma unroll
            for (int r = 0; r < 2; ++r) {
                pa[r] = *reinterpret_cast<const f64x2*>(
                    A + (k0 + A_C(r)) * lda + bm + A_R2(r));
                pb[r] = *reinterpret_cast<const f64x2*>(
                    B + (bn + B_N(r)) * ldb + k0 + B_K2(r));
            }
        }
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            double a[4], b[2];
            const int kof = kk * 4 + l4;
#pragma unroll
            for (int i = 0; i < 4; ++i)
                a[i] = As[(wr + i * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int j = 0; j < 2; ++j)
                b[j] = Bs[(wc + j * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
        if (kt + 1 < ktiles) {
#pragma unroll
            for (int r = 0; r < 2; ++r) {
                As[(A_R2(r) + 0) * LSTR + A_C(r)] = pa[r].x;
                As[(A_R2(r) + 1) * LSTR + A_C(r)] = pa[r].y;
                Bs[B_N(r) * LSTR + B_K2(r)] = pb[r].x;
                Bs[B_N(r) * LSTR + B_K2(r) + 1] = pb[r].y;
            }
        }
    }
#undef A_C
#undef A_R2
#undef B_N
#undef B_K2

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int64_t col = bn + wc + j * 16 + l16;
            double* cp = C + col * ldc + bm + wr + i * 16 + l4;
#pragma unroll
            for (int qq = 0; qq < 4; ++qq) {
                double v = alpha * acc[i][j][qq];
                cp[4 * qq] = (beta == 0.0) ? v : v + beta * cp[4 * qq];
            }
        }
    }
}


// v5: v3 + LDS double-buffering — tile kt lives in buffer kt&1, tile
// kt+1 is written to the other buffer after the MFMA phase, so each
// K-tile needs ONE barrier instead of two (the 2-barrier stage+drain
// pattern is the known ceiling of simple MFMA GEMM loops —
// cdna_hip_programming.md §5).  LDS 73.7 KB -> still 2 blocks/CU.
__global__ __launch_bounds__(512, 4)
void gemm_f64_mfma_v5(const double* __restrict__ A,
                      const double* __restrict__ B,
                      double* __restrict__ C, int64_t m, int64_t n,
                      int64_t k, int64_t lda, int64_t ldb, int64_t ldc,
                      double alpha, double beta) {
    __shared__ double lds[2 * 2 * BM * LSTR];   // [buf][A/B][...]
    // buffer b: A at lds + b*2*BM*LSTR, B at that + BM*LSTR

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int l4 = lane >> 4;
    const int wr = (wave >> 2) * 64;
    const int wc = (wave & 3) * 32;

    const int gx = gridDim.x, nwg = gridDim.x * gridDim.y;
    int w = blockIdx.y * gx + blockIdx.x;
    int q = nwg >> 3, rmd = nwg & 7, xcd = w & 7, idx = w >> 3;
    int sw = (xcd < rmd ? xcd * (q + 1) : rmd * (q + 1) + (xcd - rmd) * q)
             + idx;
    const int64_t bm = (int64_t)(sw % gx) * BM;
    const int64_t bn = (int64_t)(sw / gx) * BN;

    f64x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.0, 0.0, 0.0, 0.0};

#define A_C(r)  ((tid + (r) * 512) >> 6)
#define A_R2(r) (((tid + (r) * 512) & 63) * 2)
#define B_N(r)  ((tid + (r) * 512) >> 3)
#define B_K2(r) (((tid + (r) * 512) & 7) * 2)

    f64x2 pa[2], pb[2];
    const int64_t ktiles = k / BK;
#pragma unroll
    for (int r = 0; r < 2; ++r) {
        pa[r] = *reinterpret_cast<const f64x2*>(A + (int64_t)A_C(r) * lda
                                                + bm + A_R2(r));
        pb[r] = *reinterpret_cast<const f64x2*>(B + (bn + B_N(r)) * ldb
                                                + B_K2(r));
    }
#pragma unroll
    for (int r = 0; r < 2; ++r) {
        lds[(A_R2(r) + 0) * LSTR + A_C(r)] = pa[r].x;
        lds[(A_R2(r) + 1) * LSTR + A_C(r)] = pa[r].y;
        lds[BM * LSTR + B_N(r) * LSTR + B_K2(r)] = pb[r].x;
        lds[BM * LSTR + B_N(r) * LSTR + B_K2(r) + 1] = pb[r].y;
    }

    for (int64_t kt = 0; kt < ktiles; ++kt) {
        __syncthreads();   // buf[kt&1] complete; everyone left iter kt-1
        const double* As_c = lds + (kt & 1) * 2 * BM * LSTR;
        const double* Bs_c = As_c + BM * LSTR;
        if (kt + 1 < ktiles) {
            const int64_t k0 = (kt + 1) * BK;
#pragma unroll
            for (int r = 0; r < 2; ++r) {
                pa[r] = *reinterpret_cast<const f64x2*>(
                    A + (k0 + A_C(r)) * lda + bm + A_R2(r));
                pb[r] = *reinterpret_cast<const f64x2*>(
                    B + (bn + B_N(r)) * ldb + k0 + B_K2(r));
            }
        }
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            double a[4], b[2];
            const int kof = kk * 4 + l4;
#pragma unroll
            for (int i = 0; i < 4; ++i)
                a[i] = As_c[(wr + i * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int j = 0; j < 2; ++j)
                b[j] = Bs_c[(wc + j * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f64_16x16x4f64(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        if (kt + 1 < ktiles) {
            double* As_n = lds + ((kt + 1) & 1) * 2 * BM * LSTR;
            double* Bs_n = As_n + BM * LSTR;
#pragma unroll
            for (int r = 0; r < 2; ++r) {
                As_n[(A_R2(r) + 0) * LSTR + A_C(r)] = pa[r].x;
                As_n[(A_R2(r) + 1) * LSTR + A_C(r)] = pa[r].y;
                Bs_n[B_N(r) * LSTR + B_K2(r)] = pb[r].x;
                Bs_n[B_N(r) * LSTR + B_K2(r) + 1] = pb[r].y;
            }
        }
    }
#undef A_C
#undef A_R2
#undef B_N
#undef B_K2

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int64_t col = bn + wc + j * 16 + l16;
            double* cp = C + col * ldc + bm + wr + i * 16 + l4;
#pragma unroll
            for (int qq = 0; qq < 4; ++qq) {
                double v = alpha * acc[i][j][qq];
                cp[4 * qq] = (beta == 0.0) ? v : v + beta * cp[4 * qq];
            }
        }
    }
}


typedef float f32x4v __attribute__((ext_vector_type(4)));

__global__ void mfma_probe_f32_kernel(const float* __restrict__ A,
                                      const float* __restrict__ B,
                                      float* __restrict__ out_raw) {
    int l = threadIdx.x;
    float a = A[(l & 15) + 16 * (l >> 4)];   // A[row, k] col-major 16x4
    float b = B[(l >> 4) + 4 * (l & 15)];    // B[k, col] col-major 4x16
    f32x4v acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
#pragma unroll
    for (int q = 0; q < 4; ++q) out_raw[l * 4 + q] = acc[q];
}

extern "C" int dbg_mfma_probe_f32(const void* A, const void* B,
                                  void* out_raw);
int dbg_mfma_probe_f32_impl(const void* A, const void* B, void* out_raw,
                            hipStream_t s) {
    hipLaunchKernelGGL(mfma_probe_f32_kernel, dim3(1), dim3(64), 0, s,
                       (const float*)A, (const float*)B, (float*)out_raw);
    DA_CHECK_HIP(hipGetLastError());
    DA_CHECK_HIP(hipStreamSynchronize(s));
    return 0;
}

// Naive fallback for arbitrary shapes (small parity chunks).
__global__ void gemm_f64_naive(const double* __restrict__ A,
                               const double* __restrict__ B,
                               double* __restrict__ C,
                               int64_t m, int64_t n, int64_t k,
                               int64_t lda, int64_t ldb, int64_t ldc,
                               double alpha, double beta) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t j = (int64_t)blockIdx.y * blockDim.y + threadIdx.y;
    if (i >= m || j >= n) return;
    double s = 0.0;
    for (int64_t kk = 0; kk < k; ++kk)
        s += A[kk * lda + i] * B[j * ldb + kk];
    double v = alpha * s;
    C[j * ldc + i] = (beta == 0.0) ? v : v + beta * C[j * ldc + i];
}

// beta-only scaling when k == 0 (fill!/rmul! branch of linalg.jl:232-240)
__global__ void scale_c(double* __restrict__ C, int64_t m, int64_t n,
                        int64_t ldc, double beta) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t j = (int64_t)blockIdx.y * blockDim.y + threadIdx.y;
    if (i >= m || j >= n) return;
    C[j * ldc + i] = (beta == 0.0) ? 0.0 : beta * C[j * ldc + i];
}

// Debug probe: one v_mfma_f64_16x16x4_f64 with the lane mapping this
// file assumes.  out_c = 16x16 col-major under the assumed C map;
// out_raw = acc[q] per (lane,q) so a test can reverse-engineer the true
// map if the assumption is wrong.  Exported as dbg_mfma_probe_f64.
__global__ void mfma_probe_kernel(const double* __restrict__ A,
                                  const double* __restrict__ B,
                                  double* __restrict__ out_c,
                                  double* __restrict__ out_raw) {
    int l = threadIdx.x;
    double a = A[(l & 15) + 16 * (l >> 4)];   // A[row, k] col-major 16x4
    double b = B[(l >> 4) + 4 * (l & 15)];    // B[k, col] col-major 4x16
    f64x4 acc = {0.0, 0.0, 0.0, 0.0};
    acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
#pragma unroll
    for (int q = 0; q < 4; ++q) {
        out_raw[l * 4 + q] = acc[q];
        int row = 4 * q + (l >> 4), col = l & 15;   // probed f64 C/D map
        out_c[row + 16 * col] = acc[q];
    }
}

extern "C" int dbg_mfma_probe_f64(const void* A, const void* B,
                                  void* out_c, void* out_raw);
int dbg_mfma_probe_impl(const void* A, const void* B, void* out_c,
                        void* out_raw, hipStream_t s) {
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s,
                       (const double*)A, (const double*)B,
                       (double*)out_c, (double*)out_raw);
    DA_CHECK_HIP(hipGetLastError());
    DA_CHECK_HIP(hipStreamSynchronize(s));
    return 0;
}


// f32 GEMM on the exact f32-input MFMA (v_mfma_f32_16x16x4_f32, 157 TF
// peak = the f32 vector rate; bitwise an fmaf chain).  Same v3
// structure; C/D lane map for f32 16x16x4 probed on hardware
// (tools/probe_f32.py): col = lane&15, row = 4*(lane>>4) + q — NOTE:
// opposite q/lane roles vs the f64 instruction.
__global__ __launch_bounds__(512, 4)
void gemm_f32_mfma_v3(const float* __restrict__ A,
                      const float* __restrict__ B,
                      float* __restrict__ C, int64_t m, int64_t n,
                      int64_t k, int64_t lda, int64_t ldb, int64_t ldc,
                      float alpha, float beta) {
    typedef float f32x4 __attribute__((ext_vector_type(4)));
    __shared__ float As[BM * LSTR];
    __shared__ float Bs[BN * LSTR];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int l4 = lane >> 4;
    const int wr = (wave >> 2) * 64;
    const int wc = (wave & 3) * 32;

    const int gx = gridDim.x, nwg = gridDim.x * gridDim.y;
    int w = blockIdx.y * gx + blockIdx.x;
    int q = nwg >> 3, rmd = nwg & 7, xcd = w & 7, idx = w >> 3;
    int sw = (xcd < rmd ? xcd * (q + 1) : rmd * (q + 1) + (xcd - rmd) * q)
             + idx;
    const int64_t bm = (int64_t)(sw % gx) * BM;
    const int64_t bn = (int64_t)(sw / gx) * BN;

    f32x4 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // staging: A/B tiles are 2048 floats; 512 threads x 1 float4 each
    const int a_c = tid >> 5, a_r4 = (tid & 31) * 4;
    const int b_n = tid >> 2, b_k4 = (tid & 3) * 4;

    f32x4 pa, pb;
    const int64_t ktiles = k / BK;
    pa = *reinterpret_cast<const f32x4*>(A + (int64_t)a_c * lda + bm + a_r4);
    pb = *reinterpret_cast<const f32x4*>(B + (bn + b_n) * ldb + b_k4);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        As[(a_r4 + r) * LSTR + a_c] = pa[r];
        Bs[b_n * LSTR + b_k4 + r] = pb[r];
    }

    for (int64_t kt = 0; kt < ktiles; ++kt) {
        __syncthreads();
        if (kt + 1 < ktiles) {
            const int64_t k0 = (kt + 1) * BK;
            pa = *reinterpret_cast<const f32x4*>(
                A + (k0 + a_c) * lda + bm + a_r4);
            pb = *reinterpret_cast<const f32x4*>(
                B + (bn + b_n) * ldb + k0 + b_k4);
        }
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            float a[4], b[2];
            const int kof = kk * 4 + l4;
#pragma unroll
            for (int i = 0; i < 4; ++i)
                a[i] = As[(wr + i * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int j = 0; j < 2; ++j)
                b[j] = Bs[(wc + j * 16 + l16) * LSTR + kof];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
        if (kt + 1 < ktiles) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                As[(a_r4 + r) * LSTR + a_c] = pa[r];
                Bs[b_n * LSTR + b_k4 + r] = pb[r];
            }
        }
    }

    // f32 C/D map: row = 4*l4 + q (consecutive per lane -> one f32x4)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int64_t col = bn + wc + j * 16 + l16;
            float* cp = C + col * ldc + bm + wr + i * 16 + 4 * l4;
#pragma unroll
            for (int qq = 0; qq < 4; ++qq) {
                float v = alpha * acc[i][j][qq];
                cp[qq] = (beta == 0.f) ? v : v + beta * cp[qq];
            }
        }
    }
}

template <typename T>
__global__ void gemm_naive_t(const T* __restrict__ A,
                             const T* __restrict__ B, T* __restrict__ C,
                             int64_t m, int64_t n, int64_t k,
                             int64_t lda, int64_t ldb, int64_t ldc,
                             T alpha, T beta) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t j = (int64_t)blockIdx.y * blockDim.y + threadIdx.y;
    if (i >= m || j >= n) return;
    T s = (T)0;
    for (int64_t kk = 0; kk < k; ++kk)
        s += A[kk * lda + i] * B[j * ldb + kk];
    T v = alpha * s;
    C[j * ldc + i] = (beta == (T)0) ? v : v + beta * C[j * ldc + i];
}

template <typename T>
__global__ void scale_c_t(T* __restrict__ C, int64_t m, int64_t n,
                          int64_t ldc, T beta) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t j = (int64_t)blockIdx.y * blockDim.y + threadIdx.y;
    if (i >= m || j >= n) return;
    C[j * ldc + i] = (beta == (T)0) ? (T)0 : beta * C[j * ldc + i];
}

int launch_gemm_f32(void* Cv, const void* Av, const void* Bv,
                    int64_t m, int64_t n, int64_t k,
                    int64_t lda, int64_t ldb, int64_t ldc,
                    double alpha, double beta, hipStream_t s) {
    if (m < 0 || n < 0 || k < 0)
        return set_err(-3, "da_gemm_f32: bad shape");
    if (m == 0 || n == 0) return 0;
    const float* A = (const float*)Av;
    const float* B = (const float*)Bv;
    float* C = (float*)Cv;
    float al = (float)alpha, be = (float)beta;
    if (k == 0 || al == 0.f) {
        dim3 t(64, 4), g((m + 63) / 64, (n + 3) / 4);
        hipLaunchKernelGGL(scale_c_t<float>, g, t, 0, s, C, m, n, ldc, be);
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }
    if (m % BM == 0 && n % BN == 0 && k % BK == 0) {
        dim3 g(m / BM, n / BN);
        hipLaunchKernelGGL(gemm_f32_mfma_v3, g, dim3(512), 0, s,
                           A, B, C, m, n, k, lda, ldb, ldc, al, be);
    } else {
        dim3 t(64, 4), g((m + 63) / 64, (n + 3) / 4);
        hipLaunchKernelGGL(gemm_naive_t<float>, g, t, 0, s,
                           A, B, C, m, n, k, lda, ldb, ldc, al, be);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

int launch_gemm_i64(void* Cv, const void* Av, const void* Bv,
                    int64_t m, int64_t n, int64_t k,
                    int64_t lda, int64_t ldb, int64_t ldc,
                    int64_t alpha, int64_t beta, hipStream_t s) {
    if (m <= 0 || n <= 0) return 0;
    dim3 t(64, 4), g((m + 63) / 64, (n + 3) / 4);
    hipLaunchKernelGGL(gemm_naive_t<int64_t>, g, t, 0, s,
                       (const int64_t*)Av, (const int64_t*)Bv,
                       (int64_t*)Cv, m, n, k, lda, ldb, ldc, alpha, beta);
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

int launch_gemm_f64(void* Cv, const void* Av, const void* Bv,
                    int64_t m, int64_t n, int64_t k,
                    int64_t lda, int64_t ldb, int64_t ldc,
                    double alpha, double beta, hipStream_t s) {
    if (m < 0 || n < 0 || k < 0)
        return set_err(-3, "da_gemm_f64: bad shape");
    if (m == 0 || n == 0) return 0;
    const double* A = (const double*)Av;
    const double* B = (const double*)Bv;
    double* C = (double*)Cv;
    if (k == 0 || alpha == 0.0) {
        dim3 t(64, 4), g((m + 63) / 64, (n + 3) / 4);
        hipLaunchKernelGGL(scale_c, g, t, 0, s, C, m, n, ldc, beta);
        DA_CHECK_HIP(hipGetLastError());
        return 0;
    }
    if (m % BM == 0 && n % BN == 0 && k % BK == 0) {
        dim3 g(m / BM, n / BN);
        static int variant = -1, bk = -1;
        if (variant < 0) {
            const char* v = getenv("DA_GEMM_V");
            variant = v ? atoi(v) : 3;   // v3 measured fastest (profiles/)
            const char* b = getenv("DA_GEMM_BK");
            bk = b ? atoi(b) : 16;
        }
        if (variant == 1)
            hipLaunchKernelGGL(gemm_f64_mfma, g, dim3(256), 0, s,
                               A, B, C, m, n, k, lda, ldb, ldc, alpha, beta);
        else if (variant == 3)
            hipLaunchKernelGGL(gemm_f64_mfma_v3, g, dim3(512), 0, s,
                               A, B, C, m, n, k, lda, ldb, ldc, alpha, beta);
        else if (variant == 5)
            hipLaunchKernelGGL(gemm_f64_mfma_v5, g, dim3(512), 0, s,
                               A, B, C, m, n, k, lda, ldb, ldc, alpha, beta);
        else if (bk == 32 && k % 32 == 0)
            hipLaunchKernelGGL(gemm_f64_mfma_v2<32>, g, dim3(256), 0, s,
                               A, B, C, m, n, k, lda, ldb, ldc, alpha, beta);
        else
            hipLaunchKernelGGL(gemm_f64_mfma_v2<16>, g, dim3(256), 0, s,
                               A, B, C, m, n, k, lda, ldb, ldc, alpha, beta);
    } else {
        dim3 t(64, 4), g((m + 63) / 64, (n + 3) / 4);
        hipLaunchKernelGGL(gemm_f64_naive, g, t, 0, s,
                           A, B, C, m, n, k, lda, ldb, ldc, alpha, beta);
    }
    DA_CHECK_HIP(hipGetLastError());
    return 0;
}

} // namespace da
