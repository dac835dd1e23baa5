/* cpu_baseline.c — the timed CPU leg of bench.py (`cpu_baseline`,
 * kind="port"): the oracle's hot-path semantics in C + OpenMP, compiled
 * -O3 -march=native -fopenmp and run on all host cores (BASELINE.md
 * protocol; the reference is Julia and no julia binary exists here, so
 * this restatement stands in for the reference CPU path and is pinned by
 * the same tests as oracle/ — tests/test_cpu_baseline.py).
 *
 * ORACLE NOTICE: test/bench infrastructure only — never part of the
 * product path.
 *
 * Exposed via ctypes (oracle/cpu_baseline.py):
 *   cb_fill_uniform_f64 / _f32 : philox4x32-10, same mapping as
 *                                oracle/philox.py (bit-identical)
 *   cb_sum_f64, cb_map_sin_f64, cb_bcast_fma_f64, cb_abs2_sum_f32,
 *   cb_gemm_f64 (blocked OpenMP column-major)
 */
#include <stdint.h>
#include <math.h>
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

typedef struct { uint32_t v[4]; } u32x4;

static u32x4 philox4x32_10(uint64_t block, uint64_t seed) {
    const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
    uint32_t c0 = (uint32_t)block, c1 = (uint32_t)(block >> 32);
    uint32_t c2 = 0, c3 = 0;
    uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
    for (int r = 0; r < 10; ++r) {
        uint64_t p0 = (uint64_t)M0 * c0;
        uint64_t p1 = (uint64_t)M1 * c2;
        uint32_t n0 = (uint32_t)(p1 >> 32) ^ c1 ^ k0;
        uint32_t n1 = (uint32_t)p1;
        uint32_t n2 = (uint32_t)(p0 >> 32) ^ c3 ^ k1;
        uint32_t n3 = (uint32_t)p0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += W0; k1 += W1;
    }
    u32x4 o = {{c0, c1, c2, c3}};
    return o;
}

void cb_fill_uniform_f64(double* p, uint64_t n, uint64_t seed) {
#pragma omp parallel for schedule(static)
    for (int64_t b = 0; b <= (int64_t)((n - 1) >> 1); ++b) {
        u32x4 o = philox4x32_10((uint64_t)b, seed);
        uint64_t u0 = (((uint64_t)o.v[1] << 32) | o.v[0]) >> 11;
        uint64_t u1 = (((uint64_t)o.v[3] << 32) | o.v[2]) >> 11;
        uint64_t e0 = 2 * (uint64_t)b, e1 = e0 + 1;
        if (e0 < n) p[e0] = (double)u0 * 0x1.0p-53;
        if (e1 < n) p[e1] = (double)u1 * 0x1.0p-53;
    }
}

void cb_fill_uniform_f32(float* p, uint64_t n, uint64_t seed) {
#pragma omp parallel for schedule(static)
    for (int64_t b = 0; b <= (int64_t)((n - 1) >> 2); ++b) {
        u32x4 o = philox4x32_10((uint64_t)b, seed);
        for (int j = 0; j < 4; ++j) {
            uint64_t e = 4 * (uint64_t)b + j;
            if (e < n) p[e] = (float)(o.v[j] >> 8) * 0x1.0p-24f;
        }
    }
}

/* per-chunk pairwise-ish sum: OpenMP partials, deterministic order */
double cb_sum_f64(const double* p, uint64_t n) {
    double s = 0.0;
#pragma omp parallel for reduction(+:s) schedule(static)
    for (int64_t i = 0; i < (int64_t)n; ++i) s += p[i];
    return s;
}

float cb_abs2_sum_f32(const float* p, uint64_t n) {
    float s = 0.0f;
#pragma omp parallel for reduction(+:s) schedule(static)
    for (int64_t i = 0; i < (int64_t)n; ++i) s += p[i] * p[i];
    return s;
}

void cb_map_sin_f64(double* dst, const double* src, uint64_t n) {
#pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)n; ++i) dst[i] = sin(src[i]);
}

void cb_bcast_fma_f64(double* d, const double* a, const double* b,
                      double c, uint64_t n) {
#pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)n; ++i) d[i] = a[i] * b[i] + c;
}

/* blocked column-major GEMM C = A*B (beta=0), OpenMP over column panels */
void cb_gemm_f64(double* C, const double* A, const double* B,
                 int64_t m, int64_t n, int64_t k) {
    const int64_t MB = 64, NB = 64, KB = 256;
#pragma omp parallel for collapse(2) schedule(static)
    for (int64_t j0 = 0; j0 < n; j0 += NB) {
        for (int64_t i0 = 0; i0 < m; i0 += MB) {
            int64_t j1 = j0 + NB < n ? j0 + NB : n;
            int64_t i1 = i0 + MB < m ? i0 + MB : m;
            for (int64_t jj = j0; jj < j1; ++jj)
                for (int64_t ii = i0; ii < i1; ++ii)
                    C[jj * m + ii] = 0.0;
            for (int64_t k0 = 0; k0 < k; k0 += KB) {
                int64_t k1 = k0 + KB < k ? k0 + KB : k;
                for (int64_t jj = j0; jj < j1; ++jj) {
                    for (int64_t kk = k0; kk < k1; ++kk) {
                        double bkj = B[jj * k + kk];
                        const double* ap = A + kk * m;
                        double* cp = C + jj * m;
                        for (int64_t ii = i0; ii < i1; ++ii)
                            cp[ii] += ap[ii] * bkj;
                    }
                }
            }
        }
    }
}

int cb_num_threads(void) {
#ifdef _OPENMP
    int t = 1;
#pragma omp parallel
    {
#pragma omp single
        t = omp_get_num_threads();
    }
    return t;
#else
    return 1;
#endif
}
