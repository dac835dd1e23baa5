// fastmath.hpp — polynomial f64 sin/cos/exp for gfx950 hot maps.
//
// Round-1 profiling (profiles/r01_kernel_stats.md) showed map!(sin) on
// 2^28 f64 VALU-bound on OCML's generic sin (4.7 TB/s effective vs the
// 6.3 TB/s copy ceiling): OCML always runs its large-argument-capable
// path.  These routines restate the classic fdlibm algorithms
// (Sun Microsystems 1993, public; the basis of glibc/musl libm):
//   - sin/cos: 2-step Cody-Waite pi/2 reduction (exact for |n| < 2^20,
//     i.e. |x| < ~1.6e6) + the 13/14-degree minimax kernel polynomials
//     with the double-double tail carried through (k_sin/k_cos);
//   - exp: ln2 Cody-Waite reduction + the degree-5 rational kernel.
// Accuracy < 1 ulp on the fast range (parity-tested at 1e-13 rel vs
// numpy/libm, tests/test_gpu_parity.py).  Arguments outside the fast
// range (|x| >= 1e6 for sin/cos, |x| >= 708 for exp, NaN/Inf) fall
// back to OCML — correctness everywhere, speed where the data lives.
// Uniform-input workloads (drand chunks) take the fast path on every
// lane, so there is no divergence.
//
// Compiled -ffp-contract=off like the rest of the elementwise path;
// the algorithms are contraction-safe either way (fdlibm predates fma).
#pragma once

namespace da {
namespace fm {

// ---- fdlibm kernel sin on [-pi/4, pi/4]; y = tail of x, iy = y valid
__device__ __forceinline__ double k_sin(double x, double y, int iy) {
    const double S1 = -1.66666666666666324348e-01;
    const double S2 = 8.33333333332248946124e-03;
    const double S3 = -1.98412698298579493134e-04;
    const double S4 = 2.75573137070700676789e-06;
    const double S5 = -2.50507602534068634195e-08;
    const double S6 = 1.58969099521155010221e-10;
    double z = x * x;
    double w = z * z;
    double r = S2 + z * (S3 + z * S4) + z * w * (S5 + z * S6);
    double v = z * x;
    if (iy == 0) return x + v * (S1 + z * r);
    return x - ((z * (0.5 * y - v * r) - y) - v * S1);
}

// ---- fdlibm kernel cos on [-pi/4, pi/4]
__device__ __forceinline__ double k_cos(double x, double y) {
    const double C1 = 4.16666666666666019037e-02;
    const double C2 = -1.38888888888741095749e-03;
    const double C3 = 2.48015872894767294178e-05;
    const double C4 = -2.75573143513906633035e-07;
    const double C5 = 2.08757232129817482790e-09;
    const double C6 = -1.13596475577881948265e-11;
    double z = x * x;
    double r = z * (C1 + z * (C2 + z * (C3 + z * (C4 + z * (C5 + z * C6)))));
    double ax = fabs(x);
    // qx staves off cancellation in 1 - z/2 (0.25*ax is an exact op)
    double qx = (ax > 0.3) ? ((ax > 0.78125) ? 0.28125 : 0.25 * ax) : 0.0;
    double hz = 0.5 * z - qx;
    double a = 1.0 - qx;
    return a - (hz - (z * r - x * y));
}

// ---- 2-step Cody-Waite reduction x -> (y0, y1), quadrant n (mod 4).
// pio2_1 carries 33 bits, so fn*pio2_1 is exact for |fn| < 2^20;
// the second step leaves y0+y1 with ~86 good bits.
__device__ __forceinline__ int rem_pio2_medium(double x, double& y0,
                                               double& y1) {
    const double invpio2 = 6.36619772367581382433e-01;
    const double pio2_1 = 1.57079632673412561417e+00;
    const double pio2_1t = 6.07710050650619224932e-11;
    const double pio2_2 = 6.07710050630396597660e-11;
    const double pio2_2t = 2.02226624879595063154e-21;
    double fn = rint(x * invpio2);
    double r = x - fn * pio2_1;
    double w = fn * pio2_1t;
    double t = r;
    w = fn * pio2_2;
    r = t - w;
    w = fn * pio2_2t - ((t - r) - w);
    y0 = r - w;
    y1 = (r - y0) - w;
    return ((int)fn) & 3;
}

__device__ __forceinline__ double fast_sin(double x) {
    double ax = fabs(x);
    if (ax <= 0.7853981633974483) {        // |x| <= pi/4: no reduction
        if (ax < 7.450580596923828e-09) return x;   // 2^-27: sin(x)=x
        return k_sin(x, 0.0, 0);
    }
    if (!(ax < 1.0e6)) return ::sin(x);    // huge / NaN / Inf -> OCML
    double y0, y1;
    int n = rem_pio2_medium(x, y0, y1);
    switch (n) {
    case 0: return k_sin(y0, y1, 1);
    case 1: return k_cos(y0, y1);
    case 2: return -k_sin(y0, y1, 1);
    default: return -k_cos(y0, y1);
    }
}

__device__ __forceinline__ double fast_cos(double x) {
    double ax = fabs(x);
    if (ax <= 0.7853981633974483)
        return k_cos(x, 0.0);
    if (!(ax < 1.0e6)) return ::cos(x);
    double y0, y1;
    int n = rem_pio2_medium(x, y0, y1);
    switch (n) {
    case 0: return k_cos(y0, y1);
    case 1: return -k_sin(y0, y1, 1);
    case 2: return -k_cos(y0, y1);
    default: return k_sin(y0, y1, 1);
    }
}

__device__ __forceinline__ double fast_exp(double x) {
    const double invln2 = 1.44269504088896338700e+00;
    const double ln2HI = 6.93147180369123816490e-01;
    const double ln2LO = 1.90821492927058770002e-10;
    const double P1 = 1.66666666666666019037e-01;
    const double P2 = -2.77777777770155933842e-03;
    const double P3 = 6.61375632143793436117e-05;
    const double P4 = -1.65339022054652515390e-06;
    const double P5 = 4.13813679705723846039e-08;
    double ax = fabs(x);
    if (!(ax < 708.0)) return ::exp(x);    // over/underflow edges -> OCML
    if (ax < 3.725290298461914e-09) return 1.0 + x;   // 2^-28
    double fn = rint(x * invln2);
    int k = (int)fn;
    double hi = x - fn * ln2HI;
    double lo = fn * ln2LO;
    double xr = hi - lo;
    double t = xr * xr;
    double c = xr - t * (P1 + t * (P2 + t * (P3 + t * (P4 + t * P5))));
    double y = 1.0 - ((lo - (xr * c) / (2.0 - c)) - hi);
    return ldexp(y, k);
}

} // namespace fm

// dtype dispatch: f64 takes the polynomial path, f32 stays on OCML
// (its single-precision routines are already cheap relative to 8 B/elem
// of traffic)
__device__ __forceinline__ double da_sin(double x) { return fm::fast_sin(x); }
__device__ __forceinline__ float da_sin(float x) { return sinf(x); }
__device__ __forceinline__ double da_cos(double x) { return fm::fast_cos(x); }
__device__ __forceinline__ float da_cos(float x) { return cosf(x); }
__device__ __forceinline__ double da_exp(double x) { return fm::fast_exp(x); }
__device__ __forceinline__ float da_exp(float x) { return expf(x); }

} // namespace da
