// mapops.hpp — the unary/binary scalar functor tables shared by the
// map/map2 kernels (kernels_elementwise.hip) and the fused-broadcast
// expression interpreter (kernels_expr.hip).  ONE source of truth for
// numerics: da_expr of a single-op program is bit-identical to da_map.
//
// Semantics follow the reference's scalar-math list
// (/root/reference/test/darray.jl:775-800) with Julia conventions
// (sign/round/sinc/cosc/mod noted inline); compiled -ffp-contract=off.
#pragma once
#include "darray_hip.h"
#include "fastmath.hpp"

namespace da {

// -------------------------------------------------------------------- map
// Unary functors.  Julia-compatible semantics noted where non-obvious.
template <typename T> struct MathF;
template <> struct MathF<double> {
    static __device__ double sin_(double x) { return sin(x); }
    static __device__ double cos_(double x) { return cos(x); }
    static __device__ double tan_(double x) { return tan(x); }
    static __device__ double exp_(double x) { return exp(x); }
    static __device__ double log_(double x) { return log(x); }
    static __device__ double sqrt_(double x) { return sqrt(x); }
};

template <typename T>
__device__ __forceinline__ T apply_map(int op, T x) {
    const T one = (T)1, zero = (T)0;
    (void)zero;
    switch (op) {
    case DA_OP_IDENTITY: return x;
    case DA_OP_NEG: return -x;
    case DA_OP_ABS: return fabs(x);   // fabs: abs(-0.0) = +0.0 (Julia/numpy)
    case DA_OP_ABS2: return x * x;
    case DA_OP_INV: return one / x;
    case DA_OP_SQRT: return sqrt(x);
    case DA_OP_CBRT: return cbrt(x);
    case DA_OP_EXP: return da_exp(x);
    case DA_OP_EXP2: return exp2(x);
    case DA_OP_EXP10: return pow((T)10, x);
    case DA_OP_EXPM1: return expm1(x);
    case DA_OP_LOG: return log(x);
    case DA_OP_LOG2: return log2(x);
    case DA_OP_LOG10: return log10(x);
    case DA_OP_LOG1P: return log1p(x);
    case DA_OP_SIN: return da_sin(x);
    case DA_OP_COS: return da_cos(x);
    case DA_OP_TAN: return tan(x);
    case DA_OP_ASIN: return asin(x);
    case DA_OP_ACOS: return acos(x);
    case DA_OP_ATAN: return atan(x);
    case DA_OP_SINH: return sinh(x);
    case DA_OP_COSH: return cosh(x);
    case DA_OP_TANH: return tanh(x);
    case DA_OP_ASINH: return asinh(x);
    case DA_OP_ACOSH: return acosh(x);
    case DA_OP_ATANH: return atanh(x);
    case DA_OP_SINPI: return sinpi(x);
    case DA_OP_COSPI: return cospi(x);
    case DA_OP_FLOOR: return floor(x);
    case DA_OP_CEIL: return ceil(x);
    case DA_OP_ROUND: return rint(x);   // Julia round = half-even
    case DA_OP_TRUNC: return trunc(x);
    case DA_OP_SIGN:  // Julia sign: preserves NaN and signed zero
        return x != x ? x : (x > zero ? one : (x < zero ? -one : x));
    case DA_OP_DEG2RAD: return x * (T)(M_PI / 180.0);
    case DA_OP_RAD2DEG: return x * (T)(180.0 / M_PI);
    case DA_OP_SEC: return one / da_cos(x);
    case DA_OP_CSC: return one / da_sin(x);
    case DA_OP_COT: return one / tan(x);
    case DA_OP_ERF: return erf(x);
    case DA_OP_ERFC: return erfc(x);
    case DA_OP_ERFINV: return erfinv(x);
    case DA_OP_ERFCINV: return erfcinv(x);
    case DA_OP_ERFCX: return erfcx(x);
    case DA_OP_GAMMA: return tgamma(x);
    case DA_OP_LGAMMA: return lgamma(x);
    case DA_OP_SINC:   // Julia sinc: sin(pi x)/(pi x), 1 at 0
        return x == zero ? one : sinpi(x) / ((T)M_PI * x);
    case DA_OP_COSC:   // Julia cosc: d/dx sinc = cospi(x)/x - sinpi(x)/(pi x^2)
        return x == zero ? zero
                         : cospi(x) / x - sinpi(x) / ((T)M_PI * x * x);
    case DA_OP_SIND: return da_sin(x * (T)(M_PI / 180.0));
    case DA_OP_COSD: return da_cos(x * (T)(M_PI / 180.0));
    case DA_OP_TAND: return tan(x * (T)(M_PI / 180.0));
    case DA_OP_ASIND: return asin(x) * (T)(180.0 / M_PI);
    case DA_OP_ACOSD: return acos(x) * (T)(180.0 / M_PI);
    case DA_OP_ATAND: return atan(x) * (T)(180.0 / M_PI);
    case DA_OP_ACOT: return atan(one / x);
    case DA_OP_ACOTD: return atan(one / x) * (T)(180.0 / M_PI);
    case DA_OP_ASEC: return acos(one / x);
    case DA_OP_ACSC: return asin(one / x);
    case DA_OP_ASECH: return acosh(one / x);
    case DA_OP_ACSCH: return asinh(one / x);
    case DA_OP_ACOTH: return atanh(one / x);
    case DA_OP_ISNAN: return (T)(x != x ? 1 : 0);
    case DA_OP_ISINF: return (T)(isinf((double)x) ? 1 : 0);
    case DA_OP_ISFINITE: return (T)(isfinite((double)x) ? 1 : 0);
    }
    return x;
}

__device__ __forceinline__ int64_t apply_map_i64(int op, int64_t x) {
    switch (op) {
    case DA_OP_IDENTITY: return x;
    case DA_OP_NEG: return (int64_t)(0ull - (uint64_t)x);
    case DA_OP_ABS: return x < 0 ? (int64_t)(0ull - (uint64_t)x) : x;
    case DA_OP_ABS2: return (int64_t)((uint64_t)x * (uint64_t)x);
    case DA_OP_SIGN: return x > 0 ? 1 : (x < 0 ? -1 : 0);
    }
    return x;
}


template <typename T>
__device__ __forceinline__ T apply_map2(int op, T a, T b) {
    switch (op) {
    case DA_OP2_ADD: return a + b;
    case DA_OP2_SUB: return a - b;
    case DA_OP2_MUL: return a * b;
    case DA_OP2_DIV: return a / b;
    case DA_OP2_MIN2:  // NaN-propagating (Julia min)
        return a != a ? a : (b != b ? b : (a < b ? a : b));
    case DA_OP2_MAX2:
        return a != a ? a : (b != b ? b : (a > b ? a : b));
    case DA_OP2_REM: return fmod(a, b);
    case DA_OP2_MOD: {   // floored (Julia mod / numpy mod)
        T r = fmod(a, b);
        if (r != (T)0 && ((r < (T)0) != (b < (T)0))) r += b;
        return r;
    }
    case DA_OP2_POW: return pow(a, b);
    case DA_OP2_ATAN2: return atan2(a, b);
    }
    return a;
}

__device__ __forceinline__ int64_t apply_map2_i64(int op, int64_t a, int64_t b) {
    switch (op) {
    case DA_OP2_ADD: return (int64_t)((uint64_t)a + (uint64_t)b);
    case DA_OP2_SUB: return (int64_t)((uint64_t)a - (uint64_t)b);
    case DA_OP2_MUL: return (int64_t)((uint64_t)a * (uint64_t)b);
    case DA_OP2_IDIV: return a / b;        // truncated (Julia div)
    case DA_OP2_REM: return a % b;         // truncated (Julia rem)
    case DA_OP2_MOD: {                     // floored (Julia mod)
        int64_t r = a % b;
        if (r != 0 && ((r < 0) != (b < 0))) r += b;
        return r;
    }
    case DA_OP2_AND: return a & b;
    case DA_OP2_OR: return a | b;
    case DA_OP2_XOR: return a ^ b;
    case DA_OP2_MIN2: return a < b ? a : b;
    case DA_OP2_MAX2: return a > b ? a : b;
    }
    return a;
}

} // namespace da
