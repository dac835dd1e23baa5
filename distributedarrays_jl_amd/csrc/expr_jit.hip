// expr_jit.hip — hipRTC-compiled fused broadcast expressions (gfx950).
//
// The interpreter in kernels_expr.hip is correct for any program but
// measured VALU/occupancy-bound (the 63-case functor switch per
// instruction costs ~2.4x vs the specialized bcast_fma kernel —
// profiles/r2 probe).  This path generates, per distinct program, a
// tiny HIP kernel whose SSA chain calls apply_map<T>(CONST_OP, x) /
// apply_map2<T>(CONST_OP, a, b): with the opcode a compile-time
// constant the switch folds to a single case (the same trick as
// map_fixed_kernel, which recovered 2x in round 1), so the generated
// kernel has exactly the register pressure of the expression — and
// numerics BIT-IDENTICAL to the interpreter, because the generated
// source #includes the very same mapops.hpp/fastmath.hpp (embedded
// into the .so at build time, embedded_src.hpp).
//
// Compiled kernels are cached per (program, dtype, layout) for the
// process lifetime; constants, pointers, shapes and strides are kernel
// ARGUMENTS, so re-invoking the same expression shape never recompiles.
// Any hipRTC failure falls back to the interpreter (still a native GPU
// path) and is visible via da_expr_jit_state().  DA_EXPR_JIT=0
// disables the JIT (checked per call — tests toggle it live).
#include "common.hpp"
#include "embedded_src.hpp"
#include <hip/hiprtc.h>
#include <stdlib.h>
#include <string>
#include <unordered_map>
#include <vector>

namespace da {

// mirrored EXACTLY in the generated source (JARGS_DECL below)
static_assert(DA_EXPR_MAXARGS == 6 && DA_EXPR_MAXCONSTS == 6 &&
              DA_EXPR_MAXND == 4,
              "JARGS_DECL literal must be kept in sync");
struct JArgs {
    void* dst;
    const void* srcs[DA_EXPR_MAXARGS];
    double consts[DA_EXPR_MAXCONSTS];
    unsigned long long n;
    unsigned int dims[DA_EXPR_MAXND];
    unsigned int str[DA_EXPR_MAXARGS][DA_EXPR_MAXND];
};

namespace {

struct JitEntry {
    hipModule_t mod = nullptr;
    hipFunction_t fn = nullptr;
};

std::mutex g_jit_mtx;
std::unordered_map<std::string, JitEntry> g_jit_cache;
int g_jit_state = 1;   // 1 ready, 2 active, -1 failed (interpreter)
char g_jit_err[512] = "";

const char* JARGS_DECL =
    "struct JArgs {\n"
    "  void* dst;\n"
    "  const void* srcs[6];\n"
    "  double consts[6];\n"
    "  unsigned long long n;\n"
    "  unsigned int dims[4];\n"
    "  unsigned int str[6][4];\n"
    "};\n";

std::string gen_eval(const int32_t* prog, int plen, const char* tname,
                     bool i64, bool strided) {
    // SSA chain over a virtual stack; reg names v0..; stack holds names
    std::string s;
    s += "static __device__ __forceinline__ ";
    s += tname;
    s += strided ? " evalx(const JArgs& a, const unsigned long long* off) {\n"
                 : " evalx(const JArgs& a, unsigned long long i) {\n";
    std::vector<std::string> stk;
    int reg = 0;
    char buf[256];
    for (int pc = 0; pc < plen; ++pc) {
        int kind = prog[pc] >> 8, idx = prog[pc] & 0xff;
        if (kind == 1) {
            snprintf(buf, sizeof(buf),
                     strided
                         ? "  %s v%d = ((const %s*)a.srcs[%d])[off[%d]];\n"
                         : "  %s v%d = ((const %s*)a.srcs[%d])[i];\n",
                     tname, reg, tname, idx, idx);
            s += buf;
            stk.push_back("v" + std::to_string(reg++));
        } else if (kind == 2) {
            snprintf(buf, sizeof(buf), "  %s v%d = (%s)a.consts[%d];\n",
                     tname, reg, tname, idx);
            s += buf;
            stk.push_back("v" + std::to_string(reg++));
        } else if (kind == 0) {
            snprintf(buf, sizeof(buf),
                     "  %s v%d = da::apply_map%s(%d, %s);\n", tname, reg,
                     i64 ? "_i64" : (std::string("<") + tname + ">").c_str(),
                     idx, stk.back().c_str());
            s += buf;
            stk.back() = "v" + std::to_string(reg++);
        } else {
            std::string b = stk.back(); stk.pop_back();
            std::string a2 = stk.back(); stk.pop_back();
            snprintf(buf, sizeof(buf),
                     "  %s v%d = da::apply_map2%s(%d, %s, %s);\n", tname,
                     reg,
                     i64 ? "_i64" : (std::string("<") + tname + ">").c_str(),
                     idx, a2.c_str(), b.c_str());
            s += buf;
            stk.push_back("v" + std::to_string(reg++));
        }
    }
    s += "  return " + stk.back() + ";\n}\n";
    return s;
}

// transcendental-free programs tolerate a 4-elem unroll (measured
// 5.3 TB/s vs 4.7 at 2-elem); polynomial/OCML ops (sin, exp, ...)
// inflate registers 4x and regress — those programs stay at 2.
static bool prog_heavy(const int32_t* prog, int plen) {
    for (int pc = 0; pc < plen; ++pc) {
        int kind = prog[pc] >> 8, idx = prog[pc] & 0xff;
        if (kind == 0) {
            switch (idx) {
            case DA_OP_IDENTITY: case DA_OP_NEG: case DA_OP_ABS:
            case DA_OP_ABS2: case DA_OP_INV: case DA_OP_SQRT:
            case DA_OP_FLOOR: case DA_OP_CEIL: case DA_OP_ROUND:
            case DA_OP_TRUNC: case DA_OP_SIGN: case DA_OP_DEG2RAD:
            case DA_OP_RAD2DEG: case DA_OP_ISNAN: case DA_OP_ISINF:
            case DA_OP_ISFINITE:
                break;
            default:
                return true;
            }
        } else if (kind == 3) {
            if (idx == DA_OP2_POW || idx == DA_OP2_ATAN2 ||
                idx == DA_OP2_MOD || idx == DA_OP2_REM)
                return true;
        }
    }
    return false;
}

std::string gen_source(const int32_t* prog, int plen, int dtype, int nd,
                       int nsrcs, bool strided, int sunroll) {
    const char* tname = dtype == DA_F64 ? "double"
                        : dtype == DA_F32 ? "float" : "long long";
    bool i64 = dtype == DA_I64;
    std::string s =
        "#ifndef M_PI\n#define M_PI 3.14159265358979323846\n#endif\n"
        "#include \"mapops.hpp\"\n";
    s += JARGS_DECL;
    s += gen_eval(prog, plen, tname, i64, strided);
    char buf[4096];
    if (!strided) {
        // explicit temporaries THEN stores: dst may alias a source
        // (in-place broadcast), so an eval/store loop would serialize
        // the loads behind the stores — measured 1.21 ms (this form)
        // vs 2.13 ms (store-per-eval loop) on the fma chain
        int u = prog_heavy(prog, plen) ? 2 : 4;
        snprintf(buf, sizeof(buf),
            "extern \"C\" __global__ void ejit(JArgs a) {\n"
            "  unsigned long long i = (unsigned long long)blockIdx.x * "
            "blockDim.x + threadIdx.x;\n"
            "  unsigned long long st = (unsigned long long)gridDim.x * "
            "blockDim.x;\n"
            "  %s* dst = (%s*)a.dst;\n"
            "  unsigned long long nv = a.n / %d;\n"
            "  for (unsigned long long jp = i; jp < nv; jp += st) {\n"
            "    unsigned long long j = %d * jp;\n",
            tname, tname, u, u);
        s += buf;
        for (int q = 0; q < u; ++q) {
            snprintf(buf, sizeof(buf),
                     "    %s r%d = evalx(a, j + %d);\n", tname, q, q);
            s += buf;
        }
        for (int q = 0; q < u; ++q) {
            snprintf(buf, sizeof(buf), "    dst[j + %d] = r%d;\n", q, q);
            s += buf;
        }
        snprintf(buf, sizeof(buf),
            "  }\n"
            "  for (unsigned long long j = %d * nv + i; j < a.n; j += st)\n"
            "    dst[j] = evalx(a, j);\n"
            "}\n", u);
        s += buf;
    } else if (sunroll == 4) {
        // dim-0 extent is a multiple of 4 (checked by the launcher), so
        // every aligned group of 4 flat indices stays within one dim-0
        // run: decode once, step each operand by its dim-0 stride
        int K = nsrcs > 0 ? nsrcs : 1;
        snprintf(buf, sizeof(buf),
            "/*s4*/extern \"C\" __global__ void ejit(JArgs a) {\n"
            "  unsigned long long i0 = (unsigned long long)blockIdx.x * "
            "blockDim.x + threadIdx.x;\n"
            "  unsigned long long gs = (unsigned long long)gridDim.x * "
            "blockDim.x;\n"
            "  %s* dst = (%s*)a.dst;\n"
            "  unsigned long long ng = a.n / 4;\n"
            "  for (unsigned long long g = i0; g < ng; g += gs) {\n"
            "    unsigned long long j = 4 * g;\n"
            "    unsigned int idx[%d]; unsigned int rem = (unsigned int)j;\n"
            "    for (int d = 0; d < %d; ++d) { idx[d] = rem %% a.dims[d]; "
            "rem /= a.dims[d]; }\n"
            "    unsigned long long off[%d], o1[%d], o2[%d], o3[%d];\n"
            "    for (int k = 0; k < %d; ++k) {\n"
            "      off[k] = 0;\n"
            "      for (int d = 0; d < %d; ++d) off[k] += "
            "(unsigned long long)idx[d] * a.str[k][d];\n"
            "      o1[k] = off[k] + a.str[k][0];\n"
            "      o2[k] = o1[k] + a.str[k][0];\n"
            "      o3[k] = o2[k] + a.str[k][0];\n"
            "    }\n"
            "    %s r0 = evalx(a, off);\n"
            "    %s r1 = evalx(a, o1);\n"
            "    %s r2 = evalx(a, o2);\n"
            "    %s r3 = evalx(a, o3);\n"
            "    dst[j] = r0; dst[j + 1] = r1;\n"
            "    dst[j + 2] = r2; dst[j + 3] = r3;\n"
            "  }\n"
            "}\n", tname, tname, nd, nd, K, K, K, K, K, nd,
            tname, tname, tname, tname);
        s += buf;
    } else {
        snprintf(buf, sizeof(buf),
            "extern \"C\" __global__ void ejit(JArgs a) {\n"
            "  unsigned long long i0 = (unsigned long long)blockIdx.x * "
            "blockDim.x + threadIdx.x;\n"
            "  unsigned long long gs = (unsigned long long)gridDim.x * "
            "blockDim.x;\n"
            "  %s* dst = (%s*)a.dst;\n"
            "  for (unsigned long long i = i0; i < a.n; i += gs) {\n"
            "    unsigned int idx[%d]; unsigned int rem = (unsigned int)i;\n"
            "    for (int d = 0; d < %d; ++d) { idx[d] = rem %% a.dims[d]; "
            "rem /= a.dims[d]; }\n"
            "    unsigned long long off[%d];\n"
            "    for (int k = 0; k < %d; ++k) { off[k] = 0;\n"
            "      for (int d = 0; d < %d; ++d) off[k] += "
            "(unsigned long long)idx[d] * a.str[k][d]; }\n"
            "    dst[i] = evalx(a, off);\n"
            "  }\n"
            "}\n", tname, tname, nd, nd, nsrcs > 0 ? nsrcs : 1,
            nsrcs > 0 ? nsrcs : 1, nd);
        s += buf;
    }
    return s;
}

int jit_get(const std::string& src, hipFunction_t* out) {
    std::lock_guard<std::mutex> g(g_jit_mtx);
    auto it = g_jit_cache.find(src);
    if (it != g_jit_cache.end()) {
        *out = it->second.fn;
        return 0;
    }
    if (g_jit_cache.size() >= 256) {
        // bound module memory for workloads generating unbounded
        // distinct programs; recompilation repopulates on demand.
        // Drain the device first — a cached kernel may be in flight.
        (void)hipDeviceSynchronize();
        for (auto& kv : g_jit_cache) hipModuleUnload(kv.second.mod);
        g_jit_cache.clear();
    }
    const char* hdr_names[] = {"stdint.h", "darray_hip.h", "mapops.hpp",
                               "fastmath.hpp"};
    const char* stdint_stub =
        "#pragma once\n"
        "typedef __INT8_TYPE__ int8_t;\ntypedef __UINT8_TYPE__ uint8_t;\n"
        "typedef __INT16_TYPE__ int16_t;\ntypedef __UINT16_TYPE__ uint16_t;\n"
        "typedef __INT32_TYPE__ int32_t;\ntypedef __UINT32_TYPE__ uint32_t;\n"
        "typedef __INT64_TYPE__ int64_t;\ntypedef __UINT64_TYPE__ uint64_t;\n";
    const char* hdrs[] = {stdint_stub, embedded_darray_hip_h,
                          embedded_mapops_hpp, embedded_fastmath_hpp};
    hiprtcProgram prog;
    hiprtcResult rc = hiprtcCreateProgram(&prog, src.c_str(), "ejit.cu",
                                          4, hdrs, hdr_names);
    if (rc != HIPRTC_SUCCESS) {
        snprintf(g_jit_err, sizeof(g_jit_err), "hiprtcCreateProgram: %s",
                 hiprtcGetErrorString(rc));
        return -1;
    }
    const char* opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17",
                          "-ffp-contract=off"};
    rc = hiprtcCompileProgram(prog, 4, opts);
    if (rc != HIPRTC_SUCCESS) {
        size_t lsz = 0;
        hiprtcGetProgramLogSize(prog, &lsz);
        std::string log(lsz, 0);
        if (lsz) hiprtcGetProgramLog(prog, &log[0]);
        snprintf(g_jit_err, sizeof(g_jit_err), "hiprtc compile: %.400s",
                 log.c_str());
        hiprtcDestroyProgram(&prog);
        return -1;
    }
    size_t csz = 0;
    hiprtcGetCodeSize(prog, &csz);
    std::vector<char> code(csz);
    hiprtcGetCode(prog, code.data());
    hiprtcDestroyProgram(&prog);
    JitEntry e;
    hipError_t he = hipModuleLoadData(&e.mod, code.data());
    if (he != hipSuccess) {
        snprintf(g_jit_err, sizeof(g_jit_err), "hipModuleLoadData: %s",
                 hipGetErrorString(he));
        return -1;
    }
    he = hipModuleGetFunction(&e.fn, e.mod, "ejit");
    if (he != hipSuccess) {
        snprintf(g_jit_err, sizeof(g_jit_err), "hipModuleGetFunction: %s",
                 hipGetErrorString(he));
        hipModuleUnload(e.mod);
        return -1;
    }
    g_jit_cache[src] = e;
    *out = e.fn;
    return 0;
}

}  // namespace

// Returns 0 on success, 1 when the JIT is unavailable/disabled (caller
// should use the interpreter), <0 on launch error.
int launch_expr_jit(const int32_t* prog, int plen, void* dst,
                    const uint64_t* dst_dims, int nd,
                    void* const* srcs, const uint64_t* src_strides,
                    int nsrcs, const double* consts, int nconsts,
                    uint64_t n, int dtype, hipStream_t s) {
    const char* e = getenv("DA_EXPR_JIT");
    if (e && e[0] == '0') return 1;
    if (g_jit_state < 0) return 1;   // earlier hard failure: interpreter
    bool strided = src_strides != nullptr;
    // strided 4-wide measured a ~7% REGRESSION on the de-mean form
    // (0.99 vs 0.93 ms same-box A/B: the three extra per-operand offset
    // arrays outweigh the decode savings) — default off, kept for
    // re-evaluation via DA_EJIT_S4=1
    int sunroll = 1;
    const char* s4 = getenv("DA_EJIT_S4");
    if (s4 && s4[0] == '1' && strided && !prog_heavy(prog, plen) &&
        dst_dims && dst_dims[0] % 4 == 0 && n % 4 == 0)
        sunroll = 4;
    std::string src = gen_source(prog, plen, dtype, nd, nsrcs, strided,
                                 sunroll);
    hipFunction_t fn;
    if (jit_get(src, &fn) != 0) {
        g_jit_state = -1;            // remember; interpreter from now on
        return 1;
    }
    g_jit_state = 2;

    JArgs a;
    memset(&a, 0, sizeof(a));
    a.dst = dst;
    for (int i = 0; i < nsrcs; ++i) a.srcs[i] = srcs[i];
    for (int i = 0; i < nconsts; ++i) a.consts[i] = consts[i];
    a.n = n;
    if (strided) {
        for (int d = 0; d < nd; ++d) a.dims[d] = (unsigned)dst_dims[d];
        for (int k = 0; k < nsrcs; ++k)
            for (int d = 0; d < nd; ++d)
                a.str[k][d] = (unsigned)src_strides[(size_t)k * nd + d];
    }
    size_t asz = sizeof(a);
    void* cfg[] = {HIP_LAUNCH_PARAM_BUFFER_POINTER, &a,
                   HIP_LAUNCH_PARAM_BUFFER_SIZE, &asz,
                   HIP_LAUNCH_PARAM_END};
    uint64_t work = strided ? (sunroll == 4 ? n / 4 + 1 : n)
                            : n / 2 + 1;
    uint64_t b = (work + 255) / 256;
    if (b > 8192) b = 8192;   // 1024 workgroups/XCD fills the chip
    if (b == 0) b = 1;
    DA_CHECK_HIP(hipModuleLaunchKernel(fn, (unsigned)b, 1, 1, 256, 1, 1,
                                       0, s, nullptr, cfg));
    return 0;
}

int expr_jit_state() { return g_jit_state; }
const char* expr_jit_err() { return g_jit_err; }

} // namespace da

/* test-only: generate + hiprtc-compile a program WITHOUT loading it
 * (no GPU, no da_init needed — hipRTC is a pure compiler).  Lets the
 * CPU suite pin the codegen.  Returns 0 ok, -1 compile failure (see
 * da_expr_jit_errstr); src_out receives the generated source. */
extern "C" int dbg_expr_jit_compile(const int32_t* prog, int plen,
                                    int dtype, int nd, int nsrcs,
                                    int strided, char* src_out,
                                    int src_len) {
    using namespace da;
    std::string src = gen_source(prog, plen, dtype, nd, nsrcs,
                                 strided != 0, 1);
    if (src_out && src_len > 0) {
        strncpy(src_out, src.c_str(), src_len - 1);
        src_out[src_len - 1] = 0;
    }
    const char* hdr_names[] = {"stdint.h", "darray_hip.h", "mapops.hpp",
                               "fastmath.hpp"};
    const char* stdint_stub =
        "#pragma once\n"
        "typedef __INT8_TYPE__ int8_t;\ntypedef __UINT8_TYPE__ uint8_t;\n"
        "typedef __INT16_TYPE__ int16_t;\ntypedef __UINT16_TYPE__ uint16_t;\n"
        "typedef __INT32_TYPE__ int32_t;\ntypedef __UINT32_TYPE__ uint32_t;\n"
        "typedef __INT64_TYPE__ int64_t;\ntypedef __UINT64_TYPE__ uint64_t;\n";
    const char* hdrs[] = {stdint_stub, embedded_darray_hip_h,
                          embedded_mapops_hpp, embedded_fastmath_hpp};
    hiprtcProgram p;
    if (hiprtcCreateProgram(&p, src.c_str(), "ejit.cu", 4, hdrs,
                            hdr_names) != HIPRTC_SUCCESS)
        return -1;
    const char* opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17",
                          "-ffp-contract=off"};
    hiprtcResult rc = hiprtcCompileProgram(p, 4, opts);
    if (rc != HIPRTC_SUCCESS) {
        size_t lsz = 0;
        hiprtcGetProgramLogSize(p, &lsz);
        std::string log(lsz, 0);
        if (lsz) hiprtcGetProgramLog(p, &log[0]);
        snprintf(g_jit_err, sizeof(g_jit_err), "hiprtc compile: %.400s",
                 log.c_str());
    }
    hiprtcDestroyProgram(&p);
    return rc == HIPRTC_SUCCESS ? 0 : -1;
}
