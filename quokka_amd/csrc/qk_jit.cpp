// quokka_amd — hiprtc JIT for arbitrary filter predicates.
//
// The reference's partition_fn applies ARBITRARY predicates (polars
// expressions or DuckDB SQL, pyquokka/core.py:157-170) before
// partitioning. The static ABI covers single-column compares
// (qk_filter_*); this module runtime-compiles a fused count+scatter
// filter for any C predicate over up to QK_JIT_MAX_COLS typed columns
// (the Python side translates the reference's SQL grammar subset to the
// C expression — quokka_amd/jit.py). gfx950 code objects built with
// hiprtc; compilation needs no GPU (pure compiler), module load is lazy
// at first run.
#include <hip/hip_runtime.h>
#include <hip/hiprtc.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <string>
#include <ctype.h>
#include <vector>

#define QK_JIT_MAX_COLS 8

extern "C" const char *qk_last_error(void);
// reuse the main module's error slot via a local copy (thread-local there;
// keep jit errors separate but same contract)
static __thread char j_err[2048] = "";
extern "C" const char *qk_jit_last_error(void) { return j_err; }

static int j_fail(const char *where, const char *what) {
  snprintf(j_err, sizeof(j_err), "%s: %s", where, what);
  return 1;
}

struct QkJitProg {
  std::string code;        // gfx950 code object (hsaco)
  int ncols;
  int coltypes[QK_JIT_MAX_COLS];
  hipModule_t mod = nullptr;   // lazy-loaded on first run
  hipFunction_t f_count = nullptr, f_scatter = nullptr;
  uint64_t *scratch = nullptr; // block counts
};

static const char *type_name(int t) {
  switch (t) {
  case 0: return "int";
  case 1: return "double";
  case 2: return "unsigned char";
  case 3: return "long long";
  default: return nullptr;
  }
}

// Generated source: grid-stride chunked count + rank/scatter pair with the
// predicate inlined — same structure as the static k_filter_* kernels.
static std::string gen_source(const char *expr, int ncols,
                              const int *coltypes) {
  std::string s;
  s += "#define BLOCK 256\n#define WAVE 64\n";
  s += "typedef unsigned long long u64; typedef unsigned u32;\n";
  s += "__device__ inline void qk_atomic_mm(double* a, double v, int mx) {\n"
       "  unsigned long long* p = (unsigned long long*)a;\n"
       "  unsigned long long old = *p;\n"
       "  for (;;) {\n"
       "    double cur = __longlong_as_double(old);\n"
       "    if (mx ? (v <= cur) : (v >= cur)) return;\n"
       "    unsigned long long prev = atomicCAS(p, old,"
       " __double_as_longlong(v));\n"
       "    if (prev == old) return;\n"
       "    old = prev;\n  }\n}\n";
  s += "extern \"C\" __global__ __launch_bounds__(BLOCK) void jit_count(\n";
  s += "    u64 n, u64 chunk, u64* block_counts";
  for (int c = 0; c < ncols; c++) {
    s += ", const ";
    s += type_name(coltypes[c]);
    s += "* __restrict__ col";
    s += std::to_string(c);
  }
  s += ") {\n"
       "  u64 lo = (u64)blockIdx.x * chunk;\n"
       "  u64 hi = n < lo + chunk ? n : lo + chunk;\n"
       "  u32 cnt = 0;\n"
       "  for (u64 i = lo + threadIdx.x; i < hi; i += BLOCK) {\n";
  for (int c = 0; c < ncols; c++) {
    s += "    ";
    s += type_name(coltypes[c]);
    s += " v" + std::to_string(c) + " = col" + std::to_string(c) + "[i];\n";
  }
  s += "    if (";
  s += expr;
  s += ") cnt++;\n  }\n"
       "  __shared__ u32 lds[BLOCK / WAVE];\n"
       "  for (int off = WAVE / 2; off > 0; off >>= 1)\n"
       "    cnt += __shfl_down(cnt, off);\n"
       "  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;\n"
       "  if (lane == 0) lds[wid] = cnt;\n"
       "  __syncthreads();\n"
       "  if (threadIdx.x == 0) {\n"
       "    u64 t = 0;\n"
       "    for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];\n"
       "    block_counts[blockIdx.x] = t;\n  }\n}\n";

  s += "extern \"C\" __global__ void jit_scan(u64 nblocks, u64* counts,\n"
       "                                      u64* total) {\n"
       "  if (blockIdx.x == 0 && threadIdx.x == 0) {\n"
       "    u64 acc = 0;\n"
       "    for (u64 b = 0; b < nblocks; b++) { u64 c = counts[b];\n"
       "      counts[b] = acc; acc += c; }\n"
       "    *total = acc;\n  }\n}\n";

  s += "extern \"C\" __global__ __launch_bounds__(BLOCK) void jit_scatter(\n";
  s += "    u64 n, u64 chunk, const u64* block_offsets, u32* out_idx";
  for (int c = 0; c < ncols; c++) {
    s += ", const ";
    s += type_name(coltypes[c]);
    s += "* __restrict__ col";
    s += std::to_string(c);
  }
  s += ") {\n"
       "  u64 lo = (u64)blockIdx.x * chunk;\n"
       "  u64 hi = n < lo + chunk ? n : lo + chunk;\n"
       "  __shared__ u64 base;\n"
       "  __shared__ u32 wave_tot[BLOCK / WAVE];\n"
       "  if (threadIdx.x == 0) base = block_offsets[blockIdx.x];\n"
       "  __syncthreads();\n"
       "  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;\n"
       "  for (u64 r0 = lo; r0 < hi; r0 += BLOCK) {\n"
       "    u64 i = r0 + threadIdx.x;\n"
       "    bool pass = false;\n"
       "    if (i < hi) {\n";
  for (int c = 0; c < ncols; c++) {
    s += "      ";
    s += type_name(coltypes[c]);
    s += " v" + std::to_string(c) + " = col" + std::to_string(c) + "[i];\n";
  }
  s += "      pass = (";
  s += expr;
  s += ");\n    }\n"
       "    u64 mask = __ballot(pass);\n"
       "    u32 rank = __popcll(mask & ((1ULL << lane) - 1));\n"
       "    if (lane == 0) wave_tot[wid] = __popcll(mask);\n"
       "    __syncthreads();\n"
       "    u32 wbase = 0;\n"
       "    for (int w = 0; w < wid; w++) wbase += wave_tot[w];\n"
       "    if (pass) out_idx[base + wbase + rank] = (u32)i;\n"
       "    u32 btot = 0;\n"
       "    for (int w = 0; w < BLOCK / WAVE; w++) btot += wave_tot[w];\n"
       "    __syncthreads();\n"
       "    if (threadIdx.x == 0) base += btot;\n"
       "    __syncthreads();\n  }\n}\n";
  return s;
}

extern "C" int qk_jit_filter_build(const char *expr, int ncols,
                                   const int *coltypes, void **prog_out) {
  if (ncols < 1 || ncols > QK_JIT_MAX_COLS)
    return j_fail("qk_jit_filter_build", "ncols out of range");
  for (int c = 0; c < ncols; c++)
    if (!type_name(coltypes[c]))
      return j_fail("qk_jit_filter_build", "bad column type");
  std::string src = gen_source(expr, ncols, coltypes);

  hiprtcProgram prog;
  if (hiprtcCreateProgram(&prog, src.c_str(), "qk_jit_filter.cu", 0,
                          nullptr, nullptr) != HIPRTC_SUCCESS)
    return j_fail("qk_jit_filter_build", "hiprtcCreateProgram failed");
  const char *opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17"};
  hiprtcResult rc = hiprtcCompileProgram(prog, 3, opts);
  if (rc != HIPRTC_SUCCESS) {
    size_t lsz = 0;
    hiprtcGetProgramLogSize(prog, &lsz);
    std::string log(lsz, '\0');
    if (lsz) hiprtcGetProgramLog(prog, &log[0]);
    hiprtcDestroyProgram(&prog);
    snprintf(j_err, sizeof(j_err), "qk_jit_filter_build: compile failed: %s",
             log.c_str());
    return 2;
  }
  size_t csz = 0;
  hiprtcGetCodeSize(prog, &csz);
  QkJitProg *p = new QkJitProg();
  p->code.resize(csz);
  hiprtcGetCode(prog, &p->code[0]);
  hiprtcDestroyProgram(&prog);
  p->ncols = ncols;
  memcpy(p->coltypes, coltypes, ncols * sizeof(int));
  *prog_out = p;
  return 0;
}

static int jit_load(QkJitProg *p) {
  if (p->mod) return 0;
  if (hipModuleLoadData(&p->mod, p->code.data()) != hipSuccess)
    return j_fail("qk_jit_filter_run", "hipModuleLoadData failed");
  if (hipModuleGetFunction(&p->f_count, p->mod, "jit_count") != hipSuccess ||
      hipModuleGetFunction(&p->f_scatter, p->mod, "jit_scatter") != hipSuccess)
    return j_fail("qk_jit_filter_run", "hipModuleGetFunction failed");
  if (hipMalloc(&p->scratch, 2049 * sizeof(uint64_t)) != hipSuccess)
    return j_fail("qk_jit_filter_run", "scratch alloc failed");
  return 0;
}

extern "C" int qk_jit_filter_run(void *prog, void *stream, uint64_t n,
                                 const void *const *col_ptrs,
                                 uint32_t *out_idx, uint64_t *count_dev) {
  QkJitProg *p = (QkJitProg *)prog;
  int rc = jit_load(p);
  if (rc) return rc;
  if (!n) {
    if (hipMemsetAsync(count_dev, 0, 8, (hipStream_t)stream) != hipSuccess)
      return j_fail("qk_jit_filter_run", "memset failed");
    return 0;
  }
  uint64_t chunk = (n + 2047) / 2048;
  chunk = ((chunk + 255) / 256) * 256;
  uint32_t blocks = (uint32_t)((n + chunk - 1) / chunk);

  // hipModuleLaunchKernel takes a void*[] of pointers to each argument
  uint64_t a_n = n, a_chunk = chunk, a_nblocks = blocks;
  void *a_scratch = p->scratch;
  void *a_count = count_dev, *a_out = out_idx;
  std::vector<const void *> cols(col_ptrs, col_ptrs + p->ncols);

  std::vector<void *> args;
  args = {&a_n, &a_chunk, &a_scratch};
  for (int c = 0; c < p->ncols; c++) args.push_back((void *)&cols[c]);
  if (hipModuleLaunchKernel(p->f_count, blocks, 1, 1, 256, 1, 1, 0,
                            (hipStream_t)stream, args.data(),
                            nullptr) != hipSuccess)
    return j_fail("qk_jit_filter_run", "count launch failed");

  // single-block scan lives in the generated module too
  hipFunction_t f_scan;
  if (hipModuleGetFunction(&f_scan, p->mod, "jit_scan") != hipSuccess)
    return j_fail("qk_jit_filter_run", "scan lookup failed");
  std::vector<void *> sargs = {&a_nblocks, &a_scratch, &a_count};
  if (hipModuleLaunchKernel(f_scan, 1, 1, 1, 1, 1, 1, 0,
                            (hipStream_t)stream, sargs.data(),
                            nullptr) != hipSuccess)
    return j_fail("qk_jit_filter_run", "scan launch failed");

  args = {&a_n, &a_chunk, &a_scratch, &a_out};
  for (int c = 0; c < p->ncols; c++) args.push_back((void *)&cols[c]);
  if (hipModuleLaunchKernel(p->f_scatter, blocks, 1, 1, 256, 1, 1, 0,
                            (hipStream_t)stream, args.data(),
                            nullptr) != hipSuccess)
    return j_fail("qk_jit_filter_run", "scatter launch failed");
  return 0;
}

// ---- JIT fused scan + small-cardinality group-by partial aggregate ----
// Generates the k_q1_agg structure (per-thread REGISTER
// accumulators statically unrolled per group, wave shuffle-reduce, block
// LDS reduce, one f64 atomicAdd per accumulator per block) for ARBITRARY
// (predicate, group expression, aggregate expressions). This is how the
// reference's folded map-side DuckDB batch aggregates
// (datastream.py:795-801 + df.py:1354-1394) generalize on device.

static std::string gen_agg_source(const char *pred, const char *group_expr,
                                  int ngroups, int naggs,
                                  const char *const *agg_exprs,
                                  const int *agg_ops, int ncols,
                                  const int *coltypes) {
  auto op = [&](int a) { return agg_ops ? agg_ops[a] : 0; };
  auto init_of = [&](int a) {
    return op(a) == 1 ? "(1.0/0.0)" : op(a) == 2 ? "(-1.0/0.0)" : "0.0";
  };
  std::string s;
  s += "#define BLOCK 256\n#define WAVE 64\n";
  s += "typedef unsigned long long u64; typedef unsigned u32;\n";
  s += "__device__ inline void qk_atomic_mm(double* a, double v, int mx) {\n"
       "  unsigned long long* p = (unsigned long long*)a;\n"
       "  unsigned long long old = *p;\n"
       "  for (;;) {\n"
       "    double cur = __longlong_as_double(old);\n"
       "    if (mx ? (v <= cur) : (v >= cur)) return;\n"
       "    unsigned long long prev = atomicCAS(p, old,"
       " __double_as_longlong(v));\n"
       "    if (prev == old) return;\n"
       "    old = prev;\n  }\n}\n";
  s += "extern \"C\" __global__ __launch_bounds__(BLOCK) void jit_agg(\n";
  s += "    u64 n, double* __restrict__ out";
  for (int c = 0; c < ncols; c++) {
    s += ", const ";
    s += type_name(coltypes[c]);
    s += "* __restrict__ col";
    s += std::to_string(c);
  }
  s += ") {\n";
  s += "  typedef int v2i __attribute__((ext_vector_type(2)));\n"
       "  typedef double v2d __attribute__((ext_vector_type(2)));\n"
       "  typedef long long v2l __attribute__((ext_vector_type(2)));\n";
  for (int g = 0; g < ngroups; g++)
    for (int a = 0; a < naggs; a++)
      s += "  double acc_" + std::to_string(g) + "_" + std::to_string(a) +
           " = " + init_of(a) + ";\n";
  // one row's predicate + accumulate, parameterized by value suffix
  auto body = [&](const char *suf) {
    std::string b;
    std::string pfx = "    ";
    if (pred && pred[0]) {
      std::string p = pred;
      // rename v<k> -> v<k><suf>
      std::string q;
      for (size_t k = 0; k < p.size(); k++) {
        if (p[k] == 'v' && k + 1 < p.size() && isdigit(p[k + 1])) {
          size_t e = k + 1;
          while (e < p.size() && isdigit(p[e])) e++;
          q += p.substr(k, e - k) + suf;
          k = e - 1;
        } else q += p[k];
      }
      b += pfx + "if ((" + q + ")) {\n";
    } else
      b += pfx + "{\n";
    auto ren = [&](const char *src) {
      std::string p = src, q;
      for (size_t k = 0; k < p.size(); k++) {
        if (p[k] == 'v' && k + 1 < p.size() && isdigit(p[k + 1])) {
          size_t e = k + 1;
          while (e < p.size() && isdigit(p[e])) e++;
          q += p.substr(k, e - k) + suf;
          k = e - 1;
        } else q += p[k];
      }
      return q;
    };
    b += pfx + "  int gid = (int)(" + ren(group_expr) + ");\n";
    for (int g = 0; g < ngroups; g++) {
      b += pfx + (g == 0 ? std::string("  if (gid == 0) {\n")
                         : "  else if (gid == " + std::to_string(g) +
                               ") {\n");
      for (int a = 0; a < naggs; a++) {
        std::string acc = "acc_" + std::to_string(g) + "_" +
                          std::to_string(a);
        std::string val = "(double)(" + ren(agg_exprs[a]) + ")";
        if (op(a) == 1)
          b += pfx + "    " + acc + " = fmin(" + acc + ", " + val + ");\n";
        else if (op(a) == 2)
          b += pfx + "    " + acc + " = fmax(" + acc + ", " + val + ");\n";
        else
          b += pfx + "    " + acc + " += " + val + ";\n";
      }
      b += pfx + "  }\n";
    }
    b += pfx + "}\n";
    return b;
  };
  // 2 rows per thread: vector nt loads (16 B/lane on f64/i64 columns)
  s += "  u64 npairs = n / 2;\n"
       "  u64 stride = (u64)gridDim.x * blockDim.x;\n"
       "  for (u64 p = (u64)blockIdx.x * blockDim.x + threadIdx.x;\n"
       "       p < npairs; p += stride) {\n"
       "    u64 i = 2 * p;\n";
  for (int c = 0; c < ncols; c++) {
    std::string cn = std::to_string(c);
    switch (coltypes[c]) {
    case 0:
      s += "    v2i w" + cn + " = __builtin_nontemporal_load("
           "(const v2i*)(col" + cn + " + i));\n"
           "    int v" + cn + "_0 = w" + cn + ".x, v" + cn + "_1 = w" +
           cn + ".y;\n";
      break;
    case 1:
      s += "    v2d w" + cn + " = __builtin_nontemporal_load("
           "(const v2d*)(col" + cn + " + i));\n"
           "    double v" + cn + "_0 = w" + cn + ".x, v" + cn + "_1 = w" +
           cn + ".y;\n";
      break;
    case 3:
      s += "    v2l w" + cn + " = __builtin_nontemporal_load("
           "(const v2l*)(col" + cn + " + i));\n"
           "    long long v" + cn + "_0 = w" + cn + ".x, v" + cn +
           "_1 = w" + cn + ".y;\n";
      break;
    default: // u8: two scalar loads (sub-word vectors not worth it)
      s += "    unsigned char v" + cn + "_0 = col" + cn + "[i];\n"
           "    unsigned char v" + cn + "_1 = col" + cn + "[i + 1];\n";
    }
  }
  s += body("_0");
  s += body("_1");
  s += "  }\n";
  // odd tail handled by one thread with scalar loads
  s += "  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {\n"
       "    u64 i = n - 1;\n";
  for (int c = 0; c < ncols; c++) {
    s += "    ";
    s += type_name(coltypes[c]);
    s += " v" + std::to_string(c) + "_t = col" + std::to_string(c) +
         "[i];\n";
  }
  s += body("_t");
  s += "  }\n";
  // wave reduce -> LDS -> block reduce -> atomicAdd
  s += "  __shared__ double lds[BLOCK / WAVE][" +
       std::to_string(ngroups * naggs) + "];\n"
       "  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;\n";
  for (int g = 0; g < ngroups; g++)
    for (int a = 0; a < naggs; a++) {
      std::string acc = "acc_" + std::to_string(g) + "_" + std::to_string(a);
      const char *comb = op(a) == 1 ? "t = fmin(t, __shfl_down(t, off))"
                         : op(a) == 2 ? "t = fmax(t, __shfl_down(t, off))"
                                      : "t += __shfl_down(t, off)";
      s += "  { double t = " + acc + ";\n"
           "    for (int off = WAVE / 2; off > 0; off >>= 1)\n"
           "      " + std::string(comb) + ";\n"
           "    if (lane == 0) lds[wid][" +
           std::to_string(g * naggs + a) + "] = t; }\n";
    }
  s += "  __syncthreads();\n";
  // per-accumulator block combine + atomic tail (op-specific)
  for (int g = 0; g < ngroups; g++)
    for (int a = 0; a < naggs; a++) {
      std::string idx = std::to_string(g * naggs + a);
      std::string body;
      if (op(a) == 0)
        body = "    double t = 0;\n"
               "    for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w][" +
               idx + "];\n"
               "    if (t != 0.0) atomicAdd(&out[" + idx + "], t);\n";
      else if (op(a) == 1)
        body = "    double t = (1.0/0.0);\n"
               "    for (int w = 0; w < BLOCK / WAVE; w++)"
               " t = fmin(t, lds[w][" + idx + "]);\n"
               "    if (t < (1.0/0.0)) qk_atomic_mm(&out[" + idx +
               "], t, 0);\n";
      else
        body = "    double t = (-1.0/0.0);\n"
               "    for (int w = 0; w < BLOCK / WAVE; w++)"
               " t = fmax(t, lds[w][" + idx + "]);\n"
               "    if (t > (-1.0/0.0)) qk_atomic_mm(&out[" + idx +
               "], t, 1);\n";
      s += "  if (threadIdx.x == " + std::to_string(g * naggs + a) +
           " % BLOCK && threadIdx.x < BLOCK) {\n" + body + "  }\n";
    }
  s += "}\n";
  return s;
}

extern "C" int qk_jit_agg_build(const char *pred, const char *group_expr,
                                int ngroups, int naggs,
                                const char *const *agg_exprs,
                                const int *agg_ops, int ncols,
                                const int *coltypes, void **prog_out) {
  if (ncols < 1 || ncols > QK_JIT_MAX_COLS)
    return j_fail("qk_jit_agg_build", "ncols out of range");
  if (ngroups < 1 || naggs < 1 || ngroups * naggs > 64)
    return j_fail("qk_jit_agg_build",
                  "ngroups*naggs out of range (register accumulators; "
                  "use the hash group-by for high cardinality)");
  for (int c = 0; c < ncols; c++)
    if (!type_name(coltypes[c]))
      return j_fail("qk_jit_agg_build", "bad column type");
  if (agg_ops)
    for (int a = 0; a < naggs; a++)
      if (agg_ops[a] < 0 || agg_ops[a] > 2)
        return j_fail("qk_jit_agg_build", "bad agg op");
  std::string src = gen_agg_source(pred, group_expr, ngroups, naggs,
                                   agg_exprs, agg_ops, ncols, coltypes);
  hiprtcProgram prog;
  if (hiprtcCreateProgram(&prog, src.c_str(), "qk_jit_agg.cu", 0, nullptr,
                          nullptr) != HIPRTC_SUCCESS)
    return j_fail("qk_jit_agg_build", "hiprtcCreateProgram failed");
  const char *opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17",
                        "-munsafe-fp-atomics"};
  hiprtcResult rc = hiprtcCompileProgram(prog, 4, opts);
  if (rc != HIPRTC_SUCCESS) {
    size_t lsz = 0;
    hiprtcGetProgramLogSize(prog, &lsz);
    std::string log(lsz, '\0');
    if (lsz) hiprtcGetProgramLog(prog, &log[0]);
    hiprtcDestroyProgram(&prog);
    snprintf(j_err, sizeof(j_err), "qk_jit_agg_build: compile failed: %s",
             log.c_str());
    return 2;
  }
  size_t csz = 0;
  hiprtcGetCodeSize(prog, &csz);
  QkJitProg *p = new QkJitProg();
  p->code.resize(csz);
  hiprtcGetCode(prog, &p->code[0]);
  hiprtcDestroyProgram(&prog);
  p->ncols = ncols;
  memcpy(p->coltypes, coltypes, ncols * sizeof(int));
  *prog_out = p;
  return 0;
}

extern "C" int qk_jit_agg_run(void *prog, void *stream, uint64_t n,
                              const void *const *col_ptrs,
                              double *out_dev /* accumulates */) {
  QkJitProg *p = (QkJitProg *)prog;
  if (!p->mod) {
    if (hipModuleLoadData(&p->mod, p->code.data()) != hipSuccess)
      return j_fail("qk_jit_agg_run", "hipModuleLoadData failed");
  }
  if (!p->f_count &&
      hipModuleGetFunction(&p->f_count, p->mod, "jit_agg") != hipSuccess)
    return j_fail("qk_jit_agg_run", "hipModuleGetFunction failed");
  if (!n) return 0;
  uint64_t a_n = n;
  void *a_out = out_dev;
  std::vector<const void *> cols(col_ptrs, col_ptrs + p->ncols);
  std::vector<void *> args = {&a_n, &a_out};
  for (int c = 0; c < p->ncols; c++) args.push_back((void *)&cols[c]);
  uint32_t blocks = (uint32_t)((n + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (hipModuleLaunchKernel(p->f_count, blocks, 1, 1, 256, 1, 1, 0,
                            (hipStream_t)stream, args.data(),
                            nullptr) != hipSuccess)
    return j_fail("qk_jit_agg_run", "launch failed");
  return 0;
}

// ---- JIT elementwise transform (transform_sql map side) ---------------
// out[i] = (double)(EXPR over typed columns) — generalizes qk_mul_1md;
// the reference's with_columns_sql / transform_sql per-batch expressions
// (datastream.py:652-815).
static std::string gen_map_source(const char *expr, int ncols,
                                  const int *coltypes) {
  std::string s;
  s += "#define BLOCK 256\n";
  s += "typedef unsigned long long u64;\n";
  s += "extern \"C\" __global__ __launch_bounds__(BLOCK) void jit_map(\n";
  s += "    u64 n, double* __restrict__ out";
  for (int c = 0; c < ncols; c++) {
    s += ", const ";
    s += type_name(coltypes[c]);
    s += "* __restrict__ col";
    s += std::to_string(c);
  }
  s += ") {\n"
       "  u64 stride = (u64)gridDim.x * blockDim.x;\n"
       "  for (u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x; i < n;\n"
       "       i += stride) {\n";
  for (int c = 0; c < ncols; c++) {
    s += "    ";
    s += type_name(coltypes[c]);
    s += " v" + std::to_string(c) + " = col" + std::to_string(c) +
         "[i];\n";
  }
  s += "    out[i] = (double)(";
  s += expr;
  s += ");\n  }\n}\n";
  return s;
}

extern "C" int qk_jit_map_build(const char *expr, int ncols,
                                const int *coltypes, void **prog_out) {
  if (ncols < 1 || ncols > QK_JIT_MAX_COLS)
    return j_fail("qk_jit_map_build", "ncols out of range");
  for (int c = 0; c < ncols; c++)
    if (!type_name(coltypes[c]))
      return j_fail("qk_jit_map_build", "bad column type");
  std::string src = gen_map_source(expr, ncols, coltypes);
  hiprtcProgram prog;
  if (hiprtcCreateProgram(&prog, src.c_str(), "qk_jit_map.cu", 0, nullptr,
                          nullptr) != HIPRTC_SUCCESS)
    return j_fail("qk_jit_map_build", "hiprtcCreateProgram failed");
  const char *opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17"};
  if (hiprtcCompileProgram(prog, 3, opts) != HIPRTC_SUCCESS) {
    size_t lsz = 0;
    hiprtcGetProgramLogSize(prog, &lsz);
    std::string log(lsz, '\0');
    if (lsz) hiprtcGetProgramLog(prog, &log[0]);
    hiprtcDestroyProgram(&prog);
    snprintf(j_err, sizeof(j_err), "qk_jit_map_build: compile failed: %s",
             log.c_str());
    return 2;
  }
  size_t csz = 0;
  hiprtcGetCodeSize(prog, &csz);
  QkJitProg *p = new QkJitProg();
  p->code.resize(csz);
  hiprtcGetCode(prog, &p->code[0]);
  hiprtcDestroyProgram(&prog);
  p->ncols = ncols;
  memcpy(p->coltypes, coltypes, ncols * sizeof(int));
  *prog_out = p;
  return 0;
}

extern "C" int qk_jit_map_run(void *prog, void *stream, uint64_t n,
                              const void *const *col_ptrs, double *out_dev) {
  QkJitProg *p = (QkJitProg *)prog;
  if (!p->mod) {
    if (hipModuleLoadData(&p->mod, p->code.data()) != hipSuccess)
      return j_fail("qk_jit_map_run", "hipModuleLoadData failed");
  }
  if (!p->f_count &&
      hipModuleGetFunction(&p->f_count, p->mod, "jit_map") != hipSuccess)
    return j_fail("qk_jit_map_run", "hipModuleGetFunction failed");
  if (!n) return 0;
  uint64_t a_n = n;
  void *a_out = out_dev;
  std::vector<const void *> cols(col_ptrs, col_ptrs + p->ncols);
  std::vector<void *> args = {&a_n, &a_out};
  for (int c = 0; c < p->ncols; c++) args.push_back((void *)&cols[c]);
  uint32_t blocks = (uint32_t)((n + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (hipModuleLaunchKernel(p->f_count, blocks, 1, 1, 256, 1, 1, 0,
                            (hipStream_t)stream, args.data(),
                            nullptr) != hipSuccess)
    return j_fail("qk_jit_map_run", "launch failed");
  return 0;
}

extern "C" int qk_jit_filter_free(void *prog) {
  QkJitProg *p = (QkJitProg *)prog;
  if (p->scratch) hipFree(p->scratch);
  if (p->mod) hipModuleUnload(p->mod);
  delete p;
  return 0;
}

extern "C" uint64_t qk_jit_code_size(void *prog) {
  return ((QkJitProg *)prog)->code.size();
}
