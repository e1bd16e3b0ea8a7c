// quokka_amd — MI355X (gfx950, CDNA4) kernels for the Quokka columnar hot
// path, behind the C ABI of include/quokka_amd.h.
//
// Design: every kernel here is HBM-bound integer/byte/f64
// streaming work (scan+filter, hash build/probe, group-by, partition) — the
// operations the reference delegates to polars/duckdb.
// No MFMA (no dense contraction on this path). Rules applied from the CDNA4
// guide: 64-lane waves, 256-thread blocks, 16B/lane coalesced loads where
// layout permits, grid-stride with ~2048 blocks, per-wave shuffle reduction
// then per-block LDS reduction before any global atomic.

#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>
#include <string.h>
#include "../../include/quokka_amd.h"

#define WAVE 64

static __host__ __device__ inline uint64_t qk_min_u64(uint64_t a, uint64_t b) {
  return a < b ? a : b;
}
#define BLOCK 256
#define MAX_BLOCKS 2048

static __thread char g_err[512] = "";
static __thread int g_err_set = 0;

static int qk_fail(const char *where, hipError_t e) {
  snprintf(g_err, sizeof(g_err), "%s: %s", where, hipGetErrorString(e));
  g_err_set = 1;
  return (int)(e == hipSuccess ? hipErrorUnknown : e);
}
#define QK_TRY(where, expr)                                                    \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return qk_fail(where, _e);                           \
  } while (0)

extern "C" const char *qk_last_error(void) { return g_err_set ? g_err : ""; }
extern "C" const char *qk_build_arch(void) { return "gfx950"; }

extern "C" int qk_init(int device) {
  QK_TRY("qk_init.setDevice", hipSetDevice(device));
  QK_TRY("qk_init.warm", hipFree(nullptr));
  return 0;
}
extern "C" int qk_device_count(int *out) {
  QK_TRY("qk_device_count", hipGetDeviceCount(out));
  return 0;
}

// ---- memory -----------------------------------------------------------
extern "C" int qk_hfree(void *hptr) {
  QK_TRY("qk_hfree", hipHostFree(hptr));
  return 0;
}
extern "C" int qk_hmalloc_impl(uint64_t nbytes, void **hptr) {
  QK_TRY("qk_hmalloc", hipHostMalloc(hptr, nbytes ? nbytes : 1));
  return 0;
}
extern "C" int qk_dmalloc(uint64_t nbytes, void **dptr) {
  QK_TRY("qk_dmalloc", hipMalloc(dptr, nbytes ? nbytes : 1));
  return 0;
}
extern "C" int qk_dfree(void *dptr) {
  QK_TRY("qk_dfree", hipFree(dptr));
  return 0;
}
extern "C" int qk_h2d(void *dst, const void *src, uint64_t n) {
  QK_TRY("qk_h2d", hipMemcpy(dst, src, n, hipMemcpyHostToDevice));
  return 0;
}
extern "C" int qk_d2h(void *dst, const void *src, uint64_t n) {
  QK_TRY("qk_d2h", hipMemcpy(dst, src, n, hipMemcpyDeviceToHost));
  return 0;
}
extern "C" int qk_h2d_async(void *stream, void *dst, const void *src,
                            uint64_t n) {
  QK_TRY("qk_h2d_async", hipMemcpyAsync(dst, src, n, hipMemcpyHostToDevice,
                                        (hipStream_t)stream));
  return 0;
}
extern "C" int qk_d2h_async(void *stream, void *dst, const void *src,
                            uint64_t n) {
  QK_TRY("qk_d2h_async", hipMemcpyAsync(dst, src, n, hipMemcpyDeviceToHost,
                                        (hipStream_t)stream));
  return 0;
}
extern "C" int qk_dmemset(void *dst, int value, uint64_t n) {
  QK_TRY("qk_dmemset", hipMemset(dst, value, n));
  return 0;
}

__global__ void k_fill_i64(int64_t *dst, int64_t v, uint64_t n) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dst[i] = v;
}
extern "C" int qk_fill_i64(void *stream, int64_t *dst, int64_t v, uint64_t n) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_fill_i64, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, dst, v, n);
  QK_TRY("qk_fill_i64", hipGetLastError());
  return 0;
}

// ---- streams & timing -------------------------------------------------
extern "C" int qk_stream_create(void **stream) {
  QK_TRY("qk_stream_create", hipStreamCreate((hipStream_t *)stream));
  return 0;
}
extern "C" int qk_stream_destroy(void *stream) {
  QK_TRY("qk_stream_destroy", hipStreamDestroy((hipStream_t)stream));
  return 0;
}
extern "C" int qk_stream_sync(void *stream) {
  QK_TRY("qk_stream_sync", hipStreamSynchronize((hipStream_t)stream));
  return 0;
}

// ---- bare events: cross-stream ordering for the overlapped exchange ----
// (the comm stream records an event after each received chunk; the compute
// stream waits on it before probing that chunk — north_star's "shuffle
// overlapped with probe on a side HIP stream")
extern "C" int qk_event_create(void **ev) {
  QK_TRY("qk_event_create",
         hipEventCreateWithFlags((hipEvent_t *)ev, hipEventDisableTiming));
  return 0;
}
extern "C" int qk_event_destroy(void *ev) {
  QK_TRY("qk_event_destroy", hipEventDestroy((hipEvent_t)ev));
  return 0;
}
extern "C" int qk_event_record(void *ev, void *stream) {
  QK_TRY("qk_event_record",
         hipEventRecord((hipEvent_t)ev, (hipStream_t)stream));
  return 0;
}
extern "C" int qk_stream_wait_event(void *stream, void *ev) {
  QK_TRY("qk_stream_wait_event",
         hipStreamWaitEvent((hipStream_t)stream, (hipEvent_t)ev, 0));
  return 0;
}

struct QkTimer {
  hipEvent_t start, stop;
};
extern "C" int qk_timer_create(void **timer) {
  QkTimer *t = new QkTimer();
  QK_TRY("qk_timer_create.start", hipEventCreate(&t->start));
  QK_TRY("qk_timer_create.stop", hipEventCreate(&t->stop));
  *timer = t;
  return 0;
}
extern "C" int qk_timer_destroy(void *timer) {
  QkTimer *t = (QkTimer *)timer;
  hipEventDestroy(t->start);
  hipEventDestroy(t->stop);
  delete t;
  return 0;
}
extern "C" int qk_timer_start(void *timer, void *stream) {
  QK_TRY("qk_timer_start",
         hipEventRecord(((QkTimer *)timer)->start, (hipStream_t)stream));
  return 0;
}
extern "C" int qk_timer_stop(void *timer, void *stream) {
  QK_TRY("qk_timer_stop",
         hipEventRecord(((QkTimer *)timer)->stop, (hipStream_t)stream));
  return 0;
}
extern "C" int qk_timer_elapsed_ms(void *timer, float *out_ms) {
  QkTimer *t = (QkTimer *)timer;
  QK_TRY("qk_timer_elapsed.sync", hipEventSynchronize(t->stop));
  QK_TRY("qk_timer_elapsed", hipEventElapsedTime(out_ms, t->start, t->stop));
  return 0;
}

// ---- RNG (counter-based; device twin of oracle/executors.py splitmix64) --
__device__ __host__ inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ULL;
  x ^= x >> 30;
  x *= 0xBF58476D1CE4E5B9ULL;
  x ^= x >> 27;
  x *= 0x94D049BB133111EBULL;
  x ^= x >> 31;
  return x;
}

// TPC-H date constants (days since 1970-01-01; = oracle/tpch_gen.py values,
// pinned by tests/test_abi.py against the Python side)
#define QK_ORDERDATE_LO 8035   // 1992-01-01
#define QK_ORDERDATE_HI 10440  // 1998-08-02
#define QK_RECEIPT_CUTOFF 9298 // 1995-06-17

__global__ void k_gen_lineitem(uint64_t n, uint64_t row_offset, uint64_t seed,
                               int64_t n_parts, int64_t n_suppliers,
                               int64_t n_orders, int64_t *l_orderkey,
                               int64_t *l_suppkey, double *l_quantity,
                               double *l_extendedprice, double *l_discount,
                               double *l_tax, uint8_t *l_returnflag,
                               uint8_t *l_linestatus, int32_t *l_shipdate,
                               int32_t *l_commitdate,
                               int32_t *l_receiptdate) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t row = row_offset + i;
    uint64_t base = splitmix64(seed ^ 0x51A2B3C4D5E6F708ULL) ^
                    (row * 0x9E3779B97F4A7C15ULL);
    uint64_t h1 = splitmix64(base + 1),
             h2 = splitmix64(base + 2), h3 = splitmix64(base + 3),
             h4 = splitmix64(base + 4), h5 = splitmix64(base + 5),
             h6 = splitmix64(base + 6), h7 = splitmix64(base + 7),
             h8 = splitmix64(base + 8);
    // the ORDER's orderdate, derived with k_gen_orders' exact formula for
    // order row = row/4, so l_shipdate correlates with o_orderdate as in
    // TPC-H (Q3's date-window join selectivity depends on this)
    uint64_t order_row = (row / 4) % (uint64_t)n_orders;
    uint64_t obase = splitmix64(seed ^ 0x0DE50DE50DE50DE5ULL) ^
                     (order_row * 0x9E3779B97F4A7C15ULL);
    int32_t odate = QK_ORDERDATE_LO +
        (int32_t)(splitmix64(obase + 1) %
                  (QK_ORDERDATE_HI - QK_ORDERDATE_LO + 1));
    int32_t ship = odate + 1 + (int32_t)(h1 % 121);
    int32_t receipt = ship + 1 + (int32_t)(h2 % 30);
    // spec: L_COMMITDATE = O_ORDERDATE + U[30,90] (independent stream)
    if (l_commitdate)
      l_commitdate[i] = odate + 30 + (int32_t)(splitmix64(base + 9) % 61);
    if (l_receiptdate) l_receiptdate[i] = receipt;
    if (l_shipdate) l_shipdate[i] = ship;
    if (l_returnflag)
      l_returnflag[i] =
          (receipt <= QK_RECEIPT_CUTOFF) ? ((h3 & 1) ? (uint8_t)0 : (uint8_t)2)
                                         : (uint8_t)1;
    if (l_linestatus) l_linestatus[i] = ship > QK_RECEIPT_CUTOFF ? 1 : 0;
    double qty = (double)(1 + (int32_t)(h4 % 50));
    if (l_quantity) l_quantity[i] = qty;
    if (l_extendedprice) {
      int64_t pk = 1 + (int64_t)(h5 % (uint64_t)n_parts);
      int64_t cents = 90000 + (pk / 10) % 20001 + 100 * (pk % 1000);
      l_extendedprice[i] = qty * ((double)cents / 100.0);
    }
    if (l_discount) l_discount[i] = (double)(h6 % 11) / 100.0;
    if (l_tax) l_tax[i] = (double)(h7 % 9) / 100.0;
    if (l_suppkey) l_suppkey[i] = 1 + (int64_t)(h8 % (uint64_t)n_suppliers);
    if (l_orderkey) {
      // the ORDER row's spec-sparse key (matches k_gen_orders)
      uint64_t orow = row / 4 % (uint64_t)n_orders;
      l_orderkey[i] = (int64_t)((orow / 8) * 32 + orow % 8) + 1;
    }
  }
}
extern "C" int qk_gen_lineitem(void *stream, uint64_t n, uint64_t row_offset,
                               uint64_t seed, int64_t n_parts,
                               int64_t n_suppliers, int64_t n_orders,
                               int64_t *l_orderkey, int64_t *l_suppkey,
                               double *l_quantity, double *l_extendedprice,
                               double *l_discount, double *l_tax,
                               uint8_t *l_returnflag, uint8_t *l_linestatus,
                               int32_t *l_shipdate, int32_t *l_commitdate,
                               int32_t *l_receiptdate) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_gen_lineitem, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, row_offset, seed, n_parts,
                     n_suppliers, n_orders, l_orderkey, l_suppkey, l_quantity,
                     l_extendedprice, l_discount, l_tax, l_returnflag,
                     l_linestatus, l_shipdate, l_commitdate, l_receiptdate);
  QK_TRY("qk_gen_lineitem", hipGetLastError());
  return 0;
}

__global__ void k_gen_orders(uint64_t n, uint64_t row_offset, uint64_t seed,
                             int64_t n_customers, int64_t *o_orderkey,
                             int64_t *o_custkey, int32_t *o_orderdate,
                             int32_t *o_shippriority,
                             uint8_t *o_orderpriority, double *o_totalprice,
                             int64_t li_n_parts) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t row = row_offset + i;
    uint64_t base = splitmix64(seed ^ 0x0DE50DE50DE50DE5ULL) ^
                    (row * 0x9E3779B97F4A7C15ULL);
    uint64_t h0 = splitmix64(base + 0), h1 = splitmix64(base + 1);
    // TPC-H spec 4.2.3 sparse orderkey (8 keys per 32-key bucket) —
    // matches oracle/tpch_gen.py sparse_orderkeys
    if (o_orderkey) o_orderkey[i] = (int64_t)((row / 8) * 32 + row % 8) + 1;
    if (o_custkey) {
      int64_t ck = 1 + (int64_t)(h0 % (uint64_t)n_customers);
      // spec: custkey % 3 != 0 (customer-mortality hole); step down one
      if (ck % 3 == 0) ck -= 1;
      o_custkey[i] = ck;
    }
    if (o_orderdate)
      o_orderdate[i] = QK_ORDERDATE_LO +
          (int32_t)(h1 % (QK_ORDERDATE_HI - QK_ORDERDATE_LO + 1));
    if (o_shippriority) o_shippriority[i] = 0;
    if (o_orderpriority)
      o_orderpriority[i] = (uint8_t)(splitmix64(base + 2) % 5);
    if (o_totalprice) {
      // spec 4.2.3: derived from the order's lines — re-derive the 4
      // lines' qty/partkey/disc/tax with k_gen_lineitem's exact
      // counter-based formulas (line rows 4*row .. 4*row+3)
      double tp = 0.0;
      for (int j = 0; j < 4; j++) {
        uint64_t lrow = 4 * row + (uint64_t)j;
        uint64_t lbase = splitmix64(seed ^ 0x51A2B3C4D5E6F708ULL) ^
                         (lrow * 0x9E3779B97F4A7C15ULL);
        uint64_t h4 = splitmix64(lbase + 4), h5 = splitmix64(lbase + 5),
                 h6 = splitmix64(lbase + 6), h7 = splitmix64(lbase + 7);
        double qty = (double)(1 + (int32_t)(h4 % 50));
        int64_t pk = 1 + (int64_t)(h5 % (uint64_t)li_n_parts);
        int64_t cents = 90000 + (pk / 10) % 20001 + 100 * (pk % 1000);
        double price = qty * ((double)cents / 100.0);
        double disc = (double)(h6 % 11) / 100.0;
        double tax = (double)(h7 % 9) / 100.0;
        tp += price * (1.0 + tax) * (1.0 - disc);
      }
      o_totalprice[i] = tp;
    }
  }
}
extern "C" int qk_gen_orders(void *stream, uint64_t n, uint64_t row_offset,
                             uint64_t seed, int64_t n_customers,
                             int64_t *o_orderkey, int64_t *o_custkey,
                             int32_t *o_orderdate, int32_t *o_shippriority,
                             uint8_t *o_orderpriority, double *o_totalprice,
                             int64_t li_n_parts) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_gen_orders, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, row_offset, seed, n_customers,
                     o_orderkey, o_custkey, o_orderdate, o_shippriority,
                     o_orderpriority, o_totalprice, li_n_parts);
  QK_TRY("qk_gen_orders", hipGetLastError());
  return 0;
}

__global__ void k_gen_customer(uint64_t n, uint64_t row_offset, uint64_t seed,
                               int64_t *c_custkey, uint8_t *c_mktsegment,
                               int32_t *c_nationkey) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t row = row_offset + i;
    uint64_t h = splitmix64((seed ^ 0xC0573C0573C0573CULL) +
                            row * 0x9E3779B97F4A7C15ULL);
    if (c_custkey) c_custkey[i] = (int64_t)row + 1;
    if (c_mktsegment) c_mktsegment[i] = (uint8_t)(h % 5);
    if (c_nationkey) c_nationkey[i] = (int32_t)((h >> 32) % 25);
  }
}
extern "C" int qk_gen_customer(void *stream, uint64_t n, uint64_t row_offset,
                               uint64_t seed, int64_t *c_custkey,
                               uint8_t *c_mktsegment, int32_t *c_nationkey) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_gen_customer, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, row_offset, seed, c_custkey,
                     c_mktsegment, c_nationkey);
  QK_TRY("qk_gen_customer", hipGetLastError());
  return 0;
}

// ---- Q1 fused filter + group-by partial aggregate ---------------------
// 6 groups x 6 accumulators kept in REGISTERS per thread (statically
// indexed; runtime-indexed per-thread arrays spill to scratch on hipcc —
// guide §5.4 rule 20), divergent-but-cheap if-chain per row, then wave
// shuffle-reduce, block LDS reduce, one global f64 atomicAdd per
// accumulator per block (guide G12).

#define Q1_ACC_DECL(g)                                                         \
  double a##g##0 = 0, a##g##1 = 0, a##g##2 = 0, a##g##3 = 0, a##g##4 = 0,      \
         a##g##5 = 0;
#define Q1_ACC_ADD(g)                                                          \
  if (gid == g) {                                                              \
    a##g##0 += qty;                                                            \
    a##g##1 += price;                                                          \
    a##g##2 += disc_price;                                                     \
    a##g##3 += charge;                                                         \
    a##g##4 += disc;                                                           \
    a##g##5 += 1.0;                                                            \
  }
#define Q1_ACC_ROW()                                                           \
  do {                                                                         \
    double disc_price = price * (1.0 - disc);                                  \
    double charge = disc_price * (1.0 + tax);                                  \
    Q1_ACC_ADD(0) else Q1_ACC_ADD(1) else Q1_ACC_ADD(2) else Q1_ACC_ADD(       \
        3) else Q1_ACC_ADD(4) else Q1_ACC_ADD(5)                               \
  } while (0)

__device__ inline double wave_reduce(double v) {
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return v;
}

__global__ void __launch_bounds__(BLOCK) k_q1_agg(
    uint64_t n, const int32_t *__restrict__ shipdate,
    const double *__restrict__ quantity, const double *__restrict__ extprice,
    const double *__restrict__ discount, const double *__restrict__ tax_,
    const uint8_t *__restrict__ rflag, const uint8_t *__restrict__ lstat,
    int32_t cutoff, double *__restrict__ out) {
  Q1_ACC_DECL(0) Q1_ACC_DECL(1) Q1_ACC_DECL(2)
  Q1_ACC_DECL(3) Q1_ACC_DECL(4) Q1_ACC_DECL(5)

  uint64_t npairs = n / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; p < npairs;
       p += stride) {
    uint64_t r = 2 * p;
    // 16B/lane coalesced NON-TEMPORAL vector loads: single-pass stream,
    // keep it out of the caches (measured +8% over plain loads)
    typedef double v2d __attribute__((ext_vector_type(2)));
    typedef int v2i __attribute__((ext_vector_type(2)));
    v2d q2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(quantity + r));
    v2d p2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(extprice + r));
    v2d d2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(discount + r));
    v2d t2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(tax_ + r));
    v2i s2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2i *>(shipdate + r));
    uint8_t f0 = __builtin_nontemporal_load(&rflag[r]);
    uint8_t f1 = __builtin_nontemporal_load(&rflag[r + 1]);
    uint8_t l0 = __builtin_nontemporal_load(&lstat[r]);
    uint8_t l1 = __builtin_nontemporal_load(&lstat[r + 1]);
    if (s2.x <= cutoff) {
      int gid = (int)f0 * 2 + (int)l0;
      double qty = q2.x, price = p2.x, disc = d2.x, tax = t2.x;
      Q1_ACC_ROW();
    }
    if (s2.y <= cutoff) {
      int gid = (int)f1 * 2 + (int)l1;
      double qty = q2.y, price = p2.y, disc = d2.y, tax = t2.y;
      Q1_ACC_ROW();
    }
  }
  // odd tail row handled by global thread 0
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    uint64_t r = n - 1;
    if (shipdate[r] <= cutoff) {
      int gid = (int)rflag[r] * 2 + (int)lstat[r];
      double qty = quantity[r], price = extprice[r], disc = discount[r],
             tax = tax_[r];
      Q1_ACC_ROW();
    }
  }

  // wave shuffle-reduce each accumulator, lane 0 stages to LDS
  __shared__ double lds[BLOCK / WAVE][36];
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
#define Q1_RED(g, j, var)                                                      \
  {                                                                            \
    double s = wave_reduce(var);                                               \
    if (lane == 0) lds[wid][g * 6 + j] = s;                                    \
  }
#define Q1_RED_G(g)                                                            \
  Q1_RED(g, 0, a##g##0) Q1_RED(g, 1, a##g##1) Q1_RED(g, 2, a##g##2)            \
  Q1_RED(g, 3, a##g##3) Q1_RED(g, 4, a##g##4) Q1_RED(g, 5, a##g##5)
  Q1_RED_G(0) Q1_RED_G(1) Q1_RED_G(2) Q1_RED_G(3) Q1_RED_G(4) Q1_RED_G(5)
  __syncthreads();
  // threads 0..35 each combine the per-wave partials for one accumulator
  if (threadIdx.x < 36) {
    double s = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) s += lds[w][threadIdx.x];
    int g = threadIdx.x / 6, j = threadIdx.x % 6;
    if (s != 0.0) atomicAdd(&out[g * 8 + j], s);
  }
}

extern "C" int qk_q1_agg(void *stream, uint64_t n, const int32_t *shipdate,
                         const double *quantity, const double *extprice,
                         const double *discount, const double *tax,
                         const uint8_t *rflag, const uint8_t *lstat,
                         int32_t cutoff, double *out) {
  uint64_t npairs = n / 2;
  uint32_t blocks =
      (uint32_t)qk_min_u64(MAX_BLOCKS, (npairs + BLOCK - 1) / BLOCK);
  if (!blocks) blocks = 1;
  hipLaunchKernelGGL(k_q1_agg, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, shipdate, quantity, extprice,
                     discount, tax, rflag, lstat, cutoff, out);
  QK_TRY("qk_q1_agg", hipGetLastError());
  return 0;
}

// ---- Q6 fused filter + sum -------------------------------------------
__global__ void __launch_bounds__(BLOCK) k_q6_agg(
    uint64_t n, const int32_t *__restrict__ shipdate,
    const double *__restrict__ quantity, const double *__restrict__ extprice,
    const double *__restrict__ discount, int32_t date_lo, int32_t date_hi,
    double disc_lo, double disc_hi, double qty_hi, double *__restrict__ out) {
  double rev = 0, cnt = 0;
  uint64_t npairs = n / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; p < npairs;
       p += stride) {
    uint64_t r = 2 * p;
    typedef double v2d __attribute__((ext_vector_type(2)));
    typedef int v2i __attribute__((ext_vector_type(2)));
    v2d q2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(quantity + r));
    v2d p2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(extprice + r));
    v2d d2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2d *>(discount + r));
    v2i s2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2i *>(shipdate + r));
    if (s2.x >= date_lo && s2.x < date_hi && d2.x >= disc_lo &&
        d2.x <= disc_hi && q2.x < qty_hi) {
      rev += p2.x * d2.x;
      cnt += 1.0;
    }
    if (s2.y >= date_lo && s2.y < date_hi && d2.y >= disc_lo &&
        d2.y <= disc_hi && q2.y < qty_hi) {
      rev += p2.y * d2.y;
      cnt += 1.0;
    }
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    uint64_t r = n - 1;
    if (shipdate[r] >= date_lo && shipdate[r] < date_hi &&
        discount[r] >= disc_lo && discount[r] <= disc_hi &&
        quantity[r] < qty_hi) {
      rev += extprice[r] * discount[r];
      cnt += 1.0;
    }
  }
  __shared__ double lds[2][BLOCK / WAVE];
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  double s0 = wave_reduce(rev), s1 = wave_reduce(cnt);
  if (lane == 0) {
    lds[0][wid] = s0;
    lds[1][wid] = s1;
  }
  __syncthreads();
  if (threadIdx.x < 2) {
    double s = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) s += lds[threadIdx.x][w];
    if (s != 0.0) atomicAdd(&out[threadIdx.x], s);
  }
}
extern "C" int qk_q6_agg(void *stream, uint64_t n, const int32_t *shipdate,
                         const double *quantity, const double *extprice,
                         const double *discount, int32_t date_lo,
                         int32_t date_hi, double disc_lo, double disc_hi,
                         double qty_hi, double *out) {
  uint64_t npairs = n / 2;
  uint32_t blocks =
      (uint32_t)qk_min_u64(MAX_BLOCKS, (npairs + BLOCK - 1) / BLOCK);
  if (!blocks) blocks = 1;
  hipLaunchKernelGGL(k_q6_agg, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, shipdate, quantity, extprice,
                     discount, date_lo, date_hi, disc_lo, disc_hi, qty_hi, out);
  QK_TRY("qk_q6_agg", hipGetLastError());
  return 0;
}

// ---- generic filter (ordered compaction) ------------------------------
// Three launches: per-block count over contiguous chunks -> single-block
// exclusive scan of block counts -> re-evaluate + rank + scatter. Row order
// is preserved (polars filter semantics, core.py:170).

template <typename T>
__device__ inline bool cmp_op(T v, int op, T ref) {
  switch (op) {
  case 0: return v < ref;
  case 1: return v <= ref;
  case 2: return v > ref;
  case 3: return v >= ref;
  case 4: return v == ref;
  default: return v != ref;
  }
}

template <typename T>
__global__ void __launch_bounds__(BLOCK) k_filter_count(
    uint64_t n, const T *__restrict__ col, int op, T ref, uint64_t chunk,
    uint64_t *__restrict__ block_counts) {
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  uint32_t cnt = 0;
  for (uint64_t r = lo + threadIdx.x; r < hi; r += BLOCK)
    cnt += cmp_op(col[r], op, ref) ? 1u : 0u;
  // wave reduce then block reduce
  __shared__ uint32_t lds[BLOCK / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  if (lane == 0) lds[wid] = cnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t s = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) s += lds[w];
    block_counts[blockIdx.x] = s;
  }
}

__global__ void k_scan_blocks(uint64_t nblocks, uint64_t *__restrict__ counts,
                              uint64_t *__restrict__ total) {
  // single thread serial scan; nblocks <= 4096 so this is microseconds
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    uint64_t acc = 0;
    for (uint64_t b = 0; b < nblocks; b++) {
      uint64_t c = counts[b];
      counts[b] = acc;
      acc += c;
    }
    *total = acc;
  }
}

template <typename T>
__global__ void __launch_bounds__(BLOCK) k_filter_scatter(
    uint64_t n, const T *__restrict__ col, int op, T ref, uint64_t chunk,
    const uint64_t *__restrict__ block_offsets, uint32_t *__restrict__ out_idx) {
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  __shared__ uint64_t base;
  __shared__ uint32_t wave_tot[BLOCK / WAVE];
  if (threadIdx.x == 0) base = block_offsets[blockIdx.x];
  __syncthreads();
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  for (uint64_t r0 = lo; r0 < hi; r0 += BLOCK) {
    uint64_t r = r0 + threadIdx.x;
    bool pass = r < hi && cmp_op(col[r], op, ref);
    uint64_t mask = __ballot(pass);
    uint32_t rank = __popcll(mask & ((1ULL << lane) - 1));
    uint32_t wtot = __popcll(mask);
    if (lane == 0) wave_tot[wid] = wtot;
    __syncthreads();
    uint32_t wbase = 0;
    for (int w = 0; w < wid; w++) wbase += wave_tot[w];
    if (pass) out_idx[base + wbase + rank] = (uint32_t)r;
    uint32_t btot = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) btot += wave_tot[w];
    __syncthreads();
    if (threadIdx.x == 0) base += btot;
    __syncthreads();
  }
}

template <typename T>
static int qk_filter_impl(const char *who, void *stream, uint64_t n,
                          const T *col, int op, T value, uint32_t *out_idx,
                          uint64_t *out_count_dev) {
  if (!n) return 0;
  uint64_t rows_per_block = (n + MAX_BLOCKS - 1) / MAX_BLOCKS;
  rows_per_block = ((rows_per_block + BLOCK - 1) / BLOCK) * BLOCK;
  uint32_t blocks = (uint32_t)((n + rows_per_block - 1) / rows_per_block);
  static __thread uint64_t *scratch = nullptr;  // per-thread block_counts buf
  static __thread uint64_t scratch_cap = 0;
  if (scratch_cap < blocks) {
    if (scratch) hipFree(scratch);
    QK_TRY(who, hipMalloc(&scratch, (MAX_BLOCKS + 1) * sizeof(uint64_t)));
    scratch_cap = MAX_BLOCKS + 1;
  }
  hipLaunchKernelGGL(k_filter_count<T>, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, col, op, value, rows_per_block,
                     scratch);
  hipLaunchKernelGGL(k_scan_blocks, dim3(1), dim3(1), 0, (hipStream_t)stream,
                     (uint64_t)blocks, scratch, out_count_dev);
  hipLaunchKernelGGL(k_filter_scatter<T>, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, col, op, value, rows_per_block,
                     scratch, out_idx);
  QK_TRY(who, hipGetLastError());
  return 0;
}
extern "C" int qk_filter_i32(void *stream, uint64_t n, const int32_t *col,
                             int op, int32_t value, uint32_t *out_idx,
                             uint64_t *out_count_dev) {
  return qk_filter_impl("qk_filter_i32", stream, n, col, op, value, out_idx,
                        out_count_dev);
}
extern "C" int qk_filter_u8(void *stream, uint64_t n, const uint8_t *col,
                            int op, uint8_t value, uint32_t *out_idx,
                            uint64_t *out_count_dev) {
  return qk_filter_impl("qk_filter_u8", stream, n, col, op, value, out_idx,
                        out_count_dev);
}
// f64 variant: the threshold is a KERNEL ARGUMENT, so data-dependent
// cuts (e.g. Q22's "c_acctbal > avg" with avg computed on device) do
// not force a per-value hiprtc recompile the way a JIT literal does.
extern "C" int qk_filter_f64(void *stream, uint64_t n, const double *col,
                             int op, double value, uint32_t *out_idx,
                             uint64_t *out_count_dev) {
  return qk_filter_impl("qk_filter_f64", stream, n, col, op, value, out_idx,
                        out_count_dev);
}

// elementwise comparison flag (a > b -> 1.0) — feeds MAX-aggregated
// "any late line for this pair" dedup in Q21 (exists-subquery shape)
__global__ void k_flag_gt_i32(uint64_t n, const int32_t *__restrict__ a,
                              const int32_t *__restrict__ b,
                              double *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = a[i] > b[i] ? 1.0 : 0.0;
}
extern "C" int qk_flag_gt_i32(void *stream, uint64_t n, const int32_t *a,
                              const int32_t *b, double *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_flag_gt_i32, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, a, b, out);
  QK_TRY("qk_flag_gt_i32", hipGetLastError());
  return 0;
}

// ---- elementwise revenue ----------------------------------------------
__global__ void k_mul_1md(uint64_t n, const double *__restrict__ a,
                          const double *__restrict__ b,
                          double *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = a[i] * (1.0 - b[i]);
}
extern "C" int qk_mul_1md(void *stream, uint64_t n, const double *a,
                          const double *b, double *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_mul_1md, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, a, b, out);
  QK_TRY("qk_mul_1md", hipGetLastError());
  return 0;
}

// ---- gathers ----------------------------------------------------------
template <typename T>
__global__ void k_gather(uint64_t n, const uint32_t *__restrict__ idx,
                         const T *__restrict__ src, T *__restrict__ dst) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dst[i] = src[idx[i]];
}
template <typename T>
static int qk_gather_impl(const char *who, void *stream, uint64_t n,
                          const uint32_t *idx, const T *src, T *dst) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_gather<T>, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, idx, src, dst);
  QK_TRY(who, hipGetLastError());
  return 0;
}
extern "C" int qk_gather_i64(void *s, uint64_t n, const uint32_t *i,
                             const int64_t *src, int64_t *dst) {
  return qk_gather_impl("qk_gather_i64", s, n, i, src, dst);
}
extern "C" int qk_gather_f64(void *s, uint64_t n, const uint32_t *i,
                             const double *src, double *dst) {
  return qk_gather_impl("qk_gather_f64", s, n, i, src, dst);
}
extern "C" int qk_gather_i32(void *s, uint64_t n, const uint32_t *i,
                             const int32_t *src, int32_t *dst) {
  return qk_gather_impl("qk_gather_i32", s, n, i, src, dst);
}
extern "C" int qk_gather_u8(void *s, uint64_t n, const uint32_t *i,
                            const uint8_t *src, uint8_t *dst) {
  return qk_gather_impl("qk_gather_u8", s, n, i, src, dst);
}

// ---- hash join --------------------------------------------------------
// Open addressing, linear probing, splitmix64(key) & (cap-1). Duplicate
// build keys chain via chain_next (head swap is a single atomicExch, so
// build order within a key is unspecified — as is polars', sql_executors
// :371; parity compares multisets).

__device__ inline uint64_t slot_of(int64_t key, uint64_t cap) {
  return splitmix64((uint64_t)key) & (cap - 1);
}

// Blocked Bloom prefilter: k=2 bits in ONE u32 word (one 4 B read per
// probe, one cache line touched) — ~94% of probe misses skip the
// table-line read. bloom_mask = bit count - 1 (pow2; words = bits/32).
__device__ inline uint32_t bloom_word_mask(int64_t key, uint64_t bloom_mask,
                                           uint64_t *word_idx) {
  uint64_t h = splitmix64((uint64_t)key ^ 0xB10011B10011B100ULL);
  *word_idx = (h & bloom_mask) >> 5;
  uint32_t b1 = (uint32_t)(h >> 40) & 31u, b2 = (uint32_t)(h >> 48) & 31u;
  return (1u << b1) | (1u << b2);
}
__device__ inline void bloom_set(uint32_t *bloom, uint64_t bloom_mask,
                                 int64_t key) {
  uint64_t w;
  uint32_t m = bloom_word_mask(key, bloom_mask, &w);
  atomicOr(&bloom[w], m);
}
__device__ inline bool bloom_test(const uint32_t *bloom, uint64_t bloom_mask,
                                  int64_t key) {
  uint64_t w;
  uint32_t m = bloom_word_mask(key, bloom_mask, &w);
  return (bloom[w] & m) == m;
}

__global__ void __launch_bounds__(BLOCK) k_join_build2(
    uint64_t n, const int64_t *__restrict__ keys, uint32_t row_offset,
    int64_t *__restrict__ slot_keys, int32_t *__restrict__ slot_head,
    int32_t *__restrict__ chain_next, uint64_t cap) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t key = keys[i];
    uint64_t s = slot_of(key, cap);
    for (;;) {
      int64_t cur = slot_keys[s];
      if (cur == key) break;
      if (cur == QK_JOIN_EMPTY) {
        int64_t prev = (int64_t)atomicCAS((unsigned long long *)&slot_keys[s],
                                          (unsigned long long)QK_JOIN_EMPTY,
                                          (unsigned long long)key);
        if (prev == QK_JOIN_EMPTY || prev == key) break;
      }
      s = (s + 1) & (cap - 1);
    }
    int32_t me = (int32_t)(row_offset + i);
    int32_t old = atomicExch(&slot_head[s], me);
    chain_next[me] = old;
  }
}

extern "C" int qk_join_build(void *stream, uint64_t n, const int64_t *keys,
                             uint32_t row_offset, int64_t *slot_keys,
                             int32_t *slot_head, int32_t *chain_next,
                             uint64_t cap) {
  if (!n) return 0;
  if (cap & (cap - 1)) return qk_fail("qk_join_build.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_join_build2, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys, row_offset, slot_keys,
                     slot_head, chain_next, cap);
  QK_TRY("qk_join_build", hipGetLastError());
  return 0;
}

__global__ void __launch_bounds__(BLOCK) k_join_probe(
    uint64_t n, const int64_t *__restrict__ keys,
    const int64_t *__restrict__ slot_keys, const int32_t *__restrict__ slot_head,
    const int32_t *__restrict__ chain_next, uint64_t cap, int mode,
    uint32_t *__restrict__ out_probe, uint32_t *__restrict__ out_build,
    uint64_t out_cap, uint64_t *__restrict__ cursor) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t key = keys[i];
    uint64_t s = slot_of(key, cap);
    int32_t head = -1;
    for (;;) {
      int64_t cur = slot_keys[s];
      if (cur == key) {
        head = slot_head[s];
        break;
      }
      if (cur == QK_JOIN_EMPTY) break;
      s = (s + 1) & (cap - 1);
    }
    if (mode == 1) {  // semi
      if (head >= 0) {
        uint64_t pos = atomicAdd((unsigned long long *)cursor, 1ULL);
        if (pos < out_cap) out_probe[pos] = (uint32_t)i;
      }
      continue;
    }
    if (mode == 2) {  // anti
      if (head < 0) {
        uint64_t pos = atomicAdd((unsigned long long *)cursor, 1ULL);
        if (pos < out_cap) out_probe[pos] = (uint32_t)i;
      }
      continue;
    }
    if (head < 0) continue;
    // inner: count chain, claim a contiguous range, emit
    uint32_t cnt = 0;
    for (int32_t b = head; b >= 0; b = chain_next[b]) cnt++;
    uint64_t pos = atomicAdd((unsigned long long *)cursor, (unsigned long long)cnt);
    for (int32_t b = head; b >= 0; b = chain_next[b]) {
      if (pos < out_cap) {
        out_probe[pos] = (uint32_t)i;
        out_build[pos] = (uint32_t)b;
      }
      pos++;
    }
  }
}
extern "C" int qk_join_probe(void *stream, uint64_t n, const int64_t *keys,
                             const int64_t *slot_keys, const int32_t *slot_head,
                             const int32_t *chain_next, uint64_t cap, int mode,
                             uint32_t *out_probe, uint32_t *out_build,
                             uint64_t out_cap, uint64_t *cursor) {
  if (!n) return 0;
  if (cap & (cap - 1)) return qk_fail("qk_join_probe.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_join_probe, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys, slot_keys, slot_head,
                     chain_next, cap, mode, out_probe, out_build, out_cap,
                     cursor);
  QK_TRY("qk_join_probe", hipGetLastError());
  return 0;
}

// ---- fused Q3 path -----------------------------------------------------
// Build/probe with the surrounding filter and aggregate FUSED into the
// table pass (the reference folds the filter and per-batch partial agg
// into partition_fn the same way, core.py:152-195 + df.py:1354-1394).
// Build keys must be UNIQUE (orders/customer primary keys): slot_head
// stores the build ROW, and the orders table's slots double as group-by
// slots for the probe-side aggregate.

// customer: insert keys[i] where flag[i] == flag_val
__global__ void __launch_bounds__(BLOCK) k_build_u8eq(
    uint64_t n, const int64_t *__restrict__ keys,
    const uint8_t *__restrict__ flag, uint8_t flag_val,
    int64_t *__restrict__ slot_keys, int32_t *__restrict__ slot_head,
    uint64_t cap, uint32_t *__restrict__ bloom, uint64_t bloom_mask) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (flag[i] != flag_val) continue;
    int64_t key = keys[i];
    uint64_t s = slot_of(key, cap);
    for (;;) {
      int64_t cur = slot_keys[s];
      if (cur == key) break;
      if (cur == QK_JOIN_EMPTY) {
        int64_t prev = (int64_t)atomicCAS((unsigned long long *)&slot_keys[s],
                                          (unsigned long long)QK_JOIN_EMPTY,
                                          (unsigned long long)key);
        if (prev == QK_JOIN_EMPTY || prev == key) break;
      }
      s = (s + 1) & (cap - 1);
    }
    slot_head[s] = (int32_t)i;
    if (bloom) bloom_set(bloom, bloom_mask, key);
  }
}
extern "C" int qk_build_u8eq(void *stream, uint64_t n, const int64_t *keys,
                             const uint8_t *flag, uint8_t flag_val,
                             int64_t *slot_keys, int32_t *slot_head,
                             uint64_t cap, uint32_t *bloom,
                             uint64_t bloom_mask) {
  if (!n) return 0;
  if (cap & (cap - 1)) return qk_fail("qk_build_u8eq.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_build_u8eq, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys, flag, flag_val, slot_keys,
                     slot_head, cap, bloom, bloom_mask);
  QK_TRY("qk_build_u8eq", hipGetLastError());
  return 0;
}

__device__ inline int32_t probe_unique(const int64_t *__restrict__ slot_keys,
                                       const int32_t *__restrict__ slot_head,
                                       uint64_t cap, int64_t key,
                                       uint64_t *slot_out) {
  uint64_t s = slot_of(key, cap);
  for (;;) {
    int64_t cur = slot_keys[s];
    if (cur == key) {
      if (slot_out) *slot_out = s;
      return slot_head[s];
    }
    if (cur == QK_JOIN_EMPTY) return -1;
    s = (s + 1) & (cap - 1);
  }
}

// orders: insert o_orderkey[i] where o_orderdate[i] < date_lt AND
// o_custkey[i] hits the customer table (fused filter + semi join + build)
__device__ inline void q3_build_row(
    int64_t ck, bool pass, uint64_t i,
    const int64_t *__restrict__ o_orderkey,
    const int64_t *__restrict__ cust_keys,
    const int32_t *__restrict__ cust_head, uint64_t cust_cap,
    int64_t *__restrict__ slot_keys, int32_t *__restrict__ slot_head,
    uint64_t cap, uint32_t *__restrict__ bloom, uint64_t bloom_mask) {
  if (!pass) return;
  if (probe_unique(cust_keys, cust_head, cust_cap, ck, nullptr) < 0) return;
  int64_t key = __builtin_nontemporal_load(&o_orderkey[i]);
  uint64_t s = slot_of(key, cap);
  for (;;) {
    int64_t cur = slot_keys[s];
    if (cur == key) break;
    if (cur == QK_JOIN_EMPTY) {
      int64_t prev = (int64_t)atomicCAS((unsigned long long *)&slot_keys[s],
                                        (unsigned long long)QK_JOIN_EMPTY,
                                        (unsigned long long)key);
      if (prev == QK_JOIN_EMPTY || prev == key) break;
    }
    s = (s + 1) & (cap - 1);
  }
  slot_head[s] = (int32_t)i;
  if (bloom) bloom_set(bloom, bloom_mask, key);
}

// 2 rows/thread (vector nt loads, two cust lookups/inserts in flight)
__global__ void __launch_bounds__(BLOCK) k_q3_build_orders(
    uint64_t n, const int64_t *__restrict__ o_orderkey,
    const int64_t *__restrict__ o_custkey,
    const int32_t *__restrict__ o_orderdate, int32_t date_lt,
    const int64_t *__restrict__ cust_keys,
    const int32_t *__restrict__ cust_head, uint64_t cust_cap,
    int64_t *__restrict__ slot_keys, int32_t *__restrict__ slot_head,
    uint64_t cap, uint32_t *__restrict__ bloom, uint64_t bloom_mask,
    const uint32_t *__restrict__ cbloom, uint64_t cbloom_mask) {
  typedef int v2i __attribute__((ext_vector_type(2)));
  typedef long long v2l __attribute__((ext_vector_type(2)));
  uint64_t npairs = n / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       p < npairs; p += stride) {
    uint64_t i = 2 * p;
    v2i d2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2i *>(o_orderdate + i));
    v2l c2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2l *>(o_custkey + i));
    bool p0 = d2.x < date_lt, p1 = d2.y < date_lt;
    if (cbloom) {
      if (p0) p0 = bloom_test(cbloom, cbloom_mask, c2.x);
      if (p1) p1 = bloom_test(cbloom, cbloom_mask, c2.y);
    }
    q3_build_row(c2.x, p0, i, o_orderkey, cust_keys, cust_head, cust_cap,
                 slot_keys, slot_head, cap, bloom, bloom_mask);
    q3_build_row(c2.y, p1, i + 1, o_orderkey, cust_keys, cust_head,
                 cust_cap, slot_keys, slot_head, cap, bloom, bloom_mask);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    uint64_t i = n - 1;
    if (o_orderdate[i] < date_lt) {
      int64_t ck = o_custkey[i];
      bool pass = !cbloom || bloom_test(cbloom, cbloom_mask, ck);
      q3_build_row(ck, pass, i, o_orderkey, cust_keys, cust_head, cust_cap,
                   slot_keys, slot_head, cap, bloom, bloom_mask);
    }
  }
}
extern "C" int qk_q3_build_orders(void *stream, uint64_t n,
                                  const int64_t *o_orderkey,
                                  const int64_t *o_custkey,
                                  const int32_t *o_orderdate, int32_t date_lt,
                                  const int64_t *cust_keys,
                                  const int32_t *cust_head, uint64_t cust_cap,
                                  int64_t *slot_keys, int32_t *slot_head,
                                  uint64_t cap, uint32_t *bloom,
                                  uint64_t bloom_mask,
                                  const uint32_t *cbloom,
                                  uint64_t cbloom_mask) {
  if (!n) return 0;
  if ((cap & (cap - 1)) || (cust_cap & (cust_cap - 1)))
    return qk_fail("qk_q3_build_orders.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q3_build_orders, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, o_orderkey, o_custkey,
                     o_orderdate, date_lt, cust_keys, cust_head, cust_cap,
                     slot_keys, slot_head, cap, bloom, bloom_mask, cbloom,
                     cbloom_mask);
  QK_TRY("qk_q3_build_orders", hipGetLastError());
  return 0;
}

// count the rows k_q3_build_orders would insert (sizes the build table
// tightly so probes stay cache-resident)
__global__ void __launch_bounds__(BLOCK) k_q3_count_orders(
    uint64_t n, const int64_t *__restrict__ o_custkey,
    const int32_t *__restrict__ o_orderdate, int32_t date_lt,
    const int64_t *__restrict__ cust_keys,
    const int32_t *__restrict__ cust_head, uint64_t cust_cap,
    uint64_t *__restrict__ count, const uint32_t *__restrict__ cbloom,
    uint64_t cbloom_mask) {
  uint32_t cnt = 0;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (__builtin_nontemporal_load(&o_orderdate[i]) >= date_lt) continue;
    int64_t ck = __builtin_nontemporal_load(&o_custkey[i]);
    if (cbloom && !bloom_test(cbloom, cbloom_mask, ck)) continue;
    if (probe_unique(cust_keys, cust_head, cust_cap, ck, nullptr) >= 0)
      cnt++;
  }
  __shared__ uint32_t lds[BLOCK / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  if (lane == 0) lds[wid] = cnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t s = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) s += lds[w];
    if (s) atomicAdd((unsigned long long *)count, (unsigned long long)s);
  }
}
extern "C" int qk_q3_count_orders(void *stream, uint64_t n,
                                  const int64_t *o_custkey,
                                  const int32_t *o_orderdate, int32_t date_lt,
                                  const int64_t *cust_keys,
                                  const int32_t *cust_head, uint64_t cust_cap,
                                  uint64_t *count_dev, const uint32_t *cbloom,
                                  uint64_t cbloom_mask) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q3_count_orders, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, o_custkey, o_orderdate, date_lt,
                     cust_keys, cust_head, cust_cap, count_dev, cbloom,
                     cbloom_mask);
  QK_TRY("qk_q3_count_orders", hipGetLastError());
  return 0;
}

// lineitem: one fused pass — filter l_shipdate > date_gt, probe orders,
// revenue = price*(1-disc), atomicAdd into the orders table's slot
// (slot == group, since orderkey is unique on the build side)
__global__ void __launch_bounds__(BLOCK) k_q3_probe_agg(
    uint64_t n, const int64_t *__restrict__ l_orderkey,
    const int32_t *__restrict__ l_shipdate,
    const double *__restrict__ l_price, const double *__restrict__ l_disc,
    int32_t date_gt, const int64_t *__restrict__ slot_keys,
    const int32_t *__restrict__ slot_head, uint64_t cap,
    double *__restrict__ slot_sums, uint64_t *__restrict__ match_count) {
  uint32_t matches = 0;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (l_shipdate[i] <= date_gt) continue;
    uint64_t s;
    if (probe_unique(slot_keys, slot_head, cap, l_orderkey[i], &s) < 0)
      continue;
    matches++;
    atomicAdd(&slot_sums[s], l_price[i] * (1.0 - l_disc[i]));
  }
  if (match_count) {
    __shared__ uint32_t lds[BLOCK / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1)
      matches += __shfl_down(matches, off);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) lds[wid] = matches;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint64_t t = 0;
      for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
      if (t) atomicAdd((unsigned long long *)match_count,
                       (unsigned long long)t);
    }
  }
}
// nt variant: non-temporal loads on the streamed lineitem columns keep
// the hash table cache-resident (the streaming scan otherwise evicts it;
// probe line reads dominate actual traffic — profiles/r01_q3)
__device__ inline void q3_probe_row(
    int64_t key, bool pass, uint64_t i, const double *__restrict__ l_price,
    const double *__restrict__ l_disc, const int64_t *__restrict__ slot_keys,
    const int32_t *__restrict__ slot_head, uint64_t cap,
    double *__restrict__ slot_sums, uint32_t &matches) {
  if (!pass) return;
  uint64_t s = slot_of(key, cap);
  int32_t head = -1;
  for (;;) {
    int64_t cur = slot_keys[s];
    if (cur == key) { head = slot_head[s]; break; }
    if (cur == QK_JOIN_EMPTY) break;
    s = (s + 1) & (cap - 1);
  }
  if (head < 0) return;
  matches++;
  double price = __builtin_nontemporal_load(&l_price[i]);
  double disc = __builtin_nontemporal_load(&l_disc[i]);
  atomicAdd(&slot_sums[s], price * (1.0 - disc));
}

// 2 rows/thread: int2/long2 nt loads (8/16 B per lane, the coalescing
// sweet spot) and two independent bloom/table lookups in flight per
// thread (the probes are random-load latency-bound at full occupancy)
__global__ void __launch_bounds__(BLOCK) k_q3_probe_agg_nt(
    uint64_t n, const int64_t *__restrict__ l_orderkey,
    const int32_t *__restrict__ l_shipdate,
    const double *__restrict__ l_price, const double *__restrict__ l_disc,
    int32_t date_gt, const int64_t *__restrict__ slot_keys,
    const int32_t *__restrict__ slot_head, uint64_t cap,
    double *__restrict__ slot_sums, uint64_t *__restrict__ match_count,
    const uint32_t *__restrict__ bloom, uint64_t bloom_mask) {
  uint32_t matches = 0;
  uint64_t npairs = n / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       p < npairs; p += stride) {
    uint64_t i = 2 * p;
    typedef int v2i __attribute__((ext_vector_type(2)));
    typedef long long v2l __attribute__((ext_vector_type(2)));
    v2i s2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2i *>(l_shipdate + i));
    v2l k2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2l *>(l_orderkey + i));
    bool pass0 = s2.x > date_gt, pass1 = s2.y > date_gt;
    if (bloom) {
      if (pass0) pass0 = bloom_test(bloom, bloom_mask, k2.x);
      if (pass1) pass1 = bloom_test(bloom, bloom_mask, k2.y);
    }
    q3_probe_row(k2.x, pass0, i, l_price, l_disc, slot_keys, slot_head,
                 cap, slot_sums, matches);
    q3_probe_row(k2.y, pass1, i + 1, l_price, l_disc, slot_keys, slot_head,
                 cap, slot_sums, matches);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    uint64_t i = n - 1;
    if (l_shipdate[i] > date_gt) {
      int64_t key = l_orderkey[i];
      bool pass = !bloom || bloom_test(bloom, bloom_mask, key);
      q3_probe_row(key, pass, i, l_price, l_disc, slot_keys, slot_head,
                   cap, slot_sums, matches);
    }
  }
  if (match_count) {
    __shared__ uint32_t lds[BLOCK / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1)
      matches += __shfl_down(matches, off);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) lds[wid] = matches;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint64_t t = 0;
      for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
      if (t) atomicAdd((unsigned long long *)match_count,
                       (unsigned long long)t);
    }
  }
}

// 4 rows/thread experiment: the probe is random-load latency-bound at
// full occupancy (r01 stall anatomy), so the remaining lever is MORE
// independent load chains per lane — 4 bloom tests + up to 4 table
// walks in flight instead of 2. A/B'd against the 2-row kernel on
// MI355X; see profiles/r02 notes for the verdict.
__global__ void __launch_bounds__(BLOCK) k_q3_probe_agg_nt4(
    uint64_t n, const int64_t *__restrict__ l_orderkey,
    const int32_t *__restrict__ l_shipdate,
    const double *__restrict__ l_price, const double *__restrict__ l_disc,
    int32_t date_gt, const int64_t *__restrict__ slot_keys,
    const int32_t *__restrict__ slot_head, uint64_t cap,
    double *__restrict__ slot_sums, uint64_t *__restrict__ match_count,
    const uint32_t *__restrict__ bloom, uint64_t bloom_mask) {
  uint32_t matches = 0;
  uint64_t nquads = n / 4;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t q = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       q < nquads; q += stride) {
    uint64_t i = 4 * q;
    typedef int v4i __attribute__((ext_vector_type(4)));
    typedef long long v4l __attribute__((ext_vector_type(4)));
    v4i s4 = __builtin_nontemporal_load(
        reinterpret_cast<const v4i *>(l_shipdate + i));
    v4l k4 = __builtin_nontemporal_load(
        reinterpret_cast<const v4l *>(l_orderkey + i));
    bool pass[4] = {s4.x > date_gt, s4.y > date_gt, s4.z > date_gt,
                    s4.w > date_gt};
    int64_t keys[4] = {k4.x, k4.y, k4.z, k4.w};
    if (bloom) {
#pragma unroll
      for (int j = 0; j < 4; j++)
        if (pass[j]) pass[j] = bloom_test(bloom, bloom_mask, keys[j]);
    }
#pragma unroll
    for (int j = 0; j < 4; j++)
      q3_probe_row(keys[j], pass[j], i + j, l_price, l_disc, slot_keys,
                   slot_head, cap, slot_sums, matches);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (uint64_t i = nquads * 4; i < n; i++)
      if (l_shipdate[i] > date_gt) {
        int64_t key = l_orderkey[i];
        bool pass = !bloom || bloom_test(bloom, bloom_mask, key);
        q3_probe_row(key, pass, i, l_price, l_disc, slot_keys, slot_head,
                     cap, slot_sums, matches);
      }
  if (match_count) {
    __shared__ uint32_t lds[BLOCK / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1)
      matches += __shfl_down(matches, off);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) lds[wid] = matches;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint64_t t = 0;
      for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
      if (t) atomicAdd((unsigned long long *)match_count,
                       (unsigned long long)t);
    }
  }
}
extern "C" int qk_q3_probe_agg_nt4(void *stream, uint64_t n,
                                   const int64_t *l_orderkey,
                                   const int32_t *l_shipdate,
                                   const double *l_price,
                                   const double *l_disc, int32_t date_gt,
                                   const int64_t *slot_keys,
                                   const int32_t *slot_head, uint64_t cap,
                                   double *slot_sums, uint64_t *match_count,
                                   const uint32_t *bloom,
                                   uint64_t bloom_mask) {
  if (!n) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_q3_probe_agg_nt4.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks =
      (uint32_t)qk_min_u64(MAX_BLOCKS, (n / 4 + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q3_probe_agg_nt4, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, l_orderkey, l_shipdate,
                     l_price, l_disc, date_gt, slot_keys, slot_head, cap,
                     slot_sums, match_count, bloom, bloom_mask);
  QK_TRY("qk_q3_probe_agg_nt4", hipGetLastError());
  return 0;
}

extern "C" int qk_q3_probe_agg(void *stream, uint64_t n,
                               const int64_t *l_orderkey,
                               const int32_t *l_shipdate,
                               const double *l_price, const double *l_disc,
                               int32_t date_gt, const int64_t *slot_keys,
                               const int32_t *slot_head, uint64_t cap,
                               double *slot_sums, uint64_t *match_count) {
  if (!n) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_q3_probe_agg.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q3_probe_agg, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, l_orderkey, l_shipdate, l_price,
                     l_disc, date_gt, slot_keys, slot_head, cap, slot_sums,
                     match_count);
  QK_TRY("qk_q3_probe_agg", hipGetLastError());
  return 0;
}
extern "C" int qk_q3_probe_agg_nt(void *stream, uint64_t n,
                                  const int64_t *l_orderkey,
                                  const int32_t *l_shipdate,
                                  const double *l_price, const double *l_disc,
                                  int32_t date_gt, const int64_t *slot_keys,
                                  const int32_t *slot_head, uint64_t cap,
                                  double *slot_sums, uint64_t *match_count,
                                  const uint32_t *bloom,
                                  uint64_t bloom_mask) {
  if (!n) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_q3_probe_agg_nt.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q3_probe_agg_nt, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, l_orderkey, l_shipdate, l_price,
                     l_disc, date_gt, slot_keys, slot_head, cap, slot_sums,
                     match_count, bloom, bloom_mask);
  QK_TRY("qk_q3_probe_agg_nt", hipGetLastError());
  return 0;
}

// extract groups: slots with a nonzero revenue sum; emits the group key,
// the orders build ROW (for o_orderdate/o_shippriority attach) and the sum.
// Two passes over contiguous slot chunks (per-block count -> single-block
// scan -> rank+scatter): no shared cursor word (a single cursor measured
// 8.6 ms on 1.2M groups; wave-aggregated claims still 5.7 ms).
__global__ void __launch_bounds__(BLOCK) k_extract_count(
    const int64_t *__restrict__ slot_keys, const double *__restrict__ slot_sums,
    uint64_t cap, uint64_t chunk, uint64_t *__restrict__ block_counts) {
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(cap, lo + chunk);
  uint32_t cnt = 0;
  for (uint64_t st = lo + threadIdx.x; st < hi; st += BLOCK)
    cnt += (slot_keys[st] != QK_JOIN_EMPTY && slot_sums[st] != 0.0) ? 1u : 0u;
  __shared__ uint32_t lds[BLOCK / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  if (lane == 0) lds[wid] = cnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t t = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
    block_counts[blockIdx.x] = t;
  }
}

__global__ void __launch_bounds__(BLOCK) k_extract_scatter(
    const int64_t *__restrict__ slot_keys, const int32_t *__restrict__ slot_head,
    const double *__restrict__ slot_sums, uint64_t cap, uint64_t chunk,
    const uint64_t *__restrict__ block_offsets, int64_t *__restrict__ out_keys,
    int32_t *__restrict__ out_row, double *__restrict__ out_sums,
    uint64_t out_cap) {
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(cap, lo + chunk);
  __shared__ uint64_t base;
  __shared__ uint32_t wave_tot[BLOCK / WAVE];
  if (threadIdx.x == 0) base = block_offsets[blockIdx.x];
  __syncthreads();
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  for (uint64_t s0 = lo; s0 < hi; s0 += BLOCK) {
    uint64_t st = s0 + threadIdx.x;
    bool live = st < hi && slot_keys[st] != QK_JOIN_EMPTY &&
                slot_sums[st] != 0.0;
    uint64_t mask = __ballot(live);
    uint32_t rank = __popcll(mask & ((1ULL << lane) - 1));
    if (lane == 0) wave_tot[wid] = __popcll(mask);
    __syncthreads();
    uint32_t wbase = 0;
    for (int w = 0; w < wid; w++) wbase += wave_tot[w];
    if (live) {
      uint64_t pos = base + wbase + rank;
      if (pos < out_cap) {
        out_keys[pos] = slot_keys[st];
        out_row[pos] = slot_head[st];
        out_sums[pos] = slot_sums[st];
      }
    }
    uint32_t btot = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) btot += wave_tot[w];
    __syncthreads();
    if (threadIdx.x == 0) base += btot;
    __syncthreads();
  }
}

extern "C" int qk_q3_extract(void *stream, const int64_t *slot_keys,
                             const int32_t *slot_head,
                             const double *slot_sums, uint64_t cap,
                             int64_t *out_keys, int32_t *out_row,
                             double *out_sums, uint64_t out_cap,
                             uint64_t *cursor) {
  uint64_t chunk = (cap + MAX_BLOCKS - 1) / MAX_BLOCKS;
  chunk = ((chunk + BLOCK - 1) / BLOCK) * BLOCK;
  uint32_t blocks = (uint32_t)((cap + chunk - 1) / chunk);
  static __thread uint64_t *scratch = nullptr;
  if (!scratch) QK_TRY("qk_q3_extract", hipMalloc(&scratch,
                       (MAX_BLOCKS + 1) * sizeof(uint64_t)));
  hipLaunchKernelGGL(k_extract_count, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, slot_keys, slot_sums, cap, chunk,
                     scratch);
  hipLaunchKernelGGL(k_scan_blocks, dim3(1), dim3(1), 0, (hipStream_t)stream,
                     (uint64_t)blocks, scratch, cursor);
  hipLaunchKernelGGL(k_extract_scatter, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, slot_keys, slot_head, slot_sums,
                     cap, chunk, scratch, out_keys, out_row, out_sums,
                     out_cap);
  QK_TRY("qk_q3_extract", hipGetLastError());
  return 0;
}

// ---- fused Q5 path -----------------------------------------------------
// Key -> i32-value hash tables (custkey->nationkey, orderkey->cust_nation,
// suppkey->nationkey) and one fused lineitem probe that joins both sides
// and accumulates revenue per nation. Same unique-build-key contract as Q3.

// insert (keys[i] -> vals[i]) where bit vals[i] of accept_mask is set
// (vals must be < 32 when accept_mask != ~0u; nationkeys are 0..24)
__global__ void __launch_bounds__(BLOCK) k_build_keyval_i32(
    uint64_t n, const int64_t *__restrict__ keys,
    const int32_t *__restrict__ vals, uint32_t accept_mask,
    int64_t *__restrict__ slot_keys, int32_t *__restrict__ slot_val,
    uint64_t cap, uint32_t *__restrict__ bloom, uint64_t bloom_mask) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t v = vals[i];
    if (accept_mask != 0xFFFFFFFFu &&
        !((accept_mask >> (v & 31)) & 1u && v >= 0 && v < 32))
      continue;
    int64_t key = keys[i];
    uint64_t s = slot_of(key, cap);
    for (;;) {
      int64_t cur = slot_keys[s];
      if (cur == key) break;
      if (cur == QK_JOIN_EMPTY) {
        int64_t prev = (int64_t)atomicCAS((unsigned long long *)&slot_keys[s],
                                          (unsigned long long)QK_JOIN_EMPTY,
                                          (unsigned long long)key);
        if (prev == QK_JOIN_EMPTY || prev == key) break;
      }
      s = (s + 1) & (cap - 1);
    }
    slot_val[s] = v;
    if (bloom) bloom_set(bloom, bloom_mask, key);
  }
}
extern "C" int qk_build_keyval_i32(void *stream, uint64_t n,
                                   const int64_t *keys, const int32_t *vals,
                                   uint32_t accept_mask, int64_t *slot_keys,
                                   int32_t *slot_val, uint64_t cap,
                                   uint32_t *bloom, uint64_t bloom_mask) {
  if (!n) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_build_keyval_i32.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_build_keyval_i32, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys, vals, accept_mask,
                     slot_keys, slot_val, cap, bloom, bloom_mask);
  QK_TRY("qk_build_keyval_i32", hipGetLastError());
  return 0;
}

// orders: where date_lo <= o_orderdate < date_hi and o_custkey hits the
// customer table, insert o_orderkey -> customer's nationkey.
// count-only mode when slot_keys == NULL (writes survivor count).
__global__ void __launch_bounds__(BLOCK) k_q5_build_orders(
    uint64_t n, const int64_t *__restrict__ o_orderkey,
    const int64_t *__restrict__ o_custkey,
    const int32_t *__restrict__ o_orderdate, int32_t date_lo, int32_t date_hi,
    const int64_t *__restrict__ cust_keys, const int32_t *__restrict__ cust_val,
    uint64_t cust_cap, int64_t *__restrict__ slot_keys,
    int32_t *__restrict__ slot_val, uint64_t cap,
    uint64_t *__restrict__ count, uint32_t *__restrict__ bloom,
    uint64_t bloom_mask, const uint32_t *__restrict__ cbloom,
    uint64_t cbloom_mask) {
  uint32_t cnt = 0;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t d = __builtin_nontemporal_load(&o_orderdate[i]);
    if (d < date_lo || d >= date_hi) continue;
    int64_t ck = __builtin_nontemporal_load(&o_custkey[i]);
    if (cbloom && !bloom_test(cbloom, cbloom_mask, ck)) continue;
    int32_t nat = probe_unique(cust_keys, cust_val, cust_cap, ck, nullptr);
    if (nat < 0) continue;
    cnt++;
    if (!slot_keys) continue;
    int64_t key = o_orderkey[i];
    uint64_t s = slot_of(key, cap);
    for (;;) {
      int64_t cur = slot_keys[s];
      if (cur == key) break;
      if (cur == QK_JOIN_EMPTY) {
        int64_t prev = (int64_t)atomicCAS((unsigned long long *)&slot_keys[s],
                                          (unsigned long long)QK_JOIN_EMPTY,
                                          (unsigned long long)key);
        if (prev == QK_JOIN_EMPTY || prev == key) break;
      }
      s = (s + 1) & (cap - 1);
    }
    slot_val[s] = nat;
    if (bloom) bloom_set(bloom, bloom_mask, key);
  }
  if (count) {
    __shared__ uint32_t lds[BLOCK / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) lds[wid] = cnt;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint64_t t = 0;
      for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
      if (t) atomicAdd((unsigned long long *)count, (unsigned long long)t);
    }
  }
}
extern "C" int qk_q5_build_orders(void *stream, uint64_t n,
                                  const int64_t *o_orderkey,
                                  const int64_t *o_custkey,
                                  const int32_t *o_orderdate, int32_t date_lo,
                                  int32_t date_hi, const int64_t *cust_keys,
                                  const int32_t *cust_val, uint64_t cust_cap,
                                  int64_t *slot_keys, int32_t *slot_val,
                                  uint64_t cap, uint64_t *count_dev,
                                  uint32_t *bloom, uint64_t bloom_mask,
                                  const uint32_t *cbloom,
                                  uint64_t cbloom_mask) {
  if (!n) return 0;
  if ((cust_cap & (cust_cap - 1)) || (slot_keys && (cap & (cap - 1))))
    return qk_fail("qk_q5_build_orders.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q5_build_orders, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, o_orderkey, o_custkey,
                     o_orderdate, date_lo, date_hi, cust_keys, cust_val,
                     cust_cap, slot_keys, slot_val, cap, count_dev, bloom,
                     bloom_mask, cbloom, cbloom_mask);
  QK_TRY("qk_q5_build_orders", hipGetLastError());
  return 0;
}

// lineitem: probe orders (-> customer nation) and supplier (-> supplier
// nation); where equal, accumulate revenue into out25[nation] via
// per-block LDS f64 accumulators
__global__ void __launch_bounds__(BLOCK) k_q5_probe_agg(
    uint64_t n, const int64_t *__restrict__ l_orderkey,
    const int64_t *__restrict__ l_suppkey,
    const double *__restrict__ l_price, const double *__restrict__ l_disc,
    const int64_t *__restrict__ ord_keys, const int32_t *__restrict__ ord_val,
    uint64_t ord_cap, const int64_t *__restrict__ supp_keys,
    const int32_t *__restrict__ supp_val, uint64_t supp_cap,
    double *__restrict__ out25, uint64_t *__restrict__ match_count) {
  __shared__ double lsum[32];
  __shared__ uint32_t lcnt;
  if (threadIdx.x < 32) lsum[threadIdx.x] = 0.0;
  if (threadIdx.x == 0) lcnt = 0;
  __syncthreads();
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t cnat = probe_unique(ord_keys, ord_val, ord_cap, l_orderkey[i],
                                nullptr);
    if (cnat < 0) continue;
    int32_t snat = probe_unique(supp_keys, supp_val, supp_cap, l_suppkey[i],
                                nullptr);
    if (snat != cnat) continue;
    atomicAdd(&lsum[cnat & 31], l_price[i] * (1.0 - l_disc[i]));
    if (match_count) atomicAdd(&lcnt, 1u);
  }
  __syncthreads();
  if (threadIdx.x < 32 && lsum[threadIdx.x] != 0.0)
    atomicAdd(&out25[threadIdx.x], lsum[threadIdx.x]);
  if (match_count && threadIdx.x == 0 && lcnt)
    atomicAdd((unsigned long long *)match_count, (unsigned long long)lcnt);
}
__device__ inline void q5_probe_row(
    int64_t okey, bool pass, uint64_t i,
    const int64_t *__restrict__ l_suppkey, const double *__restrict__ l_price,
    const double *__restrict__ l_disc, const int64_t *__restrict__ ord_keys,
    const int32_t *__restrict__ ord_val, uint64_t ord_cap,
    const int64_t *__restrict__ supp_keys, const int32_t *__restrict__ supp_val,
    uint64_t supp_cap, double *lsum, uint32_t *lcnt, bool count) {
  if (!pass) return;
  int32_t cnat = probe_unique(ord_keys, ord_val, ord_cap, okey, nullptr);
  if (cnat < 0) return;
  int64_t skey = __builtin_nontemporal_load(&l_suppkey[i]);
  int32_t snat = probe_unique(supp_keys, supp_val, supp_cap, skey, nullptr);
  if (snat != cnat) return;
  double price = __builtin_nontemporal_load(&l_price[i]);
  double disc = __builtin_nontemporal_load(&l_disc[i]);
  atomicAdd(&lsum[cnat & 31], price * (1.0 - disc));
  if (count) atomicAdd(lcnt, 1u);
}

// 2 rows/thread (16 B/lane key loads, two bloom/table lookups in flight)
__global__ void __launch_bounds__(BLOCK) k_q5_probe_agg_nt(
    uint64_t n, const int64_t *__restrict__ l_orderkey,
    const int64_t *__restrict__ l_suppkey,
    const double *__restrict__ l_price, const double *__restrict__ l_disc,
    const int64_t *__restrict__ ord_keys, const int32_t *__restrict__ ord_val,
    uint64_t ord_cap, const int64_t *__restrict__ supp_keys,
    const int32_t *__restrict__ supp_val, uint64_t supp_cap,
    double *__restrict__ out25, uint64_t *__restrict__ match_count,
    const uint32_t *__restrict__ bloom, uint64_t bloom_mask) {
  __shared__ double lsum[32];
  __shared__ uint32_t lcnt;
  if (threadIdx.x < 32) lsum[threadIdx.x] = 0.0;
  if (threadIdx.x == 0) lcnt = 0;
  __syncthreads();
  bool count = match_count != nullptr;
  uint64_t npairs = n / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       p < npairs; p += stride) {
    uint64_t i = 2 * p;
    typedef long long v2l __attribute__((ext_vector_type(2)));
    v2l k2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2l *>(l_orderkey + i));
    bool pass0 = !bloom || bloom_test(bloom, bloom_mask, k2.x);
    bool pass1 = !bloom || bloom_test(bloom, bloom_mask, k2.y);
    q5_probe_row(k2.x, pass0, i, l_suppkey, l_price, l_disc, ord_keys,
                 ord_val, ord_cap, supp_keys, supp_val, supp_cap, lsum,
                 &lcnt, count);
    q5_probe_row(k2.y, pass1, i + 1, l_suppkey, l_price, l_disc, ord_keys,
                 ord_val, ord_cap, supp_keys, supp_val, supp_cap, lsum,
                 &lcnt, count);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    uint64_t i = n - 1;
    int64_t okey = l_orderkey[i];
    bool pass = !bloom || bloom_test(bloom, bloom_mask, okey);
    q5_probe_row(okey, pass, i, l_suppkey, l_price, l_disc, ord_keys,
                 ord_val, ord_cap, supp_keys, supp_val, supp_cap, lsum,
                 &lcnt, count);
  }
  __syncthreads();
  if (threadIdx.x < 32 && lsum[threadIdx.x] != 0.0)
    atomicAdd(&out25[threadIdx.x], lsum[threadIdx.x]);
  if (match_count && threadIdx.x == 0 && lcnt)
    atomicAdd((unsigned long long *)match_count, (unsigned long long)lcnt);
}

extern "C" int qk_q5_probe_agg(void *stream, uint64_t n,
                               const int64_t *l_orderkey,
                               const int64_t *l_suppkey,
                               const double *l_price, const double *l_disc,
                               const int64_t *ord_keys, const int32_t *ord_val,
                               uint64_t ord_cap, const int64_t *supp_keys,
                               const int32_t *supp_val, uint64_t supp_cap,
                               double *out25, uint64_t *match_count) {
  if (!n) return 0;
  if ((ord_cap & (ord_cap - 1)) || (supp_cap & (supp_cap - 1)))
    return qk_fail("qk_q5_probe_agg.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q5_probe_agg, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, l_orderkey, l_suppkey, l_price,
                     l_disc, ord_keys, ord_val, ord_cap, supp_keys, supp_val,
                     supp_cap, out25, match_count);
  QK_TRY("qk_q5_probe_agg", hipGetLastError());
  return 0;
}
extern "C" int qk_q5_probe_agg_nt(void *stream, uint64_t n,
                                  const int64_t *l_orderkey,
                                  const int64_t *l_suppkey,
                                  const double *l_price, const double *l_disc,
                                  const int64_t *ord_keys,
                                  const int32_t *ord_val, uint64_t ord_cap,
                                  const int64_t *supp_keys,
                                  const int32_t *supp_val, uint64_t supp_cap,
                                  double *out25, uint64_t *match_count,
                                  const uint32_t *bloom,
                                  uint64_t bloom_mask) {
  if (!n) return 0;
  if ((ord_cap & (ord_cap - 1)) || (supp_cap & (supp_cap - 1)))
    return qk_fail("qk_q5_probe_agg_nt.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_q5_probe_agg_nt, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, l_orderkey, l_suppkey, l_price,
                     l_disc, ord_keys, ord_val, ord_cap, supp_keys, supp_val,
                     supp_cap, out25, match_count, bloom, bloom_mask);
  QK_TRY("qk_q5_probe_agg_nt", hipGetLastError());
  return 0;
}

// diagnostic: bloom-test-only pass (counts survivors) — isolates the
// Bloom lookup cost from the table walk for the split-probe decision
__global__ void __launch_bounds__(BLOCK) k_bloom_count(
    uint64_t n, const int64_t *__restrict__ keys,
    const uint32_t *__restrict__ bloom, uint64_t bloom_mask,
    uint64_t *__restrict__ out) {
  uint32_t cnt = 0;
  uint64_t npairs = n / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t p = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       p < npairs; p += stride) {
    uint64_t i = 2 * p;
    typedef long long v2l __attribute__((ext_vector_type(2)));
    v2l k2 = __builtin_nontemporal_load(
        reinterpret_cast<const v2l *>(keys + i));
    cnt += bloom_test(bloom, bloom_mask, k2.x) ? 1u : 0u;
    cnt += bloom_test(bloom, bloom_mask, k2.y) ? 1u : 0u;
  }
  __shared__ uint32_t lds[BLOCK / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  if (lane == 0) lds[wid] = cnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t t = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
    if (t) atomicAdd((unsigned long long *)out, (unsigned long long)t);
  }
}
extern "C" int qk_bloom_count(void *stream, uint64_t n, const int64_t *keys,
                              const uint32_t *bloom, uint64_t bloom_mask,
                              uint64_t *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n / 2 + BLOCK - 1) / BLOCK);
  if (!blocks) blocks = 1;
  hipLaunchKernelGGL(k_bloom_count, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys, bloom, bloom_mask, out);
  QK_TRY("qk_bloom_count", hipGetLastError());
  return 0;
}

__global__ void k_gen_supplier(uint64_t n, uint64_t row_offset, uint64_t seed,
                               int64_t *s_suppkey, int32_t *s_nationkey) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t row = row_offset + i;
    uint64_t h = splitmix64((seed ^ 0x5A5A5A5A5A5A5A5AULL) +
                            row * 0x9E3779B97F4A7C15ULL);
    if (s_suppkey) s_suppkey[i] = (int64_t)row + 1;
    if (s_nationkey) s_nationkey[i] = (int32_t)(h % 25);
  }
}
// Aux column generators for at-scale runs of the wider query shapes
// (device mirrors of oracle/tpch_gen.py's independent draws; separate
// hash streams per salt so they can be added to any table without
// disturbing the existing columns' streams):
//   uniform u8 code   — l_shipmode: rng.integers(0, 7)   (tpch_gen:231)
//   bernoulli u8 flag — o_comment_special: p=0.019       (tpch_gen:165)
//   uniform f64 cents — c_acctbal: U[-999.99, 9999.99]   (tpch_gen:250)
__global__ void k_gen_aux(uint64_t n, uint64_t row_offset, uint64_t seed,
                          uint64_t salt, int mode, int64_t a, int64_t b,
                          uint8_t *out_u8, double *out_f64) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t row = row_offset + i;
    uint64_t h = splitmix64((seed ^ salt) + row * 0x9E3779B97F4A7C15ULL);
    if (mode == 0)        // uniform code in [0, a)
      out_u8[i] = (uint8_t)(h % (uint64_t)a);
    else if (mode == 1)   // bernoulli, P = a / 1e6
      out_u8[i] = (h % 1000000ULL) < (uint64_t)a ? 1 : 0;
    else                  // uniform cents in [a, b] -> dollars
      out_f64[i] = (double)(a + (int64_t)(h % (uint64_t)(b - a + 1))) / 100.0;
  }
}
extern "C" int qk_gen_aux(void *stream, uint64_t n, uint64_t row_offset,
                          uint64_t seed, uint64_t salt, int mode, int64_t a,
                          int64_t b, uint8_t *out_u8, double *out_f64) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_gen_aux, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, row_offset, seed, salt, mode,
                     a, b, out_u8, out_f64);
  QK_TRY("qk_gen_aux", hipGetLastError());
  return 0;
}

extern "C" int qk_gen_supplier(void *stream, uint64_t n, uint64_t row_offset,
                               uint64_t seed, int64_t *s_suppkey,
                               int32_t *s_nationkey) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_gen_supplier, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, row_offset, seed, s_suppkey,
                     s_nationkey);
  QK_TRY("qk_gen_supplier", hipGetLastError());
  return 0;
}

// ---- group-by i64 -> f64 sums ----------------------------------------
// f64 atomic min/max via CAS on the bit pattern (ordered-compare loop)
__device__ inline void atomic_min_f64(double *addr, double v) {
  unsigned long long *a = (unsigned long long *)addr;
  unsigned long long cur = *a;
  while (__longlong_as_double((long long)cur) > v) {
    unsigned long long prev =
        atomicCAS(a, cur, (unsigned long long)__double_as_longlong(v));
    if (prev == cur) break;
    cur = prev;
  }
}
__device__ inline void atomic_max_f64(double *addr, double v) {
  unsigned long long *a = (unsigned long long *)addr;
  unsigned long long cur = *a;
  while (__longlong_as_double((long long)cur) < v) {
    unsigned long long prev =
        atomicCAS(a, cur, (unsigned long long)__double_as_longlong(v));
    if (prev == cur) break;
    cur = prev;
  }
}

// agg_ops[c]: 0 = SUM (slot init 0), 1 = MIN (init +inf), 2 = MAX (-inf)
__global__ void __launch_bounds__(BLOCK) k_groupby_sum(
    uint64_t n, const int64_t *__restrict__ keys, const double *const *vals,
    const int32_t *__restrict__ agg_ops, int nvals, int rstride,
    int64_t *__restrict__ table, uint64_t cap,
    uint64_t *__restrict__ n_inserted) {
  // INTERLEAVED slot records: [key | v0 .. v(nvals-1) | pad] of rstride
  // 8-byte words (pow2, so a record never straddles a 128 B line for
  // rstride <= 16): find-or-insert touches ONE random HBM line per row
  // instead of one line in the key array plus one per value array —
  // the dominant cost of high-cardinality dedups (Q13/Q16/Q18/Q21).
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  uint32_t my_inserts = 0;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t key = keys[i];
    uint64_t s = slot_of(key, cap);
    int64_t *rec;
    for (;;) {
      rec = table + s * (uint64_t)rstride;
      int64_t cur = rec[0];
      if (cur == key) break;
      if (cur == QK_JOIN_EMPTY) {
        int64_t prev = (int64_t)atomicCAS((unsigned long long *)rec,
                                          (unsigned long long)QK_JOIN_EMPTY,
                                          (unsigned long long)key);
        if (prev == QK_JOIN_EMPTY) {
          // fresh slot claimed: counted per-thread, block-reduced below
          // (a single cursor word serializes at ~88 atomics/us)
          my_inserts++;
          break;
        }
        if (prev == key) break;
      }
      s = (s + 1) & (cap - 1);
    }
    double *vslot = (double *)(rec + 1);
    for (int c = 0; c < nvals; c++) {
      double v = vals[c][i];
      int op = agg_ops ? agg_ops[c] : 0;
      if (op == 1)
        atomic_min_f64(&vslot[c], v);
      else if (op == 2)
        atomic_max_f64(&vslot[c], v);
      else
        atomicAdd(&vslot[c], v);
    }
  }
  if (n_inserted) {
    __shared__ uint32_t lds[BLOCK / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1)
      my_inserts += __shfl_down(my_inserts, off);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) lds[wid] = my_inserts;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint64_t t = 0;
      for (int w = 0; w < BLOCK / WAVE; w++) t += lds[w];
      if (t) atomicAdd((unsigned long long *)n_inserted,
                       (unsigned long long)t);
    }
  }
}
__global__ void k_groupby_init(uint64_t cap, int rstride, int nvals,
                               const double *__restrict__ inits,
                               int64_t *__restrict__ table) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t s = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; s < cap;
       s += stride) {
    int64_t *rec = table + s * (uint64_t)rstride;
    rec[0] = QK_JOIN_EMPTY;
    double *v = (double *)(rec + 1);
    for (int c = 0; c < nvals; c++) v[c] = inits[c];
  }
}
extern "C" int qk_groupby_init(void *stream, int64_t *table, uint64_t cap,
                               int rstride, int nvals,
                               const double *inits_dev) {
  if (!cap) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (cap + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_groupby_init, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, cap, rstride, nvals, inits_dev,
                     table);
  QK_TRY("qk_groupby_init", hipGetLastError());
  return 0;
}
extern "C" int qk_groupby_i64_sum(void *stream, uint64_t n, const int64_t *keys,
                                  const void *vals_dev,
                                  const int32_t *agg_ops_dev, int nvals,
                                  int rstride, int64_t *table, uint64_t cap,
                                  uint64_t *n_inserted) {
  if (!n) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_groupby_i64_sum.cap_pow2", hipErrorInvalidValue);
  if (rstride < 1 + nvals || (rstride & (rstride - 1)))
    return qk_fail("qk_groupby_i64_sum.rstride", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_groupby_sum, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys,
                     (const double *const *)vals_dev, agg_ops_dev, nvals,
                     rstride, table, cap, n_inserted);
  QK_TRY("qk_groupby_i64_sum", hipGetLastError());
  return 0;
}

__global__ void k_fill_f64(double *dst, double v, uint64_t n) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dst[i] = v;
}
extern "C" int qk_fill_f64(void *stream, double *dst, double v, uint64_t n) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_fill_f64, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, dst, v, n);
  QK_TRY("qk_fill_f64", hipGetLastError());
  return 0;
}

__global__ void __launch_bounds__(BLOCK) k_groupby_extract(
    const int64_t *__restrict__ table, int rstride, int nvals, uint64_t cap,
    int64_t *__restrict__ out_keys, double *__restrict__ out_sums,
    uint64_t out_cap, uint64_t *__restrict__ cursor) {
  // wave-aggregated output cursor (one atomicAdd per wave per step);
  // reads INTERLEAVED slot records, writes COLUMN-MAJOR outputs
  // (consumers index out_sums[c * out_cap + pos])
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  int lane = (int)(threadIdx.x & (WAVE - 1));
  for (uint64_t s = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; s < cap;
       s += stride) {
    const int64_t *rec = table + s * (uint64_t)rstride;
    int64_t key = rec[0];
    bool has = key != QK_JOIN_EMPTY;
    uint64_t mask = __ballot(has);
    if (!mask) continue;
    int src = __ffsll((unsigned long long)mask) - 1;
    unsigned long long base = 0;
    if (lane == src)
      base = atomicAdd((unsigned long long *)cursor,
                       (unsigned long long)__popcll(mask));
    base = __shfl(base, src);
    if (has) {
      uint64_t pos = base + __popcll(mask & ((1ULL << lane) - 1));
      if (pos < out_cap) {
        out_keys[pos] = key;
        const double *v = (const double *)(rec + 1);
        for (int c = 0; c < nvals; c++)
          out_sums[(uint64_t)c * out_cap + pos] = v[c];
      }
    }
  }
}
extern "C" int qk_groupby_extract(void *stream, const int64_t *table,
                                  int rstride, int nvals, uint64_t cap,
                                  int64_t *out_keys, double *out_sums,
                                  uint64_t out_cap, uint64_t *cursor) {
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (cap + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_groupby_extract, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, table, rstride, nvals, cap,
                     out_keys, out_sums, out_cap, cursor);
  QK_TRY("qk_groupby_extract", hipGetLastError());
  return 0;
}

// Thresholded extract (HAVING clauses): only groups with
// sums[col] > threshold are compacted — Q18 qualifies a handful of its
// ~n_orders groups, so the d2h stays tiny instead of GBs.
__global__ void __launch_bounds__(BLOCK) k_groupby_extract_gt(
    const int64_t *__restrict__ table, int rstride, int nvals, uint64_t cap,
    int col, double threshold, int64_t *__restrict__ out_keys,
    double *__restrict__ out_sums, uint64_t out_cap,
    uint64_t *__restrict__ cursor) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  int lane = (int)(threadIdx.x & (WAVE - 1));
  for (uint64_t s = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; s < cap;
       s += stride) {
    const int64_t *rec = table + s * (uint64_t)rstride;
    int64_t key = rec[0];
    const double *v = (const double *)(rec + 1);
    bool has = key != QK_JOIN_EMPTY && v[col] > threshold;
    uint64_t mask = __ballot(has);
    if (!mask) continue;
    int src = __ffsll((unsigned long long)mask) - 1;
    unsigned long long base = 0;
    if (lane == src)
      base = atomicAdd((unsigned long long *)cursor,
                       (unsigned long long)__popcll(mask));
    base = __shfl(base, src);
    if (has) {
      uint64_t pos = base + __popcll(mask & ((1ULL << lane) - 1));
      if (pos < out_cap) {
        out_keys[pos] = key;
        for (int c = 0; c < nvals; c++)
          out_sums[(uint64_t)c * out_cap + pos] = v[c];
      }
    }
  }
}
extern "C" int qk_groupby_extract_gt(void *stream, const int64_t *table,
                                     int rstride, int nvals, uint64_t cap,
                                     int col, double threshold,
                                     int64_t *out_keys, double *out_sums,
                                     uint64_t out_cap, uint64_t *cursor) {
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (cap + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_groupby_extract_gt, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, table, rstride, nvals, cap, col,
                     threshold, out_keys, out_sums, out_cap, cursor);
  QK_TRY("qk_groupby_extract_gt", hipGetLastError());
  return 0;
}

// ---- stable radix sort (u64 keys, u32 payload) -------------------------
// Replaces the sort the reference delegates to polars (SuperFastSort,
// sql_executors.py:88-187; build-side sort :369; order-by tails). LSD,
// 8 bits/pass, pass count from the key range; stability per pass via
// (block-major offsets) x (wave-serialized in-wave ballot ranks).

#define RADIX 256

__global__ void __launch_bounds__(BLOCK) k_radix_count(
    uint64_t n, const uint64_t *__restrict__ keys, int shift, uint64_t chunk,
    uint32_t *__restrict__ counts /* [RADIX][nblocks] */, uint32_t nblocks) {
  __shared__ uint32_t hist[RADIX];
  for (int d = threadIdx.x; d < RADIX; d += BLOCK) hist[d] = 0;
  __syncthreads();
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  for (uint64_t i = lo + threadIdx.x; i < hi; i += BLOCK)
    atomicAdd(&hist[(uint32_t)(keys[i] >> shift) & 255u], 1u);
  __syncthreads();
  for (int d = threadIdx.x; d < RADIX; d += BLOCK)
    counts[(uint64_t)d * nblocks + blockIdx.x] = hist[d];
}

// one 256-thread block: thread d owns digit d's row of per-block counts;
// produces exclusive global bases per (digit, block)
__global__ void __launch_bounds__(RADIX) k_radix_scan(
    uint32_t nblocks, uint32_t *__restrict__ counts,
    uint64_t *__restrict__ bases /* [RADIX][nblocks] */) {
  __shared__ uint64_t tot[RADIX];
  int d = threadIdx.x;
  uint64_t acc = 0;
  for (uint32_t b = 0; b < nblocks; b++) {
    uint32_t c = counts[(uint64_t)d * nblocks + b];
    bases[(uint64_t)d * nblocks + b] = acc;
    acc += c;
  }
  tot[d] = acc;
  __syncthreads();
  // exclusive scan of 256 digit totals (simple doubling scan)
  for (int off = 1; off < RADIX; off <<= 1) {
    uint64_t v = d >= off ? tot[d - off] : 0;
    __syncthreads();
    tot[d] += v;
    __syncthreads();
  }
  uint64_t base = d == 0 ? 0 : tot[d - 1];
  for (uint32_t b = 0; b < nblocks; b++)
    bases[(uint64_t)d * nblocks + b] += base;
}

__global__ void __launch_bounds__(BLOCK) k_radix_scatter(
    uint64_t n, const uint64_t *__restrict__ keys_in,
    const uint32_t *__restrict__ pay_in, int shift, uint64_t chunk,
    const uint64_t *__restrict__ bases, uint32_t nblocks,
    uint64_t *__restrict__ keys_out, uint32_t *__restrict__ pay_out) {
  __shared__ uint64_t cursor[RADIX];
  for (int d = threadIdx.x; d < RADIX; d += BLOCK)
    cursor[d] = bases[(uint64_t)d * nblocks + blockIdx.x];
  __syncthreads();
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  for (uint64_t t0 = lo; t0 < hi; t0 += BLOCK) {
    uint64_t i = t0 + threadIdx.x;
    bool valid = i < hi;
    uint64_t k = valid ? keys_in[i] : ~0ULL;
    uint32_t pay = valid ? pay_in[i] : 0;
    uint32_t d8 = (uint32_t)(k >> shift) & 255u;
    // waves take turns IN ORDER: block stability = wave order x lane order
    for (int w = 0; w < BLOCK / WAVE; w++) {
      if (wid == w) {
        uint64_t same = __ballot(valid);
        for (int j = 0; j < 8; j++) {
          uint64_t bj = __ballot((d8 >> j) & 1u);
          same &= ((d8 >> j) & 1u) ? bj : ~bj;
        }
        uint32_t before = __popcll(same & ((1ULL << lane) - 1));
        uint64_t base_leader = 0;
        if (valid && before == 0)
          base_leader = atomicAdd((unsigned long long *)&cursor[d8],
                                  (unsigned long long)__popcll(same));
        int leader = __ffsll((unsigned long long)same) - 1;
        uint64_t base = __shfl(base_leader, leader < 0 ? 0 : leader);
        if (valid) {
          keys_out[base + before] = k;
          pay_out[base + before] = pay;
        }
      }
      __syncthreads();
    }
  }
}

__global__ void k_reduce_max_u64(uint64_t n, const uint64_t *__restrict__ in,
                                 unsigned long long *__restrict__ out) {
  uint64_t m = 0;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    m = in[i] > m ? in[i] : m;
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    uint64_t o = __shfl_down(m, off);
    m = o > m ? o : m;
  }
  __shared__ uint64_t lds[BLOCK / WAVE];
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  if (lane == 0) lds[wid] = m;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < BLOCK / WAVE; w++) m = lds[w] > m ? lds[w] : m;
    atomicMax(out, (unsigned long long)m);
  }
}

// order-preserving key maps: sort ascending on the u64 image == ascending
// on the source type
__global__ void k_map_f64_u64(uint64_t n, const double *__restrict__ in,
                              uint64_t *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t b = (uint64_t)__double_as_longlong(in[i]);
    out[i] = (b & 0x8000000000000000ULL) ? ~b : (b | 0x8000000000000000ULL);
  }
}
__global__ void k_map_i64_u64(uint64_t n, const int64_t *__restrict__ in,
                              uint64_t *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (uint64_t)in[i] ^ 0x8000000000000000ULL;
}
__global__ void k_bnot_u64(uint64_t n, uint64_t *__restrict__ x) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    x[i] = ~x[i];
}
__global__ void k_iota_u32(uint64_t n, uint32_t *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (uint32_t)i;
}

extern "C" int qk_map_f64_u64(void *stream, uint64_t n, const double *in,
                              uint64_t *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_map_f64_u64, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, in, out);
  QK_TRY("qk_map_f64_u64", hipGetLastError());
  return 0;
}
extern "C" int qk_map_i64_u64(void *stream, uint64_t n, const int64_t *in,
                              uint64_t *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_map_i64_u64, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, in, out);
  QK_TRY("qk_map_i64_u64", hipGetLastError());
  return 0;
}
extern "C" int qk_bnot_u64(void *stream, uint64_t n, uint64_t *x) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_bnot_u64, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, x);
  QK_TRY("qk_bnot_u64", hipGetLastError());
  return 0;
}
extern "C" int qk_iota_u32(void *stream, uint64_t n, uint32_t *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_iota_u32, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, out);
  QK_TRY("qk_iota_u32", hipGetLastError());
  return 0;
}

/* Stable ascending sort of (keys, payload) in place. keys_tmp/pay_tmp are
 * caller scratch of the same sizes. npasses -1 = derive from max key. */
extern "C" int qk_sort_pairs_u64(void *stream, uint64_t n, uint64_t *keys,
                                 uint32_t *pay, uint64_t *keys_tmp,
                                 uint32_t *pay_tmp, int npasses) {
  if (n < 2) return 0;
  uint64_t chunk = (n + MAX_BLOCKS - 1) / MAX_BLOCKS;
  chunk = ((chunk + BLOCK - 1) / BLOCK) * BLOCK;
  uint32_t nblocks = (uint32_t)((n + chunk - 1) / chunk);

  static __thread uint32_t *counts = nullptr;
  static __thread uint64_t *bases = nullptr;
  static __thread unsigned long long *dmax = nullptr;
  if (!counts) {
    QK_TRY("qk_sort_pairs_u64",
           hipMalloc(&counts, (uint64_t)RADIX * MAX_BLOCKS * 4));
    QK_TRY("qk_sort_pairs_u64",
           hipMalloc(&bases, (uint64_t)RADIX * MAX_BLOCKS * 8));
    QK_TRY("qk_sort_pairs_u64", hipMalloc(&dmax, 8));
  }
  if (npasses < 0) {
    QK_TRY("qk_sort_pairs_u64",
           hipMemsetAsync(dmax, 0, 8, (hipStream_t)stream));
    uint32_t b2 = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
    hipLaunchKernelGGL(k_reduce_max_u64, dim3(b2), dim3(BLOCK), 0,
                       (hipStream_t)stream, n, keys, dmax);
    unsigned long long hmax = 0;
    QK_TRY("qk_sort_pairs_u64",
           hipMemcpyAsync(&hmax, dmax, 8, hipMemcpyDeviceToHost,
                          (hipStream_t)stream));
    QK_TRY("qk_sort_pairs_u64", hipStreamSynchronize((hipStream_t)stream));
    npasses = 0;
    while (hmax) {
      npasses++;
      hmax >>= 8;
    }
    if (!npasses) npasses = 1;
  }
  uint64_t *ki = keys, *ko = keys_tmp;
  uint32_t *pi = pay, *po = pay_tmp;
  for (int p = 0; p < npasses; p++) {
    int shift = 8 * p;
    hipLaunchKernelGGL(k_radix_count, dim3(nblocks), dim3(BLOCK), 0,
                       (hipStream_t)stream, n, ki, shift, chunk, counts,
                       nblocks);
    hipLaunchKernelGGL(k_radix_scan, dim3(1), dim3(RADIX), 0,
                       (hipStream_t)stream, nblocks, counts, bases);
    hipLaunchKernelGGL(k_radix_scatter, dim3(nblocks), dim3(BLOCK), 0,
                       (hipStream_t)stream, n, ki, pi, shift, chunk, bases,
                       nblocks, ko, po);
    uint64_t *tk = ki; ki = ko; ko = tk;
    uint32_t *tp = pi; pi = po; po = tp;
  }
  QK_TRY("qk_sort_pairs_u64", hipGetLastError());
  if (ki != keys) {  // odd pass count: copy back
    QK_TRY("qk_sort_pairs_u64",
           hipMemcpyAsync(keys, ki, n * 8, hipMemcpyDeviceToDevice,
                          (hipStream_t)stream));
    QK_TRY("qk_sort_pairs_u64",
           hipMemcpyAsync(pay, pi, n * 4, hipMemcpyDeviceToDevice,
                          (hipStream_t)stream));
  }
  return 0;
}

// ---- RCCL exchange ----------------------------------------------------
// Grouped send/recv all-to-allv over xGMI (direct per-peer, not a ring);
// replaces the reference's Flight-based shuffle (core.py:276-376) for the
// repartition step. See include/quokka_amd.h.
#include <rccl/rccl.h>

static int qk_nccl_fail(const char *where, ncclResult_t r) {
  snprintf(g_err, sizeof(g_err), "%s: %s", where, ncclGetErrorString(r));
  g_err_set = 1;
  return 1000 + (int)r;
}
#define QK_NCCL(where, expr)                                                   \
  do {                                                                         \
    ncclResult_t _r = (expr);                                                  \
    if (_r != ncclSuccess) return qk_nccl_fail(where, _r);                     \
  } while (0)

extern "C" int qk_comm_unique_id(uint8_t out[QK_UID_BYTES]) {
  static_assert(sizeof(ncclUniqueId) == QK_UID_BYTES, "uid size");
  QK_NCCL("qk_comm_unique_id", ncclGetUniqueId((ncclUniqueId *)out));
  return 0;
}
extern "C" int qk_comm_init(int rank, int world,
                            const uint8_t uid[QK_UID_BYTES],
                            void **comm_out) {
  ncclUniqueId id;
  memcpy(&id, uid, sizeof(id));
  ncclComm_t comm;
  QK_NCCL("qk_comm_init", ncclCommInitRank(&comm, world, id, rank));
  *comm_out = (void *)comm;
  return 0;
}
extern "C" int qk_comm_destroy(void *comm) {
  QK_NCCL("qk_comm_destroy", ncclCommDestroy((ncclComm_t)comm));
  return 0;
}
// ---- GPU Parquet page decode (include/quokka_amd.h for the contract) --
// Host-built tile tables give each workgroup one bounded slice; all
// value bytes stay in HBM (the host touched only Thrift headers and RLE
// run descriptors).
template <typename T>
__device__ inline void pq_copy_vals(const uint8_t *__restrict__ s,
                                    uint8_t *__restrict__ d, uint64_t cnt) {
  for (uint64_t i = threadIdx.x; i < cnt; i += BLOCK) {
    T v;
    __builtin_memcpy(&v, s + i * sizeof(T), sizeof(T));  // src unaligned
    *reinterpret_cast<T *>(d + i * sizeof(T)) = v;       // dst aligned
  }
}
__global__ void __launch_bounds__(BLOCK) k_pq_plain_copy(
    uint64_t ntiles, const uint64_t *__restrict__ tiles,
    const uint8_t *__restrict__ src, uint8_t *__restrict__ dst,
    uint32_t es) {
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    uint64_t so = tiles[t * 3], eo = tiles[t * 3 + 1], cnt = tiles[t * 3 + 2];
    const uint8_t *s = src + so;
    uint8_t *d = dst + eo * es;
    switch (es) {
      case 8: pq_copy_vals<uint64_t>(s, d, cnt); break;
      case 4: pq_copy_vals<uint32_t>(s, d, cnt); break;
      case 2: pq_copy_vals<uint16_t>(s, d, cnt); break;
      default: pq_copy_vals<uint8_t>(s, d, cnt); break;
    }
  }
}
extern "C" int qk_pq_plain_copy(void *stream, uint64_t ntiles,
                                const uint64_t *tiles, const void *src_bytes,
                                void *dst, uint32_t elem_size) {
  if (!ntiles) return 0;
  if (elem_size != 1 && elem_size != 2 && elem_size != 4 && elem_size != 8)
    return qk_fail("qk_pq_plain_copy.elem_size", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, ntiles);
  hipLaunchKernelGGL(k_pq_plain_copy, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, ntiles, tiles,
                     (const uint8_t *)src_bytes, (uint8_t *)dst, elem_size);
  QK_TRY("qk_pq_plain_copy", hipGetLastError());
  return 0;
}
// One WAVE per page: lane 0 parses the RLE/bit-packed run headers
// (varints — inherently sequential within a page), broadcasts via
// __shfl (wave-lockstep, no barriers), and all 64 lanes expand the run.
// Low-cardinality columns emit millions of tiny runs (measured 1.3M for
// a 24M-row 2-value column) — parsing them in host Python cost seconds;
// per-page parsing on-device leaves the host with ~1 descriptor per page.
__global__ void __launch_bounds__(WAVE) k_pq_rle_pages(
    uint64_t npages, const uint64_t *__restrict__ ents,
    const uint8_t *__restrict__ src, uint32_t *__restrict__ out) {
  for (uint64_t pg = blockIdx.x; pg < npages; pg += gridDim.x) {
    const uint64_t *E = ents + pg * 6;
    uint64_t pos = E[0], end = E[1], dst = E[2], remaining = E[3];
    uint32_t bw = (uint32_t)E[4];
    uint32_t idx_off = (uint32_t)E[5];
    if (bw == 0) {                       // all indices are 0, no stream
      for (uint64_t i = threadIdx.x; i < remaining; i += WAVE)
        out[dst + i] = idx_off;
      continue;
    }
    uint64_t mask = (1ull << bw) - 1;
    while (remaining && pos < end) {
      uint64_t h = 0, npos = pos, val = 0;
      if (threadIdx.x == 0) {
        uint32_t shift = 0;
        uint8_t b;
        do {
          b = src[npos++];
          h |= (uint64_t)(b & 0x7F) << shift;
          shift += 7;
        } while (b & 0x80);
        if (!(h & 1)) {
          uint32_t nb = (bw + 7) >> 3;
          for (uint32_t i = 0; i < nb; i++)
            val |= (uint64_t)src[npos++] << (8 * i);
        }
      }
      h = (uint64_t)__shfl((long long)h, 0);
      npos = (uint64_t)__shfl((long long)npos, 0);
      uint64_t n;
      if (h & 1) {
        uint64_t ngroups = h >> 1;
        n = qk_min_u64(ngroups * 8, remaining);
        uint64_t bitbase = npos * 8;
        for (uint64_t i = threadIdx.x; i < n; i += WAVE) {
          uint64_t bit = bitbase + i * (uint64_t)bw;
          uint64_t w;
          __builtin_memcpy(&w, src + (bit >> 3), 8);  // unaligned, slack
          out[dst + i] = (uint32_t)((w >> (bit & 7)) & mask) + idx_off;
        }
        npos += ngroups * bw;
      } else {
        val = (uint64_t)__shfl((long long)val, 0);
        n = qk_min_u64(h >> 1, remaining);
        for (uint64_t i = threadIdx.x; i < n; i += WAVE)
          out[dst + i] = (uint32_t)val + idx_off;
      }
      dst += n;
      remaining -= n;
      pos = npos;
    }
  }
}
extern "C" int qk_pq_rle_pages(void *stream, uint64_t npages,
                               const uint64_t *ents, const uint8_t *src_bytes,
                               uint32_t *out) {
  if (!npages) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(64 * 1024, npages);
  hipLaunchKernelGGL(k_pq_rle_pages, dim3(blocks), dim3(WAVE), 0,
                     (hipStream_t)stream, npages, ents, src_bytes, out);
  QK_TRY("qk_pq_rle_pages", hipGetLastError());
  return 0;
}

// ---- device string dictionary ------------------------------------------
// General variable-width string keys for join/group-by (the reference
// joins/groups on arbitrary key types via polars, sql_executors.py:
// 325-377/:556-599; round 1 supported only <=256-entry host dicts).
// Design: an open-addressing table keyed by a 64-bit byte hash with EXACT
// byte verification against an on-device string ARENA (first occurrence
// of each distinct string is copied in), assigning dense u32 codes that
// stay consistent across batches because the table and arena persist.
// Equal strings => equal codes, distinct => distinct (bytes verified, no
// hash-collision false sharing), so string joins/group-bys reduce to the
// existing integer kernels over the codes.
//
// slot_hash: u64[cap], 0 = empty (real hash 0 remapped to 1)
// slot_code: i32[cap], -1 until the owning lane publishes
// code_off/code_len: per-code arena offset/length (cap_codes entries)
// arena/arena_cursor: byte arena; HOST guarantees capacity >= cursor +
//   batch bytes before each launch (no mid-kernel growth path needed)
__device__ inline uint64_t qk_str_hash(const uint8_t *p, uint32_t len) {
  uint64_t h = 0x9E3779B97F4A7C15ULL ^ len;
  uint32_t i = 0;
  for (; i + 8 <= len; i += 8) {
    uint64_t w;
    __builtin_memcpy(&w, p + i, 8);
    h = splitmix64(h ^ w);
  }
  uint64_t tail = 0;
  for (uint32_t k = 0; i < len; i++, k++)
    tail |= (uint64_t)p[i] << (8 * k);
  h = splitmix64(h ^ tail);
  return h ? h : 1;
}

__global__ void __launch_bounds__(BLOCK) k_str_dict_encode(
    uint64_t n, const int64_t *__restrict__ offsets,
    const uint8_t *__restrict__ bytes, uint64_t *slot_hash,
    int32_t *slot_code, uint64_t cap, uint64_t *code_off,
    uint32_t *code_len, uint8_t *arena, uint64_t *arena_cursor,
    uint32_t *counter, uint32_t *__restrict__ out_codes) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t o0 = offsets[i], o1 = offsets[i + 1];
    uint32_t len = (uint32_t)(o1 - o0);
    const uint8_t *p = bytes + o0;
    uint64_t h = qk_str_hash(p, len);
    uint64_t s = h & (cap - 1);
    for (;;) {
      uint64_t cur = __atomic_load_n((unsigned long long *)&slot_hash[s],
                                     __ATOMIC_RELAXED);
      if (cur == 0) {
        uint64_t prev = atomicCAS((unsigned long long *)&slot_hash[s],
                                  0ULL, (unsigned long long)h);
        if (prev == 0) {
          // this lane owns the slot: copy bytes, assign the code, then
          // PUBLISH via slot_code (readers spin until != -1)
          uint64_t aoff =
              atomicAdd((unsigned long long *)arena_cursor,
                        (unsigned long long)len);
          for (uint32_t k = 0; k < len; k++) arena[aoff + k] = p[k];
          uint32_t code = atomicAdd(counter, 1u);
          code_off[code] = aoff;
          code_len[code] = len;
          __threadfence();
          __atomic_store_n(&slot_code[s], (int32_t)code,
                           __ATOMIC_RELEASE);
          out_codes[i] = code;
          break;
        }
        cur = prev;
      }
      if (cur == h) {
        int32_t code;
        do {
          code = __atomic_load_n(&slot_code[s], __ATOMIC_ACQUIRE);
        } while (code < 0);
        if (code_len[code] == len) {
          const uint8_t *q = arena + code_off[code];
          uint32_t k = 0;
          while (k < len && q[k] == p[k]) k++;
          if (k == len) {
            out_codes[i] = (uint32_t)code;
            break;
          }
        }
        // same 64-bit hash, different bytes: keep probing
      }
      s = (s + 1) & (cap - 1);
    }
  }
}
extern "C" int qk_str_dict_encode(void *stream, uint64_t n,
                                  const int64_t *offsets,
                                  const uint8_t *bytes, uint64_t *slot_hash,
                                  int32_t *slot_code, uint64_t cap,
                                  uint64_t *code_off, uint32_t *code_len,
                                  uint8_t *arena, uint64_t *arena_cursor,
                                  uint32_t *counter, uint32_t *out_codes) {
  if (!n) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_str_dict_encode.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_str_dict_encode, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, offsets, bytes, slot_hash,
                     slot_code, cap, code_off, code_len, arena,
                     arena_cursor, counter, out_codes);
  QK_TRY("qk_str_dict_encode", hipGetLastError());
  return 0;
}

// Growth path: re-seat (hash, code) pairs into a larger table. Codes and
// the arena are immutable — previously returned codes stay valid.
__global__ void __launch_bounds__(BLOCK) k_str_dict_rehash(
    uint32_t ncodes, const uint64_t *__restrict__ code_off,
    const uint32_t *__restrict__ code_len, const uint8_t *__restrict__ arena,
    uint64_t *slot_hash, int32_t *slot_code, uint64_t cap) {
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t c = blockIdx.x * blockDim.x + threadIdx.x; c < ncodes;
       c += stride) {
    uint64_t h = qk_str_hash(arena + code_off[c], code_len[c]);
    uint64_t s = h & (cap - 1);
    for (;;) {
      uint64_t prev = atomicCAS((unsigned long long *)&slot_hash[s], 0ULL,
                                (unsigned long long)h);
      if (prev == 0) {
        __threadfence();
        __atomic_store_n(&slot_code[s], (int32_t)c, __ATOMIC_RELEASE);
        break;
      }
      s = (s + 1) & (cap - 1);
    }
  }
}
extern "C" int qk_str_dict_rehash(void *stream, uint32_t ncodes,
                                  const uint64_t *code_off,
                                  const uint32_t *code_len,
                                  const uint8_t *arena, uint64_t *slot_hash,
                                  int32_t *slot_code, uint64_t cap) {
  if (!ncodes) return 0;
  if (cap & (cap - 1))
    return qk_fail("qk_str_dict_rehash.cap_pow2", hipErrorInvalidValue);
  uint32_t blocks =
      (uint32_t)qk_min_u64(MAX_BLOCKS, (ncodes + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_str_dict_rehash, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, ncodes, code_off, code_len, arena,
                     slot_hash, slot_code, cap);
  QK_TRY("qk_str_dict_rehash", hipGetLastError());
  return 0;
}

__global__ void __launch_bounds__(BLOCK) k_i64_combine(
    uint64_t n, const int64_t *__restrict__ x, const int64_t *__restrict__ y,
    int64_t scale, int64_t *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = x[i] * scale + y[i];
}
// Composite i64 key: out = x * scale + y (e.g. (partkey, suppkey) group
// keys for Q20/Q21-style per-pair aggregates — exact for |x*scale+y| < 2^63)
extern "C" int qk_i64_combine(void *stream, uint64_t n, const int64_t *x,
                              const int64_t *y, int64_t scale, int64_t *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_i64_combine, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, x, y, scale, out);
  QK_TRY("qk_i64_combine", hipGetLastError());
  return 0;
}

__global__ void __launch_bounds__(BLOCK) k_i64_shr(
    uint64_t n, const int64_t *__restrict__ x, int shift,
    int64_t *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = x[i] >> shift;
}
// Arithmetic shift right (decompose pow2-scaled composite keys on device)
extern "C" int qk_i64_shr(void *stream, uint64_t n, const int64_t *x,
                          int shift, int64_t *out) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_i64_shr, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, x, shift, out);
  QK_TRY("qk_i64_shr", hipGetLastError());
  return 0;
}

extern "C" int qk_d2d(void *dst, const void *src, uint64_t nbytes) {
  QK_TRY("qk_d2d", hipMemcpy(dst, src, nbytes, hipMemcpyDeviceToDevice));
  return 0;
}

// ---- GPU snappy page decompression -------------------------------------
// Parquet SNAPPY-compressed pages (the reference reads them transparently
// through pyarrow, unordered_readers.py:51; real TPC-H datasets ship
// compressed). One WAVE per page: lane 0 walks the snappy element stream
// (format: varint uncompressed length, then tagged elements — tag low
// 2 bits: 0 literal, 1 copy with 11-bit offset, 2 copy with 2-byte
// offset, 3 copy with 4-byte offset) and broadcasts each element; all 64
// lanes execute the copy cooperatively. Match copies may read bytes this
// wave wrote earlier, so a single-wave __syncthreads() (block == one
// wave) orders them; literals read only the source and skip the barrier.
// Pages decode independently — thousands in flight fill the chip.
//
// After decompression, v1 data pages carry their definition-levels block
// INSIDE the decompressed bytes; for max_def == 1 lane 0 verifies
// "no nulls" (bit-packed groups all 0xFF / fill runs == 1, mirroring
// parquet_gpu._check_levels_v1) and reports where the values start.
//
// desc per page (8 u64): [src_off, src_len, dst_off, uncompressed_len,
//   mode (0 = raw, 1 = verify v1 def-levels for max_def 1), num_values,
//   0, 0]
// out per page (4 i64): [data_off_rel (after levels), err (0 ok,
//   1 corrupt stream, 2 length mismatch, 3 nulls present),
//   first_byte_after_levels (the RLE bit-width byte), 0]
__global__ void k_snappy_pages(uint64_t npages,
                               const uint64_t *__restrict__ descs,
                               const uint8_t *__restrict__ src,
                               uint8_t *__restrict__ dst,
                               int64_t *__restrict__ out) {
  uint64_t page = blockIdx.x;
  if (page >= npages) return;
  const uint64_t *d = descs + page * 8;
  const uint8_t *ip = src + d[0];
  const uint8_t *iend = ip + d[1];
  uint8_t *op0 = dst + d[2];
  uint64_t expect = d[3];
  int lane = threadIdx.x;
  int64_t *o = out + page * 4;

  // varint preamble: uncompressed length (lane 0 parses; all lanes
  // recompute — cheap, keeps them in lockstep without a broadcast)
  uint64_t ulen = 0;
  {
    int shift = 0;
    const uint8_t *p = ip;
    while (p < iend) {
      uint8_t b = *p++;
      ulen |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    ip = p;
  }
  if (ulen != expect) {
    if (lane == 0) { o[0] = 0; o[1] = 2; o[2] = 0; }
    return;
  }

  uint64_t opos = 0;
  int err = 0;
  while (ip < iend && opos < ulen) {
    // every lane parses the tag identically (uniform scalar work) — no
    // broadcast needed, the wave stays converged
    uint8_t tag = *ip++;
    uint64_t len;
    uint64_t cof = 0;       // copy offset (0 => literal)
    if ((tag & 3) == 0) {                         // literal
      len = (tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        len = 0;
        for (int k = 0; k < nb; k++) len |= (uint64_t)ip[k] << (8 * k);
        len += 1;
        ip += nb;
      }
      if (ip + len > iend || opos + len > ulen) { err = 1; break; }
      for (uint64_t k = lane; k < len; k += 64) op0[opos + k] = ip[k];
      ip += len;
    } else {
      if ((tag & 3) == 1) {                       // copy1: 4..11 bytes
        len = 4 + ((tag >> 2) & 7);
        cof = ((uint64_t)(tag >> 5) << 8) | *ip;
        ip += 1;
      } else if ((tag & 3) == 2) {                // copy2
        len = (tag >> 2) + 1;
        cof = (uint64_t)ip[0] | ((uint64_t)ip[1] << 8);
        ip += 2;
      } else {                                    // copy4
        len = (tag >> 2) + 1;
        cof = (uint64_t)ip[0] | ((uint64_t)ip[1] << 8) |
              ((uint64_t)ip[2] << 16) | ((uint64_t)ip[3] << 24);
        ip += 4;
      }
      if (cof == 0 || cof > opos || opos + len > ulen) { err = 1; break; }
      // the match may reference bytes other lanes wrote: order them
      __threadfence_block();
      __syncthreads();
      if (cof >= len) {
        for (uint64_t k = lane; k < len; k += 64)
          op0[opos + k] = op0[opos - cof + k];
      } else {
        // overlapped match = repeating pattern of period cof: every
        // output byte maps to the PRE-MATCH region, no intra-copy RAW
        for (uint64_t k = lane; k < len; k += 64)
          op0[opos + k] = op0[opos - cof + (k % cof)];
      }
      __threadfence_block();
      __syncthreads();
    }
    opos += len;
  }
  if (!err && opos != ulen) err = 1;

  if (lane == 0) {
    int64_t data_off = 0;
    int64_t first = -1;
    if (!err && d[4] == 1) {
      // v1 definition-levels block: [u32 len][RLE runs], max_def == 1
      if (ulen < 4) {
        err = 1;
      } else {
        uint64_t ln = (uint64_t)op0[0] | ((uint64_t)op0[1] << 8) |
                      ((uint64_t)op0[2] << 16) | ((uint64_t)op0[3] << 24);
        uint64_t p = 4, lend = 4 + ln;
        uint64_t seen = 0, nvals = d[5];
        if (lend > ulen) err = 1;
        while (!err && seen < nvals && p < lend) {
          // varint run header (mirrors _check_levels_v1 exactly)
          uint64_t h = 0;
          int shift = 0;
          while (p < lend) {
            uint8_t b = op0[p++];
            h |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
          }
          if (h & 1) {                    // bit-packed groups, width 1
            uint64_t ngroups = h >> 1;
            uint64_t nv = ngroups * 8;
            if (nv > nvals - seen) nv = nvals - seen;
            uint64_t full = nv / 8, rem = nv % 8;
            if (p + ngroups > lend) { err = 1; break; }
            for (uint64_t g = 0; g < full; g++)
              if (op0[p + g] != 0xFF) { err = 3; break; }
            if (!err && rem &&
                (op0[p + full] & ((1u << rem) - 1)) != ((1u << rem) - 1))
              err = 3;
            p += ngroups;
            seen += nv;
          } else {                        // fill run
            if (p >= lend) { err = 1; break; }
            uint64_t nv = h >> 1;
            if (nv > nvals - seen) nv = nvals - seen;
            if (nv && op0[p] != 1) err = 3;
            p += 1;
            seen += nv;
          }
        }
        data_off = (int64_t)lend;
      }
    }
    if (!err && (uint64_t)data_off < ulen) first = op0[data_off];
    o[0] = data_off;
    o[1] = err;
    o[2] = first;
    o[3] = 0;
  }
}
extern "C" int qk_snappy_pages(void *stream, uint64_t npages,
                               const uint64_t *descs_dev,
                               const uint8_t *src_dev, uint8_t *dst_dev,
                               int64_t *out_dev) {
  if (!npages) return 0;
  hipLaunchKernelGGL(k_snappy_pages, dim3((uint32_t)npages), dim3(64), 0,
                     (hipStream_t)stream, npages, descs_dev, src_dev,
                     dst_dev, out_dev);
  QK_TRY("qk_snappy_pages", hipGetLastError());
  return 0;
}

// ---- GPU gzip (DEFLATE, RFC 1951/1952) page decompression --------------
// Parquet's GZIP codec wraps each page in a gzip stream. Entropy decode
// is inherently sequential, so ONE LANE decodes a page's bit stream
// (canonical Huffman via per-length first/count tables built in LDS);
// pages decode independently and hundreds in flight fill the chip —
// the same division of labor as qk_snappy_pages, with the sequential
// part larger. Supported: stored/fixed/dynamic blocks, standard gzip
// headers (FEXTRA/FNAME/FCOMMENT/FHCRC skipped); errors are reported
// per page, never guessed.
namespace qkgz {
struct BitReader {
  const uint8_t *p, *end;
  uint32_t bitbuf;
  int bitcnt;
  bool fail;
  __device__ void init(const uint8_t *s, const uint8_t *e) {
    p = s; end = e; bitbuf = 0; bitcnt = 0; fail = false;
  }
  __device__ uint32_t bits(int n) {       // n <= 24, LSB-first
    while (bitcnt < n) {
      if (p >= end) { fail = true; return 0; }
      bitbuf |= (uint32_t)(*p++) << bitcnt;
      bitcnt += 8;
    }
    uint32_t v = bitbuf & ((1u << n) - 1);
    bitbuf >>= n;
    bitcnt -= n;
    return v;
  }
  __device__ void align_byte() {
    bitbuf = 0; bitcnt = 0;
  }
  __device__ uint32_t peek(int n) {      // pads zeros past the end
    while (bitcnt < n && p < end) {
      bitbuf |= (uint32_t)(*p++) << bitcnt;
      bitcnt += 8;
    }
    return bitbuf & ((1u << n) - 1);
  }
  __device__ void consume(int n) {
    bitbuf >>= n;
    bitcnt -= n;
    if (bitcnt < 0) { fail = true; bitcnt = 0; }
  }
};

#define QK_GZ_LUT_BITS 10

// canonical Huffman decode tables (per length 1..15)
struct Huff {
  uint16_t count[16];     // codes of each length
  uint16_t first[16];     // first canonical code of each length
  uint16_t offset[16];    // index into syms of first code of each length
  uint16_t syms[288];
  __device__ bool build(const uint8_t *lens, int n) {
    for (int i = 0; i < 16; i++) count[i] = 0;
    for (int i = 0; i < n; i++) count[lens[i]]++;
    count[0] = 0;
    uint32_t code = 0;
    int total = 0;
    for (int l = 1; l < 16; l++) {
      code = (code + count[l - 1]) << 1;
      first[l] = (uint16_t)code;
      offset[l] = total;
      total += count[l];
      if (code + count[l] > (1u << l)) return false;   // over-subscribed
      code += 0;                                        // (first advanced below)
    }
    // recompute properly: first[l] must accumulate counts of SHORTER
    // lengths only (the loop above already does via code shifting)
    uint16_t idx[16];
    for (int l = 0; l < 16; l++) idx[l] = offset[l];
    for (int i = 0; i < n; i++)
      if (lens[i]) syms[idx[lens[i]]++] = (uint16_t)i;
    return true;
  }
  __device__ int decode(BitReader &br) {
    uint32_t code = 0;
    for (int l = 1; l < 16; l++) {
      code |= br.bits(1);
      if (br.fail) return -1;
      if (count[l] && code < (uint32_t)first[l] + count[l] &&
          code >= first[l])
        return syms[offset[l] + (code - first[l])];
      code <<= 1;
    }
    return -1;
  }
  // 10-bit direct lookup (zlib-style): entry = sym | (len << 9), 0 =
  // escape to the bit-by-bit path (codes longer than 10 bits). The
  // stream is LSB-first while Huffman codes are MSB-first, so each
  // code is bit-reversed into the index space.
  __device__ void build_lut(uint16_t *lut) {
    for (int i = 0; i < (1 << QK_GZ_LUT_BITS); i++) lut[i] = 0;
    for (int l = 1; l <= QK_GZ_LUT_BITS; l++)
      for (int k = 0; k < count[l]; k++) {
        uint32_t code = first[l] + k;
        uint16_t sym = syms[offset[l] + k];
        uint32_t rev = 0;
        for (int b = 0; b < l; b++)
          rev |= ((code >> b) & 1u) << (l - 1 - b);
        uint16_t e = (uint16_t)(sym | (l << 9));
        for (uint32_t idx = rev; idx < (1u << QK_GZ_LUT_BITS);
             idx += (1u << l))
          lut[idx] = e;
      }
  }
  __device__ int decode_fast(BitReader &br, const uint16_t *lut) {
    uint16_t e = lut[br.peek(QK_GZ_LUT_BITS)];
    if (e) {
      br.consume(e >> 9);
      return br.fail ? -1 : (e & 0x1FF);
    }
    return decode(br);
  }
};

__constant__ uint16_t LEN_BASE[29] = {
    3, 4, 5, 6, 7, 8, 9, 10, 11, 13, 15, 17, 19, 23, 27, 31,
    35, 43, 51, 59, 67, 83, 99, 115, 131, 163, 195, 227, 258};
__constant__ uint8_t LEN_EXTRA[29] = {
    0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 2, 2,
    3, 3, 3, 3, 4, 4, 4, 4, 5, 5, 5, 5, 0};
__constant__ uint16_t DIST_BASE[30] = {
    1, 2, 3, 4, 5, 7, 9, 13, 17, 25, 33, 49, 65, 97, 129, 193,
    257, 385, 513, 769, 1025, 1537, 2049, 3073, 4097, 6145, 8193,
    12289, 16385, 24577};
__constant__ uint8_t DIST_EXTRA[30] = {
    0, 0, 0, 0, 1, 1, 2, 2, 3, 3, 4, 4, 5, 5, 6, 6,
    7, 7, 8, 8, 9, 9, 10, 10, 11, 11, 12, 12, 13, 13};
__constant__ uint8_t CLC_ORDER[19] = {
    16, 17, 18, 0, 8, 7, 9, 6, 10, 5, 11, 4, 12, 3, 13, 2, 14, 1, 15};
}  // namespace qkgz

// LDS scratch per workgroup (one wave): the two Huffman tables + the
// code-length array for dynamic blocks
struct QkGzLds {
  qkgz::Huff lit, dist;
  uint8_t lens[320];
  uint16_t lut_lit[1 << QK_GZ_LUT_BITS];
  uint16_t lut_dist[1 << QK_GZ_LUT_BITS];
};

// desc per page (8 u64): [src_off, src_len, dst_off, uncompressed_len,
//   mode (0 raw deflate / 1 gzip wrapper), 0, 0, 0]
// out per page (4 i64): [written, err, 0, 0]; err 0 ok, 1 corrupt,
//   2 length mismatch, 3 unsupported
__global__ void __launch_bounds__(64) k_gzip_pages(
    uint64_t npages, const uint64_t *__restrict__ descs,
    const uint8_t *__restrict__ src, uint8_t *__restrict__ dst,
    int64_t *__restrict__ out) {
  using namespace qkgz;
  __shared__ QkGzLds L;
  uint64_t page = blockIdx.x;
  if (page >= npages) return;
  if (threadIdx.x != 0) return;   // entropy decode is sequential; other
                                  // lanes idle (pages fill the chip)
  const uint64_t *d = descs + page * 8;
  const uint8_t *ip = src + d[0];
  const uint8_t *iend = ip + d[1];
  uint8_t *op = dst + d[2];
  uint64_t ocap = d[3];
  int64_t *o = out + page * 4;
  o[0] = 0; o[1] = 0; o[2] = 0; o[3] = 0;

  if (d[4] == 1) {                       // gzip wrapper (RFC 1952)
    if (iend - ip < 10 || ip[0] != 0x1f || ip[1] != 0x8b || ip[2] != 8) {
      o[1] = 3; return;
    }
    uint8_t flg = ip[3];
    ip += 10;
    if (flg & 4) {                       // FEXTRA
      if (iend - ip < 2) { o[1] = 1; return; }
      uint32_t xl = ip[0] | ((uint32_t)ip[1] << 8);
      ip += 2 + xl;
    }
    if (flg & 8)  while (ip < iend && *ip++) {}   // FNAME
    if (flg & 16) while (ip < iend && *ip++) {}   // FCOMMENT
    if (flg & 2)  ip += 2;                        // FHCRC
    if (ip >= iend) { o[1] = 1; return; }
  }

  BitReader br;
  br.init(ip, iend);
  uint64_t w = 0;
  for (;;) {
    uint32_t bfinal = br.bits(1);
    uint32_t btype = br.bits(2);
    if (br.fail) { o[1] = 1; return; }
    if (btype == 0) {                    // stored
      br.align_byte();
      // br.p already consumed whole bytes; realign: p points past
      // consumed bytes (bitbuf dropped) — p is correct since bits()
      // only advances p per byte
      if (br.end - br.p < 4) { o[1] = 1; return; }
      uint32_t len = br.p[0] | ((uint32_t)br.p[1] << 8);
      uint32_t nlen = br.p[2] | ((uint32_t)br.p[3] << 8);
      br.p += 4;
      if ((len ^ 0xFFFF) != nlen || br.end - br.p < (long)len ||
          w + len > ocap) { o[1] = 1; return; }
      for (uint32_t i = 0; i < len; i++) op[w + i] = br.p[i];
      br.p += len;
      w += len;
    } else if (btype == 1 || btype == 2) {
      if (btype == 1) {                  // fixed tables
        for (int i = 0; i < 144; i++) L.lens[i] = 8;
        for (int i = 144; i < 256; i++) L.lens[i] = 9;
        for (int i = 256; i < 280; i++) L.lens[i] = 7;
        for (int i = 280; i < 288; i++) L.lens[i] = 8;
        if (!L.lit.build(L.lens, 288)) { o[1] = 1; return; }
        for (int i = 0; i < 30; i++) L.lens[i] = 5;
        if (!L.dist.build(L.lens, 30)) { o[1] = 1; return; }
        L.lit.build_lut(L.lut_lit);
        L.dist.build_lut(L.lut_dist);
      } else {                           // dynamic tables
        uint32_t hlit = br.bits(5) + 257;
        uint32_t hdist = br.bits(5) + 1;
        uint32_t hclen = br.bits(4) + 4;
        if (br.fail || hlit > 288 || hdist > 30) { o[1] = 1; return; }
        uint8_t cl_lens[19];
        for (int i = 0; i < 19; i++) cl_lens[i] = 0;
        for (uint32_t i = 0; i < hclen; i++)
          cl_lens[CLC_ORDER[i]] = (uint8_t)br.bits(3);
        Huff clh;
        if (br.fail || !clh.build(cl_lens, 19)) { o[1] = 1; return; }
        uint32_t nl = hlit + hdist;
        uint32_t i = 0;
        while (i < nl) {
          int s = clh.decode(br);
          if (s < 0) { o[1] = 1; return; }
          if (s < 16) {
            L.lens[i++] = (uint8_t)s;
          } else if (s == 16) {
            if (i == 0) { o[1] = 1; return; }
            uint32_t r = 3 + br.bits(2);
            uint8_t v = L.lens[i - 1];
            while (r-- && i < nl) L.lens[i++] = v;
          } else if (s == 17) {
            uint32_t r = 3 + br.bits(3);
            while (r-- && i < nl) L.lens[i++] = 0;
          } else {
            uint32_t r = 11 + br.bits(7);
            while (r-- && i < nl) L.lens[i++] = 0;
          }
          if (br.fail) { o[1] = 1; return; }
        }
        if (!L.lit.build(L.lens, hlit)) { o[1] = 1; return; }
        if (!L.dist.build(L.lens + hlit, hdist)) { o[1] = 1; return; }
        L.lit.build_lut(L.lut_lit);
        L.dist.build_lut(L.lut_dist);
      }
      for (;;) {                         // decode symbols
        int s = L.lit.decode_fast(br, L.lut_lit);
        if (s < 0) { o[1] = 1; return; }
        if (s < 256) {
          if (w >= ocap) { o[1] = 1; return; }
          op[w++] = (uint8_t)s;
        } else if (s == 256) {
          break;
        } else {
          s -= 257;
          if (s >= 29) { o[1] = 1; return; }
          uint32_t len = LEN_BASE[s] + br.bits(LEN_EXTRA[s]);
          int ds = L.dist.decode_fast(br, L.lut_dist);
          if (ds < 0 || ds >= 30) { o[1] = 1; return; }
          uint32_t distv = DIST_BASE[ds] + br.bits(DIST_EXTRA[ds]);
          if (br.fail || distv > w || w + len > ocap) {
            o[1] = 1; return;
          }
          for (uint32_t k = 0; k < len; k++, w++)
            op[w] = op[w - distv];
        }
      }
    } else {
      o[1] = 3; return;                  // reserved btype
    }
    if (bfinal) break;
  }
  if (w != d[3]) { o[1] = 2; return; }
  o[0] = (int64_t)w;
  o[2] = w ? (int64_t)op[0] : -1;   // the RLE bit-width byte for
                                    // dictionary-index pages
}
extern "C" int qk_gzip_pages(void *stream, uint64_t npages,
                             const uint64_t *descs_dev,
                             const uint8_t *src_dev, uint8_t *dst_dev,
                             int64_t *out_dev) {
  if (!npages) return 0;
  hipLaunchKernelGGL(k_gzip_pages, dim3((uint32_t)npages), dim3(64), 0,
                     (hipStream_t)stream, npages, descs_dev, src_dev,
                     dst_dev, out_dev);
  QK_TRY("qk_gzip_pages", hipGetLastError());
  return 0;
}

// ---- host-side Thrift page-header walker -------------------------------
// Parquet page headers are Thrift compact-protocol structs between pages
// (parquet-format PageHeader). pyarrow does not expose them, and walking
// thousands of them in Python measured ~0.2 s per GB of file — host C
// here, value decode on the GPU (the division of labor of the whole
// parquet_gpu path; mirrors parquet_thrift.py, which stays as the
// documented reference of the format subset).
namespace pqwalk {
struct Cur {
  const uint8_t *buf;
  uint64_t pos, end;
  bool fail;
};
static inline uint64_t rvarint(Cur &c) {
  uint64_t x = 0;
  int shift = 0;
  while (true) {
    if (c.pos >= c.end || shift > 63) { c.fail = true; return 0; }
    uint8_t b = c.buf[c.pos++];
    x |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) return x;
    shift += 7;
  }
}
static inline int64_t rzigzag(Cur &c) {
  uint64_t x = rvarint(c);
  return (int64_t)(x >> 1) ^ -(int64_t)(x & 1);
}
static void skip_struct(Cur &c);
static void skip_field(Cur &c, int ftype) {
  switch (ftype) {
    case 1: case 2: return;                       // bool true/false
    case 3: if (c.pos >= c.end) c.fail = true; else c.pos++; return;
    case 4: case 5: case 6: rvarint(c); return;   // i16/i32/i64
    case 7: c.pos += 8; if (c.pos > c.end) c.fail = true; return;
    case 8: {                                     // binary
      uint64_t n = rvarint(c);
      c.pos += n;
      if (c.pos > c.end) c.fail = true;
      return;
    }
    case 9: case 10: {                            // list/set
      if (c.pos >= c.end) { c.fail = true; return; }
      uint8_t h = c.buf[c.pos++];
      uint64_t size = h >> 4;
      if (size == 0xF) size = rvarint(c);
      for (uint64_t i = 0; i < size && !c.fail; i++)
        skip_field(c, h & 0xF);
      return;
    }
    case 12: skip_struct(c); return;
    default: c.fail = true; return;
  }
}
static void skip_struct(Cur &c) {
  int64_t fid = 0;
  while (!c.fail) {
    if (c.pos >= c.end) { c.fail = true; return; }
    uint8_t b = c.buf[c.pos++];
    if (b == 0) return;
    int delta = b >> 4;
    if (delta == 0) fid = rzigzag(c); else fid += delta;
    skip_field(c, b & 0xF);
  }
}
// parse one struct keeping integer fields by id into out[0..maxf);
// nested structs in `nest` parsed recursively with an id offset
struct Want {
  int64_t v[16];
  bool has[16];
};
static void read_struct(Cur &c, Want &top, Want *d5, Want *d7, Want *d8) {
  int64_t fid = 0;
  while (!c.fail) {
    if (c.pos >= c.end) { c.fail = true; return; }
    uint8_t b = c.buf[c.pos++];
    if (b == 0) return;
    int delta = b >> 4;
    int ftype = b & 0xF;
    if (delta == 0) fid = rzigzag(c); else fid += delta;
    Want *sub = (fid == 5) ? d5 : (fid == 7) ? d7 : (fid == 8) ? d8
                                                              : nullptr;
    if (ftype == 12 && sub) {
      read_struct(c, *sub, nullptr, nullptr, nullptr);
    } else if (fid < 16 && ftype >= 3 && ftype <= 6) {
      if (ftype == 3) {
        if (c.pos >= c.end) { c.fail = true; return; }
        top.v[fid] = c.buf[c.pos++];
      } else {
        top.v[fid] = rzigzag(c);
      }
      top.has[fid] = true;
    } else if (ftype == 1 || ftype == 2) {
      if (fid < 16) { top.v[fid] = (ftype == 1); top.has[fid] = true; }
    } else {
      skip_field(c, ftype);
    }
  }
}
}  // namespace pqwalk

/* out: QK_PQ_PAGE_FIELDS i64 per page —
 * [kind, num_values, encoding, def_enc, data_off, data_len,
 *  v2_levels_len, num_nulls, uncompressed_len, 0]. See header. */
extern "C" int qk_pq_walk_pages(const uint8_t *buf, uint64_t start,
                                uint64_t total_len, int64_t num_values,
                                int64_t *out, int64_t max_pages,
                                int64_t *n_out) {
  using namespace pqwalk;
  Cur c{buf, start, start + total_len, false};
  int64_t seen = 0, np = 0;
  while (seen < num_values && c.pos < c.end) {
    if (np >= max_pages)
      return qk_fail("qk_pq_walk_pages.max_pages", hipErrorInvalidValue);
    Want top = {}, d5 = {}, d7 = {}, d8 = {};
    read_struct(c, top, &d5, &d7, &d8);
    if (c.fail || !top.has[1] || !top.has[3])
      return qk_fail("qk_pq_walk_pages.header", hipErrorInvalidValue);
    int64_t kind = top.v[1];
    int64_t *o = out + np * 10;
    o[0] = kind;
    o[4] = (int64_t)c.pos;          // data_off
    o[5] = top.v[3];                // data_len (compressed size)
    o[6] = 0;                       // v2_levels_len
    o[7] = 0;                       // num_nulls
    o[8] = top.has[2] ? top.v[2] : top.v[3];  // uncompressed size
    o[9] = 1;  // page data compressed under the chunk codec (v2 may opt out)
    if (kind == 0) {                // data page v1
      if (!d5.has[1] || !d5.has[2])
        return qk_fail("qk_pq_walk_pages.v1", hipErrorInvalidValue);
      o[1] = d5.v[1];
      o[2] = d5.v[2];
      o[3] = d5.has[3] ? d5.v[3] : -1;
      seen += o[1];
    } else if (kind == 2) {         // dictionary page
      if (!d7.has[1] || !d7.has[2])
        return qk_fail("qk_pq_walk_pages.dict", hipErrorInvalidValue);
      o[1] = d7.v[1];
      o[2] = d7.v[2];
      o[3] = -1;
    } else if (kind == 3) {         // data page v2
      if (!d8.has[1] || !d8.has[4])
        return qk_fail("qk_pq_walk_pages.v2", hipErrorInvalidValue);
      o[1] = d8.v[1];
      o[2] = d8.v[4];
      o[3] = 3;
      o[6] = (d8.has[5] ? d8.v[5] : 0) + (d8.has[6] ? d8.v[6] : 0);
      o[7] = d8.has[2] ? d8.v[2] : 0;
      // DataPageHeaderV2.is_compressed (optional bool, default true):
      // writers leave incompressible pages raw and clear this flag
      o[9] = d8.has[7] ? d8.v[7] : 1;
      seen += o[1];
    } else {
      return qk_fail("qk_pq_walk_pages.page_type", hipErrorInvalidValue);
    }
    c.pos += top.v[3];
    np++;
  }
  *n_out = np;
  return 0;
}

// ---- range partition ids (include/quokka_amd.h for the contract) ------
__global__ void __launch_bounds__(BLOCK) k_range_part_ids(
    uint64_t n, const int64_t *__restrict__ keys, int64_t per_range,
    uint32_t nparts, int64_t *__restrict__ out) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    int64_t k = keys[i] - 1;
    int64_t q = k / per_range;
    if (k % per_range < 0) q--;               // floor division
    if (q < 0) q = 0;
    if (q >= (int64_t)nparts) q = (int64_t)nparts - 1;
    out[i] = q;
  }
}
extern "C" int qk_range_part_ids(void *stream, uint64_t n,
                                 const int64_t *keys, int64_t per_range,
                                 uint32_t nparts, int64_t *out) {
  if (!n) return 0;
  if (per_range <= 0 || !nparts)
    return qk_fail("qk_range_part_ids", hipErrorInvalidValue);
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_range_part_ids, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, n, keys, per_range, nparts, out);
  QK_TRY("qk_range_part_ids", hipGetLastError());
  return 0;
}

// ---- GPU CSV parse (include/quokka_amd.h for the contract) ------------
// The reference's CSV scan reads byte ranges and hands them to
// polars.read_csv on the CPU (unordered_readers.py:273-442, :438). Here
// the raw bytes go to HBM and two kernels do the work: an ordered
// newline index (same count/scan/scatter shape as the filter), then a
// thread-per-row typed field parser.

__global__ void __launch_bounds__(BLOCK) k_csv_nl_count(
    uint64_t lo0, uint64_t n, const uint8_t *__restrict__ b, uint64_t chunk,
    uint64_t *__restrict__ block_counts) {
  uint64_t lo = lo0 + (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  uint32_t cnt = 0;
  for (uint64_t r = lo + threadIdx.x; r < hi; r += BLOCK)
    cnt += b[r] == '\n' ? 1u : 0u;
  __shared__ uint32_t lds[BLOCK / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  if (lane == 0) lds[wid] = cnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t s = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) s += lds[w];
    block_counts[blockIdx.x] = s;
  }
}

__global__ void __launch_bounds__(BLOCK) k_csv_nl_scatter(
    uint64_t lo0, uint64_t n, const uint8_t *__restrict__ b, uint64_t chunk,
    const uint64_t *__restrict__ block_offsets, uint64_t *__restrict__ out) {
  uint64_t lo = lo0 + (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  __shared__ uint64_t base;
  __shared__ uint32_t wave_tot[BLOCK / WAVE];
  if (threadIdx.x == 0) base = block_offsets[blockIdx.x];
  __syncthreads();
  int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
  for (uint64_t r0 = lo; r0 < hi; r0 += BLOCK) {
    uint64_t r = r0 + threadIdx.x;
    bool pass = r < hi && b[r] == '\n';
    uint64_t mask = __ballot(pass);
    uint32_t rank = __popcll(mask & ((1ULL << lane) - 1));
    uint32_t wtot = __popcll(mask);
    if (lane == 0) wave_tot[wid] = wtot;
    __syncthreads();
    uint32_t wbase = 0;
    for (int w = 0; w < wid; w++) wbase += wave_tot[w];
    if (pass) out[base + wbase + rank] = r;
    uint32_t btot = 0;
    for (int w = 0; w < BLOCK / WAVE; w++) btot += wave_tot[w];
    __syncthreads();
    if (threadIdx.x == 0) base += btot;
    __syncthreads();
  }
}

extern "C" int qk_csv_newlines(void *stream, uint64_t data_start, uint64_t n,
                               const uint8_t *bytes, uint64_t *out_pos,
                               uint64_t *out_count_dev) {
  if (n <= data_start) return 0;
  uint64_t span = n - data_start;
  uint64_t chunk = (span + MAX_BLOCKS - 1) / MAX_BLOCKS;
  chunk = ((chunk + BLOCK - 1) / BLOCK) * BLOCK;
  uint32_t blocks = (uint32_t)((span + chunk - 1) / chunk);
  static __thread uint64_t *scratch = nullptr;
  if (!scratch)
    QK_TRY("qk_csv_newlines",
           hipMalloc(&scratch, (MAX_BLOCKS + 1) * sizeof(uint64_t)));
  hipLaunchKernelGGL(k_csv_nl_count, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, data_start, n, bytes, chunk,
                     scratch);
  hipLaunchKernelGGL(k_scan_blocks, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, (uint64_t)blocks, scratch,
                     out_count_dev);
  hipLaunchKernelGGL(k_csv_nl_scatter, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, data_start, n, bytes, chunk,
                     scratch, out_pos);
  QK_TRY("qk_csv_newlines", hipGetLastError());
  return 0;
}

// Quote-aware newline indexing (RFC-4180): a '\n' ends a row only when
// the count of quote chars before it is even ("" escapes toggle twice
// and cancel). One WAVE per ~24 KB unit, walked 64 bytes per step with
// coalesced lane loads; quote parity propagates across the wave through
// __ballot + prefix-popcount (quotes below my lane this step) and a
// running per-step parity, so the only truly sequential work is the
// tiny prefix scan over per-unit quote counts between passes:
//   1. quotes per unit   2. prefix parity   3. valid-\n count per unit
//   4. k_scan_blocks exclusive offsets      5. scatter positions
__global__ void k_scan_parity(uint64_t nblocks, const uint64_t *counts,
                              uint8_t *parity) {
  uint64_t acc = 0;
  for (uint64_t i = 0; i < nblocks; i++) {
    parity[i] = (uint8_t)(acc & 1);
    acc += counts[i];
  }
}
__global__ void __launch_bounds__(BLOCK) k_csv_quote_count_w(
    uint64_t lo0, uint64_t n, const uint8_t *__restrict__ b, uint64_t unit,
    uint64_t nunits, uint8_t quote, uint64_t *__restrict__ counts) {
  uint64_t u = ((uint64_t)blockIdx.x * BLOCK + threadIdx.x) / WAVE;
  if (u >= nunits) return;
  int lane = threadIdx.x & (WAVE - 1);
  uint64_t lo = lo0 + u * unit, hi = qk_min_u64(n, lo + unit);
  uint32_t cnt = 0;
  for (uint64_t r = lo + lane; r < hi; r += WAVE)
    cnt += b[r] == quote ? 1u : 0u;
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  if (lane == 0) counts[u] = cnt;
}
__global__ void __launch_bounds__(BLOCK) k_csv_nl_count_qw(
    uint64_t lo0, uint64_t n, const uint8_t *__restrict__ b, uint64_t unit,
    uint64_t nunits, uint8_t quote, const uint8_t *__restrict__ parity,
    uint64_t *__restrict__ unit_counts) {
  uint64_t u = ((uint64_t)blockIdx.x * BLOCK + threadIdx.x) / WAVE;
  if (u >= nunits) return;
  int lane = threadIdx.x & (WAVE - 1);
  uint64_t lo = lo0 + u * unit, hi = qk_min_u64(n, lo + unit);
  uint32_t in_q = parity[u];
  uint32_t cnt = 0;
  uint64_t below = (1ULL << lane) - 1;
  for (uint64_t r0 = lo; r0 < hi; r0 += WAVE) {
    uint64_t r = r0 + lane;
    uint8_t c = r < hi ? b[r] : (uint8_t)0;
    uint64_t qmask = __ballot(c == quote);
    bool valid = c == '\n' &&
                 (((in_q + __popcll(qmask & below)) & 1u) == 0u);
    cnt += valid ? 1u : 0u;
    in_q ^= (uint32_t)(__popcll(qmask) & 1u);
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) cnt += __shfl_down(cnt, off);
  if (lane == 0) unit_counts[u] = cnt;
}
__global__ void __launch_bounds__(BLOCK) k_csv_nl_scatter_qw(
    uint64_t lo0, uint64_t n, const uint8_t *__restrict__ b, uint64_t unit,
    uint64_t nunits, uint8_t quote, const uint8_t *__restrict__ parity,
    const uint64_t *__restrict__ unit_offsets, uint64_t *__restrict__ out) {
  uint64_t u = ((uint64_t)blockIdx.x * BLOCK + threadIdx.x) / WAVE;
  if (u >= nunits) return;
  int lane = threadIdx.x & (WAVE - 1);
  uint64_t lo = lo0 + u * unit, hi = qk_min_u64(n, lo + unit);
  uint32_t in_q = parity[u];
  uint64_t w = unit_offsets[u];
  uint64_t below = (1ULL << lane) - 1;
  for (uint64_t r0 = lo; r0 < hi; r0 += WAVE) {
    uint64_t r = r0 + lane;
    uint8_t c = r < hi ? b[r] : (uint8_t)0;
    uint64_t qmask = __ballot(c == quote);
    bool valid = c == '\n' &&
                 (((in_q + __popcll(qmask & below)) & 1u) == 0u);
    uint64_t vmask = __ballot(valid);
    if (valid) out[w + __popcll(vmask & below)] = r;
    w += __popcll(vmask);
    in_q ^= (uint32_t)(__popcll(qmask) & 1u);
  }
}
extern "C" int qk_csv_newlines_quoted(void *stream, uint64_t data_start,
                                      uint64_t n, const uint8_t *bytes,
                                      uint8_t quote, uint64_t *out_pos,
                                      uint64_t *out_count_dev) {
  if (n <= data_start) return 0;
  uint64_t span = n - data_start;
  const uint64_t MAX_UNITS = 16384;     // waves; 16K x 64 lanes ~ 8x CU fill
  uint64_t unit = (span + MAX_UNITS - 1) / MAX_UNITS;
  unit = ((unit + (uint64_t)WAVE - 1) / WAVE) * WAVE;
  if (unit == 0) unit = WAVE;
  uint64_t units = (span + unit - 1) / unit;
  uint32_t blocks =
      (uint32_t)((units * WAVE + BLOCK - 1) / BLOCK);
  static __thread uint64_t *scratch = nullptr;
  static __thread uint8_t *par = nullptr;
  if (!scratch)
    QK_TRY("qk_csv_newlines_quoted",
           hipMalloc(&scratch, (MAX_UNITS + 1) * sizeof(uint64_t)));
  if (!par)
    QK_TRY("qk_csv_newlines_quoted", hipMalloc(&par, MAX_UNITS + 1));
  hipLaunchKernelGGL(k_csv_quote_count_w, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, data_start, n, bytes, unit,
                     units, quote, scratch);
  hipLaunchKernelGGL(k_scan_parity, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, units, scratch, par);
  hipLaunchKernelGGL(k_csv_nl_count_qw, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, data_start, n, bytes, unit,
                     units, quote, par, scratch);
  hipLaunchKernelGGL(k_scan_blocks, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, units, scratch, out_count_dev);
  hipLaunchKernelGGL(k_csv_nl_scatter_qw, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, data_start, n, bytes, unit,
                     units, quote, par, scratch, out_pos);
  QK_TRY("qk_csv_newlines_quoted", hipGetLastError());
  return 0;
}

// field parsers. Numeric parses are BIT-EXACT vs strtod for <= 15
// significant digits: the digit string becomes an exact int64, the
// scale 10^k is an exact double (k <= 22), and one correctly-rounded
// division yields the correctly-rounded decimal value — the same double
// strtod returns. More digits -> parse error (row reported), never a
// silently different value.
__device__ inline bool csv_i64(const uint8_t *b, uint64_t s, uint64_t e,
                               int64_t *out) {
  if (s >= e) return false;
  bool neg = b[s] == '-';
  if (neg || b[s] == '+') s++;
  if (s >= e) return false;
  int64_t v = 0;
  int nd = 0;
  for (; s < e; s++) {
    uint8_t c = b[s] - '0';
    if (c > 9) return false;
    v = v * 10 + c;
    if (++nd > 18) return false;
  }
  *out = neg ? -v : v;
  return true;
}
__device__ inline bool csv_f64(const uint8_t *b, uint64_t s, uint64_t e,
                               double *out) {
  static const double P10[19] = {1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7,
                                 1e8, 1e9, 1e10, 1e11, 1e12, 1e13, 1e14,
                                 1e15, 1e16, 1e17, 1e18};
  if (s >= e) return false;
  bool neg = b[s] == '-';
  if (neg || b[s] == '+') s++;
  if (s >= e) return false;
  int64_t m = 0;
  int nd = 0, k = 0;
  bool dot = false;
  for (; s < e; s++) {
    uint8_t c = b[s];
    if (c == '.') {
      if (dot) return false;
      dot = true;
      continue;
    }
    c -= '0';
    if (c > 9) return false;
    if (nd || c) nd++;                    // significant digits
    m = m * 10 + c;
    if (dot) k++;
    if (nd > 15 || k > 18) return false;  // beyond bit-exact range
  }
  double v = (double)m / P10[k];
  *out = neg ? -v : v;
  return true;
}
__device__ inline bool csv_date32(const uint8_t *b, uint64_t s, uint64_t e,
                                  int32_t *out) {
  if (e - s != 10 || b[s + 4] != '-' || b[s + 7] != '-') return false;
  int y = 0, mo = 0, d = 0;
  for (int i = 0; i < 4; i++) {
    uint8_t c = b[s + i] - '0';
    if (c > 9) return false;
    y = y * 10 + c;
  }
  for (int i = 5; i < 7; i++) {
    uint8_t c = b[s + i] - '0';
    if (c > 9) return false;
    mo = mo * 10 + c;
  }
  for (int i = 8; i < 10; i++) {
    uint8_t c = b[s + i] - '0';
    if (c > 9) return false;
    d = d * 10 + c;
  }
  if (mo < 1 || mo > 12 || d < 1 || d > 31) return false;
  // Howard Hinnant days_from_civil (public-domain algorithm)
  int yy = y - (mo <= 2);
  int era = (yy >= 0 ? yy : yy - 399) / 400;
  unsigned yoe = (unsigned)(yy - era * 400);
  unsigned doy = (153u * (unsigned)(mo + (mo > 2 ? -3 : 9)) + 2u) / 5u +
                 (unsigned)d - 1u;
  unsigned doe = yoe * 365u + yoe / 4u - yoe / 100u + doy;
  *out = (int32_t)(era * 146097 + (int)doe - 719468);
  return true;
}

#define QK_CSV_MAX_DICT 32

__global__ void __launch_bounds__(BLOCK) k_csv_parse(
    uint64_t nrows, const uint8_t *__restrict__ b, uint64_t data_start,
    const uint64_t *__restrict__ nl, uint8_t sep, uint8_t quote, int ncols,
    const int *__restrict__ coltypes, void *const *__restrict__ outs,
    const uint64_t *__restrict__ dict_cands,
    const uint8_t *__restrict__ dict_lens, const int *__restrict__ ncands,
    unsigned long long *__restrict__ err_row) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       r < nrows; r += stride) {
    uint64_t pos = r == 0 ? data_start : nl[r - 1] + 1;
    uint64_t row_end = nl[r];
    if (row_end > pos && b[row_end - 1] == '\r') row_end--;
    bool ok = true;
    for (int c = 0; c < ncols && ok; c++) {
      uint64_t fs = pos, fe;
      if (quote && pos < row_end && b[pos] == quote) {
        // RFC-4180 quoted field: content [pos+1, closing quote);
        // doubled quotes escape (and stay in the slice — numeric/dict
        // parses of escaped content fail loudly, candidates are plain)
        fs = pos + 1;
        fe = fs;
        while (fe < row_end) {
          if (b[fe] == quote) {
            if (fe + 1 < row_end && b[fe + 1] == quote) fe += 2;
            else break;
          } else {
            fe++;
          }
        }
        uint64_t after = fe < row_end ? fe + 1 : row_end;
        pos = (after < row_end && b[after] == sep) ? after + 1 : after;
      } else {
        fe = pos;
        while (fe < row_end && b[fe] != sep) fe++;
        pos = fe < row_end ? fe + 1 : row_end;
      }
      switch (coltypes[c]) {
        case 0: ok = csv_i64(b, fs, fe, (int64_t *)outs[c] + r); break;
        case 1: ok = csv_f64(b, fs, fe, (double *)outs[c] + r); break;
        case 2: ok = csv_date32(b, fs, fe, (int32_t *)outs[c] + r); break;
        case 3: {
          uint64_t len = fe - fs;
          uint64_t w = 0;
          for (uint64_t i = 0; i < 8 && i < len; i++)
            w |= (uint64_t)b[fs + i] << (8 * i);
          int code = -1;
          for (int j = 0; j < ncands[c]; j++)
            if (dict_cands[c * QK_CSV_MAX_DICT + j] == w &&
                dict_lens[c * QK_CSV_MAX_DICT + j] ==
                    (uint8_t)(len > 255 ? 255 : len)) {
              code = j;
              break;
            }
          if (code < 0) { ok = false; break; }
          ((uint8_t *)outs[c])[r] = (uint8_t)code;
          break;
        }
        default: break;                  // 4 = skip
      }
    }
    if (!ok) atomicMin(err_row, (unsigned long long)r);
  }
}

extern "C" int qk_csv_parse(void *stream, uint64_t nrows,
                            const uint8_t *bytes, uint64_t data_start,
                            const uint64_t *nl_pos, uint8_t sep,
                            uint8_t quote, int ncols,
                            const int *coltypes, void *const *out_ptrs,
                            const uint64_t *dict_cands,
                            const uint8_t *dict_lens, const int *ncands,
                            uint64_t *err_row) {
  if (!nrows) return 0;
  uint32_t blocks =
      (uint32_t)qk_min_u64(MAX_BLOCKS, (nrows + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_csv_parse, dim3(blocks), dim3(BLOCK), 0,
                     (hipStream_t)stream, nrows, bytes, data_start, nl_pos,
                     sep, quote, ncols, coltypes, out_ptrs, dict_cands,
                     dict_lens, ncands, (unsigned long long *)err_row);
  QK_TRY("qk_csv_parse", hipGetLastError());
  return 0;
}

// RCCL 2.27 p2p SILENTLY DELIVERS ONLY THE FIRST HALF of a send whose
// byte count exceeds 1 GiB (measured on MI355X: exact at 2^30 B, exactly
// half the rows at every size above — scripts/diag_exchange.py). Two
// defenses: the self-partition never touches RCCL (plain async D2D copy),
// and peer pieces are split into <=QK_P2P_CHUNK-byte sends/recvs — both
// ranks derive the identical split from the exchanged counts, so the
// grouped send/recv matching stays aligned.
static const uint64_t QK_P2P_CHUNK = 256ull << 20;

extern "C" int qk_alltoallv(void *stream, void *comm, int world,
                            uint32_t elem_size, const void *send_buf,
                            const uint64_t *send_offsets,
                            const uint64_t *send_counts, void *recv_buf,
                            const uint64_t *recv_offsets,
                            const uint64_t *recv_counts) {
  int rank = 0;
  QK_NCCL("qk_alltoallv.rank",
          ncclCommUserRank((ncclComm_t)comm, &rank));
  if (send_counts[rank]) {
    if (send_counts[rank] != recv_counts[rank])
      return qk_fail("qk_alltoallv.self_count", hipErrorInvalidValue);
    QK_TRY("qk_alltoallv.self_copy",
           hipMemcpyAsync(
               (char *)recv_buf + recv_offsets[rank] * elem_size,
               (const char *)send_buf + send_offsets[rank] * elem_size,
               send_counts[rank] * elem_size, hipMemcpyDeviceToDevice,
               (hipStream_t)stream));
  }
  QK_NCCL("qk_alltoallv.group_start", ncclGroupStart());
  for (int p = 0; p < world; p++) {
    if (p == rank) continue;
    for (uint64_t off = 0, nb = send_counts[p] * elem_size; off < nb;
         off += QK_P2P_CHUNK) {
      uint64_t m = nb - off < QK_P2P_CHUNK ? nb - off : QK_P2P_CHUNK;
      QK_NCCL("qk_alltoallv.send",
              ncclSend((const char *)send_buf +
                           send_offsets[p] * elem_size + off,
                       m, ncclUint8, p, (ncclComm_t)comm,
                       (hipStream_t)stream));
    }
    for (uint64_t off = 0, nb = recv_counts[p] * elem_size; off < nb;
         off += QK_P2P_CHUNK) {
      uint64_t m = nb - off < QK_P2P_CHUNK ? nb - off : QK_P2P_CHUNK;
      QK_NCCL("qk_alltoallv.recv",
              ncclRecv((char *)recv_buf +
                           recv_offsets[p] * elem_size + off,
                       m, ncclUint8, p, (ncclComm_t)comm,
                       (hipStream_t)stream));
    }
  }
  QK_NCCL("qk_alltoallv.group_end", ncclGroupEnd());
  return 0;
}
extern "C" int qk_allreduce_f64(void *stream, void *comm, double *buf,
                                uint64_t n) {
  QK_NCCL("qk_allreduce_f64",
          ncclAllReduce(buf, buf, n, ncclDouble, ncclSum, (ncclComm_t)comm,
                        (hipStream_t)stream));
  return 0;
}

// ---- hash partition (int key: part = key % nparts, quokka_runtime:222) --
// Negative keys: the reference's polars `%` yields NEGATIVE partition ids
// and its runtime breaks on them; we keep equal-keys-colocate semantics
// with the mathematical (non-negative) mod instead of corrupting memory.
__device__ inline uint32_t part_of(int64_t key, uint32_t nparts) {
  int64_t p = key % (int64_t)nparts;
  return (uint32_t)(p < 0 ? p + (int64_t)nparts : p);
}

__global__ void __launch_bounds__(BLOCK) k_partition_hist(
    uint64_t n, const int64_t *__restrict__ keys, uint32_t nparts,
    uint64_t *__restrict__ hist) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint32_t *lh = (uint32_t *)smem;
  for (uint32_t p = threadIdx.x; p < nparts; p += BLOCK) lh[p] = 0;
  __syncthreads();
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    atomicAdd(&lh[part_of(keys[i], nparts)], 1u);
  __syncthreads();
  for (uint32_t p = threadIdx.x; p < nparts; p += BLOCK)
    if (lh[p]) atomicAdd((unsigned long long *)&hist[p], (unsigned long long)lh[p]);
}
extern "C" int qk_partition_hist(void *stream, uint64_t n, const int64_t *keys,
                                 uint32_t nparts, uint64_t *hist) {
  if (!n) return 0;
  uint32_t blocks = (uint32_t)qk_min_u64(MAX_BLOCKS, (n + BLOCK - 1) / BLOCK);
  hipLaunchKernelGGL(k_partition_hist, dim3(blocks), dim3(BLOCK),
                     nparts * sizeof(uint32_t), (hipStream_t)stream, n, keys,
                     nparts, hist);
  QK_TRY("qk_partition_hist", hipGetLastError());
  return 0;
}

// Block-aggregated scatter: per-chunk LDS histogram, ONE global cursor
// atomicAdd per (block, partition), then LDS-cursor placement. (The naive
// per-row global atomic measured 1.4 s at nparts=1 over 120M rows: a
// single cursor word takes ~88 atomics/us.) nparts <= 512.
__global__ void __launch_bounds__(BLOCK) k_partition_scatter(
    uint64_t n, const int64_t *__restrict__ keys, uint32_t nparts,
    uint64_t *__restrict__ cursors, uint32_t *__restrict__ out_idx,
    uint64_t chunk) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint32_t *lcnt = (uint32_t *)smem;             // [nparts]
  uint64_t *lbase = (uint64_t *)(smem + ((nparts * 4 + 15) & ~15u));
  uint64_t lo = (uint64_t)blockIdx.x * chunk;
  uint64_t hi = qk_min_u64(n, lo + chunk);
  for (uint64_t c0 = lo; c0 < hi; c0 += 65536) {
    uint64_t c1 = qk_min_u64(hi, c0 + 65536);
    for (uint32_t p = threadIdx.x; p < nparts; p += BLOCK) lcnt[p] = 0;
    __syncthreads();
    for (uint64_t i = c0 + threadIdx.x; i < c1; i += BLOCK)
      atomicAdd(&lcnt[part_of(keys[i], nparts)], 1u);
    __syncthreads();
    for (uint32_t p = threadIdx.x; p < nparts; p += BLOCK) {
      lbase[p] = lcnt[p]
                     ? atomicAdd((unsigned long long *)&cursors[p],
                                 (unsigned long long)lcnt[p])
                     : 0;
      lcnt[p] = 0;
    }
    __syncthreads();
    for (uint64_t i = c0 + threadIdx.x; i < c1; i += BLOCK) {
      uint32_t p = part_of(keys[i], nparts);
      uint32_t r = atomicAdd(&lcnt[p], 1u);
      out_idx[lbase[p] + r] = (uint32_t)i;
    }
    __syncthreads();
  }
}
extern "C" int qk_partition_scatter(void *stream, uint64_t n,
                                    const int64_t *keys, uint32_t nparts,
                                    uint64_t *cursors, uint32_t *out_idx) {
  if (!n) return 0;
  if (nparts > 512)
    return qk_fail("qk_partition_scatter.nparts", hipErrorInvalidValue);
  uint64_t chunk = (n + MAX_BLOCKS - 1) / MAX_BLOCKS;
  chunk = ((chunk + BLOCK - 1) / BLOCK) * BLOCK;
  uint32_t blocks = (uint32_t)((n + chunk - 1) / chunk);
  uint32_t lds_bytes = ((nparts * 4 + 15) & ~15u) + nparts * 8;
  hipLaunchKernelGGL(k_partition_scatter, dim3(blocks), dim3(BLOCK),
                     lds_bytes, (hipStream_t)stream, n, keys, nparts,
                     cursors, out_idx, chunk);
  QK_TRY("qk_partition_scatter", hipGetLastError());
  return 0;
}
