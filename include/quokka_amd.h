/* quokka_amd — C ABI of the MI355X-native columnar execution path.
 *
 * This is the drop-in boundary beneath the reference's (marsupialtail/quokka)
 * Executor / partition-function plugin API. The reference is pure Python
 * (pyquokka/executors/base_executor.py:26-32 `Executor.execute/done`,
 * pyquokka/core.py:152-195 `partition_fn`) delegating all arithmetic to
 * polars/duckdb; this library replaces exactly that arithmetic with HIP
 * kernels for gfx950. The Python classes in quokka_amd/executors.py bind
 * these entry points via ctypes and present the reference's own interface
 * (same names / argument meaning / error behaviour); see INTEGRATION.md for
 * the binding a pyquokka maintainer would add.
 *
 * Conventions:
 *  - every function returns 0 on success, nonzero HIP/ABI error code;
 *    qk_last_error() returns a static string for the calling thread's last
 *    failure. No function falls back to CPU: with no usable GPU, calls fail.
 *  - `stream` is an opaque HIP stream handle from qk_stream_create (NULL =
 *    default stream). All kernel entry points are asynchronous on `stream`.
 *  - device pointers are void* from qk_dmalloc; columns are dense device
 *    arrays (no nulls — the reference's TPC-H hot path has none).
 *  - dates are int32 days since 1970-01-01 (Arrow date32), strings are
 *    host-side dictionary codes (u8), matching the staging layer.
 */
#ifndef QUOKKA_AMD_H
#define QUOKKA_AMD_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- context ------------------------------------------------------- */
int qk_init(int device);                 /* set + warm the HIP device      */
int qk_device_count(int *out);
const char *qk_last_error(void);
const char *qk_build_arch(void);         /* "gfx950"                       */

/* ---- device memory -------------------------------------------------- */
int qk_dmalloc(uint64_t nbytes, void **dptr);
int qk_hmalloc_impl(uint64_t nbytes, void **hptr); /* pinned host memory */
int qk_hfree(void *hptr);                          /* free pinned memory */
int qk_dfree(void *dptr);
int qk_h2d(void *dst_dev, const void *src_host, uint64_t nbytes);
int qk_d2h(void *dst_host, const void *src_dev, uint64_t nbytes);
/* Async variants on a stream; src/dst host memory must be PINNED
 * (qk_hmalloc_impl) or the copy silently degrades to staged+sync. Used
 * by the double-buffered staging bounce (shim._PinnedBounce): host
 * memmove of chunk k+1 overlaps the DMA of chunk k on alternating
 * streams. */
int qk_h2d_async(void *stream, void *dst_dev, const void *src_host,
                 uint64_t nbytes);
int qk_d2h_async(void *stream, void *dst_host, const void *src_dev,
                 uint64_t nbytes);
int qk_dmemset(void *dst_dev, int value, uint64_t nbytes);
int qk_fill_i64(void *stream, int64_t *dst_dev, int64_t value, uint64_t n);

/* ---- streams & timing ----------------------------------------------- */
int qk_stream_create(void **stream);
int qk_stream_destroy(void *stream);
int qk_stream_sync(void *stream);
/* Bare (non-timing) HIP events for cross-stream ordering: the overlapped
 * shuffle records an event on the comm stream after each received chunk and
 * the compute stream waits on it before probing that chunk (north_star:
 * RCCL repartition overlapped with probe on a side HIP stream; replaces the
 * reference's 8-thread Flight do_put pool overlap, core.py:324-371). */
int qk_event_create(void **ev);
int qk_event_destroy(void *ev);
int qk_event_record(void *ev, void *stream);
int qk_stream_wait_event(void *stream, void *ev);
/* HIP-event timer pair on the launching stream (roofline measurement). */
int qk_timer_create(void **timer);
int qk_timer_destroy(void *timer);
int qk_timer_start(void *timer, void *stream);
int qk_timer_stop(void *timer, void *stream);
int qk_timer_elapsed_ms(void *timer, float *out_ms);   /* syncs the stop event */

/* ---- synthetic data (bench only; distributions = oracle/tpch_gen.py) */
/* Fill lineitem columns for rows [row_offset, row_offset+n) with the TPC-H
 * distributions of oracle/tpch_gen.py (counter-based RNG: value depends only
 * on (seed, global row)). Any output pointer may be NULL to skip a column.
 * n_parts/n_suppliers/n_orders scale key ranges (pass tpch_gen values). */
int qk_gen_lineitem(void *stream, uint64_t n, uint64_t row_offset, uint64_t seed,
                    int64_t n_parts, int64_t n_suppliers, int64_t n_orders,
                    int64_t *l_orderkey, int64_t *l_suppkey,
                    double *l_quantity, double *l_extendedprice,
                    double *l_discount, double *l_tax,
                    uint8_t *l_returnflag, uint8_t *l_linestatus,
                    int32_t *l_shipdate, int32_t *l_commitdate,
                    int32_t *l_receiptdate);

/* Orders columns for rows [row_offset, row_offset+n): o_orderkey SPARSE
 * per TPC-H spec 4.2.3 (8 keys per 32-key bucket, = oracle/tpch_gen
 * sparse_orderkeys), o_custkey uniform 1..n_customers with the spec's
 * custkey%3 != 0 mortality hole, o_orderdate uniform spec range,
 * o_shippriority 0. Consistent with qk_gen_lineitem's l_orderkey (order
 * row = lineitem row/4: every order has exactly 4 lines). */
/* o_orderpriority: uniform u8 code 0..4; o_totalprice: derived from the
 * order's 4 lines by re-deriving their qty/partkey/disc/tax with
 * qk_gen_lineitem's counter-based formulas (pass the same li_n_parts). */
int qk_gen_orders(void *stream, uint64_t n, uint64_t row_offset, uint64_t seed,
                  int64_t n_customers, int64_t *o_orderkey, int64_t *o_custkey,
                  int32_t *o_orderdate, int32_t *o_shippriority,
                  uint8_t *o_orderpriority, double *o_totalprice,
                  int64_t li_n_parts);
/* Customer columns: c_custkey dense row+1, c_mktsegment uniform u8 0..4,
 * c_nationkey uniform i32 0..24. Any output may be NULL. */
int qk_gen_customer(void *stream, uint64_t n, uint64_t row_offset,
                    uint64_t seed, int64_t *c_custkey, uint8_t *c_mktsegment,
                    int32_t *c_nationkey);
/* Supplier columns: s_suppkey dense row+1, s_nationkey uniform 0..24. */
int qk_gen_supplier(void *stream, uint64_t n, uint64_t row_offset,
                    uint64_t seed, int64_t *s_suppkey, int32_t *s_nationkey);
/* Aux independent-draw column generator (device mirror of
 * oracle/tpch_gen.py's independent columns). mode 0: out_u8 = uniform
 * code in [0,a) (l_shipmode, tpch_gen:231). mode 1: out_u8 = bernoulli
 * flag, P = a/1e6 (o_comment_special, tpch_gen:165). mode 2: out_f64 =
 * uniform cents in [a,b] / 100 (c_acctbal, tpch_gen:250). */
int qk_gen_aux(void *stream, uint64_t n, uint64_t row_offset, uint64_t seed,
               uint64_t salt, int mode, int64_t a, int64_t b,
               uint8_t *out_u8, double *out_f64);

/* ---- TPC-H Q1: fused filter + group-by partial aggregate ------------- *
 * Replaces the map-side partial agg the reference folds into partition_fn
 * (pyquokka/core.py:173-176 + df.py:1354-1394: per-batch DuckDB
 * "SUM/COUNT group by l_returnflag,l_linestatus") and the probe-side concat
 * of SQLAggExecutor.execute (sql_executors.py:587-590), in one pass.
 * out_dev = device f64[6 groups][8]:
 *   [sum_qty, sum_base_price, sum_disc_price, sum_charge, sum_disc, count,
 *    pad, pad]; group id = l_returnflag*2 + l_linestatus.
 * ACCUMULATES into out_dev (caller zeroes once; repeated calls = executor
 * state accumulation across batches, sql_executors.py:587-590). */
int qk_q1_agg(void *stream, uint64_t n,
              const int32_t *l_shipdate, const double *l_quantity,
              const double *l_extendedprice, const double *l_discount,
              const double *l_tax, const uint8_t *l_returnflag,
              const uint8_t *l_linestatus, int32_t cutoff_date,
              double *out_dev /* f64[48] */);

/* ---- TPC-H Q6: fused filter + sum ------------------------------------ *
 * tpch_ref.py:171-183 semantics. out_dev = f64[2] {sum_revenue, count};
 * accumulates like qk_q1_agg. Bounds are the exact f64 constants of the
 * reference SQL expressions. */
int qk_q6_agg(void *stream, uint64_t n,
              const int32_t *l_shipdate, const double *l_quantity,
              const double *l_extendedprice, const double *l_discount,
              int32_t date_lo, int32_t date_hi,
              double disc_lo, double disc_hi, double qty_hi,
              double *out_dev /* f64[2] */);

/* ---- generic filter: compare + compaction ----------------------------- *
 * Replaces the predicate filter of partition_fn (core.py:170, polars
 * DataFrame.filter) for single-column compares; emits the ORDERED indices
 * of passing rows (row order preserved, as polars filter does).
 * op: 0 '<', 1 '<=', 2 '>', 3 '>=', 4 '==', 5 '!='.
 * out_count_dev: device u64, set to number of passing rows (must be zeroed).
 * out_idx capacity must be >= n. */
int qk_filter_i32(void *stream, uint64_t n, const int32_t *col, int op,
                  int32_t value, uint32_t *out_idx, uint64_t *out_count_dev);
int qk_filter_u8(void *stream, uint64_t n, const uint8_t *col, int op,
                 uint8_t value, uint32_t *out_idx, uint64_t *out_count_dev);
/* f64 variant; threshold is a kernel argument (no JIT literal). */
int qk_filter_f64(void *stream, uint64_t n, const double *col, int op,
                  double value, uint32_t *out_idx, uint64_t *out_count_dev);

/* elementwise revenue: out[i] = a[i] * (1 - b[i]) (the per-row product the
 * reference computes before summing, apps/tpc-h/tpch.py:151) */
int qk_flag_gt_i32(void *stream, uint64_t n, const int32_t *a,
                   const int32_t *b, double *out); /* out=a>b?1:0 (Q21) */
int qk_mul_1md(void *stream, uint64_t n, const double *a, const double *b,
               double *out);

/* gather: dst[i] = src[idx[i]] (column compaction / join payload gather) */
int qk_gather_i64(void *stream, uint64_t n_idx, const uint32_t *idx,
                  const int64_t *src, int64_t *dst);
int qk_gather_f64(void *stream, uint64_t n_idx, const uint32_t *idx,
                  const double *src, double *dst);
int qk_gather_i32(void *stream, uint64_t n_idx, const uint32_t *idx,
                  const int32_t *src, int32_t *dst);
int qk_gather_u8(void *stream, uint64_t n_idx, const uint32_t *idx,
                 const uint8_t *src, uint8_t *dst);

/* ---- hash join build / probe (i64 keys) ------------------------------- *
 * Replaces BuildProbeJoinExecutor (sql_executors.py:325-377): build side =
 * stream 1 (vstacked state), probe = stream 0, how in {inner,semi,anti}
 * (left via inner + host fill). Open-addressing table, linear probing,
 * capacity a power of two >= 2x build rows. Duplicate build keys chain
 * through chain_next. slot_keys must be pre-filled with QK_JOIN_EMPTY
 * (qk_fill_i64), slot_head with -1 bytes (qk_dmemset 0xff).
 * Build may be called repeatedly (batch accumulation): build_row_offset is
 * added to local row indices so chains index the concatenated build side. */
#define QK_JOIN_EMPTY INT64_MIN
int qk_join_build(void *stream, uint64_t n_build, const int64_t *keys,
                  uint32_t build_row_offset, int64_t *slot_keys,
                  int32_t *slot_head, int32_t *chain_next, uint64_t capacity);
/* mode: 0 = inner (emit probe_idx,build_idx pairs), 1 = semi (probe_idx
 * only), 2 = anti (probe_idx only). out_cursor_dev (u64, zeroed) returns the
 * TOTAL match count even when it exceeds out_capacity (no silent
 * truncation: caller re-runs with larger buffers when cursor > capacity). */
int qk_join_probe(void *stream, uint64_t n_probe, const int64_t *keys,
                  const int64_t *slot_keys, const int32_t *slot_head,
                  const int32_t *chain_next, uint64_t capacity, int mode,
                  uint32_t *out_probe_idx, uint32_t *out_build_idx,
                  uint64_t out_capacity, uint64_t *out_cursor_dev);

/* ---- fused Q3 path ---------------------------------------------------- *
 * Filter + semi-join + build and filter + probe + group-by aggregate fused
 * into single passes (the reference folds filters and per-batch partial
 * aggs into partition_fn the same way, core.py:152-195 + df.py:1354-1394).
 * Build keys must be UNIQUE (orders/customer primary keys); slot_head
 * stores the build ROW index; the orders table's slots double as group-by
 * slots for the probe aggregate. Tables pre-filled like qk_join_build's. */
int qk_build_u8eq(void *stream, uint64_t n, const int64_t *keys,
                  const uint8_t *flag, uint8_t flag_val, int64_t *slot_keys,
                  int32_t *slot_head, uint64_t capacity, uint32_t *bloom,
                  uint64_t bloom_mask);
/* bloom/bloom_mask (nullable/0): optional Bloom prefilter the probe-side
 * kernels test before the table walk (bits = pow2, mask = bits-1). */
int qk_q3_build_orders(void *stream, uint64_t n, const int64_t *o_orderkey,
                       const int64_t *o_custkey, const int32_t *o_orderdate,
                       int32_t date_lt, const int64_t *cust_keys,
                       const int32_t *cust_head, uint64_t cust_cap,
                       int64_t *slot_keys, int32_t *slot_head,
                       uint64_t capacity, uint32_t *bloom,
                       uint64_t bloom_mask, const uint32_t *cust_bloom,
                       uint64_t cust_bloom_mask);
/* Count the rows qk_q3_build_orders would insert (for tight table sizing;
 * count_dev u64, zeroed). */
int qk_q3_count_orders(void *stream, uint64_t n, const int64_t *o_custkey,
                       const int32_t *o_orderdate, int32_t date_lt,
                       const int64_t *cust_keys, const int32_t *cust_head,
                       uint64_t cust_cap, uint64_t *count_dev,
                       const uint32_t *cust_bloom, uint64_t cust_bloom_mask);
int qk_q3_probe_agg(void *stream, uint64_t n, const int64_t *l_orderkey,
                    const int32_t *l_shipdate, const double *l_price,
                    const double *l_disc, int32_t date_gt,
                    const int64_t *slot_keys, const int32_t *slot_head,
                    uint64_t capacity, double *slot_sums /* f64[capacity],
                    zeroed; groups keyed by slot */,
                    uint64_t *match_count_dev /* nullable, zeroed u64 */);
/* Non-temporal-load variant of qk_q3_probe_agg: streams the lineitem
 * columns without polluting the caches so the build table stays resident
 * (same results; perf variant, A/B-measured in profiles/). */
int qk_q3_probe_agg_nt(void *stream, uint64_t n, const int64_t *l_orderkey,
                       const int32_t *l_shipdate, const double *l_price,
                       const double *l_disc, int32_t date_gt,
                       const int64_t *slot_keys, const int32_t *slot_head,
                       uint64_t capacity, double *slot_sums,
                       uint64_t *match_count_dev, const uint32_t *bloom,
                       uint64_t bloom_mask);
/* 4-rows-per-lane ILP variant: doubles the independent bloom/table load
 * chains per lane (the probe is random-load latency-bound at full
 * occupancy — r01 stall anatomy); same results, A/B in profiles/r02. */
int qk_q3_probe_agg_nt4(void *stream, uint64_t n, const int64_t *l_orderkey,
                        const int32_t *l_shipdate, const double *l_price,
                        const double *l_disc, int32_t date_gt,
                        const int64_t *slot_keys, const int32_t *slot_head,
                        uint64_t capacity, double *slot_sums,
                        uint64_t *match_count_dev, const uint32_t *bloom,
                        uint64_t bloom_mask);
/* Emit (orderkey, orders_build_row, revenue) for slots with sum != 0.
 * cursor (u64, zeroed) = group count (counted even past out_cap). */
int qk_q3_extract(void *stream, const int64_t *slot_keys,
                  const int32_t *slot_head, const double *slot_sums,
                  uint64_t capacity, int64_t *out_keys, int32_t *out_row,
                  double *out_sums, uint64_t out_cap, uint64_t *cursor_dev);

/* ---- fused Q5 path ----------------------------------------------------- *
 * Key -> i32-value hash tables and a fused lineitem probe for the 6-table
 * chain (tpch_ref.py:142-169). Unique build keys (primary keys). */
/* Insert keys[i] -> vals[i] where bit vals[i] of accept_mask is set
 * (accept_mask = 0xFFFFFFFF accepts all; else vals must be in [0,32)). */
int qk_build_keyval_i32(void *stream, uint64_t n, const int64_t *keys,
                        const int32_t *vals, uint32_t accept_mask,
                        int64_t *slot_keys, int32_t *slot_val,
                        uint64_t capacity, uint32_t *bloom,
                        uint64_t bloom_mask);
/* Orders in [date_lo, date_hi) whose o_custkey hits the customer table:
 * insert o_orderkey -> customer nationkey. slot_keys == NULL -> count-only
 * (count_dev gets the survivor count either way when non-NULL). */
int qk_q5_build_orders(void *stream, uint64_t n, const int64_t *o_orderkey,
                       const int64_t *o_custkey, const int32_t *o_orderdate,
                       int32_t date_lo, int32_t date_hi,
                       const int64_t *cust_keys, const int32_t *cust_val,
                       uint64_t cust_cap, int64_t *slot_keys,
                       int32_t *slot_val, uint64_t capacity,
                       uint64_t *count_dev, uint32_t *bloom,
                       uint64_t bloom_mask, const uint32_t *cust_bloom,
                       uint64_t cust_bloom_mask);
/* Fused probe: join lineitem to orders (-> customer nation) and supplier
 * (-> supplier nation); where equal accumulate revenue into out25[nation].
 * out25: f64[32], zeroed (slots 25..31 unused). */
int qk_q5_probe_agg(void *stream, uint64_t n, const int64_t *l_orderkey,
                    const int64_t *l_suppkey, const double *l_price,
                    const double *l_disc, const int64_t *ord_keys,
                    const int32_t *ord_val, uint64_t ord_cap,
                    const int64_t *supp_keys, const int32_t *supp_val,
                    uint64_t supp_cap, double *out25,
                    uint64_t *match_count_dev /* nullable */);

/* Non-temporal-load variant of qk_q5_probe_agg (see qk_q3_probe_agg_nt). */
int qk_q5_probe_agg_nt(void *stream, uint64_t n, const int64_t *l_orderkey,
                       const int64_t *l_suppkey, const double *l_price,
                       const double *l_disc, const int64_t *ord_keys,
                       const int32_t *ord_val, uint64_t ord_cap,
                       const int64_t *supp_keys, const int32_t *supp_val,
                       uint64_t supp_cap, double *out25,
                       uint64_t *match_count_dev, const uint32_t *bloom,
                       uint64_t bloom_mask);

/* ---- group-by (i64 key) sum ------------------------------------------- *
 * Replaces SQLAggExecutor's DuckDB group-by (sql_executors.py:592-599) for
 * distributive SUM over an i64 key (the post-rewrite partial form,
 * sql_utils.py:299-413). Open-addressing accumulate table; slot_keys
 * pre-filled with QK_JOIN_EMPTY, slot_sums zeroed. Repeated calls
 * accumulate. nvals value columns share one key lookup (vals/slot_sums are
 * arrays-of-columns, each `capacity` doubles: slot_sums[c*capacity+slot]). */
/* agg_ops_dev (device i32[nvals], nullable -> all SUM): per value column
 * 0 = SUM (slot init 0), 1 = MIN (init +inf), 2 = MAX (init -inf) — the
 * distributive ops the two-phase rewrite emits (sql_utils.py:299-413). */
/* n_inserted_dev (u64, nullable): incremented once per NEW group claimed —
 * the host reads it after each batch and rebuilds into a larger table
 * before cumulative distinct keys approach capacity (the find-or-insert
 * probe loop would otherwise spin forever on a full table). */
int qk_groupby_i64_sum(void *stream, uint64_t n, const int64_t *keys,
                       const void *vals_dev /* dev double*[nvals] */,
                       const int32_t *agg_ops_dev /* NULL = all SUM */,
                       int nvals, int rstride /* pow2 words/slot >= 1+nvals */,
                       int64_t *table, uint64_t cap, uint64_t *n_inserted);
/* Initialize an INTERLEAVED group-by table: every slot record =
 * [key=EMPTY | v0..v(nvals-1) = inits[c] | pad] of rstride 8-byte words
 * (pow2 so a record never straddles a 128 B HBM line: find-or-insert
 * touches ONE random line per row, not one per array). */
int qk_groupby_init(void *stream, int64_t *table, uint64_t cap, int rstride,
                    int nvals, const double *inits_dev);
int qk_fill_f64(void *stream, double *dst_dev, double value, uint64_t n);
/* Compact occupied slots to out_keys/out_sums (unordered); out_cursor_dev
 * (u64, zeroed) = number of groups. out capacity must be >= group count. */
int qk_groupby_extract(void *stream, const int64_t *table, int rstride,
                       int nvals, uint64_t cap, int64_t *out_keys,
                       double *out_sums /* column-major, stride out_cap */,
                       uint64_t out_cap, uint64_t *cursor);
/* Thresholded extract: only groups whose sums[col] > threshold are
 * compacted (HAVING clauses, e.g. Q18's sum(l_quantity) > 300 over
 * ~n_orders groups — d2h of only the qualifying handful). */
int qk_groupby_extract_gt(void *stream, const int64_t *table, int rstride,
                          int nvals, uint64_t cap, int col, double threshold,
                          int64_t *out_keys, double *out_sums,
                          uint64_t out_cap, uint64_t *cursor);

/* ---- hash partition (shuffle map side) -------------------------------- *
 * Replaces partition_key_str (quokka_runtime.py:217-231). Int key semantics
 * bit-exact with the reference (:222): part = key % nparts (non-negative
 * keys). hist_dev: u64[nparts], zeroed. */
int qk_partition_hist(void *stream, uint64_t n, const int64_t *keys,
                      uint32_t nparts, uint64_t *hist_dev);
/* Scatter row indices grouped by partition: out_idx[offsets[p] ... ] = rows
 * of partition p (order within a partition unspecified). cursors_dev must be
 * initialized to the exclusive prefix sum of hist (u64[nparts]). */
int qk_partition_scatter(void *stream, uint64_t n, const int64_t *keys,
                         uint32_t nparts, uint64_t *cursors_dev,
                         uint32_t *out_idx);

/* ---- stable radix sort ------------------------------------------------ *
 * Replaces the sorts the reference delegates to polars
 * (SuperFastSortExecutor sql_executors.py:88-187; build-side sort :369;
 * order-by tails). Stable ascending LSD sort of (u64 key, u32 payload);
 * keys_tmp/pay_tmp: caller scratch of same sizes; npasses -1 derives the
 * pass count from the max key. Order-preserving key maps for f64/i64 and
 * an iota payload helper included. */
int qk_sort_pairs_u64(void *stream, uint64_t n, uint64_t *keys,
                      uint32_t *payload, uint64_t *keys_tmp,
                      uint32_t *pay_tmp, int npasses);
int qk_map_f64_u64(void *stream, uint64_t n, const double *in, uint64_t *out);
int qk_map_i64_u64(void *stream, uint64_t n, const int64_t *in,
                   uint64_t *out);
int qk_iota_u32(void *stream, uint64_t n, uint32_t *out);
int qk_bnot_u64(void *stream, uint64_t n, uint64_t *x); /* stable desc sort */

/* ---- hiprtc JIT (arbitrary predicates / transforms / partial aggs) --- *
 * Runtime-compiled gfx950 kernels for the reference's ARBITRARY
 * filter_sql predicates (core.py:157-170), transform_sql expressions
 * (datastream.py:652-815) and folded map-side batch aggregates
 * (df.py:1354-1394) — quokka_amd/jit.py translates the SQL grammar to
 * the C expressions these take. Compilation needs no GPU; module load
 * is lazy at first run. coltypes: 0=i32, 1=f64, 2=u8, 3=i64.
 * qk_jit_last_error() returns the hiprtc log on compile failure. */
const char *qk_jit_last_error(void);
int qk_jit_filter_build(const char *c_expr, int ncols, const int *coltypes,
                        void **prog_out);
int qk_jit_filter_run(void *prog, void *stream, uint64_t n,
                      const void *const *col_ptrs, uint32_t *out_idx,
                      uint64_t *count_dev);
int qk_jit_map_build(const char *c_expr, int ncols, const int *coltypes,
                     void **prog_out);
int qk_jit_map_run(void *prog, void *stream, uint64_t n,
                   const void *const *col_ptrs, double *out_dev);
/* group_expr: C int expression over v<i>; ngroups*naggs <= 64 (register
 * accumulators; use qk_groupby_* for high cardinality). out accumulates
 * ngroups*naggs f64 (row-major [group][agg]). agg_ops (or NULL = all
 * SUM): 0=SUM, 1=MIN, 2=MAX per aggregate — the caller must initialize
 * `out` to the matching identity (0 / +inf / -inf) before the first
 * run; untouched MIN/MAX groups keep the identity. */
int qk_jit_agg_build(const char *pred_or_empty, const char *group_expr,
                     int ngroups, int naggs, const char *const *agg_exprs,
                     const int *agg_ops, int ncols, const int *coltypes,
                     void **prog_out);
int qk_jit_agg_run(void *prog, void *stream, uint64_t n,
                   const void *const *col_ptrs, double *out_dev);
int qk_jit_filter_free(void *prog);   /* frees any qk_jit_* program */
uint64_t qk_jit_code_size(void *prog);

/* ---- RCCL exchange (multi-GPU hash repartition) ----------------------- *
 * Replaces the reference's shuffle data plane (core.py:276-376 push ->
 * Arrow Flight do_put/do_get, flight.py) for the join/group-by
 * repartition: per-rank column buffers, partition-ordered by
 * `key % world` (quokka_runtime.py:222), exchanged with grouped
 * ncclSend/ncclRecv over xGMI — direct per-peer sends (7 p2p links/GPU),
 * not a ring (SURVEY.md §5 backend note). One rank per GPU; the 128-byte
 * unique id is broadcast host-side (gloo) by the caller. */
#define QK_UID_BYTES 128
int qk_comm_unique_id(uint8_t out[QK_UID_BYTES]);
int qk_comm_init(int rank, int world, const uint8_t uid[QK_UID_BYTES],
                 void **comm_out);
int qk_comm_destroy(void *comm);
/* All-to-all of one column: for each peer p, send send_counts[p] elements
 * of `elem_size` bytes from send_buf + send_offsets[p]*elem_size and
 * receive recv_counts[p] into recv_buf + recv_offsets[p]*elem_size.
 * Counts/offsets are HOST arrays (exchanged by the caller beforehand).
 * Grouped send/recv, completes on `stream`. */
int qk_alltoallv(void *stream, void *comm, int world, uint32_t elem_size,
                 const void *send_buf, const uint64_t *send_offsets,
                 const uint64_t *send_counts, void *recv_buf,
                 const uint64_t *recv_offsets, const uint64_t *recv_counts);
/* All-reduce SUM of an f64 device buffer (tiny partial-agg combine). */
int qk_allreduce_f64(void *stream, void *comm, double *buf, uint64_t n);

/* ---- GPU Parquet page decode ------------------------------------------ *
 * The reference scans Parquet with pyarrow's CPU reader
 * (pyquokka/dataset.py InputParquetDataset.get_next_batch ->
 * pyarrow.parquet). Here the host parses only METADATA (footer via
 * pyarrow; per-page Thrift headers and RLE run boundaries in
 * quokka_amd/parquet_thrift.py + parquet_gpu.py) and the raw column
 * chunk bytes are uploaded once; these kernels do all VALUE decoding in
 * HBM. Host-built tile tables (uploaded as u64 arrays) give each
 * workgroup one bounded slice of work.
 *
 * PLAIN pages (fixed-width): tiles = ntiles x 3 u64
 * [src_byte_off, dst_elem_off, count]; elem-wise copy of `count` values
 * of `elem_size` bytes from src_bytes+src_byte_off (byte-aligned, may be
 * unaligned) to dst + dst_elem_off. */
int qk_pq_plain_copy(void *stream, uint64_t ntiles, const uint64_t *tiles,
                     const void *src_bytes, void *dst, uint32_t elem_size);
/* RLE/bit-packed hybrid dictionary indices (parquet encoding.md),
 * parsed AND expanded on-device, one wave per page (low-cardinality
 * columns emit millions of tiny runs — host-side run parsing cost
 * seconds, so the host ships one descriptor per page): ents = npages x
 * 6 u64 [src_byte_off (first run header, after the bit-width byte),
 * src_byte_end, dst_off, count, bit_width, idx_off]. Emits u32
 * (index + idx_off) — idx_off rebases each chunk's dictionary into a
 * column-global concatenated dictionary so a whole multi-row-group
 * column expands in ONE launch. bw 0 => all idx_off. src_bytes needs
 * >= 8 bytes of slack after the end. */
int qk_pq_rle_pages(void *stream, uint64_t npages, const uint64_t *ents,
                    const uint8_t *src_bytes, uint32_t *out);
/* ---- device string dictionary ---------------------------------------- *
 * General variable-width string keys for join/group-by (the reference
 * joins/groups on arbitrary key types via polars, sql_executors.py:
 * 325-377/:556-599). Open-addressing table keyed by a 64-bit byte hash
 * with EXACT byte verification against an on-device arena; assigns dense
 * u32 codes, consistent across calls because table+arena persist. Equal
 * strings => equal codes (bytes verified), so string joins/group-bys run
 * on the existing integer kernels over the codes.
 * slot_hash u64[cap] zeroed (0 = empty), slot_code i32[cap] filled -1,
 * code_off u64 / code_len u32 per code, arena + arena_cursor_dev (u64),
 * counter_dev (u32) = number of codes. HOST must guarantee, before each
 * call: table load < 1/2 after worst-case n new codes, code arrays hold
 * counter + n, arena holds cursor + sum(len). */
int qk_str_dict_encode(void *stream, uint64_t n, const int64_t *offsets,
                       const uint8_t *bytes, uint64_t *slot_hash,
                       int32_t *slot_code, uint64_t capacity,
                       uint64_t *code_off, uint32_t *code_len,
                       uint8_t *arena, uint64_t *arena_cursor_dev,
                       uint32_t *counter_dev, uint32_t *out_codes);
/* Growth: re-seat (hash, code) pairs into a larger zeroed/-1 table;
 * codes and arena unchanged (previously returned codes stay valid). */
int qk_str_dict_rehash(void *stream, uint32_t ncodes,
                       const uint64_t *code_off, const uint32_t *code_len,
                       const uint8_t *arena, uint64_t *slot_hash,
                       int32_t *slot_code, uint64_t capacity);
/* Composite i64 key column: out = x * scale + y (per-(a,b) group keys,
 * e.g. Q20's (partkey, suppkey) quantity sums). */
int qk_i64_combine(void *stream, uint64_t n, const int64_t *x,
                   const int64_t *y, int64_t scale, int64_t *out);
/* Arithmetic shift right: out = x >> shift (decompose pow2-scaled
 * composite keys on device, e.g. pair key -> orderkey). */
int qk_i64_shr(void *stream, uint64_t n, const int64_t *x, int shift,
               int64_t *out);
/* Synchronous device-to-device copy (dict growth, arena relocation). */
int qk_d2d(void *dst_dev, const void *src_dev, uint64_t nbytes);

/* GPU snappy decompression of Parquet pages (the reference reads
 * compressed files transparently through pyarrow,
 * unordered_readers.py:51). One wave per page, pages independent.
 * descs_dev: npages x 8 u64 [src_off, src_len, dst_off,
 * uncompressed_len, mode (0 raw / 1 verify v1 def-levels max_def==1),
 * num_values, 0, 0]; out_dev: npages x 4 i64 [data_off_rel (start of
 * values after the in-page levels block), err (0 ok / 1 corrupt /
 * 2 length mismatch / 3 nulls present), first_byte_after_levels (the
 * RLE bit-width byte, -1 if none), 0]. */
int qk_snappy_pages(void *stream, uint64_t npages, const uint64_t *descs_dev,
                    const uint8_t *src_dev, uint8_t *dst_dev,
                    int64_t *out_dev);
/* GPU gzip (DEFLATE) decompression of Parquet pages: entropy decode is
 * sequential, so lane 0 decodes a page's bit stream (canonical Huffman
 * tables built in LDS) and pages decode independently in parallel.
 * descs_dev: npages x 8 u64 [src_off, src_len, dst_off,
 * uncompressed_len, mode (1 = gzip wrapper / 0 = raw deflate), 0,0,0];
 * out_dev: npages x 4 i64 [written, err (0 ok/1 corrupt/2 length
 * mismatch/3 unsupported), first_byte, 0]. */
int qk_gzip_pages(void *stream, uint64_t npages, const uint64_t *descs_dev,
                  const uint8_t *src_dev, uint8_t *dst_dev,
                  int64_t *out_dev);
/* Host-side Thrift compact-protocol walk of one column chunk's page
 * headers (parquet-format PageHeader; the metadata side of the decode —
 * pyarrow exposes only the footer, and walking thousands of headers in
 * Python measured ~0.2 s/GB of file, dominating warm end-to-end scans).
 * out: max_pages x 10 i64 [kind(0 data v1/2 dict/3 data v2), num_values,
 * encoding, def_level_encoding(-1 none/3 RLE), data_off(abs),
 * data_len(compressed), v2_levels_len, num_nulls, uncompressed_len, 0];
 * *n_out = pages written. Walks until num_values data values covered. */
#define QK_PQ_PAGE_FIELDS 10
int qk_pq_walk_pages(const uint8_t *buf, uint64_t start,
                     uint64_t total_len, int64_t num_values, int64_t *out,
                     int64_t max_pages, int64_t *n_out);

/* ---- GPU CSV parse ---------------------------------------------------- *
 * The reference's CSV scan splits files into byte ranges and decodes on
 * the CPU with polars.read_csv (unordered_readers.py:273-442, :438).
 * Here the raw bytes live in HBM: qk_csv_newlines builds the ORDERED
 * newline index (u64 positions; count written to out_count_dev), then
 * qk_csv_parse decodes one row per thread. coltypes: 0=i64, 1=f64
 * (decimal, <=15 significant digits — parsed as exact-int / exact
 * power-of-ten division, BIT-EXACT vs strtod; more digits = row error),
 * 2=date32 ("YYYY-MM-DD" -> days since epoch), 3=u8 dictionary code
 * (candidates as first-8-bytes-LE u64 + length, <=32 per column,
 * unknown value = row error), 4=skip. Trailing '\r' stripped; trailing
 * separators (dbgen .tbl) tolerated. err_row: device u64 initialized to
 * UINT64_MAX; the LOWEST failing row index lands there. */
/* Range-partition ids (quokka_runtime.py:234-243 partition_key_range):
 * id = (key - 1) / (total_range / nparts), floor division, clamped into
 * [0, nparts-1] (the reference misroutes keys outside [1, total_range];
 * we clamp — documented divergence, same key->same channel). Feed the
 * ids to qk_partition_hist/scatter (id % nparts == id). */
int qk_range_part_ids(void *stream, uint64_t n, const int64_t *keys,
                      int64_t per_range, uint32_t nparts, int64_t *out);

#define QK_CSV_MAX_DICT 32
int qk_csv_newlines(void *stream, uint64_t data_start, uint64_t n,
                    const uint8_t *bytes, uint64_t *out_pos,
                    uint64_t *out_count_dev);
/* Quote-aware variant (RFC-4180): a newline ends a row only when the
 * count of `quote` chars before it is even ("" escapes toggle twice and
 * cancel). Slower 5-pass path; callers use it only when the buffer
 * contains the quote char at all. */
int qk_csv_newlines_quoted(void *stream, uint64_t data_start, uint64_t n,
                           const uint8_t *bytes_dev, uint8_t quote_char,
                           uint64_t *out_pos_dev, uint64_t *out_count_dev);
int qk_csv_parse(void *stream, uint64_t nrows, const uint8_t *bytes,
                 uint64_t data_start, const uint64_t *nl_pos, uint8_t sep,
                 uint8_t quote_char /* 0 = no quoting */, int ncols,
                 const int *coltypes, void *const *out_ptrs,
                 const uint64_t *dict_cands, const uint8_t *dict_lens,
                 const int *ncands, uint64_t *err_row);

#ifdef __cplusplus
}
#endif
#endif /* QUOKKA_AMD_H */
