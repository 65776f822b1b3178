/*
 * gpuq.h — C-ABI of the MI355X-native columnar execution engine (libgpuq.so).
 *
 * This is the drop-in boundary beneath Spark's columnar plugin API: a Scala
 * host layer (SparkSessionExtensions.injectColumnar + ColumnarRule,
 * sql/core/src/main/scala/org/apache/spark/sql/SparkSessionExtensions.scala:168
 * and sql/core/.../execution/Columnar.scala:36-50) binds these entry points
 * over JNI and swaps SortExec / HashAggregateExec / ShuffledHashJoinExec /
 * ShuffleExchangeExec for GPU exec nodes whose doExecuteColumnar() calls
 * land here. The JNI stub a Spark maintainer would add is shown in
 * INTEGRATION.md; in this repo the same ABI is driven by the Python host
 * mirror (spark_amd/) used for tests and benchmarks.
 *
 * Conventions:
 *  - all device pointers are raw HIP device pointers on the CALLER's current
 *    HIP device; `stream` is a hipStream_t (pass 0 for the default stream).
 *    One executor task = one stream; calls are thread-safe per stream.
 *  - the caller owns every buffer, including scratch workspaces, sized via
 *    the gpuq_*_workspace_bytes() helpers. No allocation inside libgpuq.
 *  - column buffers follow the Arrow layout used by Spark's ColumnVector
 *    API (sql/catalyst/src/main/java/org/apache/spark/sql/vectorized/
 *    ColumnVector.java:58-366): a dense data buffer plus an optional
 *    validity bitmap (1 bit/row, LSB-first; NULL bitmap ptr = no nulls).
 *  - return value: 0 on success, non-zero error code; gpuq_last_error()
 *    returns a thread-local message (mapped to exceptions by the host,
 *    mirroring SparkOutOfMemoryError semantics for device OOM).
 */
#ifndef GPUQ_H
#define GPUQ_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- error handling ---- */
#define GPUQ_OK 0
#define GPUQ_ERR_HIP 1          /* underlying HIP call failed           */
#define GPUQ_ERR_INVALID 2      /* bad argument / unsupported combo     */
#define GPUQ_ERR_OVERFLOW 3     /* hash table or output buffer overflow */

const char* gpuq_last_error(void);
int gpuq_device_count(void);

/* ---- optional per-kernel profiling (bench roofline evidence) ----
 * When enabled, each kernel launch is bracketed by HIP events on its own
 * stream; gpuq_kernel_stats synchronizes pending events and returns the
 * accumulated GPU time and launch count for a kernel tag (e.g.
 * "radix_scatter", "radix_hist", "agg_build", "join_probe"). */
void gpuq_profiling(int enable);
void gpuq_kernel_stats_reset(void);
int gpuq_kernel_stats(const char* name, double* total_ms, long long* count);

/* ---- dtypes ---- */
#define GPUQ_INT64 0
#define GPUQ_FLOAT64 1
#define GPUQ_INT32 2

/* One column of a ColumnarBatch (ColumnarBatch.java:61-123 access contract,
 * device-resident). */
typedef struct gpuq_col {
  void* data;                 /* device ptr to dense values              */
  const uint8_t* validity;    /* device ptr to Arrow validity bitmap or NULL */
  int32_t dtype;              /* GPUQ_* */
} gpuq_col;

/* ---------------------------------------------------------------- */
/* Synthetic data generation (bench only; bit-identical to the CPU  */
/* oracle's splitmix64 generator — DESIGN.md §data).                 */
/* ---------------------------------------------------------------- */
int gpuq_gen_i64_range(void* stream, uint64_t seed, uint64_t start,
                       int64_t nrows, uint64_t range, int64_t* out);
int gpuq_gen_f64_unit(void* stream, uint64_t seed, uint64_t start,
                      int64_t nrows, double* out);

/* RangeExec scan feed (basicPhysicalOperators.scala:630):
 * out[i] = start + i*step */
int gpuq_range_i64(void* stream, int64_t nrows, int64_t start, int64_t step,
                   int64_t* out);

/* ---------------------------------------------------------------- */
/* SORT — replaces SortExec (execution/SortExec.scala:75-126): the   */
/* radix-eligible single-key path (canUseRadixSort, SortExec.scala   */
/* :82-83) over an int64 or float64 key column. Produces the sorted  */
/* row permutation; gather materializes payload columns.             */
/* Key encoding on device matches PrefixComparators.java:66-83       */
/* (double bijection) / SignedPrefixComparator (int64); sort is an   */
/* LSB radix over 8-bit digits with the reference's skip-uniform-    */
/* byte optimization (RadixSort.java:108-139) and is STABLE (ties    */
/* keep row order, as the reference's (prefix,pointer) sort does).   */
/* NULL keys are placed per nulls_first in input order               */
/* (SortOrder.scala:35-45 defaults: asc=>first, desc=>last).         */
/* ---------------------------------------------------------------- */

/* workspace bytes for gpuq_sort_perm at nrows (nrows < 2^32) */
int64_t gpuq_sort_workspace_bytes(int64_t nrows);

/* Sort one key column; writes the sorted permutation (row ids into the
 * input) to out_perm[nrows] (uint32) and, if out_keys != NULL, the sorted
 * key column. With key.validity set, NULL rows are placed per nulls_first
 * in input order and out_keys carries the input's raw values at those
 * positions (callers track NULL-ness via the permuted validity). */
int gpuq_sort_perm(void* stream, int64_t nrows, gpuq_col key,
                   int32_t desc, int32_t nulls_first,
                   uint32_t* out_perm, void* out_keys,
                   void* workspace, int64_t workspace_bytes);

/* out[i] = col[perm[i]] — payload materialization after sort/partition. */
int gpuq_gather(void* stream, int64_t nrows, gpuq_col col,
                const uint32_t* perm, void* out);

/* two 8-byte columns through one permutation in one kernel */
int gpuq_gather2_i64(void* stream, int64_t nrows, const void* a, const void* b,
                     const uint32_t* perm, void* out_a, void* out_b);

/* two 8-byte columns through one permutation via an interleaved staging
 * buffer (scratch: nrows * 16 B device): one streaming interleave pass,
 * then ONE random b128 load per row serving both columns — halves the
 * random-request count and line amplification of the plain two-column
 * gather. */
int gpuq_gather2_i64_fast(void* stream, int64_t nrows, const void* a,
                          const void* b, const uint32_t* perm,
                          void* out_a, void* out_b, void* scratch);

/* the two halves separately: the interleave depends only on the payload
 * columns, so a host can run it on a SIDE stream concurrently with the
 * sort passes and hide it entirely (event-sync before the pairs gather) */
int gpuq_interleave2_i64(void* stream, int64_t nrows, const void* a,
                         const void* b, void* pairs);
int gpuq_gather2_pairs(void* stream, int64_t nrows, const void* pairs,
                       const uint32_t* perm, void* out_a, void* out_b);

/* ---------------------------------------------------------------- */
/* HASH AGGREGATE — replaces HashAggregateExec                       */
/* (execution/aggregate/HashAggregateExec.scala:99-151) for          */
/* GROUP BY int64 key -> SUM(float64) / COUNT. Open-address table    */
/* (the GPU analog of BytesToBytesMap, core/.../unsafe/map/          */
/* BytesToBytesMap.java:50-56) keyed by Murmur3(key,42), linear      */
/* probe, device-scope atomics. SUM accumulation order is device     */
/* order => float64 parity within 1e-6 relative (north star); keys   */
/* and COUNT bit-exact. NULL keys form one group; NULL values are    */
/* skipped (Sum.scala:113-141).                                      */
/* ---------------------------------------------------------------- */

/* capacity must be a power of two >= 2 * expected distinct groups.   */
int64_t gpuq_hash_agg_workspace_bytes(int64_t capacity);

/* aggregate ops bitmask */
#define GPUQ_AGG_SUM 1
#define GPUQ_AGG_COUNT 2

/* Aggregate nrows of (key,val) into the workspace table, then compact:
 * writes ngroups rows of (key, key_valid, sum, sum_valid, count) into the
 * caller's output arrays (each sized for max possible groups) and returns
 * the group count via *out_ngroups. Emission order is nondeterministic
 * (the parity checker is order-insensitive, as QueryTest.checkAnswer is).
 * Multiple (key,val) batches can be accumulated before compaction:
 * pass finalize=0 to accumulate only, finalize=1 to also compact.
 * ops: GPUQ_AGG_* bitmask; SUM-only mode (no COUNT) requires non-null
 * values (sum NULL-ness tracking needs COUNT) and allows out_counts=NULL. */
int gpuq_hash_agg_i64_f64(void* stream, int64_t nrows,
                          gpuq_col key, gpuq_col val,
                          void* workspace, int64_t capacity, int32_t first_batch,
                          int32_t finalize, int32_t ops,
                          int64_t* out_keys, uint8_t* out_key_valid,
                          double* out_sums, uint8_t* out_sum_valid,
                          int64_t* out_counts, int64_t* out_ngroups);

/* Partitioned aggregation: same results contract as gpuq_hash_agg_i64_f64
 * for a single batch with NON-NULL values — rows are bucket-partitioned
 * by 13 Murmur bits in ONE non-stable pass (histogram + per-block range
 * reservation + contiguous per-bucket runs), then each 16 K-row chunk
 * aggregates in an LDS table that flushes to the global table mid-chunk
 * if a skewed/boundary chunk overflows it; sidesteps the global-atomic
 * throughput wall at mid/high cardinality and never needs a fallback. */
int64_t gpuq_hash_agg_part_workspace_bytes(int64_t nrows, int64_t capacity);
int gpuq_hash_agg_partitioned(void* stream, int64_t nrows,
                              gpuq_col key, gpuq_col val,
                              void* workspace, int64_t capacity, int32_t ops,
                              int64_t* out_keys, uint8_t* out_key_valid,
                              double* out_sums, uint8_t* out_sum_valid,
                              int64_t* out_counts, int64_t* out_ngroups);

/* Multi-aggregate: one pass computing up to 6 accumulators per group
 * (HashAggregateExec evaluates a list of aggregate expressions,
 * HashAggregateExec.scala:68-76 — e.g. TPC-H Q1's 8 aggregates).
 * spec_ops[j]: 0=SUM(float64 col), 1=COUNT(col), 2=COUNT(*),
 * 3=SUM(int64 col) -> int64 (Sum.scala LongType result; non-ansi
 * overflow wraps), 4=MIN(int64), 5=MAX(int64), 6=MIN(float64),
 * 7=MAX(float64) (Min/Max.scala; float ordering = Java Double.compare:
 * NaN greatest, -0.0 < 0.0);
 * up to 12 specs; spec_cols[j] indexes vals[] (ignored for COUNT(*)). out_accs[j] is a
 * device array per spec: f64 for SUM_F64/MIN_F64/MAX_F64, i64 otherwise.
 * Small tables (cap*(1+nspecs)*8 <= 64 KB) aggregate per-block in LDS
 * first. Value ops skip NULL rows; an all-NULL group emits the op's
 * neutral init (0 for SUM/COUNT, extremes for MIN/MAX) — pair with a
 * COUNT spec to realize SQL NULL results (Sum.scala: NULL iff no
 * non-null input). */
int64_t gpuq_hash_agg_multi_workspace_bytes(int64_t capacity, int32_t nspecs);
int gpuq_hash_agg_multi(void* stream, int64_t nrows, gpuq_col key,
                        const gpuq_col* vals, const int32_t* spec_ops,
                        const int32_t* spec_cols, int32_t nspecs,
                        void* workspace, int64_t capacity,
                        int32_t first_batch, int32_t finalize,
                        int64_t* out_keys, uint8_t* out_key_valid,
                        void* const* out_accs, int64_t* out_ngroups);

/* Composite-key aggregate: GROUP BY (k1..kK), K <= 4 int64 columns with
 * independent NULLability (the reference groups by an UnsafeRow key tuple,
 * UnsafeFixedWidthAggregationMap.java:39). Slots claim by CAS + publish;
 * tuple equality is verified against stored keys, so the full domain is
 * exact. out_keys[c]: device i64 array per key column; out_kmask: one byte
 * per group, bit c = key column c non-NULL. Same spec_ops as
 * gpuq_hash_agg_multi. */
int64_t gpuq_hash_agg_keys_workspace_bytes(int64_t capacity, int32_t nkeys,
                                           int32_t nspecs);
int gpuq_hash_agg_keys(void* stream, int64_t nrows,
                       const gpuq_col* key_cols, int32_t nkeys,
                       const gpuq_col* vals, const int32_t* spec_ops,
                       const int32_t* spec_cols, int32_t nspecs,
                       void* workspace, int64_t capacity,
                       int32_t first_batch, int32_t finalize,
                       void* const* out_keys, uint8_t* out_kmask,
                       void* const* out_accs, int64_t* out_ngroups);

/* ---------------------------------------------------------------- */
/* PARTITION — replaces ShuffleExchangeExec's partition-id + write   */
/* path (exchange/ShuffleExchangeExec.scala:357-470 +                */
/* core/src/main/java/.../shuffle/sort/UnsafeShuffleWriter.java):    */
/* pid = Pmod(Murmur3Hash(key,42), n) (partitioning.scala:328-330),  */
/* stable radix-partition into per-partition contiguous runs. The    */
/* cross-GPU exchange itself is RCCL all-to-all over xGMI, driven by */
/* the host layer (torch.distributed / ncclGroup of send-recv) on    */
/* the per-partition runs this produces.                             */
/* ---------------------------------------------------------------- */

int64_t gpuq_partition_workspace_bytes(int64_t nrows, int32_t num_parts);

/* Computes the stable permutation that groups rows by partition id and the
 * per-partition row counts. out_perm[nrows] (uint32), out_counts[num_parts]
 * (int64, device). Gather columns with gpuq_gather afterwards.
 * num_parts <= 65536 (two radix passes above 256). */
int gpuq_partition_perm(void* stream, int64_t nrows, gpuq_col key,
                        int32_t num_parts, uint32_t* out_perm,
                        int64_t* out_counts,
                        void* workspace, int64_t workspace_bytes);

/* Multi-column partition keys: pid = Pmod(Murmur3Hash(k1..kK, 42), n)
 * with the hash seed-chained column-wise and NULL columns passing the
 * running seed through (hash.scala:849-860 HashExpression.eval;
 * partitioning.scala:328). nkeys <= 4, each int64. */
int gpuq_partition_perm_multi(void* stream, int64_t nrows,
                              const gpuq_col* key_cols, int32_t nkeys,
                              int32_t num_parts, uint32_t* out_perm,
                              int64_t* out_counts,
                              void* workspace, int64_t workspace_bytes);

/* Range partition (global ORDER BY across GPUs): bin = first bound >=
 * key in the sort order (RangePartitioning, ShuffleExchangeExec.scala:
 * 379-400 — the host layer samples keys and picks quantile bounds);
 * NULL keys go to the first/last partition per nulls_first. bounds:
 * device array of nbounds entries, same dtype as the key; out_counts:
 * nbounds+1 entries. Stable, like the hash partition. */
int gpuq_range_partition_perm(void* stream, int64_t nrows, gpuq_col key,
                              int32_t desc, int32_t nulls_first,
                              const void* bounds, int32_t nbounds,
                              uint32_t* out_perm, int64_t* out_counts,
                              void* workspace, int64_t workspace_bytes);

/* ---------------------------------------------------------------- */
/* HASH JOIN — replaces ShuffledHashJoinExec inner join              */
/* (joins/ShuffledHashJoinExec.scala:103-132; build side analog of   */
/* LongHashedRelation, joins/HashedRelation.scala:993 — duplicates   */
/* chained per key). NULL keys never match. Emission order is        */
/* nondeterministic (checker sorts, as QueryTest does).              */
/* ---------------------------------------------------------------- */

int64_t gpuq_join_build_workspace_bytes(int64_t build_rows, int64_t capacity);
int64_t gpuq_join_probe_workspace_bytes(int64_t probe_rows);

/* Build the hash table over the build-side key column. capacity: power of
 * two >= ~1.6*build_rows. The workspace holds the table + chains (and, for
 * large NULL-free tables, the hash-ordered copy of the build side that
 * keeps probes L3-local) and must stay alive through probes. */
int gpuq_join_build_i64(void* stream, int64_t build_rows, gpuq_col build_key,
                        void* workspace, int64_t capacity);

/* Probe: emits matching (probe_rid, build_rid) uint32 pairs into the
 * caller's buffers (out_cap entries each); *out_nmatches returns the total
 * match count. If the count exceeds out_cap, returns GPUQ_ERR_OVERFLOW
 * after setting *out_nmatches (call again with bigger buffers).
 * probe_workspace (gpuq_join_probe_workspace_bytes) enables the
 * hash-ordered probe stream; pass NULL to probe in input order. */
/* join_type (ShuffledHashJoinExec.scala joinType dispatch): 0=Inner,
 * 1=probe-side Outer (unmatched probe rows pair with build rid 0xFFFFFFFF
 * => NULL build columns; gather with gpuq_gather_nullable), 2=LeftSemi
 * (probe row once iff matched; out_b = 0xFFFFFFFF), 3=LeftAnti (iff
 * unmatched; NULL probe keys emit — the non-null-aware anti). */
int gpuq_join_probe_i64_typed(void* stream, int64_t probe_rows, gpuq_col probe_key,
                              const void* workspace, int64_t capacity,
                              int64_t build_rows, void* probe_workspace,
                              int64_t probe_workspace_bytes, int32_t join_type,
                              uint32_t* out_probe_rid, uint32_t* out_build_rid,
                              int64_t out_capacity, int64_t* out_nmatches);

/* inner-join compatibility wrapper (join_type = 0) */
int gpuq_join_probe_i64(void* stream, int64_t probe_rows, gpuq_col probe_key,
                        const void* workspace, int64_t capacity, int64_t build_rows,
                        void* probe_workspace, int64_t probe_ws_bytes,
                        uint32_t* out_probe_rid, uint32_t* out_build_rid,
                        int64_t out_cap, int64_t* out_nmatches);

/* ---------------------------------------------------------------- */
/* FILTER / PROJECT — SURVEY §8(f).2: FilterExec / ProjectExec on    */
/* columnar batches, so multi-operator plans stay on-device.         */
/* Filter: single comparison col OP literal; WHERE keeps only TRUE   */
/* (NULL comparison result drops the row). Stable compaction: the    */
/* first *out_count entries of out_perm are the passing rows in      */
/* input order. Project: elementwise a OP b (column or literal).     */
/* ---------------------------------------------------------------- */

#define GPUQ_CMP_EQ 0
#define GPUQ_CMP_LT 1
#define GPUQ_CMP_LE 2
#define GPUQ_CMP_GT 3
#define GPUQ_CMP_GE 4
#define GPUQ_CMP_NE 5

int64_t gpuq_filter_workspace_bytes(int64_t nrows);
int gpuq_filter_cmp(void* stream, int64_t nrows, gpuq_col col, int32_t op,
                    double lit_f64, int64_t lit_i64,
                    uint32_t* out_perm, int64_t* out_count /* device, u64 */,
                    void* workspace, int64_t workspace_bytes);

#define GPUQ_BINOP_ADD 0
#define GPUQ_BINOP_SUB 1
#define GPUQ_BINOP_MUL 2
#define GPUQ_BINOP_DIV 3
#define GPUQ_BINOP_RSUB 4  /* out = b - a (literal - column) */

int gpuq_project_binop(void* stream, int64_t nrows, gpuq_col a,
                       const void* b /* second column's data or NULL for literal */,
                       double lit_f64, int64_t lit_i64, int32_t op, void* out);

/* int64 -> float64 cast (AVG evaluation: sum / cast(count)) */
int gpuq_cast_i64_f64(void* stream, int64_t nrows, const int64_t* in, double* out);

/* ---------------------------------------------------------------- */
/* VALIDITY BITMAP UTILITIES — move Arrow validity bitmaps           */
/* (LSB-first, ColumnVector null contract, ColumnVector.java:58-366) */
/* through permutations and across the RCCL exchange (bitmaps travel */
/* as u8 columns: all-to-all row splits are not byte-aligned).       */
/* ---------------------------------------------------------------- */

/* out bit i = src bit perm[i] (permute a validity bitmap alongside its
 * data column) */
int gpuq_gather_bits(void* stream, int64_t nrows, const uint8_t* src_bits,
                     const uint32_t* perm, uint8_t* out_bits);

/* bitmap <-> one-byte-per-row (exchange transport form) */
int gpuq_bits_to_u8(void* stream, int64_t nrows, const uint8_t* bits,
                    uint8_t* out);
int gpuq_u8_to_bits(void* stream, int64_t nrows, const uint8_t* u8,
                    uint8_t* out_bits);

/* out[i] = col[perm[i]], with perm[i] == 0xFFFFFFFF producing NULL (the
 * outer-join build side); out_bits = src validity AND not-NIL. */
int gpuq_gather_nullable(void* stream, int64_t nrows, gpuq_col col,
                         const uint32_t* perm, void* out, uint8_t* out_bits);

/* validity bitmap for key column `bit` of a composite-key result:
 * out bit i = (mask[i] >> bit) & 1 (see gpuq_hash_agg_keys out_kmask) */
int gpuq_maskbit_to_bits(void* stream, int64_t nrows, const uint8_t* mask,
                         int32_t bit, uint8_t* out_bits);

/* bit i = (in[i] != 0): merged-COUNT column -> NULL-ness of merged
 * SUM/MIN/MAX results (final-mode aggregation, Sum.scala merge) */
int gpuq_nonzero_to_bits(void* stream, int64_t nrows, const int64_t* in,
                         uint8_t* out_bits);

/* min/max/valid-count reduction of an int64 column. out_dev: device u64[3]
 * = {min encoded (encode_i64), max encoded, count of valid rows}. */
int gpuq_minmax_i64(void* stream, int64_t nrows, gpuq_col col,
                    unsigned long long* out_dev);

/* Pack two narrow int64 key columns into one: out = (a - a_bias) << shift
 * | (b - b_bias). Caller proves the ranges via gpuq_minmax_i64 first.
 * Feeds composite GROUP BY keys into the single-key fast path (per-block
 * LDS tables at low cardinality). */
int gpuq_pack2_i64(void* stream, int64_t nrows, const int64_t* a,
                   const int64_t* b, int64_t a_bias, int64_t b_bias,
                   int32_t shift, int64_t* out);
int gpuq_unpack2_i64(void* stream, int64_t nrows, const int64_t* in,
                     int64_t a_bias, int64_t b_bias, int32_t shift,
                     int64_t* out_a, int64_t* out_b);

#ifdef __cplusplus
}
#endif
#endif /* GPUQ_H */
