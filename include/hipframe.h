/* hipframe.h — C-ABI of libhipframe.so: the MI355X-native partition-operator
 * engine behind the modin_amd backend.
 *
 * This boundary replaces Modin's per-partition "engine deploy" call: in the
 * reference, every operator template executes a pandas kernel on a partition's
 * pandas.DataFrame through
 *   modin/core/execution/python/common/engine_wrapper.py:17-42 (PythonWrapper.deploy)
 *   modin/core/dataframe/pandas/partitioning/partition.py:114   (apply)
 * Here the partition payload is a set of device-resident column buffers
 * (hf_col) on one MI355X, and each operator template dispatches one of the
 * entry points below (hand-written gfx950 HIP kernels) instead of a pandas
 * call.  Host-side callers bind via ctypes (see modin_amd/core/lib.py and
 * INTEGRATION.md for the stub a Modin maintainer would add).
 *
 * Conventions:
 *   - every function returns int status (0 = HF_OK, nonzero = error) unless
 *     stated; hf_last_error() gives a thread-local message for the last
 *     failure.
 *   - all kernels run on one module-owned HIP stream per process;
 *     hf_sync() drains it.  Host calls are not thread-safe (Modin's L4/L5
 *     calls all arrive on the user thread — partition_manager.py semantics).
 *   - ownership: every hf_col* returned by the library is owned by the caller
 *     and freed with hf_col_free().
 *   - there is NO CPU fallback behind any entry point: on a machine without a
 *     visible GPU, hf_init fails and every compute call after a failed init
 *     fails loudly.
 */
#ifndef HIPFRAME_H
#define HIPFRAME_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status ---- */
enum {
  HF_OK = 0,
  HF_ERR_HIP = 1,        /* HIP runtime error; hf_last_error() has detail   */
  HF_ERR_ARG = 2,        /* bad argument (dtype/len mismatch, null, range)  */
  HF_ERR_NOINIT = 3,     /* hf_init not called / failed                     */
  HF_ERR_UNSUPPORTED = 4 /* op/dtype combination not implemented            */
};

/* ---- dtypes (column element types; SoA device layout) ---- */
enum {
  HF_INT64 = 0,
  HF_FLOAT64 = 1
};

/* Opaque column: device pointer + dtype + length + owning GPU.
 * Replaces the per-partition pandas block of
 * modin/core/execution/python/.../partition.py:22 (partition wraps a concrete
 * pandas.DataFrame); here a partition wraps a set of hf_col. */
typedef struct hf_col hf_col;

/* ---- lifecycle ---- */
int  hf_init(int gpu);              /* select device, create stream + pool   */
int  hf_shutdown(void);
int  hf_device_count(int* out);     /* works without init                    */
const char* hf_last_error(void);
int  hf_sync(void);                 /* drain the module stream               */

/* ---- column memory / transfer ----
 * Replace PandasDataframePartition.put / get:
 *   modin/core/dataframe/pandas/partitioning/partition.py:277 (put)
 *   partition_manager.py:1070 from_pandas -> H2D columnar upload. */
int  hf_put(const void* host, int64_t len, int dtype, hf_col** out); /* H2D  */
int  hf_get(const hf_col* col, void* host);                          /* D2H  */
int  hf_col_alloc(int64_t len, int dtype, hf_col** out);  /* uninitialised   */
int  hf_col_free(hf_col* col);
int64_t   hf_col_len(const hf_col* col);
int       hf_col_dtype(const hf_col* col);
uintptr_t hf_col_dptr(const hf_col* col);  /* raw device ptr (RCCL/torch
                                              interop plumbing only)         */

/* Raw device buffer alloc/free for dense groupby tables when the caller does
 * not hand in an externally allocated (e.g. torch) buffer. */
int  hf_alloc_raw(int64_t bytes, uintptr_t* dptr);
int  hf_free_raw(uintptr_t dptr);
int  hf_memset_raw(uintptr_t dptr, int value, int64_t bytes);

/* ---- Map: elementwise scalar ops ----
 * Device form of the Map operator template
 * (modin/core/dataframe/algebra/map.py:28-70 + query_compiler.py:2036 abs,
 * :2710 fillna, binary.py:449 scalar branch): one coalesced HBM scan,
 * out[i] = op(in[i], scalar).  f64 and i64 columns. */
enum {
  HF_MAP_ADD = 0,   /* x + s       */
  HF_MAP_SUB = 1,   /* x - s       */
  HF_MAP_RSUB = 2,  /* s - x       */
  HF_MAP_MUL = 3,   /* x * s       */
  HF_MAP_DIV = 4,   /* x / s  (f64 only)                        */
  HF_MAP_RDIV = 5,  /* s / x  (f64 only)                        */
  HF_MAP_FILLNA = 6,/* isnan(x) ? s : x  (f64 only)             */
  HF_MAP_ABS = 7,   /* |x|         */
  HF_MAP_NEG = 8,   /* -x          */
  HF_MAP_CAST_F64 = 9, /* (double)x : i64 -> f64; scalar ignored */
  HF_MAP_CAST_I64 = 10, /* (int64)x : f64 -> i64, C truncation (astype)    */
  HF_MAP_SQRT = 11, /* sqrt(x), f64 only (std = sqrt(var))                  */
  HF_MAP_MIN = 12,  /* fmin(x, s) — clip upper; NaN passes through (f64)    */
  HF_MAP_MAX = 13,  /* fmax(x, s) — clip lower; NaN passes through (f64)    */
  HF_MAP_ROUND = 14,/* rint(x*s)/s, f64 only (pandas round(d): s = 10^d;
                       half-even like numpy)                               */
  HF_MAP_IDIV = 15, /* Python floordiv(x, s), int64 only — exact (the f64
                       DIV path rounds beyond 2^53; datetime calendar math
                       needs exactness)                                    */
  HF_MAP_IMOD = 16  /* Python mod(x, s) (sign of s), int64 only           */
};
int hf_map_scalar(int op, const hf_col* in, double scalar, hf_col** out);
/* i64 column with an exact int64 scalar (double cannot hold all int64). */
int hf_map_scalar_i64(int op, const hf_col* in, int64_t scalar, hf_col** out);

/* ---- Binary: elementwise column<op>column ----
 * Device form of algebra/binary.py:293-459 frame branch (n_ary_op zip-apply,
 * dataframe.py:3851) for two row-aligned columns of equal length. */
enum {
  HF_BIN_ADD = 0, HF_BIN_SUB = 1, HF_BIN_MUL = 2, HF_BIN_DIV = 3,
  HF_BIN_MIN = 4, /* fmin / integer min — NaN-skipping (axis=1 folds)   */
  HF_BIN_MAX = 5  /* fmax / integer max                                  */
};
int hf_binary(int op, const hf_col* a, const hf_col* b, hf_col** out);

/* ---- TreeReduce: full-column reductions ----
 * Device form of TreeReduce map+reduce (algebra/tree_reduce.py:29-82,
 * dataframe.py:2208-2250; pandas nan-skipping semantics of
 * DataFrame.sum/count/min/max, query_compiler.py:984).
 * One pass returns all four partials so mean/count need no extra scan:
 *   sum   = sum of non-NaN elements (0.0 if none — pandas sum min_count=0)
 *   count = number of non-NaN elements
 *   mn/mx = min/max over non-NaN (caller maps count==0 -> NaN)
 * For HF_INT64 columns the partials are exact (sum may wrap like pandas/numpy
 * int64); mn/mx returned via the f64 slots losslessly for |x| < 2^53 and via
 * imn/imx exactly. */
typedef struct {
  double  sum;
  int64_t count;
  double  mn, mx;     /* valid when count > 0 */
  int64_t isum, imn, imx; /* exact int64 partials for HF_INT64 columns */
} hf_reduce_result;
int hf_reduce(const hf_col* in, hf_reduce_result* out); /* syncs the stream */

/* ---- GroupByReduce: dense-key hash/array aggregation ----
 * Device form of GroupByReduce.map / .reduce
 * (modin/core/dataframe/algebra/groupby.py:124-208 map = per-partition
 *  groupby(...).sum(); :211-300 reduce = concat partials + groupby(level=0))
 * redesigned MI355X-first: the "map" phase accumulates every partition into
 * ONE dense key-indexed table per GPU (keys must lie in
 * [key_min, key_min + n_slots)); the "reduce" phase is a table merge (RCCL
 * all-reduce across GPUs — done by the caller over the raw table buffers) +
 * on-device compaction to sorted present keys.
 *
 * Table layout (all device memory, caller-allocated so it can live in an
 * RCCL-reducible tensor):
 *   sums   : double[nvals][n_slots]   per-value-column NaN-skipping sums
 *   rowcnt : int64 [n_slots]          rows seen per key (NaN rows included —
 *                                     defines group presence like pandas)
 *   counts : int64 [nvals][n_slots]   non-NaN count per value column
 *                                     (optional: pass 0 to skip; needed for
 *                                     count/mean)
 * All buffers must be zeroed by the caller before the first accumulate. */
enum { HF_AGG_SUM = 0, HF_AGG_MIN = 1, HF_AGG_MAX = 2,
       HF_AGG_PROD = 3 /* scans only (hf_cumsum / hf_seg_cumsum) */ };
/* agg_op picks the per-slot combine for the value table (the "sums" buffer
 * doubles as the min/max table; initialize it to 0 / +inf / -inf with
 * hf_fill_f64 before the first accumulate — GroupbyReduceImpl's other
 * map/reduce pairs, storage_formats/pandas/groupby.py:237-248). */
int hf_groupby_accum(const hf_col* keys,            /* HF_INT64, len n        */
                     const hf_col* const* vals,     /* nvals HF_FLOAT64 cols  */
                     int nvals, int agg_op,
                     int64_t key_min, int64_t n_slots,
                     uintptr_t sums, uintptr_t rowcnt, uintptr_t counts);

int hf_fill_f64(uintptr_t dptr, double value, int64_t n);
int hf_fill_i64(uintptr_t dptr, int64_t value, int64_t n);

/* ---- deterministic device-side random fills (synthetic data generation:
 * the bench's frames are born in HBM; oracle.rand_* mirrors the splitmix64
 * formulas bit-exactly so the verify gate can regenerate expectations on the
 * host — the device analog of the reference benchmarks' np.random frames,
 * e.g. modin/tests/pandas/test_groupby.py fixtures).
 *   randint : col[i] = lo + splitmix64(seed+i) % (hi-lo)     (HF_INT64)
 *   randf64 : col[i] = (splitmix64(seed+i) >> 11) * 2^-53    (HF_FLOAT64)
 *   randcdf : u as randf64; col[i] = searchsorted(cdf, u, side='right')
 *             over a sorted f64 cdf column (zipf & friends)  (HF_INT64) */
int hf_fill_randint(hf_col* col, uint64_t seed, int64_t lo, int64_t hi);
int hf_fill_randf64(hf_col* col, uint64_t seed);
int hf_fill_randcdf(hf_col* col, uint64_t seed, const hf_col* cdf);

/* ---- hash-table groupby (unbounded key ranges, single rank) ----
 * Open-addressing table of H power-of-2 slots (+1 special slot for the
 * INT64_MIN sentinel key): tkey i64[H+1] initialised to INT64_MIN via
 * hf_fill_i64; sums f64[nvals][H+1] to the agg identity; rowcnt/counts
 * i64[...] zeroed.  Insert = splitmix64 probe + atomicCAS claim; a probe
 * sweep that wraps the table reports "hash table full" (caller grows H and
 * retries).  Compaction filters present slots, SORTS the surviving keys
 * (the wide radix sort), and gathers — same output shape as the dense
 * compact.  Cross-rank hash groupby needs the shuffle exchange (later
 * round); the dense table remains the multi-GPU path. */
int hf_groupby_hash_accum(const hf_col* keys, const hf_col* const* vals,
                          int nvals, int agg_op, int64_t H,
                          uintptr_t tkey, uintptr_t sums, uintptr_t rowcnt,
                          uintptr_t counts);
int hf_groupby_hash_compact(uintptr_t tkey, uintptr_t sums, uintptr_t rowcnt,
                            uintptr_t counts, int nvals, int64_t H,
                            hf_col** out_keys, hf_col** out_sums,
                            hf_col** out_counts, int64_t* n_groups);

/* ---- sort-based general groupby (any cardinality, single rank) ----
 * Input columns must already be KEY-SORTED (hf_sort_perm + hf_gather);
 * runs of equal keys become groups: run heads are detected in place, a
 * tile scan assigns run ids, and per-run aggregates accumulate with one
 * device atomic per row (the extreme-cardinality fallback — runs are short
 * there, so contention is low). */
int hf_groupby_sorted(const hf_col* sorted_keys, const hf_col* const* vals,
                      int nvals, int agg_op, int want_counts,
                      hf_col** out_keys, hf_col** out_sums,
                      hf_col** out_counts, int64_t* n_groups);

/* out[i] = count[i] != 0 ? val[i] : NaN — min/max of an empty (all-NaN)
 * group is NaN in pandas. */
int hf_fixup_empty(const hf_col* val, const hf_col* cnt, hf_col** out);

/* Compact a (merged) table to pandas-groupby-shaped output: ascending present
 * keys (rowcnt>0), per-column sums, optional counts.  Returns n_groups and
 * caller-owned output columns (out_counts may be NULL if counts==0). */
int hf_groupby_compact(uintptr_t sums, uintptr_t rowcnt, uintptr_t counts,
                       int nvals, int64_t key_min, int64_t n_slots,
                       hf_col** out_keys,          /* HF_INT64 [n_groups]     */
                       hf_col** out_sums,          /* nvals cols, caller array*/
                       hf_col** out_counts,        /* nvals cols or NULL      */
                       int64_t* n_groups);

/* ---- Join: broadcast-right hash (dense-range CSR) inner join ----
 * Device form of MergeImpl.row_axis_merge
 * (modin/core/storage_formats/pandas/merge.py:104-178: materialize the right
 * frame once, broadcast it to every left partition, per-partition
 * pandas.merge).  Here the "broadcast right" is a device-resident CSR over
 * the dense key range [key_min, key_min+n_slots): per-key row lists in
 * right-row order (pandas inner-merge match order), with the right value
 * columns gathered into CSR order so probing streams sequentially.
 * The probe preserves pandas semantics: output rows ordered by left row,
 * then right row within a key; result index is a fresh RangeIndex. */
typedef struct hf_join hf_join;

int hf_join_build(const hf_col* rkeys,          /* HF_INT64                 */
                  const hf_col* const* rvals,   /* nr HF_FLOAT64 columns    */
                  int nr, int64_t key_min, int64_t n_slots, hf_join** out);
int hf_join_free(hf_join* j);

/* Probe with a left partition.  Outputs: out_keys (match keys), out_lidx
 * (left row index per match, for gathering left payload columns), and the
 * nr right columns in match order.  Caller owns all returned columns. */
int hf_join_probe(const hf_join* j, const hf_col* lkeys,
                  hf_col** out_keys,            /* HF_INT64 [n_out]         */
                  hf_col** out_lidx,            /* HF_INT64 [n_out]         */
                  hf_col** out_rcols,           /* nr cols, caller array    */
                  int64_t* n_out);

/* Gather col[idx[i]] -> out[i] (materialize left payload columns of a join
 * result; also the generic `take` kernel). */
int hf_gather(const hf_col* col, const hf_col* idx, hf_col** out);

/* Concatenate columns device-to-device (the device form of the partition
 * concat in deploy_axis_func, axis_partition.py:449 — used to materialize a
 * multi-partition right frame for the broadcast join, combine() at
 * dataframe.py:2918). */
int hf_col_concat(const hf_col* const* cols, int ncols, hf_col** out);

/* Row slice [start, start+len) of a column (head/tail/iloc ranges). */
int hf_col_slice(const hf_col* col, int64_t start, int64_t len, hf_col** out);

/* ---- Compare + Filter (df[df.v > x], dropna) ----
 * Compare is a Map-shaped elementwise kernel producing an int64 0/1 mask
 * (query_compiler comparison bindings feed Binary.register in the
 * reference; the mask column is the device form of the boolean Series).
 * Filter is the device form of `PandasDataframe.filter`/`mask`
 * (partition.py:224): a plan (per-tile kept-row offsets from one mask scan)
 * applied per column as a ballot-ranked compaction, preserving row order;
 * hf_filter_iota materializes the kept original positions (the pandas
 * result index). */
enum {
  HF_CMP_GT = 0, HF_CMP_GE = 1, HF_CMP_LT = 2, HF_CMP_LE = 3,
  HF_CMP_EQ = 4, HF_CMP_NE = 5,
  HF_CMP_NOTNA = 6  /* x == x (dropna/notna; scalar ignored) */
};
int hf_compare_scalar(int op, const hf_col* col, double scalar,
                      hf_col** out);  /* int64 0/1 mask; NaN compares false
                                         except NE (pandas semantics) */

typedef struct hf_filterplan hf_filterplan;
int hf_filter_plan(const hf_col* mask, hf_filterplan** out, int64_t* n_kept);
int hf_filter_apply(const hf_filterplan* plan, const hf_col* col,
                    hf_col** out);
int hf_filter_iota(const hf_filterplan* plan, int64_t base, hf_col** out);
int hf_filter_plan_free(hf_filterplan* plan);

/* ---- Sort: stable permutation by an int64 key column ----
 * Device form of PandasDataframe.sort_by (dataframe.py:2742 ->
 * _apply_func_to_range_partitioning), restricted to one
 * int64 key with bounded range (key range <= 2^27) this round: an LSD radix
 * sort over packed (u32 shifted-key, u32 origin) pairs, 8-bit digits,
 * wave-private tiles (in-order lanes make each pass stable, so the result
 * equals pandas sort_values(kind="stable")).  ascending=0 sorts by
 * (key_max - key), which preserves pandas' stable-descending tie order.
 * Returns the permutation as an int64 column of original positions; the
 * caller gathers payload columns and the index with hf_gather. */
int hf_sort_perm(const hf_col* keys, int ascending, hf_col** out_perm);

/* ---- multi-GPU range shuffle (device form of the reference's
 * RangePartitioning shuffle, partition_manager.py:1937 ``shuffle_partitions``
 * / experimental range-partitioning groupby): per-row destination-rank
 * binning against host-supplied SORTED splitters.
 * dest[i] = #{j : splitters[j] <= key[i]}  (int64 compares — exact over the
 * full key range; groups never straddle a boundary because the rule depends
 * only on the key value).  nsplit in [0, 63]; keys int64. */
int hf_shuffle_dest(const hf_col* keys, const int64_t* splitters, int nsplit,
                    hf_col** dest);

/* Order-preserving f64 <-> i64 bit transform (total order trick: flip all
 * bits of negatives, flip only the sign of non-negatives), so float sort /
 * groupby / merge keys ride the int64 radix machinery.  -0.0 normalizes to
 * +0.0 first (pandas groups them together); NaN maps above +inf (pandas
 * na_position='last').  direction=0: f64 col -> ordered i64 col;
 * direction=1: ordered i64 col -> f64 col (inverse). */
int hf_ordered_i64(const hf_col* col, int direction, hf_col** out);

/* Inclusive prefix scan down the column (pandas cumsum/cummin/cummax,
 * axis=0; agg_op = HF_AGG_SUM/MIN/MAX): f64 skips NaN (NaN rows stay NaN,
 * later results unaffected), int64 is exact.  Device three-phase scan:
 * per-tile combine -> single-workgroup exclusive tile scan -> per-tile
 * apply. */
int hf_cumsum(const hf_col* col, int agg_op, hf_col** out);

/* SEGMENTED inclusive prefix scan: restart at every row whose `heads` entry
 * is nonzero (int64 0/1 column, same length).  The device form of the
 * pandas groupby transform family (DataFrameGroupBy.cumsum/cummin/cummax,
 * modin/pandas/groupby.py) applied after a stable sort by key: key-run
 * head flags delimit the segments.  NaN rows stay NaN and contribute the
 * identity (pandas skipna); int64 is exact.  The combine is the classic
 * segmented-scan pair operator ((fa,va),(fb,vb)) -> (fa|fb, fb ? vb :
 * comb(va,vb)) — associative, not commutative, so every fold is ordered. */
int hf_seg_cumsum(const hf_col* col, const hf_col* heads, int agg_op,
                  hf_col** out);

/* Inverse-permutation scatter: out[idx[i]] = col[i]; idx must be a
 * permutation of [0, len) (a hf_sort_perm result).  Restores original row
 * order after a sort-then-transform composition — the device analog of the
 * reference reindexing a transform result back to the caller's index. */
int hf_scatter(const hf_col* col, const hf_col* idx, hf_col** out);

/* Cartesian-product gather indices (pandas merge how='cross'):
 * lidx[i] = i / nr, ridx[i] = i % nr over n = nl*nr rows. */
int hf_cross_idx(int64_t nl, int64_t nr, hf_col** lidx, hf_col** ridx);

/* Exact-match binary search: out[i] = j with sorted[j] == keys[i], else -1.
 * Densifies unbounded int64 join keys through the sorted distinct right
 * keys (lower_bound per row), so the dense-range CSR join
 * (hf_join_build's 2^27 slot cap) covers ANY key span with <= 2^27
 * DISTINCT build keys — the device form of the reference's hash-join key
 * lookup (storage_formats/pandas/merge.py row_axis_merge). */
int hf_search_sorted(const hf_col* keys, const hf_col* sorted_uniq,
                     hf_col** out);

/* Raw device-to-device copy on the hipframe stream — the interop bridge to
 * RCCL-visible torch buffers (exchange_splits): hf columns are copied into /
 * out of torch-allocated device tensors by address. */
int hf_memcpy_dd(uintptr_t dst, uintptr_t src, int64_t bytes);

/* ---- profiling (bench.py roofline leg) ----
 * When enabled, every kernel launch is bracketed by HIP events on the module
 * stream; hf_kernel_stats returns the accumulated count and total ms for the
 * named kernel since the last hf_kernel_stats_reset. */
int hf_profiling(int enable);
int hf_kernel_stats(const char* name, int64_t* launches, double* total_ms);
int hf_kernel_stats_reset(void);

#ifdef __cplusplus
}
#endif
#endif /* HIPFRAME_H */
