/* vmgpu.h — C-ABI of the MI355X-native vmselect rollup/aggregation engine.
 *
 * This is the drop-in boundary described in SURVEY.md §8b: it replaces the
 * worker fan-out seam inside the reference's evalRollupFuncNoCache
 * (app/vmselect/promql/eval.go:1899-1904 → evalRollupWithIncrementalAggregate
 * eval.go:1927 / evalRollupNoIncrementalAggregate eval.go:1968).  Everything
 * above the seam (parse, plan, SearchQuery, rollupResultCache probe, memory
 * limiter) and below it (TSDB block scan) stays in the host process; the host
 * hands decoded columnar series (CSR (timestamps[], values[]) batches, the
 * callback input of netstorage.Results.RunParallel, netstorage.go:219) to
 * this library and receives the rollup grid / aggregated group matrix back.
 *
 * The intended host binding is Go cgo (see INTEGRATION.md for the stub a
 * vmselect maintainer would add); the in-repo host mirror is Python ctypes
 * (victoriametrics_amd/engine.py) because this image carries no Go toolchain.
 *
 * Plain C types only: no HIP or torch types cross this boundary.
 */
#ifndef VMGPU_H
#define VMGPU_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Rollup function ids — numerically identical to the oracle's vm_func_id
 * (oracle/vm_oracle.h) and named after the reference's rollupFuncs map keys
 * (app/vmselect/promql/rollup.go:24-108). */
typedef int32_t vmgpu_func_id;

/* Cross-series incremental aggregate op ids — same numbering as the oracle's
 * vm_aggr_op (aggr_incremental.go:18-66 callback table). */
enum {
  VMGPU_AGGR_NONE = 0,
  VMGPU_AGGR_SUM = 1,
  VMGPU_AGGR_MIN = 2,
  VMGPU_AGGR_MAX = 3,
  VMGPU_AGGR_AVG = 4,
  VMGPU_AGGR_COUNT = 5,
  VMGPU_AGGR_SUM2 = 6,
  VMGPU_AGGR_GEOMEAN = 7,
  VMGPU_AGGR_GROUP = 8,
};

/* Evaluation plan — the fields of promql.rollupConfig (rollup.go:574-606)
 * that shape the computation, plus the preFunc toggles that getRollupConfigs
 * (rollup.go:374-516) would install and the aggregate op. */
#define VMGPU_PRE_NONE 0
#define VMGPU_PRE_DELTA 1            /* deltaValues (rollup.go:963) */
#define VMGPU_PRE_DERIV 2            /* derivValues (rollup.go:979) */
#define VMGPU_PRE_SCRAPE_INTERVAL 3  /* rollup_scrape_interval preFunc */

typedef struct vmgpu_plan {
  int32_t func;                   /* vmgpu_func_id */
  int32_t aggr;                   /* VMGPU_AGGR_* (NONE => per-series output) */
  int64_t start, end, step;       /* grid [start:end:step], ms (eval.go:234) */
  int64_t window;                 /* lookbehind window ms; 0 => auto-adjust */
  int64_t lookback_delta;         /* rc.LookbackDelta */
  int64_t min_staleness_interval; /* -search.minStalenessInterval */
  int64_t max_staleness_interval; /* staleness arg of removeCounterResets
                                   * (= lookback_delta + window when
                                   *  lookback_delta != 0; rollup.go:380-387) */
  int32_t may_adjust_window;      /* rollupFuncsCanAdjustWindow[func] */
  int32_t is_default_rollup;
  int32_t remove_counter_resets;  /* rollupFuncsRemoveCounterResets[func] */
  int32_t drop_stale_nans;        /* dropStaleNaNs unless -search.noStaleMarkers
                                   * or func==default_rollup (eval.go:2108) */
  int32_t samples_scanned_per_call; /* rollupFuncsSamplesScannedPerCall */
  int32_t skip_finalize;          /* leave (values,counts) un-finalized so the
                                   * caller can all-reduce across shards first
                                   *  */
  int32_t pre_func;               /* VMGPU_PRE_*: per-series value transform
                                   * applied after removeCounterResets, for
                                   * the rollup_rate/rollup_delta/
                                   * rollup_scrape_interval families
                                   * (getRollupConfigs, rollup.go:436-516) */
  double  arg;                    /* phi / le / gt / eq / secs for arg funcs */
  double  arg2;                   /* second scalar (holt_winters tf) */
} vmgpu_plan;

/* Select the device and create the library context.  One process drives one
 * GPU (one process per GPU over RCCL is the scaling model); n_devices must
 * currently be 1.  Returns 0 on success. */
int vmgpu_init(const int* device_ids, int n_devices);
int vmgpu_shutdown(void);

/* Upload a decoded-series batch (CSR layout: series s occupies
 * [offsets[s], offsets[s+1]) in ts/vals; offsets has n_series+1 entries;
 * group_ids may be NULL) to the device.  Returns a handle via *out_handle. */
int vmgpu_batch_create(const int64_t* ts, const double* vals,
                       const uint64_t* offsets, uint32_t n_series,
                       const int32_t* group_ids, uint32_t n_groups,
                       uint64_t* out_handle,
                       char* errbuf, size_t errbuf_len);
int vmgpu_batch_destroy(uint64_t handle);

/* Evaluate the plan over an uploaded batch.
 * aggr==NONE: out is [n_series x n_grid] (row-major, n_grid =
 *   1 + (end-start)/step, eval.go:247); out_counts unused.
 * aggr!=NONE: out and out_counts are [n_groups x n_grid]; with
 *   plan->skip_finalize they hold the raw partial matrices (identity-filled
 *   where empty) ready for a cross-shard all-reduce.
 * out/out_counts may be NULL to leave results on the device (fetch with
 * vmgpu_batch_fetch_out) — that is how benches time the kernel without PCIe.
 * Returns 0 on success; nonzero + errbuf otherwise. */
int vmgpu_rollup_exec(const vmgpu_plan* plan, uint64_t handle,
                      double* out, double* out_counts,
                      uint64_t* out_samples_scanned,
                      char* errbuf, size_t errbuf_len);

/* Download the last exec's outputs for this batch (sizes in elements). */
int vmgpu_batch_fetch_out(uint64_t handle, double* dst, size_t n_elems,
                          double* dst_counts, size_t n_count_elems);

/* Apply the aggregate finalize step (finalizeAggrCommon/Avg/Count/Group/
 * Geomean, aggr_incremental.go:141-168 + 189-512) to host matrices — used
 * after the cross-shard all-reduce when exec ran with skip_finalize. */
void vmgpu_aggr_finalize_host(int32_t aggr, double* values, double* counts,
                              uint64_t n_elems);

/* One-shot convenience wrapper (create + exec + destroy), the literal shape
 * a cgo shim binds. */
int vmgpu_rollup_eval(const vmgpu_plan* plan,
                      const int64_t* ts, const double* vals,
                      const uint64_t* offsets, uint32_t n_series,
                      const int32_t* group_ids, uint32_t n_groups,
                      double* out, double* out_counts,
                      uint64_t* out_samples_scanned,
                      char* errbuf, size_t errbuf_len);

/* topk family (newAggrFuncTopK / getRangeTopKTimeseries, aggr.go:646-741)
 * over the last evaluated output of a batch.
 * vmgpu_topk_range: ranks rows by a whole-range summary (summary_op:
 *   0=avg 1=min 2=max 3=median 4=last), returns up to k row ids in the
 *   reference's output order and optionally the per-point remaining sum.
 * vmgpu_topk_pointwise: per-grid-point top/bottom-k across rows; all other
 *   values become NaN (fillNaNsAtIdx); out may be NULL to keep on device. */
int vmgpu_topk_range(uint64_t handle, double k, int32_t summary_op,
                     int32_t reverse, int64_t* out_sel, int64_t* out_n_sel,
                     double* out_remaining, char* errbuf, size_t errbuf_len);
int vmgpu_topk_pointwise(uint64_t handle, double k, int32_t reverse,
                         double* out, char* errbuf, size_t errbuf_len);

/* histogram_quantile (transform.go:992-1137) over grouped le-bucket rows
 * (rows sorted by (group, le), same-le rows pre-merged by the host;
 * group_offsets has n_groups+1 entries).  bucket_values is
 * [rows x n_grid] host memory; outputs are [n_groups x n_grid]. */
/* Non-incremental cross-series aggregates (aggr.go long tail; the
 * incremental ops fuse into the rollup kernels).  Per (group, grid point)
 * over member rows (CSR group_rows/group_offsets into the values matrix):
 * median/quantile/mad/mode/distinct/stddev/stdvar reduce to
 * out[n_groups x n_grid]; share/zscore rewrite member rows into
 * values_out [n_series x n_grid]; iqr_bounds emits lower/upper rows
 * consumed by vmgpu_colagg_filter (outliers_iqr / outliers_mad). */
#define VMGPU_COLAGG_MEDIAN 0
#define VMGPU_COLAGG_QUANTILE 1
#define VMGPU_COLAGG_MAD 2
#define VMGPU_COLAGG_STDDEV 3
#define VMGPU_COLAGG_STDVAR 4
#define VMGPU_COLAGG_MODE 5
#define VMGPU_COLAGG_DISTINCT 6
#define VMGPU_COLAGG_SHARE 7
#define VMGPU_COLAGG_ZSCORE 8
#define VMGPU_COLAGG_IQR_BOUNDS 9
/* per-point column forms of the simple aggregates (aggrFuncSum/Min/Max/
 * Avg/Count/Sum2/Geomean/Group, aggr.go:316-452) for series-level
 * aggregation of resident result sets (the rollup-fused incremental forms
 * cover the rollup->aggregate path) */
#define VMGPU_COLAGG_SUM 10
#define VMGPU_COLAGG_MIN 11
#define VMGPU_COLAGG_MAX 12
#define VMGPU_COLAGG_AVG 13
#define VMGPU_COLAGG_COUNT 14
#define VMGPU_COLAGG_SUM2 15
#define VMGPU_COLAGG_GEOMEAN 16
#define VMGPU_COLAGG_GROUP 17

int vmgpu_colagg(int32_t op, const double* values, uint32_t n_series,
                 uint32_t n_grid, const uint32_t* group_rows,
                 const uint64_t* group_offsets, uint32_t n_groups,
                 double phi, double* out, double* out2,
                 double* values_out, char* errbuf, size_t errbuf_len);

/* mode 0 = IQR (b1=lower, b2=upper; aggrFuncOutliersIQR), mode 1 = MAD
 * (b1=medians, b2=mads*tolerance; aggrFuncOutliersMAD).  flags[s]=1 when
 * series s has any point outside its group's bounds. */
int vmgpu_colagg_filter(int32_t mode, const double* values,
                        const int32_t* group_of, uint32_t n_series,
                        uint32_t n_grid, const double* b1, const double* b2,
                        uint32_t n_groups, uint8_t* flags,
                        char* errbuf, size_t errbuf_len);

/* histogram_avg/stddev/stdvar (transformHistogramAvg/Stddev/Stdvar +
 * avgForLeTimeseries/stdvarForLeTimeseries): mode 0 avg, 1 stddev,
 * 2 stdvar.  Same CSR bucket layout as vmgpu_histogram_quantile. */
int vmgpu_histogram_stat(int32_t mode, const double* bucket_values,
                         const double* les, const uint64_t* group_offsets,
                         uint32_t n_groups, int32_t n_grid, double* out,
                         char* errbuf, size_t errbuf_len);

/* histogram_share (transformHistogramShare): le_req is the per-grid scalar
 * row (getScalar semantics); out_lower/out_upper nullable bounds. */
int vmgpu_histogram_share(const double* le_req, const double* bucket_values,
                          const double* les, const uint64_t* group_offsets,
                          uint32_t n_groups, int32_t n_grid, double* out,
                          double* out_lower, double* out_upper,
                          char* errbuf, size_t errbuf_len);

int vmgpu_histogram_quantile(double phi, const double* bucket_values,
                             const double* les, const uint64_t* group_offsets,
                             uint32_t n_groups, int32_t n_grid,
                             double* out, double* out_lower, double* out_upper,
                             char* errbuf, size_t errbuf_len);

/* GPU block decode — the fetch path's
 * Block.UnmarshalData (lib/storage/block.go:250) + lib/encoding varint/
 * delta codecs + decimal->float, fused on device.  zstd frames (marshal
 * types 1 and 4) are decompressed on the HOST before this call (standard
 * CPU format), so descriptors carry post-zstd types
 * 2/3/5/6.  e10 = pow(10, |scale|) computed by the caller with libm so
 * device results are bit-identical to the CPU path. */
typedef struct vmgpu_block_desc {
  uint64_t ts_data_off;   /* into the shared payload buffer */
  uint64_t val_data_off;
  uint64_t out_off;       /* row offset of this block in the output columns */
  int64_t min_timestamp;  /* blockHeader fields (lib/storage/block_header.go) */
  int64_t max_timestamp;
  int64_t first_value;
  double e10;
  uint32_t ts_data_len;
  uint32_t val_data_len;
  uint32_t rows;
  int32_t scale;
  uint8_t ts_mt;
  uint8_t val_mt;
  uint8_t precision_bits;
  uint8_t _pad;
} vmgpu_block_desc;

int vmgpu_decode_blocks(const uint8_t* payload, uint64_t payload_len,
                        const vmgpu_block_desc* blocks, uint32_t n_blocks,
                        uint64_t total_rows,
                        int64_t* out_ts, double* out_vals,
                        char* errbuf, size_t errbuf_len);

/* Per-series k-way merge of decoded blocks + optional dedup — the GPU
 * equivalent of app/vmselect/netstorage/netstorage.go:564 mergeSortBlocks
 * and lib/storage/dedup.go:29 DeduplicateSamples.  Input: total_rows decoded
 * samples in ts/vals, partitioned into n_blocks sorted blocks by
 * block_offsets[n_blocks+1]; series s owns blocks
 * [series_block_start[s], series_block_start[s+1]).  Output: merged (and
 * deduped when dedup_interval>0) samples, densely packed; out_offsets has
 * n_series+1 entries, out_counts[s] = merged length of series s.  Blocks
 * with disjoint time ranges (the common LSM-part case) take a wave-parallel
 * concatenation path; overlapping blocks take the exact heap-merge path. */
/* Binary operators on the result grid (binary_op.go + metricsql/binaryop).
 * Label matching (adjustBinaryOpTags, binary_op.go:271) is host metadata
 * work; these entry points do the per-point math over matched pair lists.
 * Op ids follow binaryOpFuncs (binary_op.go:15-43). */
#define VMGPU_BINOP_PLUS 0
#define VMGPU_BINOP_MINUS 1
#define VMGPU_BINOP_MUL 2
#define VMGPU_BINOP_DIV 3
#define VMGPU_BINOP_MOD 4
#define VMGPU_BINOP_POW 5
#define VMGPU_BINOP_ATAN2 6
#define VMGPU_BINOP_EQ 7
#define VMGPU_BINOP_NEQ 8
#define VMGPU_BINOP_GT 9
#define VMGPU_BINOP_LT 10
#define VMGPU_BINOP_GTE 11
#define VMGPU_BINOP_LTE 12
#define VMGPU_BINOP_DEFAULT 13
#define VMGPU_BINOP_IF 14
#define VMGPU_BINOP_IFNOT 15
#define VMGPU_BINOP_AND 16
#define VMGPU_BINOP_OR 17

/* newBinaryOpFunc's value loop (binary_op.go:162-236): pair p reads row
 * left_idx[p] of left_vals (NULL idx = p) and right_idx[p] of right_vals;
 * fill_left/fill_right mirror the fill()/fill_left()/fill_right()
 * modifiers; drop_nan_right the vector-comparison NaN rule (:199). */
int vmgpu_binop_eval(int32_t op, int32_t is_bool, int32_t drop_nan_right,
                     const double* left_vals, uint32_t n_left_rows,
                     const uint32_t* left_idx,
                     const double* right_vals, uint32_t n_right_rows,
                     const uint32_t* right_idx,
                     uint32_t n_pairs, uint32_t n_grid,
                     int32_t has_fill_left, double fill_left,
                     int32_t has_fill_right, double fill_right,
                     double* out, char* errbuf, size_t errbuf_len);

/* Set-op masking: mode 0 = and/if (addRightNaNsToLeft, binary_op.go:549),
 * 1 = unless/ifnot (addLeftNaNsIfNoRightNaNs, :729), 2 = default fill
 * (fillLeftNaNsWithRightValues, :622).  left row l is masked against the
 * right rows of its key group (CSR: group_offsets/group_rows); left_vals
 * modified in place. */
int vmgpu_binop_mask(int32_t mode, double* left_vals, uint32_t n_left,
                     const uint32_t* left_group,
                     const double* right_vals, uint32_t n_right,
                     const uint32_t* group_offsets, uint32_t n_groups,
                     const uint32_t* group_rows,
                     uint32_t n_grid, char* errbuf, size_t errbuf_len);

/* `or` merge walk (fillLeftNaNsWithRightValuesOrMerge, binary_op.go:645):
 * per key group, the exact left-outer/right-inner consume-and-fill order;
 * can_merge[(l,r)] (row-major per group, bases in merge_offsets) is
 * precomputed on the host from marshaled metric names.  Both value
 * matrices are modified in place. */
int vmgpu_binop_or(double* left_vals, uint32_t n_left,
                   double* right_vals, uint32_t n_right,
                   const uint32_t* lgroup_offsets, const uint32_t* lgroup_rows,
                   const uint32_t* rgroup_offsets, const uint32_t* rgroup_rows,
                   const uint8_t* can_merge, const uint64_t* merge_offsets,
                   uint64_t merge_len, uint32_t n_groups, uint32_t n_grid,
                   char* errbuf, size_t errbuf_len);

/* Transform functions on the result grid (transform.go).  Elementwise ids
 * below VMGPU_TF_SERIES_BASE run flat over the matrix; per-series ids walk
 * columns sequentially (one thread per series).  arg1/arg2 are per-grid
 * scalar-arg rows (getScalar semantics); scalar_arg the phi/k/z scalars. */
enum {
  VMGPU_TF_ABS = 0, VMGPU_TF_CEIL, VMGPU_TF_FLOOR, VMGPU_TF_EXP,
  VMGPU_TF_LN, VMGPU_TF_LOG2, VMGPU_TF_LOG10, VMGPU_TF_SQRT,
  VMGPU_TF_SIN, VMGPU_TF_COS, VMGPU_TF_TAN, VMGPU_TF_ASIN, VMGPU_TF_ACOS,
  VMGPU_TF_ATAN, VMGPU_TF_SINH, VMGPU_TF_COSH, VMGPU_TF_TANH,
  VMGPU_TF_ASINH, VMGPU_TF_ACOSH, VMGPU_TF_ATANH, VMGPU_TF_DEG,
  VMGPU_TF_RAD, VMGPU_TF_SGN, VMGPU_TF_CLAMP, VMGPU_TF_CLAMP_MIN,
  VMGPU_TF_CLAMP_MAX, VMGPU_TF_ROUND, VMGPU_TF_BITMAP_AND,
  VMGPU_TF_BITMAP_OR, VMGPU_TF_BITMAP_XOR, VMGPU_TF_DAY_OF_MONTH,
  VMGPU_TF_DAY_OF_WEEK, VMGPU_TF_DAY_OF_YEAR, VMGPU_TF_DAYS_IN_MONTH,
  VMGPU_TF_HOUR, VMGPU_TF_MINUTE, VMGPU_TF_MONTH, VMGPU_TF_YEAR,

  VMGPU_TF_SERIES_BASE = 100,
  VMGPU_TF_KEEP_LAST_VALUE = 100, VMGPU_TF_KEEP_NEXT_VALUE,
  VMGPU_TF_INTERPOLATE, VMGPU_TF_RUNNING_SUM, VMGPU_TF_RUNNING_MIN,
  VMGPU_TF_RUNNING_MAX, VMGPU_TF_RUNNING_AVG, VMGPU_TF_RANGE_SUM,
  VMGPU_TF_RANGE_MIN, VMGPU_TF_RANGE_MAX, VMGPU_TF_RANGE_AVG,
  VMGPU_TF_RANGE_FIRST, VMGPU_TF_RANGE_LAST, VMGPU_TF_RANGE_NORMALIZE,
  VMGPU_TF_RANGE_ZSCORE, VMGPU_TF_RANGE_TRIM_ZSCORE, VMGPU_TF_RANGE_STDDEV,
  VMGPU_TF_RANGE_STDVAR, VMGPU_TF_RANGE_LINREG, VMGPU_TF_RANGE_MAD,
  VMGPU_TF_RANGE_TRIM_OUTLIERS, VMGPU_TF_RANGE_TRIM_SPIKES,
  VMGPU_TF_RANGE_QUANTILE, VMGPU_TF_SMOOTH_EXPONENTIAL,
  VMGPU_TF_REMOVE_RESETS
};

/* values: [n_series x n_grid] f64, transformed in place.  ts: the shared
 * grid timestamps (range_linear_regression).  keep_flags (nullable,
 * [n_series]): 0 = series dropped from the result (range_normalize on an
 * all-NaN series, transform.go:1383). */
int vmgpu_transform(int32_t func, double* values, uint32_t n_series,
                    uint32_t n_grid, const int64_t* ts, const double* arg1,
                    const double* arg2, double scalar_arg,
                    uint8_t* keep_flags, char* errbuf, size_t errbuf_len);

int vmgpu_merge_blocks(const int64_t* ts, const double* vals,
                       const uint64_t* block_offsets, uint32_t n_blocks,
                       const uint32_t* series_block_start, uint32_t n_series,
                       int64_t dedup_interval,
                       int64_t* out_ts, double* out_vals,
                       uint64_t* out_offsets, uint64_t* out_counts,
                       char* errbuf, size_t errbuf_len);

/* Fused cold-cache fetch: compressed-block payload -> decode -> per-series
 * merge+dedup -> device-resident rollup batch, decoded columns never
 * crossing PCIe.  blocks[b].out_off must be the running
 * row offset (CSR over blocks); series s owns blocks
 * [series_block_start[s], series_block_start[s+1]).  out_offsets
 * ([n_series+1], caller-alloc) receives the merged per-series CSR. */
int vmgpu_batch_create_from_blocks(
    const uint8_t* payload, uint64_t payload_len,
    const vmgpu_block_desc* blocks, uint32_t n_blocks, uint64_t total_rows,
    const uint32_t* series_block_start, uint32_t n_series,
    int64_t dedup_interval, const int32_t* group_ids, uint32_t n_groups,
    uint64_t* out_handle, uint64_t* out_offsets,
    char* errbuf, size_t errbuf_len);

/* Packed block stream: the single-buffer wire form a fetch layer hands
 * over for a whole cold query (replaces the reference's per-series
 * []sortedBlock handoff, netstorage.go:423-614).  n_blocks records, each a
 * 48-byte vmgpu_packed_block_hdr followed by ts_data (ts_data_len bytes)
 * then val_data (val_data_len bytes); record i belongs to the series given
 * by series_block_start (blocks appear grouped by series, in order).
 * Marshal types are post-zstd (2/3/5/6 and const/delta-const). */
typedef struct vmgpu_packed_block_hdr {
  int64_t min_timestamp;
  int64_t max_timestamp;
  int64_t first_value;   /* values column first value; the timestamps
                            column's first value is min_timestamp */
  uint32_t rows;
  int32_t scale;
  uint32_t ts_data_len;
  uint32_t val_data_len;
  uint8_t ts_mt;
  uint8_t val_mt;
  uint8_t precision_bits;
  uint8_t _pad[5];
} vmgpu_packed_block_hdr;

/* Native descriptor build + fused decode/merge/batch-create from a packed
 * block stream: walks the records in C (no per-block marshaling in the
 * host-language layer), then runs the vmgpu_batch_create_from_blocks
 * pipeline against the stream buffer itself as the device payload. */
int vmgpu_batch_create_packed(
    const uint8_t* packed, uint64_t packed_len, uint64_t n_blocks,
    const uint32_t* series_block_start, uint32_t n_series,
    int64_t dedup_interval, const int32_t* group_ids, uint32_t n_groups,
    uint64_t* out_handle, uint64_t* out_offsets,
    char* errbuf, size_t errbuf_len);

/* Pinned host buffers for PCIe-rate staging of payloads and results (the
 * cgo layer would pool these; hipHostMalloc/-Free underneath). */
int vmgpu_host_alloc(uint64_t nbytes, void** out_ptr);
int vmgpu_host_free(void* ptr);

/* Wall time of the rollup kernels inside the last vmgpu_rollup_exec on this
 * thread's context, measured with hipEvents on the launch stream (for the
 * bench's roofline accounting). */
int vmgpu_last_kernel_ms(double* out_ms);

/* Device properties the bench needs for roofline context. */
int vmgpu_device_info(char* name, size_t name_len, double* hbm_gib,
                      int* cu_count);

#ifdef __cplusplus
}
#endif
#endif /* VMGPU_H */
