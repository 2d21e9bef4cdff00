/* vm_oracle.h — CPU oracle for the vmselect rollup/aggregation hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is a faithful CPU restatement of the
 * reference Go algorithms (VictoriaMetrics vmselect PromQL rollup path) used
 * (a) as the parity oracle for the HIP/CDNA4 engine and (b) as the reported
 * `cpu_baseline` leg of bench.py. It must never be imported, linked or called
 * by the product path (victoriametrics_amd/ engine): only tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may use it.
 *
 * Pinned against the reference's own in-repo golden vectors (see
 * tests/golden/ JSON files, transcribed from app/vmselect/promql/rollup_test.go and
 * friends). Each function cites the reference file:line it restates.
 *
 * Reference: /root/reference (VictoriaMetrics/VictoriaMetrics, 2026-08-21).
 */
#ifndef VM_ORACLE_H
#define VM_ORACLE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Rollup function ids. Names follow the reference's rollupFuncs map keys
 * (app/vmselect/promql/rollup.go:24-108). */
typedef enum {
  VM_FN_RATE = 0,            /* rollupDerivFast, rollup.go:1954 */
  VM_FN_INCREASE,            /* rollupDelta, rollup.go:1859 */
  VM_FN_INCREASE_PURE,       /* rollupIncreasePure, rollup.go:1835 */
  VM_FN_DELTA,               /* rollupDelta (no counter-reset removal) */
  VM_FN_DELTA_PROMETHEUS,    /* rollupDeltaPrometheus, rollup.go:1903 */
  VM_FN_RATE_PROMETHEUS,     /* rollupDerivFastPrometheus, rollup.go:1946 */
  VM_FN_IRATE,               /* rollupIderiv, rollup.go:1991 */
  VM_FN_IDERIV,              /* rollupIderiv (no counter-reset removal) */
  VM_FN_IDELTA,              /* rollupIdelta, rollup.go:1915 */
  VM_FN_DERIV_FAST,          /* rollupDerivFast (no counter-reset removal) */
  VM_FN_AVG,                 /* rollupAvg, rollup.go:1541 */
  VM_FN_MIN,                 /* rollupMin, rollup.go:1561 */
  VM_FN_MAX,                 /* rollupMax, rollup.go:1580 */
  VM_FN_SUM,                 /* rollupSum, rollup.go:1690 */
  VM_FN_SUM2,                /* rollupSum2, rollup.go:1727 */
  VM_FN_COUNT,               /* rollupCount, rollup.go:1771 */
  VM_FN_FIRST,               /* rollupFirst */
  VM_FN_LAST,                /* rollupLast (also default_rollup body) */
  VM_FN_QUANTILE,            /* newRollupQuantile, rollup.go:1450 (arg=phi) */
  VM_FN_MEDIAN,              /* rollupMedian, rollup.go:1599 */
  VM_FN_STDDEV,              /* rollupStddev, rollup.go:1795 */
  VM_FN_STDVAR,              /* rollupStdvar, rollup.go:1799 */
  VM_FN_CHANGES,             /* rollupChanges */
  VM_FN_CHANGES_PROMETHEUS,  /* rollupChangesPrometheus, rollup.go:2082 */
  VM_FN_RESETS,              /* rollupResets */
  VM_FN_LAG,                 /* rollupLag, rollup.go:2055 */
  VM_FN_LIFETIME,            /* rollupLifetime, rollup.go:2040 */
  VM_FN_SCRAPE_INTERVAL,     /* rollupScrapeInterval, rollup.go:2067 */
  VM_FN_RATE_OVER_SUM,       /* rollupRateOverSum, rollup.go:1707 */
  VM_FN_RANGE,               /* rollupRange, rollup.go:1721 */
  VM_FN_TFIRST,              /* rollupTfirst, rollup.go:1643 */
  VM_FN_TLAST,               /* rollupTlast (also `timestamp`), rollup.go:1656 */
  VM_FN_TMIN,                /* rollupTmin, rollup.go:1603 */
  VM_FN_TMAX,                /* rollupTmax, rollup.go:1623 */
  VM_FN_TLAST_CHANGE,        /* rollupTlastChange, rollup.go:1669 */
  VM_FN_GEOMEAN,             /* rollupGeomean, rollup.go:1741 */
  VM_FN_PRESENT,             /* rollupPresent, rollup.go:1762 */
  VM_FN_ABSENT,              /* rollupAbsent, rollup.go:1755 */
  VM_FN_STALE_SAMPLES,       /* rollupStaleSamples, rollup.go:1781 */
  VM_FN_COUNT_LE,            /* newRollupCountLE (arg=le) */
  VM_FN_COUNT_GT,            /* newRollupCountGT (arg=gt) */
  VM_FN_COUNT_EQ,            /* newRollupCountEQ (arg=eq) */
  VM_FN_COUNT_NE,            /* newRollupCountNE (arg=ne) */
  VM_FN_SHARE_LE,            /* newRollupShareLE */
  VM_FN_SHARE_GT,            /* newRollupShareGT */
  VM_FN_SHARE_EQ,            /* newRollupShareEQ */
  VM_FN_SUM_LE,              /* newRollupSumLE */
  VM_FN_SUM_GT,              /* newRollupSumGT */
  VM_FN_SUM_EQ,              /* newRollupSumEQ */
  VM_FN_DERIV,               /* rollupDerivSlow (linear regression), rollup.go:1939 */
  VM_FN_PREDICT_LINEAR,      /* newRollupPredictLinear (arg=secs) */
  VM_FN_ASCENT,              /* rollupAscentOverTime */
  VM_FN_DESCENT,             /* rollupDescentOverTime */
  VM_FN_ZSCORE,              /* rollupZScoreOverTime */
  VM_FN_INTEGRATE,           /* rollupIntegrate */
  VM_FN_DISTINCT,            /* rollupDistinct */
  VM_FN_INCREASES,           /* rollupIncreases */
  VM_FN_DECREASES,           /* rollupDecreases */
  VM_FN_MAD,                 /* rollupMAD, rollup.go:1469 */
  VM_FN_DEFAULT_ROLLUP,      /* rollupDefault (last value in window) */
  VM_FN_MODE,                /* rollupModeOverTime, rollup.go:2293 + modeNoNaNs aggr.go:541 */
  VM_FN_DURATION,            /* newRollupDurationOverTime, rollup.go:1151 (arg=dMax secs) */
  VM_FN_OUTLIER_IQR,         /* rollupOutlierIQR, rollup.go:1427 */
  VM_FN_OPEN,                /* rollupOpen, rollup.go:2227 (candlestick) */
  VM_FN_CLOSE,               /* rollupClose, rollup.go:2238 */
  VM_FN_LOW,                 /* rollupLow, rollup.go:2262 */
  VM_FN_HIGH,                /* rollupHigh, rollup.go:2245 */
  VM_FN_HOLT_WINTERS,        /* newRollupHoltWinters, rollup.go:1030 (arg=sf, arg2=tf) */
  VM_FN_HOEFFDING_LOWER,     /* newRollupHoeffdingBoundLower, rollup.go:1323 (arg=phi) */
  VM_FN_HOEFFDING_UPPER,     /* newRollupHoeffdingBoundUpper, rollup.go:1338 */
  VM_FN__COUNT
} vm_func_id;

/* Cross-series incremental aggregate ops
 * (app/vmselect/promql/aggr_incremental.go:18-66). */
typedef enum {
  VM_AGGR_NONE = 0,
  VM_AGGR_SUM,
  VM_AGGR_MIN,
  VM_AGGR_MAX,
  VM_AGGR_AVG,
  VM_AGGR_COUNT,
  VM_AGGR_SUM2,
  VM_AGGR_GEOMEAN,
  VM_AGGR_GROUP,
  VM_AGGR__COUNT
} vm_aggr_op;

/* Mirror of promql.rollupConfig (rollup.go:574-606) restricted to fields that
 * shape the computation. */
typedef struct {
  int32_t func;                  /* vm_func_id */
  int32_t may_adjust_window;     /* rollupFuncsCanAdjustWindow[func] */
  int64_t start, end, step;      /* grid [start:end:step], ms */
  int64_t window;                /* lookbehind window, ms; 0 => auto-adjust */
  int64_t lookback_delta;        /* rc.LookbackDelta, ms */
  int64_t min_staleness_interval;/* -search.minStalenessInterval flag, ms */
  int32_t is_default_rollup;
  int32_t samples_scanned_per_call; /* rollupFuncsSamplesScannedPerCall */
  double  arg;                   /* phi / le / gt / eq / secs for arg funcs */
  double  arg2;                  /* second scalar (holt_winters tf) */
} vm_rollup_config;

/* getTimestamps, eval.go:234-254. Returns number of grid points. */
int64_t vm_grid_points(int64_t start, int64_t end, int64_t step);
void vm_get_timestamps(int64_t start, int64_t end, int64_t step, int64_t* dst);

/* removeCounterResets, rollup.go:921-958. In-place. */
void vm_remove_counter_resets(double* values, const int64_t* timestamps, int64_t n,
                              int64_t max_staleness_interval);

/* deltaValues / derivValues preFuncs, rollup.go:960-999. In-place. */
void vm_delta_values(double* values, int64_t n);
void vm_deriv_values(double* values, const int64_t* timestamps, int64_t n);

/* dropStaleNaNs, eval.go:2108-2132 (slow path; caller checks fast path).
 * Compacts in place, returns new length. */
int64_t vm_drop_stale_nans(double* values, int64_t* timestamps, int64_t n);

/* decimal.StaleNaN (lib/decimal/decimal.go:403-419). */
int vm_is_stale_nan(double v);
double vm_stale_nan(void);

/* rollupConfig.doInternal, rollup.go:701-823: one series over the grid.
 * dst must have vm_grid_points() elements. Returns samplesScanned. */
uint64_t vm_rollup_do(const vm_rollup_config* rc, const double* values,
                      const int64_t* timestamps, int64_t n, double* dst);

/* Raw rollup-func invocation with an explicit rollupFuncArg — mirrors the
 * reference's own unit-test harness (rollup_test.go:223-259 testRollupFunc and
 * the direct rfa tests). For funcs in rollupFuncsRemoveCounterResets the
 * caller applies vm_remove_counter_resets first, like the harness does. */
double vm_call_rollup_fn(int32_t func, double prev_value, int64_t prev_timestamp,
                         const double* values, const int64_t* timestamps, int64_t n,
                         double real_prev_value, double real_next_value,
                         int64_t curr_timestamp, int64_t idx, int64_t window, double arg,
                         double arg2);

/* quantile over unsorted values with NaN filtering (aggr.go:870-876) and
 * quantileSorted (aggr.go:922-940). */
double vm_quantile(double phi, const double* values, int64_t n);
double vm_quantile_sorted(double phi, const double* values, int64_t n);

/* getScrapeInterval / getMaxPrevInterval, rollup.go:871-919 (exposed for
 * unit parity with the GPU per-series preamble). */
int64_t vm_get_scrape_interval(const int64_t* timestamps, int64_t n, int64_t default_interval);
int64_t vm_get_max_prev_interval(int64_t scrape_interval);

/* Incremental aggregate callbacks over a dense [n_grid] row
 * (aggr_incremental.go:200-512). dstv/dstc are the running value/count rows. */
void vm_aggr_update(int aggr_op, double* dstv, double* dstc, const double* values, int64_t n_grid);
void vm_aggr_merge(int aggr_op, double* dstv, double* dstc,
                   const double* srcv, const double* srcc, int64_t n_grid);
void vm_aggr_finalize(int aggr_op, double* dstv, double* dstc, int64_t n_grid);

/* Batch evaluation over CSR series — the same shape as the product C-ABI
 * (include/vmgpu.h: vmgpu_rollup_eval), used as the cpu_baseline leg and as
 * the full-size parity reference. Series s occupies [offsets[s], offsets[s+1])
 * in ts/vals. remove_counter_resets/max_staleness_interval mirror the preFunc
 * from getRollupConfigs (rollup.go:389-393). If group_ids is NULL the output
 * is [n_series × n_grid]; otherwise aggregated [n_groups × n_grid] with
 * out_counts [n_groups × n_grid] (may be NULL for aggr==NONE).
 * n_threads: OpenMP threads (1 = serial). Returns 0 on success. */
int vm_rollup_eval_batch(const vm_rollup_config* rc,
                         int32_t remove_counter_resets,
                         int64_t max_staleness_interval,
                         int32_t drop_stale_nans,
                         int32_t pre_func,
                         const int64_t* ts, const double* vals,
                         const uint64_t* offsets, uint32_t n_series,
                         const int32_t* group_ids, uint32_t n_groups, int32_t aggr_op,
                         double* out, double* out_counts,
                         uint64_t* out_samples_scanned,
                         int n_threads);

/* ---- topk family (aggr.go:646-741) + histogram_quantile (transform.go:992)
 * — dense-matrix restatements; see oracle/topk.c ---- */
void vm_topk_pointwise(double* values, int64_t n_series, int64_t n_grid,
                       double k, int32_t reverse);
double vm_topk_summary(int32_t op, const double* v, int64_t n);
int64_t vm_topk_range(const double* values, int64_t n_series, int64_t n_grid,
                      double k, int32_t summary_op, int32_t reverse,
                      int64_t* out_sel, double* out_remaining);
void vm_histogram_quantile(double phi, const double* bucket_values,
                           const double* les, const uint64_t* group_offsets,
                           int64_t n_groups, int64_t n_grid,
                           double* out, double* out_lower, double* out_upper);

#ifdef __cplusplus
}
#endif
#endif /* VM_ORACLE_H */
