/* aggr.c — CPU oracle for cross-series incremental aggregation + the batch
 * CSR entry used as the cpu_baseline leg of bench.py.
 *
 * TEST INFRASTRUCTURE ONLY (see vm_oracle.h header note).
 *
 * Faithful C restatement of app/vmselect/promql/aggr_incremental.go:
 *   updateAggrSum 200, mergeAggrSum 218, updateAggrMin 239, updateAggrMax 282,
 *   updateAggrAvg 325, updateAggrCount 381, updateAggrSum2 421,
 *   updateAggrGeomean 460, finalizeAggrCommon 189, finalizeAggrAvg 368,
 *   finalizeAggrCount 401, finalizeAggrGroup 410, finalizeAggrGeomean 501,
 * and the orchestration seam evalRollupWithIncrementalAggregate
 * (eval.go:1927-1966) / evalRollupNoIncrementalAggregate (eval.go:1968-2006)
 * reduced to its dense-group-matrix form (labels → dense group ids are
 * assigned by the host).
 */
#include "vm_oracle.h"
#include <math.h>
#include <string.h>
#include <stdlib.h>

#ifdef _OPENMP
#include <omp.h>
#endif

static const double NAN_V = NAN;

void vm_aggr_update(int aggr_op, double* dstv, double* dstc, const double* values, int64_t n_grid) {
  switch (aggr_op) {
    case VM_AGGR_SUM:
      for (int64_t i = 0; i < n_grid; i++) {
        double v = values[i];
        if (isnan(v)) continue;
        if (dstc[i] == 0) { dstv[i] = v; dstc[i] = 1; continue; }
        dstv[i] += v;
      }
      break;
    case VM_AGGR_MIN:
      for (int64_t i = 0; i < n_grid; i++) {
        double v = values[i];
        if (isnan(v)) continue;
        if (dstc[i] == 0) { dstv[i] = v; dstc[i] = 1; continue; }
        if (v < dstv[i]) dstv[i] = v;
      }
      break;
    case VM_AGGR_MAX:
      for (int64_t i = 0; i < n_grid; i++) {
        double v = values[i];
        if (isnan(v)) continue;
        if (dstc[i] == 0) { dstv[i] = v; dstc[i] = 1; continue; }
        if (v > dstv[i]) dstv[i] = v;
      }
      break;
    case VM_AGGR_AVG:
      for (int64_t i = 0; i < n_grid; i++) {
        double v = values[i];
        if (isnan(v)) continue;
        if (dstc[i] == 0) { dstv[i] = v; dstc[i] = 1; continue; }
        dstv[i] += v;
        dstc[i]++;
      }
      break;
    case VM_AGGR_COUNT: /* also `group` (aggr_incremental.go:61-65) */
    case VM_AGGR_GROUP:
      for (int64_t i = 0; i < n_grid; i++) {
        if (isnan(values[i])) continue;
        dstv[i]++;
      }
      break;
    case VM_AGGR_SUM2:
      for (int64_t i = 0; i < n_grid; i++) {
        double v = values[i];
        if (isnan(v)) continue;
        if (dstc[i] == 0) { dstv[i] = v * v; dstc[i] = 1; continue; }
        dstv[i] += v * v;
      }
      break;
    case VM_AGGR_GEOMEAN:
      for (int64_t i = 0; i < n_grid; i++) {
        double v = values[i];
        if (isnan(v)) continue;
        if (dstc[i] == 0) { dstv[i] = v; dstc[i] = 1; continue; }
        dstv[i] *= v;
        dstc[i]++;
      }
      break;
    default:
      break;
  }
}

void vm_aggr_merge(int aggr_op, double* dstv, double* dstc,
                   const double* srcv, const double* srcc, int64_t n_grid) {
  switch (aggr_op) {
    case VM_AGGR_SUM:
    case VM_AGGR_SUM2: /* mergeAggrSum2 adds already-squared src values */
      for (int64_t i = 0; i < n_grid; i++) {
        if (srcc[i] == 0) continue;
        if (dstc[i] == 0) { dstv[i] = srcv[i]; dstc[i] = 1; continue; }
        dstv[i] += srcv[i];
      }
      break;
    case VM_AGGR_MIN:
      for (int64_t i = 0; i < n_grid; i++) {
        if (srcc[i] == 0) continue;
        if (dstc[i] == 0) { dstv[i] = srcv[i]; dstc[i] = 1; continue; }
        if (srcv[i] < dstv[i]) dstv[i] = srcv[i];
      }
      break;
    case VM_AGGR_MAX:
      for (int64_t i = 0; i < n_grid; i++) {
        if (srcc[i] == 0) continue;
        if (dstc[i] == 0) { dstv[i] = srcv[i]; dstc[i] = 1; continue; }
        if (srcv[i] > dstv[i]) dstv[i] = srcv[i];
      }
      break;
    case VM_AGGR_AVG:
    case VM_AGGR_GEOMEAN:
      for (int64_t i = 0; i < n_grid; i++) {
        if (srcc[i] == 0) continue;
        if (dstc[i] == 0) { dstv[i] = srcv[i]; dstc[i] = srcc[i]; continue; }
        if (aggr_op == VM_AGGR_AVG) dstv[i] += srcv[i];
        else dstv[i] *= srcv[i];
        dstc[i] += srcc[i];
      }
      break;
    case VM_AGGR_COUNT:
    case VM_AGGR_GROUP:
      for (int64_t i = 0; i < n_grid; i++) dstv[i] += srcv[i];
      break;
    default:
      break;
  }
}

void vm_aggr_finalize(int aggr_op, double* dstv, double* dstc, int64_t n_grid) {
  switch (aggr_op) {
    case VM_AGGR_SUM:
    case VM_AGGR_MIN:
    case VM_AGGR_MAX:
    case VM_AGGR_SUM2: /* finalizeAggrCommon */
      for (int64_t i = 0; i < n_grid; i++)
        if (dstc[i] == 0) dstv[i] = NAN_V;
      break;
    case VM_AGGR_AVG:
      for (int64_t i = 0; i < n_grid; i++) {
        if (dstc[i] == 0) { dstv[i] = NAN_V; continue; }
        dstv[i] /= dstc[i];
      }
      break;
    case VM_AGGR_COUNT:
      for (int64_t i = 0; i < n_grid; i++)
        if (dstv[i] == 0) dstv[i] = NAN_V;
      break;
    case VM_AGGR_GROUP:
      for (int64_t i = 0; i < n_grid; i++) {
        if (dstv[i] == 0) dstv[i] = NAN_V;
        else dstv[i] = 1;
      }
      break;
    case VM_AGGR_GEOMEAN:
      for (int64_t i = 0; i < n_grid; i++) {
        if (dstc[i] == 0) { dstv[i] = NAN_V; continue; }
        dstv[i] = pow(dstv[i], 1.0 / dstc[i]);
      }
      break;
    default:
      break;
  }
}

/* Batch evaluation over CSR series — the cpu_baseline leg.
 *
 * Orchestration mirrors RunParallel's per-series worker callback
 * (netstorage.go:219 + eval.go:1937-1950): per series, dropStaleNaNs →
 * preFunc(removeCounterResets) → rollupConfig.Do → per-series output row or
 * incremental aggregate update into a per-thread [n_groups × n_grid] matrix,
 * merged at the end (finalizeTimeseries, aggr_incremental.go:141-168).
 */
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
                         int n_threads) {
  int64_t n_grid = vm_grid_points(rc->start, rc->end, rc->step);
  if (n_grid <= 0) return 1;
  if (n_threads < 1) n_threads = 1;

  int grouped = (group_ids != NULL && aggr_op != VM_AGGR_NONE);
  size_t gmat = (size_t)n_groups * (size_t)n_grid;

  double* partial_v = NULL;
  double* partial_c = NULL;
  if (grouped) {
    partial_v = (double*)calloc((size_t)n_threads * gmat, sizeof(double));
    partial_c = (double*)calloc((size_t)n_threads * gmat, sizeof(double));
    if (!partial_v || !partial_c) { free(partial_v); free(partial_c); return 2; }
  }

  uint64_t samples_total = 0;

#ifdef _OPENMP
#pragma omp parallel num_threads(n_threads) reduction(+ : samples_total)
#endif
  {
    int tid = 0;
#ifdef _OPENMP
    tid = omp_get_thread_num();
#endif
    double* scratch_v = NULL;
    int64_t* scratch_t = NULL;
    double* row = NULL;
    size_t scratch_cap = 0;
    row = (double*)malloc((size_t)n_grid * sizeof(double));

#ifdef _OPENMP
#pragma omp for schedule(dynamic, 64)
#endif
    for (int64_t s = 0; s < (int64_t)n_series; s++) {
      uint64_t lo = offsets[s], hi = offsets[s + 1];
      int64_t n = (int64_t)(hi - lo);
      const double* v = vals + lo;
      const int64_t* t = ts + lo;
      if (remove_counter_resets || drop_stale_nans || pre_func) {
        if ((size_t)n > scratch_cap) {
          scratch_cap = (size_t)(n > 64 ? n : 64);
          scratch_v = (double*)realloc(scratch_v, scratch_cap * sizeof(double));
          scratch_t = (int64_t*)realloc(scratch_t, scratch_cap * sizeof(int64_t));
        }
        memcpy(scratch_v, v, (size_t)n * sizeof(double));
        memcpy(scratch_t, t, (size_t)n * sizeof(int64_t));
        if (drop_stale_nans) n = vm_drop_stale_nans(scratch_v, scratch_t, n);
        if (remove_counter_resets)
          vm_remove_counter_resets(scratch_v, scratch_t, n, max_staleness_interval);
        /* preFunc transforms for the rollup_* pseudo-functions
         * (getRollupConfigs, rollup.go:436-516) */
        if (pre_func == 1) vm_delta_values(scratch_v, n);
        else if (pre_func == 2) vm_deriv_values(scratch_v, scratch_t, n);
        else if (pre_func == 3 && n > 0) {
          /* rollup_scrape_interval preFunc (rollup.go:476-494) */
          double prev_secs = nan("");
          for (int64_t i = 0; i < n; i++) {
            double secs = (double)scratch_t[i] / 1000.0;
            scratch_v[i] = secs - prev_secs;
            prev_secs = secs;
          }
          if (n > 1) scratch_v[0] = scratch_v[1];
        }
        v = scratch_v;
        t = scratch_t;
      }
      double* dst = grouped ? row : (out + (size_t)s * (size_t)n_grid);
      samples_total += vm_rollup_do(rc, v, t, n, dst);
      if (grouped) {
        int32_t g = group_ids[s];
        if (g >= 0 && (uint32_t)g < n_groups) {
          vm_aggr_update(aggr_op, partial_v + (size_t)tid * gmat + (size_t)g * (size_t)n_grid,
                         partial_c + (size_t)tid * gmat + (size_t)g * (size_t)n_grid, row, n_grid);
        }
      }
    }
    free(scratch_v);
    free(scratch_t);
    free(row);
  }

  if (grouped) {
    memcpy(out, partial_v, gmat * sizeof(double));
    double* cnts = out_counts ? out_counts : (double*)calloc(gmat, sizeof(double));
    memcpy(cnts, partial_c, gmat * sizeof(double));
    for (int tid = 1; tid < n_threads; tid++) {
      for (uint32_t g = 0; g < n_groups; g++) {
        vm_aggr_merge(aggr_op, out + (size_t)g * (size_t)n_grid, cnts + (size_t)g * (size_t)n_grid,
                      partial_v + (size_t)tid * gmat + (size_t)g * (size_t)n_grid,
                      partial_c + (size_t)tid * gmat + (size_t)g * (size_t)n_grid, n_grid);
      }
    }
    for (uint32_t g = 0; g < n_groups; g++) {
      vm_aggr_finalize(aggr_op, out + (size_t)g * (size_t)n_grid,
                       cnts + (size_t)g * (size_t)n_grid, n_grid);
    }
    if (!out_counts) free(cnts);
    free(partial_v);
    free(partial_c);
  }

  if (out_samples_scanned) *out_samples_scanned = samples_total;
  return 0;
}
