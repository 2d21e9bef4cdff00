/* rollup.c — CPU oracle for the vmselect rollup hot loop.
 *
 * TEST INFRASTRUCTURE ONLY (see vm_oracle.h header note).
 *
 * Faithful C restatement of:
 *   app/vmselect/promql/rollup.go:
 *     doInternal 701-823, seekFirstTimestampIdxAfter 825-855,
 *     getScrapeInterval 871-897, getMaxPrevInterval 899-919,
 *     removeCounterResets 921-958, deltaValues 960-974, derivValues 976-999,
 *     the rollup funcs (cited per function below)
 *   app/vmselect/promql/eval.go: getTimestamps 234-254, dropStaleNaNs 2108-2132
 *   app/vmselect/promql/aggr.go: quantile 870, quantileSorted 922-940,
 *     modeNoNaNs 541-564
 *   lib/decimal/decimal.go: StaleNaN 403-419
 */
#include "vm_oracle.h"
#include <math.h>
#include <stdlib.h>
#include <string.h>

static const double NAN_V = NAN;

/* ---------- decimal.StaleNaN (lib/decimal/decimal.go:403-419) ---------- */

/* Prometheus staleness marker: NaN with payload bits 0x7ff0000000000002. */
static const uint64_t STALE_NAN_BITS = 0x7ff0000000000002ULL;

double vm_stale_nan(void) {
  double v;
  uint64_t b = STALE_NAN_BITS;
  memcpy(&v, &b, 8);
  return v;
}

int vm_is_stale_nan(double v) {
  uint64_t b;
  memcpy(&b, &v, 8);
  return b == STALE_NAN_BITS;
}

/* ---------- getTimestamps (eval.go:234-254) ---------- */

int64_t vm_grid_points(int64_t start, int64_t end, int64_t step) {
  if (step <= 0 || start > end) return -1;
  return 1 + (end - start) / step;
}

void vm_get_timestamps(int64_t start, int64_t end, int64_t step, int64_t* dst) {
  int64_t n = vm_grid_points(start, end, step);
  for (int64_t i = 0; i < n; i++) {
    dst[i] = start;
    start += step;
  }
}

/* ---------- removeCounterResets (rollup.go:921-958) ---------- */

void vm_remove_counter_resets(double* values, const int64_t* timestamps, int64_t n,
                              int64_t max_staleness_interval) {
  if (n == 0) return;
  double correction = 0;
  double prev_value = values[0];
  for (int64_t i = 0; i < n; i++) {
    double v = values[i];
    double d = v - prev_value;
    if (d < 0) {
      if ((-d * 8) < prev_value) {
        /* likely partial counter reset (issue 2787) */
        correction += prev_value - v;
      } else {
        correction += prev_value;
      }
    }
    if (i > 0 && max_staleness_interval > 0) {
      int64_t gap = timestamps[i] - timestamps[i - 1];
      if (gap > max_staleness_interval) {
        /* reset correction when the gap exceeds the staleness interval
         * (issue 8072); note: values[i] is left as-is and the monotonic
         * clamp below is skipped for this element. */
        correction = 0;
        prev_value = v;
        continue;
      }
    }
    prev_value = v;
    values[i] = v + correction;
    /* guard against float precision (issue 5571) */
    if (i > 0 && values[i] < values[i - 1]) {
      values[i] = values[i - 1];
    }
  }
}

/* ---------- deltaValues / derivValues (rollup.go:960-999) ---------- */

void vm_delta_values(double* values, int64_t n) {
  if (n == 0) return;
  double prev_delta = 0;
  double prev_value = values[0];
  for (int64_t i = 1; i < n; i++) {
    double v = values[i];
    prev_delta = v - prev_value;
    values[i - 1] = prev_delta;
    prev_value = v;
  }
  values[n - 1] = prev_delta;
}

void vm_deriv_values(double* values, const int64_t* timestamps, int64_t n) {
  if (n == 0) return;
  double prev_deriv = 0;
  double prev_value = values[0];
  int64_t prev_ts = timestamps[0];
  for (int64_t i = 1; i < n; i++) {
    double v = values[i];
    int64_t ts = timestamps[i];
    if (ts == prev_ts) {
      /* use the previous value for duplicate timestamps */
      values[i - 1] = prev_deriv;
      continue;
    }
    double dt = (double)(ts - prev_ts) / 1e3;
    prev_deriv = (v - prev_value) / dt;
    values[i - 1] = prev_deriv;
    prev_value = v;
    prev_ts = ts;
  }
  values[n - 1] = prev_deriv;
}

/* ---------- dropStaleNaNs slow path (eval.go:2121-2131) ---------- */

int64_t vm_drop_stale_nans(double* values, int64_t* timestamps, int64_t n) {
  int64_t k = 0;
  for (int64_t i = 0; i < n; i++) {
    if (vm_is_stale_nan(values[i])) continue;
    values[k] = values[i];
    timestamps[k] = timestamps[i];
    k++;
  }
  return k;
}

/* ---------- quantile (aggr.go:861-940) ---------- */

static int cmp_f64(const void* a, const void* b) {
  double x = *(const double*)a, y = *(const double*)b;
  if (x < y) return -1;
  if (x > y) return 1;
  return 0;
}

double vm_quantile_sorted(double phi, const double* values, int64_t n) {
  if (n == 0 || isnan(phi)) return NAN_V;
  if (phi < 0) return -INFINITY;
  if (phi > 1) return INFINITY;
  double nn = (double)n;
  double rank = phi * (nn - 1);
  double lower_index = fmax(0, floor(rank));
  double upper_index = fmin(nn - 1, lower_index + 1);
  double weight = rank - floor(rank);
  return values[(int64_t)lower_index] * (1 - weight) + values[(int64_t)upper_index] * weight;
}

/* quantile() copies values, drops NaNs, sorts, then quantileSorted
 * (aggr.go:870-890). Scratch is heap-allocated per call: the oracle is a
 * checker, not the product. */
double vm_quantile(double phi, const double* values, int64_t n) {
  double* tmp = (double*)malloc((size_t)(n > 0 ? n : 1) * sizeof(double));
  int64_t k = 0;
  for (int64_t i = 0; i < n; i++) {
    if (isnan(values[i])) continue;
    tmp[k++] = values[i];
  }
  qsort(tmp, (size_t)k, sizeof(double), cmp_f64);
  double q = vm_quantile_sorted(phi, tmp, k);
  free(tmp);
  return q;
}

/* ---------- getScrapeInterval / getMaxPrevInterval (rollup.go:871-919) ---- */

int64_t vm_get_scrape_interval(const int64_t* timestamps, int64_t n, int64_t default_interval) {
  if (n < 2) return default_interval;
  /* 0.6 quantile of the last up-to-20 intervals, collected newest-first
   * (ordering is irrelevant after the sort inside quantile). */
  double intervals[21];
  int64_t ts_prev = timestamps[n - 1];
  int64_t m = n - 1; /* timestamps[:n-1] */
  int64_t lo = m > 20 ? m - 20 : 0;
  int64_t cnt = 0;
  for (int64_t i = m - 1; i >= lo; i--) {
    intervals[cnt++] = (double)(ts_prev - timestamps[i]);
    ts_prev = timestamps[i];
  }
  int64_t scrape_interval = (int64_t)vm_quantile(0.6, intervals, cnt);
  if (scrape_interval <= 0) return default_interval;
  return scrape_interval;
}

int64_t vm_get_max_prev_interval(int64_t scrape_interval) {
  if (scrape_interval <= 2 * 1000) return scrape_interval + 4 * scrape_interval;
  if (scrape_interval <= 4 * 1000) return scrape_interval + 2 * scrape_interval;
  if (scrape_interval <= 8 * 1000) return scrape_interval + scrape_interval;
  if (scrape_interval <= 16 * 1000) return scrape_interval + scrape_interval / 2;
  if (scrape_interval <= 32 * 1000) return scrape_interval + scrape_interval / 4;
  return scrape_interval + scrape_interval / 8;
}

/* ---------- rollup func argument (rollup.go:523-556) ---------- */

typedef struct {
  double prev_value;
  int64_t prev_timestamp;
  const double* values;
  const int64_t* timestamps;
  int64_t nvalues;
  double real_prev_value;
  double real_next_value;
  int64_t curr_timestamp;
  int64_t idx;
  int64_t window;
  double arg;  /* phi / le / gt / eq / secs for parameterized funcs */
  double arg2; /* holt_winters tf */
} vm_rfa;

/* ---------- individual rollup funcs ---------- */

/* rollupDerivFast (rollup.go:1954-1989) — rate()/deriv_fast() */
static double fn_deriv_fast(const vm_rfa* rfa) {
  const double* values = rfa->values;
  const int64_t* timestamps = rfa->timestamps;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  int64_t prev_timestamp = rfa->prev_timestamp;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    if (n == 1) return NAN_V;
    prev_value = values[0];
    prev_timestamp = timestamps[0];
  } else if (n == 0) {
    return 0;
  }
  double v_end = values[n - 1];
  int64_t t_end = timestamps[n - 1];
  double dv = v_end - prev_value;
  double dt = (double)(t_end - prev_timestamp) / 1e3;
  return dv / dt;
}

/* rollupDelta (rollup.go:1859-1901) — delta()/increase() */
static double fn_delta(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    if (!isnan(rfa->real_prev_value)) {
      return values[n - 1] - rfa->real_prev_value;
    }
    double d = 0;
    if (n > 1) {
      d = values[1] - values[0];
    } else if (!isnan(rfa->real_next_value)) {
      d = rfa->real_next_value - values[0];
    }
    if (fabs(values[0]) < 10 * (fabs(d) + 1)) {
      prev_value = 0;
    } else {
      prev_value = values[0];
      values++;
      n--;
    }
  }
  if (n == 0) return 0;
  return values[n - 1] - prev_value;
}

/* rollupIncreasePure (rollup.go:1835-1857) */
static double fn_increase_pure(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    prev_value = 0;
    if (!isnan(rfa->real_prev_value)) prev_value = rfa->real_prev_value;
  }
  if (n == 0) return 0;
  return values[n - 1] - prev_value;
}

/* rollupDeltaPrometheus (rollup.go:1903-1913) */
static double fn_delta_prometheus(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n < 2) return NAN_V;
  return rfa->values[n - 1] - rfa->values[0];
}

/* rollupDerivFastPrometheus (rollup.go:1946-1952) */
static double fn_rate_prometheus(const vm_rfa* rfa) {
  double delta = fn_delta_prometheus(rfa);
  if (isnan(delta) || rfa->window == 0) return NAN_V;
  return delta / ((double)rfa->window / 1e3);
}

/* rollupIdelta (rollup.go:1915-1937) */
static double fn_idelta(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  if (n == 0) {
    if (isnan(rfa->prev_value)) return NAN_V;
    return 0;
  }
  double last_value = values[n - 1];
  n--;
  if (n == 0) {
    double prev_value = rfa->prev_value;
    if (isnan(prev_value)) return last_value;
    return last_value - prev_value;
  }
  return last_value - values[n - 1];
}

/* rollupIderiv (rollup.go:1991-2038) — irate()/ideriv() */
static double fn_ideriv(const vm_rfa* rfa) {
  const double* values = rfa->values;
  const int64_t* timestamps = rfa->timestamps;
  int64_t n = rfa->nvalues;
  if (n < 2) {
    if (n == 0) return NAN_V;
    if (isnan(rfa->prev_value)) return NAN_V;
    return (values[0] - rfa->prev_value) / ((double)(timestamps[0] - rfa->prev_timestamp) / 1e3);
  }
  double v_end = values[n - 1];
  int64_t t_end = timestamps[n - 1];
  int64_t m = n - 1; /* drop last */
  /* skip data points with duplicate timestamps */
  while (m > 0 && timestamps[m - 1] >= t_end) m--;
  int64_t t_start;
  double v_start;
  if (m == 0) {
    if (isnan(rfa->prev_value)) return 0;
    t_start = rfa->prev_timestamp;
    v_start = rfa->prev_value;
  } else {
    t_start = timestamps[m - 1];
    v_start = values[m - 1];
  }
  double dv = v_end - v_start;
  int64_t dt = t_end - t_start;
  return dv / ((double)dt / 1e3);
}

/* rollupAvg (rollup.go:1541-1559) */
static double fn_avg(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double sum = 0;
  for (int64_t i = 0; i < n; i++) sum += rfa->values[i];
  return sum / (double)n;
}

/* rollupMin (rollup.go:1561-1578) */
static double fn_min(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double m = rfa->values[0];
  for (int64_t i = 0; i < n; i++)
    if (rfa->values[i] < m) m = rfa->values[i];
  return m;
}

/* rollupMax (rollup.go:1580-1597) */
static double fn_max(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double m = rfa->values[0];
  for (int64_t i = 0; i < n; i++)
    if (rfa->values[i] > m) m = rfa->values[i];
  return m;
}

/* rollupSum (rollup.go:1690-1705) */
static double fn_sum(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double sum = 0;
  for (int64_t i = 0; i < n; i++) sum += rfa->values[i];
  return sum;
}

/* rollupSum2 (rollup.go:1727-1739) */
static double fn_sum2(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double sum2 = 0;
  for (int64_t i = 0; i < n; i++) sum2 += rfa->values[i] * rfa->values[i];
  return sum2;
}

/* rollupCount (rollup.go:1771-1779) */
static double fn_count(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  return (double)rfa->nvalues;
}

/* rollupFirst (rollup.go:2375-2386) */
static double fn_first(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  return rfa->values[0];
}

/* rollupDefault / rollupLast (rollup.go:2388-2401) */
static double fn_last(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  return rfa->values[rfa->nvalues - 1];
}

/* newRollupQuantile (rollup.go:1450-1467) */
static double fn_quantile(const vm_rfa* rfa) {
  return vm_quantile(rfa->arg, rfa->values, rfa->nvalues);
}

/* rollupMedian (rollup.go:1599-1601) */
static double fn_median(const vm_rfa* rfa) {
  return vm_quantile(0.5, rfa->values, rfa->nvalues);
}

/* stdvar (rollup.go:1808-1833) — Welford */
static double fn_stdvar(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  if (n == 1) return 0;
  double avg = 0, count = 0, q = 0;
  for (int64_t i = 0; i < n; i++) {
    double v = rfa->values[i];
    if (isnan(v)) continue;
    count++;
    double avg_new = avg + (v - avg) / count;
    q += (v - avg) * (v - avg_new);
    avg = avg_new;
  }
  if (count == 0) return NAN_V;
  return q / count;
}

static double fn_stddev(const vm_rfa* rfa) { return sqrt(fn_stdvar(rfa)); }

/* rollupChanges (rollup.go:2106-2137) */
static double fn_changes(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  int64_t cnt = 0;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    if (!isnan(rfa->real_prev_value)) {
      prev_value = rfa->real_prev_value;
    } else {
      cnt++;
      prev_value = values[0];
      values++;
      n--;
    }
  }
  for (int64_t i = 0; i < n; i++) {
    double v = values[i];
    if (v != prev_value) {
      if (fabs(v - prev_value) < 1e-12 * fabs(v)) continue;
      cnt++;
      prev_value = v;
    }
  }
  return (double)cnt;
}

/* rollupChangesPrometheus (rollup.go:2082-2104) */
static double fn_changes_prometheus(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  if (n < 1) return NAN_V;
  double prev_value = values[0];
  int64_t cnt = 0;
  for (int64_t i = 1; i < n; i++) {
    double v = values[i];
    if (v != prev_value) {
      if (fabs(v - prev_value) < 1e-12 * fabs(v)) continue;
      cnt++;
      prev_value = v;
    }
  }
  return (double)cnt;
}

/* rollupResets (rollup.go:2175-2204) — also decreases_over_time */
static double fn_resets(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  if (n == 0) {
    if (isnan(rfa->prev_value)) return NAN_V;
    return 0;
  }
  double prev_value = rfa->prev_value;
  if (isnan(prev_value)) {
    prev_value = values[0];
    values++;
    n--;
  }
  if (n == 0) return 0;
  int64_t cnt = 0;
  for (int64_t i = 0; i < n; i++) {
    double v = values[i];
    if (v < prev_value) {
      /* precision-error guard: skips the count AND the prev update */
      if (fabs(v - prev_value) < 1e-12 * fabs(v)) continue;
      cnt++;
    }
    prev_value = v;
  }
  return (double)cnt;
}

/* rollupIncreases (rollup.go:2139-2169) */
static double fn_increases(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  if (n == 0) {
    if (isnan(rfa->prev_value)) return NAN_V;
    return 0;
  }
  double prev_value = rfa->prev_value;
  if (isnan(prev_value)) {
    prev_value = values[0];
    values++;
    n--;
  }
  if (n == 0) return 0;
  int64_t cnt = 0;
  for (int64_t i = 0; i < n; i++) {
    double v = values[i];
    if (v > prev_value) {
      /* precision-error guard: skips the count AND the prev update */
      if (fabs(v - prev_value) < 1e-12 * fabs(v)) continue;
      cnt++;
    }
    prev_value = v;
  }
  return (double)cnt;
}

/* rollupLag (rollup.go:2055-2065) */
static double fn_lag(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) {
    if (isnan(rfa->prev_value)) return NAN_V;
    return (double)(rfa->curr_timestamp - rfa->prev_timestamp) / 1e3;
  }
  return (double)(rfa->curr_timestamp - rfa->timestamps[n - 1]) / 1e3;
}

/* rollupLifetime (rollup.go:2040-2053) */
static double fn_lifetime(const vm_rfa* rfa) {
  const int64_t* timestamps = rfa->timestamps;
  int64_t n = rfa->nvalues;
  if (isnan(rfa->prev_value)) {
    if (n < 2) return NAN_V;
    return (double)(timestamps[n - 1] - timestamps[0]) / 1e3;
  }
  if (n == 0) return NAN_V;
  return (double)(timestamps[n - 1] - rfa->prev_timestamp) / 1e3;
}

/* rollupScrapeInterval (rollup.go:2067-2080) */
static double fn_scrape_interval(const vm_rfa* rfa) {
  const int64_t* timestamps = rfa->timestamps;
  int64_t n = rfa->nvalues;
  if (isnan(rfa->prev_value)) {
    if (n < 2) return NAN_V;
    return ((double)(timestamps[n - 1] - timestamps[0]) / 1e3) / (double)(n - 1);
  }
  if (n == 0) return NAN_V;
  return ((double)(timestamps[n - 1] - rfa->prev_timestamp) / 1e3) / (double)n;
}

/* rollupRateOverSum (rollup.go:1707-1719) */
static double fn_rate_over_sum(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  double sum = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++) sum += rfa->values[i];
  return sum / ((double)rfa->window / 1e3);
}

/* rollupRange (rollup.go:1721-1725) */
static double fn_range(const vm_rfa* rfa) { return fn_max(rfa) - fn_min(rfa); }

/* rollupTfirst (rollup.go:1643-1654) */
static double fn_tfirst(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  return (double)rfa->timestamps[0] / 1e3;
}

/* rollupTlast (rollup.go:1656-1667) */
static double fn_tlast(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  return (double)rfa->timestamps[rfa->nvalues - 1] / 1e3;
}

/* rollupTmin (rollup.go:1603-1621) */
static double fn_tmin(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double min_value = rfa->values[0];
  int64_t min_ts = rfa->timestamps[0];
  for (int64_t i = 0; i < n; i++) {
    if (rfa->values[i] <= min_value) {
      min_value = rfa->values[i];
      min_ts = rfa->timestamps[i];
    }
  }
  return (double)min_ts / 1e3;
}

/* rollupTmax (rollup.go:1623-1641) */
static double fn_tmax(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double max_value = rfa->values[0];
  int64_t max_ts = rfa->timestamps[0];
  for (int64_t i = 0; i < n; i++) {
    if (rfa->values[i] >= max_value) {
      max_value = rfa->values[i];
      max_ts = rfa->timestamps[i];
    }
  }
  return (double)max_ts / 1e3;
}

/* rollupTlastChange (rollup.go:1669-1688) */
static double fn_tlast_change(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double last_value = rfa->values[n - 1];
  for (int64_t i = n - 2; i >= 0; i--) {
    if (rfa->values[i] != last_value) {
      return (double)rfa->timestamps[i + 1] / 1e3;
    }
  }
  if (isnan(rfa->prev_value) || rfa->prev_value != last_value) {
    return (double)rfa->timestamps[0] / 1e3;
  }
  return NAN_V;
}

/* rollupGeomean (rollup.go:1741-1753) */
static double fn_geomean(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double p = 1.0;
  for (int64_t i = 0; i < n; i++) p *= rfa->values[i];
  return pow(p, 1.0 / (double)n);
}

/* rollupPresent / rollupAbsent (rollup.go:1755-1769) */
static double fn_present(const vm_rfa* rfa) { return rfa->nvalues > 0 ? 1 : NAN_V; }
static double fn_absent(const vm_rfa* rfa) { return rfa->nvalues == 0 ? 1 : NAN_V; }

/* rollupStaleSamples (rollup.go:1781-1793) */
static double fn_stale_samples(const vm_rfa* rfa) {
  if (rfa->nvalues == 0) return NAN_V;
  int64_t cnt = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (vm_is_stale_nan(rfa->values[i])) cnt++;
  return (double)cnt;
}

/* count/share/sum filters (rollup.go:1185-1276) */
static double count_le(const vm_rfa* rfa) {
  int64_t n = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] <= rfa->arg) n++;
  return (double)n;
}
static double count_gt(const vm_rfa* rfa) {
  int64_t n = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] > rfa->arg) n++;
  return (double)n;
}
static double count_eq(const vm_rfa* rfa) {
  int64_t n = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] == rfa->arg) n++;
  return (double)n;
}
static double count_ne(const vm_rfa* rfa) {
  int64_t n = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] != rfa->arg) n++;
  return (double)n;
}
static double share_le(const vm_rfa* rfa) { return count_le(rfa) / (double)rfa->nvalues; }
static double share_gt(const vm_rfa* rfa) { return count_gt(rfa) / (double)rfa->nvalues; }
static double share_eq(const vm_rfa* rfa) { return count_eq(rfa) / (double)rfa->nvalues; }
static double sum_le(const vm_rfa* rfa) {
  double s = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] <= rfa->arg) s += rfa->values[i];
  return s;
}
static double sum_gt(const vm_rfa* rfa) {
  double s = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] > rfa->arg) s += rfa->values[i];
  return s;
}
static double sum_eq(const vm_rfa* rfa) {
  double s = 0;
  for (int64_t i = 0; i < rfa->nvalues; i++)
    if (rfa->values[i] == rfa->arg) s += rfa->values[i];
  return s;
}

/* linearRegression (rollup.go:1099-1136) */
static void linear_regression(const double* values, const int64_t* timestamps, int64_t n,
                              int64_t intercept_time, double* out_v, double* out_k) {
  if (n == 0) { *out_v = NAN_V; *out_k = NAN_V; return; }
  int is_const = 1;
  for (int64_t i = 1; i < n; i++)
    if (values[i] != values[i - 1]) { is_const = 0; break; }
  if (is_const) { *out_v = values[0]; *out_k = 0; return; }
  double v_sum = 0, t_sum = 0, tv_sum = 0, tt_sum = 0;
  int64_t cnt = 0;
  for (int64_t i = 0; i < n; i++) {
    double v = values[i];
    if (isnan(v)) continue;
    double dt = (double)(timestamps[i] - intercept_time) / 1e3;
    v_sum += v;
    t_sum += dt;
    tv_sum += dt * v;
    tt_sum += dt * dt;
    cnt++;
  }
  if (cnt == 0) { *out_v = NAN_V; *out_k = NAN_V; return; }
  double k = 0;
  double t_diff = tt_sum - t_sum * t_sum / (double)cnt;
  if (fabs(t_diff) >= 1e-6) {
    k = (tv_sum - t_sum * v_sum / (double)cnt) / t_diff;
  }
  *out_v = v_sum / (double)cnt - k * t_sum / (double)cnt;
  *out_k = k;
}

/* rollupDerivSlow (rollup.go:1939-1944) */
static double fn_deriv(const vm_rfa* rfa) {
  double v, k;
  linear_regression(rfa->values, rfa->timestamps, rfa->nvalues, rfa->curr_timestamp, &v, &k);
  return k;
}

/* newRollupPredictLinear (rollup.go:1080-1097); rfa->arg = secs */
static double fn_predict_linear(const vm_rfa* rfa) {
  double v, k;
  linear_regression(rfa->values, rfa->timestamps, rfa->nvalues, rfa->curr_timestamp, &v, &k);
  if (isnan(v)) return NAN_V;
  return v + k * rfa->arg;
}

/* rollupAscentOverTime / rollupDescentOverTime (rollup.go:2317-2357) */
static double fn_ascent(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    prev_value = values[0];
    values++;
    n--;
  }
  double s = 0;
  for (int64_t i = 0; i < n; i++) {
    double d = values[i] - prev_value;
    if (d > 0) s += d;
    prev_value = values[i];
  }
  return s;
}
static double fn_descent(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    prev_value = values[0];
    values++;
    n--;
  }
  double s = 0;
  for (int64_t i = 0; i < n; i++) {
    double d = prev_value - values[i];
    if (d > 0) s += d;
    prev_value = values[i];
  }
  return s;
}

/* rollupZScoreOverTime (rollup.go:2359-2373) */
static double fn_zscore(const vm_rfa* rfa) {
  double scrape_interval = fn_scrape_interval(rfa);
  double lag = fn_lag(rfa);
  if (isnan(scrape_interval) || isnan(lag) || lag > scrape_interval) return NAN_V;
  double d = fn_last(rfa) - fn_avg(rfa);
  if (d == 0) return 0;
  return d / fn_stddev(rfa);
}

/* rollupIntegrate (rollup.go:2417-2450) */
static double fn_integrate(const vm_rfa* rfa) {
  const double* values = rfa->values;
  const int64_t* timestamps = rfa->timestamps;
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  int64_t prev_timestamp = rfa->curr_timestamp - rfa->window;
  if (isnan(prev_value)) {
    if (n == 0) return NAN_V;
    prev_value = values[0];
    prev_timestamp = timestamps[0];
    values++;
    timestamps++;
    n--;
  }
  double sum = 0;
  for (int64_t i = 0; i < n; i++) {
    double dt = (double)(timestamps[i] - prev_timestamp) / 1e3;
    sum += prev_value * dt;
    prev_timestamp = timestamps[i];
    prev_value = values[i];
  }
  if (!isnan(rfa->real_next_value)) {
    double dt = (double)(rfa->curr_timestamp - prev_timestamp) / 1e3;
    sum += prev_value * dt;
  }
  return sum;
}

/* rollupDistinct (rollup.go:2403-2415) — set cardinality via sort */
static double fn_distinct(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double* tmp = (double*)malloc((size_t)n * sizeof(double));
  memcpy(tmp, rfa->values, (size_t)n * sizeof(double));
  qsort(tmp, (size_t)n, sizeof(double), cmp_f64);
  int64_t cnt = 0;
  for (int64_t i = 0; i < n; i++) {
    if (i == 0 || tmp[i] != tmp[i - 1]) cnt++;
  }
  /* Go map keys: NaN != NaN so each NaN is distinct — values are clean of
   * NaNs on this path, so sorted-unique matches the map cardinality. */
  free(tmp);
  return (double)cnt;
}

/* rollupMAD (rollup.go:1469-1488) */
static double fn_mad(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  double median = vm_quantile(0.5, rfa->values, n);
  double* ds = (double*)malloc((size_t)(n > 0 ? n : 1) * sizeof(double));
  for (int64_t i = 0; i < n; i++) ds[i] = fabs(rfa->values[i] - median);
  double v = vm_quantile(0.5, ds, n);
  free(ds);
  return v;
}

/* rollupModeOverTime (rollup.go:2293-2303) + modeNoNaNs (aggr.go:541-564) */
static double fn_mode(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  double prev_value = rfa->prev_value;
  if (n == 0) return prev_value;
  double* a = (double*)malloc((size_t)n * sizeof(double));
  memcpy(a, rfa->values, (size_t)n * sizeof(double));
  qsort(a, (size_t)n, sizeof(double), cmp_f64);
  int64_t j = -1;
  int64_t d_max = 0;
  double mode = prev_value;
  for (int64_t i = 0; i < n; i++) {
    double v = a[i];
    if (prev_value == v) continue;
    int64_t d = i - j;
    if (d > d_max || isnan(mode)) {
      d_max = d;
      mode = prev_value;
    }
    j = i;
    prev_value = v;
  }
  int64_t d = n - j;
  if (d > d_max || isnan(mode)) mode = prev_value;
  free(a);
  return mode;
}

/* newRollupDurationOverTime (rollup.go:1151-1180); rfa->arg = dMax secs */
static double fn_duration(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  int64_t t_prev = rfa->timestamps[0];
  int64_t d_sum = 0;
  int64_t d_max = (int64_t)(rfa->arg * 1000);
  for (int64_t i = 0; i < n; i++) {
    int64_t d = rfa->timestamps[i] - t_prev;
    if (d <= d_max) d_sum += d;
    t_prev = rfa->timestamps[i];
  }
  return (double)d_sum / 1000;
}

/* rollupOutlierIQR (rollup.go:1427-1448) */
static double fn_outlier_iqr(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  if (n < 2) return NAN_V;
  double q25 = vm_quantile(0.25, rfa->values, n);
  double q75 = vm_quantile(0.75, rfa->values, n);
  double iqr = 1.5 * (q75 - q25);
  double v = rfa->values[n - 1];
  if (v > q75 + iqr || v < q25 - iqr) return v;
  return NAN_V;
}

/* candlestick helpers (rollup.go:2206-2291): the window is
 * (currTimestamp-window, currTimestamp) EXCLUSIVE of currTimestamp, and the
 * pre-window sample participates when it falls inside the window. */
static int64_t candlestick_len(const vm_rfa* rfa) {
  int64_t n = rfa->nvalues;
  while (n > 0 && rfa->timestamps[n - 1] >= rfa->curr_timestamp) n--;
  return n;
}
static double candlestick_first(const vm_rfa* rfa) {
  if (rfa->prev_timestamp + rfa->window >= rfa->curr_timestamp) return rfa->prev_value;
  return NAN_V;
}
static double fn_open(const vm_rfa* rfa) {
  double v = candlestick_first(rfa);
  if (!isnan(v)) return v;
  int64_t n = candlestick_len(rfa);
  if (n == 0) return NAN_V;
  return rfa->values[0];
}
static double fn_close(const vm_rfa* rfa) {
  int64_t n = candlestick_len(rfa);
  if (n == 0) return candlestick_first(rfa);
  return rfa->values[n - 1];
}
static double fn_high(const vm_rfa* rfa) {
  int64_t n = candlestick_len(rfa);
  const double* values = rfa->values;
  double maxv = candlestick_first(rfa);
  int64_t i = 0;
  if (isnan(maxv)) {
    if (n == 0) return NAN_V;
    maxv = values[0];
    i = 1;
  }
  for (; i < n; i++)
    if (values[i] > maxv) maxv = values[i];
  return maxv;
}
static double fn_low(const vm_rfa* rfa) {
  int64_t n = candlestick_len(rfa);
  const double* values = rfa->values;
  double minv = candlestick_first(rfa);
  int64_t i = 0;
  if (isnan(minv)) {
    if (n == 0) return NAN_V;
    minv = values[0];
    i = 1;
  }
  for (; i < n; i++)
    if (values[i] < minv) minv = values[i];
  return minv;
}

/* newRollupHoltWinters (rollup.go:1030-1077); arg=sf, arg2=tf */
static double fn_holt_winters(const vm_rfa* rfa) {
  const double* values = rfa->values;
  int64_t n = rfa->nvalues;
  if (n == 0) return NAN_V;
  double sf = rfa->arg;
  if (sf < 0 || sf > 1) return NAN_V;
  double tf = rfa->arg2;
  if (tf < 0 || tf > 1) return NAN_V;
  double s0 = rfa->prev_value;
  if (isnan(s0)) {
    s0 = values[0];
    values++;
    n--;
    if (n == 0) return s0;
  }
  double b0 = values[0] - s0;
  for (int64_t i = 0; i < n; i++) {
    double v = values[i];
    double s1 = sf * v + (1 - sf) * (s0 + b0);
    double b1 = tf * (s1 - s0) + (1 - tf) * b0;
    s0 = s1;
    b0 = b1;
  }
  return s0;
}

/* rollupHoeffdingBoundInternal (rollup.go:1353-1381); arg=phi */
static void hoeffding_internal(const vm_rfa* rfa, double* out_bound, double* out_avg) {
  int64_t n = rfa->nvalues;
  if (n == 0) { *out_bound = NAN_V; *out_avg = NAN_V; return; }
  if (n == 1) { *out_bound = 0; *out_avg = rfa->values[0]; return; }
  double v_max = fn_max(rfa);
  double v_min = fn_min(rfa);
  double v_avg = fn_avg(rfa);
  double v_range = v_max - v_min;
  if (v_range <= 0) { *out_bound = 0; *out_avg = v_avg; return; }
  double phi = rfa->arg;
  if (phi >= 1) { *out_bound = INFINITY; *out_avg = v_avg; return; }
  if (phi <= 0) { *out_bound = 0; *out_avg = v_avg; return; }
  *out_bound = v_range * sqrt(log(1 / (1 - phi)) / (2 * (double)n));
  *out_avg = v_avg;
}
static double fn_hoeffding_lower(const vm_rfa* rfa) {
  double b, a;
  hoeffding_internal(rfa, &b, &a);
  return a - b;
}
static double fn_hoeffding_upper(const vm_rfa* rfa) {
  double b, a;
  hoeffding_internal(rfa, &b, &a);
  return a + b;
}

typedef double (*vm_rollup_fn)(const vm_rfa*);

static vm_rollup_fn fn_table(int32_t func) {
  switch (func) {
    case VM_FN_RATE: return fn_deriv_fast;
    case VM_FN_INCREASE: return fn_delta;
    case VM_FN_INCREASE_PURE: return fn_increase_pure;
    case VM_FN_DELTA: return fn_delta;
    case VM_FN_DELTA_PROMETHEUS: return fn_delta_prometheus;
    case VM_FN_RATE_PROMETHEUS: return fn_rate_prometheus;
    case VM_FN_IRATE: return fn_ideriv;
    case VM_FN_IDERIV: return fn_ideriv;
    case VM_FN_IDELTA: return fn_idelta;
    case VM_FN_DERIV_FAST: return fn_deriv_fast;
    case VM_FN_AVG: return fn_avg;
    case VM_FN_MIN: return fn_min;
    case VM_FN_MAX: return fn_max;
    case VM_FN_SUM: return fn_sum;
    case VM_FN_SUM2: return fn_sum2;
    case VM_FN_COUNT: return fn_count;
    case VM_FN_FIRST: return fn_first;
    case VM_FN_LAST: return fn_last;
    case VM_FN_QUANTILE: return fn_quantile;
    case VM_FN_MEDIAN: return fn_median;
    case VM_FN_STDDEV: return fn_stddev;
    case VM_FN_STDVAR: return fn_stdvar;
    case VM_FN_CHANGES: return fn_changes;
    case VM_FN_CHANGES_PROMETHEUS: return fn_changes_prometheus;
    case VM_FN_RESETS: return fn_resets;
    case VM_FN_LAG: return fn_lag;
    case VM_FN_LIFETIME: return fn_lifetime;
    case VM_FN_SCRAPE_INTERVAL: return fn_scrape_interval;
    case VM_FN_RATE_OVER_SUM: return fn_rate_over_sum;
    case VM_FN_RANGE: return fn_range;
    case VM_FN_TFIRST: return fn_tfirst;
    case VM_FN_TLAST: return fn_tlast;
    case VM_FN_TMIN: return fn_tmin;
    case VM_FN_TMAX: return fn_tmax;
    case VM_FN_TLAST_CHANGE: return fn_tlast_change;
    case VM_FN_GEOMEAN: return fn_geomean;
    case VM_FN_PRESENT: return fn_present;
    case VM_FN_ABSENT: return fn_absent;
    case VM_FN_STALE_SAMPLES: return fn_stale_samples;
    case VM_FN_COUNT_LE: return count_le;
    case VM_FN_COUNT_GT: return count_gt;
    case VM_FN_COUNT_EQ: return count_eq;
    case VM_FN_COUNT_NE: return count_ne;
    case VM_FN_SHARE_LE: return share_le;
    case VM_FN_SHARE_GT: return share_gt;
    case VM_FN_SHARE_EQ: return share_eq;
    case VM_FN_SUM_LE: return sum_le;
    case VM_FN_SUM_GT: return sum_gt;
    case VM_FN_SUM_EQ: return sum_eq;
    case VM_FN_DERIV: return fn_deriv;
    case VM_FN_PREDICT_LINEAR: return fn_predict_linear;
    case VM_FN_ASCENT: return fn_ascent;
    case VM_FN_DESCENT: return fn_descent;
    case VM_FN_ZSCORE: return fn_zscore;
    case VM_FN_INTEGRATE: return fn_integrate;
    case VM_FN_DISTINCT: return fn_distinct;
    case VM_FN_INCREASES: return fn_increases;
    case VM_FN_DECREASES: return fn_resets;
    case VM_FN_MAD: return fn_mad;
    case VM_FN_DEFAULT_ROLLUP: return fn_last;
    case VM_FN_MODE: return fn_mode;
    case VM_FN_DURATION: return fn_duration;
    case VM_FN_OUTLIER_IQR: return fn_outlier_iqr;
    case VM_FN_OPEN: return fn_open;
    case VM_FN_CLOSE: return fn_close;
    case VM_FN_LOW: return fn_low;
    case VM_FN_HIGH: return fn_high;
    case VM_FN_HOLT_WINTERS: return fn_holt_winters;
    case VM_FN_HOEFFDING_LOWER: return fn_hoeffding_lower;
    case VM_FN_HOEFFDING_UPPER: return fn_hoeffding_upper;
    default: return NULL;
  }
}

double vm_call_rollup_fn(int32_t func, double prev_value, int64_t prev_timestamp,
                         const double* values, const int64_t* timestamps, int64_t n,
                         double real_prev_value, double real_next_value,
                         int64_t curr_timestamp, int64_t idx, int64_t window, double arg,
                         double arg2) {
  vm_rollup_fn f = fn_table(func);
  if (!f) return NAN_V;
  vm_rfa rfa;
  rfa.prev_value = prev_value;
  rfa.prev_timestamp = prev_timestamp;
  rfa.values = values;
  rfa.timestamps = timestamps;
  rfa.nvalues = n;
  rfa.real_prev_value = real_prev_value;
  rfa.real_next_value = real_next_value;
  rfa.curr_timestamp = curr_timestamp;
  rfa.idx = idx;
  rfa.window = window;
  rfa.arg = arg;
  rfa.arg2 = arg2;
  return f(&rfa);
}

/* seekFirstTimestampIdxAfter (rollup.go:825-855): first index in
 * timestamps[0..n) with timestamps[idx] > seek. The reference's ±2-hint fast
 * path only narrows the search range; the result is the plain upper bound. */
static int64_t upper_bound_i64(const int64_t* timestamps, int64_t n, int64_t seek) {
  int64_t i = 0, j = n;
  while (i < j) {
    int64_t h = (i + j) >> 1;
    if (timestamps[h] <= seek) i = h + 1;
    else j = h;
  }
  return i;
}

/* doInternal (rollup.go:701-823) */
uint64_t vm_rollup_do(const vm_rollup_config* rc, const double* values,
                      const int64_t* timestamps, int64_t n, double* dst) {
  int64_t n_grid = vm_grid_points(rc->start, rc->end, rc->step);
  if (n_grid < 0) return 0;

  int64_t max_prev_interval = rc->step;
  if (rc->start < rc->end) {
    int64_t scrape_interval = vm_get_scrape_interval(timestamps, n, rc->step);
    max_prev_interval = vm_get_max_prev_interval(scrape_interval);
  }
  if (rc->lookback_delta > 0 && max_prev_interval > rc->lookback_delta) {
    max_prev_interval = rc->lookback_delta;
  }
  if (rc->min_staleness_interval > 0 && max_prev_interval < rc->min_staleness_interval) {
    max_prev_interval = rc->min_staleness_interval;
  }
  int64_t window = rc->window;
  if (window <= 0) {
    window = rc->step;
    if (rc->may_adjust_window && window < max_prev_interval) {
      window = max_prev_interval;
    }
    if (rc->is_default_rollup && rc->lookback_delta > 0 && window > rc->lookback_delta) {
      window = rc->lookback_delta;
    }
  }

  vm_rfa rfa;
  rfa.window = window;
  rfa.arg = rc->arg;
  rfa.arg2 = rc->arg2;
  vm_rollup_fn f = fn_table(rc->func);
  if (!f) return 0;

  int64_t i = 0, j = 0;
  uint64_t samples_scanned = (uint64_t)n;
  uint64_t per_call = (uint64_t)rc->samples_scanned_per_call;
  int64_t t_end = rc->start;
  for (int64_t g = 0; g < n_grid; g++, t_end += rc->step) {
    int64_t t_start = t_end - window;
    i += upper_bound_i64(timestamps + i, n - i, t_start);
    if (j < i) j = i;
    j += upper_bound_i64(timestamps + j, n - j, t_end);

    rfa.prev_value = NAN_V;
    rfa.prev_timestamp = t_start - max_prev_interval;
    if (i < n && i > 0 && timestamps[i - 1] > rfa.prev_timestamp) {
      rfa.prev_value = values[i - 1];
      rfa.prev_timestamp = timestamps[i - 1];
    }
    rfa.values = values + i;
    rfa.timestamps = timestamps + i;
    rfa.nvalues = j - i;
    rfa.real_prev_value = NAN_V;
    if (i > 0) {
      /* realPrevValue eligibility (rollup.go:787-804) */
      int64_t curr_timestamp = t_start;
      if (rfa.nvalues > 0) curr_timestamp = rfa.timestamps[0];
      if (rc->lookback_delta == 0 || (curr_timestamp - timestamps[i - 1]) < rc->lookback_delta) {
        rfa.real_prev_value = values[i - 1];
      }
    }
    rfa.real_next_value = (j < n) ? values[j] : NAN_V;
    rfa.curr_timestamp = t_end;
    rfa.idx = g;
    dst[g] = f(&rfa);
    if (per_call > 0) samples_scanned += per_call;
    else samples_scanned += (uint64_t)rfa.nvalues;
  }
  return samples_scanned;
}
