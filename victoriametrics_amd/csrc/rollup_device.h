/* rollup_device.h — device-side rollup function set for gfx950.
 *
 * Independent MI355X-native implementation of the reference's rollup
 * functions (app/vmselect/promql/rollup.go; file:line cited per function).
 * Deliberately NOT shared with the CPU oracle (oracle/rollup.c): the oracle
 * is the checker, this is the product; keeping the implementations separate
 * is what makes the parity tests meaningful.
 *
 * Each grid point is evaluated by ONE lane with left-to-right serial window
 * loops, so arithmetic-only functions are bit-exact against the sequential
 * reference semantics.  Functions needing an order statistic (quantile,
 * median, mad, mode) use O(w^2) count-based selection instead of a sort —
 * same result, no per-lane scratch.
 */
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define VM_DEV __device__ __forceinline__

static VM_DEV double vm_dnan() { return __longlong_as_double(0x7ff8000000000000LL); }
static VM_DEV double vm_dinf() { return __longlong_as_double(0x7ff0000000000000LL); }
static VM_DEV bool vm_isnan(double v) { return v != v; }
static VM_DEV bool vm_is_stale_nan(double v) {
  /* lib/decimal/decimal.go:403-419 */
  return __double_as_longlong(v) == 0x7ff0000000000002LL;
}

/* Function ids — numbering matches oracle/vm_oracle.h vm_func_id. */
enum {
  VMF_RATE = 0, VMF_INCREASE, VMF_INCREASE_PURE, VMF_DELTA,
  VMF_DELTA_PROMETHEUS, VMF_RATE_PROMETHEUS, VMF_IRATE, VMF_IDERIV,
  VMF_IDELTA, VMF_DERIV_FAST, VMF_AVG, VMF_MIN, VMF_MAX, VMF_SUM, VMF_SUM2,
  VMF_COUNT, VMF_FIRST, VMF_LAST, VMF_QUANTILE, VMF_MEDIAN, VMF_STDDEV,
  VMF_STDVAR, VMF_CHANGES, VMF_CHANGES_PROMETHEUS, VMF_RESETS, VMF_LAG,
  VMF_LIFETIME, VMF_SCRAPE_INTERVAL, VMF_RATE_OVER_SUM, VMF_RANGE,
  VMF_TFIRST, VMF_TLAST, VMF_TMIN, VMF_TMAX, VMF_TLAST_CHANGE, VMF_GEOMEAN,
  VMF_PRESENT, VMF_ABSENT, VMF_STALE_SAMPLES, VMF_COUNT_LE, VMF_COUNT_GT,
  VMF_COUNT_EQ, VMF_COUNT_NE, VMF_SHARE_LE, VMF_SHARE_GT, VMF_SHARE_EQ,
  VMF_SUM_LE, VMF_SUM_GT, VMF_SUM_EQ, VMF_DERIV, VMF_PREDICT_LINEAR,
  VMF_ASCENT, VMF_DESCENT, VMF_ZSCORE, VMF_INTEGRATE, VMF_DISTINCT,
  VMF_INCREASES, VMF_DECREASES, VMF_MAD, VMF_DEFAULT_ROLLUP, VMF_MODE,
  VMF_DURATION, VMF_OUTLIER_IQR, VMF_OPEN, VMF_CLOSE, VMF_LOW, VMF_HIGH,
  VMF_HOLT_WINTERS, VMF_HOEFFDING_LOWER, VMF_HOEFFDING_UPPER,
};

/* rollupFuncArg (rollup.go:523-556) restricted to what the funcs read. */
struct VmRfa {
  double prev_value;
  int64_t prev_timestamp;
  const double* values;
  const int64_t* timestamps;
  int n;
  double real_prev_value;
  double real_next_value;
  int64_t curr_timestamp;
  int64_t window;
  double arg;
  double arg2;
};

/* ---- order-statistic helpers (count-based selection, no sort) ---- */

/* k-th smallest (0-based) among the non-NaN elements of v[0..n). */
static VM_DEV double vm_select_kth(const double* v, int n, int k) {
  for (int c = 0; c < n; c++) {
    double x = v[c];
    if (vm_isnan(x)) continue;
    int less = 0, eq = 0;
    for (int i = 0; i < n; i++) {
      double y = v[i];
      if (vm_isnan(y)) continue;
      if (y < x) less++;
      else if (y == x) eq++;
    }
    if (less <= k && k < less + eq) return x;
  }
  return vm_dnan();
}

/* quantileSorted over the sorted non-NaN elements (aggr.go:922-940),
 * with the NaN filtering of quantile (aggr.go:870-890). */
static VM_DEV double vm_dev_quantile(double phi, const double* v, int n) {
  int m = 0;
  for (int i = 0; i < n; i++)
    if (!vm_isnan(v[i])) m++;
  if (m == 0 || vm_isnan(phi)) return vm_dnan();
  if (phi < 0) return -vm_dinf();
  if (phi > 1) return vm_dinf();
  double nn = (double)m;
  double rank = phi * (nn - 1);
  double lower_index = fmax(0.0, floor(rank));
  double upper_index = fmin(nn - 1, lower_index + 1);
  double weight = rank - floor(rank);
  double lo = vm_select_kth(v, n, (int)lower_index);
  double hi = vm_select_kth(v, n, (int)upper_index);
  return lo * (1 - weight) + hi * weight;
}

/* k-th smallest of |v[i]-med| (for MAD), NaN-skipping. */
static VM_DEV double vm_select_kth_absdev(const double* v, int n, double med, int k) {
  for (int c = 0; c < n; c++) {
    if (vm_isnan(v[c])) continue;
    double x = fabs(v[c] - med);
    int less = 0, eq = 0;
    for (int i = 0; i < n; i++) {
      if (vm_isnan(v[i])) continue;
      double y = fabs(v[i] - med);
      if (y < x) less++;
      else if (y == x) eq++;
    }
    if (less <= k && k < less + eq) return x;
  }
  return vm_dnan();
}

static VM_DEV double vm_dev_quantile_absdev(double phi, const double* v, int n, double med) {
  int m = 0;
  for (int i = 0; i < n; i++)
    if (!vm_isnan(v[i])) m++;
  if (m == 0 || vm_isnan(phi)) return vm_dnan();
  double nn = (double)m;
  double rank = phi * (nn - 1);
  double li = fmax(0.0, floor(rank));
  double ui = fmin(nn - 1, li + 1);
  double w = rank - floor(rank);
  double lo = vm_select_kth_absdev(v, n, med, (int)li);
  double hi = vm_select_kth_absdev(v, n, med, (int)ui);
  return lo * (1 - w) + hi * w;
}

/* ---- individual funcs ---- */

/* rollupDerivFast (rollup.go:1954-1989) */
static VM_DEV double vmf_deriv_fast(const VmRfa* r) {
  const double* values = r->values;
  const int64_t* ts = r->timestamps;
  int n = r->n;
  double pv = r->prev_value;
  int64_t pt = r->prev_timestamp;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    if (n == 1) return vm_dnan();
    pv = values[0];
    pt = ts[0];
  } else if (n == 0) {
    return 0;
  }
  double dv = values[n - 1] - pv;
  double dt = (double)(ts[n - 1] - pt) / 1e3;
  return dv / dt;
}

/* rollupDelta (rollup.go:1859-1901) */
static VM_DEV double vmf_delta(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  double pv = r->prev_value;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    if (!vm_isnan(r->real_prev_value)) return values[n - 1] - r->real_prev_value;
    double d = 0;
    if (n > 1) d = values[1] - values[0];
    else if (!vm_isnan(r->real_next_value)) d = r->real_next_value - values[0];
    if (fabs(values[0]) < 10 * (fabs(d) + 1)) {
      pv = 0;
    } else {
      pv = values[0];
      values++;
      n--;
    }
  }
  if (n == 0) return 0;
  return values[n - 1] - pv;
}

/* rollupIncreasePure (rollup.go:1835-1857) */
static VM_DEV double vmf_increase_pure(const VmRfa* r) {
  int n = r->n;
  double pv = r->prev_value;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    pv = 0;
    if (!vm_isnan(r->real_prev_value)) pv = r->real_prev_value;
  }
  if (n == 0) return 0;
  return r->values[n - 1] - pv;
}

/* rollupDeltaPrometheus (rollup.go:1903-1913) */
static VM_DEV double vmf_delta_prometheus(const VmRfa* r) {
  if (r->n < 2) return vm_dnan();
  return r->values[r->n - 1] - r->values[0];
}

/* rollupDerivFastPrometheus (rollup.go:1946-1952) */
static VM_DEV double vmf_rate_prometheus(const VmRfa* r) {
  double d = vmf_delta_prometheus(r);
  if (vm_isnan(d) || r->window == 0) return vm_dnan();
  return d / ((double)r->window / 1e3);
}

/* rollupIdelta (rollup.go:1915-1937) */
static VM_DEV double vmf_idelta(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  if (n == 0) {
    if (vm_isnan(r->prev_value)) return vm_dnan();
    return 0;
  }
  double last = values[n - 1];
  n--;
  if (n == 0) {
    if (vm_isnan(r->prev_value)) return last;
    return last - r->prev_value;
  }
  return last - values[n - 1];
}

/* rollupIderiv (rollup.go:1991-2038) */
static VM_DEV double vmf_ideriv(const VmRfa* r) {
  const double* values = r->values;
  const int64_t* ts = r->timestamps;
  int n = r->n;
  if (n < 2) {
    if (n == 0) return vm_dnan();
    if (vm_isnan(r->prev_value)) return vm_dnan();
    return (values[0] - r->prev_value) / ((double)(ts[0] - r->prev_timestamp) / 1e3);
  }
  double v_end = values[n - 1];
  int64_t t_end = ts[n - 1];
  int m = n - 1;
  while (m > 0 && ts[m - 1] >= t_end) m--;
  int64_t t_start;
  double v_start;
  if (m == 0) {
    if (vm_isnan(r->prev_value)) return 0;
    t_start = r->prev_timestamp;
    v_start = r->prev_value;
  } else {
    t_start = ts[m - 1];
    v_start = values[m - 1];
  }
  return (v_end - v_start) / ((double)(t_end - t_start) / 1e3);
}

/* rollupAvg/Min/Max/Sum/Sum2/Count (rollup.go:1541-1790) */
static VM_DEV double vmf_avg(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double s = 0;
  for (int i = 0; i < r->n; i++) s += r->values[i];
  return s / (double)r->n;
}
static VM_DEV double vmf_min(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double m = r->values[0];
  for (int i = 0; i < r->n; i++)
    if (r->values[i] < m) m = r->values[i];
  return m;
}
static VM_DEV double vmf_max(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double m = r->values[0];
  for (int i = 0; i < r->n; i++)
    if (r->values[i] > m) m = r->values[i];
  return m;
}
static VM_DEV double vmf_sum(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double s = 0;
  for (int i = 0; i < r->n; i++) s += r->values[i];
  return s;
}
static VM_DEV double vmf_sum2(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double s = 0;
  for (int i = 0; i < r->n; i++) s += r->values[i] * r->values[i];
  return s;
}
static VM_DEV double vmf_count(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  return (double)r->n;
}
static VM_DEV double vmf_first(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  return r->values[0];
}
static VM_DEV double vmf_last(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  return r->values[r->n - 1];
}

/* stdvar/stddev (rollup.go:1799-1833, Welford) */
static VM_DEV double vmf_stdvar(const VmRfa* r) {
  int n = r->n;
  if (n == 0) return vm_dnan();
  if (n == 1) return 0;
  double avg = 0, count = 0, q = 0;
  for (int i = 0; i < n; i++) {
    double v = r->values[i];
    if (vm_isnan(v)) continue;
    count += 1;
    double avg_new = avg + (v - avg) / count;
    q += (v - avg) * (v - avg_new);
    avg = avg_new;
  }
  if (count == 0) return vm_dnan();
  return q / count;
}
static VM_DEV double vmf_stddev(const VmRfa* r) { return sqrt(vmf_stdvar(r)); }

/* rollupChanges (rollup.go:2106-2137) */
static VM_DEV double vmf_changes(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  double pv = r->prev_value;
  int cnt = 0;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    if (!vm_isnan(r->real_prev_value)) {
      pv = r->real_prev_value;
    } else {
      cnt++;
      pv = values[0];
      values++;
      n--;
    }
  }
  for (int i = 0; i < n; i++) {
    double v = values[i];
    if (v != pv) {
      if (fabs(v - pv) < 1e-12 * fabs(v)) continue;
      cnt++;
      pv = v;
    }
  }
  return (double)cnt;
}

/* rollupChangesPrometheus (rollup.go:2082-2104) */
static VM_DEV double vmf_changes_prometheus(const VmRfa* r) {
  if (r->n < 1) return vm_dnan();
  double pv = r->values[0];
  int cnt = 0;
  for (int i = 1; i < r->n; i++) {
    double v = r->values[i];
    if (v != pv) {
      if (fabs(v - pv) < 1e-12 * fabs(v)) continue;
      cnt++;
      pv = v;
    }
  }
  return (double)cnt;
}

/* rollupResets (rollup.go:2175-2204); also decreases_over_time */
static VM_DEV double vmf_resets(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  if (n == 0) {
    if (vm_isnan(r->prev_value)) return vm_dnan();
    return 0;
  }
  double pv = r->prev_value;
  if (vm_isnan(pv)) {
    pv = values[0];
    values++;
    n--;
  }
  if (n == 0) return 0;
  int cnt = 0;
  for (int i = 0; i < n; i++) {
    double v = values[i];
    if (v < pv) {
      if (fabs(v - pv) < 1e-12 * fabs(v)) continue;
      cnt++;
    }
    pv = v;
  }
  return (double)cnt;
}

/* rollupIncreases (rollup.go:2139-2169) */
static VM_DEV double vmf_increases(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  if (n == 0) {
    if (vm_isnan(r->prev_value)) return vm_dnan();
    return 0;
  }
  double pv = r->prev_value;
  if (vm_isnan(pv)) {
    pv = values[0];
    values++;
    n--;
  }
  if (n == 0) return 0;
  int cnt = 0;
  for (int i = 0; i < n; i++) {
    double v = values[i];
    if (v > pv) {
      if (fabs(v - pv) < 1e-12 * fabs(v)) continue;
      cnt++;
    }
    pv = v;
  }
  return (double)cnt;
}

/* rollupLag / rollupLifetime / rollupScrapeInterval (rollup.go:2040-2080) */
static VM_DEV double vmf_lag(const VmRfa* r) {
  if (r->n == 0) {
    if (vm_isnan(r->prev_value)) return vm_dnan();
    return (double)(r->curr_timestamp - r->prev_timestamp) / 1e3;
  }
  return (double)(r->curr_timestamp - r->timestamps[r->n - 1]) / 1e3;
}
static VM_DEV double vmf_lifetime(const VmRfa* r) {
  int n = r->n;
  if (vm_isnan(r->prev_value)) {
    if (n < 2) return vm_dnan();
    return (double)(r->timestamps[n - 1] - r->timestamps[0]) / 1e3;
  }
  if (n == 0) return vm_dnan();
  return (double)(r->timestamps[n - 1] - r->prev_timestamp) / 1e3;
}
static VM_DEV double vmf_scrape_interval(const VmRfa* r) {
  int n = r->n;
  if (vm_isnan(r->prev_value)) {
    if (n < 2) return vm_dnan();
    return ((double)(r->timestamps[n - 1] - r->timestamps[0]) / 1e3) / (double)(n - 1);
  }
  if (n == 0) return vm_dnan();
  return ((double)(r->timestamps[n - 1] - r->prev_timestamp) / 1e3) / (double)n;
}

/* rollupRateOverSum (rollup.go:1707-1719) */
static VM_DEV double vmf_rate_over_sum(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double s = 0;
  for (int i = 0; i < r->n; i++) s += r->values[i];
  return s / ((double)r->window / 1e3);
}

static VM_DEV double vmf_range(const VmRfa* r) { return vmf_max(r) - vmf_min(r); }

/* t-funcs (rollup.go:1603-1688) */
static VM_DEV double vmf_tfirst(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  return (double)r->timestamps[0] / 1e3;
}
static VM_DEV double vmf_tlast(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  return (double)r->timestamps[r->n - 1] / 1e3;
}
static VM_DEV double vmf_tmin(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double mv = r->values[0];
  int64_t mt = r->timestamps[0];
  for (int i = 0; i < r->n; i++) {
    if (r->values[i] <= mv) { mv = r->values[i]; mt = r->timestamps[i]; }
  }
  return (double)mt / 1e3;
}
static VM_DEV double vmf_tmax(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double mv = r->values[0];
  int64_t mt = r->timestamps[0];
  for (int i = 0; i < r->n; i++) {
    if (r->values[i] >= mv) { mv = r->values[i]; mt = r->timestamps[i]; }
  }
  return (double)mt / 1e3;
}
static VM_DEV double vmf_tlast_change(const VmRfa* r) {
  int n = r->n;
  if (n == 0) return vm_dnan();
  double last = r->values[n - 1];
  for (int i = n - 2; i >= 0; i--) {
    if (r->values[i] != last) return (double)r->timestamps[i + 1] / 1e3;
  }
  if (vm_isnan(r->prev_value) || r->prev_value != last)
    return (double)r->timestamps[0] / 1e3;
  return vm_dnan();
}

/* rollupGeomean / Present / Absent / StaleSamples (rollup.go:1741-1793) */
static VM_DEV double vmf_geomean(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  double p = 1.0;
  for (int i = 0; i < r->n; i++) p *= r->values[i];
  return pow(p, 1.0 / (double)r->n);
}
static VM_DEV double vmf_present(const VmRfa* r) { return r->n > 0 ? 1.0 : vm_dnan(); }
static VM_DEV double vmf_absent(const VmRfa* r) { return r->n == 0 ? 1.0 : vm_dnan(); }
static VM_DEV double vmf_stale_samples(const VmRfa* r) {
  if (r->n == 0) return vm_dnan();
  int c = 0;
  for (int i = 0; i < r->n; i++)
    if (vm_is_stale_nan(r->values[i])) c++;
  return (double)c;
}

/* filters (rollup.go:1185-1276) */
static VM_DEV double vmf_count_le(const VmRfa* r) {
  int c = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] <= r->arg) c++;
  return (double)c;
}
static VM_DEV double vmf_count_gt(const VmRfa* r) {
  int c = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] > r->arg) c++;
  return (double)c;
}
static VM_DEV double vmf_count_eq(const VmRfa* r) {
  int c = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] == r->arg) c++;
  return (double)c;
}
static VM_DEV double vmf_count_ne(const VmRfa* r) {
  int c = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] != r->arg) c++;
  return (double)c;
}
static VM_DEV double vmf_sum_le(const VmRfa* r) {
  double s = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] <= r->arg) s += r->values[i];
  return s;
}
static VM_DEV double vmf_sum_gt(const VmRfa* r) {
  double s = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] > r->arg) s += r->values[i];
  return s;
}
static VM_DEV double vmf_sum_eq(const VmRfa* r) {
  double s = 0;
  for (int i = 0; i < r->n; i++)
    if (r->values[i] == r->arg) s += r->values[i];
  return s;
}

/* linearRegression (rollup.go:1099-1136) */
static VM_DEV void vm_dev_linreg(const double* values, const int64_t* ts, int n,
                                 int64_t intercept_time, double* out_v, double* out_k) {
  if (n == 0) { *out_v = vm_dnan(); *out_k = vm_dnan(); return; }
  bool is_const = true;
  for (int i = 1; i < n; i++)
    if (values[i] != values[i - 1]) { is_const = false; break; }
  if (is_const) { *out_v = values[0]; *out_k = 0; return; }
  double v_sum = 0, t_sum = 0, tv_sum = 0, tt_sum = 0;
  int cnt = 0;
  for (int i = 0; i < n; i++) {
    double v = values[i];
    if (vm_isnan(v)) continue;
    double dt = (double)(ts[i] - intercept_time) / 1e3;
    v_sum += v;
    t_sum += dt;
    tv_sum += dt * v;
    tt_sum += dt * dt;
    cnt++;
  }
  if (cnt == 0) { *out_v = vm_dnan(); *out_k = vm_dnan(); return; }
  double k = 0;
  double t_diff = tt_sum - t_sum * t_sum / (double)cnt;
  if (fabs(t_diff) >= 1e-6) k = (tv_sum - t_sum * v_sum / (double)cnt) / t_diff;
  *out_v = v_sum / (double)cnt - k * t_sum / (double)cnt;
  *out_k = k;
}

static VM_DEV double vmf_deriv(const VmRfa* r) {
  double v, k;
  vm_dev_linreg(r->values, r->timestamps, r->n, r->curr_timestamp, &v, &k);
  return k;
}
static VM_DEV double vmf_predict_linear(const VmRfa* r) {
  double v, k;
  vm_dev_linreg(r->values, r->timestamps, r->n, r->curr_timestamp, &v, &k);
  if (vm_isnan(v)) return vm_dnan();
  return v + k * r->arg;
}

/* ascent/descent (rollup.go:2317-2357) */
static VM_DEV double vmf_ascent(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  double pv = r->prev_value;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    pv = values[0];
    values++;
    n--;
  }
  double s = 0;
  for (int i = 0; i < n; i++) {
    double d = values[i] - pv;
    if (d > 0) s += d;
    pv = values[i];
  }
  return s;
}
static VM_DEV double vmf_descent(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  double pv = r->prev_value;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    pv = values[0];
    values++;
    n--;
  }
  double s = 0;
  for (int i = 0; i < n; i++) {
    double d = pv - values[i];
    if (d > 0) s += d;
    pv = values[i];
  }
  return s;
}

/* zscore_over_time (rollup.go:2359-2373) */
static VM_DEV double vmf_zscore(const VmRfa* r) {
  double si = vmf_scrape_interval(r);
  double lag = vmf_lag(r);
  if (vm_isnan(si) || vm_isnan(lag) || lag > si) return vm_dnan();
  double d = vmf_last(r) - vmf_avg(r);
  if (d == 0) return 0;
  return d / vmf_stddev(r);
}

/* rollupIntegrate (rollup.go:2417-2450) */
static VM_DEV double vmf_integrate(const VmRfa* r) {
  const double* values = r->values;
  const int64_t* ts = r->timestamps;
  int n = r->n;
  double pv = r->prev_value;
  int64_t pt = r->curr_timestamp - r->window;
  if (vm_isnan(pv)) {
    if (n == 0) return vm_dnan();
    pv = values[0];
    pt = ts[0];
    values++;
    ts++;
    n--;
  }
  double s = 0;
  for (int i = 0; i < n; i++) {
    double dt = (double)(ts[i] - pt) / 1e3;
    s += pv * dt;
    pt = ts[i];
    pv = values[i];
  }
  if (!vm_isnan(r->real_next_value)) {
    double dt = (double)(r->curr_timestamp - pt) / 1e3;
    s += pv * dt;
  }
  return s;
}

/* rollupDistinct (rollup.go:2403-2415): first-occurrence count.  Go float64
 * map keys: +0==-0 collapse, each NaN distinct. */
static VM_DEV double vmf_distinct(const VmRfa* r) {
  int n = r->n;
  if (n == 0) return vm_dnan();
  int cnt = 0;
  for (int i = 0; i < n; i++) {
    double x = r->values[i];
    if (vm_isnan(x)) { cnt++; continue; }
    bool first = true;
    for (int j = 0; j < i; j++)
      if (r->values[j] == x) { first = false; break; }
    if (first) cnt++;
  }
  return (double)cnt;
}

/* rollupMAD (rollup.go:1469-1488) */
static VM_DEV double vmf_mad(const VmRfa* r) {
  double med = vm_dev_quantile(0.5, r->values, r->n);
  return vm_dev_quantile_absdev(0.5, r->values, r->n, med);
}

/* rollupModeOverTime (rollup.go:2293-2303) + modeNoNaNs (aggr.go:541-564):
 * the sorted-run scan re-expressed as a run iteration in ascending value
 * order (no sort needed; results identical). */
static VM_DEV double vmf_mode(const VmRfa* r) {
  int n = r->n;
  double prev = r->prev_value;
  if (n == 0) return prev;
  double mode = prev;
  int d_max = 0;
  int j = -1;
  int i = 0; /* running sorted index */
  /* iterate distinct values ascending */
  double cur = vm_dinf();
  for (int t = 0; t < n; t++)
    if (!vm_isnan(r->values[t]) && r->values[t] < cur) cur = r->values[t];
  /* NaNs cannot appear here on the storage path; they sort nowhere in Go's
   * sort.Float64s anyway, so ignore them like the values-are-clean contract */
  while (i < n) {
    int c = 0;
    for (int t = 0; t < n; t++)
      if (r->values[t] == cur) c++;
    if (c == 0) break;
    if (!(cur == prev)) {
      int d = i - j;
      if (d > d_max || vm_isnan(mode)) { d_max = d; mode = prev; }
      j = i;
      prev = cur;
    }
    i += c;
    /* next distinct value above cur */
    double nxt = vm_dinf();
    bool found = false;
    for (int t = 0; t < n; t++) {
      double x = r->values[t];
      if (!vm_isnan(x) && x > cur && (!found || x < nxt)) { nxt = x; found = true; }
    }
    if (!found) break;
    cur = nxt;
  }
  int d = n - j;
  if (d > d_max || vm_isnan(mode)) mode = prev;
  return mode;
}

/* duration_over_time (rollup.go:1151-1180); arg = dMax seconds */
static VM_DEV double vmf_duration(const VmRfa* r) {
  int n = r->n;
  if (n == 0) return vm_dnan();
  int64_t t_prev = r->timestamps[0];
  int64_t d_sum = 0;
  int64_t d_max = (int64_t)(r->arg * 1000);
  for (int i = 0; i < n; i++) {
    int64_t d = r->timestamps[i] - t_prev;
    if (d <= d_max) d_sum += d;
    t_prev = r->timestamps[i];
  }
  return (double)d_sum / 1000;
}

/* rollupOutlierIQR (rollup.go:1427-1448) */
static VM_DEV double vmf_outlier_iqr(const VmRfa* r) {
  int n = r->n;
  if (n < 2) return vm_dnan();
  double q25 = vm_dev_quantile(0.25, r->values, n);
  double q75 = vm_dev_quantile(0.75, r->values, n);
  double iqr = 1.5 * (q75 - q25);
  double v = r->values[n - 1];
  if (v > q75 + iqr || v < q25 - iqr) return v;
  return vm_dnan();
}

/* candlestick family (rollup.go:2206-2291): window excludes currTimestamp;
 * the pre-window sample opens the candle when within the window. */
static VM_DEV int vmf_candlestick_len(const VmRfa* r) {
  int n = r->n;
  while (n > 0 && r->timestamps[n - 1] >= r->curr_timestamp) n--;
  return n;
}
static VM_DEV double vmf_candlestick_first(const VmRfa* r) {
  if (r->prev_timestamp + r->window >= r->curr_timestamp) return r->prev_value;
  return vm_dnan();
}
static VM_DEV double vmf_open(const VmRfa* r) {
  double v = vmf_candlestick_first(r);
  if (!vm_isnan(v)) return v;
  int n = vmf_candlestick_len(r);
  if (n == 0) return vm_dnan();
  return r->values[0];
}
static VM_DEV double vmf_close(const VmRfa* r) {
  int n = vmf_candlestick_len(r);
  if (n == 0) return vmf_candlestick_first(r);
  return r->values[n - 1];
}
static VM_DEV double vmf_high(const VmRfa* r) {
  int n = vmf_candlestick_len(r);
  double m = vmf_candlestick_first(r);
  int i = 0;
  if (vm_isnan(m)) {
    if (n == 0) return vm_dnan();
    m = r->values[0];
    i = 1;
  }
  for (; i < n; i++)
    if (r->values[i] > m) m = r->values[i];
  return m;
}
static VM_DEV double vmf_low(const VmRfa* r) {
  int n = vmf_candlestick_len(r);
  double m = vmf_candlestick_first(r);
  int i = 0;
  if (vm_isnan(m)) {
    if (n == 0) return vm_dnan();
    m = r->values[0];
    i = 1;
  }
  for (; i < n; i++)
    if (r->values[i] < m) m = r->values[i];
  return m;
}

/* holt_winters (rollup.go:1030-1077); arg=sf, arg2=tf */
static VM_DEV double vmf_holt_winters(const VmRfa* r) {
  const double* values = r->values;
  int n = r->n;
  if (n == 0) return vm_dnan();
  double sf = r->arg;
  if (sf < 0 || sf > 1) return vm_dnan();
  double tf = r->arg2;
  if (tf < 0 || tf > 1) return vm_dnan();
  double s0 = r->prev_value;
  if (vm_isnan(s0)) {
    s0 = values[0];
    values++;
    n--;
    if (n == 0) return s0;
  }
  double b0 = values[0] - s0;
  for (int i = 0; i < n; i++) {
    double v = values[i];
    double s1 = sf * v + (1 - sf) * (s0 + b0);
    double b1 = tf * (s1 - s0) + (1 - tf) * b0;
    s0 = s1;
    b0 = b1;
  }
  return s0;
}

/* hoeffding bounds (rollup.go:1323-1381); arg=phi */
static VM_DEV double vmf_hoeffding(const VmRfa* r, int upper) {
  int n = r->n;
  if (n == 0) return vm_dnan();
  if (n == 1) return r->values[0];
  double v_max = vmf_max(r);
  double v_min = vmf_min(r);
  double v_avg = vmf_avg(r);
  double v_range = v_max - v_min;
  double bound;
  if (v_range <= 0) bound = 0;
  else if (r->arg >= 1) bound = vm_dinf();
  else if (r->arg <= 0) bound = 0;
  else bound = v_range * sqrt(log(1 / (1 - r->arg)) / (2 * (double)n));
  return upper ? v_avg + bound : v_avg - bound;
}

static VM_DEV double vm_eval_rollup_fn(int32_t func, const VmRfa* r) {
  switch (func) {
    case VMF_RATE: case VMF_DERIV_FAST: return vmf_deriv_fast(r);
    case VMF_INCREASE: case VMF_DELTA: return vmf_delta(r);
    case VMF_INCREASE_PURE: return vmf_increase_pure(r);
    case VMF_DELTA_PROMETHEUS: return vmf_delta_prometheus(r);
    case VMF_RATE_PROMETHEUS: return vmf_rate_prometheus(r);
    case VMF_IRATE: case VMF_IDERIV: return vmf_ideriv(r);
    case VMF_IDELTA: return vmf_idelta(r);
    case VMF_AVG: return vmf_avg(r);
    case VMF_MIN: return vmf_min(r);
    case VMF_MAX: return vmf_max(r);
    case VMF_SUM: return vmf_sum(r);
    case VMF_SUM2: return vmf_sum2(r);
    case VMF_COUNT: return vmf_count(r);
    case VMF_FIRST: return vmf_first(r);
    case VMF_LAST: case VMF_DEFAULT_ROLLUP: return vmf_last(r);
    case VMF_QUANTILE: return vm_dev_quantile(r->arg, r->values, r->n);
    case VMF_MEDIAN: return vm_dev_quantile(0.5, r->values, r->n);
    case VMF_STDDEV: return vmf_stddev(r);
    case VMF_STDVAR: return vmf_stdvar(r);
    case VMF_CHANGES: return vmf_changes(r);
    case VMF_CHANGES_PROMETHEUS: return vmf_changes_prometheus(r);
    case VMF_RESETS: case VMF_DECREASES: return vmf_resets(r);
    case VMF_LAG: return vmf_lag(r);
    case VMF_LIFETIME: return vmf_lifetime(r);
    case VMF_SCRAPE_INTERVAL: return vmf_scrape_interval(r);
    case VMF_RATE_OVER_SUM: return vmf_rate_over_sum(r);
    case VMF_RANGE: return vmf_range(r);
    case VMF_TFIRST: return vmf_tfirst(r);
    case VMF_TLAST: return vmf_tlast(r);
    case VMF_TMIN: return vmf_tmin(r);
    case VMF_TMAX: return vmf_tmax(r);
    case VMF_TLAST_CHANGE: return vmf_tlast_change(r);
    case VMF_GEOMEAN: return vmf_geomean(r);
    case VMF_PRESENT: return vmf_present(r);
    case VMF_ABSENT: return vmf_absent(r);
    case VMF_STALE_SAMPLES: return vmf_stale_samples(r);
    case VMF_COUNT_LE: return vmf_count_le(r);
    case VMF_COUNT_GT: return vmf_count_gt(r);
    case VMF_COUNT_EQ: return vmf_count_eq(r);
    case VMF_COUNT_NE: return vmf_count_ne(r);
    case VMF_SHARE_LE: return vmf_count_le(r) / (double)r->n;
    case VMF_SHARE_GT: return vmf_count_gt(r) / (double)r->n;
    case VMF_SHARE_EQ: return vmf_count_eq(r) / (double)r->n;
    case VMF_SUM_LE: return vmf_sum_le(r);
    case VMF_SUM_GT: return vmf_sum_gt(r);
    case VMF_SUM_EQ: return vmf_sum_eq(r);
    case VMF_DERIV: return vmf_deriv(r);
    case VMF_PREDICT_LINEAR: return vmf_predict_linear(r);
    case VMF_ASCENT: return vmf_ascent(r);
    case VMF_DESCENT: return vmf_descent(r);
    case VMF_ZSCORE: return vmf_zscore(r);
    case VMF_INTEGRATE: return vmf_integrate(r);
    case VMF_DISTINCT: return vmf_distinct(r);
    case VMF_INCREASES: return vmf_increases(r);
    case VMF_MAD: return vmf_mad(r);
    case VMF_MODE: return vmf_mode(r);
    case VMF_DURATION: return vmf_duration(r);
    case VMF_OUTLIER_IQR: return vmf_outlier_iqr(r);
    case VMF_OPEN: return vmf_open(r);
    case VMF_CLOSE: return vmf_close(r);
    case VMF_LOW: return vmf_low(r);
    case VMF_HIGH: return vmf_high(r);
    case VMF_HOLT_WINTERS: return vmf_holt_winters(r);
    case VMF_HOEFFDING_LOWER: return vmf_hoeffding(r, 0);
    case VMF_HOEFFDING_UPPER: return vmf_hoeffding(r, 1);
    default: return vm_dnan();
  }
}
