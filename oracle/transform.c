/* oracle/transform.c — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
 *
 * CPU restatement of the transform-function value math
 * (app/vmselect/promql/transform.go) used to pin the GPU transform kernels:
 * one-arg math funcs (:26-131), clamp family (:271), round (:2340), sgn
 * (:2374), bitmap (:2745), UTC date-time funcs (:333), and the per-series
 * walks (keep_last, keep_next, interpolate, running, range,
 * smooth_exponential, remove_resets; :1232-1731, :2929).  Func ids match
 * include/vmgpu.h VMGPU_TF_* so tests drive both sides with one table.
 */
#include <math.h>
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

#ifndef M_PI
#define M_PI 3.14159265358979323846
#endif

#define TF_ABS 0
#define TF_CEIL 1
#define TF_FLOOR 2
#define TF_EXP 3
#define TF_LN 4
#define TF_LOG2 5
#define TF_LOG10 6
#define TF_SQRT 7
#define TF_SIN 8
#define TF_COS 9
#define TF_TAN 10
#define TF_ASIN 11
#define TF_ACOS 12
#define TF_ATAN 13
#define TF_SINH 14
#define TF_COSH 15
#define TF_TANH 16
#define TF_ASINH 17
#define TF_ACOSH 18
#define TF_ATANH 19
#define TF_DEG 20
#define TF_RAD 21
#define TF_SGN 22
#define TF_CLAMP 23
#define TF_CLAMP_MIN 24
#define TF_CLAMP_MAX 25
#define TF_ROUND 26
#define TF_BITMAP_AND 27
#define TF_BITMAP_OR 28
#define TF_BITMAP_XOR 29
#define TF_DAY_OF_MONTH 30
#define TF_DAY_OF_WEEK 31
#define TF_DAY_OF_YEAR 32
#define TF_DAYS_IN_MONTH 33
#define TF_HOUR 34
#define TF_MINUTE 35
#define TF_MONTH 36
#define TF_YEAR 37
#define TF_SERIES_BASE 100
#define TF_KEEP_LAST_VALUE 100
#define TF_KEEP_NEXT_VALUE 101
#define TF_INTERPOLATE 102
#define TF_RUNNING_SUM 103
#define TF_RUNNING_MIN 104
#define TF_RUNNING_MAX 105
#define TF_RUNNING_AVG 106
#define TF_RANGE_SUM 107
#define TF_RANGE_MIN 108
#define TF_RANGE_MAX 109
#define TF_RANGE_AVG 110
#define TF_RANGE_FIRST 111
#define TF_RANGE_LAST 112
#define TF_RANGE_NORMALIZE 113
#define TF_RANGE_ZSCORE 114
#define TF_RANGE_TRIM_ZSCORE 115
#define TF_RANGE_STDDEV 116
#define TF_RANGE_STDVAR 117
#define TF_RANGE_LINREG 118
#define TF_RANGE_MAD 119
#define TF_RANGE_TRIM_OUTLIERS 120
#define TF_RANGE_TRIM_SPIKES 121
#define TF_RANGE_QUANTILE 122
#define TF_SMOOTH_EXPONENTIAL 123
#define TF_REMOVE_RESETS 124

static double tf_nan(void) { return nan(""); }

/* Go int64(v)/uint64(v) conversions, amd64 semantics */
static long long tf_go_i64(double v) {
  if (isnan(v) || v >= 9.223372036854775808e18 || v < -9.223372036854775808e18)
    return (long long)0x8000000000000000LL;
  return (long long)v;
}

static unsigned long long tf_go_u64(double v) {
  if (v < 9.223372036854775808e18) return (unsigned long long)tf_go_i64(v);
  return (unsigned long long)tf_go_i64(v - 9.223372036854775808e18) +
         0x8000000000000000ULL;
}

typedef struct { int y, m, d, yd; } TfCivil;

static TfCivil tf_civil_from_unix(long long sec, long long* rem) {
  long long days = sec / 86400;
  long long r = sec % 86400;
  if (r < 0) { r += 86400; days -= 1; }
  *rem = r;
  long long z = days + 719468;
  long long era = (z >= 0 ? z : z - 146096) / 146097;
  unsigned doe = (unsigned)(z - era * 146097);
  unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  long long y = (long long)yoe + era * 400;
  unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  unsigned mp = (5 * doy + 2) / 153;
  unsigned d = doy - (153 * mp + 2) / 5 + 1;
  unsigned m = mp < 10 ? mp + 3 : mp - 9;
  TfCivil c;
  c.y = (int)(y + (m <= 2));
  c.m = (int)m;
  c.d = (int)d;
  int leap = (c.y % 4 == 0 && (c.y % 100 != 0 || c.y % 400 == 0)) ? 1 : 0;
  static const int cum[13] = {0, 0, 31, 59, 90, 120, 151, 181, 212, 243,
                              273, 304, 334};
  c.yd = cum[c.m] + (c.m > 2 ? leap : 0) + c.d;
  return c;
}

static int tf_days_in_month(int y, int m) {
  static const int dm[13] = {0, 31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31};
  int leap = (y % 4 == 0 && (y % 100 != 0 || y % 400 == 0)) ? 1 : 0;
  return (m == 2) ? 28 + leap : dm[m];
}

double vm_tf_elementwise(int32_t f, double v, double a1, double a2) {
  switch (f) {
    case TF_ABS:   return fabs(v);
    case TF_CEIL:  return ceil(v);
    case TF_FLOOR: return floor(v);
    case TF_EXP:   return exp(v);
    case TF_LN:    return log(v);
    case TF_LOG2:  return log2(v);
    case TF_LOG10: return log10(v);
    case TF_SQRT:  return sqrt(v);
    case TF_SIN:   return sin(v);
    case TF_COS:   return cos(v);
    case TF_TAN:   return tan(v);
    case TF_ASIN:  return asin(v);
    case TF_ACOS:  return acos(v);
    case TF_ATAN:  return atan(v);
    case TF_SINH:  return sinh(v);
    case TF_COSH:  return cosh(v);
    case TF_TANH:  return tanh(v);
    case TF_ASINH: return asinh(v);
    case TF_ACOSH: return acosh(v);
    case TF_ATANH: return atanh(v);
    case TF_DEG:   return v * 180.0 / M_PI;
    case TF_RAD:   return v * M_PI / 180.0;
    case TF_SGN:   return (v < 0) ? -1.0 : (v > 0 ? 1.0 : 0.0);
    case TF_CLAMP:
      if (v < a1) v = a1;
      if (v > a2) v = a2;
      return v;
    case TF_CLAMP_MIN: return (v < a1) ? a1 : v;
    case TF_CLAMP_MAX: return (v > a1) ? a1 : v;
    case TF_ROUND: {
      v += 0.5 * copysign(a1, v);
      v -= fmod(v, a1);
      double ip;
      (void)modf(v * a2, &ip);
      return ip / a2;
    }
    case TF_BITMAP_AND:
      return (isnan(v) || isnan(a1)) ? tf_nan()
             : (double)(tf_go_u64(v) & tf_go_u64(a1));
    case TF_BITMAP_OR:
      return (isnan(v) || isnan(a1)) ? tf_nan()
             : (double)(tf_go_u64(v) | tf_go_u64(a1));
    case TF_BITMAP_XOR:
      return (isnan(v) || isnan(a1)) ? tf_nan()
             : (double)(tf_go_u64(v) ^ tf_go_u64(a1));
    default: break;
  }
  if (isnan(v)) return v;
  long long rem;
  TfCivil c = tf_civil_from_unix(tf_go_i64(v), &rem);
  switch (f) {
    case TF_DAY_OF_MONTH:  return (double)c.d;
    case TF_DAY_OF_WEEK: {
      long long days = tf_go_i64(v) / 86400;
      if (tf_go_i64(v) % 86400 < 0) days -= 1;
      long long w = (days + 4) % 7;
      if (w < 0) w += 7;
      return (double)w;
    }
    case TF_DAY_OF_YEAR:   return (double)c.yd;
    case TF_DAYS_IN_MONTH: return (double)tf_days_in_month(c.y, c.m);
    case TF_HOUR:          return (double)(rem / 3600);
    case TF_MINUTE:        return (double)((rem % 3600) / 60);
    case TF_MONTH:         return (double)c.m;
    case TF_YEAR:          return (double)c.y;
    default:               return tf_nan();
  }
}

static double tf_quantile_sorted(double phi, const double* a, int n) {
  if (n == 0 || isnan(phi)) return tf_nan();
  if (phi < 0) return -INFINITY;
  if (phi > 1) return INFINITY;
  double rank = phi * (double)(n - 1);
  double lower_idx = fmax(0.0, floor(rank));
  double upper_idx = fmin((double)(n - 1), lower_idx + 1.0);
  double weight = rank - floor(rank);
  return a[(int)lower_idx] * (1.0 - weight) + a[(int)upper_idx] * weight;
}

static int tf_sorted_nonnan(const double* row, int64_t n, double* scratch) {
  int cnt = 0;
  for (int64_t i = 0; i < n; i++) {
    double v = row[i];
    if (isnan(v)) continue;
    int j = cnt;
    while (j > 0 && scratch[j - 1] > v) {
      scratch[j] = scratch[j - 1];
      j--;
    }
    scratch[j] = v;
    cnt++;
  }
  return cnt;
}

static double tf_mean(const double* row, int64_t n) {
  double sum = 0.0;
  int64_t cnt = 0;
  for (int64_t i = 0; i < n; i++)
    if (!isnan(row[i])) { sum += row[i]; cnt++; }
  return sum / (double)cnt;
}

static double tf_stdvar(const double* row, int64_t n) {
  double avg = 0.0, count = 0.0, q = 0.0;
  for (int64_t i = 0; i < n; i++) {
    double v = row[i];
    if (isnan(v)) continue;
    count++;
    double avg_new = avg + (v - avg) / count;
    q += (v - avg) * (v - avg_new);
    avg = avg_new;
  }
  if (count == 0.0) return tf_nan();
  return q / count;
}

/* one series row, in place; returns keep flag (0 = drop series) */
int32_t vm_tf_series(int32_t func, double* row, int64_t n, const int64_t* ts,
                     const double* arg1, double scalar_arg) {
  if (func == TF_KEEP_LAST_VALUE) {
    if (n == 0) return 1;
    double last = row[0];
    for (int64_t i = 0; i < n; i++) {
      if (!isnan(row[i])) last = row[i];
      else row[i] = last;
    }
    return 1;
  }
  if (func == TF_KEEP_NEXT_VALUE) {
    if (n == 0) return 1;
    double next = row[n - 1];
    for (int64_t i = n - 1; i >= 0; i--) {
      if (!isnan(row[i])) next = row[i];
      else row[i] = next;
    }
    return 1;
  }
  if (func == TF_INTERPOLATE) {
    int64_t lo = 0, hi = n;
    while (lo < hi && isnan(row[lo])) lo++;
    while (hi > lo && isnan(row[hi - 1])) hi--;
    double prev = tf_nan();
    for (int64_t i = lo; i < hi; i++) {
      if (!isnan(row[i])) continue;
      if (i > lo) prev = row[i - 1];
      int64_t j = i + 1;
      while (j < hi && isnan(row[j])) j++;
      double next = (j >= hi) ? prev : row[j];
      if (isnan(prev)) prev = next;
      double delta = (next - prev) / (double)(j - i + 1);
      while (i < j) {
        prev += delta;
        row[i] = prev;
        i++;
      }
    }
    return 1;
  }
  if (func >= TF_RUNNING_SUM && func <= TF_RANGE_AVG) {
    int64_t lo = 0;
    while (lo < n && isnan(row[lo])) lo++;
    if (lo >= n) return 1;
    double prev = row[lo];
    for (int64_t i = lo + 1; i < n; i++) {
      double v = row[i];
      int idx = (int)(i - lo);
      if (!isnan(v)) {
        switch (func) {
          case TF_RUNNING_SUM: case TF_RANGE_SUM: prev = prev + v; break;
          case TF_RUNNING_MIN: case TF_RANGE_MIN:
            prev = (prev < v) ? prev : v; break;
          case TF_RUNNING_MAX: case TF_RANGE_MAX:
            prev = (prev > v) ? prev : v; break;
          default: prev = prev + (v - prev) / (double)(idx + 1); break;
        }
      }
      row[i] = prev;
    }
    if (func >= TF_RANGE_SUM) {
      int64_t hi = n;
      while (hi > 0 && isnan(row[hi - 1])) hi--;
      if (hi == 0) return 1;
      double last = row[hi - 1];
      for (int64_t i = 0; i < n; i++) row[i] = last;
    }
    return 1;
  }
  if (func == TF_RANGE_FIRST) {
    int64_t lo = 0;
    while (lo < n && isnan(row[lo])) lo++;
    if (lo >= n) return 1;
    double first = row[lo];
    for (int64_t i = 0; i < n; i++) row[i] = first;
    return 1;
  }
  if (func == TF_RANGE_LAST) {
    int64_t hi = n;
    while (hi > 0 && isnan(row[hi - 1])) hi--;
    if (hi == 0) return 1;
    double last = row[hi - 1];
    for (int64_t i = 0; i < n; i++) row[i] = last;
    return 1;
  }
  if (func == TF_RANGE_NORMALIZE) {
    double vmin = INFINITY, vmax = -INFINITY;
    for (int64_t i = 0; i < n; i++) {
      double v = row[i];
      if (isnan(v)) continue;
      if (v < vmin) vmin = v;
      if (v > vmax) vmax = v;
    }
    double d = vmax - vmin;
    if (isinf(d)) return 0;
    for (int64_t i = 0; i < n; i++) row[i] = (row[i] - vmin) / d;
    return 1;
  }
  if (func == TF_RANGE_ZSCORE || func == TF_RANGE_TRIM_ZSCORE) {
    double sd = sqrt(tf_stdvar(row, n));
    double avg = tf_mean(row, n);
    if (func == TF_RANGE_ZSCORE) {
      for (int64_t i = 0; i < n; i++) row[i] = (row[i] - avg) / sd;
    } else {
      double z = fabs(scalar_arg);
      for (int64_t i = 0; i < n; i++)
        if (fabs(row[i] - avg) / sd > z) row[i] = tf_nan();
    }
    return 1;
  }
  if (func == TF_RANGE_STDDEV || func == TF_RANGE_STDVAR) {
    double v = tf_stdvar(row, n);
    if (func == TF_RANGE_STDDEV) v = sqrt(v);
    for (int64_t i = 0; i < n; i++) row[i] = v;
    return 1;
  }
  if (func == TF_RANGE_LINREG) {
    if (n == 0) return 1;
    int64_t t0 = ts[0];
    int all_const = 1;
    for (int64_t i = 1; i < n && all_const; i++)
      all_const = (row[i] == row[0]);
    double v0, k;
    if (all_const) {
      v0 = row[0];
      k = 0.0;
    } else {
      double vsum = 0, tsum = 0, tvsum = 0, ttsum = 0, cnt = 0;
      for (int64_t i = 0; i < n; i++) {
        double v = row[i];
        if (isnan(v)) continue;
        double dt = (double)(ts[i] - t0) / 1e3;
        cnt++;
        vsum += v;
        tsum += dt;
        tvsum += dt * v;
        ttsum += dt * dt;
      }
      if (cnt == 0) {
        v0 = tf_nan();
        k = tf_nan();
      } else {
        k = 0.0;
        double tdiff = ttsum - tsum * tsum / cnt;
        if (fabs(tdiff) >= 1e-6) k = (tvsum - tsum * vsum / cnt) / tdiff;
        v0 = vsum / cnt - k * tsum / cnt;
      }
    }
    for (int64_t i = 0; i < n; i++)
      row[i] = v0 + k * (double)(ts[i] - t0) / 1e3;
    return 1;
  }
  if (func == TF_RANGE_MAD || func == TF_RANGE_TRIM_OUTLIERS) {
    double* sc = (double*)malloc((size_t)(n > 0 ? n : 1) * 8);
    int cnt = tf_sorted_nonnan(row, n, sc);
    double med = tf_quantile_sorted(0.5, sc, cnt);
    for (int i = 0; i < cnt; i++) sc[i] = fabs(sc[i] - med);
    for (int i = 1; i < cnt; i++) {
      double v = sc[i];
      int j = i;
      while (j > 0 && sc[j - 1] > v) { sc[j] = sc[j - 1]; j--; }
      sc[j] = v;
    }
    double madv = tf_quantile_sorted(0.5, sc, cnt);
    if (func == TF_RANGE_MAD) {
      for (int64_t i = 0; i < n; i++) row[i] = madv;
    } else {
      double dmax = scalar_arg * madv;
      for (int64_t i = 0; i < n; i++)
        if (fabs(row[i] - med) > dmax) row[i] = tf_nan();
    }
    free(sc);
    return 1;
  }
  if (func == TF_RANGE_TRIM_SPIKES) {
    double* sc = (double*)malloc((size_t)(n > 0 ? n : 1) * 8);
    int cnt = tf_sorted_nonnan(row, n, sc);
    double phi = scalar_arg / 2.0;
    double vmax = tf_quantile_sorted(1.0 - phi, sc, cnt);
    double vmin = tf_quantile_sorted(phi, sc, cnt);
    for (int64_t i = 0; i < n; i++) {
      double v = row[i];
      if (isnan(v)) continue;
      if (v > vmax || v < vmin) row[i] = tf_nan();
    }
    free(sc);
    return 1;
  }
  if (func == TF_RANGE_QUANTILE) {
    double* sc = (double*)malloc((size_t)(n > 0 ? n : 1) * 8);
    int cnt = 0;
    int64_t last_idx = -1;
    for (int64_t i = 0; i < n; i++) {
      double v = row[i];
      if (isnan(v)) continue;
      int j = cnt;
      while (j > 0 && sc[j - 1] > v) { sc[j] = sc[j - 1]; j--; }
      sc[j] = v;
      cnt++;
      last_idx = i;
    }
    if (last_idx >= 0) row[last_idx] = tf_quantile_sorted(scalar_arg, sc, cnt);
    free(sc);
    int64_t hi = n;
    while (hi > 0 && isnan(row[hi - 1])) hi--;
    if (hi == 0) return 1;
    double last = row[hi - 1];
    for (int64_t i = 0; i < n; i++) row[i] = last;
    return 1;
  }
  if (func == TF_SMOOTH_EXPONENTIAL) {
    int64_t lo = 0;
    while (lo < n && isnan(row[lo])) lo++;
    while (lo < n && isinf(row[lo])) lo++;
    if (lo >= n) return 1;
    double avg = row[lo];
    for (int64_t i = lo + 1; i < n; i++) {
      double v = row[i];
      if (isnan(v)) continue;
      if (isinf(v)) { row[i] = avg; continue; }
      double sf = arg1 ? arg1[i] : 1.0;
      if (isnan(sf)) sf = 1.0;
      if (sf < 0.0) sf = 0.0;
      if (sf > 1.0) sf = 1.0;
      avg = avg * (1.0 - sf) + v * sf;
      row[i] = avg;
    }
    return 1;
  }
  if (func == TF_REMOVE_RESETS) {
    int64_t lo = 0;
    while (lo < n && isnan(row[lo])) lo++;
    if (lo >= n) return 1;
    double corr = 0.0;
    double prev = row[lo];
    for (int64_t i = lo; i < n; i++) {
      double v = row[i];
      if (isnan(v)) continue;
      double d = v - prev;
      if (d < 0.0) {
        if ((-d * 8.0) < prev) corr += prev - v;
        else corr += prev;
      }
      prev = v;
      row[i] = v + corr;
    }
    return 1;
  }
  return 1;
}

void vm_tf_apply(int32_t func, double* values, int64_t n_series,
                 int64_t n_grid, const int64_t* ts, const double* arg1,
                 const double* arg2, double scalar_arg, uint8_t* keep) {
  if (func < TF_SERIES_BASE) {
    for (int64_t s = 0; s < n_series; s++)
      for (int64_t g = 0; g < n_grid; g++)
        values[s * n_grid + g] = vm_tf_elementwise(
            func, values[s * n_grid + g], arg1 ? arg1[g] : 0.0,
            arg2 ? arg2[g] : 0.0);
    if (keep) memset(keep, 1, (size_t)n_series);
    return;
  }
  for (int64_t s = 0; s < n_series; s++) {
    int32_t k = vm_tf_series(func, values + s * n_grid, n_grid, ts, arg1,
                             scalar_arg);
    if (keep) keep[s] = (uint8_t)k;
  }
}
