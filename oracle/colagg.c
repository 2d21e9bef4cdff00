/* oracle/colagg.c — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
 *
 * CPU restatement of the non-incremental cross-series aggregates
 * (app/vmselect/promql/aggr.go): per-point median/quantile (quantile() +
 * quantileSorted, aggr.go:900-940), mad (getPerPointMedians/MADs :985),
 * stddev/stdvar (:352,:371 Welford), mode (modeNoNaNs :541), distinct
 * (:423), share (:462), zscore (:493), iqr bounds (getPerPointIQRBounds
 * :975) and the outlier filters (:952,:1022).  Op ids match
 * include/vmgpu.h VMGPU_COLAGG_*.
 */
#include <math.h>
#include <stdint.h>
#include <stdlib.h>

static double ca_nan(void) { return nan(""); }

static double ca_quantile_sorted(double phi, const double* a, int n) {
  if (n == 0 || isnan(phi)) return ca_nan();
  if (phi < 0) return -INFINITY;
  if (phi > 1) return INFINITY;
  double rank = phi * (double)(n - 1);
  double lower_idx = fmax(0.0, floor(rank));
  double upper_idx = fmin((double)(n - 1), lower_idx + 1.0);
  double weight = rank - floor(rank);
  return a[(int)lower_idx] * (1.0 - weight) + a[(int)upper_idx] * weight;
}

static int ca_sorted_col(const double* values, uint32_t n_grid,
                         const uint32_t* rows, uint32_t lo, uint32_t hi,
                         uint32_t g, double* sc) {
  int cnt = 0;
  for (uint32_t k = lo; k < hi; k++) {
    double v = values[(size_t)rows[k] * n_grid + g];
    if (isnan(v)) continue;
    int j = cnt;
    while (j > 0 && sc[j - 1] > v) {
      sc[j] = sc[j - 1];
      j--;
    }
    sc[j] = v;
    cnt++;
  }
  return cnt;
}

void vm_colagg(int32_t op, const double* values, uint32_t n_series,
               uint32_t n_grid, const uint32_t* group_rows,
               const uint64_t* group_offsets, uint32_t n_groups, double phi,
               double* out, double* out2, double* values_out) {
  uint32_t max_members = 0;
  for (uint32_t grp = 0; grp < n_groups; grp++) {
    uint32_t m = (uint32_t)(group_offsets[grp + 1] - group_offsets[grp]);
    if (m > max_members) max_members = m;
  }
  double* sc = (double*)malloc((size_t)(max_members ? max_members : 1) * 8);
  for (uint32_t grp = 0; grp < n_groups; grp++) {
    uint32_t lo = (uint32_t)group_offsets[grp];
    uint32_t hi = (uint32_t)group_offsets[grp + 1];
    for (uint32_t g = 0; g < n_grid; g++) {
      size_t e = (size_t)grp * n_grid + g;
      switch (op) {
        case 0:   /* median */
        case 1: { /* quantile */
          int cnt = ca_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
          out[e] = ca_quantile_sorted(op == 0 ? 0.5 : phi, sc, cnt);
          break;
        }
        case 2: { /* mad */
          int cnt = ca_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
          double med = ca_quantile_sorted(0.5, sc, cnt);
          for (int i = 0; i < cnt; i++) sc[i] = fabs(sc[i] - med);
          for (int i = 1; i < cnt; i++) {
            double v = sc[i];
            int j = i;
            while (j > 0 && sc[j - 1] > v) { sc[j] = sc[j - 1]; j--; }
            sc[j] = v;
          }
          out[e] = ca_quantile_sorted(0.5, sc, cnt);
          break;
        }
        case 3:   /* stddev */
        case 4: { /* stdvar */
          double avg = 0, count = 0, q = 0;
          for (uint32_t k = lo; k < hi; k++) {
            double v = values[(size_t)group_rows[k] * n_grid + g];
            if (isnan(v)) continue;
            count++;
            double avg_new = avg + (v - avg) / count;
            q += (v - avg) * (v - avg_new);
            avg = avg_new;
          }
          if (count == 0) q = ca_nan();
          double r = q / count;
          out[e] = (op == 3) ? sqrt(r) : r;
          break;
        }
        case 5: { /* mode */
          int cnt = ca_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
          double prev = ca_nan();
          double mode = ca_nan();
          if (cnt > 0) {
            int j = -1;
            int dmax = 0;
            for (int i = 0; i < cnt; i++) {
              double v = sc[i];
              if (prev == v) continue;
              int d = i - j;
              if (d > dmax || isnan(mode)) {
                dmax = d;
                mode = prev;
              }
              j = i;
              prev = v;
            }
            int d = cnt - j;
            if (d > dmax || isnan(mode)) mode = prev;
          }
          out[e] = mode;
          break;
        }
        case 6: { /* distinct */
          int cnt = ca_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
          int n = 0;
          for (int i = 0; i < cnt; i++)
            if (i == 0 || sc[i] != sc[i - 1]) n++;
          out[e] = (n == 0) ? ca_nan() : (double)n;
          break;
        }
        case 7: { /* share */
          double sum = 0;
          for (uint32_t k = lo; k < hi; k++) {
            double v = values[(size_t)group_rows[k] * n_grid + g];
            if (isnan(v) || v < 0) continue;
            sum += v;
          }
          for (uint32_t k = lo; k < hi; k++) {
            size_t idx = (size_t)group_rows[k] * n_grid + g;
            double v = values[idx];
            values_out[idx] = (isnan(v) || v < 0) ? ca_nan() : v / sum;
          }
          break;
        }
        case 8: { /* zscore */
          double avg = 0, count = 0, q = 0;
          for (uint32_t k = lo; k < hi; k++) {
            double v = values[(size_t)group_rows[k] * n_grid + g];
            if (isnan(v)) continue;
            count++;
            double avg_new = avg + (v - avg) / count;
            q += (v - avg) * (v - avg_new);
            avg = avg_new;
          }
          if (count == 0) {
            for (uint32_t k = lo; k < hi; k++) {
              size_t idx = (size_t)group_rows[k] * n_grid + g;
              values_out[idx] = values[idx];
            }
            break;
          }
          double sd = sqrt(q / count);
          for (uint32_t k = lo; k < hi; k++) {
            size_t idx = (size_t)group_rows[k] * n_grid + g;
            double v = values[idx];
            values_out[idx] = isnan(v) ? v : (v - avg) / sd;
          }
          break;
        }
        case 10: case 11: case 12: case 13:
        case 14: case 15: case 16: case 17: { /* sum..group */
          double sum = 0, prod = 1, mn = ca_nan(), mx = ca_nan();
          double cnt = 0;
          for (uint32_t k = lo; k < hi; k++) {
            double v = values[(size_t)group_rows[k] * n_grid + g];
            if (isnan(v)) continue;
            cnt++;
            sum += (op == 15) ? v * v : v;
            prod *= v;
            if (isnan(mn) || v < mn) mn = v;
            if (isnan(mx) || v > mx) mx = v;
          }
          double r;
          switch (op) {
            case 10: case 15: r = (cnt == 0) ? ca_nan() : sum; break;
            case 11: r = mn; break;
            case 12: r = mx; break;
            case 13: r = (cnt == 0) ? ca_nan() : sum / cnt; break;
            case 14: r = (cnt == 0) ? ca_nan() : cnt; break;
            case 17: r = (cnt == 0) ? ca_nan() : 1.0; break;
            default: r = (cnt == 0) ? ca_nan() : pow(prod, 1.0 / cnt); break;
          }
          out[e] = r;
          break;
        }
        case 9: { /* iqr bounds */
          int cnt = ca_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
          double q25 = ca_quantile_sorted(0.25, sc, cnt);
          double q75 = ca_quantile_sorted(0.75, sc, cnt);
          double iqr = 1.5 * (q75 - q25);
          out[e] = q25 - iqr;
          out2[e] = q75 + iqr;
          break;
        }
        default:
          break;
      }
    }
  }
  free(sc);
  (void)n_series;
}

void vm_colagg_filter(int32_t mode, const double* values,
                      const int32_t* group_of, uint32_t n_series,
                      uint32_t n_grid, const double* b1, const double* b2,
                      uint8_t* flags) {
  for (uint32_t s = 0; s < n_series; s++) {
    int32_t grp = group_of[s];
    uint8_t f = 0;
    if (grp >= 0) {
      const double* row = values + (size_t)s * n_grid;
      const double* r1 = b1 + (size_t)grp * n_grid;
      const double* r2 = b2 + (size_t)grp * n_grid;
      for (uint32_t g = 0; g < n_grid && !f; g++) {
        double v = row[g];
        if (mode == 0) {
          if (v > r2[g] || v < r1[g]) f = 1;
        } else {
          if (fabs(v - r1[g]) > r2[g]) f = 1;
        }
      }
    }
    flags[s] = f;
  }
}
