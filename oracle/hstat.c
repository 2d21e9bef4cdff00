/* oracle/hstat.c — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
 *
 * CPU restatement of histogram_avg / histogram_stddev / histogram_stdvar
 * (transform.go: avgForLeTimeseries / stdvarForLeTimeseries) and
 * histogram_share (transformHistogramShare's `share` closure), over the
 * CSR grouped-bucket layout the GPU kernels take.
 */
#include <math.h>
#include <stdint.h>

static double hs_nan(void) { return nan(""); }

void vm_histogram_stat(int32_t mode, const double* bv, const double* les,
                       const uint64_t* goff, int64_t n_groups, int64_t n_grid,
                       double* out) {
  for (int64_t grp = 0; grp < n_groups; grp++) {
    int64_t lo = (int64_t)goff[grp];
    int64_t n_les = (int64_t)(goff[grp + 1] - lo);
    for (int64_t g = 0; g < n_grid; g++) {
      double le_prev = 0, v_prev = 0, sum = 0, sum2 = 0, wt = 0;
      for (int64_t j = 0; j < n_les; j++) {
        double le = les[lo + j];
        if (isinf(le)) continue;
        double v = bv[(lo + j) * n_grid + g];
        double n = (le + le_prev) / 2;
        double w = v - v_prev;
        sum += n * w;
        sum2 += n * n * w;
        wt += w;
        le_prev = le;
        v_prev = v;
      }
      double r;
      if (wt == 0) {
        r = hs_nan();
      } else if (mode == 0) {
        r = sum / wt;
      } else {
        double avg = sum / wt;
        double sv = sum2 / wt - avg * avg;
        if (sv < 0) sv = 0;
        r = (mode == 1) ? sqrt(sv) : sv;
      }
      out[grp * n_grid + g] = r;
    }
  }
}

void vm_histogram_share(const double* le_req, const double* bv,
                        const double* les, const uint64_t* goff,
                        int64_t n_groups, int64_t n_grid, double* out,
                        double* out_lo, double* out_hi) {
  for (int64_t grp = 0; grp < n_groups; grp++) {
    int64_t lo = (int64_t)goff[grp];
    int64_t n_les = (int64_t)(goff[grp + 1] - lo);
    for (int64_t g = 0; g < n_grid; g++) {
      double req = le_req[g];
      double q = hs_nan(), lb = hs_nan(), ub = hs_nan();
      if (!isnan(req) && n_les > 0) {
        if (req < 0) {
          q = 0; lb = 0; ub = 0;
        } else if (isinf(req) && req > 0) {
          q = 1; lb = 1; ub = 1;
        } else {
          double v_last = 0;
          {
            double fix_prev = 0;
            for (int64_t j = 0; j < n_les; j++) {
              double v = bv[(lo + j) * n_grid + g];
              if (j == 0) fix_prev = isnan(v) ? 0 : v;
              else if (!(isnan(v) || fix_prev > v)) fix_prev = v;
            }
            v_last = fix_prev;
          }
          double vp = 0, lep = 0, fix_prev = 0;
          int done = 0;
          for (int64_t j = 0; j < n_les && !done; j++) {
            double raw = bv[(lo + j) * n_grid + g];
            double v;
            if (j == 0) v = isnan(raw) ? 0 : raw;
            else v = (isnan(raw) || fix_prev > raw) ? fix_prev : raw;
            fix_prev = v;
            double le = les[lo + j];
            if (req >= le) {
              vp = v;
              lep = le;
              continue;
            }
            lb = vp / v_last;
            if (isinf(le) && le > 0) {
              q = lb;
              ub = 1;
            } else if (lep == req) {
              q = lb;
              ub = lb;
            } else {
              ub = v / v_last;
              q = lb + (v - vp) / v_last * (req - lep) / (le - lep);
            }
            done = 1;
          }
          if (!done) { q = 1; lb = 1; ub = 1; }
        }
      }
      out[grp * n_grid + g] = q;
      if (out_lo) out_lo[grp * n_grid + g] = lb;
      if (out_hi) out_hi[grp * n_grid + g] = ub;
    }
  }
}
