/* aggrcol.hip — non-incremental cross-series aggregates (aggr.go).
 *
 * The incremental ops (sum/min/max/avg/count/sum2/geomean/any/group,
 * aggr_incremental.go) fuse into the rollup kernels; everything else in
 * aggr.go needs the whole per-point column of member values at once:
 *   median / quantile     aggrFuncMedian :1230, aggrFuncQuantile via
 *                         newAggrQuantileFunc :1240 + quantile()
 *   mad                   aggrFuncMAD :942 (getPerPointMedians/MADs)
 *   stddev / stdvar       aggrFuncStddev :352 / aggrFuncStdvar :371
 *                         (per-point Welford)
 *   mode                  aggrFuncMode :446 + modeNoNaNs :541
 *   distinct              aggrFuncDistinct :423
 *   share                 aggrFuncShare :462 (per-series output)
 *   zscore                aggrFuncZScore :493 (per-series output)
 *   outliers_iqr bounds   getPerPointIQRBounds :975 (lower/upper rows;
 *                         series filtering via the filter kernel)
 *   outliers_mad filter   aggrFuncOutliersMAD :1022
 *
 * One thread per (group, grid point); member values gathered by row index
 * (CSR group_rows).  Sort-based ops insertion-sort the non-NaN column into
 * a per-thread global scratch slot — these run on grouped RESULT sets
 * (groups of 10-1000 members), not the raw series count.
 */
#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdio>

#include "../../include/vmgpu.h"

namespace {

int cset_err(char* errbuf, size_t len, const char* msg) {
  if (errbuf && len) snprintf(errbuf, len, "%s", msg);
  return 1;
}

int chip_err(char* errbuf, size_t len, const char* what, hipError_t e) {
  if (errbuf && len) snprintf(errbuf, len, "%s: %s", what, hipGetErrorString(e));
  return 2;
}

#define CHIP_TRY(expr, what)                                               \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) return chip_err(errbuf, errbuf_len, what, _e);   \
  } while (0)

struct CDevBuf {
  void* p = nullptr;
  ~CDevBuf() { if (p) (void)hipFree(p); }
  hipError_t alloc(size_t n) { return hipMalloc(&p, n ? n : 1); }
};

}  // namespace

static __device__ __forceinline__ double c_nan() {
  return __longlong_as_double(0x7ff8000000000000LL);
}

/* quantileSorted (aggr.go:922), bit-exact formulation */
static __device__ double c_quantile_sorted(double phi, const double* a, int n) {
  if (n == 0 || isnan(phi)) return c_nan();
  if (phi < 0) return __longlong_as_double(0xfff0000000000000LL);
  if (phi > 1) return __longlong_as_double(0x7ff0000000000000LL);
  double rank = phi * (double)(n - 1);
  double lower_idx = fmax(0.0, floor(rank));
  double upper_idx = fmin((double)(n - 1), lower_idx + 1.0);
  double weight = rank - floor(rank);
  return a[(int)lower_idx] * (1.0 - weight) + a[(int)upper_idx] * weight;
}

/* gather + insertion-sort the non-NaN member column into scratch */
static __device__ int c_sorted_col(const double* values, uint32_t n_grid,
                                   const uint32_t* rows, uint32_t lo,
                                   uint32_t hi, uint32_t g, double* sc) {
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

__global__ void colagg_kernel(int32_t op, const double* values,
                              double* values_out, /* per-series ops */
                              const uint32_t* group_rows,
                              const uint64_t* group_offsets,
                              uint32_t n_groups, uint32_t n_grid,
                              double phi, double* scratch,
                              uint32_t scratch_slots, uint32_t max_members,
                              double* out, double* out2) {
  uint64_t total = (uint64_t)n_groups * n_grid;
  uint64_t tid0 = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (uint64_t e = tid0; e < total; e += (uint64_t)gridDim.x * blockDim.x) {
    uint32_t grp = (uint32_t)(e / n_grid);
    uint32_t g = (uint32_t)(e % n_grid);
    uint32_t lo = (uint32_t)group_offsets[grp];
    uint32_t hi = (uint32_t)group_offsets[grp + 1];
    double* sc = scratch ? scratch + (tid0 % scratch_slots) * (size_t)max_members
                         : nullptr;
    switch (op) {
      case VMGPU_COLAGG_MEDIAN:
      case VMGPU_COLAGG_QUANTILE: {
        /* quantile() (rollup.go) = filter NaN, sort, quantileSorted */
        int cnt = c_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
        out[e] = c_quantile_sorted(
            op == VMGPU_COLAGG_MEDIAN ? 0.5 : phi, sc, cnt);
        break;
      }
      case VMGPU_COLAGG_MAD: {
        /* getPerPointMedians + getPerPointMADs (aggr.go) */
        int cnt = c_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
        double med = c_quantile_sorted(0.5, sc, cnt);
        for (int i = 0; i < cnt; i++) sc[i] = fabs(sc[i] - med);
        for (int i = 1; i < cnt; i++) {
          double v = sc[i];
          int j = i;
          while (j > 0 && sc[j - 1] > v) { sc[j] = sc[j - 1]; j--; }
          sc[j] = v;
        }
        out[e] = c_quantile_sorted(0.5, sc, cnt);
        break;
      }
      case VMGPU_COLAGG_STDDEV:
      case VMGPU_COLAGG_STDVAR: {
        /* aggrFuncStdvar: per-point Welford; single-member groups are
         * exactly zero (and NaN stays NaN) on both paths */
        double avg = 0, count = 0, q = 0;
        for (uint32_t k = lo; k < hi; k++) {
          double v = values[(size_t)group_rows[k] * n_grid + g];
          if (isnan(v)) continue;
          count++;
          double avg_new = avg + (v - avg) / count;
          q += (v - avg) * (v - avg_new);
          avg = avg_new;
        }
        if (count == 0) q = c_nan();
        double r = q / count;
        out[e] = (op == VMGPU_COLAGG_STDDEV) ? sqrt(r) : r;
        break;
      }
      case VMGPU_COLAGG_MODE: {
        /* modeNoNaNs (aggr.go:541) over the sorted non-NaN column */
        int cnt = c_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
        double prev = c_nan();
        double mode = c_nan();
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
      case VMGPU_COLAGG_DISTINCT: {
        int cnt = c_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
        int n = 0;
        for (int i = 0; i < cnt; i++)
          if (i == 0 || sc[i] != sc[i - 1]) n++;
        out[e] = (n == 0) ? c_nan() : (double)n;
        break;
      }
      case VMGPU_COLAGG_SHARE: {
        /* aggrFuncShare: non-negative sum, then v/sum per member */
        double sum = 0;
        for (uint32_t k = lo; k < hi; k++) {
          double v = values[(size_t)group_rows[k] * n_grid + g];
          if (isnan(v) || v < 0) continue;
          sum += v;
        }
        for (uint32_t k = lo; k < hi; k++) {
          size_t idx = (size_t)group_rows[k] * n_grid + g;
          double v = values[idx];
          values_out[idx] = (isnan(v) || v < 0) ? c_nan() : v / sum;
        }
        break;
      }
      case VMGPU_COLAGG_ZSCORE: {
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
      case VMGPU_COLAGG_SUM:
      case VMGPU_COLAGG_MIN:
      case VMGPU_COLAGG_MAX:
      case VMGPU_COLAGG_AVG:
      case VMGPU_COLAGG_COUNT:
      case VMGPU_COLAGG_SUM2:
      case VMGPU_COLAGG_GEOMEAN:
      case VMGPU_COLAGG_GROUP: {
        /* aggrFuncSum/Min/Max/Avg/Count/Sum2/Geomean/Group (aggr.go):
         * left-to-right member order (the reference's own order is Go map
         * iteration, i.e. unspecified) */
        double sum = 0, prod = 1, mn = c_nan(), mx = c_nan();
        double cnt = 0;
        for (uint32_t k = lo; k < hi; k++) {
          double v = values[(size_t)group_rows[k] * n_grid + g];
          if (isnan(v)) continue;
          cnt++;
          sum += (op == VMGPU_COLAGG_SUM2) ? v * v : v;
          prod *= v;
          if (isnan(mn) || v < mn) mn = v;
          if (isnan(mx) || v > mx) mx = v;
        }
        double r;
        switch (op) {
          case VMGPU_COLAGG_SUM:
          case VMGPU_COLAGG_SUM2:   r = (cnt == 0) ? c_nan() : sum; break;
          case VMGPU_COLAGG_MIN:    r = mn; break;
          case VMGPU_COLAGG_MAX:    r = mx; break;
          case VMGPU_COLAGG_AVG:    r = (cnt == 0) ? c_nan() : sum / cnt; break;
          case VMGPU_COLAGG_COUNT:  r = (cnt == 0) ? c_nan() : cnt; break;
          case VMGPU_COLAGG_GROUP:  r = (cnt == 0) ? c_nan() : 1.0; break;
          default:  /* geomean */
            r = (cnt == 0) ? c_nan() : pow(prod, 1.0 / cnt); break;
        }
        out[e] = r;
        break;
      }
      case VMGPU_COLAGG_IQR_BOUNDS: {
        /* getPerPointIQRBounds (aggr.go:975): q25/q75 +/- 1.5*iqr */
        int cnt = c_sorted_col(values, n_grid, group_rows, lo, hi, g, sc);
        double q25 = c_quantile_sorted(0.25, sc, cnt);
        double q75 = c_quantile_sorted(0.75, sc, cnt);
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

/* per-series outlier filter: flags[s]=1 when any point of series s lies
 * outside its group's bounds (aggrFuncOutliersIQR :952 / OutliersMAD
 * :1022 — mode 0: v > upper || v < lower; mode 1: |v - median| > mad) */
__global__ void colagg_filter_kernel(int32_t mode, const double* values,
                                     const int32_t* group_of,
                                     const double* b1, const double* b2,
                                     uint32_t n_series, uint32_t n_grid,
                                     uint8_t* flags) {
  for (uint64_t s = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       s < n_series; s += (uint64_t)gridDim.x * blockDim.x) {
    int32_t grp = group_of[s];
    uint8_t f = 0;
    if (grp >= 0) {
      const double* row = values + s * (size_t)n_grid;
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

extern "C" {

int vmgpu_colagg(int32_t op, const double* values, uint32_t n_series,
                 uint32_t n_grid, const uint32_t* group_rows,
                 const uint64_t* group_offsets, uint32_t n_groups,
                 double phi, double* out, double* out2,
                 double* values_out, char* errbuf, size_t errbuf_len) {
  if (!values || !group_rows || !group_offsets || n_groups == 0 || n_grid == 0)
    return cset_err(errbuf, errbuf_len, "vmgpu: bad colagg args");
  hipStream_t st = 0;
  size_t vbytes = (size_t)n_series * n_grid * 8;
  uint64_t n_rows = group_offsets[n_groups];
  uint32_t max_members = 0;
  for (uint32_t grp = 0; grp < n_groups; grp++) {
    uint32_t m = (uint32_t)(group_offsets[grp + 1] - group_offsets[grp]);
    if (m > max_members) max_members = m;
  }
  bool needs_scratch = (op == VMGPU_COLAGG_MEDIAN || op == VMGPU_COLAGG_QUANTILE ||
                        op == VMGPU_COLAGG_MAD || op == VMGPU_COLAGG_MODE ||
                        op == VMGPU_COLAGG_DISTINCT || op == VMGPU_COLAGG_IQR_BOUNDS);
  bool per_series = (op == VMGPU_COLAGG_SHARE || op == VMGPU_COLAGG_ZSCORE);
  size_t out_elems = (size_t)n_groups * n_grid;
  CDevBuf dv, dvo, dgr, dgo, dout, dout2, dscr;
  CHIP_TRY(dv.alloc(vbytes), "alloc colagg vals");
  CHIP_TRY(hipMemcpyAsync(dv.p, values, vbytes, hipMemcpyHostToDevice, st), "ul vals");
  CHIP_TRY(dgr.alloc((size_t)n_rows * 4), "alloc colagg rows");
  CHIP_TRY(dgo.alloc((size_t)(n_groups + 1) * 8), "alloc colagg off");
  CHIP_TRY(hipMemcpyAsync(dgr.p, group_rows, (size_t)n_rows * 4, hipMemcpyHostToDevice, st), "ul rows");
  CHIP_TRY(hipMemcpyAsync(dgo.p, group_offsets, (size_t)(n_groups + 1) * 8, hipMemcpyHostToDevice, st), "ul off");
  uint32_t blocks = (uint32_t)std::min<uint64_t>((out_elems + 255) / 256, 2048);
  uint32_t scratch_slots = 0;
  if (needs_scratch) {
    scratch_slots = blocks * 256;
    size_t sbytes = (size_t)scratch_slots * max_members * 8;
    while (sbytes > (size_t)2 << 30 && blocks > 64) {  /* cap scratch at 2 GB */
      blocks /= 2;
      scratch_slots = blocks * 256;
      sbytes = (size_t)scratch_slots * max_members * 8;
    }
    CHIP_TRY(dscr.alloc(sbytes), "alloc colagg scratch");
  }
  if (per_series) CHIP_TRY(dvo.alloc(vbytes), "alloc colagg vout");
  if (out) CHIP_TRY(dout.alloc(out_elems * 8), "alloc colagg out");
  if (out2) CHIP_TRY(dout2.alloc(out_elems * 8), "alloc colagg out2");
  hipLaunchKernelGGL(colagg_kernel, dim3(blocks), dim3(256), 0, st, op,
                     (const double*)dv.p, (double*)dvo.p,
                     (const uint32_t*)dgr.p, (const uint64_t*)dgo.p,
                     n_groups, n_grid, phi, (double*)dscr.p, scratch_slots,
                     max_members, (double*)dout.p, (double*)dout2.p);
  if (out) CHIP_TRY(hipMemcpyAsync(out, dout.p, out_elems * 8, hipMemcpyDeviceToHost, st), "dl out");
  if (out2) CHIP_TRY(hipMemcpyAsync(out2, dout2.p, out_elems * 8, hipMemcpyDeviceToHost, st), "dl out2");
  if (per_series && values_out)
    CHIP_TRY(hipMemcpyAsync(values_out, dvo.p, vbytes, hipMemcpyDeviceToHost, st), "dl vout");
  CHIP_TRY(hipStreamSynchronize(st), "sync colagg");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess) return chip_err(errbuf, errbuf_len, "colagg kernel", kerr);
  return 0;
}

int vmgpu_colagg_filter(int32_t mode, const double* values,
                        const int32_t* group_of, uint32_t n_series,
                        uint32_t n_grid, const double* b1, const double* b2,
                        uint32_t n_groups, uint8_t* flags,
                        char* errbuf, size_t errbuf_len) {
  if (!values || !group_of || !b1 || !b2 || !flags || n_series == 0)
    return cset_err(errbuf, errbuf_len, "vmgpu: bad colagg filter args");
  hipStream_t st = 0;
  size_t vbytes = (size_t)n_series * n_grid * 8;
  size_t bbytes = (size_t)n_groups * n_grid * 8;
  CDevBuf dv, dgo, db1, db2, df;
  CHIP_TRY(dv.alloc(vbytes), "alloc filt vals");
  CHIP_TRY(dgo.alloc((size_t)n_series * 4), "alloc filt grp");
  CHIP_TRY(db1.alloc(bbytes), "alloc filt b1");
  CHIP_TRY(db2.alloc(bbytes), "alloc filt b2");
  CHIP_TRY(df.alloc(n_series), "alloc filt flags");
  CHIP_TRY(hipMemcpyAsync(dv.p, values, vbytes, hipMemcpyHostToDevice, st), "ul vals");
  CHIP_TRY(hipMemcpyAsync(dgo.p, group_of, (size_t)n_series * 4, hipMemcpyHostToDevice, st), "ul grp");
  CHIP_TRY(hipMemcpyAsync(db1.p, b1, bbytes, hipMemcpyHostToDevice, st), "ul b1");
  CHIP_TRY(hipMemcpyAsync(db2.p, b2, bbytes, hipMemcpyHostToDevice, st), "ul b2");
  uint32_t blocks = std::min<uint32_t>((n_series + 255) / 256, 2048);
  hipLaunchKernelGGL(colagg_filter_kernel, dim3(blocks), dim3(256), 0, st,
                     mode, (const double*)dv.p, (const int32_t*)dgo.p,
                     (const double*)db1.p, (const double*)db2.p,
                     n_series, n_grid, (uint8_t*)df.p);
  CHIP_TRY(hipMemcpyAsync(flags, df.p, n_series, hipMemcpyDeviceToHost, st), "dl flags");
  CHIP_TRY(hipStreamSynchronize(st), "sync filt");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess) return chip_err(errbuf, errbuf_len, "filt kernel", kerr);
  return 0;
}

}  /* extern "C" */
