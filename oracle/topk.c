/* topk.c — CPU oracle for the topk/bottomk aggregate family and
 * histogram_quantile.
 *
 * TEST INFRASTRUCTURE ONLY (see vm_oracle.h header note).
 * Faithful C restatement of:
 *   app/vmselect/promql/aggr.go: newAggrFuncTopK 646-675 (per-point sort +
 *     fillNaNsAtIdx 786-791 + getIntK 793-802), getRangeTopKTimeseries
 *     704-741 (+ summary funcs minValue/maxValue/avgValue/medianValue/
 *     lastValue 804-858, getRemainingSumTimeseries 751-784),
 *     lessWithNaNs/greaterWithNaNs 1259-1279
 *   app/vmselect/promql/transform.go: transformHistogramQuantile quantile
 *     walk 1028-1074, fixBrokenBuckets 1140-1167, mergeSameLE 1169-1187
 *
 * Dense-matrix form: series labels are ids; "series" = row of the
 * [n_series x n_grid] rollup output.
 */
#include "vm_oracle.h"
#include <math.h>
#include <stdlib.h>
#include <string.h>

static const double NAN_V = NAN;

/* lessWithNaNs (aggr.go:1259): NaN < everything */
static int less_with_nans(double a, double b) {
  if (isnan(a)) return !isnan(b);
  if (isnan(b)) return 0;
  return a < b;
}
static int greater_with_nans(double a, double b) {
  if (isnan(a)) return !isnan(b);
  if (isnan(b)) return 0;
  return a > b;
}

/* getIntK (aggr.go:793-802) */
static int64_t get_int_k(double k, int64_t max_v) {
  if (isnan(k)) return 0;
  int64_t kn = (int64_t)k;
  if (k > 9.2e18) kn = max_v; /* floatToIntBounded clamp */
  if (kn < 0) return 0;
  if (kn > max_v) return max_v;
  return kn;
}

typedef struct {
  double v;
  int64_t idx;
} kv_t;

static int g_reverse = 0;
static int cmp_kv(const void* pa, const void* pb) {
  const kv_t* a = (const kv_t*)pa;
  const kv_t* b = (const kv_t*)pb;
  int lt = g_reverse ? greater_with_nans(a->v, b->v) : less_with_nans(a->v, b->v);
  int gt = g_reverse ? greater_with_nans(b->v, a->v) : less_with_nans(b->v, a->v);
  if (lt) return -1;
  if (gt) return 1;
  /* stabilize ties by original index so results are deterministic (the
   * reference's sort.Slice is unstable: tie order there is unspecified) */
  return (a->idx < b->idx) ? -1 : (a->idx > b->idx ? 1 : 0);
}

/* Per-point topk (newAggrFuncTopK): for every grid point keep the k top
 * (bottom if reverse) values across series, NaN-fill the rest, in place. */
void vm_topk_pointwise(double* values, int64_t n_series, int64_t n_grid,
                       double k, int32_t reverse) {
  if (n_series == 0) return;
  kv_t* arr = (kv_t*)malloc((size_t)n_series * sizeof(kv_t));
  int64_t kn = get_int_k(k, n_series);
  for (int64_t gpt = 0; gpt < n_grid; gpt++) {
    for (int64_t s = 0; s < n_series; s++) {
      arr[s].v = values[s * n_grid + gpt];
      arr[s].idx = s;
    }
    g_reverse = reverse;
    qsort(arr, (size_t)n_series, sizeof(kv_t), cmp_kv);
    /* fillNaNsAtIdx: NaN all but the last kn after the sort */
    for (int64_t s = 0; s < n_series - kn; s++) {
      values[arr[s].idx * n_grid + gpt] = NAN_V;
    }
  }
  free(arr);
}

/* series summary funcs (aggr.go:804-858), NaN-skipping */
static double summary_min(const double* v, int64_t n) {
  double m = NAN_V;
  int64_t i = 0;
  while (i < n && isnan(m)) m = v[i++];
  for (; i < n; i++)
    if (!isnan(v[i]) && v[i] < m) m = v[i];
  return m;
}
static double summary_max(const double* v, int64_t n) {
  double m = NAN_V;
  int64_t i = 0;
  while (i < n && isnan(m)) m = v[i++];
  for (; i < n; i++)
    if (!isnan(v[i]) && v[i] > m) m = v[i];
  return m;
}
static double summary_avg(const double* v, int64_t n) {
  double s = 0;
  int64_t c = 0;
  for (int64_t i = 0; i < n; i++) {
    if (isnan(v[i])) continue;
    c++;
    s += v[i];
  }
  if (c == 0) return NAN_V;
  return s / (double)c;
}
static double summary_last(const double* v, int64_t n) {
  while (n > 0 && isnan(v[n - 1])) n--;
  if (n == 0) return NAN_V;
  return v[n - 1];
}
static double summary_median(const double* v, int64_t n) {
  return vm_quantile(0.5, v, n);
}

double vm_topk_summary(int32_t op, const double* v, int64_t n) {
  switch (op) {
    case 0: return summary_avg(v, n);
    case 1: return summary_min(v, n);
    case 2: return summary_max(v, n);
    case 3: return summary_median(v, n);
    case 4: return summary_last(v, n);
    default: return NAN_V;
  }
}

/* Range topk (getRangeTopKTimeseries, aggr.go:704-741): rank series by a
 * whole-range summary, keep the top k.  Outputs:
 *   out_sel       [<=k]  selected series ids in the reference's output order
 *                        (descending rank after the final reverseSeries)
 *   out_remaining [n_grid] per-point sum over the NON-selected series
 *                        (NaN where no non-NaN values), if non-NULL
 * Returns the number of selected series. */
int64_t vm_topk_range(const double* values, int64_t n_series, int64_t n_grid,
                      double k, int32_t summary_op, int32_t reverse,
                      int64_t* out_sel, double* out_remaining) {
  kv_t* arr = (kv_t*)malloc((size_t)n_series * sizeof(kv_t));
  for (int64_t s = 0; s < n_series; s++) {
    arr[s].v = vm_topk_summary(summary_op, values + s * n_grid, n_grid);
    arr[s].idx = s;
  }
  g_reverse = reverse;
  qsort(arr, (size_t)n_series, sizeof(kv_t), cmp_kv);
  int64_t kn = get_int_k(k, n_series);
  if (out_remaining) {
    for (int64_t gpt = 0; gpt < n_grid; gpt++) {
      double sum = 0;
      int64_t cnt = 0;
      for (int64_t s = 0; s < n_series - kn; s++) {
        double v = values[arr[s].idx * n_grid + gpt];
        if (isnan(v)) continue;
        sum += v;
        cnt++;
      }
      out_remaining[gpt] = (cnt == 0) ? NAN_V : sum;
    }
  }
  /* kept = last kn of the sorted order; output order = reversed (best first) */
  int64_t m = 0;
  for (int64_t s = n_series - 1; s >= n_series - kn; s--) {
    out_sel[m++] = arr[s].idx;
  }
  free(arr);
  return m;
}

/* ---- histogram_quantile ---- */

/* fixBrokenBuckets (transform.go:1140-1167) on one grid point of a bucket
 * column (values already sorted by le, same-le merged). */
static void fix_broken_buckets(double* col, int64_t n_les) {
  if (n_les < 1) return;
  double v_prev = col[0];
  if (isnan(v_prev)) {
    v_prev = 0;
    col[0] = 0;
  }
  for (int64_t j = 1; j < n_les; j++) {
    double v = col[j];
    if (isnan(v) || v_prev > v) col[j] = v_prev;
    else v_prev = v;
  }
}

/* quantile walk (transform.go:1028-1074) for one grid point.
 * les/col hold the group's buckets sorted by le (same-le pre-merged). */
static double hq_point(double phi, const double* les, double* col, int64_t n_les,
                       double* out_lower, double* out_upper) {
  double lower = NAN_V, upper = NAN_V, q = NAN_V;
  if (isnan(phi)) goto done;
  fix_broken_buckets(col, n_les);
  {
    double v_last = (n_les > 0) ? col[n_les - 1] : 0;
    if (v_last == 0) goto done;
    if (phi < 0) {
      q = -INFINITY;
      lower = -INFINITY;
      upper = col[0];
      goto done;
    }
    if (phi > 1) {
      q = INFINITY;
      lower = v_last;
      upper = INFINITY;
      goto done;
    }
    double v_req = v_last * phi;
    double v_prev = 0, le_prev = 0;
    for (int64_t j = 0; j < n_les; j++) {
      double v = col[j];
      double le = les[j];
      if (v <= 0) {
        le_prev = le;
        continue;
      }
      if (v < v_req) {
        v_prev = v;
        le_prev = le;
        continue;
      }
      if (isinf(le)) break;
      if (v == v_prev) {
        q = le_prev;
        lower = le_prev;
        upper = v;
        goto done;
      }
      q = le_prev + (le - le_prev) * (v_req - v_prev) / (v - v_prev);
      lower = le_prev;
      upper = le;
      goto done;
    }
    /* lastNonInf (transform.go:1019-1027) */
    {
      int64_t j = n_les;
      double vv = NAN_V;
      while (j > 0) {
        if (!isinf(les[j - 1])) {
          vv = les[j - 1];
          break;
        }
        j--;
      }
      q = vv;
      lower = vv;
      upper = INFINITY;
    }
  }
done:
  if (out_lower) *out_lower = lower;
  if (out_upper) *out_upper = upper;
  return q;
}

/* histogram_quantile over grouped bucket rows.  Rows must be sorted by
 * (group, le) with same-le rows pre-merged (mergeSameLE adds values — the
 * host mirror does that; vm_hq_merge_same_le below is the helper).
 * bucket_values: [n_rows x n_grid]; les: [n_rows]; group row ranges via
 * group_offsets [n_groups+1].  out/out_lower/out_upper: [n_groups x n_grid]
 * (bounds optional). */
void vm_histogram_quantile(double phi, const double* bucket_values,
                           const double* les, const uint64_t* group_offsets,
                           int64_t n_groups, int64_t n_grid,
                           double* out, double* out_lower, double* out_upper) {
  int64_t max_les = 0;
  for (int64_t g = 0; g < n_groups; g++) {
    int64_t n = (int64_t)(group_offsets[g + 1] - group_offsets[g]);
    if (n > max_les) max_les = n;
  }
  double* col = (double*)malloc((size_t)(max_les > 0 ? max_les : 1) * sizeof(double));
  for (int64_t g = 0; g < n_groups; g++) {
    int64_t lo = (int64_t)group_offsets[g];
    int64_t n_les = (int64_t)(group_offsets[g + 1] - lo);
    for (int64_t gpt = 0; gpt < n_grid; gpt++) {
      for (int64_t j = 0; j < n_les; j++)
        col[j] = bucket_values[(lo + j) * n_grid + gpt];
      double lowv, upv;
      double q = hq_point(phi, les + lo, col, n_les, &lowv, &upv);
      out[g * n_grid + gpt] = q;
      if (out_lower) out_lower[g * n_grid + gpt] = lowv;
      if (out_upper) out_upper[g * n_grid + gpt] = upv;
    }
  }
  free(col);
}
