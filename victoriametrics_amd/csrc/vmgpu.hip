/* vmgpu.hip — MI355X-native (gfx950/CDNA4) rollup + aggregation engine.
 *
 * Replaces the reference's per-series worker fan-out
 * (netstorage.Results.RunParallel, netstorage.go:219 +
 * evalRollup{With,No}IncrementalAggregate, eval.go:1927/1968) with three HIP
 * kernels picked by series length, all sharing the device rollup set in
 * rollup_device.h:
 *
 *   rollup_wave_kernel  — series with <= CHUNK_WAVE samples: one 64-lane
 *       wavefront per series.  The wave stages the series' (ts, vals)
 *       columns into LDS with coalesced loads, optionally drops Prometheus
 *       stale NaNs (eval.go:2108) and applies removeCounterResets
 *       (rollup.go:921) as an EXACT-order fused scan (sparse sequential
 *       correction walk over ballot'd reset/gap events + segmented prefix
 *       max for the monotonic clamp), then evaluates the [start:end:step]
 *       grid with one lane per grid point (binary-search window seek in LDS,
 *       left-to-right serial window reductions => bit-exact vs the
 *       sequential reference semantics).
 *   rollup_block_kernel — series up to CHUNK_BLOCK samples: one 256-thread
 *       workgroup per series, series staged in up to 128 KiB of LDS.
 *   rollup_huge_kernel  — longer series: per-series preprocessing into a
 *       global scratch column, grid evaluated from L2-cached global memory.
 *
 * Cross-series by-label aggregation (aggr_incremental.go) is fused into the
 * grid loop: identity-initialized [n_groups x n_grid] value/count matrices
 * updated with device-scope f64 atomics (CAS loops for min/max/product), so
 * the 1M-series rollup never materializes per-series grids in HBM.  The
 * multi-GPU merge all-reduces those matrices BEFORE the
 * finalize step; plan->skip_finalize exposes exactly that cut.
 */
#include <hip/hip_runtime.h>
#include <algorithm>
#include <atomic>
#include <cstdlib>
#include <cstring>
#include <cstdio>
#include <map>
#include <mutex>
#include <vector>

#include "devalloc.h"
#include "rollup_device.h"
#include "../../include/vmgpu.h"

#define WAVE 64
#define BLOCK_THREADS 256
#define WAVES_PER_BLOCK (BLOCK_THREADS / WAVE)
#define CHUNK_WAVE 512
/* block-kernel series cap: 16 B/sample of LDS + 16 B control word must stay
 * within the 64 KiB dynamic-LDS launch limit (the 160 KiB/CU budget is a
 * static-declaration property on gfx950) */
#define CHUNK_BLOCK 4064
#define MAX_WAVE_BLOCKS 4096

/* ------------------------------------------------------------------ */
/* device helpers                                                     */
/* ------------------------------------------------------------------ */

static VM_DEV void wave_lds_sync() {
  /* All DS and VMEM ops of this wave complete + compiler barrier: safe
   * cross-lane visibility within one wavefront for LDS *and* for the huge
   * kernel's global-scratch variant (no workgroup barrier needed).  Used
   * between phases only, never inside hot loops. */
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
}

static VM_DEV void wave_ds_sync() {
  /* LDS-only phase barrier for one wavefront: DS ops (including the
   * ds_bpermute behind __shfl) are lgkm-counted, so this gives cross-lane
   * LDS visibility while deliberately leaving prefetched global loads
   * (vmcnt) in flight — the point of the software-pipelined kernel. */
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

/* STR = element stride in int64 units: 1 for column layout, 2 for the pipe
 * kernel's (t, v) pair-interleaved LDS (one ds_read2/b128 per sample). */
template <int STR = 1>
static VM_DEV int vm_upper_bound(const int64_t* ts, int n, int64_t seek) {
  /* first index with ts[idx] > seek (seekFirstTimestampIdxAfter semantics,
   * rollup.go:825-855 — the reference's ±2 hint only narrows the range). */
  int i = 0, j = n;
  while (i < j) {
    int h = (i + j) >> 1;
    if (ts[h * STR] <= seek) i = h + 1;
    else j = h;
  }
  return i;
}

/* Per-lane upper bound from a density-interpolated guess: a bounded walk
 * resolves near-uniform sampling in 1-2 probes; anything irregular falls
 * back to a plain binary search.  Result identical to vm_upper_bound. */
template <int STR = 1>
static VM_DEV int vm_ub_hint(const int64_t* ts, int n, int64_t seek, int g) {
  if (g < 0) g = 0;
  if (g > n) g = n;
  for (int steps = 0; steps < 4; steps++) {
    bool gt_here = (g >= n) || (ts[g * STR] > seek);
    if (!gt_here) { g++; continue; }       /* result is above g */
    if (g == 0 || ts[(g - 1) * STR] <= seek) return g;
    g--;                                   /* result is below g */
  }
  return vm_upper_bound<STR>(ts, n, seek);
}

/* Branch-free variant: four independent probes decide among {g-1, g, g+1}
 * with pure selects; lanes outside the +-1 window (rare on real scrape
 * cadences) take a wave-coordinated binary-search fallback.  Result
 * identical to vm_upper_bound. */
template <int STR = 1>
static VM_DEV int vm_ub_hint_fast(const int64_t* ts, int n, int64_t seek, int g) {
  if (g < 0) g = 0;
  if (g > n) g = n;
  int i_m2 = g - 2 < 0 ? 0 : g - 2;
  int i_m1 = g - 1 < 0 ? 0 : g - 1;
  int i_0 = g < n - 1 ? g : (n - 1 < 0 ? 0 : n - 1);
  int i_p1 = g + 1 < n - 1 ? g + 1 : (n - 1 < 0 ? 0 : n - 1);
  int64_t t_m2 = ts[i_m2 * STR];
  int64_t t_m1 = ts[i_m1 * STR];
  int64_t t_0 = ts[i_0 * STR];
  int64_t t_p1 = ts[i_p1 * STR];
  bool le_m2 = (g - 2 < 0) || (t_m2 <= seek);  /* ts[g-2] <= seek (vacuous at edge) */
  bool le_m1 = (g - 1 < 0) || (g - 1 >= n) || (t_m1 <= seek);
  bool gt_m1 = (g - 1 >= 0) && (g - 1 < n) && (t_m1 > seek);
  bool le_0 = (g < n) && (t_0 <= seek);
  bool gt_0 = (g >= n) || (t_0 > seek);
  bool gt_p1 = (g + 1 >= n) || (t_p1 > seek);
  /* ok(x): (x==0 or ts[x-1]<=seek) and (x==n or ts[x]>seek) */
  bool ok_g = le_m1 && gt_0;
  bool ok_p1 = (g + 1 <= n) && le_0 && gt_p1;
  bool ok_m1 = (g - 1 >= 0) && le_m2 && gt_m1;
  int r = ok_g ? g : (ok_p1 ? g + 1 : (ok_m1 ? g - 1 : -1));
  if (__any(r < 0)) {
    if (r < 0) r = vm_upper_bound<STR>(ts, n, seek);
  }
  return r;
}

static VM_DEV void vm_atomic_min_f64(double* addr, double val) {
  unsigned long long* p = (unsigned long long*)addr;
  unsigned long long old = __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  while (val < __longlong_as_double((long long)old)) {
    unsigned long long assumed = old;
    old = atomicCAS(p, assumed, (unsigned long long)__double_as_longlong(val));
    if (old == assumed) break;
  }
}

static VM_DEV void vm_atomic_max_f64(double* addr, double val) {
  unsigned long long* p = (unsigned long long*)addr;
  unsigned long long old = __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  while (val > __longlong_as_double((long long)old)) {
    unsigned long long assumed = old;
    old = atomicCAS(p, assumed, (unsigned long long)__double_as_longlong(val));
    if (old == assumed) break;
  }
}

static VM_DEV void vm_atomic_mul_f64(double* addr, double val) {
  unsigned long long* p = (unsigned long long*)addr;
  unsigned long long old = __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  while (true) {
    unsigned long long assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    old = atomicCAS(p, assumed, (unsigned long long)__double_as_longlong(cur * val));
    if (old == assumed) break;
  }
}

/* Kernel-side plan (host plan flattened; all wave-uniform). */
struct KPlan {
  int64_t start, end, step;
  int64_t window;          /* plan window (0 => auto-adjust per series) */
  int64_t lookback_delta;
  int64_t min_staleness;
  int64_t max_staleness;   /* removeCounterResets staleness interval */
  int32_t n_grid;
  int32_t func;
  int32_t aggr;
  int32_t sspc;            /* samplesScannedPerCall */
  int32_t may_adjust;
  int32_t is_default;
  int32_t rcr;
  int32_t drop_stale;
  int32_t chunk_wave;      /* LDS samples per wave (wave kernel), 64-aligned */
  int32_t chunk_block;     /* LDS samples for the block kernel, sized to the
                              batch's longest block-class series */
  int32_t pre_func;        /* VMGPU_PRE_* value transform after rcr */
  int32_t jbuf_elems;      /* LDS boundary-cache elements for the rate paths */
  int32_t jbuf_mode;       /* 0 = none; 1 = u16 j-cache (shared-boundary);
                              2 = sample-scatter Et/Ev/J boundary map */
  double arg;
  double arg2;
};

/* Block-kernel LDS: 32B header + chunk_block*16 series data + 256B scratch
 * + (when it fits the 64 KiB dynamic limit and a rate j-cache is wanted)
 * n_grid u16 window-end indices.  Host sizing and kernel carve must agree. */
static __host__ __device__ inline size_t vm_block_lds_bytes(
    int32_t chunk_block, int32_t jbuf_mode, int32_t n_grid, int* has_jbuf) {
  size_t base = 32 + (size_t)chunk_block * 16 + 256;
  size_t jb = ((size_t)n_grid * 2 + 15) & ~(size_t)15;
  if (jbuf_mode >= 1 && base + jb <= 64 * 1024) {
    if (has_jbuf) *has_jbuf = 1;
    return base + jb;
  }
  if (has_jbuf) *has_jbuf = 0;
  return base;
}

/* LDS bytes for the rate boundary cache (host sizing and kernel carve must
 * agree) */
static __host__ __device__ inline size_t vm_jbuf_bytes(int32_t mode,
                                                       int32_t elems) {
  size_t a2 = ((size_t)elems * 2 + 15) & ~(size_t)15;
  if (mode == 2) return a2;            /* J u16 over [-dg, n_grid) */
  if (mode == 1) return a2;            /* J u16 over [0, n_grid) */
  return 0;
}

struct KIO {
  const int64_t* ts = nullptr;
  const double* vals = nullptr;
  const uint64_t* offsets = nullptr;
  const uint32_t* series_sel = nullptr; /* list of series ids for this kernel */
  uint32_t n_sel = 0;
  const int32_t* group_ids = nullptr;   /* may be null */
  const int64_t* series_si = nullptr;   /* batch scrape-interval index (raw
                                           0.6-quantile per series; 0 = use
                                           the plan default).  Valid only
                                           when no samples were dropped. */
  double* out = nullptr;                /* per-series [n_series x n_grid] or group values */
  double* out_counts = nullptr;         /* group counts */
  unsigned long long* samples_scanned = nullptr;
  /* huge-kernel scratch */
  int64_t* scr_ts = nullptr;
  double* scr_vals = nullptr;
  const uint64_t* scr_offsets = nullptr; /* per huge-series scratch base */
};

/* --- phase A: compacted load (executed by ONE wave) ----------------- */
/* Copies [g_ts,g_vals)[0..n) to dst, dropping stale NaNs if requested.
 * Returns new count. dst may be LDS or global. */
struct __align__(16) vm_i64x2 { int64_t x, y; };

template <int STR = 1>
static __device__ int load_compact_wave(const int64_t* g_ts, const double* g_vals,
                                        int64_t n, int64_t* d_ts, double* d_vals,
                                        bool drop_stale, int lane) {
  int count = 0;
  int64_t src = 0;
  /* vector fast path: 128-element tiles via 16-B loads/stores (the staging
   * loop is instruction-issue-bound, not bandwidth-bound — see profiles/).
   * Falls to the scalar tile path when stale NaNs must be compacted out or
   * the running count is odd (LDS b128 stores need 16-B alignment);
   * column layout only. */
  if (STR == 1 && (((uintptr_t)g_ts | (uintptr_t)g_vals) & 15) == 0) {
    const uint64_t lt_mask = (lane == 63) ? 0x7fffffffffffffffULL
                                          : ((1ULL << lane) - 1);
    while (src + 2 <= n) {
      int64_t pairs = (n - src) >> 1;
      if (pairs > 64) pairs = 64;
      bool act = lane < pairs;
      double2 v = {0.0, 0.0};
      vm_i64x2 t = {0, 0};
      if (act) {
        v = *(const double2*)(g_vals + src + 2 * lane);
        t = *(const vm_i64x2*)(g_ts + src + 2 * lane);
      }
      bool k0 = act && !(drop_stale && vm_is_stale_nan(v.x));
      bool k1 = act && !(drop_stale && vm_is_stale_nan(v.y));
      uint64_t m0 = __ballot(k0), m1 = __ballot(k1);
      uint64_t full = (pairs == 64) ? ~0ULL : ((1ULL << pairs) - 1);
      if ((m0 & m1) == full && (count & 1) == 0) {
        if (act) {
          *(double2*)(d_vals + count + 2 * lane) = v;
          *(vm_i64x2*)(d_ts + count + 2 * lane) = t;
        }
        count += (int)(2 * pairs);
      } else {
        int below = __popcll(m0 & lt_mask) + __popcll(m1 & lt_mask);
        int dst0 = count + below;
        if (k0) { d_ts[dst0] = t.x; d_vals[dst0] = v.x; }
        if (k1) { int d1 = dst0 + (k0 ? 1 : 0); d_ts[d1] = t.y; d_vals[d1] = v.y; }
        count += __popcll(m0) + __popcll(m1);
      }
      src += 2 * pairs;
    }
  }
  for (; src < n; src += WAVE) {
    int64_t k = src + lane;
    bool active = k < n;
    double v = 0.0;
    int64_t t = 0;
    if (active) {
      v = g_vals[k];
      t = g_ts[k];
    }
    bool keep = active && !(drop_stale && vm_is_stale_nan(v));
    uint64_t m = __ballot(keep);
    if (keep) {
      int dst = count + __popcll(m & ((lane == 63) ? 0x7fffffffffffffffULL
                                                   : ((1ULL << lane) - 1)));
      d_ts[dst * STR] = t;
      d_vals[dst * STR] = v;
    }
    count += __popcll(m);
  }
  return count;
}

/* --- fused phase A+B: staging + removeCounterResets in one pass ------ */
/* When no sample is dropped (no stale NaNs — the overwhelmingly common
 * case), the counter-reset scan can run on the staging registers, saving
 * the whole LDS re-read pass.  Returns the kept count on success, or -1 to
 * signal "drops present, redo via load_compact_wave + rcr_scan_wave"
 * (wave-uniform).  Scan semantics identical to rcr_scan_wave below. */
static __device__ int load_rcr_fused_wave(const int64_t* g_ts, const double* g_vals,
                                          int64_t n, int64_t* d_ts, double* d_vals,
                                          bool drop_stale, int64_t msi, int lane) {
  double corr = 0.0;
  double prev_raw = 0.0;
  int64_t prev_ts = 0;
  double prev_fin = 0.0;
  for (int64_t base = 0; base < n; base += WAVE) {
    int64_t k = base + lane;
    bool active = k < n;
    double v = 0.0;
    int64_t t = 0;
    if (active) {
      v = g_vals[k];
      t = g_ts[k];
    }
    if (drop_stale && __ballot(active && vm_is_stale_nan(v)) != 0) return -1;
    if (active) d_ts[k] = t;
    /* scan round (identical to rcr_scan_wave, values from registers) */
    double pv = __shfl_up(v, 1);
    int64_t pt = __shfl_up(t, 1);
    if (lane == 0) { pv = prev_raw; pt = prev_ts; }
    bool isfirst = (k == 0);
    double d = v - pv;
    double inc = 0.0;
    if (!isfirst && d < 0) inc = ((-d * 8) < pv) ? (pv - v) : pv;
    bool gap = (!isfirst && msi > 0 && (t - pt) > msi);
    uint64_t em = __ballot(active && (gap || inc != 0.0));
    uint64_t dm = __ballot(active && !isfirst && d < 0);
    int last = (int)(n - base - 1);
    if (last > 63) last = 63;
    if (em == 0 && dm == 0 &&
        (base == 0 || __shfl(v, 0) + corr >= prev_fin)) {
      if (active) d_vals[k] = v + corr;
      prev_raw = __shfl(v, last);
      prev_ts = __shfl(t, last);
      prev_fin = prev_raw + corr;
      continue;
    }
    double c = corr;
    double mycorr = corr;
    while (em) {
      int b = __ffsll((unsigned long long)em) - 1;
      em &= em - 1;
      double ib = __shfl(inc, b);
      int gb = __shfl((int)gap, b);
      double cn = gb ? 0.0 : (c + ib);
      if (lane >= b) mycorr = cn;
      c = cn;
    }
    double fin = v + mycorr;
    bool bnd = isfirst || gap;
    double x = fin;
    int f = bnd ? 1 : 0;
    if (lane == 0 && !bnd) x = fmax(x, prev_fin);
    for (int dlt = 1; dlt < WAVE; dlt <<= 1) {
      double xo = __shfl_up(x, dlt);
      int fo = __shfl_up(f, dlt);
      if (lane >= dlt) {
        if (!f) x = fmax(x, xo);
        f = f | fo;
      }
    }
    if (active) d_vals[k] = x;
    corr = c;
    prev_raw = __shfl(v, last);
    prev_ts = __shfl(t, last);
    prev_fin = __shfl(x, last);
  }
  return (int)n;
}

/* --- phase B: removeCounterResets over the dense column -------------- */
/* EXACT restatement of rollup.go:921-958 as a wave scan; see file header. */
template <int STR = 1>
static __device__ void rcr_scan_wave(int64_t* d_ts, double* d_vals, int count,
                                     int64_t msi, int lane) {
  double corr = 0.0;       /* running correction */
  double prev_raw = 0.0;   /* raw value of previous element */
  int64_t prev_ts = 0;
  double prev_fin = 0.0;   /* clamped output of previous element */
  for (int base = 0; base < count; base += WAVE) {
    int k = base + lane;
    bool active = k < count;
    double v = active ? d_vals[k * STR] : 0.0;
    int64_t t = active ? d_ts[k * STR] : 0;
    double pv = __shfl_up(v, 1);
    int64_t pt = __shfl_up(t, 1);
    if (lane == 0) { pv = prev_raw; pt = prev_ts; }
    bool isfirst = (k == 0);
    double d = v - pv;
    double inc = 0.0;
    if (!isfirst && d < 0) inc = ((-d * 8) < pv) ? (pv - v) : pv;
    bool gap = (!isfirst && msi > 0 && (t - pt) > msi);
    /* corrections in exact sequential order: only reset/gap events change the
     * running correction, and f64 `c + 0.0` is an identity, so walking the
     * (rare) event lanes reproduces the serial left-to-right sum bitwise. */
    uint64_t em = __ballot(active && (gap || inc != 0.0));
    /* fast path: no events, no negative deltas, and the chunk starts at or
     * above the carried clamp level -> corrections are uniform and the
     * monotonic clamp is a no-op. */
    uint64_t dm = __ballot(active && !isfirst && d < 0);
    int last_fast = count - base - 1;
    if (last_fast > 63) last_fast = 63;
    if (em == 0 && dm == 0 &&
        (base == 0 || __shfl(v, 0) + corr >= prev_fin)) {
      if (corr != 0.0 && active) d_vals[k * STR] = v + corr;
      prev_raw = __shfl(v, last_fast);
      prev_ts = __shfl(t, last_fast);
      prev_fin = prev_raw + corr;
      continue;
    }
    double c = corr;
    double mycorr = corr;
    while (em) {
      int b = __ffsll((unsigned long long)em) - 1;
      em &= em - 1;
      double ib = __shfl(inc, b);
      int gb = __shfl((int)gap, b);
      double cn = gb ? 0.0 : (c + ib);
      if (lane >= b) mycorr = cn;
      c = cn;
    }
    double fin = v + mycorr;
    /* monotonic clamp (rollup.go:952-956) = segmented prefix max; a gap
     * element keeps its raw value and is exempt (the Go `continue`). */
    bool bnd = isfirst || gap;
    double x = fin;
    int f = bnd ? 1 : 0;
    if (lane == 0 && !bnd) x = fmax(x, prev_fin);
    for (int dlt = 1; dlt < WAVE; dlt <<= 1) {
      double xo = __shfl_up(x, dlt);
      int fo = __shfl_up(f, dlt);
      if (lane >= dlt) {
        if (!f) x = fmax(x, xo);
        f = f | fo;
      }
    }
    if (active) d_vals[k * STR] = x;
    int last = count - base - 1;
    if (last > 63) last = 63;
    corr = c;
    prev_raw = __shfl(v, last);
    prev_ts = __shfl(t, last);
    prev_fin = __shfl(x, last);
  }
}

/* --- preFunc value transforms (rollup_* families) -------------------- */
/* deltaValues (rollup.go:960), derivValues (:976) and the
 * rollup_scrape_interval preFunc (:476), applied in place after
 * removeCounterResets exactly as getRollupConfigs composes them.  Fast
 * paths are wave-elementwise; derivValues' duplicate-timestamp carry is
 * replayed serially on lane 0 when duplicates exist (rare — dedup
 * upstream normally removes them). */
static VM_DEV void pre_func_wave(const int64_t* d_ts, double* d_vals,
                                 int count, int mode, int lane) {
  if (count <= 0 || mode == VMGPU_PRE_NONE) return;
  if (mode == VMGPU_PRE_SCRAPE_INTERVAL) {
    /* values[i] = ts[i]/1e3 - ts[i-1]/1e3 (NaN seed); values[0]=values[1] */
    for (int base = 0; base < count; base += WAVE) {
      int k = base + lane;
      if (k < count) {
        double cur = (double)d_ts[k] / 1000.0;
        double prev = (k > 0) ? (double)d_ts[k - 1] / 1000.0 : vm_dnan();
        d_vals[k] = cur - prev;
      }
      wave_lds_sync();
    }
    if (lane == 0 && count > 1) d_vals[0] = d_vals[1];
    wave_lds_sync();
    return;
  }
  if (mode == VMGPU_PRE_DELTA) {
    double last = 0.0;
    if (count >= 2) last = d_vals[count - 1] - d_vals[count - 2];
    for (int base = 0; base + 1 < count; base += WAVE) {
      int k = base + lane;
      double a = 0, b = 0;
      bool act = (k + 1 < count);
      if (act) {
        a = d_vals[k];
        b = d_vals[k + 1];
      }
      wave_lds_sync();      /* all reads land before any write */
      if (act) d_vals[k] = b - a;
      wave_lds_sync();
    }
    if (lane == 0) d_vals[count - 1] = last;
    wave_lds_sync();
    return;
  }
  /* VMGPU_PRE_DERIV */
  bool dup = false;
  for (int base = 0; base + 1 < count; base += WAVE) {
    int k = base + lane;
    bool d = (k + 1 < count) && (d_ts[k + 1] == d_ts[k]);
    if (__ballot(d) != 0) dup = true;
  }
  if (!dup) {
    double last = 0.0;
    if (count >= 2)
      last = (d_vals[count - 1] - d_vals[count - 2]) /
             ((double)(d_ts[count - 1] - d_ts[count - 2]) / 1e3);
    for (int base = 0; base + 1 < count; base += WAVE) {
      int k = base + lane;
      double a = 0, b = 0;
      int64_t ta = 0, tb = 0;
      bool act = (k + 1 < count);
      if (act) {
        a = d_vals[k];
        b = d_vals[k + 1];
        ta = d_ts[k];
        tb = d_ts[k + 1];
      }
      wave_lds_sync();
      if (act) d_vals[k] = (b - a) / ((double)(tb - ta) / 1e3);
      wave_lds_sync();
    }
    if (lane == 0) d_vals[count - 1] = last;
    wave_lds_sync();
    return;
  }
  if (lane == 0) {
    /* exact serial replay of derivValues with the duplicate-ts carry */
    double prev_deriv = 0.0;
    double prev_value = d_vals[0];
    int64_t prev_ts = d_ts[0];
    for (int i = 0; i + 1 < count; i++) {
      double v = d_vals[i + 1];
      int64_t t = d_ts[i + 1];
      if (t == prev_ts) {
        d_vals[i] = prev_deriv;
        continue;
      }
      double dt = (double)(t - prev_ts) / 1e3;
      prev_deriv = (v - prev_value) / dt;
      d_vals[i] = prev_deriv;
      prev_value = v;
      prev_ts = t;
    }
    d_vals[count - 1] = prev_deriv;
  }
  wave_lds_sync();
}

/* --- per-series window parameters (doInternal preamble) -------------- */
/* getScrapeInterval (rollup.go:871-897): 0.6 quantile of the last <=20
 * sample gaps, computed wave-cooperatively with a rank-based selection
 * (identical result to sort + quantileSorted). */
template <bool LDS_ONLY, int STR = 1>
static __device__ int64_t scrape_interval_wave_t(const int64_t* d_ts, int count,
                                                 int64_t default_interval, int lane,
                                                 double* scratch) {
  if (count < 2) return default_interval;
  int cnt = count - 1;
  if (cnt > 20) cnt = 20;
  bool active = lane < cnt;
  double gap = 0.0;
  if (active) gap = (double)(d_ts[(count - 1 - lane) * STR] -
                             d_ts[(count - 2 - lane) * STR]);
  if (active) scratch[lane] = gap;
  if (LDS_ONLY) wave_ds_sync(); else wave_lds_sync();
  /* rank of this lane's gap among the cnt gaps (ties broken by lane);
   * scratch broadcast keeps the cnt probes independent (one LDS round trip)
   * instead of cnt serial shuffles */
  int rank = 0;
  for (int i = 0; i < cnt; i++) {
    double gi = scratch[i];
    if (active && (gi < gap || (gi == gap && i < lane))) rank++;
  }
  double nn = (double)cnt;
  double q_rank = 0.6 * (nn - 1);
  int li = (int)fmax(0.0, floor(q_rank));
  int ui = (int)fmin(nn - 1, (double)(li + 1));
  double w = q_rank - floor(q_rank);
  uint64_t mlo = __ballot(active && rank == li);
  uint64_t mhi = __ballot(active && rank == ui);
  int lane_lo = __ffsll((unsigned long long)mlo) - 1;
  int lane_hi = __ffsll((unsigned long long)mhi) - 1;
  double lo = __shfl(gap, lane_lo);
  double hi = __shfl(gap, lane_hi);
  double q = lo * (1 - w) + hi * w;
  int64_t si = (int64_t)q;
  if (si <= 0) return default_interval;
  return si;
}

static __device__ int64_t scrape_interval_wave(const int64_t* d_ts, int count,
                                               int64_t default_interval, int lane,
                                               double* scratch) {
  return scrape_interval_wave_t<false>(d_ts, count, default_interval, lane, scratch);
}

/* Batch scrape-interval index: getScrapeInterval's 0.6-quantile is a pure
 * function of the series' immutable resident timestamps, so it is computed
 * ONCE at batch creation (like the length-partition lists) and each query
 * reads it instead of re-deriving it per series.  Stored RAW (0 = "fewer
 than 2 samples or non-positive quantile": the kernels substitute the
 * plan's default, exactly as rollup.go:871-897 does).  Queries that drop
 * stale NaNs fall back to the in-kernel computation whenever the compacted
 * count differs from the stored one (the quantile could differ). */
__global__ __launch_bounds__(BLOCK_THREADS) void si_prep_kernel(
    const int64_t* ts, const uint64_t* offsets, uint32_t n_series,
    int64_t* out_si) {
  __shared__ double scratch[WAVES_PER_BLOCK][32];
  const int lane = threadIdx.x % WAVE;
  const int wv = threadIdx.x / WAVE;
  uint32_t wid = blockIdx.x * WAVES_PER_BLOCK + wv;
  uint32_t stride = gridDim.x * WAVES_PER_BLOCK;
  for (uint32_t i = wid; i < n_series; i += stride) {
    uint64_t lo = offsets[i];
    int count = (int)(offsets[i + 1] - lo);
    int64_t si = scrape_interval_wave_t<false>(ts + lo, count, 0, lane, scratch[wv]);
    if (lane == 0) out_si[i] = si;
  }
}

static VM_DEV int64_t max_prev_interval_tiers(int64_t si) {
  /* getMaxPrevInterval (rollup.go:899-919) */
  if (si <= 2000) return si + 4 * si;
  if (si <= 4000) return si + 2 * si;
  if (si <= 8000) return si + si;
  if (si <= 16000) return si + si / 2;
  if (si <= 32000) return si + si / 4;
  return si + si / 8;
}

struct SeriesWindow {
  int64_t window;
  int64_t max_prev_interval;
};

static VM_DEV SeriesWindow series_window(const KPlan& p, int64_t scrape_interval_est) {
  /* doInternal window setup (rollup.go:719-756) */
  SeriesWindow sw;
  int64_t mpi = p.step;
  if (p.start < p.end) mpi = max_prev_interval_tiers(scrape_interval_est);
  if (p.lookback_delta > 0 && mpi > p.lookback_delta) mpi = p.lookback_delta;
  if (p.min_staleness > 0 && mpi < p.min_staleness) mpi = p.min_staleness;
  int64_t window = p.window;
  if (window <= 0) {
    window = p.step;
    if (p.may_adjust && window < mpi) window = mpi;
    if (p.is_default && p.lookback_delta > 0 && window > p.lookback_delta)
      window = p.lookback_delta;
  }
  sw.window = window;
  sw.max_prev_interval = mpi;
  return sw;
}

/* --- grid-point evaluation (one lane, one grid point) -----------------
 * FUNC_CT >= 0 folds the rollup-function dispatch at compile time (the hot
 * functions get specialized kernels with small register footprints);
 * FUNC_CT == -1 is the generic runtime-dispatch fallback. */
/* Route one grid value to the per-series matrix or the grouped aggregate
 * (the eval seam of evalRollupNoIncrementalAggregate vs
 * evalRollupWithIncrementalAggregate). */
static VM_DEV void vm_emit_value(const KPlan& p, const KIO& io, uint32_t s,
                                 int g, double v) {
  if (p.aggr == VMGPU_AGGR_NONE) {
    io.out[(size_t)s * (size_t)p.n_grid + (size_t)g] = v;
    return;
  }
  int grp = io.group_ids ? io.group_ids[s] : -1;
  if (grp >= 0 && !vm_isnan(v)) {
    double* gv = io.out + (size_t)grp * (size_t)p.n_grid + (size_t)g;
    double* gc = io.out_counts + (size_t)grp * (size_t)p.n_grid + (size_t)g;
    /* counts are true contribution counts only for avg (finalize divides);
     * every other op uses them as a presence gate, where an idempotent
     * plain store of 1.0 replaces the second atomic (cross-shard merge
     * still works: the all-reduce SUM of flags stays nonzero) */
    switch (p.aggr) {
      case VMGPU_AGGR_SUM: atomicAdd(gv, v); *gc = 1.0; break;
      case VMGPU_AGGR_AVG: atomicAdd(gv, v); atomicAdd(gc, 1.0); break;
      case VMGPU_AGGR_MIN: vm_atomic_min_f64(gv, v); *gc = 1.0; break;
      case VMGPU_AGGR_MAX: vm_atomic_max_f64(gv, v); *gc = 1.0; break;
      case VMGPU_AGGR_COUNT:
      case VMGPU_AGGR_GROUP: atomicAdd(gv, 1.0); *gc = 1.0; break;
      case VMGPU_AGGR_SUM2: atomicAdd(gv, v * v); *gc = 1.0; break;
      /* geomean finalize takes pow(product, 1/count): true counts */
      case VMGPU_AGGR_GEOMEAN: vm_atomic_mul_f64(gv, v); atomicAdd(gc, 1.0); break;
      default: break;
    }
  }
}

/* Fused branch-free rate/deriv_fast evaluator (rollupDerivFast,
 * rollup.go:1954-1989 + the rfa construction of doInternal:779-810 reduced
 * to the fields rate reads): six independent LDS loads + selects. */
#ifndef VMGPU_PIPE_NO_TV
#define VMGPU_PIPE_NO_TV 0
#endif

/* 16-byte (t, v) record of the pair-interleaved LDS slab */
struct __align__(16) vm_tv {
  int64_t t;
  double v;
};

template <int STR = 1>
static VM_DEV double eval_rate_fused(const KPlan& p, const SeriesWindow& sw,
                                     const int64_t* ts, const double* vals,
                                     int count, int i, int j, int64_t t_start) {
  int im1 = i - 1 < 0 ? 0 : i - 1;
  int ii = i < count - 1 ? i : (count - 1 < 0 ? 0 : count - 1);
  int jm1 = j - 1 < 0 ? 0 : j - 1;
  double v_prevc, v_first, v_end;
  int64_t t_prevc, t_first, t_end_s;
  if constexpr (STR == 2 && !VMGPU_PIPE_NO_TV) {
    /* one b128 LDS read per boundary role instead of two scattered b64s */
    const vm_tv* tv = (const vm_tv*)ts;
    vm_tv a = tv[im1], b = tv[ii], c = tv[jm1];
    t_prevc = a.t;
    v_prevc = a.v;
    t_first = b.t;
    v_first = b.v;
    t_end_s = c.t;
    v_end = c.v;
  } else {
    v_prevc = vals[im1 * STR];
    t_prevc = ts[im1 * STR];
    v_first = vals[ii * STR];
    t_first = ts[ii * STR];
    v_end = vals[jm1 * STR];
    t_end_s = ts[jm1 * STR];
  }
  int n = j - i;
  bool has_prev = (i < count) && (i > 0) &&
                  (t_prevc > t_start - sw.max_prev_interval);
  double pv = has_prev ? v_prevc : v_first;
  int64_t ptm = has_prev ? t_prevc : t_first;
  double slope = (v_end - pv) / ((double)(t_end_s - ptm) / 1e3);
  double res_prev = (n == 0) ? 0.0 : slope;
  double res_nop = (n <= 1) ? vm_dnan() : slope;
  return has_prev ? res_prev : res_nop;
}

template <int FUNC_CT>
static VM_DEV uint64_t eval_grid_point_ij(const KPlan& p, const SeriesWindow& sw,
                                          const int64_t* ts, const double* vals,
                                          int count, int g, uint32_t s,
                                          const KIO& io, int i, int j) {
  int64_t t_end = p.start + (int64_t)g * p.step;
  int64_t t_start = t_end - sw.window;
  VmRfa r;
  r.window = sw.window;
  r.arg = p.arg;
  r.arg2 = p.arg2;
  r.prev_value = vm_dnan();
  r.prev_timestamp = t_start - sw.max_prev_interval;
  if (i < count && i > 0 && ts[i - 1] > r.prev_timestamp) {
    r.prev_value = vals[i - 1];
    r.prev_timestamp = ts[i - 1];
  }
  r.values = vals + i;
  r.timestamps = ts + i;
  r.n = j - i;
  r.real_prev_value = vm_dnan();
  if (i > 0) {
    int64_t curr = (r.n > 0) ? ts[i] : t_start;
    if (p.lookback_delta == 0 || (curr - ts[i - 1]) < p.lookback_delta)
      r.real_prev_value = vals[i - 1];
  }
  r.real_next_value = (j < count) ? vals[j] : vm_dnan();
  r.curr_timestamp = t_end;
  double v = vm_eval_rollup_fn(FUNC_CT >= 0 ? FUNC_CT : p.func, &r);
  (void)t_start;
  vm_emit_value(p, io, s, g, v);
  return (p.sspc > 0) ? (uint64_t)p.sspc : (uint64_t)(j - i);
}

template <int FUNC_CT>
static VM_DEV uint64_t eval_grid_point(const KPlan& p, const SeriesWindow& sw,
                                       const int64_t* ts, const double* vals,
                                       int count, int g, uint32_t s,
                                       const KIO& io) {
  int64_t t_end = p.start + (int64_t)g * p.step;
  int64_t t_start = t_end - sw.window;
  int i = vm_upper_bound(ts, count, t_start);
  int j = vm_upper_bound(ts, count, t_end);
  return eval_grid_point_ij<FUNC_CT>(p, sw, ts, vals, count, g, s, io, i, j);
}

/* ------------------------------------------------------------------ */
/* kernel 1: one wave per series (n <= CHUNK_WAVE)                    */
/* ------------------------------------------------------------------ */

template <int FUNC_CT, bool GACC = false, bool PREF = false>
__global__ __launch_bounds__(BLOCK_THREADS) void rollup_wave_kernel(KPlan p, KIO io) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int wave_in_block = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const size_t jbuf_bytes = vm_jbuf_bytes(p.jbuf_mode, p.jbuf_elems);
  const size_t wave_bytes = (size_t)p.chunk_wave * 16 + 256 + jbuf_bytes;
  int64_t* lts = (int64_t*)(smem + (size_t)wave_in_block * wave_bytes);
  double* lvs = (double*)(smem + (size_t)wave_in_block * wave_bytes +
                          (size_t)p.chunk_wave * 8);
  double* lscratch = (double*)(smem + (size_t)wave_in_block * wave_bytes +
                               (size_t)p.chunk_wave * 16);
  char* jbuf_region = smem + (size_t)wave_in_block * wave_bytes +
                      (size_t)p.chunk_wave * 16 + 256;
  /* mode-1 layout (u16 j-cache); in mode 2 the same base is re-carved by
   * the scatter path below */
  uint16_t* jbuf = (uint16_t*)jbuf_region;
  uint64_t scanned = 0;
  const uint32_t wave_id = blockIdx.x * WAVES_PER_BLOCK + wave_in_block;
  const uint32_t wave_stride = gridDim.x * WAVES_PER_BLOCK;

  /* GACC (compile-time variant, launched only for grouped rate plans with
   * n_grid <= 4*WAVE on a group-relayouted batch): per-point results
   * accumulate in registers across each group's contiguous series run and
   * flush with ONE atomic per (group, point).  Commutativity makes partial
   * runs safe: flushing is only needed on group change and at kernel end. */
  double accv[GACC ? 4 : 1], accc[GACC ? 4 : 1];
  int cur_grp = -1;
  double acc_init = 0.0;
  if constexpr (GACC) {
    acc_init = (p.aggr == VMGPU_AGGR_MIN) ? vm_dinf()
               : (p.aggr == VMGPU_AGGR_MAX) ? -vm_dinf()
               : (p.aggr == VMGPU_AGGR_GEOMEAN) ? 1.0 : 0.0;
#pragma unroll
    for (int u = 0; u < 4; u++) { accv[u] = acc_init; accc[u] = 0.0; }
  }
  auto acc_flush = [&]() {
    if constexpr (GACC) {
      if (cur_grp < 0) return;
#pragma unroll
      for (int u = 0; u < 4; u++) {
        int g = u * WAVE + lane;
        if (g < p.n_grid && accc[u] > 0.0) {
          double* gv = io.out + (size_t)cur_grp * (size_t)p.n_grid + (size_t)g;
          double* gc = io.out_counts + (size_t)cur_grp * (size_t)p.n_grid + (size_t)g;
          switch (p.aggr) {
            case VMGPU_AGGR_SUM:  atomicAdd(gv, accv[u]); *gc = 1.0; break;
            case VMGPU_AGGR_AVG:  atomicAdd(gv, accv[u]); atomicAdd(gc, accc[u]); break;
            case VMGPU_AGGR_MIN:  vm_atomic_min_f64(gv, accv[u]); *gc = 1.0; break;
            case VMGPU_AGGR_MAX:  vm_atomic_max_f64(gv, accv[u]); *gc = 1.0; break;
            case VMGPU_AGGR_COUNT:
            case VMGPU_AGGR_GROUP: atomicAdd(gv, accc[u]); *gc = 1.0; break;
            case VMGPU_AGGR_SUM2: atomicAdd(gv, accv[u]); *gc = 1.0; break;
            case VMGPU_AGGR_GEOMEAN:
              vm_atomic_mul_f64(gv, accv[u]); atomicAdd(gc, accc[u]); break;
            default: break;
          }
        }
        accv[u] = acc_init;
        accc[u] = 0.0;
      }
      cur_grp = -1;
    }
  };

  /* GACC: contiguous spans keep the group-relayouted runs within one wave;
   * otherwise strided (XCD-spread) assignment as before */
  uint32_t ws_begin, ws_end, ws_step;
  if constexpr (GACC) {
    uint32_t span = (io.n_sel + wave_stride - 1) / wave_stride;
    ws_begin = wave_id * span;
    ws_end = ws_begin + span;
    if (ws_end > io.n_sel) ws_end = io.n_sel;
    if (ws_begin > io.n_sel) ws_begin = io.n_sel;
    ws_step = 1;
  } else {
    ws_begin = wave_id;
    ws_end = io.n_sel;
    ws_step = wave_stride;
  }
  for (uint32_t ws = ws_begin; ws < ws_end; ws += ws_step) {
    uint32_t s = io.series_sel ? io.series_sel[ws] : ws;
    uint64_t lo = io.offsets[s];
    int64_t n = (int64_t)(io.offsets[s + 1] - lo);

#ifdef VMGPU_ABL_RAW_COPY
    int count = (int)n;
    for (int64_t k = lane; k < n; k += WAVE) {
      lts[k] = io.ts[lo + k];
      lvs[k] = io.vals[lo + k];
    }
    wave_lds_sync();
#ifndef VMGPU_ABL_NO_RCR
    if (p.rcr) rcr_scan_wave(lts, lvs, count, p.max_staleness, lane);
    wave_lds_sync();
#endif
#else
    int count = -1;
#ifndef VMGPU_ABL_NO_RCR
    if (p.rcr)
      count = load_rcr_fused_wave(io.ts + lo, io.vals + lo, n, lts, lvs,
                                  p.drop_stale != 0, p.max_staleness, lane);
#endif
    if (count < 0) {
      /* no rcr, or stale NaNs present: compact first, then scan */
      count = load_compact_wave(io.ts + lo, io.vals + lo, n, lts, lvs,
                                p.drop_stale != 0, lane);
      wave_lds_sync();
#ifndef VMGPU_ABL_NO_RCR
      if (p.rcr) rcr_scan_wave(lts, lvs, count, p.max_staleness, lane);
#endif
    }
    wave_lds_sync();
#endif
    /* compile-time gated: the inlined preFunc body costs ~10% on the hot
     * rate path when merely PRESENT (VGPR 32 -> 44), so it lives in its
     * own instantiation like GACC */
    if constexpr (PREF) {
      if (p.pre_func) pre_func_wave(lts, lvs, count, p.pre_func, lane);
    }

    int64_t si = p.step;
#ifndef VMGPU_ABL_NO_SCRAPE
    if (p.start < p.end) {
      if (io.series_si && count == (int)n) {
        int64_t c = io.series_si[s];
        si = c > 0 ? c : p.step;
      } else {
        si = scrape_interval_wave(lts, count, p.step, lane, lscratch);
      }
    }
#endif
    SeriesWindow sw = series_window(p, si);

    if (lane == 0) scanned += (uint64_t)count;
    /* window seek: a per-series linear time->index map gives each lane a
     * density-interpolated starting guess; vm_ub_hint's bounded walk
     * resolves it (binary-search fallback keeps irregular series exact). */
    double idx_per_ms = 0.0;
    int64_t ts0 = 0;
    if (count > 1) {
      ts0 = lts[0];
      int64_t span_ms = lts[count - 1] - ts0;
      idx_per_ms = span_ms > 0 ? (double)(count - 1) / (double)span_ms : 0.0;
    }
#if !defined(VMGPU_ABL_NO_SEEK) && !defined(VMGPU_ABL_NO_EVAL)
    if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
      /* Both fast paths exploit i(g) = j(g-dg) when the window is a step
       * multiple (the standard rate(m[5m]) @ 15s grid): the window START
       * boundary of point g is the window END boundary of point g-dg. */
      int dg64 = (sw.window > 0 && p.step > 0 && sw.window % p.step == 0)
                     ? (int)(sw.window / p.step) : 0;
      if (p.jbuf_mode == 2 && dg64 > 0 && count <= 65534 &&
          p.n_grid + dg64 <= p.jbuf_elems) {
        /* sample-scatter boundary map (J-only): each sample computes the
         * grid range where it is the last sample <= t_end (one boundary
         * per sample — t_hi(k) = t_lo(k+1)) and scatters index+1 into J
         * over the extended range [-dg, n_grid), replacing BOTH per-point
         * seeks; eval gathers values by index as before.  Exactly ub()
         * semantics: J[g] = #samples <= t_end(g).  (A fatter variant also
         * caching (ts, value) rows in LDS measured SLOWER — the extra
         * 4.6 KB/wave of LDS cost 3 waves/SIMD of occupancy.) */
        uint16_t* Jb = (uint16_t*)jbuf_region;
        const int ext = p.n_grid + dg64;
        for (int e = lane; e < ext; e += WAVE) Jb[e] = 0;
        wave_lds_sync();
        const double inv_step = 1.0 / (double)p.step;
        for (int base = 0; base < count; base += WAVE) {
          int k = base + lane;
          bool active = k < count;
          int64_t t_k = active ? lts[k] : 0;
          int64_t t_n = (active && k + 1 < count) ? lts[k + 1] : 0;
          /* g_lo = ceil((t_k - start)/step): float estimate, exact int fixup */
          int g_lo = (int)floor((double)(t_k - p.start) * inv_step) - 1;
          g_lo += (p.start + (int64_t)g_lo * p.step < t_k);
          g_lo += (p.start + (int64_t)g_lo * p.step < t_k);
          g_lo += (p.start + (int64_t)g_lo * p.step < t_k);
          int g_hi;
          if (k + 1 < count) {
            g_hi = (int)floor((double)(t_n - p.start) * inv_step) - 1;
            g_hi += (p.start + (int64_t)g_hi * p.step < t_n);
            g_hi += (p.start + (int64_t)g_hi * p.step < t_n);
            g_hi += (p.start + (int64_t)g_hi * p.step < t_n);
          } else {
            g_hi = p.n_grid;
          }
          if (g_lo < -dg64) g_lo = -dg64;
          if (g_hi > p.n_grid) g_hi = p.n_grid;
          if (active) {
            for (int g = g_lo; g < g_hi; g++) Jb[g + dg64] = (uint16_t)(k + 1);
          }
        }
        wave_lds_sync();
        for (int g0 = 0; g0 < p.n_grid; g0 += 4 * WAVE) {
#pragma unroll
          for (int u = 0; u < 4; u++) {
            int g = g0 + u * WAVE + lane;
            if (g < p.n_grid) {
              int64_t t_end = p.start + (int64_t)g * p.step;
              int64_t t_start = t_end - sw.window;
              int j = Jb[g + dg64];
              int i = Jb[g];                     /* = J[(g-dg)+dg] */
              vm_emit_value(p, io, s, g,
                            eval_rate_fused(p, sw, lts, lvs, count, i, j, t_start));
              scanned += 2;
            }
          }
        }
        wave_lds_sync();
        continue;
      }
      if (p.jbuf_mode >= 1 &&
          (size_t)p.n_grid * 2 <= vm_jbuf_bytes(p.jbuf_mode, p.jbuf_elems) &&
          dg64 > 0 && count <= 65535) {
        bool use_acc = false;
        if constexpr (GACC) {
          int sgrp = io.group_ids ? io.group_ids[s] : -1;
          use_acc = (sgrp >= 0);
          if (use_acc && sgrp != cur_grp) {
            acc_flush();
            cur_grp = sgrp;
          }
        }
        for (int g = lane; g < p.n_grid; g += WAVE) {
          int64_t t_end = p.start + (int64_t)g * p.step;
          int gj = (int)((double)(t_end - ts0) * idx_per_ms) + 1;
          jbuf[g] = (uint16_t)vm_ub_hint_fast(lts, count, t_end, gj);
        }
        wave_lds_sync();
        for (int g0 = 0; g0 < p.n_grid; g0 += 4 * WAVE) {
#pragma unroll
          for (int u = 0; u < 4; u++) {
            int g = g0 + u * WAVE + lane;
            if (g < p.n_grid) {
              int64_t t_end = p.start + (int64_t)g * p.step;
              int64_t t_start = t_end - sw.window;
              int j = jbuf[g];
              int i;
              if (g >= dg64) {
                i = jbuf[g - dg64];
              } else {
                int gi = (int)((double)(t_start - ts0) * idx_per_ms) + 1;
                i = vm_ub_hint_fast(lts, count, t_start, gi);
              }
              double v = eval_rate_fused(p, sw, lts, lvs, count, i, j, t_start);
              if (GACC && use_acc) {
                if constexpr (GACC) {
                  if (!vm_isnan(v)) {
                    switch (p.aggr) {
                      case VMGPU_AGGR_SUM:
                      case VMGPU_AGGR_AVG:  accv[u] += v; break;
                      case VMGPU_AGGR_MIN:
                        accv[u] = (v < accv[u]) ? v : accv[u]; break;
                      case VMGPU_AGGR_MAX:
                        accv[u] = (v > accv[u]) ? v : accv[u]; break;
                      case VMGPU_AGGR_SUM2: accv[u] += v * v; break;
                      case VMGPU_AGGR_GEOMEAN: accv[u] *= v; break;
                      default: break; /* COUNT/GROUP count only */
                    }
                    accc[u] += 1.0;
                  }
                }
              } else {
                vm_emit_value(p, io, s, g, v);
              }
              scanned += 2;
            }
          }
        }
        wave_lds_sync();
        continue;
      }
    }
#endif
    for (int g0 = 0; g0 < p.n_grid; g0 += 4 * WAVE) {
      /* four grid points per lane per outer iteration: independent seek and
       * eval chains that the scheduler interleaves (the phases are LDS-
       * latency chains, not bandwidth-bound) */
#pragma unroll
      for (int u = 0; u < 4; u++) {
        int g = g0 + u * WAVE + lane;
        if (g < p.n_grid) {
          int64_t t_end = p.start + (int64_t)g * p.step;
          int64_t t_start = t_end - sw.window;
          int gi = (int)((double)(t_start - ts0) * idx_per_ms) + 1;
          int gj = (int)((double)(t_end - ts0) * idx_per_ms) + 1;
#ifdef VMGPU_ABL_NO_SEEK
          int i = gi < 0 ? 0 : (gi > count ? count : gi);
          int j = gj < 0 ? 0 : (gj > count ? count : gj);
          if (j < i) j = i;
#else
          int i, j;
          if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
            i = vm_ub_hint_fast(lts, count, t_start, gi);
            j = vm_ub_hint_fast(lts, count, t_end, gj);
          } else {
            i = vm_ub_hint(lts, count, t_start, gi);
            j = vm_ub_hint(lts, count, t_end, gj);
          }
#endif
#ifdef VMGPU_ABL_NO_EVAL
          io.out[(size_t)s * (size_t)p.n_grid + (size_t)g] = (j > 0 && j <= count) ? lvs[j - 1] : 0.0;
          scanned += 2;
#else
          if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
            vm_emit_value(p, io, s, g,
                          eval_rate_fused(p, sw, lts, lvs, count, i, j, t_start));
            scanned += 2; /* samplesScannedPerCall for rate/deriv_fast */
          } else {
            scanned += eval_grid_point_ij<FUNC_CT>(p, sw, lts, lvs, count, g, s, io, i, j);
          }
#endif
        }
      }
    }
    wave_lds_sync();
  }
  acc_flush();
  /* reduce samplesScanned: wave shuffle + one atomic per wave */
  for (int d = 32; d > 0; d >>= 1) scanned += __shfl_down((unsigned long long)scanned, d);
  if (lane == 0 && scanned) atomicAdd(io.samples_scanned, (unsigned long long)scanned);
}

/* ------------------------------------------------------------------ */
/* kernel 1b: software-pipelined register-staged wave kernel          */
/*                                                                    */
/* The wave kernel above is instruction-ISSUE/latency bound, not HBM  */
/* bound (profiles/round1_final.md §5: FETCH/WRITE ≈ algorithmic but  */
/* ACTIVE_INST_ANY ≈ 30% at 7 waves/SIMD).  This variant restructures */
/* the per-series phases so the memory system never drains:           */
/*   - the series' columns are loaded straight into REGISTERS (up to  */
/*     PIPE_CHUNKS coalesced b64 loads per column in flight at once), */
/*   - the removeCounterResets scan consumes those registers while    */
/*     writing the corrected columns to LDS (no LDS re-read pass),    */
/*   - the NEXT series' loads are issued as soon as the registers die,*/
/*     overlapping scrape/seek/eval/emit of the current series,       */
/*   - every phase barrier is lgkm-only (wave_ds_sync), so the        */
/*     prefetched loads stay outstanding across the whole evaluation. */
/* Results are bit-identical to the wave kernel: same scan, same      */
/* j-cache seeks, same evaluators.                                    */
/* ------------------------------------------------------------------ */

#define PIPE_CHUNKS 4 /* series cap = PIPE_CHUNKS*WAVE = 256 samples */

/* Pair-rounds removeCounterResets over an LDS/global COLUMN (stride STR
 * in int64 units): the rcr_scan_pairs logic with per-round pair loads
 * instead of register sources — 128 samples per round instead of 64, so
 * the serial-round chain halves.  Bit-exact with rcr_scan_wave (same
 * element-order event walk, same segmented-max clamp). */
template <int STR = 1>
static __device__ void rcr_scan_col_pairs(int64_t* d_ts, double* d_vals,
                                          int count, int64_t msi, int lane) {
  double corr = 0.0;
  double prev_raw = 0.0;
  int64_t prev_ts = 0;
  double prev_fin = 0.0;
  for (int base = 0; base < count; base += 2 * WAVE) {
    int k0 = base + 2 * lane;
    int k1 = k0 + 1;
    bool a0 = k0 < count;
    bool a1 = k1 < count;
    double v0 = a0 ? d_vals[k0 * STR] : 0.0;
    double v1 = a1 ? d_vals[k1 * STR] : 0.0;
    int64_t t0 = a0 ? d_ts[k0 * STR] : 0;
    int64_t t1 = a1 ? d_ts[k1 * STR] : 0;
    int rem = count - base;
    if (rem > 2 * WAVE) rem = 2 * WAVE;
    int lastl = (rem - 1) >> 1;
    bool last_is_e1 = ((rem - 1) & 1) != 0;
    double pv0 = __shfl_up(v1, 1);
    int64_t pt0 = __shfl_up(t1, 1);
    if (lane == 0) { pv0 = prev_raw; pt0 = prev_ts; }
    bool isfirst0 = (k0 == 0);
    double d0 = v0 - pv0;
    double d1 = v1 - v0;
    double inc0 = 0.0, inc1 = 0.0;
    if (!isfirst0 && d0 < 0) inc0 = ((-d0 * 8) < pv0) ? (pv0 - v0) : pv0;
    if (d1 < 0) inc1 = ((-d1 * 8) < v0) ? (v0 - v1) : v0;
    bool gap0 = (!isfirst0 && msi > 0 && (t0 - pt0) > msi);
    bool gap1 = (msi > 0 && (t1 - t0) > msi);
    bool evt0 = a0 && (gap0 || inc0 != 0.0);
    bool evt1 = a1 && (gap1 || inc1 != 0.0);
    uint64_t em = __ballot(evt0 || evt1);
    uint64_t dm = __ballot((a0 && !isfirst0 && d0 < 0) || (a1 && d1 < 0));
    if (em == 0 && dm == 0 &&
        (base == 0 || __shfl(v0, 0) + corr >= prev_fin)) {
      if (corr != 0.0) {
        if (a0) d_vals[k0 * STR] = v0 + corr;
        if (a1) d_vals[k1 * STR] = v1 + corr;
      }
      double lv = last_is_e1 ? v1 : v0;
      int64_t lt = last_is_e1 ? t1 : t0;
      prev_raw = __shfl(lv, lastl);
      prev_ts = __shfl(lt, lastl);
      prev_fin = prev_raw + corr;
      continue;
    }
    double cc = corr;
    double mc0 = corr, mc1 = corr;
    uint64_t w = em;
    while (w) {
      int b = __ffsll((unsigned long long)w) - 1;
      w &= w - 1;
      double i0b = __shfl(inc0, b);
      double i1b = __shfl(inc1, b);
      int fl = __shfl((int)gap0 | ((int)gap1 << 1) | ((int)evt0 << 2) |
                          ((int)evt1 << 3), b);
      if (fl & 4) {
        double cn = (fl & 1) ? 0.0 : (cc + i0b);
        if (lane >= b) { mc0 = cn; mc1 = cn; }
        cc = cn;
      }
      if (fl & 8) {
        double cn = (fl & 2) ? 0.0 : (cc + i1b);
        if (lane > b) mc0 = cn;
        if (lane >= b) mc1 = cn;
        cc = cn;
      }
    }
    double fin0 = v0 + mc0;
    double fin1 = v1 + mc1;
    bool f0 = isfirst0 || gap0;
    bool f1 = gap1;
    double x0 = fin0, x1 = fin1;
    if (lane == 0 && !f0) x0 = fmax(x0, prev_fin);
    double pm = f1 ? x1 : fmax(x0, x1);
    int pf = (f0 || f1) ? 1 : 0;
    for (int dlt = 1; dlt < WAVE; dlt <<= 1) {
      double pmo = __shfl_up(pm, dlt);
      int pfo = __shfl_up(pf, dlt);
      if (lane >= dlt) {
        if (!pf) pm = fmax(pm, pmo);
        pf |= pfo;
      }
    }
    double inc_m = __shfl_up(pm, 1);
    if (lane > 0 && !f0) x0 = fmax(x0, inc_m);
    if (!f1) x1 = fmax(x1, x0);
    if (a0) d_vals[k0 * STR] = x0;
    if (a1) d_vals[k1 * STR] = x1;
    corr = cc;
    {
      double lvr = last_is_e1 ? v1 : v0;
      int64_t ltr = last_is_e1 ? t1 : t0;
      double lxf = last_is_e1 ? x1 : x0;
      prev_raw = __shfl(lvr, lastl);
      prev_ts = __shfl(ltr, lastl);
      prev_fin = __shfl(lxf, lastl);
    }
  }
}

#define PIPE_PCHUNKS 2 /* pair chunks: 2 x 128 samples = 256 cap */

/* Pair-register removeCounterResets: lane l holds samples (2l, 2l+1) of
 * each 128-sample chunk (ONE b128 global load per column per chunk).
 * Exactly the rcr_scan_wave semantics (rollup.go:921-958): the running
 * correction visits the (rare) event ELEMENTS in index order bitwise, and
 * the monotonic clamp is the same segmented prefix max, evaluated as
 * in-lane pair combines + the cross-lane scan (fmax is exactly
 * associative).  Writes the pair-interleaved LDS slab; returns n, or -1
 * when a stale NaN must be compacted out (caller falls back to the global
 * compact path). */
static VM_DEV int rcr_scan_pairs(const int64_t* t0a, const int64_t* t1a,
                                 const double* v0a, const double* v1a,
                                 int64_t n, int64_t* d_ts, double* d_vals,
                                 bool drop_stale, int64_t msi, int lane) {
  double corr = 0.0;
  double prev_raw = 0.0;
  int64_t prev_ts = 0;
  double prev_fin = 0.0;
#pragma unroll
  for (int c = 0; c < PIPE_PCHUNKS; c++) {
    int64_t base = (int64_t)c * 2 * WAVE;
    if (base >= n) continue;
    int64_t k0 = base + 2 * lane;
    int64_t k1 = k0 + 1;
    bool a0 = k0 < n;
    bool a1 = k1 < n;
    double v0 = a0 ? v0a[c] : 0.0;
    double v1 = a1 ? v1a[c] : 0.0;
    int64_t t0 = a0 ? t0a[c] : 0;
    int64_t t1 = a1 ? t1a[c] : 0;
    if (drop_stale &&
        __ballot((a0 && vm_is_stale_nan(v0)) ||
                 (a1 && vm_is_stale_nan(v1))) != 0)
      return -1;
    if (a0) d_ts[k0 * 2] = t0;
    if (a1) d_ts[k1 * 2] = t1;
    int rem = (int)(n - base);
    if (rem > 2 * WAVE) rem = 2 * WAVE;
    int lastl = (rem - 1) >> 1;
    bool last_is_e1 = ((rem - 1) & 1) != 0;
    /* previous element: elem0's is the neighbor lane's elem1; elem1's is
     * the in-lane elem0 (no cross-lane op) */
    double pv0 = __shfl_up(v1, 1);
    int64_t pt0 = __shfl_up(t1, 1);
    if (lane == 0) { pv0 = prev_raw; pt0 = prev_ts; }
    bool isfirst0 = (k0 == 0);
    double d0 = v0 - pv0;
    double d1 = v1 - v0;
    double inc0 = 0.0, inc1 = 0.0;
    if (!isfirst0 && d0 < 0) inc0 = ((-d0 * 8) < pv0) ? (pv0 - v0) : pv0;
    if (d1 < 0) inc1 = ((-d1 * 8) < v0) ? (v0 - v1) : v0;
    bool gap0 = (!isfirst0 && msi > 0 && (t0 - pt0) > msi);
    bool gap1 = (msi > 0 && (t1 - t0) > msi);
    bool evt0 = a0 && (gap0 || inc0 != 0.0);
    bool evt1 = a1 && (gap1 || inc1 != 0.0);
    uint64_t em = __ballot(evt0 || evt1);
    uint64_t dm = __ballot((a0 && !isfirst0 && d0 < 0) || (a1 && d1 < 0));
    if (em == 0 && dm == 0 &&
        (base == 0 || __shfl(v0, 0) + corr >= prev_fin)) {
      if (a0) d_vals[k0 * 2] = v0 + corr;
      if (a1) d_vals[k1 * 2] = v1 + corr;
      double lv = last_is_e1 ? v1 : v0;
      int64_t lt = last_is_e1 ? t1 : t0;
      prev_raw = __shfl(lv, lastl);
      prev_ts = __shfl(lt, lastl);
      prev_fin = prev_raw + corr;
      continue;
    }
    /* correction walk in exact element order (elem0 of lane b, then its
     * elem1) — identical serial sum to the reference */
    double cc = corr;
    double mc0 = corr, mc1 = corr;
    uint64_t w = em;
    while (w) {
      int b = __ffsll((unsigned long long)w) - 1;
      w &= w - 1;
      double i0b = __shfl(inc0, b);
      double i1b = __shfl(inc1, b);
      int fl = __shfl((int)gap0 | ((int)gap1 << 1) | ((int)evt0 << 2) |
                          ((int)evt1 << 3), b);
      if (fl & 4) { /* event at element 0 of lane b: applies to k >= 2b */
        double cn = (fl & 1) ? 0.0 : (cc + i0b);
        if (lane >= b) { mc0 = cn; mc1 = cn; }
        cc = cn;
      }
      if (fl & 8) { /* event at element 1 of lane b: applies to k >= 2b+1 */
        double cn = (fl & 2) ? 0.0 : (cc + i1b);
        if (lane > b) mc0 = cn;
        if (lane >= b) mc1 = cn;
        cc = cn;
      }
    }
    double fin0 = v0 + mc0;
    double fin1 = v1 + mc1;
    /* monotonic clamp = segmented prefix max; boundary elements (series
     * start / staleness gaps) keep their value and start a new segment */
    bool f0 = isfirst0 || gap0;
    bool f1 = gap1;
    double x0 = fin0, x1 = fin1;
    if (lane == 0 && !f0) x0 = fmax(x0, prev_fin);
    double pm = f1 ? x1 : fmax(x0, x1);
    int pf = (f0 || f1) ? 1 : 0;
    for (int dlt = 1; dlt < WAVE; dlt <<= 1) {
      double pmo = __shfl_up(pm, dlt);
      int pfo = __shfl_up(pf, dlt);
      if (lane >= dlt) {
        if (!pf) pm = fmax(pm, pmo);
        pf |= pfo;
      }
    }
    double inc_m = __shfl_up(pm, 1); /* exclusive incoming (lane 0 unused) */
    if (lane > 0 && !f0) x0 = fmax(x0, inc_m);
    if (!f1) x1 = fmax(x1, x0);
    if (a0) d_vals[k0 * 2] = x0;
    if (a1) d_vals[k1 * 2] = x1;
    corr = cc;
    {
      double lvr = last_is_e1 ? v1 : v0;
      int64_t ltr = last_is_e1 ? t1 : t0;
      double lxf = last_is_e1 ? x1 : x0;
      prev_raw = __shfl(lvr, lastl);
      prev_ts = __shfl(ltr, lastl);
      prev_fin = __shfl(lxf, lastl);
    }
  }
  return (int)n;
}

/* VMGPU_PIPE_MINWAVES forces an occupancy floor (waves/SIMD) on the pipe
 * kernel for A/B builds; VMGPU_PIPE_UNROLL (2 or 4) sets the eval ILP. */
#ifndef VMGPU_PIPE_UNROLL
#define VMGPU_PIPE_UNROLL 4
#endif

template <int FUNC_CT, bool GROUPED>
__global__
#ifdef VMGPU_PIPE_MINWAVES
__launch_bounds__(BLOCK_THREADS, VMGPU_PIPE_MINWAVES)
#else
__launch_bounds__(BLOCK_THREADS)
#endif
void rollup_pipe_kernel(KPlan p, KIO io) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int wave_in_block = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const size_t jbuf_bytes = vm_jbuf_bytes(p.jbuf_mode, p.jbuf_elems);
  const size_t wave_bytes = (size_t)p.chunk_wave * 16 + 256 + jbuf_bytes;
  /* PAIR-interleaved series slab: sample k = 16-byte {t, v} at offset 16k
   * (stride template parameter STR=2 on every accessor) — the eval
   * gathers fetch one b128 LDS read per boundary role instead of two
   * scattered b64 reads */
  int64_t* lts = (int64_t*)(smem + (size_t)wave_in_block * wave_bytes);
  double* lvs = (double*)(smem + (size_t)wave_in_block * wave_bytes + 8);
  double* lscratch = (double*)(smem + (size_t)wave_in_block * wave_bytes +
                               (size_t)p.chunk_wave * 16);
  uint16_t* jbuf = (uint16_t*)(smem + (size_t)wave_in_block * wave_bytes +
                               (size_t)p.chunk_wave * 16 + 256);
  uint64_t scanned = 0;
  const uint32_t wave_id = blockIdx.x * WAVES_PER_BLOCK + wave_in_block;
  const uint32_t wave_stride = gridDim.x * WAVES_PER_BLOCK;

  /* pair staging registers: lane l holds samples (2l, 2l+1) of each
   * 128-sample chunk, loaded with ONE b128 per column per chunk */
  int64_t rt0[PIPE_PCHUNKS], rt1[PIPE_PCHUNKS];
  double rv0[PIPE_PCHUNKS], rv1[PIPE_PCHUNKS];

  /* J-scatter mode (A/B builds only: -DVMGPU_PIPE_SCATTER): plan-uniform
   * window that is a step multiple + an ext jbuf (jbuf_mode 2).  The
   * staging scan then builds the full upper-bound map J[g+sdg] =
   * #samples <= t_end(g) and the eval phase does no seeks at all.
   * Measured SLOWER than the probe j-cache at config 2 (the divergent
   * range writes sit on the serial scan chain) — compiled out so the
   * default kernel does not pay its register pressure. */
#ifdef VMGPU_PIPE_SCATTER
  int scat_dg = 0;
  if (p.jbuf_mode == 2 && p.window > 0 && p.step > 0 &&
      p.window % p.step == 0) {
    int64_t d = p.window / p.step;
    if (d > 0 && p.n_grid + d <= p.jbuf_elems) scat_dg = (int)d;
  }
  const double inv_gstep = 1.0 / (double)p.step;
#else
  const int scat_dg = 0;
  const double inv_gstep = 0.0;
#endif

  /* wave_id is wave-uniform by construction; readfirstlane makes that
   * provable, so the per-series descriptor reads below become scalar
   * (s_load, lgkm-counted) instead of divergent vector loads, and the
   * staging loads below are unconditional with clamped indices — one
   * basic block, so the compiler can track outstanding vmcnt precisely
   * instead of falling back to vmcnt(0) at every use. */
  uint32_t ws = __builtin_amdgcn_readfirstlane(wave_id);
  uint32_t s = 0;
  uint64_t lo = 0;
  int64_t n = 0;
  int64_t si_raw = 0;
  if (ws < io.n_sel) {
    s = __builtin_amdgcn_readfirstlane(io.series_sel ? io.series_sel[ws] : ws);
    lo = io.offsets[s];
    n = (int64_t)(io.offsets[s + 1] - lo);
    if (io.series_si) si_raw = io.series_si[s];
    if (n > 0 && (lo & 1) == 0) {
      const int64_t* gts = io.ts + lo;
      const double* gvs = io.vals + lo;
      const int64_t nm2 = (n - 1) & ~(int64_t)1;
#pragma unroll
      for (int c = 0; c < PIPE_PCHUNKS; c++) {
        int64_t k = (int64_t)c * 2 * WAVE + 2 * lane;
        if (k > nm2) k = nm2; /* batch columns carry 16 B slack for this */
#ifdef VMGPU_PIPE_NT
        typedef long long v2i64 __attribute__((ext_vector_type(2)));
        typedef double v2f64 __attribute__((ext_vector_type(2)));
        v2i64 tpv = __builtin_nontemporal_load((const v2i64*)(gts + k));
        v2f64 vpv = __builtin_nontemporal_load((const v2f64*)(gvs + k));
        vm_i64x2 tp = {tpv.x, tpv.y};
        double2 vp = {vpv.x, vpv.y};
#else
        vm_i64x2 tp = *(const vm_i64x2*)(gts + k);
        double2 vp = *(const double2*)(gvs + k);
#endif
        rt0[c] = tp.x;
        rt1[c] = tp.y;
        rv0[c] = vp.x;
        rv1[c] = vp.y;
      }
    }
  }
  while (ws < io.n_sel) {
    const uint32_t ws_n = ws + wave_stride;
    const bool has_next = ws_n < io.n_sel;
    uint32_t s_n = 0;
    uint64_t lo_n = 0;
    int64_t n_n = 0;
    int64_t si_raw_n = 0;
    if (has_next) {
      s_n = __builtin_amdgcn_readfirstlane(
          io.series_sel ? io.series_sel[ws_n] : ws_n);
      lo_n = io.offsets[s_n];
      n_n = (int64_t)(io.offsets[s_n + 1] - lo_n);
      if (io.series_si) si_raw_n = io.series_si[s_n];
    }
    /* J-scatter init: zero = "no sample <= t_end" for positions before the
     * first sample's range (upper bound 0) */
    if (scat_dg > 0) {
      const int ext = p.n_grid + scat_dg;
      for (int e = lane; e < ext; e += WAVE) jbuf[e] = 0;
      wave_ds_sync();
    }
    /* stage the current series into LDS from registers (rcr fused);
     * odd-offset series (no 16-B-aligned pair loads) fall to the global
     * compact path below */
    int count = -1;
    const bool pair_ok = ((lo & 1) == 0);
#ifdef VMGPU_PIPE_ABL_NO_SCAN
    if (false) {
#else
    if (p.rcr && pair_ok) {
#endif
      count = rcr_scan_pairs(rt0, rt1, rv0, rv1, n, lts, lvs,
                             p.drop_stale != 0, p.max_staleness, lane);
    } else if (pair_ok) {
      bool stale = false;
      if (p.drop_stale) {
#pragma unroll
        for (int c = 0; c < PIPE_PCHUNKS; c++) {
          int64_t k = (int64_t)c * 2 * WAVE + 2 * lane;
          if (__ballot((k < n && vm_is_stale_nan(rv0[c])) ||
                       (k + 1 < n && vm_is_stale_nan(rv1[c]))) != 0)
            stale = true;
        }
      }
      if (!stale) {
#pragma unroll
        for (int c = 0; c < PIPE_PCHUNKS; c++) {
          int64_t k = (int64_t)c * 2 * WAVE + 2 * lane;
          if (k < n) { lts[k * 2] = rt0[c]; lvs[k * 2] = rv0[c]; }
          if (k + 1 < n) {
            lts[(k + 1) * 2] = rt1[c];
            lvs[(k + 1) * 2] = rv1[c];
          }
        }
        count = (int)n;
      }
    }
    /* the staging registers are dead now — prefetch the NEXT series into
     * them; these loads stay in flight across scrape/seek/eval/emit */
    if (has_next && n_n > 0 && (lo_n & 1) == 0) {
      const int64_t* gts = io.ts + lo_n;
      const double* gvs = io.vals + lo_n;
      const int64_t nm2 = (n_n - 1) & ~(int64_t)1;
#pragma unroll
      for (int c = 0; c < PIPE_PCHUNKS; c++) {
        int64_t k = (int64_t)c * 2 * WAVE + 2 * lane;
        if (k > nm2) k = nm2;
#ifdef VMGPU_PIPE_NT
        typedef long long v2i64 __attribute__((ext_vector_type(2)));
        typedef double v2f64 __attribute__((ext_vector_type(2)));
        v2i64 tpv = __builtin_nontemporal_load((const v2i64*)(gts + k));
        v2f64 vpv = __builtin_nontemporal_load((const v2f64*)(gvs + k));
        vm_i64x2 tp = {tpv.x, tpv.y};
        double2 vp = {vpv.x, vpv.y};
#else
        vm_i64x2 tp = *(const vm_i64x2*)(gts + k);
        double2 vp = *(const double2*)(gvs + k);
#endif
        rt0[c] = tp.x;
        rt1[c] = tp.y;
        rv0[c] = vp.x;
        rv1[c] = vp.y;
      }
    }
    if (count < 0) {
      /* stale NaNs present or odd-offset series (rare): from global */
      count = load_compact_wave<2>(io.ts + lo, io.vals + lo, n, lts, lvs,
                                   p.drop_stale != 0, lane);
      wave_ds_sync();
      if (p.rcr) rcr_scan_wave<2>(lts, lvs, count, p.max_staleness, lane);
      if (scat_dg > 0) {
        /* the register scatter saw pre-compaction indices: rebuild J by
         * direct binary search over the compacted column */
        wave_ds_sync();
        const int ext = p.n_grid + scat_dg;
        for (int e = lane; e < ext; e += WAVE) {
          int64_t t_end = p.start + (int64_t)(e - scat_dg) * p.step;
          jbuf[e] = (uint16_t)vm_upper_bound<2>(lts, count, t_end);
        }
      }
    }
    wave_ds_sync();

#ifdef VMGPU_PIPE_ABL_STAGE
    /* stage-only ablation: keep the output write traffic, skip the rest */
    for (int g = lane; g < p.n_grid; g += WAVE) {
      int gc = g < count ? g : (count > 0 ? count - 1 : 0);
      io.out[(size_t)s * (size_t)p.n_grid + (size_t)g] =
          count ? lvs[gc * 2] : 0.0;
    }
    wave_ds_sync();
    ws = ws_n;
    s = s_n;
    lo = lo_n;
    n = n_n;
    si_raw = si_raw_n;
    continue;
#endif

    int64_t si = p.step;
#ifndef VMGPU_ABL_NO_SCRAPE
    if (p.start < p.end) {
      if (io.series_si && count == (int)n) {
        si = si_raw > 0 ? si_raw : p.step;
      } else {
        si = scrape_interval_wave_t<true, 2>(lts, count, p.step, lane, lscratch);
      }
    }
#endif
    SeriesWindow sw = series_window(p, si);
    if (lane == 0) scanned += (uint64_t)count;

    double idx_per_ms = 0.0;
    int64_t ts0 = 0;
    if (count > 1) {
      ts0 = lts[0];
      int64_t span_ms = lts[(count - 1) * 2] - ts0;
      idx_per_ms = span_ms > 0 ? (double)(count - 1) / (double)span_ms : 0.0;
    }
    /* emit: GROUPED folds the aggregate switch in with the per-series
     * group row hoisted out of the point loop; the ungrouped build is a
     * plain coalesced row store (no per-point branch, no switch). */
    double* out_row = io.out + (size_t)s * (size_t)p.n_grid;
    double* grp_vrow = nullptr;
    double* grp_crow = nullptr;
    if constexpr (GROUPED) {
      int grp = io.group_ids ? io.group_ids[s] : -1;
      if (grp >= 0) {
        grp_vrow = io.out + (size_t)grp * (size_t)p.n_grid;
        grp_crow = io.out_counts + (size_t)grp * (size_t)p.n_grid;
      }
    }
    auto emit = [&](int g, double v) {
      if constexpr (GROUPED) {
        if (grp_vrow && !vm_isnan(v)) {
          double* gv = grp_vrow + g;
          double* gc = grp_crow + g;
          switch (p.aggr) {
            case VMGPU_AGGR_SUM: atomicAdd(gv, v); *gc = 1.0; break;
            case VMGPU_AGGR_AVG: atomicAdd(gv, v); atomicAdd(gc, 1.0); break;
            case VMGPU_AGGR_MIN: vm_atomic_min_f64(gv, v); *gc = 1.0; break;
            case VMGPU_AGGR_MAX: vm_atomic_max_f64(gv, v); *gc = 1.0; break;
            case VMGPU_AGGR_COUNT:
            case VMGPU_AGGR_GROUP: atomicAdd(gv, 1.0); *gc = 1.0; break;
            case VMGPU_AGGR_SUM2: atomicAdd(gv, v * v); *gc = 1.0; break;
            case VMGPU_AGGR_GEOMEAN:
              vm_atomic_mul_f64(gv, v); atomicAdd(gc, 1.0); break;
            default: break;
          }
        }
      } else {
#ifdef VMGPU_PIPE_NT
        __builtin_nontemporal_store(v, out_row + g);
#else
        out_row[g] = v;
#endif
      }
    };
    const int64_t t_step_wave = (int64_t)WAVE * p.step;
    bool done = false;
    if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
      if (scat_dg > 0) {
        /* scatter-J eval: J was built during staging; no seeks at all. */
        int gs = lane;
        int64_t te_s = p.start + (int64_t)lane * p.step;
        for (; gs < p.n_grid; gs += WAVE, te_s += t_step_wave) {
          int j = jbuf[gs + scat_dg];
          int i = jbuf[gs];
#ifdef VMGPU_PIPE_ABL_NO_EVAL
          emit(gs, (j > 0 && j <= count) ? lvs[(j - 1) * 2] : 0.0);
          (void)i;
#else
          emit(gs, eval_rate_fused<2>(p, sw, lts, lvs, count, i, j,
                                      te_s - sw.window));
#endif
        }
        done = true;
      }
      int dg64 = (sw.window > 0 && p.step > 0 && sw.window % p.step == 0)
                     ? (int)(sw.window / p.step) : 0;
      if (!done && p.jbuf_mode >= 1 &&
          (size_t)p.n_grid * 2 <= vm_jbuf_bytes(p.jbuf_mode, p.jbuf_elems) &&
          dg64 > 0 && count <= 65535) {
        if (dg64 < WAVE) {
          /* FUSED seek+eval: with dg < 64 the window-START boundary
           * j(g - dg) lives in THIS wave's just-computed j registers (or
           * the previous round's) — one shuffle replaces the whole jbuf
           * write/sync/read round trip.  Head lanes of round 0 (window
           * start before the grid) probe for i directly. */
          int j_prev = 0;
          int g = lane;
          int64_t te = p.start + (int64_t)lane * p.step;
          for (int r = 0; r * WAVE < p.n_grid;
               r++, g += WAVE, te += t_step_wave) {
            bool act = g < p.n_grid;
            int gj = (int)((double)(te - ts0) * idx_per_ms) + 1;
            int j_cur = act
                            ? vm_ub_hint_fast<2>(lts, count, te, gj)
                            : count;
            int i;
            int i_same = __shfl(j_cur, lane - dg64);
            int i_prev = __shfl(j_prev, lane + WAVE - dg64);
            i = (lane >= dg64) ? i_same : i_prev;
            if (r == 0 && lane < dg64) {
              int64_t t_start = te - sw.window;
              int gi = (int)((double)(t_start - ts0) * idx_per_ms) + 1;
              i = vm_ub_hint_fast<2>(lts, count, t_start, gi);
            }
            if (act) {
#ifdef VMGPU_PIPE_ABL_NO_EVAL
              emit(g, (j_cur > 0 && j_cur <= count) ? lvs[(j_cur - 1) * 2] : 0.0);
              (void)i;
#else
              emit(g, eval_rate_fused<2>(p, sw, lts, lvs, count, i, j_cur,
                                         te - sw.window));
#endif
            }
            j_prev = j_cur;
          }
          done = true;
        }
        if (!done) {
        /* j-cache fill: full lane-blocks without bounds checks, one masked
         * tail; t_end is strength-reduced (+64*step per round) */
        {
          int gf = lane;
          int64_t te = p.start + (int64_t)lane * p.step;
          const int fill_full = (p.n_grid / WAVE) * WAVE;
          for (; gf < fill_full; gf += WAVE, te += t_step_wave) {
            int gj = (int)((double)(te - ts0) * idx_per_ms) + 1;
#ifdef VMGPU_PIPE_WALK_PROBE
            jbuf[gf] = (uint16_t)vm_ub_hint<2>(lts, count, te, gj);
#else
            jbuf[gf] = (uint16_t)vm_ub_hint_fast<2>(lts, count, te, gj);
#endif
          }
          if (gf < p.n_grid) {
            int gj = (int)((double)(te - ts0) * idx_per_ms) + 1;
#ifdef VMGPU_PIPE_WALK_PROBE
            jbuf[gf] = (uint16_t)vm_ub_hint<2>(lts, count, te, gj);
#else
            jbuf[gf] = (uint16_t)vm_ub_hint_fast<2>(lts, count, te, gj);
#endif
          }
        }
        wave_ds_sync();
        /* the first dg points (window start before the grid) probe for i;
         * keeping them out of the main loop keeps the hot body free of the
         * probe's registers and branches */
        const int ghead = dg64 < p.n_grid ? dg64 : p.n_grid;
        for (int g = lane; g < ghead; g += WAVE) {
          int64_t t_end = p.start + (int64_t)g * p.step;
          int64_t t_start = t_end - sw.window;
          int j = jbuf[g];
          int gi = (int)((double)(t_start - ts0) * idx_per_ms) + 1;
          int i = vm_ub_hint_fast<2>(lts, count, t_start, gi);
#ifdef VMGPU_PIPE_ABL_NO_EVAL
          emit(g, (j > 0 && j <= count) ? lvs[(j - 1) * 2] : 0.0);
          (void)i;
#else
          emit(g, eval_rate_fused<2>(p, sw, lts, lvs, count, i, j, t_start));
#endif
        }
        /* main eval: full lane-blocks (no per-point exec masking), then one
         * masked tail round */
        {
          int g = ghead + lane;
          int64_t te = p.start + (int64_t)(ghead + lane) * p.step;
          const int span = p.n_grid - ghead;
          const int main_end = ghead + (span / WAVE) * WAVE;
          for (; g < main_end; g += WAVE, te += t_step_wave) {
            int j = jbuf[g];
            int i = jbuf[g - dg64];
#ifdef VMGPU_PIPE_ABL_NO_EVAL
            emit(g, (j > 0 && j <= count) ? lvs[(j - 1) * 2] : 0.0);
            (void)i;
#else
            emit(g, eval_rate_fused<2>(p, sw, lts, lvs, count, i, j,
                                       te - sw.window));
#endif
          }
          if (g < p.n_grid) {
            int j = jbuf[g];
            int i = jbuf[g - dg64];
#ifdef VMGPU_PIPE_ABL_NO_EVAL
            emit(g, (j > 0 && j <= count) ? lvs[(j - 1) * 2] : 0.0);
            (void)i;
#else
            emit(g, eval_rate_fused<2>(p, sw, lts, lvs, count, i, j,
                                       te - sw.window));
#endif
          }
        }
        }
        done = true;
      }
    }
    if (!done) {
      /* correctness fallback only (per-series window not a step multiple —
       * the host gates the pipe launch on plan shapes where the jbuf path
       * applies, so this stays cold): plain binary-search seeks, minimal
       * register footprint. */
      for (int g = lane; g < p.n_grid; g += WAVE) {
        int64_t t_end = p.start + (int64_t)g * p.step;
        int64_t t_start = t_end - sw.window;
        int i = vm_upper_bound<2>(lts, count, t_start);
        int j = vm_upper_bound<2>(lts, count, t_end);
        emit(g, eval_rate_fused<2>(p, sw, lts, lvs, count, i, j, t_start));
      }
    }
    /* samplesScanned: every grid point of this series contributes exactly
     * samplesScannedPerCall (= 2 for rate/deriv_fast) */
    if (lane == 0) scanned += 2ull * (unsigned long long)p.n_grid;
    wave_ds_sync();
    ws = ws_n;
    s = s_n;
    lo = lo_n;
    n = n_n;
    si_raw = si_raw_n;
  }
  for (int d = 32; d > 0; d >>= 1) scanned += __shfl_down((unsigned long long)scanned, d);
  if (lane == 0 && scanned) atomicAdd(io.samples_scanned, (unsigned long long)scanned);
}

/* ------------------------------------------------------------------ */
/* kernel 2: one 256-thread block per series (n <= CHUNK_BLOCK)       */
/* ------------------------------------------------------------------ */

template <int FUNC_CT>
__global__ __launch_bounds__(BLOCK_THREADS) void rollup_block_kernel(KPlan p, KIO io) {
  /* ALL LDS carved from one dynamic region (no static __shared__ in front —
   * guide §6 G17: statics shift the 16-B-aligned dynamic base).
   * layout: [0,8) count (int) + si (packed), [16, 16+CB*8) ts, then vals. */
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int32_t CB = p.chunk_block;
  int* sh_count = (int*)smem;
  int64_t* sh_si = (int64_t*)(smem + 8);
  unsigned long long* sh_sum = (unsigned long long*)(smem + 16);
  int64_t* lts = (int64_t*)(smem + 32);
  double* lvs = (double*)(smem + 32 + (size_t)CB * 8);
  double* lscratch = (double*)(smem + 32 + (size_t)CB * 16);
  int has_jbuf = 0;
  (void)vm_block_lds_bytes(CB, p.jbuf_mode, p.n_grid, &has_jbuf);
  uint16_t* jbuf = (uint16_t*)(smem + 32 + (size_t)CB * 16 + 256);
  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  uint64_t scanned = 0;

  for (uint32_t bs = blockIdx.x; bs < io.n_sel; bs += gridDim.x) {
    uint32_t s = io.series_sel[bs];
    uint64_t lo = io.offsets[s];
    int64_t n = (int64_t)(io.offsets[s + 1] - lo);

    {
      /* block-wide copy + one-pass stale detection; the wave-0 ballot
       * compaction (in place, forward) only runs when a stale NaN is
       * actually present */
      int* sh_stale = (int*)(sh_sum);  /* reuse the 8-byte scratch word */
      if (tid == 0) *sh_stale = 0;
      __syncthreads();
      int loc = 0;
      for (int64_t k = tid; k < n; k += BLOCK_THREADS) {
        int64_t t = io.ts[lo + k];
        double v = io.vals[lo + k];
        lts[k] = t;
        lvs[k] = v;
        if (p.drop_stale && vm_is_stale_nan(v)) loc = 1;
      }
      if (loc) atomicExch(sh_stale, 1);
      __syncthreads();
      if (*sh_stale) {
        if (wave == 0) {
          int c = load_compact_wave(lts, lvs, n, lts, lvs, true, lane);
          if (lane == 0) *sh_count = c;
        }
      } else if (tid == 0) {
        *sh_count = (int)n;
      }
    }
    __syncthreads();
    int count = *sh_count;
    /* the counter-reset scan only touches VALUES and the j-cache fill only
     * reads TIMESTAMPS: wave 0 scans (pair rounds — half the serial chain)
     * while waves 1-3 fill the rate j-cache concurrently */
    if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
      if (has_jbuf && count > 1 && count <= 65535 && wave != 0) {
        int64_t ts0f = lts[0];
        int64_t span_ms = lts[count - 1] - ts0f;
        double ipm = span_ms > 0 ? (double)(count - 1) / (double)span_ms : 0.0;
        for (int g = (wave - 1) * WAVE + lane; g < p.n_grid;
             g += (BLOCK_THREADS - WAVE)) {
          int64_t t_end = p.start + (int64_t)g * p.step;
          int gj = (int)((double)(t_end - ts0f) * ipm) + 1;
          jbuf[g] = (uint16_t)vm_ub_hint_fast(lts, count, t_end, gj);
        }
      }
    }
#ifndef VMGPU_ABL_NO_RCR
    if (p.rcr && wave == 0)
      rcr_scan_col_pairs<1>(lts, lvs, count, p.max_staleness, lane);
#endif
    __syncthreads();
    if (p.pre_func && wave == 0)
      pre_func_wave(lts, lvs, count, p.pre_func, lane);
    __syncthreads();
    if (wave == 0) {
      int64_t si = p.step;
      if (p.start < p.end) {
        if (io.series_si && count == (int)n) {
          int64_t c = io.series_si[s];
          si = c > 0 ? c : p.step;
        } else {
          si = scrape_interval_wave(lts, count, p.step, lane, lscratch);
        }
      }
      if (lane == 0) *sh_si = si;
    }
    __syncthreads();
    SeriesWindow sw = series_window(p, *sh_si);

    if (tid == 0) scanned += (uint64_t)count;
    if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
      /* shared-boundary j-cache (see the wave kernel): one seek per point,
       * i(g) = j(g - window/step) when the window is a step multiple */
      int dg64 = (sw.window > 0 && p.step > 0 && sw.window % p.step == 0)
                     ? (int)(sw.window / p.step) : 0;
      if (has_jbuf && dg64 > 0 && count <= 65535) {
        double idx_per_ms = 0.0;
        int64_t ts0 = 0;
        if (count > 1) {
          ts0 = lts[0];
          int64_t span_ms = lts[count - 1] - ts0;
          idx_per_ms = span_ms > 0 ? (double)(count - 1) / (double)span_ms : 0.0;
        }
        /* count > 1 mirrors the concurrent prefill's gate exactly: the
         * j-cache was already built by waves 1-3 during the scan */
        if (count <= 1) {
          for (int g = tid; g < p.n_grid; g += BLOCK_THREADS) {
            int64_t t_end = p.start + (int64_t)g * p.step;
            int gj = (int)((double)(t_end - ts0) * idx_per_ms) + 1;
            jbuf[g] = (uint16_t)vm_ub_hint_fast(lts, count, t_end, gj);
          }
        }
        __syncthreads();
        for (int g = tid; g < p.n_grid; g += BLOCK_THREADS) {
          int64_t t_end = p.start + (int64_t)g * p.step;
          int64_t t_start = t_end - sw.window;
          int j = jbuf[g];
          int i;
          if (g >= dg64) {
            i = jbuf[g - dg64];
          } else {
            int gi = (int)((double)(t_start - ts0) * idx_per_ms) + 1;
            i = vm_ub_hint_fast(lts, count, t_start, gi);
          }
          vm_emit_value(p, io, s, g,
                        eval_rate_fused(p, sw, lts, lvs, count, i, j, t_start));
          scanned += 2;
        }
        __syncthreads();
        continue;
      }
    }
    for (int g0 = 0; g0 < p.n_grid; g0 += BLOCK_THREADS) {
      int g = g0 + tid;
      if (g < p.n_grid) scanned += eval_grid_point<FUNC_CT>(p, sw, lts, lvs, count, g, s, io);
    }
    __syncthreads();
  }
  for (int d = 32; d > 0; d >>= 1) scanned += __shfl_down((unsigned long long)scanned, d);
  if (tid == 0) *sh_sum = 0;
  __syncthreads();
  if (lane == 0 && scanned) atomicAdd(sh_sum, (unsigned long long)scanned);
  __syncthreads();
  if (tid == 0 && *sh_sum) atomicAdd(io.samples_scanned, *sh_sum);
}

/* ------------------------------------------------------------------ */
/* kernel 3: huge series via global scratch                           */
/* ------------------------------------------------------------------ */

template <int FUNC_CT>
__global__ __launch_bounds__(BLOCK_THREADS) void rollup_huge_kernel(KPlan p, KIO io) {
  __shared__ int sh_count;
  __shared__ int sh_has_stale;
  __shared__ int64_t sh_si;
  __shared__ double sh_scratch[32];
  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  uint64_t scanned = 0;

  for (uint32_t bs = blockIdx.x; bs < io.n_sel; bs += gridDim.x) {
    uint32_t s = io.series_sel[bs];
    uint64_t lo = io.offsets[s];
    int64_t n = (int64_t)(io.offsets[s + 1] - lo);
    int64_t* dts = io.scr_ts + io.scr_offsets[bs];
    double* dvs = io.scr_vals + io.scr_offsets[bs];

    const int64_t* uts;
    const double* uvs;
    if (p.drop_stale || p.rcr || p.pre_func) {
      /* block-wide copy to the global scratch + stale detection in one
       * pass (the old wave-0-only copy serialized ~n*16 B per series on
       * one wave); stale compaction (rare) falls back to an IN-PLACE
       * wave-0 forward compaction — within each tile the wave's loads all
       * precede its stores and destinations never pass sources. */
      if (tid == 0) sh_has_stale = 0;
      __syncthreads();
      int loc = 0;
      for (int64_t k = tid; k < n; k += BLOCK_THREADS) {
        int64_t t = io.ts[lo + k];
        double v = io.vals[lo + k];
        dts[k] = t;
        dvs[k] = v;
        if (p.drop_stale && vm_is_stale_nan(v)) loc = 1;
      }
      if (loc) atomicExch(&sh_has_stale, 1);
      __syncthreads();
      if (wave == 0) {
        int c = (int)n;
        if (sh_has_stale)
          c = load_compact_wave(dts, dvs, n, dts, dvs, true, lane);
        if (p.rcr) rcr_scan_col_pairs<1>(dts, dvs, c, p.max_staleness, lane);
        if (p.pre_func) pre_func_wave(dts, dvs, c, p.pre_func, lane);
        if (lane == 0) sh_count = c;
      }
      __syncthreads();
      /* wave 0's global-scratch stores must be visible to the whole block:
       * same-CU reads, but L1 is per-CU so plain loads after syncthreads are
       * coherent within the block's CU. */
      uts = dts;
      uvs = dvs;
    } else {
      if (tid == 0) sh_count = (int)n;
      __syncthreads();
      uts = io.ts + lo;
      uvs = io.vals + lo;
    }
    int count = sh_count;
    if (wave == 0) {
      int64_t si = p.step;
      if (p.start < p.end) {
        if (io.series_si && count == (int)n) {
          int64_t c = io.series_si[s];
          si = c > 0 ? c : p.step;
        } else {
          /* scrape interval needs the tail of the (possibly compacted) column */
          si = scrape_interval_wave(uts, count, p.step, lane, sh_scratch);
        }
      }
      if (lane == 0) sh_si = si;
    }
    __syncthreads();
    SeriesWindow sw = series_window(p, sh_si);

    if (tid == 0) scanned += (uint64_t)count;
    /* density-interpolated seek hints (same scheme as the wave kernel):
     * a full binary search over a 10^4-sample column costs ~15 dependent
     * global loads per boundary; the hint + bounded walk resolves in ~1 */
    double idx_per_ms = 0.0;
    int64_t ts0 = 0;
    if (count > 1) {
      ts0 = uts[0];
      int64_t span_ms = uts[count - 1] - ts0;
      idx_per_ms = span_ms > 0 ? (double)(count - 1) / (double)span_ms : 0.0;
    }
    for (int g0 = 0; g0 < p.n_grid; g0 += BLOCK_THREADS) {
      int g = g0 + tid;
      if (g < p.n_grid) {
        int64_t t_end = p.start + (int64_t)g * p.step;
        int64_t t_start = t_end - sw.window;
        int gi = (int)((double)(t_start - ts0) * idx_per_ms) + 1;
        int gj = (int)((double)(t_end - ts0) * idx_per_ms) + 1;
        int i = vm_ub_hint(uts, count, t_start, gi);
        int j = vm_ub_hint(uts, count, t_end, gj);
        if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
          vm_emit_value(p, io, s, g,
                        eval_rate_fused(p, sw, uts, uvs, count, i, j, t_start));
          scanned += 2;
        } else {
          scanned += eval_grid_point_ij<FUNC_CT>(p, sw, uts, uvs, count, g, s, io, i, j);
        }
      }
    }
    __syncthreads();
  }
  for (int d = 32; d > 0; d >>= 1) scanned += __shfl_down((unsigned long long)scanned, d);
  __shared__ unsigned long long block_sum;
  if (tid == 0) block_sum = 0;
  __syncthreads();
  if (lane == 0 && scanned) atomicAdd(&block_sum, (unsigned long long)scanned);
  __syncthreads();
  if (tid == 0 && block_sum) atomicAdd(io.samples_scanned, block_sum);
}

/* ------------------------------------------------------------------ */
/* aggregate identity init + finalize                                 */
/* ------------------------------------------------------------------ */

__global__ void aggr_init_kernel(double* values, double* counts, uint64_t n, int32_t aggr) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  double ident = 0.0;
  if (aggr == VMGPU_AGGR_MIN) ident = vm_dinf();
  else if (aggr == VMGPU_AGGR_MAX) ident = -vm_dinf();
  else if (aggr == VMGPU_AGGR_GEOMEAN) ident = 1.0;
  for (; i < n; i += stride) {
    values[i] = ident;
    counts[i] = 0.0;
  }
}

__global__ void aggr_finalize_kernel(double* values, double* counts, uint64_t n, int32_t aggr) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    double c = counts[i];
    switch (aggr) {
      case VMGPU_AGGR_SUM:
      case VMGPU_AGGR_MIN:
      case VMGPU_AGGR_MAX:
      case VMGPU_AGGR_SUM2:
        if (c == 0) values[i] = vm_dnan();
        break;
      case VMGPU_AGGR_AVG:
        values[i] = (c == 0) ? vm_dnan() : values[i] / c;
        break;
      case VMGPU_AGGR_COUNT:
        if (values[i] == 0) values[i] = vm_dnan();
        break;
      case VMGPU_AGGR_GROUP:
        values[i] = (values[i] == 0) ? vm_dnan() : 1.0;
        break;
      case VMGPU_AGGR_GEOMEAN:
        values[i] = (c == 0) ? vm_dnan() : pow(values[i], 1.0 / c);
        break;
      default:
        break;
    }
  }
}


/* ------------------------------------------------------------------ */
/* topk family (aggr.go:646-741) + histogram_quantile (transform.go)  */
/* ------------------------------------------------------------------ */

/* Order-preserving u64 key for f64 under lessWithNaNs / greaterWithNaNs
 * (aggr.go:1259-1279): selection always keeps the "k largest keys".  NaN
 * maps to the SMALLEST key in both directions — lessWithNaNs treats NaN as
 * smaller than any number (so topk never keeps it) and greaterWithNaNs
 * treats it as bigger (so it sorts to the FRONT of the bottomk order and
 * is again never kept). */
static VM_DEV unsigned long long vm_topk_key(double v, int reverse) {
  if (vm_isnan(v)) return 0ULL;
  unsigned long long b = (unsigned long long)__double_as_longlong(v);
  unsigned long long ord = (b >> 63) ? ~b : (b | 0x8000000000000000ULL);
  /* the NaN sentinels 0 and ~0 are unreachable for non-NaN doubles: their
   * preimages under the order map are NaN bit patterns */
  return reverse ? ~ord : ord;
}

/* inverse of vm_topk_key's order map (non-NaN keys only) */
static VM_DEV double vm_topk_key_inv(unsigned long long ord) {
  unsigned long long b = (ord & 0x8000000000000000ULL)
                             ? (ord ^ 0x8000000000000000ULL)
                             : ~ord;
  return __longlong_as_double((long long)b);
}

/* k-th smallest (0-based) non-NaN key of a row: 8 passes of 256-bin LDS
 * histogram refinement over the order-preserving u64 keys (exact — every
 * byte of the result is pinned by counts).  hist: per-wave uint32[256]. */
static __device__ unsigned long long topk_row_kth(
    const double* row, int32_t n_grid, uint32_t k, int lane, uint32_t* hist) {
  unsigned long long prefix = 0;
  for (int byte = 7; byte >= 0; byte--) {
    const int shift = byte * 8;
    for (int b = lane; b < 256; b += WAVE) hist[b] = 0;
    wave_ds_sync();
    const unsigned long long pmask =
        (byte == 7) ? 0ULL : (~0ULL << (shift + 8));
    for (int g = lane; g < n_grid; g += WAVE) {
      double v = row[g];
      if (vm_isnan(v)) continue;
      unsigned long long key = vm_topk_key(v, 0);
      if ((key & pmask) != prefix) continue;
      atomicAdd(&hist[(key >> shift) & 0xff], 1u);
    }
    wave_ds_sync();
    /* each lane owns 4 consecutive bins; exclusive wave prefix of totals */
    uint32_t b0 = hist[lane * 4], b1 = hist[lane * 4 + 1];
    uint32_t b2 = hist[lane * 4 + 2], b3 = hist[lane * 4 + 3];
    uint32_t lt = b0 + b1 + b2 + b3;
    uint32_t ex = lt;
    for (int d = 1; d < WAVE; d <<= 1) {
      uint32_t xo = __shfl_up(ex, d);
      if (lane >= d) ex += xo;
    }
    ex -= lt;
    uint32_t c0 = ex, c1 = c0 + b0, c2 = c1 + b1, c3 = c2 + b2;
    int pick = -1;
    if (k >= c0 && k < c0 + b0) pick = 0;
    else if (k >= c1 && k < c1 + b1) pick = 1;
    else if (k >= c2 && k < c2 + b2) pick = 2;
    else if (k >= c3 && k < c3 + b3) pick = 3;
    uint64_t mb = __ballot(pick >= 0);
    int srcl = __ffsll((unsigned long long)mb) - 1;
    int pickv = __shfl(pick, srcl);
    uint32_t below = __shfl(pickv == 0 ? c0 : pickv == 1 ? c1
                            : pickv == 2 ? c2 : c3, srcl);
    int bin = srcl * 4 + pickv;
    k -= below;
    prefix |= ((unsigned long long)(unsigned)bin) << shift;
    wave_ds_sync();
  }
  return prefix;
}

/* per-series range summaries (aggr.go:804-858), one wave per series row */
__global__ __launch_bounds__(BLOCK_THREADS) void topk_summary_kernel(
    const double* values, uint32_t n_series, int32_t n_grid, int32_t op,
    int32_t reverse, unsigned long long* keys) {
  __shared__ uint32_t sh_hist[WAVES_PER_BLOCK][256];
  const int wave_in_block = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  uint32_t wid = blockIdx.x * WAVES_PER_BLOCK + wave_in_block;
  uint32_t stride = gridDim.x * WAVES_PER_BLOCK;
  for (uint32_t s = wid; s < n_series; s += stride) {
    const double* row = values + (size_t)s * n_grid;
    double acc = vm_dnan();
    if (op == 0 || op == 1 || op == 2) { /* avg / min / max */
      double sum = 0, cnt = 0;
      double mn = vm_dnan(), mx = vm_dnan();
      for (int g = lane; g < n_grid; g += WAVE) {
        double v = row[g];
        if (vm_isnan(v)) continue;
        cnt += 1;
        sum += v;
        if (vm_isnan(mn) || v < mn) mn = v;
        if (vm_isnan(mx) || v > mx) mx = v;
      }
      for (int d = 32; d > 0; d >>= 1) {
        double so = __shfl_down(sum, d);
        double co = __shfl_down(cnt, d);
        double mno = __shfl_down(mn, d);
        double mxo = __shfl_down(mx, d);
        sum += so;
        cnt += co;
        if (vm_isnan(mn) || (!vm_isnan(mno) && mno < mn)) mn = mno;
        if (vm_isnan(mx) || (!vm_isnan(mxo) && mxo > mx)) mx = mxo;
      }
      if (op == 0) acc = (cnt == 0) ? vm_dnan() : sum / cnt;
      else if (op == 1) acc = mn;
      else acc = mx;
    } else if (op == 3) { /* median = quantile(0.5) (aggr.go:848,922) */
      uint32_t m = 0;
      for (int g = lane; g < n_grid; g += WAVE)
        if (!vm_isnan(row[g])) m++;
      for (int d = 32; d > 0; d >>= 1) m += __shfl_down(m, d);
      m = __shfl(m, 0);
      if (m > 0) {
        uint32_t li = (m - 1) / 2; /* floor(0.5*(m-1)) */
        unsigned long long kv =
            topk_row_kth(row, n_grid, li, lane, sh_hist[wave_in_block]);
        double vlo = vm_topk_key_inv(kv);
        if (m & 1) {
          acc = vlo; /* odd count: weight 0 */
        } else {
          /* upper = positional li+1 in the sorted row: equals vlo inside
           * an equal run, else the smallest key > kv */
          uint32_t cle = 0; /* #keys <= kv */
          unsigned long long nxt = ~0ULL;
          for (int g = lane; g < n_grid; g += WAVE) {
            double v = row[g];
            if (vm_isnan(v)) continue;
            unsigned long long key = vm_topk_key(v, 0);
            if (key <= kv) cle++;
            else if (key < nxt) nxt = key;
          }
          for (int d = 32; d > 0; d >>= 1) {
            cle += __shfl_down(cle, d);
            unsigned long long no = __shfl_down(nxt, d);
            if (no < nxt) nxt = no;
          }
          cle = __shfl(cle, 0);
          nxt = __shfl(nxt, 0);
          double vhi = (li + 1 < cle) ? vlo : vm_topk_key_inv(nxt);
          /* quantileSorted: v[lower]*(1-weight) + v[upper]*weight, w=0.5 */
          acc = vlo * 0.5 + vhi * 0.5;
        }
      }
    } else if (op == 4) { /* last non-NaN */
      int base = ((n_grid + WAVE - 1) / WAVE - 1) * WAVE;
      for (; base >= 0; base -= WAVE) {
        int g = base + lane;
        double v = (g < n_grid) ? row[g] : vm_dnan();
        uint64_t m = __ballot(!vm_isnan(v));
        if (m) {
          int hi = 63 - __clzll((unsigned long long)m);
          acc = __shfl(v, hi);
          break;
        }
      }
    }
    if (lane == 0) keys[s] = vm_topk_key(acc, reverse);
  }
}

__global__ void topk_hist_kernel(const unsigned long long* keys, uint32_t n,
                                 unsigned long long prefix, int shift,
                                 uint32_t* hist) {
  /* 16-bit histogram of keys whose bits above shift+16 equal prefix */
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t stride = gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned long long k = keys[i];
    if (shift < 48 && (k >> (shift + 16)) != prefix) continue;
    atomicAdd(&hist[(k >> shift) & 0xffff], 1u);
  }
}

__global__ void topk_collect_kernel(const unsigned long long* keys, uint32_t n,
                                    unsigned long long kstar, uint32_t ties_quota,
                                    uint32_t* counters, uint32_t* sel,
                                    uint32_t sel_cap) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t stride = gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned long long k = keys[i];
    if (k > kstar) {
      uint32_t slot = atomicAdd(&counters[0], 1u);
      if (slot < sel_cap) sel[slot] = i;
    } else if (k == kstar) {
      uint32_t t = atomicAdd(&counters[1], 1u);
      if (t < ties_quota) {
        uint32_t slot = atomicAdd(&counters[0], 1u);
        if (slot < sel_cap) sel[slot] = i;
      }
    }
  }
}

__global__ void topk_remaining_kernel(const double* values, uint32_t n_series,
                                      int32_t n_grid, const uint8_t* selected,
                                      double* rem_sum, double* rem_cnt) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_series * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    uint32_t s = (uint32_t)(i / n_grid);
    int32_t g = (int32_t)(i % n_grid);
    if (selected[s]) continue;
    double v = values[i];
    if (vm_isnan(v)) continue;
    atomicAdd(&rem_sum[g], v);
    atomicAdd(&rem_cnt[g], 1.0);
  }
}

/* ---- pointwise topk over [n_series x n_grid] ---- */

__global__ void topk_col_hist_kernel(const double* values, uint32_t n_series,
                                     int32_t n_grid, int32_t reverse,
                                     uint32_t* hists) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_series * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int32_t g = (int32_t)(i % n_grid);
    unsigned long long k = vm_topk_key(values[i], reverse);
    atomicAdd(&hists[(size_t)g * 65536 + (k >> 48)], 1u);
  }
}

__global__ void topk_col_threshold_kernel(const uint32_t* hists, int32_t n_grid,
                                          uint32_t k, uint32_t* bin_of_col,
                                          uint32_t* above_of_col) {
  int g = blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= n_grid) return;
  const uint32_t* h = hists + (size_t)g * 65536;
  uint32_t cum = 0;
  int bin = 0;
  for (int b = 65535; b >= 0; b--) {
    uint32_t c = h[b];
    if (cum + c >= k) {
      bin = b;
      break;
    }
    cum += c;
  }
  bin_of_col[g] = (uint32_t)bin;
  above_of_col[g] = cum;
}

__global__ void topk_col_candidates_kernel(const double* values, uint32_t n_series,
                                           int32_t n_grid, int32_t reverse,
                                           const uint32_t* bin_of_col,
                                           unsigned long long* cand,
                                           uint32_t* cand_n, uint32_t cap,
                                           uint32_t* overflow) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_series * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int32_t g = (int32_t)(i % n_grid);
    unsigned long long k = vm_topk_key(values[i], reverse);
    if ((uint32_t)(k >> 48) != bin_of_col[g]) continue;
    uint32_t slot = atomicAdd(&cand_n[g], 1u);
    if (slot < cap) cand[(size_t)g * cap + slot] = k;
    else atomicAdd(overflow, 1u);
  }
}

__global__ __launch_bounds__(256) void topk_col_kstar_kernel(
    const unsigned long long* cand, const uint32_t* cand_n, uint32_t cap,
    const uint32_t* above_of_col, uint32_t k, int32_t n_grid,
    unsigned long long* kstar_of_col, uint32_t* ties_of_col) {
  int g = blockIdx.x;
  if (g >= n_grid) return;
  uint32_t n = cand_n[g];
  if (n > cap) n = cap;
  uint32_t need = k > above_of_col[g] ? k - above_of_col[g] : 0;
  const unsigned long long* c = cand + (size_t)g * cap;
  __shared__ unsigned long long sh_lo, sh_hi, sh_mid;
  __shared__ uint32_t sh_cnt;
  if (threadIdx.x == 0) {
    sh_lo = 0;
    sh_hi = ~0ULL;
  }
  __syncthreads();
  if (need == 0 || n == 0) {
    if (threadIdx.x == 0) {
      kstar_of_col[g] = ~0ULL;
      ties_of_col[g] = 0;
    }
    return;
  }
  for (int it = 0; it < 64; it++) {
    if (threadIdx.x == 0) {
      sh_mid = sh_lo + ((sh_hi - sh_lo) >> 1);
      sh_cnt = 0;
    }
    __syncthreads();
    unsigned long long mid = sh_mid;
    uint32_t local = 0;
    for (uint32_t t = threadIdx.x; t < n; t += blockDim.x)
      if (c[t] >= mid) local++;
    atomicAdd(&sh_cnt, local);
    __syncthreads();
    bool done = false;
    if (threadIdx.x == 0) {
      if (sh_cnt >= need) sh_lo = sh_mid + 1;
      else sh_hi = sh_mid;
    }
    __syncthreads();
    if (sh_lo >= sh_hi) { done = true; }
    if (done) break;
  }
  unsigned long long kstar = sh_lo - 1;
  if (threadIdx.x == 0) sh_cnt = 0;
  __syncthreads();
  uint32_t local = 0;
  for (uint32_t t = threadIdx.x; t < n; t += blockDim.x)
    if (c[t] > kstar) local++;
  atomicAdd(&sh_cnt, local);
  __syncthreads();
  if (threadIdx.x == 0) {
    kstar_of_col[g] = kstar;
    ties_of_col[g] = need - sh_cnt;
  }
}

__global__ void topk_col_fill_kernel(double* values, uint32_t n_series,
                                     int32_t n_grid, int32_t reverse, uint32_t k,
                                     const uint32_t* bin_of_col,
                                     const unsigned long long* kstar_of_col,
                                     uint32_t* ties_taken,
                                     const uint32_t* ties_of_col) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_series * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int32_t g = (int32_t)(i % n_grid);
    unsigned long long key = vm_topk_key(values[i], reverse);
    uint32_t bin = bin_of_col[g];
    uint32_t kb = (uint32_t)(key >> 48);
    bool keep;
    if (kb > bin) {
      keep = true;
    } else if (kb < bin) {
      keep = false;
    } else {
      unsigned long long kstar = kstar_of_col[g];
      if (key > kstar) keep = true;
      else if (key == kstar) {
        uint32_t t = atomicAdd(&ties_taken[g], 1u);
        keep = t < ties_of_col[g];
      } else {
        keep = false;
      }
    }
    if (!keep) values[i] = vm_dnan();
  }
}

/* histogram_quantile (transform.go:1028-1074 + fixBrokenBuckets 1140):
 * one thread per (group, grid point) */
/* histogram_avg / histogram_stddev / histogram_stdvar
 * (transform.go:transformHistogramAvg/Stddev/Stdvar + avgForLeTimeseries /
 * stdvarForLeTimeseries): (le+lePrev)/2-weighted moments over the bucket
 * column, +/-Inf les skipped, NO fixBrokenBuckets (the reference reads raw
 * cumulative values here). mode: 0 avg, 1 stddev, 2 stdvar. */
__global__ void hstat_kernel(int32_t mode, const double* bv, const double* les,
                             const uint64_t* goff, int64_t n_groups,
                             int32_t n_grid, double* out) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_groups * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int64_t grp = (int64_t)(i / n_grid);
    int32_t g = (int32_t)(i % n_grid);
    int64_t lo = (int64_t)goff[grp];
    int64_t n_les = (int64_t)(goff[grp + 1] - lo);
    double le_prev = 0, v_prev = 0, sum = 0, sum2 = 0, wt = 0;
    for (int64_t j = 0; j < n_les; j++) {
      double le = les[lo + j];
      if (isinf(le)) continue;
      double v = bv[(size_t)(lo + j) * n_grid + g];
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
      r = vm_dnan();
    } else if (mode == 0) {
      r = sum / wt;
    } else {
      double avg = sum / wt;
      double sv = sum2 / wt - avg * avg;
      if (sv < 0) sv = 0;
      r = (mode == 1) ? sqrt(sv) : sv;
    }
    out[i] = r;
  }
}

/* histogram_share (transform.go:transformHistogramShare): the quantile
 * walk's dual — share of samples <= leReq[g], with lower/upper bounds.
 * fixBrokenBuckets applied on the fly as in hq_kernel. */
__global__ void hshare_kernel(const double* le_req, const double* bv,
                              const double* les, const uint64_t* goff,
                              int64_t n_groups, int32_t n_grid, double* out,
                              double* out_lo, double* out_hi) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_groups * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int64_t grp = (int64_t)(i / n_grid);
    int32_t g = (int32_t)(i % n_grid);
    int64_t lo = (int64_t)goff[grp];
    int64_t n_les = (int64_t)(goff[grp + 1] - lo);
    double req = le_req[g];
    double q = vm_dnan(), lb = vm_dnan(), ub = vm_dnan();
    if (!vm_isnan(req) && n_les > 0) {
      if (req < 0) {
        q = 0; lb = 0; ub = 0;
      } else if (isinf(req) && req > 0) {
        q = 1; lb = 1; ub = 1;
      } else {
        /* fixBrokenBuckets pass: running-max with NaN->prev */
        double v_last = 0;
        {
          double fix_prev = 0;
          for (int64_t j = 0; j < n_les; j++) {
            double v = bv[(size_t)(lo + j) * n_grid + g];
            if (j == 0) fix_prev = vm_isnan(v) ? 0 : v;
            else if (!(vm_isnan(v) || fix_prev > v)) fix_prev = v;
          }
          v_last = fix_prev;
        }
        double vp = 0, lep = 0;
        double fix_prev = 0;
        bool done = false;
        for (int64_t j = 0; j < n_les && !done; j++) {
          double raw = bv[(size_t)(lo + j) * n_grid + g];
          double v;
          if (j == 0) v = vm_isnan(raw) ? 0 : raw;
          else v = (vm_isnan(raw) || fix_prev > raw) ? fix_prev : raw;
          fix_prev = v;
          double le = les[lo + j];
          if (req >= le) {
            vp = v;
            lep = le;
            continue;
          }
          /* lePrev <= req < le */
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
          done = true;
        }
        if (!done) { q = 1; lb = 1; ub = 1; }
      }
    }
    out[i] = q;
    if (out_lo) out_lo[i] = lb;
    if (out_hi) out_hi[i] = ub;
  }
}

__global__ void hq_kernel(double phi, const double* bv, const double* les,
                          const uint64_t* goff, int64_t n_groups, int32_t n_grid,
                          double* out, double* out_lo, double* out_hi) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)n_groups * n_grid;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int64_t grp = (int64_t)(i / n_grid);
    int32_t g = (int32_t)(i % n_grid);
    int64_t lo = (int64_t)goff[grp];
    int64_t n_les = (int64_t)(goff[grp + 1] - lo);
    double q = vm_dnan(), lb = vm_dnan(), ub = vm_dnan();
    if (!vm_isnan(phi)) {
      double fix_prev = 0;
      double v_last = 0;
      for (int64_t j = 0; j < n_les; j++) {
        double v = bv[(size_t)(lo + j) * n_grid + g];
        if (j == 0) fix_prev = vm_isnan(v) ? 0 : v;
        else if (!(vm_isnan(v) || fix_prev > v)) fix_prev = v;
        v_last = fix_prev;
      }
      if (v_last != 0) {
        if (phi < 0) {
          q = -vm_dinf();
          lb = -vm_dinf();
          double v0 = bv[(size_t)lo * n_grid + g];
          ub = vm_isnan(v0) ? 0 : v0;
        } else if (phi > 1) {
          q = vm_dinf();
          lb = v_last;
          ub = vm_dinf();
        } else {
          double v_req = v_last * phi;
          double vp = 0, lep = 0;
          fix_prev = 0;
          bool done = false;
          for (int64_t j = 0; j < n_les && !done; j++) {
            double raw = bv[(size_t)(lo + j) * n_grid + g];
            double v;
            if (j == 0) v = vm_isnan(raw) ? 0 : raw;
            else v = (vm_isnan(raw) || fix_prev > raw) ? fix_prev : raw;
            fix_prev = v;
            double le = les[lo + j];
            if (v <= 0) {
              lep = le;
              continue;
            }
            if (v < v_req) {
              vp = v;
              lep = le;
              continue;
            }
            if (isinf(le)) break;
            if (v == vp) {
              q = lep;
              lb = lep;
              ub = v;
              done = true;
              break;
            }
            q = lep + (le - lep) * (v_req - vp) / (v - vp);
            lb = lep;
            ub = le;
            done = true;
          }
          if (!done) {
            double vv = vm_dnan();
            for (int64_t j = n_les - 1; j >= 0; j--) {
              if (!isinf(les[lo + j])) {
                vv = les[lo + j];
                break;
              }
            }
            q = vv;
            lb = vv;
            ub = vm_dinf();
          }
        }
      }
    }
    out[i] = q;
    if (out_lo) out_lo[i] = lb;
    if (out_hi) out_hi[i] = ub;
  }
}

/* ------------------------------------------------------------------ */
/* launch dispatch: compile-time specialization for hot funcs         */
/* ------------------------------------------------------------------ */

template <int FUNC_CT>
static void launch_rollup_t(int which, uint32_t blocks, size_t lds,
                            const KPlan& p, const KIO& w, hipStream_t stream) {
  if (which == 0) {
    /* grouped rate plans with grids the accumulator covers take the
     * register-run variant (GACC) — valid because grouped batches are
     * physically relayouted by group at creation.  preFunc plans take the
     * PREF instantiation (compile-time gated for the same reason). */
    /* A/B measured (profiles/round1_final.md §4): the register-run
     * accumulator (GACC) is 0.4 ms SLOWER than plain per-point atomics at
     * config 3 (2.89 vs 3.30 ms) — the accumulate/flush instruction cost
     * exceeds the scattered-atomic cost on this issue-bound kernel.  The
     * instantiation stays compiled (VMGPU_ABL_GACC re-enables it for
     * future A/Bs) but plain atomics are the default. */
    bool gacc = false;
#ifdef VMGPU_ABL_GACC
    gacc = (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) &&
           p.aggr != VMGPU_AGGR_NONE && w.group_ids != nullptr &&
           p.n_grid <= 4 * WAVE;
#endif
    if (p.pre_func != 0) {
      if (gacc)
        hipLaunchKernelGGL((rollup_wave_kernel<FUNC_CT, true, true>),
                           dim3(blocks), dim3(BLOCK_THREADS), lds, stream, p, w);
      else
        hipLaunchKernelGGL((rollup_wave_kernel<FUNC_CT, false, true>),
                           dim3(blocks), dim3(BLOCK_THREADS), lds, stream, p, w);
    } else if (gacc) {
      hipLaunchKernelGGL((rollup_wave_kernel<FUNC_CT, true, false>), dim3(blocks),
                         dim3(BLOCK_THREADS), lds, stream, p, w);
    } else {
      hipLaunchKernelGGL((rollup_wave_kernel<FUNC_CT, false, false>), dim3(blocks),
                         dim3(BLOCK_THREADS), lds, stream, p, w);
    }
  } else if (which == 1) {
    hipLaunchKernelGGL(rollup_block_kernel<FUNC_CT>, dim3(blocks),
                       dim3(BLOCK_THREADS), lds, stream, p, w);
  } else if (which == 3) {
    /* software-pipelined register-staged wave variant (series <= 256
     * samples); instantiated only for the hot specializations it is
     * measured on — everything else falls back to the wave kernel */
    if constexpr (FUNC_CT == VMF_RATE || FUNC_CT == VMF_DERIV_FAST) {
      if (p.aggr != VMGPU_AGGR_NONE && w.group_ids)
        hipLaunchKernelGGL((rollup_pipe_kernel<FUNC_CT, true>), dim3(blocks),
                           dim3(BLOCK_THREADS), lds, stream, p, w);
      else
        hipLaunchKernelGGL((rollup_pipe_kernel<FUNC_CT, false>), dim3(blocks),
                           dim3(BLOCK_THREADS), lds, stream, p, w);
    } else {
      launch_rollup_t<FUNC_CT>(0, blocks, lds, p, w, stream);
    }
  } else {
    hipLaunchKernelGGL(rollup_huge_kernel<FUNC_CT>, dim3(blocks),
                       dim3(BLOCK_THREADS), lds, stream, p, w);
  }
}

static hipStream_t launch_stream();

static void launch_rollup(int which, uint32_t blocks, size_t lds,
                          const KPlan& p, const KIO& w) {
  hipStream_t s = launch_stream();
  switch (p.func) {
    case VMF_RATE: launch_rollup_t<VMF_RATE>(which, blocks, lds, p, w, s); break;
    case VMF_INCREASE: launch_rollup_t<VMF_INCREASE>(which, blocks, lds, p, w, s); break;
    case VMF_INCREASE_PURE: launch_rollup_t<VMF_INCREASE_PURE>(which, blocks, lds, p, w, s); break;
    case VMF_DELTA: launch_rollup_t<VMF_DELTA>(which, blocks, lds, p, w, s); break;
    case VMF_AVG: launch_rollup_t<VMF_AVG>(which, blocks, lds, p, w, s); break;
    case VMF_MIN: launch_rollup_t<VMF_MIN>(which, blocks, lds, p, w, s); break;
    case VMF_MAX: launch_rollup_t<VMF_MAX>(which, blocks, lds, p, w, s); break;
    case VMF_SUM: launch_rollup_t<VMF_SUM>(which, blocks, lds, p, w, s); break;
    case VMF_COUNT: launch_rollup_t<VMF_COUNT>(which, blocks, lds, p, w, s); break;
    case VMF_LAST: launch_rollup_t<VMF_LAST>(which, blocks, lds, p, w, s); break;
    case VMF_DEFAULT_ROLLUP: launch_rollup_t<VMF_DEFAULT_ROLLUP>(which, blocks, lds, p, w, s); break;
    case VMF_QUANTILE: launch_rollup_t<VMF_QUANTILE>(which, blocks, lds, p, w, s); break;
    case VMF_DERIV_FAST: launch_rollup_t<VMF_DERIV_FAST>(which, blocks, lds, p, w, s); break;
    case VMF_FIRST: launch_rollup_t<VMF_FIRST>(which, blocks, lds, p, w, s); break;
    default: launch_rollup_t<-1>(which, blocks, lds, p, w, s); break;
  }
}

/* ------------------------------------------------------------------ */
/* host side                                                          */
/* ------------------------------------------------------------------ */

namespace {

struct Batch {
  int64_t* d_ts = nullptr;
  double* d_vals = nullptr;
  uint64_t* d_offsets = nullptr;
  int32_t* d_group_ids = nullptr;
  uint32_t n_series = 0;
  uint32_t n_groups = 0;
  uint64_t n_samples = 0;
  /* series partition by length */
  int64_t* d_si = nullptr;   /* per-series scrape-interval index */
  uint32_t* d_wave_list = nullptr;
  uint32_t* d_block_list = nullptr;
  uint32_t* d_huge_list = nullptr;
  uint32_t n_wave = 0, n_block = 0, n_huge = 0;
  uint32_t max_wave_len = 0;
  uint32_t max_block_len = 0;
  bool wave_is_identity = false; /* all series small: skip the list */
  uint64_t* d_huge_scr_offsets = nullptr;
  uint64_t huge_scratch_elems = 0;
  int64_t* d_scr_ts = nullptr;
  double* d_scr_vals = nullptr;
  /* outputs of the last exec */
  double* d_out = nullptr;
  size_t out_elems = 0;
  double* d_counts = nullptr;
  size_t count_elems = 0;
  unsigned long long* d_scanned = nullptr;
  /* shape of the last exec's output */
  uint32_t last_rows = 0;
  int32_t last_grid = 0;
  /* grouped batches are physically relayouted so each group's series are
   * contiguous in HBM (locality for the grouped-run accumulation);
   * perm[i] = original series id of physical row i (empty = identity) */
  std::vector<uint32_t> perm;
};

struct Ctx {
  std::mutex mu;
  bool inited = false;
  int device = 0;
  hipStream_t stream = nullptr;
  hipEvent_t ev_start = nullptr, ev_stop = nullptr;
  double last_kernel_ms = 0.0;
  std::map<uint64_t, Batch> batches;
  uint64_t next_handle = 1;
};

Ctx g_ctx;
}  // namespace

static hipStream_t launch_stream() { return g_ctx.stream; }

/* stream-ordered pool allocation (devalloc.h) */
hipError_t vm_dev_malloc_raw(void** p, size_t n) {
  return hipMallocAsync(p, n, g_ctx.stream);
}
hipError_t vm_dev_free_raw(void* p) {
  if (!p) return hipSuccess;
  return hipFreeAsync(p, g_ctx.stream);
}

hipStream_t vm_ctx_stream(void) { return g_ctx.stream; }

namespace {

int set_err(char* errbuf, size_t len, const char* msg) {
  if (errbuf && len) {
    snprintf(errbuf, len, "%s", msg);
  }
  return 1;
}

int hip_err(char* errbuf, size_t len, const char* what, hipError_t e) {
  if (errbuf && len) {
    snprintf(errbuf, len, "%s: %s", what, hipGetErrorString(e));
  }
  return 2;
}

#define HIP_TRY(expr, what)                              \
  do {                                                   \
    hipError_t _e = (expr);                              \
    if (_e != hipSuccess) return hip_err(errbuf, errbuf_len, what, _e); \
  } while (0)

/* copy series perm[i] of the source CSR into dense row i of the new CSR */
__global__ __launch_bounds__(BLOCK_THREADS) void relayout_kernel(
    const int64_t* src_ts, const double* src_vals, const uint64_t* src_off,
    const uint32_t* perm, const uint64_t* dst_off, uint32_t n_series,
    int64_t* dst_ts, double* dst_vals) {
  const int lane = threadIdx.x % WAVE;
  const int wv = threadIdx.x / WAVE;
  uint32_t wid = blockIdx.x * (BLOCK_THREADS / WAVE) + wv;
  uint32_t stride = gridDim.x * (BLOCK_THREADS / WAVE);
  for (uint32_t i = wid; i < n_series; i += stride) {
    uint32_t s = perm[i];
    uint64_t slo = src_off[s];
    uint64_t dlo = dst_off[i];
    uint64_t n = src_off[s + 1] - slo;
    for (uint64_t k = lane; k < n; k += WAVE) {
      dst_ts[dlo + k] = src_ts[slo + k];
      dst_vals[dlo + k] = src_vals[slo + k];
    }
  }
}

/* Physically reorder a grouped batch so each group's series are contiguous
 * (stable by (group, series)).  b.d_ts/d_vals must hold the ORIGINAL CSR
 * (offsets_orig); on success they point at the relayouted CSR, b.perm maps
 * physical row -> original series, new_offsets holds the physical CSR, and
 * b.d_group_ids is uploaded in physical order.  Identity permutations skip
 * all work.  Returns 0 or an error code. */
static int relayout_by_group(Batch& b, const int32_t* group_ids,
                             const uint64_t* offsets_orig, uint32_t n_series,
                             std::vector<uint64_t>& new_offsets,
                             char* errbuf, size_t errbuf_len) {
  std::vector<uint32_t> perm(n_series);
  for (uint32_t s = 0; s < n_series; s++) perm[s] = s;
  std::stable_sort(perm.begin(), perm.end(), [&](uint32_t a, uint32_t c) {
    return group_ids[a] < group_ids[c];
  });
  bool identity = true;
  for (uint32_t s = 0; s < n_series && identity; s++)
    identity = (perm[s] == s);
  new_offsets.resize(n_series + 1);
  new_offsets[0] = 0;
  for (uint32_t i = 0; i < n_series; i++)
    new_offsets[i + 1] = new_offsets[i] +
                         (offsets_orig[perm[i] + 1] - offsets_orig[perm[i]]);
  if (identity) {
    for (uint32_t i = 0; i <= n_series; i++) new_offsets[i] = offsets_orig[i];
    return 0;
  }
  uint64_t total = offsets_orig[n_series];
  int64_t* d_ts2 = nullptr;
  double* d_vals2 = nullptr;
  uint64_t* d_soff = nullptr;
  uint64_t* d_doff = nullptr;
  uint32_t* d_perm = nullptr;
  HIP_TRY(vm_dev_malloc(&d_ts2, total * 8 + 16), "alloc relayout ts");
  HIP_TRY(vm_dev_malloc(&d_vals2, total * 8 + 16), "alloc relayout vals");
  HIP_TRY(vm_dev_malloc(&d_soff, (size_t)(n_series + 1) * 8), "alloc relayout soff");
  HIP_TRY(vm_dev_malloc(&d_doff, (size_t)(n_series + 1) * 8), "alloc relayout doff");
  HIP_TRY(vm_dev_malloc(&d_perm, (size_t)n_series * 4), "alloc relayout perm");
  hipStream_t st = g_ctx.stream;
  HIP_TRY(hipMemcpyAsync(d_soff, offsets_orig, (size_t)(n_series + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul soff");
  HIP_TRY(hipMemcpyAsync(d_doff, new_offsets.data(), (size_t)(n_series + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul doff");
  HIP_TRY(hipMemcpyAsync(d_perm, perm.data(), (size_t)n_series * 4,
                         hipMemcpyHostToDevice, st), "ul perm");
  uint32_t blocks = std::min<uint32_t>(
      (n_series + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK, 4096);
  hipLaunchKernelGGL(relayout_kernel, dim3(blocks), dim3(BLOCK_THREADS), 0, st,
                     b.d_ts, b.d_vals, d_soff, d_perm, d_doff, n_series,
                     d_ts2, d_vals2);
  HIP_TRY(hipStreamSynchronize(st), "sync relayout");
  hipError_t kerr = hipGetLastError();
  (void)vm_dev_free(d_soff);
  (void)vm_dev_free(d_doff);
  (void)vm_dev_free(d_perm);
  if (kerr != hipSuccess) {
    (void)vm_dev_free(d_ts2);
    (void)vm_dev_free(d_vals2);
    return hip_err(errbuf, errbuf_len, "relayout kernel", kerr);
  }
  (void)vm_dev_free(b.d_ts);
  (void)vm_dev_free(b.d_vals);
  b.d_ts = d_ts2;
  b.d_vals = d_vals2;
  b.perm = std::move(perm);
  return 0;
}

int build_si_index(Batch& b, char* errbuf, size_t errbuf_len) {
  HIP_TRY(vm_dev_malloc(&b.d_si, (size_t)b.n_series * 8), "alloc si index");
  uint32_t blocks = std::min<uint32_t>(
      (b.n_series + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK, 4096);
  hipLaunchKernelGGL(si_prep_kernel, dim3(blocks), dim3(BLOCK_THREADS), 0,
                     g_ctx.stream, b.d_ts, b.d_offsets, b.n_series, b.d_si);
  HIP_TRY(hipStreamSynchronize(g_ctx.stream), "sync si index");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess)
    return hip_err(errbuf, errbuf_len, "si index kernel", kerr);
  return 0;
}

void free_batch(Batch& b) {
  (void)vm_dev_free(b.d_ts);
  (void)vm_dev_free(b.d_vals);
  (void)vm_dev_free(b.d_offsets);
  (void)vm_dev_free(b.d_group_ids);
  (void)vm_dev_free(b.d_si);
  (void)vm_dev_free(b.d_wave_list);
  (void)vm_dev_free(b.d_block_list);
  (void)vm_dev_free(b.d_huge_list);
  (void)vm_dev_free(b.d_huge_scr_offsets);
  (void)vm_dev_free(b.d_scr_ts);
  (void)vm_dev_free(b.d_scr_vals);
  (void)vm_dev_free(b.d_out);
  (void)vm_dev_free(b.d_counts);
  (void)vm_dev_free(b.d_scanned);
  b = Batch();
}

}  // namespace

extern "C" {

int vmgpu_init(const int* device_ids, int n_devices) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (g_ctx.inited) return 0;
  if (n_devices != 1) return 1; /* one process per GPU (RCCL scaling model) */
  int dev = device_ids ? device_ids[0] : 0;
  if (hipSetDevice(dev) != hipSuccess) return 2;
  if (hipStreamCreate(&g_ctx.stream) != hipSuccess) return 3;
  {
    /* retain freed device memory in the pool: repeat queries re-use the
     * same multi-GB carve instead of re-mapping it */
    hipMemPool_t pool = nullptr;
    if (hipDeviceGetDefaultMemPool(&pool, dev) == hipSuccess && pool) {
      uint64_t thr = ~0ULL;
      (void)hipMemPoolSetAttribute(pool, hipMemPoolAttrReleaseThreshold, &thr);
    }
  }
  if (hipEventCreate(&g_ctx.ev_start) != hipSuccess) return 4;
  if (hipEventCreate(&g_ctx.ev_stop) != hipSuccess) return 5;
  g_ctx.device = dev;
  g_ctx.inited = true;
  return 0;
}

int vmgpu_shutdown(void) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return 0;
  for (auto& kv : g_ctx.batches) free_batch(kv.second);
  g_ctx.batches.clear();
  (void)hipEventDestroy(g_ctx.ev_start);
  (void)hipEventDestroy(g_ctx.ev_stop);
  (void)hipStreamDestroy(g_ctx.stream);
  g_ctx.stream = nullptr;
  g_ctx.inited = false;
  return 0;
}

int vmgpu_batch_create(const int64_t* ts, const double* vals,
                       const uint64_t* offsets, uint32_t n_series,
                       const int32_t* group_ids, uint32_t n_groups,
                       uint64_t* out_handle,
                       char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  if (!ts || !vals || !offsets || !out_handle || n_series == 0)
    return set_err(errbuf, errbuf_len, "vmgpu: bad args");
  Batch b;
  b.n_series = n_series;
  b.n_groups = n_groups;
  b.n_samples = offsets[n_series];

  /* +16 B slack: the pipe kernel's clamped tail PAIR load may touch one
   * element past the last series */
  HIP_TRY(vm_dev_malloc(&b.d_ts, b.n_samples * sizeof(int64_t) + 16), "alloc ts");
  HIP_TRY(vm_dev_malloc(&b.d_vals, b.n_samples * sizeof(double) + 16), "alloc vals");
  HIP_TRY(hipMemcpy(b.d_ts, ts, b.n_samples * sizeof(int64_t), hipMemcpyHostToDevice), "upload ts");
  HIP_TRY(hipMemcpy(b.d_vals, vals, b.n_samples * sizeof(double), hipMemcpyHostToDevice), "upload vals");

  /* grouped batches: physically relayout so each group's series are
   * contiguous in HBM (see relayout_by_group).  eff_* describe the
   * PHYSICAL CSR used by the kernels. */
  std::vector<uint64_t> phys_offsets;
  std::vector<int32_t> phys_gids;
  const uint64_t* eff_offsets = offsets;
  const int32_t* eff_gids = group_ids;
  if (group_ids && n_groups > 0) {
    int rrc = relayout_by_group(b, group_ids, offsets, n_series,
                                phys_offsets, errbuf, errbuf_len);
    if (rrc != 0) {
      free_batch(b);
      return rrc;
    }
    eff_offsets = phys_offsets.data();
    if (!b.perm.empty()) {
      phys_gids.resize(n_series);
      for (uint32_t i = 0; i < n_series; i++)
        phys_gids[i] = group_ids[b.perm[i]];
      eff_gids = phys_gids.data();
    }
  }

  /* partition series by length (host pass over the physical offsets) */
  std::vector<uint32_t> wave_list, block_list, huge_list;
  std::vector<uint64_t> huge_scr_off;
  uint64_t huge_total = 0;
  for (uint32_t s = 0; s < n_series; s++) {
    uint64_t n = eff_offsets[s + 1] - eff_offsets[s];
    if (n <= CHUNK_WAVE) {
      wave_list.push_back(s);
      if ((uint32_t)n > b.max_wave_len) b.max_wave_len = (uint32_t)n;
    } else if (n <= CHUNK_BLOCK) {
      block_list.push_back(s);
      if ((uint32_t)n > b.max_block_len) b.max_block_len = (uint32_t)n;
    } else {
      huge_list.push_back(s);
      huge_scr_off.push_back(huge_total);
      huge_total += n;
    }
  }
  b.n_wave = (uint32_t)wave_list.size();
  b.n_block = (uint32_t)block_list.size();
  b.n_huge = (uint32_t)huge_list.size();
  b.wave_is_identity = (b.n_wave == n_series);
  b.huge_scratch_elems = huge_total;

  HIP_TRY(vm_dev_malloc(&b.d_offsets, (n_series + 1) * sizeof(uint64_t)), "alloc offsets");
  HIP_TRY(hipMemcpy(b.d_offsets, eff_offsets, (n_series + 1) * sizeof(uint64_t), hipMemcpyHostToDevice), "upload offsets");
  if (group_ids) {
    HIP_TRY(vm_dev_malloc(&b.d_group_ids, n_series * sizeof(int32_t)), "alloc gids");
    HIP_TRY(hipMemcpy(b.d_group_ids, eff_gids, n_series * sizeof(int32_t), hipMemcpyHostToDevice), "upload gids");
  }
  if (!b.wave_is_identity && b.n_wave) {
    HIP_TRY(vm_dev_malloc(&b.d_wave_list, b.n_wave * 4), "alloc wave list");
    HIP_TRY(hipMemcpy(b.d_wave_list, wave_list.data(), b.n_wave * 4, hipMemcpyHostToDevice), "upload wave list");
  }
  if (b.n_block) {
    HIP_TRY(vm_dev_malloc(&b.d_block_list, b.n_block * 4), "alloc block list");
    HIP_TRY(hipMemcpy(b.d_block_list, block_list.data(), b.n_block * 4, hipMemcpyHostToDevice), "upload block list");
  }
  if (b.n_huge) {
    HIP_TRY(vm_dev_malloc(&b.d_huge_list, b.n_huge * 4), "alloc huge list");
    HIP_TRY(hipMemcpy(b.d_huge_list, huge_list.data(), b.n_huge * 4, hipMemcpyHostToDevice), "upload huge list");
    HIP_TRY(vm_dev_malloc(&b.d_huge_scr_offsets, b.n_huge * 8), "alloc huge offsets");
    HIP_TRY(hipMemcpy(b.d_huge_scr_offsets, huge_scr_off.data(), b.n_huge * 8, hipMemcpyHostToDevice), "upload huge offsets");
    HIP_TRY(vm_dev_malloc(&b.d_scr_ts, huge_total * sizeof(int64_t)), "alloc scratch ts");
    HIP_TRY(vm_dev_malloc(&b.d_scr_vals, huge_total * sizeof(double)), "alloc scratch vals");
  }
  HIP_TRY(vm_dev_malloc(&b.d_scanned, sizeof(unsigned long long)), "alloc scanned");
  {
    int src_rc = build_si_index(b, errbuf, errbuf_len);
    if (src_rc != 0) { free_batch(b); return src_rc; }
  }

  uint64_t h = g_ctx.next_handle++;
  g_ctx.batches[h] = b;
  *out_handle = h;
  return 0;
}

/* decode.hip seam: decode blocks, merge+dedup, compact — device-resident */
int vmdec_decode_merge_device(
    const uint8_t* payload, uint64_t payload_len,
    const vmgpu_block_desc* blocks, uint32_t n_blocks, uint64_t total_rows,
    const uint32_t* series_block_start, uint32_t n_series,
    int64_t dedup_interval, hipStream_t st,
    int64_t** out_d_ts, double** out_d_vals, uint64_t* h_final_offsets,
    char* errbuf, size_t errbuf_len);

/* Cold-cache fetch path fused on device (SURVEY.md §8f(1)+(2) motivation):
 * compressed-block payload in, resident rollup batch out — the decoded
 * columns never cross PCIe.  out_offsets[n_series+1] reports the merged
 * CSR so the host can interpret per-series results. */
int vmgpu_batch_create_from_blocks(
    const uint8_t* payload, uint64_t payload_len,
    const vmgpu_block_desc* blocks, uint32_t n_blocks, uint64_t total_rows,
    const uint32_t* series_block_start, uint32_t n_series,
    int64_t dedup_interval, const int32_t* group_ids, uint32_t n_groups,
    uint64_t* out_handle, uint64_t* out_offsets,
    char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  if (!out_handle || !out_offsets || n_series == 0)
    return set_err(errbuf, errbuf_len, "vmgpu: bad args");
  int64_t* d_ts = nullptr;
  double* d_vals = nullptr;
  int rc = vmdec_decode_merge_device(payload, payload_len, blocks, n_blocks,
                                     total_rows, series_block_start, n_series,
                                     dedup_interval, g_ctx.stream,
                                     &d_ts, &d_vals, out_offsets,
                                     errbuf, errbuf_len);
  if (rc != 0) return rc;

  Batch b;
  b.n_series = n_series;
  b.n_groups = n_groups;
  b.n_samples = out_offsets[n_series];
  b.d_ts = d_ts;
  b.d_vals = d_vals;

  /* grouped: physically relayout (see relayout_by_group); out_offsets
   * reported to the caller stay in ORIGINAL series order */
  std::vector<uint64_t> phys_offsets;
  std::vector<int32_t> phys_gids;
  const uint64_t* eff_offsets = out_offsets;
  const int32_t* eff_gids = group_ids;
  if (group_ids && n_groups > 0) {
    int rrc = relayout_by_group(b, group_ids, out_offsets, n_series,
                                phys_offsets, errbuf, errbuf_len);
    if (rrc != 0) {
      free_batch(b);
      return rrc;
    }
    eff_offsets = phys_offsets.data();
    if (!b.perm.empty()) {
      phys_gids.resize(n_series);
      for (uint32_t i = 0; i < n_series; i++)
        phys_gids[i] = group_ids[b.perm[i]];
      eff_gids = phys_gids.data();
    }
  }

  std::vector<uint32_t> wave_list, block_list, huge_list;
  std::vector<uint64_t> huge_scr_off;
  uint64_t huge_total = 0;
  for (uint32_t s = 0; s < n_series; s++) {
    uint64_t n = eff_offsets[s + 1] - eff_offsets[s];
    if (n <= CHUNK_WAVE) {
      wave_list.push_back(s);
      if ((uint32_t)n > b.max_wave_len) b.max_wave_len = (uint32_t)n;
    } else if (n <= CHUNK_BLOCK) {
      block_list.push_back(s);
    } else {
      huge_list.push_back(s);
      huge_scr_off.push_back(huge_total);
      huge_total += n;
    }
  }
  b.n_wave = (uint32_t)wave_list.size();
  b.n_block = (uint32_t)block_list.size();
  b.n_huge = (uint32_t)huge_list.size();
  b.wave_is_identity = (b.n_wave == n_series);
  b.huge_scratch_elems = huge_total;

#define FBB_TRY(expr, what)                                                  \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) { free_batch(b);                                   \
      return hip_err(errbuf, errbuf_len, what, _e); }                        \
  } while (0)

  FBB_TRY(vm_dev_malloc(&b.d_offsets, (n_series + 1) * sizeof(uint64_t)), "alloc offsets");
  FBB_TRY(hipMemcpy(b.d_offsets, eff_offsets, (n_series + 1) * sizeof(uint64_t),
                    hipMemcpyHostToDevice), "upload offsets");
  if (group_ids) {
    FBB_TRY(vm_dev_malloc(&b.d_group_ids, n_series * sizeof(int32_t)), "alloc gids");
    FBB_TRY(hipMemcpy(b.d_group_ids, eff_gids, n_series * sizeof(int32_t),
                      hipMemcpyHostToDevice), "upload gids");
  }
  if (!b.wave_is_identity && b.n_wave) {
    FBB_TRY(vm_dev_malloc(&b.d_wave_list, b.n_wave * 4), "alloc wave list");
    FBB_TRY(hipMemcpy(b.d_wave_list, wave_list.data(), b.n_wave * 4,
                      hipMemcpyHostToDevice), "upload wave list");
  }
  if (b.n_block) {
    FBB_TRY(vm_dev_malloc(&b.d_block_list, b.n_block * 4), "alloc block list");
    FBB_TRY(hipMemcpy(b.d_block_list, block_list.data(), b.n_block * 4,
                      hipMemcpyHostToDevice), "upload block list");
  }
  if (b.n_huge) {
    FBB_TRY(vm_dev_malloc(&b.d_huge_list, b.n_huge * 4), "alloc huge list");
    FBB_TRY(hipMemcpy(b.d_huge_list, huge_list.data(), b.n_huge * 4,
                      hipMemcpyHostToDevice), "upload huge list");
    FBB_TRY(vm_dev_malloc(&b.d_huge_scr_offsets, b.n_huge * 8), "alloc huge offsets");
    FBB_TRY(hipMemcpy(b.d_huge_scr_offsets, huge_scr_off.data(), b.n_huge * 8,
                      hipMemcpyHostToDevice), "upload huge offsets");
    FBB_TRY(vm_dev_malloc(&b.d_scr_ts, huge_total * sizeof(int64_t)), "alloc scratch ts");
    FBB_TRY(vm_dev_malloc(&b.d_scr_vals, huge_total * sizeof(double)), "alloc scratch vals");
  }
  FBB_TRY(vm_dev_malloc(&b.d_scanned, sizeof(unsigned long long)), "alloc scanned");
#undef FBB_TRY
  {
    int src_rc = build_si_index(b, errbuf, errbuf_len);
    if (src_rc != 0) { free_batch(b); return src_rc; }
  }

  uint64_t h = g_ctx.next_handle++;
  g_ctx.batches[h] = b;
  *out_handle = h;
  return 0;
}

/* Native descriptor build from a packed block stream (vmgpu.h): the C
 * parse that the reference does in Go when unpackWorker walks its
 * []sortedBlock (netstorage.go:423-614).  The stream buffer itself is the
 * device payload — data offsets point into it, nothing is re-copied on the
 * host. */
int vmgpu_batch_create_packed(
    const uint8_t* packed, uint64_t packed_len, uint64_t n_blocks,
    const uint32_t* series_block_start, uint32_t n_series,
    int64_t dedup_interval, const int32_t* group_ids, uint32_t n_groups,
    uint64_t* out_handle, uint64_t* out_offsets,
    char* errbuf, size_t errbuf_len) {
  if (!packed || !series_block_start || !out_handle || !out_offsets ||
      n_series == 0 || n_blocks == 0)
    return set_err(errbuf, errbuf_len, "vmgpu: bad args");
  std::vector<vmgpu_block_desc> descs(n_blocks);
  uint64_t off = 0;
  uint64_t total_rows = 0;
  for (uint64_t i = 0; i < n_blocks; i++) {
    if (off + sizeof(vmgpu_packed_block_hdr) > packed_len)
      return set_err(errbuf, errbuf_len, "vmgpu: packed stream truncated");
    vmgpu_packed_block_hdr h;
    memcpy(&h, packed + off, sizeof(h)); /* stream need not be aligned */
    off += sizeof(h);
    vmgpu_block_desc& d = descs[i];
    d.ts_data_off = off;
    d.ts_data_len = h.ts_data_len;
    off += h.ts_data_len;
    d.val_data_off = off;
    d.val_data_len = h.val_data_len;
    off += h.val_data_len;
    if (off > packed_len)
      return set_err(errbuf, errbuf_len, "vmgpu: packed stream truncated");
    if (h.rows == 0 || h.rows > 8192)
      return set_err(errbuf, errbuf_len, "vmgpu: bad packed rows");
    d.out_off = total_rows;
    d.min_timestamp = h.min_timestamp;
    d.max_timestamp = h.max_timestamp;
    d.first_value = h.first_value;
    d.scale = h.scale;
    d.e10 = (h.scale == 0)
                ? 1.0
                : pow(10.0, (double)(h.scale < 0 ? -h.scale : h.scale));
    d.rows = h.rows;
    d.ts_mt = h.ts_mt;
    d.val_mt = h.val_mt;
    d.precision_bits = h.precision_bits;
    d._pad = 0;
    total_rows += h.rows;
  }
  return vmgpu_batch_create_from_blocks(
      packed, packed_len, descs.data(), (uint32_t)n_blocks, total_rows,
      series_block_start, n_series, dedup_interval, group_ids, n_groups,
      out_handle, out_offsets, errbuf, errbuf_len);
}

/* Pinned host memory for PCIe-rate payload/result staging (the cgo
 * production layer would hold these in a pool, like netstorage's
 * result buffer pools). */
int vmgpu_host_alloc(uint64_t nbytes, void** out_ptr) {
  if (!out_ptr || nbytes == 0) return 1;
  *out_ptr = nullptr;
  return hipHostMalloc(out_ptr, nbytes, hipHostMallocDefault) == hipSuccess
             ? 0 : 2;
}

int vmgpu_host_free(void* ptr) {
  return hipHostFree(ptr) == hipSuccess ? 0 : 1;
}

/* Physical-row -> original-series mapping of a (relayouted) grouped batch:
 * out_perm[n_series].  Identity batches fill 0..n-1.  Callers need this to
 * place per-series outputs of vmgpu_rollup_exec back in request order. */
int vmgpu_batch_perm(uint64_t handle, uint32_t* out_perm) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  auto it = g_ctx.batches.find(handle);
  if (it == g_ctx.batches.end() || !out_perm) return 1;
  const Batch& b = it->second;
  if (b.perm.empty()) {
    for (uint32_t i = 0; i < b.n_series; i++) out_perm[i] = i;
  } else {
    for (uint32_t i = 0; i < b.n_series; i++) out_perm[i] = b.perm[i];
  }
  return 0;
}

int vmgpu_batch_destroy(uint64_t handle) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  auto it = g_ctx.batches.find(handle);
  if (it == g_ctx.batches.end()) return 1;
  free_batch(it->second);
  g_ctx.batches.erase(it);
  return 0;
}

int vmgpu_rollup_exec(const vmgpu_plan* plan, uint64_t handle,
                      double* out, double* out_counts,
                      uint64_t* out_samples_scanned,
                      char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  auto it = g_ctx.batches.find(handle);
  if (it == g_ctx.batches.end()) return set_err(errbuf, errbuf_len, "vmgpu: bad handle");
  Batch& b = it->second;
  if (plan->step <= 0 || plan->start > plan->end)
    return set_err(errbuf, errbuf_len, "vmgpu: bad grid");
  int64_t n_grid64 = 1 + (plan->end - plan->start) / plan->step;
  if (n_grid64 > (int64_t)1 << 30)
    return set_err(errbuf, errbuf_len, "vmgpu: grid too large");
  int32_t n_grid = (int32_t)n_grid64;
  bool grouped = plan->aggr != VMGPU_AGGR_NONE;
  if (grouped && (!b.d_group_ids || b.n_groups == 0))
    return set_err(errbuf, errbuf_len, "vmgpu: aggr plan needs group_ids");

  size_t out_elems = grouped ? (size_t)b.n_groups * n_grid
                             : (size_t)b.n_series * n_grid;
  size_t count_elems = grouped ? out_elems : 0;
  if (b.out_elems < out_elems) {
    (void)vm_dev_free(b.d_out);
    b.d_out = nullptr;
    HIP_TRY(vm_dev_malloc(&b.d_out, out_elems * sizeof(double)), "alloc out");
    b.out_elems = out_elems;
  }
  if (count_elems && b.count_elems < count_elems) {
    (void)vm_dev_free(b.d_counts);
    b.d_counts = nullptr;
    HIP_TRY(vm_dev_malloc(&b.d_counts, count_elems * sizeof(double)), "alloc counts");
    b.count_elems = count_elems;
  }
  HIP_TRY(hipMemsetAsync(b.d_scanned, 0, 8, g_ctx.stream), "zero scanned");

  KPlan p;
  p.start = plan->start;
  p.end = plan->end;
  p.step = plan->step;
  p.window = plan->window;
  p.lookback_delta = plan->lookback_delta;
  p.min_staleness = plan->min_staleness_interval;
  p.max_staleness = plan->max_staleness_interval;
  p.n_grid = n_grid;
  p.func = plan->func;
  p.aggr = plan->aggr;
  p.sspc = plan->samples_scanned_per_call;
  p.may_adjust = plan->may_adjust_window;
  p.is_default = plan->is_default_rollup;
  p.rcr = plan->remove_counter_resets;
  p.drop_stale = plan->drop_stale_nans;
  p.pre_func = plan->pre_func;
  p.chunk_wave = (int32_t)std::min<uint32_t>(
      CHUNK_WAVE, std::max<uint32_t>(64, (b.max_wave_len + 63) & ~63u));
  p.chunk_block = (int32_t)std::min<uint32_t>(
      CHUNK_BLOCK, std::max<uint32_t>(64, (b.max_block_len + 63) & ~63u));
  /* rate boundary caches (see the wave kernel): the u16 j-cache
   * (shared-boundary path) is the measured default — both scatter
   * variants benched SLOWER at config 2 (full Et/Ev/J: 3.27 ms from LDS
   * occupancy loss; J-only scatter: 2.77 ms from the boundary-fixup +
   * divergent write cost) vs 2.51 ms for the j-cache.  Mode 2 remains
   * selectable for grids too large for the u16 cache. */
  /* pipe eligibility decides the jbuf flavor: the pipelined kernel builds
   * the full ext J map during its staging scan (mode 2), everything else
   * uses the probe-filled u16 j-cache (mode 1). */
  bool use_pipe = false;
  if ((plan->func == VMF_RATE || plan->func == VMF_DERIV_FAST) && b.n_wave &&
      b.max_wave_len <= (uint32_t)(PIPE_CHUNKS * WAVE) && plan->pre_func == 0 &&
      plan->window > 0 && plan->step > 0 && plan->window % plan->step == 0 &&
      n_grid > 1) {
    static int pipe_disabled = -1;
    if (pipe_disabled < 0) {
      const char* e = getenv("VMGPU_DISABLE_PIPE");
      pipe_disabled = (e && e[0] == '1') ? 1 : 0;
    }
#ifdef VMGPU_ABL_GACC
    pipe_disabled = 1; /* the GACC A/B runs through the wave kernel */
#endif
    use_pipe = !pipe_disabled;
  }
  p.jbuf_elems = 0;
  p.jbuf_mode = 0;
  if ((plan->func == VMF_RATE || plan->func == VMF_DERIV_FAST) && n_grid > 1) {
    size_t base = (size_t)WAVES_PER_BLOCK * ((size_t)p.chunk_wave * 16 + 256);
    int64_t dgp = (plan->window > 0 && plan->step > 0 &&
                   plan->window % plan->step == 0)
                      ? plan->window / plan->step : 0;
    /* A/B measured (round 2): the in-scan J scatter costs MORE inside the
     * staging chain (+0.46 ms at config 2) than the probe j-cache it
     * replaces (~0.42 ms in an overlappable phase) — the divergent range
     * writes and boundary fixups sit on the serial scan path.  Probe mode
     * stays the default; VMGPU_PIPE_SCATTER=1 re-enables the scatter. */
#ifdef VMGPU_PIPE_SCATTER
    static int pipe_scatter = -1;
    if (pipe_scatter < 0) {
      const char* e2 = getenv("VMGPU_PIPE_SCATTER");
      pipe_scatter = (e2 && e2[0] == '1') ? 1 : 0;
    }
#else
    const int pipe_scatter = 0;
#endif
    if (use_pipe && pipe_scatter && dgp > 0 && n_grid + dgp <= 32000 &&
        base + WAVES_PER_BLOCK * vm_jbuf_bytes(2, (int32_t)(n_grid + dgp)) <=
            64 * 1024) {
      p.jbuf_mode = 2;
      p.jbuf_elems = (int32_t)(n_grid + dgp);
    } else if (n_grid <= 4096 &&
        base + WAVES_PER_BLOCK * vm_jbuf_bytes(1, n_grid) <= 64 * 1024) {
      p.jbuf_mode = 1;
      p.jbuf_elems = n_grid;
    } else if (dgp > 0 && n_grid + dgp <= 32000) {
      int32_t elems = (int32_t)(n_grid + dgp);
      if (base + WAVES_PER_BLOCK * vm_jbuf_bytes(2, elems) <= 64 * 1024) {
        p.jbuf_mode = 2;
        p.jbuf_elems = elems;
      }
    }
  }
  /* the pipe kernel carries only the j-cache rate path */
  if (p.jbuf_mode == 0) use_pipe = false;
  p.arg = plan->arg;
  p.arg2 = plan->arg2;

  KIO io;
  io.ts = b.d_ts;
  io.vals = b.d_vals;
  io.offsets = b.d_offsets;
  io.group_ids = b.d_group_ids;
  io.series_si = b.d_si;
  io.out = b.d_out;
  io.out_counts = b.d_counts;
  io.samples_scanned = b.d_scanned;
  io.scr_ts = b.d_scr_ts;
  io.scr_vals = b.d_scr_vals;
  io.scr_offsets = b.d_huge_scr_offsets;

  if (grouped) {
    uint64_t n = out_elems;
    int blocks = (int)std::min<uint64_t>((n + 255) / 256, 2048);
    hipLaunchKernelGGL(aggr_init_kernel, dim3(blocks), dim3(256), 0, g_ctx.stream,
                       b.d_out, b.d_counts, n, plan->aggr);
  }

  HIP_TRY(hipEventRecord(g_ctx.ev_start, g_ctx.stream), "event start");

  /* compile-time specializations for the hot functions (see launch_rollup) */
  if (b.n_wave) {
    KIO w = io;
    w.series_sel = b.wave_is_identity ? nullptr : b.d_wave_list;
    w.n_sel = b.n_wave;
    uint32_t blocks = std::min<uint32_t>((b.n_wave + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK,
                                         MAX_WAVE_BLOCKS);
    size_t lds = (size_t)WAVES_PER_BLOCK *
                 ((size_t)p.chunk_wave * 16 + 256 +
                  vm_jbuf_bytes(p.jbuf_mode, p.jbuf_elems));
    /* software-pipelined register-staged variant for short series (the
     * config-2 flagship shape); VMGPU_DISABLE_PIPE=1 A/Bs the wave kernel */
    launch_rollup(use_pipe ? 3 : 0, blocks, lds, p, w);
  }
  if (b.n_block) {
    KIO w = io;
    w.series_sel = b.d_block_list;
    w.n_sel = b.n_block;
    uint32_t blocks = std::min<uint32_t>(b.n_block, 2048);
    size_t lds = vm_block_lds_bytes(p.chunk_block, p.jbuf_mode, p.n_grid,
                                    nullptr);
    launch_rollup(1, blocks, lds, p, w);
  }
  if (b.n_huge) {
    KIO w = io;
    w.series_sel = b.d_huge_list;
    w.n_sel = b.n_huge;
    uint32_t blocks = std::min<uint32_t>(b.n_huge, 2048);
    launch_rollup(2, blocks, 0, p, w);
  }

  HIP_TRY(hipEventRecord(g_ctx.ev_stop, g_ctx.stream), "event stop");

  if (grouped && !plan->skip_finalize) {
    uint64_t n = out_elems;
    int blocks = (int)std::min<uint64_t>((n + 255) / 256, 2048);
    hipLaunchKernelGGL(aggr_finalize_kernel, dim3(blocks), dim3(256), 0, g_ctx.stream,
                       b.d_out, b.d_counts, n, plan->aggr);
  }

  if (out) {
    HIP_TRY(hipMemcpyAsync(out, b.d_out, out_elems * sizeof(double),
                           hipMemcpyDeviceToHost, g_ctx.stream), "download out");
  }
  if (out_counts && count_elems) {
    HIP_TRY(hipMemcpyAsync(out_counts, b.d_counts, count_elems * sizeof(double),
                           hipMemcpyDeviceToHost, g_ctx.stream), "download counts");
  }
  unsigned long long scanned_h = 0;
  HIP_TRY(hipMemcpyAsync(&scanned_h, b.d_scanned, 8, hipMemcpyDeviceToHost, g_ctx.stream),
          "download scanned");
  HIP_TRY(hipStreamSynchronize(g_ctx.stream), "sync");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess) return hip_err(errbuf, errbuf_len, "kernel", kerr);

  float ms = 0;
  HIP_TRY(hipEventElapsedTime(&ms, g_ctx.ev_start, g_ctx.ev_stop), "event elapsed");
  g_ctx.last_kernel_ms = (double)ms;

  if (out_samples_scanned) *out_samples_scanned = (uint64_t)scanned_h;
  b.last_rows = grouped ? b.n_groups : b.n_series;
  b.last_grid = n_grid;
  return 0;
}

int vmgpu_batch_fetch_out(uint64_t handle, double* dst, size_t n_elems,
                          double* dst_counts, size_t n_count_elems) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  auto it = g_ctx.batches.find(handle);
  if (it == g_ctx.batches.end()) return 1;
  Batch& b = it->second;
  if (dst && n_elems) {
    if (n_elems > b.out_elems) return 2;
    if (hipMemcpy(dst, b.d_out, n_elems * sizeof(double), hipMemcpyDeviceToHost) != hipSuccess)
      return 3;
  }
  if (dst_counts && n_count_elems) {
    if (n_count_elems > b.count_elems) return 4;
    if (hipMemcpy(dst_counts, b.d_counts, n_count_elems * sizeof(double),
                  hipMemcpyDeviceToHost) != hipSuccess)
      return 5;
  }
  return 0;
}

void vmgpu_aggr_finalize_host(int32_t aggr, double* values, double* counts,
                              uint64_t n_elems) {
  const double dnan = __builtin_nan("");
  for (uint64_t i = 0; i < n_elems; i++) {
    double c = counts ? counts[i] : 0.0;
    switch (aggr) {
      case VMGPU_AGGR_SUM:
      case VMGPU_AGGR_MIN:
      case VMGPU_AGGR_MAX:
      case VMGPU_AGGR_SUM2:
        if (c == 0) values[i] = dnan;
        break;
      case VMGPU_AGGR_AVG:
        values[i] = (c == 0) ? dnan : values[i] / c;
        break;
      case VMGPU_AGGR_COUNT:
        if (values[i] == 0) values[i] = dnan;
        break;
      case VMGPU_AGGR_GROUP:
        values[i] = (values[i] == 0) ? dnan : 1.0;
        break;
      case VMGPU_AGGR_GEOMEAN:
        values[i] = (c == 0) ? dnan : pow(values[i], 1.0 / c);
        break;
      default:
        break;
    }
  }
}

int vmgpu_rollup_eval(const vmgpu_plan* plan,
                      const int64_t* ts, const double* vals,
                      const uint64_t* offsets, uint32_t n_series,
                      const int32_t* group_ids, uint32_t n_groups,
                      double* out, double* out_counts,
                      uint64_t* out_samples_scanned,
                      char* errbuf, size_t errbuf_len) {
  uint64_t h = 0;
  int rc = vmgpu_batch_create(ts, vals, offsets, n_series, group_ids, n_groups,
                              &h, errbuf, errbuf_len);
  if (rc) return rc;
  rc = vmgpu_rollup_exec(plan, h, out, out_counts, out_samples_scanned,
                         errbuf, errbuf_len);
  vmgpu_batch_destroy(h);
  return rc;
}


int vmgpu_topk_range(uint64_t handle, double k, int32_t summary_op,
                     int32_t reverse, int64_t* out_sel, int64_t* out_n_sel,
                     double* out_remaining, char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  auto it = g_ctx.batches.find(handle);
  if (it == g_ctx.batches.end()) return set_err(errbuf, errbuf_len, "vmgpu: bad handle");
  Batch& b = it->second;
  if (!b.d_out || b.last_rows == 0)
    return set_err(errbuf, errbuf_len, "vmgpu: no evaluated output on this batch");
  uint32_t n = b.last_rows;
  int32_t n_grid = b.last_grid;
  uint32_t kk = 0;
  if (k == k && k > 0) kk = (k > (double)n) ? n : (uint32_t)k;
  if (out_n_sel) *out_n_sel = 0;
  if (kk == 0) return 0;

  hipStream_t st = g_ctx.stream;
  unsigned long long* d_keys = nullptr;
  uint32_t* d_hist = nullptr;
  uint32_t* d_misc = nullptr; /* counters */
  uint32_t* d_sel = nullptr;
  HIP_TRY(vm_dev_malloc(&d_keys, (size_t)n * 8), "alloc keys");
  HIP_TRY(vm_dev_malloc(&d_hist, 65536 * 4), "alloc hist");
  HIP_TRY(vm_dev_malloc(&d_misc, 8), "alloc counters");
  HIP_TRY(vm_dev_malloc(&d_sel, (size_t)kk * 4), "alloc sel");
  uint32_t blocks = std::min<uint32_t>((n + BLOCK_THREADS - 1) / BLOCK_THREADS * WAVES_PER_BLOCK, 2048);
  hipLaunchKernelGGL(topk_summary_kernel, dim3(blocks), dim3(BLOCK_THREADS), 0, st,
                     b.d_out, n, n_grid, summary_op, reverse, d_keys);
  /* refine the k-th largest key threshold 16 bits at a time */
  unsigned long long prefix = 0;
  uint32_t need_above = kk; /* how many keys >= threshold so far required */
  std::vector<uint32_t> hist(65536);
  uint32_t above_total = 0;
  for (int shift = 48; shift >= 0; shift -= 16) {
    HIP_TRY(hipMemsetAsync(d_hist, 0, 65536 * 4, st), "zero hist");
    uint32_t hblocks = std::min<uint32_t>((n + 255) / 256, 2048);
    hipLaunchKernelGGL(topk_hist_kernel, dim3(hblocks), dim3(256), 0, st,
                       d_keys, n, prefix, shift, d_hist);
    HIP_TRY(hipMemcpyAsync(hist.data(), d_hist, 65536 * 4, hipMemcpyDeviceToHost, st), "dl hist");
    HIP_TRY(hipStreamSynchronize(st), "sync hist");
    uint32_t cum = 0;
    int bin = 0;
    for (int bb = 65535; bb >= 0; bb--) {
      uint32_t c = hist[bb];
      if (cum + c >= need_above - above_total) { bin = bb; break; }
      cum += c;
    }
    above_total += cum;
    prefix = (prefix << 16) | (unsigned long long)bin;
  }
  unsigned long long kstar = prefix;
  uint32_t ties_quota = need_above - above_total;
  HIP_TRY(hipMemsetAsync(d_misc, 0, 8, st), "zero counters");
  uint32_t cblocks = std::min<uint32_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(topk_collect_kernel, dim3(cblocks), dim3(256), 0, st,
                     d_keys, n, kstar, ties_quota, d_misc, d_sel, kk);
  std::vector<uint32_t> sel(kk);
  uint32_t nsel_h[2];
  HIP_TRY(hipMemcpyAsync(sel.data(), d_sel, (size_t)kk * 4, hipMemcpyDeviceToHost, st), "dl sel");
  HIP_TRY(hipMemcpyAsync(nsel_h, d_misc, 8, hipMemcpyDeviceToHost, st), "dl counters");
  HIP_TRY(hipStreamSynchronize(st), "sync sel");
  uint32_t m = std::min<uint32_t>(nsel_h[0], kk);
  /* order the selection by key descending (reference output order after
   * reverseSeries): download keys of the selected rows */
  std::vector<unsigned long long> skeys(m);
  for (uint32_t x = 0; x < m; x++) {
    HIP_TRY(hipMemcpyAsync(&skeys[x], d_keys + sel[x], 8, hipMemcpyDeviceToHost, st), "dl key");
  }
  HIP_TRY(hipStreamSynchronize(st), "sync keys");
  std::vector<uint32_t> order(m);
  for (uint32_t x = 0; x < m; x++) order[x] = x;
  std::sort(order.begin(), order.end(), [&](uint32_t a, uint32_t bx) {
    if (skeys[a] != skeys[bx]) return skeys[a] > skeys[bx];
    /* ties: descending row id, matching the oracle's reversed ascending
     * sort (the reference's own sort is unstable: tie order unspecified) */
    return sel[a] > sel[bx];
  });
  for (uint32_t x = 0; x < m; x++) out_sel[x] = (int64_t)sel[order[x]];
  if (out_n_sel) *out_n_sel = (int64_t)m;

  if (out_remaining) {
    uint8_t* d_mask = nullptr;
    double* d_rem = nullptr;
    HIP_TRY(vm_dev_malloc(&d_mask, n), "alloc mask");
    HIP_TRY(vm_dev_malloc(&d_rem, (size_t)n_grid * 16), "alloc rem");
    HIP_TRY(hipMemsetAsync(d_mask, 0, n, st), "zero mask");
    HIP_TRY(hipMemsetAsync(d_rem, 0, (size_t)n_grid * 16, st), "zero rem");
    std::vector<uint8_t> mask(n, 0);
    for (uint32_t x = 0; x < m; x++) mask[sel[x]] = 1;
    HIP_TRY(hipMemcpyAsync(d_mask, mask.data(), n, hipMemcpyHostToDevice, st), "ul mask");
    uint32_t rblocks = std::min<uint32_t>(
        (uint32_t)(((size_t)n * n_grid + 255) / 256), 4096u);
    hipLaunchKernelGGL(topk_remaining_kernel, dim3(rblocks), dim3(256), 0, st,
                       b.d_out, n, n_grid, d_mask, d_rem, d_rem + n_grid);
    std::vector<double> rem(2 * (size_t)n_grid);
    HIP_TRY(hipMemcpyAsync(rem.data(), d_rem, (size_t)n_grid * 16, hipMemcpyDeviceToHost, st), "dl rem");
    HIP_TRY(hipStreamSynchronize(st), "sync rem");
    for (int32_t g = 0; g < n_grid; g++)
      out_remaining[g] = (rem[n_grid + g] == 0) ? __builtin_nan("") : rem[g];
    (void)vm_dev_free(d_mask);
    (void)vm_dev_free(d_rem);
  }
  (void)vm_dev_free(d_keys);
  (void)vm_dev_free(d_hist);
  (void)vm_dev_free(d_misc);
  (void)vm_dev_free(d_sel);
  return 0;
}

int vmgpu_topk_pointwise(uint64_t handle, double k, int32_t reverse,
                         double* out, char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  auto it = g_ctx.batches.find(handle);
  if (it == g_ctx.batches.end()) return set_err(errbuf, errbuf_len, "vmgpu: bad handle");
  Batch& b = it->second;
  if (!b.d_out || b.last_rows == 0)
    return set_err(errbuf, errbuf_len, "vmgpu: no evaluated output on this batch");
  uint32_t n = b.last_rows;
  int32_t n_grid = b.last_grid;
  uint32_t kk = 0;
  if (k == k && k > 0) kk = (k > (double)n) ? n : (uint32_t)k;
  hipStream_t st = g_ctx.stream;
  if (kk >= n || n == 0) {
    if (out) {
      HIP_TRY(hipMemcpyAsync(out, b.d_out, (size_t)n * n_grid * 8,
                             hipMemcpyDeviceToHost, st), "dl out");
      HIP_TRY(hipStreamSynchronize(st), "sync");
    }
    return 0;
  }
  const uint32_t CAND_CAP = 8192;
  uint32_t* d_hists = nullptr;
  uint32_t* d_bins = nullptr;
  unsigned long long* d_cand = nullptr;
  unsigned long long* d_kstar = nullptr;
  uint32_t* d_small = nullptr; /* cand_n[n_grid], ties[n_grid], taken[n_grid], overflow */
  HIP_TRY(vm_dev_malloc(&d_hists, (size_t)n_grid * 65536 * 4), "alloc col hists");
  HIP_TRY(vm_dev_malloc(&d_bins, (size_t)n_grid * 8), "alloc bins");
  HIP_TRY(vm_dev_malloc(&d_cand, (size_t)n_grid * CAND_CAP * 8), "alloc cands");
  HIP_TRY(vm_dev_malloc(&d_kstar, (size_t)n_grid * 8), "alloc kstar");
  HIP_TRY(vm_dev_malloc(&d_small, (size_t)n_grid * 12 + 4), "alloc small");
  uint32_t* d_cand_n = d_small;
  uint32_t* d_ties = d_small + n_grid;
  uint32_t* d_taken = d_small + 2 * (size_t)n_grid;
  uint32_t* d_overflow = d_small + 3 * (size_t)n_grid;
  HIP_TRY(hipMemsetAsync(d_hists, 0, (size_t)n_grid * 65536 * 4, st), "zero hists");
  HIP_TRY(hipMemsetAsync(d_small, 0, (size_t)n_grid * 12 + 4, st), "zero small");
  size_t total = (size_t)n * n_grid;
  uint32_t eblocks = (uint32_t)std::min<size_t>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(topk_col_hist_kernel, dim3(eblocks), dim3(256), 0, st,
                     b.d_out, n, n_grid, reverse, d_hists);
  hipLaunchKernelGGL(topk_col_threshold_kernel,
                     dim3((n_grid + 255) / 256), dim3(256), 0, st,
                     d_hists, n_grid, kk, d_bins, d_bins + n_grid);
  hipLaunchKernelGGL(topk_col_candidates_kernel, dim3(eblocks), dim3(256), 0, st,
                     b.d_out, n, n_grid, reverse, d_bins, d_cand, d_cand_n,
                     CAND_CAP, d_overflow);
  uint32_t overflow_h = 0;
  HIP_TRY(hipMemcpyAsync(&overflow_h, d_overflow, 4, hipMemcpyDeviceToHost, st), "dl ovf");
  HIP_TRY(hipStreamSynchronize(st), "sync ovf");
  if (overflow_h) {
    (void)vm_dev_free(d_hists); (void)vm_dev_free(d_bins); (void)vm_dev_free(d_cand);
    (void)vm_dev_free(d_kstar); (void)vm_dev_free(d_small);
    return set_err(errbuf, errbuf_len,
                   "vmgpu: topk candidate overflow (massive ties); unsupported");
  }
  hipLaunchKernelGGL(topk_col_kstar_kernel, dim3(n_grid), dim3(256), 0, st,
                     d_cand, d_cand_n, CAND_CAP, d_bins + n_grid, kk, n_grid,
                     d_kstar, d_ties);
  hipLaunchKernelGGL(topk_col_fill_kernel, dim3(eblocks), dim3(256), 0, st,
                     b.d_out, n, n_grid, reverse, kk, d_bins, d_kstar, d_taken,
                     d_ties);
  if (out) {
    HIP_TRY(hipMemcpyAsync(out, b.d_out, total * 8, hipMemcpyDeviceToHost, st), "dl out");
  }
  HIP_TRY(hipStreamSynchronize(st), "sync fill");
  hipError_t kerr = hipGetLastError();
  (void)vm_dev_free(d_hists); (void)vm_dev_free(d_bins); (void)vm_dev_free(d_cand);
  (void)vm_dev_free(d_kstar); (void)vm_dev_free(d_small);
  if (kerr != hipSuccess) return hip_err(errbuf, errbuf_len, "topk kernel", kerr);
  return 0;
}

int vmgpu_histogram_stat(int32_t mode, const double* bucket_values,
                         const double* les, const uint64_t* group_offsets,
                         uint32_t n_groups, int32_t n_grid, double* out,
                         char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  if (!bucket_values || !les || !group_offsets || !out || n_groups == 0 || n_grid <= 0)
    return set_err(errbuf, errbuf_len, "vmgpu: bad args");
  hipStream_t st = g_ctx.stream;
  uint64_t n_rows = group_offsets[n_groups];
  double* d_bv = nullptr;
  double* d_les = nullptr;
  uint64_t* d_off = nullptr;
  double* d_out = nullptr;
  size_t out_elems = (size_t)n_groups * n_grid;
  HIP_TRY(vm_dev_malloc(&d_bv, (size_t)n_rows * n_grid * 8), "alloc bv");
  HIP_TRY(vm_dev_malloc(&d_les, (size_t)n_rows * 8), "alloc les");
  HIP_TRY(vm_dev_malloc(&d_off, (size_t)(n_groups + 1) * 8), "alloc off");
  HIP_TRY(vm_dev_malloc(&d_out, out_elems * 8), "alloc out");
  HIP_TRY(hipMemcpyAsync(d_bv, bucket_values, (size_t)n_rows * n_grid * 8,
                         hipMemcpyHostToDevice, st), "ul bv");
  HIP_TRY(hipMemcpyAsync(d_les, les, (size_t)n_rows * 8, hipMemcpyHostToDevice, st), "ul les");
  HIP_TRY(hipMemcpyAsync(d_off, group_offsets, (size_t)(n_groups + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul off");
  uint32_t blocks = (uint32_t)std::min<size_t>((out_elems + 255) / 256, 4096);
  hipLaunchKernelGGL(hstat_kernel, dim3(blocks), dim3(256), 0, st, mode, d_bv,
                     d_les, d_off, (int64_t)n_groups, n_grid, d_out);
  HIP_TRY(hipMemcpyAsync(out, d_out, out_elems * 8, hipMemcpyDeviceToHost, st), "dl out");
  HIP_TRY(hipStreamSynchronize(st), "sync hstat");
  hipError_t kerr = hipGetLastError();
  (void)vm_dev_free(d_bv); (void)vm_dev_free(d_les); (void)vm_dev_free(d_off);
  (void)vm_dev_free(d_out);
  if (kerr != hipSuccess) return hip_err(errbuf, errbuf_len, "hstat kernel", kerr);
  return 0;
}

int vmgpu_histogram_share(const double* le_req, const double* bucket_values,
                          const double* les, const uint64_t* group_offsets,
                          uint32_t n_groups, int32_t n_grid, double* out,
                          double* out_lower, double* out_upper,
                          char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  if (!le_req || !bucket_values || !les || !group_offsets || !out ||
      n_groups == 0 || n_grid <= 0)
    return set_err(errbuf, errbuf_len, "vmgpu: bad args");
  hipStream_t st = g_ctx.stream;
  uint64_t n_rows = group_offsets[n_groups];
  double* d_req = nullptr;
  double* d_bv = nullptr;
  double* d_les = nullptr;
  uint64_t* d_off = nullptr;
  double* d_out = nullptr;
  double* d_lo = nullptr;
  double* d_hi = nullptr;
  size_t out_elems = (size_t)n_groups * n_grid;
  HIP_TRY(vm_dev_malloc(&d_req, (size_t)n_grid * 8), "alloc req");
  HIP_TRY(vm_dev_malloc(&d_bv, (size_t)n_rows * n_grid * 8), "alloc bv");
  HIP_TRY(vm_dev_malloc(&d_les, (size_t)n_rows * 8), "alloc les");
  HIP_TRY(vm_dev_malloc(&d_off, (size_t)(n_groups + 1) * 8), "alloc off");
  HIP_TRY(vm_dev_malloc(&d_out, out_elems * 8), "alloc out");
  if (out_lower) HIP_TRY(vm_dev_malloc(&d_lo, out_elems * 8), "alloc lo");
  if (out_upper) HIP_TRY(vm_dev_malloc(&d_hi, out_elems * 8), "alloc hi");
  HIP_TRY(hipMemcpyAsync(d_req, le_req, (size_t)n_grid * 8,
                         hipMemcpyHostToDevice, st), "ul req");
  HIP_TRY(hipMemcpyAsync(d_bv, bucket_values, (size_t)n_rows * n_grid * 8,
                         hipMemcpyHostToDevice, st), "ul bv");
  HIP_TRY(hipMemcpyAsync(d_les, les, (size_t)n_rows * 8, hipMemcpyHostToDevice, st), "ul les");
  HIP_TRY(hipMemcpyAsync(d_off, group_offsets, (size_t)(n_groups + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul off");
  uint32_t blocks = (uint32_t)std::min<size_t>((out_elems + 255) / 256, 4096);
  hipLaunchKernelGGL(hshare_kernel, dim3(blocks), dim3(256), 0, st, d_req,
                     d_bv, d_les, d_off, (int64_t)n_groups, n_grid,
                     d_out, d_lo, d_hi);
  HIP_TRY(hipMemcpyAsync(out, d_out, out_elems * 8, hipMemcpyDeviceToHost, st), "dl out");
  if (out_lower) HIP_TRY(hipMemcpyAsync(out_lower, d_lo, out_elems * 8, hipMemcpyDeviceToHost, st), "dl lo");
  if (out_upper) HIP_TRY(hipMemcpyAsync(out_upper, d_hi, out_elems * 8, hipMemcpyDeviceToHost, st), "dl hi");
  HIP_TRY(hipStreamSynchronize(st), "sync hshare");
  hipError_t kerr = hipGetLastError();
  (void)vm_dev_free(d_req); (void)vm_dev_free(d_bv); (void)vm_dev_free(d_les);
  (void)vm_dev_free(d_off); (void)vm_dev_free(d_out); (void)vm_dev_free(d_lo);
  (void)vm_dev_free(d_hi);
  if (kerr != hipSuccess) return hip_err(errbuf, errbuf_len, "hshare kernel", kerr);
  return 0;
}

int vmgpu_histogram_quantile(double phi, const double* bucket_values,
                             const double* les, const uint64_t* group_offsets,
                             uint32_t n_groups, int32_t n_grid,
                             double* out, double* out_lower, double* out_upper,
                             char* errbuf, size_t errbuf_len) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return set_err(errbuf, errbuf_len, "vmgpu: not initialized");
  if (!bucket_values || !les || !group_offsets || !out || n_groups == 0 || n_grid <= 0)
    return set_err(errbuf, errbuf_len, "vmgpu: bad args");
  hipStream_t st = g_ctx.stream;
  uint64_t n_rows = group_offsets[n_groups];
  double* d_bv = nullptr;
  double* d_les = nullptr;
  uint64_t* d_off = nullptr;
  double* d_out = nullptr;
  double* d_lo = nullptr;
  double* d_hi = nullptr;
  size_t out_elems = (size_t)n_groups * n_grid;
  HIP_TRY(vm_dev_malloc(&d_bv, (size_t)n_rows * n_grid * 8), "alloc bv");
  HIP_TRY(vm_dev_malloc(&d_les, (size_t)n_rows * 8), "alloc les");
  HIP_TRY(vm_dev_malloc(&d_off, (size_t)(n_groups + 1) * 8), "alloc off");
  HIP_TRY(vm_dev_malloc(&d_out, out_elems * 8), "alloc out");
  if (out_lower) HIP_TRY(vm_dev_malloc(&d_lo, out_elems * 8), "alloc lo");
  if (out_upper) HIP_TRY(vm_dev_malloc(&d_hi, out_elems * 8), "alloc hi");
  HIP_TRY(hipMemcpyAsync(d_bv, bucket_values, (size_t)n_rows * n_grid * 8,
                         hipMemcpyHostToDevice, st), "ul bv");
  HIP_TRY(hipMemcpyAsync(d_les, les, (size_t)n_rows * 8, hipMemcpyHostToDevice, st), "ul les");
  HIP_TRY(hipMemcpyAsync(d_off, group_offsets, (size_t)(n_groups + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul off");
  uint32_t blocks = (uint32_t)std::min<size_t>((out_elems + 255) / 256, 4096);
  hipLaunchKernelGGL(hq_kernel, dim3(blocks), dim3(256), 0, st, phi, d_bv, d_les,
                     d_off, (int64_t)n_groups, n_grid, d_out, d_lo, d_hi);
  HIP_TRY(hipMemcpyAsync(out, d_out, out_elems * 8, hipMemcpyDeviceToHost, st), "dl out");
  if (out_lower) HIP_TRY(hipMemcpyAsync(out_lower, d_lo, out_elems * 8, hipMemcpyDeviceToHost, st), "dl lo");
  if (out_upper) HIP_TRY(hipMemcpyAsync(out_upper, d_hi, out_elems * 8, hipMemcpyDeviceToHost, st), "dl hi");
  HIP_TRY(hipStreamSynchronize(st), "sync hq");
  hipError_t kerr = hipGetLastError();
  (void)vm_dev_free(d_bv); (void)vm_dev_free(d_les); (void)vm_dev_free(d_off);
  (void)vm_dev_free(d_out); (void)vm_dev_free(d_lo); (void)vm_dev_free(d_hi);
  if (kerr != hipSuccess) return hip_err(errbuf, errbuf_len, "hq kernel", kerr);
  return 0;
}

int vmgpu_last_kernel_ms(double* out_ms) {
  std::lock_guard<std::mutex> lock(g_ctx.mu);
  if (!g_ctx.inited) return 1;
  *out_ms = g_ctx.last_kernel_ms;
  return 0;
}

int vmgpu_device_info(char* name, size_t name_len, double* hbm_gib, int* cu_count) {
  hipDeviceProp_t prop;
  int dev = 0;
  (void)hipGetDevice(&dev);
  if (hipGetDeviceProperties(&prop, dev) != hipSuccess) return 1;
  if (name && name_len) snprintf(name, name_len, "%s", prop.name);
  if (hbm_gib) *hbm_gib = (double)prop.totalGlobalMem / (1024.0 * 1024.0 * 1024.0);
  if (cu_count) *cu_count = prop.multiProcessorCount;
  return 0;
}

}  /* extern "C" */
