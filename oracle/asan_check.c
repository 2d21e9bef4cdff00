/* asan_check — sanitizer harness over the CPU oracle (SURVEY.md §5's
 * sanitizer-build promise).  Built with -fsanitize=address,undefined by the
 * Makefile's `asan` target and executed by tests/test_sanitizer.py; any
 * out-of-bounds access, leak, or UB in the C restatement aborts the run.
 *
 * Exercises, with deliberately odd sizes (0, 1, primes, chunk boundaries):
 *   - codec roundtrips: varint, nearest-delta(2), marshal_int64_array,
 *     the packed block stream (vm_pack_blocks)
 *   - the scan/preFunc passes: removeCounterResets, delta/derivValues,
 *     dropStaleNaNs
 *   - rollup evaluation over the grid for a set of representative funcs
 *     (incl. grouped aggregation) with ragged CSR batches
 *   - topk range/pointwise selection and histogram_quantile
 */
#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "vm_oracle.h"

int64_t vm_pack_blocks(const int64_t* ts, const int64_t* vals,
                       const uint64_t* offsets, uint32_t n_series,
                       uint8_t precision_bits, uint8_t* dst, int64_t cap,
                       uint32_t* series_block_start);
int64_t vm_marshal_int64_array(uint8_t* dst, const int64_t* a, int64_t n,
                               uint8_t precision_bits, uint8_t* out_mt,
                               int64_t* out_first);
int vm_unmarshal_int64_array(int64_t* dst, int64_t items, const uint8_t* src,
                             size_t src_len, uint8_t mt, int64_t first_value);

static uint64_t rng_state = 8428;
static uint64_t rnd(void) {
  rng_state ^= rng_state << 13;
  rng_state ^= rng_state >> 7;
  rng_state ^= rng_state << 17;
  return rng_state;
}

static void check(int cond, const char* what) {
  if (!cond) {
    fprintf(stderr, "asan_check FAILED: %s\n", what);
    exit(1);
  }
}

static void codec_roundtrips(void) {
  static const int64_t sizes[] = {1, 2, 3, 7, 63, 64, 65, 127, 997, 8192};
  for (size_t si = 0; si < sizeof(sizes) / sizeof(sizes[0]); si++) {
    int64_t n = sizes[si];
    int64_t* a = malloc(n * sizeof(int64_t));
    int64_t v = 1000;
    for (int64_t i = 0; i < n; i++) {
      v += (int64_t)(rnd() % 2000) - 980;
      a[i] = v;
    }
    uint8_t* buf = malloc((size_t)n * 10 + 64);
    uint8_t mt = 0;
    int64_t first = 0;
    int64_t len = vm_marshal_int64_array(buf, a, n, 64, &mt, &first);
    check(len >= 0, "marshal");
    int64_t* back = malloc(n * sizeof(int64_t));
    check(vm_unmarshal_int64_array(back, n, buf, (size_t)len, mt, first) == 0,
          "unmarshal");
    check(memcmp(a, back, (size_t)n * 8) == 0, "codec roundtrip");
    free(back);
    free(buf);
    free(a);
  }
}

static void packed_stream(void) {
  enum { NS = 17 };
  uint64_t offsets[NS + 1];
  offsets[0] = 0;
  for (int s = 0; s < NS; s++) offsets[s + 1] = offsets[s] + 1 + rnd() % 300;
  uint64_t total = offsets[NS];
  int64_t* ts = malloc(total * 8);
  int64_t* vals = malloc(total * 8);
  for (int s = 0; s < NS; s++) {
    int64_t t = 1600000000000LL;
    int64_t c = 0;
    for (uint64_t k = offsets[s]; k < offsets[s + 1]; k++) {
      t += 14500 + (int64_t)(rnd() % 1000);
      c += (int64_t)(rnd() % 500);
      ts[k] = t;
      vals[k] = c;
    }
  }
  int64_t cap = (int64_t)total * 22 + NS * 192 + 64;
  uint8_t* dst = malloc((size_t)cap);
  uint32_t sbs[NS + 1];
  int64_t w = vm_pack_blocks(ts, vals, offsets, NS, 64, dst, cap, sbs);
  check(w > 0, "pack_blocks");
  free(dst);
  free(vals);
  free(ts);
}

static void scans_and_rollups(void) {
  static const int funcs[] = {VM_FN_RATE,           VM_FN_INCREASE,
                              VM_FN_AVG,  VM_FN_QUANTILE,
                              VM_FN_DEFAULT_ROLLUP, VM_FN_CHANGES,
                              VM_FN_MAD,  VM_FN_HOLT_WINTERS,
                              VM_FN_MODE, VM_FN_OUTLIER_IQR};
  enum { NS = 9 };
  uint64_t offsets[NS + 1];
  offsets[0] = 0;
  for (int s = 0; s < NS; s++)
    offsets[s + 1] = offsets[s] + (s == 4 ? 0 : rnd() % 260);
  uint64_t total = offsets[NS];
  int64_t* ts = malloc((total ? total : 1) * 8);
  double* vals = malloc((total ? total : 1) * 8);
  for (int s = 0; s < NS; s++) {
    int64_t t = 1600000000000LL;
    double c = 0;
    for (uint64_t k = offsets[s]; k < offsets[s + 1]; k++) {
      t += 14000 + (int64_t)(rnd() % 2000);
      c = (rnd() % 100 == 0) ? 0.0 : c + (double)(rnd() % 300);
      ts[k] = t;
      vals[k] = (rnd() % 50 == 0) ? vm_stale_nan() : c;
    }
  }
  /* standalone scan passes over the first non-empty series */
  for (int s = 0; s < NS; s++) {
    int64_t n = (int64_t)(offsets[s + 1] - offsets[s]);
    if (!n) continue;
    double* vcopy = malloc(n * 8);
    int64_t* tcopy = malloc(n * 8);
    memcpy(vcopy, vals + offsets[s], (size_t)n * 8);
    memcpy(tcopy, ts + offsets[s], (size_t)n * 8);
    int64_t m = vm_drop_stale_nans(vcopy, tcopy, n);
    vm_remove_counter_resets(vcopy, tcopy, m, 300000);
    vm_delta_values(vcopy, m);
    vm_deriv_values(vcopy, tcopy, m);
    free(tcopy);
    free(vcopy);
  }
  int32_t gids[NS];
  for (int s = 0; s < NS; s++) gids[s] = s % 3;
  vm_rollup_config rc;
  memset(&rc, 0, sizeof(rc));
  rc.start = 1600000000000LL + 600000;
  rc.end = rc.start + 49 * 15000;
  rc.step = 15000;
  rc.window = 300000;
  rc.arg = 0.9;
  rc.arg2 = 0.1;
  int64_t n_grid = vm_grid_points(rc.start, rc.end, rc.step);
  for (size_t fi = 0; fi < sizeof(funcs) / sizeof(funcs[0]); fi++) {
    rc.func = funcs[fi];
    double* out = malloc((size_t)NS * (size_t)n_grid * 8);
    uint64_t scanned = 0;
    check(vm_rollup_eval_batch(&rc, rc.func == VM_FN_RATE, 300000, 1, 0, ts,
                               vals, offsets, NS, NULL, 0, VM_AGGR_NONE, out,
                               NULL, &scanned, 2) == 0,
          "rollup_eval_batch");
    free(out);
    /* grouped */
    double* gout = malloc(3 * (size_t)n_grid * 8);
    double* gcnt = malloc(3 * (size_t)n_grid * 8);
    check(vm_rollup_eval_batch(&rc, 0, 0, 1, 0, ts, vals, offsets, NS, gids,
                               3, VM_AGGR_AVG, gout, gcnt, &scanned, 2) == 0,
          "grouped eval");
    free(gcnt);
    free(gout);
  }
  /* topk + histogram_quantile over a dense matrix */
  int64_t rows = 23, cols = 31;
  double* m = malloc((size_t)rows * cols * 8);
  for (int64_t i = 0; i < rows * cols; i++)
    m[i] = (rnd() % 7 == 0) ? NAN : (double)(rnd() % 1000) - 500;
  int64_t sel[23];
  double rem[31];
  for (int op = 0; op <= 4; op++) {
    int64_t ns = vm_topk_range(m, rows, cols, 5, op, op & 1, sel, rem);
    check(ns >= 0 && ns <= 5, "topk_range");
  }
  vm_topk_pointwise(m, rows, cols, 4, 0);
  uint64_t goff[4] = {0, 7, 15, 23};
  double les[23];
  for (int i = 0; i < 23; i++) les[i] = (double)(i + 1);
  double* hq = malloc(3 * (size_t)cols * 8);
  for (int64_t i = 0; i < rows * cols; i++) m[i] = fabs(m[i]);
  vm_histogram_quantile(0.95, m, les, goff, 3, cols, hq, NULL, NULL);
  free(hq);
  free(m);
  free(vals);
  free(ts);
}

/* Decode mutated/truncated/garbage byte streams: whatever the return
 * code, the decoder must never read or write out of bounds (ASan checks
 * that), and a success must still produce exactly `items` outputs. */
static void malformed_decode_fuzz(void) {
  enum { N = 257 };
  int64_t a[N];
  int64_t v = -37;
  for (int64_t i = 0; i < N; i++) {
    v += (int64_t)(rnd() % 5000) - 2400;
    a[i] = v;
  }
  uint8_t base[N * 10 + 64];
  uint8_t mt = 0;
  int64_t first = 0;
  int64_t len = vm_marshal_int64_array(base, a, N, 64, &mt, &first);
  check(len > 0, "fuzz marshal");
  int64_t dst[N];
  for (int iter = 0; iter < 4000; iter++) {
    /* heap copy sized EXACTLY to the (possibly truncated) length so ASan
     * sees one-past-the-end reads */
    size_t cut = (size_t)(rnd() % (uint64_t)(len + 1));
    uint8_t* buf = malloc(cut ? cut : 1);
    memcpy(buf, base, cut);
    int nmut = (int)(rnd() % 4);
    for (int m = 0; m < nmut && cut; m++)
      buf[rnd() % cut] = (uint8_t)rnd();
    uint8_t use_mt = (iter % 3 == 0) ? (uint8_t)rnd() : mt;
    int64_t items = (iter % 5 == 0) ? (int64_t)(rnd() % (2 * N)) : N;
    if (items > N) items = N;  /* dst capacity */
    (void)vm_unmarshal_int64_array(dst, items, buf, cut, use_mt, first);
    free(buf);
  }
  /* pure garbage bytes under every marshal type */
  for (int iter = 0; iter < 1000; iter++) {
    size_t glen = 1 + (size_t)(rnd() % 300);
    uint8_t* buf = malloc(glen);
    for (size_t i = 0; i < glen; i++) buf[i] = (uint8_t)rnd();
    (void)vm_unmarshal_int64_array(dst, (int64_t)(rnd() % N), buf, glen,
                                   (uint8_t)(rnd() % 8), (int64_t)rnd());
    free(buf);
  }
}

int main(void) {
  codec_roundtrips();
  packed_stream();
  scans_and_rollups();
  malformed_decode_fuzz();
  printf("asan_check OK\n");
  return 0;
}
