/* encoding.c — CPU oracle for lib/encoding (block codec) + the fetch/decode
 * merge (netstorage mergeSortBlocks) + dedup.
 *
 * TEST INFRASTRUCTURE ONLY (see vm_oracle.h header note).
 * Faithful C restatement of:
 *   lib/encoding/int.go: MarshalVarInt64s 107-168, UnmarshalVarInt64s 184-290
 *   lib/encoding/nearest_delta.go: 15-144
 *   lib/encoding/nearest_delta2.go: 16-90
 *   lib/encoding/encoding.go: marshalInt64Array 119-173,
 *     unmarshalInt64Array 175-250, EnsureNonDecreasingSequence 255-286,
 *     isConst 288-306, isDeltaConst 309-323, isGauge 325-366,
 *     getCompressLevel 368-382
 *   lib/storage/dedup.go: DeduplicateSamples 29-92
 *   app/vmselect/netstorage/netstorage.go: mergeSortBlocks 564-614,
 *     equalSamplesPrefix 618-642, binarySearchTimestamps 644-660,
 *     sortBlocksHeap (Go container/heap semantics) 686-734
 * zstd frames use the system libzstd (dlopen "libzstd.so.1") — the format is
 * standard; compressed BYTES may differ from the reference's vendored zstd
 * build, round-trip and cross-decode are what parity requires.
 */
#include "vm_decimal.h"
#include <dlfcn.h>
#include <math.h>
#include <stdlib.h>
#include <string.h>

/* ---------------- varint (zigzag + LEB128) ---------------- */

static size_t put_uvarint(uint8_t* dst, uint64_t u) {
  size_t i = 0;
  while (u >= 0x80) {
    dst[i++] = (uint8_t)(u | 0x80);
    u >>= 7;
  }
  dst[i++] = (uint8_t)u;
  return i;
}

size_t vm_marshal_varint64s(uint8_t* dst, const int64_t* vs, int64_t n) {
  size_t off = 0;
  for (int64_t i = 0; i < n; i++) {
    int64_t v = vs[i];
    uint64_t u = ((uint64_t)v << 1) ^ (uint64_t)(v >> 63);
    off += put_uvarint(dst + off, u);
  }
  return off;
}

int64_t vm_unmarshal_varint64s(int64_t* dst, int64_t n, const uint8_t* src, size_t src_len) {
  size_t idx = 0;
  for (int64_t i = 0; i < n; i++) {
    uint64_t u = 0;
    int shift = 0;
    int nbytes = 0;
    for (;;) {
      if (idx >= src_len) return -1;
      uint8_t c = src[idx++];
      nbytes++;
      if (nbytes > 10) return -2;
      if (nbytes == 10 && c > 1) return -3;
      u |= (uint64_t)(c & 0x7f) << shift;
      if (c < 0x80) break;
      shift += 7;
    }
    dst[i] = (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
  }
  return (int64_t)idx;
}

/* ---------------- nearest delta (nearest_delta.go) ---------------- */

static uint8_t bits_len64(uint64_t v) {
  return (uint8_t)(v == 0 ? 0 : 64 - __builtin_clzll(v));
}

static uint8_t get_trailing_zeros(int64_t v, uint8_t precision_bits) {
  if (v < 0) v = -v;
  uint8_t vbits = bits_len64((uint64_t)v);
  if (vbits <= precision_bits) return 0;
  return (uint8_t)(vbits - precision_bits);
}

static uint8_t dec_if_nonzero(uint8_t n) { return n == 0 ? 0 : (uint8_t)(n - 1); }

static int64_t nearest_delta(int64_t next, int64_t prev, uint8_t precision_bits,
                             uint8_t prev_tz, uint8_t* out_tz) {
  int64_t d = next - prev;
  if (d == 0) {
    *out_tz = dec_if_nonzero(prev_tz);
    return 0;
  }
  int64_t origin = next;
  if (origin < 0) origin = -origin;
  uint8_t origin_bits = bits_len64((uint64_t)origin);
  if (origin_bits <= precision_bits) {
    *out_tz = dec_if_nonzero(prev_tz);
    return d;
  }
  uint8_t tz = (uint8_t)(origin_bits - precision_bits);
  if (tz > (uint8_t)(prev_tz + 4)) {
    *out_tz = (uint8_t)(prev_tz + 2);
    return d;
  }
  if ((uint8_t)(tz + 4) < prev_tz) {
    *out_tz = (uint8_t)(prev_tz - 2);
    return d;
  }
  int minus = 0;
  if (d < 0) {
    minus = 1;
    d = -d;
  }
  int64_t nd = (int64_t)((uint64_t)d & (~0ULL << tz));
  if (minus) nd = -nd;
  *out_tz = tz;
  return nd;
}

size_t vm_marshal_nearest_delta(uint8_t* dst, const int64_t* src, int64_t n,
                                uint8_t precision_bits, int64_t* out_first) {
  *out_first = src[0];
  int64_t v = src[0];
  size_t off = 0;
  if (precision_bits == 64) {
    for (int64_t i = 1; i < n; i++) {
      int64_t d = src[i] - v;
      v += d;
      int64_t tmp = d;
      off += vm_marshal_varint64s(dst + off, &tmp, 1);
    }
  } else {
    uint8_t tz = get_trailing_zeros(v, precision_bits);
    for (int64_t i = 1; i < n; i++) {
      int64_t d = nearest_delta(src[i], v, precision_bits, tz, &tz);
      v += d;
      int64_t tmp = d;
      off += vm_marshal_varint64s(dst + off, &tmp, 1);
    }
  }
  return off;
}

int vm_unmarshal_nearest_delta(int64_t* dst, const uint8_t* src, size_t src_len,
                               int64_t first_value, int64_t items) {
  if (items < 1) return 1;
  int64_t* deltas = (int64_t*)malloc((size_t)(items > 1 ? items - 1 : 1) * 8);
  int64_t used = vm_unmarshal_varint64s(deltas, items - 1, src, src_len);
  if (used < 0 || (size_t)used != src_len) {
    free(deltas);
    return 2;
  }
  int64_t v = first_value;
  dst[0] = v;
  for (int64_t i = 1; i < items; i++) {
    v += deltas[i - 1];
    dst[i] = v;
  }
  free(deltas);
  return 0;
}

/* ---------------- nearest delta2 (nearest_delta2.go) ---------------- */

size_t vm_marshal_nearest_delta2(uint8_t* dst, const int64_t* src, int64_t n,
                                 uint8_t precision_bits, int64_t* out_first) {
  *out_first = src[0];
  int64_t d1 = src[1] - src[0];
  size_t off = vm_marshal_varint64s(dst, &d1, 1);
  int64_t v = src[1];
  if (precision_bits == 64) {
    for (int64_t i = 2; i < n; i++) {
      int64_t d2 = src[i] - v - d1;
      d1 += d2;
      v += d1;
      off += vm_marshal_varint64s(dst + off, &d2, 1);
    }
  } else {
    uint8_t tz = get_trailing_zeros(v, precision_bits);
    for (int64_t i = 2; i < n; i++) {
      int64_t d2 = nearest_delta(src[i] - v, d1, precision_bits, tz, &tz);
      d1 += d2;
      v += d1;
      off += vm_marshal_varint64s(dst + off, &d2, 1);
    }
  }
  return off;
}

int vm_unmarshal_nearest_delta2(int64_t* dst, const uint8_t* src, size_t src_len,
                                int64_t first_value, int64_t items) {
  if (items < 2) return 1;
  int64_t* is = (int64_t*)malloc((size_t)(items - 1) * 8);
  int64_t used = vm_unmarshal_varint64s(is, items - 1, src, src_len);
  if (used < 0 || (size_t)used != src_len) {
    free(is);
    return 2;
  }
  int64_t v = first_value;
  int64_t d1 = is[0];
  dst[0] = v;
  v += d1;
  dst[1] = v;
  for (int64_t i = 2; i < items; i++) {
    d1 += is[i - 1];
    v += d1;
    dst[i] = v;
  }
  free(is);
  return 0;
}

/* ---------------- detectors (encoding.go) ---------------- */

int vm_is_const(const int64_t* a, int64_t n) {
  if (n == 0) return 0;
  for (int64_t i = 1; i < n; i++)
    if (a[i] != a[0]) return 0;
  return 1;
}

int vm_is_delta_const(const int64_t* a, int64_t n) {
  if (n < 2) return 0;
  int64_t d1 = a[1] - a[0];
  for (int64_t i = 2; i < n; i++)
    if (a[i] - a[i - 1] != d1) return 0;
  return 1;
}

int vm_is_gauge(const int64_t* a, int64_t n) {
  if (n < 2) return 0;
  int64_t resets = 0;
  int64_t v_prev = a[0];
  if (v_prev < 0) return 1;
  for (int64_t i = 1; i < n; i++) {
    int64_t v = a[i];
    if (v < v_prev) {
      if (v < 0) return 1;
      if (v > (v_prev >> 3)) return 1;
      resets++;
    }
    v_prev = v;
  }
  if (resets <= 2) return 0;
  return resets > (n >> 3);
}

static int get_compress_level(int64_t items) {
  if (items <= 1 << 6) return 1;
  if (items <= 1 << 8) return 2;
  if (items <= 1 << 10) return 3;
  if (items <= 1 << 12) return 4;
  return 5;
}

/* ---------------- zstd via dlopen ---------------- */

typedef size_t (*zstd_compress_fn)(void*, size_t, const void*, size_t, int);
typedef size_t (*zstd_decompress_fn)(void*, size_t, const void*, size_t);
typedef size_t (*zstd_bound_fn)(size_t);
typedef unsigned (*zstd_iserr_fn)(size_t);
typedef unsigned long long (*zstd_framesize_fn)(const void*, size_t);

static zstd_compress_fn z_compress;
static zstd_decompress_fn z_decompress;
static zstd_bound_fn z_bound;
static zstd_iserr_fn z_iserr;
static zstd_framesize_fn z_framesize;
static int z_loaded = 0;

static int load_zstd(void) {
  if (z_loaded) return z_compress != NULL;
  z_loaded = 1;
  void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
  if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
  if (!h) return 0;
  z_compress = (zstd_compress_fn)dlsym(h, "ZSTD_compress");
  z_decompress = (zstd_decompress_fn)dlsym(h, "ZSTD_decompress");
  z_bound = (zstd_bound_fn)dlsym(h, "ZSTD_compressBound");
  z_iserr = (zstd_iserr_fn)dlsym(h, "ZSTD_isError");
  z_framesize = (zstd_framesize_fn)dlsym(h, "ZSTD_getFrameContentSize");
  return z_compress && z_decompress && z_bound && z_iserr && z_framesize;
}

/* helper for tests: decompress one zstd frame with the dlopen'd libzstd
 * (mirrors what the host engine does before vmgpu_decode_blocks) */
int64_t vm_zstd_decompress(uint8_t* dst, size_t cap, const uint8_t* src, size_t n) {
  if (!load_zstd()) return -1;
  unsigned long long fs = z_framesize(src, n);
  if (fs == (unsigned long long)-1 || fs == (unsigned long long)-2) return -2;
  if (fs > cap) return -3;
  size_t dlen = z_decompress(dst, cap, src, n);
  if (z_iserr(dlen)) return -4;
  return (int64_t)dlen;
}

/* ---------------- marshalInt64Array (encoding.go:119-173) ---------------- */

#define MIN_COMPRESSIBLE_BLOCK_SIZE 128

int64_t vm_marshal_int64_array(uint8_t* dst, const int64_t* a, int64_t n,
                               uint8_t precision_bits, uint8_t* out_mt,
                               int64_t* out_first) {
  if (n == 0) return -2;
  if (vm_is_const(a, n)) {
    *out_first = a[0];
    *out_mt = VM_MT_CONST;
    return 0;
  }
  if (vm_is_delta_const(a, n)) {
    *out_first = a[0];
    *out_mt = VM_MT_DELTA_CONST;
    int64_t d = a[1] - a[0];
    return (int64_t)vm_marshal_varint64s(dst, &d, 1);
  }
  size_t cap = (size_t)n * 10 + 64;
  uint8_t* plain = (uint8_t*)malloc(cap);
  size_t plain_len;
  uint8_t mt;
  if (vm_is_gauge(a, n)) {
    mt = VM_MT_ZSTD_NEAREST_DELTA;
    uint8_t pb = precision_bits;
    if (pb < 6) pb = (uint8_t)(pb + 2); /* gauges get 2 extra bits */
    plain_len = vm_marshal_nearest_delta(plain, a, n, pb, out_first);
  } else {
    mt = VM_MT_ZSTD_NEAREST_DELTA2;
    plain_len = vm_marshal_nearest_delta2(plain, a, n, precision_bits, out_first);
  }
  int64_t out_len = -1;
  if (plain_len >= MIN_COMPRESSIBLE_BLOCK_SIZE && load_zstd()) {
    size_t bound = z_bound(plain_len);
    uint8_t* comp = (uint8_t*)malloc(bound);
    size_t clen = z_compress(comp, bound, plain, plain_len,
                             get_compress_level(n));
    if (!z_iserr(clen) && (double)clen <= 0.9 * (double)plain_len) {
      memcpy(dst, comp, clen);
      out_len = (int64_t)clen;
    }
    free(comp);
  }
  if (out_len < 0) {
    /* ineffective compression (or block too small): plain data */
    mt = (mt == VM_MT_ZSTD_NEAREST_DELTA2) ? VM_MT_NEAREST_DELTA2
                                           : VM_MT_NEAREST_DELTA;
    memcpy(dst, plain, plain_len);
    out_len = (int64_t)plain_len;
  }
  free(plain);
  *out_mt = mt;
  return out_len;
}

int vm_unmarshal_int64_array(int64_t* dst, int64_t items, const uint8_t* src,
                             size_t src_len, uint8_t mt, int64_t first_value) {
  switch (mt) {
    case VM_MT_ZSTD_NEAREST_DELTA:
    case VM_MT_ZSTD_NEAREST_DELTA2: {
      if (!load_zstd()) return -1;
      unsigned long long fs = z_framesize(src, src_len);
      if (fs == (unsigned long long)-1 || fs == (unsigned long long)-2)
        return 3;
      uint8_t* plain = (uint8_t*)malloc(fs ? fs : 1);
      size_t dlen = z_decompress(plain, fs, src, src_len);
      if (z_iserr(dlen)) {
        free(plain);
        return 4;
      }
      int rc = (mt == VM_MT_ZSTD_NEAREST_DELTA)
                   ? vm_unmarshal_nearest_delta(dst, plain, dlen, first_value, items)
                   : vm_unmarshal_nearest_delta2(dst, plain, dlen, first_value, items);
      free(plain);
      return rc;
    }
    case VM_MT_NEAREST_DELTA:
      return vm_unmarshal_nearest_delta(dst, src, src_len, first_value, items);
    case VM_MT_NEAREST_DELTA2:
      return vm_unmarshal_nearest_delta2(dst, src, src_len, first_value, items);
    case VM_MT_CONST: {
      if (src_len > 0) return 5;
      for (int64_t i = 0; i < items; i++) dst[i] = first_value;
      return 0;
    }
    case VM_MT_DELTA_CONST: {
      int64_t d;
      int64_t used = vm_unmarshal_varint64s(&d, 1, src, src_len);
      if (used <= 0 || (size_t)used != src_len) return 6;
      int64_t v = first_value;
      for (int64_t i = 0; i < items; i++) {
        dst[i] = v;
        v += d;
      }
      return 0;
    }
    default:
      return 7;
  }
}

/* ---------------- EnsureNonDecreasingSequence (encoding.go:255-286) ------ */

void vm_ensure_non_decreasing(int64_t* a, int64_t n, int64_t v_min, int64_t v_max) {
  if (n == 0) return;
  if (a[0] != v_min) a[0] = v_min;
  int64_t v_prev = a[0];
  for (int64_t i = 1; i < n; i++) {
    if (a[i] < v_prev) a[i] = v_prev;
    v_prev = a[i];
  }
  int64_t i = n - 1;
  if (a[i] != v_max) {
    a[i] = v_max;
    i--;
    while (i >= 0 && a[i] > v_max) {
      a[i] = v_max;
      i--;
    }
  }
}

/* ---------------- DeduplicateSamples (dedup.go:29-92) ---------------- */

static int is_stale_nan_f(double v) {
  uint64_t b;
  memcpy(&b, &v, 8);
  return b == 0x7ff0000000000002ULL;
}

static int needs_dedup(const int64_t* ts, int64_t n, int64_t dedup_interval) {
  /* needsDedup (dedup.go:149-174): pairs closer than the interval, aligned
   * buckets */
  if (n < 2 || dedup_interval <= 0) return 0;
  int64_t ts_next = ts[0] + dedup_interval - 1;
  ts_next -= ts_next % dedup_interval;
  for (int64_t i = 1; i < n; i++) {
    if (ts[i] <= ts_next) return 1;
    ts_next += dedup_interval;
    if (ts_next < ts[i]) {
      ts_next = ts[i] + dedup_interval - 1;
      ts_next -= ts_next % dedup_interval;
    }
  }
  return 0;
}

int64_t vm_deduplicate_samples(int64_t* ts, double* vals, int64_t n, int64_t dedup_interval) {
  if (!needs_dedup(ts, n, dedup_interval)) return n;
  int64_t ts_next = ts[0] + dedup_interval - 1;
  ts_next -= ts_next % dedup_interval;
  int64_t k = 0;
  for (int64_t i = 1; i < n; i++) {
    int64_t t = ts[i];
    if (t <= ts_next) continue;
    /* choose the max non-stale value among samples sharing ts[i-1] */
    int64_t j = i - 1;
    int64_t ts_prev = ts[j];
    double v_prev = vals[j];
    while (j > 0 && ts[j - 1] == ts_prev) {
      j--;
      if (is_stale_nan_f(vals[j])) continue;
      if (is_stale_nan_f(v_prev)) {
        v_prev = vals[j];
        continue;
      }
      if (vals[j] > v_prev) v_prev = vals[j];
    }
    ts[k] = ts_prev;
    vals[k] = v_prev;
    k++;
    ts_next += dedup_interval;
    if (ts_next < t) {
      ts_next = t + dedup_interval - 1;
      ts_next -= ts_next % dedup_interval;
    }
  }
  int64_t j = n - 1;
  int64_t ts_prev = ts[j];
  double v_prev = vals[j];
  while (j > 0 && ts[j - 1] == ts_prev) {
    j--;
    if (is_stale_nan_f(vals[j])) continue;
    if (is_stale_nan_f(v_prev)) {
      v_prev = vals[j];
      continue;
    }
    if (vals[j] > v_prev) v_prev = vals[j];
  }
  ts[k] = ts_prev;
  vals[k] = v_prev;
  k++;
  return k;
}

/* ---------------- mergeSortBlocks (netstorage.go:564-614) ---------------- */

typedef struct {
  const int64_t* ts;
  const double* vals;
  int64_t len;
  int64_t next;
} SB;

static int sb_less(SB** sbs, int64_t i, int64_t j) {
  return sbs[i]->ts[sbs[i]->next] < sbs[j]->ts[sbs[j]->next];
}

/* Go container/heap siftDown/siftUp semantics */
static void sb_down(SB** sbs, int64_t i0, int64_t n) {
  int64_t i = i0;
  for (;;) {
    int64_t j1 = 2 * i + 1;
    if (j1 >= n || j1 < 0) break;
    int64_t j = j1;
    int64_t j2 = j1 + 1;
    if (j2 < n && sb_less(sbs, j2, j1)) j = j2;
    if (!sb_less(sbs, j, i)) break;
    SB* t = sbs[i];
    sbs[i] = sbs[j];
    sbs[j] = t;
    i = j;
  }
}

static void sb_up(SB** sbs, int64_t j) {
  while (j > 0) {
    int64_t i = (j - 1) / 2;
    if (i == j || !sb_less(sbs, j, i)) break;
    SB* t = sbs[i];
    sbs[i] = sbs[j];
    sbs[j] = t;
    j = i;
  }
}

static int64_t equal_samples_prefix(const SB* a, const SB* b) {
  int64_t an = a->len - a->next, bn = b->len - b->next;
  int64_t n = 0;
  while (n < an && n < bn && a->ts[a->next + n] == b->ts[b->next + n]) n++;
  int64_t m = 0;
  while (m < n) {
    uint64_t av, bv;
    memcpy(&av, &a->vals[a->next + m], 8);
    memcpy(&bv, &b->vals[b->next + m], 8);
    if (av != bv) break;
    m++;
  }
  return m;
}

static int64_t bsearch_ts(const int64_t* ts, int64_t n, int64_t seek) {
  if (n > 0 && ts[n - 1] <= seek) return n;
  int64_t i = 0, j = n;
  while (i < j) {
    int64_t h = (i + j) >> 1;
    if (ts[h] <= seek) i = h + 1;
    else j = h;
  }
  return i;
}

int64_t vm_merge_sort_blocks(const int64_t* ts, const double* vals,
                             const uint64_t* offsets, int32_t n_blocks,
                             int64_t dedup_interval,
                             int64_t* dst_ts, double* dst_vals) {
  SB* blocks = (SB*)malloc((size_t)n_blocks * sizeof(SB));
  SB** sbs = (SB**)malloc((size_t)n_blocks * sizeof(SB*));
  int64_t nsb = 0;
  for (int32_t b = 0; b < n_blocks; b++) {
    int64_t lo = (int64_t)offsets[b], hi = (int64_t)offsets[b + 1];
    if (hi == lo) continue; /* skip empty blocks */
    blocks[nsb].ts = ts + lo;
    blocks[nsb].vals = vals + lo;
    blocks[nsb].len = hi - lo;
    blocks[nsb].next = 0;
    sbs[nsb] = &blocks[nsb];
    nsb++;
  }
  int64_t out = 0;
  if (nsb == 0) {
    free(blocks);
    free(sbs);
    return 0;
  }
  /* heap.Init */
  for (int64_t i = nsb / 2 - 1; i >= 0; i--) sb_down(sbs, i, nsb);
  for (;;) {
    SB* top = sbs[0];
    if (nsb == 1) {
      int64_t rem = top->len - top->next;
      memcpy(dst_ts + out, top->ts + top->next, (size_t)rem * 8);
      memcpy(dst_vals + out, top->vals + top->next, (size_t)rem * 8);
      out += rem;
      break;
    }
    /* getNextBlock (netstorage.go:692-706): smaller of the root's children */
    SB* sb_next;
    if (nsb < 3) {
      sb_next = sbs[1];
    } else {
      sb_next = sb_less(sbs, 2, 1) ? sbs[2] : sbs[1];
    }
    int64_t ts_next = sb_next->ts[sb_next->next];
    int64_t top_next_idx = top->next;
    int64_t n_eq = equal_samples_prefix(top, sb_next);
    if (n_eq > 0 && dedup_interval > 0) {
      /* skip replicated samples at top */
      top->next = top_next_idx + n_eq;
    } else {
      int64_t adv = bsearch_ts(top->ts + top_next_idx, top->len - top_next_idx, ts_next);
      top->next = top_next_idx + adv;
      memcpy(dst_ts + out, top->ts + top_next_idx, (size_t)adv * 8);
      memcpy(dst_vals + out, top->vals + top_next_idx, (size_t)adv * 8);
      out += adv;
    }
    if (top->next < top->len) {
      /* heap.Fix(0) */
      sb_down(sbs, 0, nsb);
      sb_up(sbs, 0);
    } else {
      /* heap.Pop */
      SB* t = sbs[0];
      sbs[0] = sbs[nsb - 1];
      sbs[nsb - 1] = t;
      nsb--;
      if (nsb > 0) sb_down(sbs, 0, nsb);
    }
  }
  out = vm_deduplicate_samples(dst_ts, dst_vals, out, dedup_interval);
  free(blocks);
  free(sbs);
  return out;
}

/* ---------------- packed block stream (bench input generator) -------------
 * Builds the wire form the product's vmgpu_batch_create_packed parses: a
 * sequence of records, each a 48-byte header followed by ts_data then
 * val_data (post-zstd marshal types only — the device decode contract).
 * This is TEST INFRASTRUCTURE: the cold-query bench uses it to synthesize
 * the compressed payload a vmselect fetch would hand over (the write/encode
 * path itself is out of scope).  Header layout must match
 * include/vmgpu.h vmgpu_packed_block_hdr. */
typedef struct {
  int64_t min_timestamp;
  int64_t max_timestamp;
  int64_t first_value;
  uint32_t rows;
  int32_t scale;
  uint32_t ts_data_len;
  uint32_t val_data_len;
  uint8_t ts_mt;
  uint8_t val_mt;
  uint8_t precision_bits;
  uint8_t pad[5];
} vm_packed_hdr;

/* marshal without the zstd stage (types 0/2/3/5/6 only) */
static int64_t marshal_nozstd(uint8_t* dst, const int64_t* a, int64_t n,
                              uint8_t precision_bits, uint8_t* out_mt,
                              int64_t* out_first) {
  if (n == 0) return -2;
  if (vm_is_const(a, n)) {
    *out_first = a[0];
    *out_mt = VM_MT_CONST;
    return 0;
  }
  if (vm_is_delta_const(a, n)) {
    *out_first = a[0];
    *out_mt = VM_MT_DELTA_CONST;
    int64_t d = a[1] - a[0];
    return (int64_t)vm_marshal_varint64s(dst, &d, 1);
  }
  if (vm_is_gauge(a, n)) {
    uint8_t pb = precision_bits;
    if (pb < 6) pb = (uint8_t)(pb + 2);
    *out_mt = VM_MT_NEAREST_DELTA;
    return (int64_t)vm_marshal_nearest_delta(dst, a, n, pb, out_first);
  }
  *out_mt = VM_MT_NEAREST_DELTA2;
  return (int64_t)vm_marshal_nearest_delta2(dst, a, n, precision_bits, out_first);
}

/* CSR int64 columns (scale 0) -> packed stream, one block per series.
 * Returns packed bytes written, or -1 if cap would overflow.
 * series_block_start (n_series+1, may be NULL) gets the identity CSR. */
int64_t vm_pack_blocks(const int64_t* ts, const int64_t* vals,
                       const uint64_t* offsets, uint32_t n_series,
                       uint8_t precision_bits, uint8_t* dst, int64_t cap,
                       uint32_t* series_block_start) {
  int64_t w = 0;
  for (uint32_t s = 0; s < n_series; s++) {
    if (series_block_start) series_block_start[s] = s;
    uint64_t lo = offsets[s];
    int64_t n = (int64_t)(offsets[s + 1] - lo);
    if (n == 0 || n > 8192) return -2; /* one block per series only */
    int64_t worst = (int64_t)sizeof(vm_packed_hdr) + 2 * (n * 10 + 64);
    if (w + worst > cap) return -1;
    uint8_t* body = dst + w + sizeof(vm_packed_hdr);
    int64_t tfirst = 0, vfirst = 0;
    uint8_t tmt = 0, vmt = 0;
    int64_t tlen = marshal_nozstd(body, ts + lo, n, 64, &tmt, &tfirst);
    if (tlen < 0) return -3;
    int64_t vlen = marshal_nozstd(body + tlen, vals + lo, n, precision_bits,
                                  &vmt, &vfirst);
    if (vlen < 0) return -3;
    /* the stream position is not 8-aligned in general: build the header
     * locally and memcpy (the parser reads it the same way) */
    vm_packed_hdr h;
    memset(&h, 0, sizeof(h));
    h.min_timestamp = ts[lo];
    h.max_timestamp = ts[lo + n - 1];
    h.first_value = vfirst;
    h.rows = (uint32_t)n;
    h.scale = 0;
    h.ts_data_len = (uint32_t)tlen;
    h.val_data_len = (uint32_t)vlen;
    h.ts_mt = tmt;
    h.val_mt = vmt;
    h.precision_bits = precision_bits;
    memcpy(dst + w, &h, sizeof(h));
    w += (int64_t)sizeof(vm_packed_hdr) + tlen + vlen;
  }
  if (series_block_start) series_block_start[n_series] = n_series;
  return w;
}
