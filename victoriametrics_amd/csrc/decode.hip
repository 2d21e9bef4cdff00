/* decode.hip — GPU block decode for the vmselect fetch path.
 *
 * Independent MI355X-native implementation of the reference's block codec
 * read path (lib/encoding + lib/decimal + lib/storage/block.go:250
 * UnmarshalData), plus a host-side encoder mirroring the write path
 * (marshalInt64Array, encoding.go:119 — needed to produce realistic blocks
 * for benches/tests) and the per-series k-way merge
 * (mergeSortBlocks, netstorage.go:564).  zstd frames are a CPU format and
 * stay on the host (decompressed before upload).
 *
 * Device decode design (one 256-thread workgroup per storage block, blocks
 * looped grid-stride):
 *   phase 1  terminator-mask build: one ballot per 64 payload bytes, mask
 *            words + per-word terminator counts staged in LDS, then an
 *            exact integer prefix over words.
 *   phase 2  wave-parallel varint decode: each thread locates the start of
 *            its varint via the word prefix (binary search + bit-select)
 *            and decodes <=10 bytes from the L2-resident payload; zigzag to
 *            signed deltas in a per-workgroup global scratch column.
 *   phase 3  block-wide exact integer prefix sum reconstructs values
 *            (nearest-delta; run twice for nearest-delta2's second-order
 *            deltas).  int64 wrap-around matches Go via uint64 arithmetic.
 *   phase 4  timestamps: EnsureNonDecreasingSequence (encoding.go:255) for
 *            precisionBits<64 or bounds validation; values: fused
 *            decimal->float (AppendDecimalToFloat, decimal.go:100) with the
 *            10^scale double precomputed on the host so results are
 *            bit-identical to the CPU path.
 */
#include <hip/hip_runtime.h>
#include "devalloc.h"
#include <algorithm>
#include <cstdio>
#include <cstring>
#include <mutex>
#include <vector>

#include "../../include/vmgpu.h"

#define DWAVE 64
#define DBLOCK 256
#define DMAX_ROWS 8192          /* maxRowsPerBlock, lib/storage/block.go:14 */
#define DMAX_PAYLOAD (DMAX_ROWS * 10 + 16)
#define DGRID 1024              /* workgroups; each owns a scratch column */

namespace {

int dset_err(char* errbuf, size_t len, const char* msg) {
  if (errbuf && len) snprintf(errbuf, len, "%s", msg);
  return 1;
}

int dhip_err(char* errbuf, size_t len, const char* what, hipError_t e) {
  if (errbuf && len) snprintf(errbuf, len, "%s: %s", what, hipGetErrorString(e));
  return 2;
}

#define DHIP_TRY(expr, what)                                               \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) return dhip_err(errbuf, errbuf_len, what, _e);   \
  } while (0)

}  // namespace

/* decimal special values (lib/decimal/decimal.go:403-419) */
#define D_VINF_POS 0x7fffffffffffffffLL
#define D_VINF_NEG (-0x7fffffffffffffffLL - 1)
#define D_VSTALE   0x7ffffffffffffffeLL
#define D_VMAX     0x7ffffffffffffffdLL
#define D_VMIN     (-0x7fffffffffffffffLL)

static __device__ __forceinline__ double dec_to_float(long long v, double e10,
                                                      int e_neg) {
  if (v > D_VMAX || v < D_VMIN) {
    if (v == D_VINF_POS) return __longlong_as_double(0x7ff0000000000000LL);
    if (v == D_VINF_NEG) return __longlong_as_double(0xfff0000000000000LL);
    return __longlong_as_double(0x7ff0000000000002LL); /* StaleNaN */
  }
  double f = (double)v;
  return e_neg ? f / e10 : f * e10;
}

struct DecState {
  /* per-workgroup scratch column [DMAX_ROWS] of decoded deltas/values */
  long long* scratch;
};

/* mask words + per-word counts + prefix in LDS:
 * words = ceil(payload_len/64) <= DMAX_PAYLOAD/64 = 1282 */
#define DMAX_WORDS ((DMAX_PAYLOAD + 63) / 64)

struct __align__(16) DecLds {
  unsigned long long mask[DMAX_WORDS];
  unsigned short wcount[DMAX_WORDS];  /* terminators per word */
  unsigned int wprefix[DMAX_WORDS + 1];
  unsigned int carry_lo[DBLOCK / DWAVE];
  long long carry[DBLOCK / DWAVE + 1];
  int err;
};

/* exact u64 wave inclusive prefix sum (int wrap == Go semantics) */
static __device__ __forceinline__ unsigned long long wave_prefix_u64(
    unsigned long long x, int lane) {
  for (int d = 1; d < DWAVE; d <<= 1) {
    unsigned long long o = __shfl_up(x, d);
    if (lane >= d) x += o;
  }
  return x;
}

__global__ __launch_bounds__(DBLOCK) void decode_blocks_kernel(
    const uint8_t* payload, const vmgpu_block_desc* blocks, uint32_t n_blocks,
    long long* scratch_all, int64_t* out_ts, double* out_vals, int* err_flag) {
  __shared__ DecLds L;
  const int tid = threadIdx.x;
  const int lane = tid % DWAVE;
  const int wv = tid / DWAVE;

  for (uint32_t bi = blockIdx.x; bi < n_blocks; bi += gridDim.x) {
    const vmgpu_block_desc bd = blocks[bi];
    long long* scratch = scratch_all + (size_t)blockIdx.x * DMAX_ROWS;
    /* two streams per block: timestamps then values */
    for (int stream = 0; stream < 2; stream++) {
      const uint8_t mt = stream ? bd.val_mt : bd.ts_mt;
      const uint64_t off = stream ? bd.val_data_off : bd.ts_data_off;
      const uint32_t len = stream ? bd.val_data_len : bd.ts_data_len;
      const long long first = stream ? bd.first_value : bd.min_timestamp;
      const uint32_t rows = bd.rows;
      const uint8_t* src = payload + off;

      int nvarints = 0;
      if (mt == 3) {                      /* const */
        nvarints = 0;
      } else if (mt == 2) {               /* delta const */
        nvarints = 1;
      } else if (mt == 5 || mt == 6) {    /* nearest delta2 / nearest delta */
        nvarints = (int)rows - 1;
      } else {
        if (tid == 0) L.err = 10 + mt;
        __syncthreads();
        atomicExch(err_flag, L.err);
        return;
      }

      /* ---- phase 1: terminator masks ---- */
      int words = (int)((len + 63) / 64);
      if (words > DMAX_WORDS) {
        if (tid == 0) atomicExch(err_flag, 2);
        return;
      }
      for (int w = wv; w < words; w += DBLOCK / DWAVE) {
        uint32_t byte_idx = (uint32_t)w * 64 + lane;
        bool term = byte_idx < len && src[byte_idx] < 0x80;
        unsigned long long m = __ballot(term);
        if (lane == 0) {
          L.mask[w] = m;
          L.wcount[w] = (unsigned short)__popcll(m);
        }
      }
      __syncthreads();
      /* exact prefix over words: one wave, carry across rounds */
      if (wv == 0) {
        unsigned int carry = 0;
        for (int base = 0; base < words; base += DWAVE) {
          int w = base + lane;
          unsigned long long c = (w < words) ? L.wcount[w] : 0;
          unsigned long long p = wave_prefix_u64(c, lane);
          if (w < words) L.wprefix[w + 1] = carry + (unsigned int)p;
          carry += (unsigned int)__shfl(p, DWAVE - 1);
        }
        if (lane == 0) {
          L.wprefix[0] = 0;
          L.err = 0;
        }
      }
      __syncthreads();
      int total_terms = (words > 0) ? (int)L.wprefix[words] : 0;
      if (total_terms != nvarints) {
        /* payload length / varint count mismatch */
        if (tid == 0) atomicExch(err_flag, 3);
        return;
      }

      /* ---- phase 2: per-varint decode into scratch ---- */
      for (int v = tid; v < nvarints; v += DBLOCK) {
        uint32_t start;
        if (v == 0) {
          start = 0;
        } else {
          /* locate terminator #(v-1): binary search words by prefix */
          int lo = 0, hi = words - 1;
          while (lo < hi) {
            int mid = (lo + hi) >> 1;
            if ((int)L.wprefix[mid + 1] <= v - 1) lo = mid + 1;
            else hi = mid;
          }
          int within = (v - 1) - (int)L.wprefix[lo];
          unsigned long long m = L.mask[lo];
          /* select the (within)-th set bit */
          for (int s = 0; s < within; s++) m &= m - 1;
          int bit = __ffsll((unsigned long long)m) - 1;
          start = (uint32_t)lo * 64 + (uint32_t)bit + 1;
        }
        unsigned long long u = 0;
        int shift = 0;
        uint32_t p = start;
        for (;;) {
          if (p >= len || shift > 63) {
            atomicExch(err_flag, 4);
            break;
          }
          uint8_t c = src[p++];
          u |= (unsigned long long)(c & 0x7f) << shift;
          if (c < 0x80) break;
          shift += 7;
        }
        /* zigzag (int.go:184) */
        long long d = (long long)(u >> 1) ^ -(long long)(u & 1);
        scratch[v] = d;
      }
      __syncthreads();

      /* ---- phase 3: reconstruct via exact prefix sums ---- */
      /* inclusive prefix over scratch[0..nv): block-cooperative, one wave
       * per 64-chunk with cross-wave carries through LDS */
      int passes = (mt == 5) ? 2 : (mt == 6 ? 1 : 0);
      for (int pass = 0; pass < passes; pass++) {
        /* for delta2: pass 0 turns d2[] into d1[] (starting at scratch[0]
         * = d1 already: prefix starting from index 0 across all entries);
         * pass 1 prefixes d1[] into cumulative deltas */
        long long run = 0;
        for (int base = 0; base < nvarints; base += DWAVE) {
          /* single wave (wave 0) processes sequential chunks to keep the
           * carry exact and simple; other waves idle in this loop */
          if (wv == 0) {
            int k = base + lane;
            unsigned long long x = (k < nvarints) ? (unsigned long long)scratch[k] : 0;
            unsigned long long pfx = wave_prefix_u64(x, lane);
            if (k < nvarints)
              scratch[k] = (long long)((unsigned long long)run + pfx);
            run = (long long)((unsigned long long)run +
                              (unsigned long long)__shfl(pfx, DWAVE - 1));
          }
        }
        __syncthreads();
      }

      /* ---- phase 4: materialize rows ---- */
      int64_t* dst_ts = out_ts + bd.out_off;
      double* dst_vals = out_vals + bd.out_off;
      const double e10 = bd.e10;
      const int e_neg = bd.scale < 0;
      if (mt == 3) { /* const */
        for (uint32_t r = tid; r < rows; r += DBLOCK) {
          if (stream) dst_vals[r] = dec_to_float(first, e10, e_neg);
          else dst_ts[r] = first;
        }
      } else if (mt == 2) { /* delta const: v += d each row */
        long long d = 0;
        /* scratch[0] holds the delta after phase 2 (single varint) */
        d = scratch[0];
        for (uint32_t r = tid; r < rows; r += DBLOCK) {
          long long v = (long long)((unsigned long long)first +
                                    (unsigned long long)d * r);
          if (stream) dst_vals[r] = dec_to_float(v, e10, e_neg);
          else dst_ts[r] = v;
        }
      } else {
        /* nearest delta / delta2: row 0 = first; row r>0 = first + scratch[r-1] */
        for (uint32_t r = tid; r < rows; r += DBLOCK) {
          long long v = (r == 0) ? first
                                 : (long long)((unsigned long long)first +
                                               (unsigned long long)scratch[r - 1]);
          if (stream) dst_vals[r] = dec_to_float(v, e10, e_neg);
          else dst_ts[r] = v;
        }
      }
      __syncthreads();

      /* timestamp post-processing */
      if (!stream) {
        if (bd.precision_bits < 64) {
          /* EnsureNonDecreasingSequence (encoding.go:255-286): prefix max
           * with first/last pinned — single wave, exact */
          if (wv == 0) {
            if (lane == 0) {
              dst_ts[0] = bd.min_timestamp;
              dst_ts[rows - 1] = bd.max_timestamp;
            }
          }
          __syncthreads();
          if (wv == 0) {
            long long runmax = (long long)0x8000000000000000LL;
            for (uint32_t base = 0; base < rows; base += DWAVE) {
              uint32_t k = base + lane;
              long long t = (k < rows) ? dst_ts[k] : (long long)0x8000000000000000LL;
              /* inclusive prefix max */
              long long x = t;
              for (int d = 1; d < DWAVE; d <<= 1) {
                long long o = __shfl_up(x, d);
                if (lane >= d && o > x) x = o;
              }
              if (runmax > x) x = runmax;
              if (k < rows) dst_ts[k] = x;
              runmax = __shfl(x, DWAVE - 1);
            }
            /* re-pin the tail below vMax (second half of Ensure...) */
            long long vmax = bd.max_timestamp;
            for (uint32_t k = lane; k < rows; k += DWAVE) {
              long long t = dst_ts[k];
              if (t > vmax) dst_ts[k] = vmax;
            }
          }
          __syncthreads();
        } else if (mt == 5 || mt == 6) {
          /* checkTimestampsBounds (block.go:329-350) */
          bool bad = false;
          for (uint32_t k = tid; k < rows; k += DBLOCK) {
            long long t = dst_ts[k];
            if (t < bd.min_timestamp || t > bd.max_timestamp) bad = true;
            if (k + 1 < rows && dst_ts[k + 1] < t) bad = true;
          }
          if (bad) atomicExch(err_flag, 5);
          __syncthreads();
        }
      }
      __syncthreads();
    }
  }
}

extern "C" {

int vmgpu_decode_blocks(const uint8_t* payload, uint64_t payload_len,
                        const vmgpu_block_desc* blocks, uint32_t n_blocks,
                        uint64_t total_rows,
                        int64_t* out_ts, double* out_vals,
                        char* errbuf, size_t errbuf_len) {
  if (!payload || !blocks || !out_ts || !out_vals || n_blocks == 0)
    return dset_err(errbuf, errbuf_len, "vmgpu: bad args");
  hipStream_t st = vm_ctx_stream();
  uint8_t* d_payload = nullptr;
  vmgpu_block_desc* d_blocks = nullptr;
  long long* d_scratch = nullptr;
  int64_t* d_ts = nullptr;
  double* d_vals = nullptr;
  int* d_err = nullptr;
  DHIP_TRY(vm_dev_malloc(&d_payload, payload_len ? payload_len : 1), "alloc payload");
  DHIP_TRY(vm_dev_malloc(&d_blocks, (size_t)n_blocks * sizeof(vmgpu_block_desc)), "alloc descs");
  DHIP_TRY(vm_dev_malloc(&d_scratch, (size_t)DGRID * DMAX_ROWS * 8), "alloc scratch");
  DHIP_TRY(vm_dev_malloc(&d_ts, (size_t)total_rows * 8), "alloc ts");
  DHIP_TRY(vm_dev_malloc(&d_vals, (size_t)total_rows * 8), "alloc vals");
  DHIP_TRY(vm_dev_malloc(&d_err, 4), "alloc err");
  DHIP_TRY(hipMemcpyAsync(d_payload, payload, payload_len, hipMemcpyHostToDevice, st), "ul payload");
  DHIP_TRY(hipMemcpyAsync(d_blocks, blocks, (size_t)n_blocks * sizeof(vmgpu_block_desc),
                          hipMemcpyHostToDevice, st), "ul descs");
  DHIP_TRY(hipMemsetAsync(d_err, 0, 4, st), "zero err");
  uint32_t grid = std::min<uint32_t>(n_blocks, DGRID);
  hipLaunchKernelGGL(decode_blocks_kernel, dim3(grid), dim3(DBLOCK), 0, st,
                     d_payload, d_blocks, n_blocks, d_scratch, d_ts, d_vals, d_err);
  int err_h = 0;
  DHIP_TRY(hipMemcpyAsync(&err_h, d_err, 4, hipMemcpyDeviceToHost, st), "dl err");
  DHIP_TRY(hipMemcpyAsync(out_ts, d_ts, (size_t)total_rows * 8, hipMemcpyDeviceToHost, st), "dl ts");
  DHIP_TRY(hipMemcpyAsync(out_vals, d_vals, (size_t)total_rows * 8, hipMemcpyDeviceToHost, st), "dl vals");
  DHIP_TRY(hipStreamSynchronize(st), "sync");
  hipError_t kerr = hipGetLastError();
  (void)vm_dev_free(d_payload);
  (void)vm_dev_free(d_blocks);
  (void)vm_dev_free(d_scratch);
  (void)vm_dev_free(d_ts);
  (void)vm_dev_free(d_vals);
  (void)vm_dev_free(d_err);
  if (kerr != hipSuccess) return dhip_err(errbuf, errbuf_len, "decode kernel", kerr);
  if (err_h != 0) {
    char msg[64];
    snprintf(msg, sizeof(msg), "vmgpu: block decode error %d", err_h);
    return dset_err(errbuf, errbuf_len, msg);
  }
  return 0;
}

}  /* extern "C" */


/* ------------------------------------------------------------------ */
/* per-series k-way merge + dedup (netstorage.go:564 mergeSortBlocks, */
/* lib/storage/dedup.go:29 DeduplicateSamples)                        */
/* ------------------------------------------------------------------ */

/* One wave per series.  Fast path: when the series' blocks are disjoint in
 * time (the overwhelmingly common case: LSM parts hold distinct time
 * ranges), the merge is a concatenation of the blocks in min-timestamp
 * order — wave-parallel coalesced copies.  Overlapping blocks fall back to
 * the exact serial heap merge on lane 0 (matching the reference's
 * container/heap order for duplicate timestamps).  Dedup (when enabled)
 * runs as a lane-cooperative pass identical to DeduplicateSamples. */

#define MAX_BLOCKS_PER_SERIES 64

static __device__ __forceinline__ void d_wave_sync() {
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
}

static __device__ int d_is_stale(double v) {
  return __double_as_longlong(v) == 0x7ff0000000000002LL;
}

__global__ __launch_bounds__(DBLOCK) void merge_blocks_kernel(
    const int64_t* ts, const double* vals,
    const uint64_t* block_offsets,      /* [n_blocks+1] into ts/vals */
    const uint32_t* series_block_start, /* [n_series+1]: block range per series */
    const uint64_t* out_offsets,        /* [n_series+1] capacity offsets */
    uint32_t n_series, int64_t dedup_interval,
    int64_t* out_ts, double* out_vals,
    uint64_t* out_counts,               /* [n_series] merged lengths */
    int* err_flag) {
  const int lane = threadIdx.x % DWAVE;
  const int wv = threadIdx.x / DWAVE;
  uint32_t wid = blockIdx.x * (DBLOCK / DWAVE) + wv;
  uint32_t stride = gridDim.x * (DBLOCK / DWAVE);
  for (uint32_t s = wid; s < n_series; s += stride) {
    uint32_t b0 = series_block_start[s];
    uint32_t b1 = series_block_start[s + 1];
    uint32_t nb = b1 - b0;
    int64_t* dts = out_ts + out_offsets[s];
    double* dvs = out_vals + out_offsets[s];
    if (nb == 0) {
      if (lane == 0) out_counts[s] = 0;
      continue;
    }
    if (nb > MAX_BLOCKS_PER_SERIES) {
      if (lane == 0) atomicExch(err_flag, 20);
      continue;
    }
    /* order blocks by first timestamp (insertion order on lane 0, tiny nb),
     * detect overlap */
    uint32_t order[MAX_BLOCKS_PER_SERIES];
    uint32_t nb_nonempty = 0;
    bool overlap = false;
    if (lane == 0) {
      for (uint32_t b = 0; b < nb; b++) {
        uint64_t lo = block_offsets[b0 + b], hi = block_offsets[b0 + b + 1];
        if (hi == lo) continue; /* skip empty blocks (mergeSortBlocks:570) */
        uint32_t idx = nb_nonempty++;
        order[idx] = b;
        while (idx > 0) {
          int64_t cur_first = ts[block_offsets[b0 + order[idx]]];
          int64_t prev_first = ts[block_offsets[b0 + order[idx - 1]]];
          if (prev_first <= cur_first) break;
          uint32_t t = order[idx];
          order[idx] = order[idx - 1];
          order[idx - 1] = t;
          idx--;
        }
      }
      for (uint32_t i = 1; i < nb_nonempty; i++) {
        uint64_t prev_end = block_offsets[b0 + order[i - 1] + 1];
        uint64_t cur_lo = block_offsets[b0 + order[i]];
        if (ts[prev_end - 1] >= ts[cur_lo]) overlap = true;
      }
    }
    nb_nonempty = __shfl(nb_nonempty, 0);
    overlap = __shfl((int)overlap, 0);
    for (uint32_t i = 0; i < nb_nonempty; i++) order[i] = __shfl(order[i], 0);

    uint64_t n_merged = 0;
    if (!overlap) {
      /* concatenate in order: wave-parallel coalesced copy */
      for (uint32_t i = 0; i < nb_nonempty; i++) {
        uint64_t lo = block_offsets[b0 + order[i]];
        uint64_t hi = block_offsets[b0 + order[i] + 1];
        for (uint64_t k = lane; k < hi - lo; k += DWAVE) {
          dts[n_merged + k] = ts[lo + k];
          dvs[n_merged + k] = vals[lo + k];
        }
        n_merged += hi - lo;
      }
      d_wave_sync();
    } else if (lane == 0) {
      /* exact serial heap merge (mergeSortBlocks + getNextBlock +
       * equalSamplesPrefix semantics) — Go container/heap order */
      uint64_t next[MAX_BLOCKS_PER_SERIES];
      uint64_t lim[MAX_BLOCKS_PER_SERIES];
      uint32_t heap[MAX_BLOCKS_PER_SERIES];
      uint32_t hn = 0;
      /* heap seeded in INPUT order (skipping empties), exactly as the
       * reference builds sbs before heap.Init — initial array order decides
       * tie-breaking for equal timestamps */
      for (uint32_t b = 0; b < nb; b++) {
        uint64_t lo = block_offsets[b0 + b], hi = block_offsets[b0 + b + 1];
        if (hi == lo) continue;
        next[b] = lo;
        lim[b] = hi;
        heap[hn++] = b;
      }
      /* heap.Init: siftDown from hn/2-1 */
      for (int i = (int)hn / 2 - 1; i >= 0; i--) {
        uint32_t j = (uint32_t)i;
        for (;;) {
          uint32_t c = 2 * j + 1;
          if (c >= hn) break;
          if (c + 1 < hn && ts[next[heap[c + 1]]] < ts[next[heap[c]]]) c++;
          if (ts[next[heap[c]]] >= ts[next[heap[j]]]) break;
          uint32_t t = heap[j]; heap[j] = heap[c]; heap[c] = t;
          j = c;
        }
      }
      while (hn > 0) {
        uint32_t top = heap[0];
        if (hn == 1) {
          for (uint64_t k = next[top]; k < lim[top]; k++) {
            dts[n_merged] = ts[k];
            dvs[n_merged] = vals[k];
            n_merged++;
          }
          break;
        }
        uint32_t nxt = (hn < 3) ? heap[1]
                       : (ts[next[heap[2]]] < ts[next[heap[1]]] ? heap[2] : heap[1]);
        int64_t ts_next = ts[next[nxt]];
        /* equalSamplesPrefix: identical (ts, value-bits) prefixes are
         * replicated samples; skip them at top when dedup is enabled */
        uint64_t neq = 0;
        {
          uint64_t a = next[top], b = next[nxt];
          while (a + neq < lim[top] && b + neq < lim[nxt] &&
                 ts[a + neq] == ts[b + neq] &&
                 __double_as_longlong(vals[a + neq]) ==
                     __double_as_longlong(vals[b + neq]))
            neq++;
        }
        if (neq > 0 && dedup_interval > 0) {
          next[top] += neq;
        } else {
          while (next[top] < lim[top] && ts[next[top]] <= ts_next) {
            dts[n_merged] = ts[next[top]];
            dvs[n_merged] = vals[next[top]];
            n_merged++;
            next[top]++;
          }
        }
        if (next[top] < lim[top]) {
          /* heap.Fix(0) = siftDown then siftUp (top only needs down) */
          uint32_t j = 0;
          for (;;) {
            uint32_t c = 2 * j + 1;
            if (c >= hn) break;
            if (c + 1 < hn && ts[next[heap[c + 1]]] < ts[next[heap[c]]]) c++;
            if (ts[next[heap[c]]] >= ts[next[heap[j]]]) break;
            uint32_t t = heap[j]; heap[j] = heap[c]; heap[c] = t;
            j = c;
          }
        } else {
          /* heap.Pop */
          heap[0] = heap[hn - 1];
          hn--;
          uint32_t j = 0;
          for (;;) {
            uint32_t c = 2 * j + 1;
            if (c >= hn) break;
            if (c + 1 < hn && ts[next[heap[c + 1]]] < ts[next[heap[c]]]) c++;
            if (ts[next[heap[c]]] >= ts[next[heap[j]]]) break;
            uint32_t t = heap[j]; heap[j] = heap[c]; heap[c] = t;
            j = c;
          }
        }
      }
    }
    n_merged = __shfl((unsigned long long)n_merged, 0);
    d_wave_sync();

    /* DeduplicateSamples (dedup.go:29-92), serial exact on lane 0 (rare:
     * needsDedup gate runs wave-parallel first) */
    if (dedup_interval > 0 && n_merged >= 2) {
      bool needs = false;
      /* needsDedup (dedup.go:149): sequential bucket walk — cheap serial */
      if (lane == 0) {
        int64_t ts_next_b = dts[0] + dedup_interval - 1;
        ts_next_b -= ts_next_b % dedup_interval;
        for (uint64_t i = 1; i < n_merged; i++) {
          int64_t t = dts[i];
          if (t <= ts_next_b) { needs = true; break; }
          ts_next_b += dedup_interval;
          if (ts_next_b < t) {
            ts_next_b = t + dedup_interval - 1;
            ts_next_b -= ts_next_b % dedup_interval;
          }
        }
      }
      needs = __shfl((int)needs, 0);
      if (needs && lane == 0) {
        int64_t ts_next_b = dts[0] + dedup_interval - 1;
        ts_next_b -= ts_next_b % dedup_interval;
        uint64_t k = 0;
        for (uint64_t i = 1; i < n_merged; i++) {
          int64_t t = dts[i];
          if (t <= ts_next_b) continue;
          uint64_t j = i - 1;
          int64_t tp = dts[j];
          double vp = dvs[j];
          while (j > 0 && dts[j - 1] == tp) {
            j--;
            if (d_is_stale(dvs[j])) continue;
            if (d_is_stale(vp)) { vp = dvs[j]; continue; }
            if (dvs[j] > vp) vp = dvs[j];
          }
          dts[k] = tp;
          dvs[k] = vp;
          k++;
          ts_next_b += dedup_interval;
          if (ts_next_b < t) {
            ts_next_b = t + dedup_interval - 1;
            ts_next_b -= ts_next_b % dedup_interval;
          }
        }
        uint64_t j = n_merged - 1;
        int64_t tp = dts[j];
        double vp = dvs[j];
        while (j > 0 && dts[j - 1] == tp) {
          j--;
          if (d_is_stale(dvs[j])) continue;
          if (d_is_stale(vp)) { vp = dvs[j]; continue; }
          if (dvs[j] > vp) vp = dvs[j];
        }
        dts[k] = tp;
        dvs[k] = vp;
        k++;
        n_merged = k;
      }
      n_merged = __shfl((unsigned long long)n_merged, 0);
    }
    if (lane == 0) out_counts[s] = n_merged;
    d_wave_sync();
  }
}

extern "C" {

int vmgpu_merge_blocks(const int64_t* ts, const double* vals,
                       const uint64_t* block_offsets, uint32_t n_blocks,
                       const uint32_t* series_block_start, uint32_t n_series,
                       int64_t dedup_interval,
                       int64_t* out_ts, double* out_vals,
                       uint64_t* out_offsets, uint64_t* out_counts,
                       char* errbuf, size_t errbuf_len) {
  if (!ts || !vals || !block_offsets || !series_block_start || !out_ts ||
      !out_vals || !out_offsets || !out_counts || n_series == 0)
    return dset_err(errbuf, errbuf_len, "vmgpu: bad args");
  hipStream_t st = vm_ctx_stream();
  uint64_t total = block_offsets[n_blocks];
  /* capacity offsets = pre-merge block extents per series */
  std::vector<uint64_t> cap_off(n_series + 1);
  cap_off[0] = 0;
  for (uint32_t s = 0; s < n_series; s++) {
    uint64_t lo = block_offsets[series_block_start[s]];
    uint64_t hi = block_offsets[series_block_start[s + 1]];
    cap_off[s + 1] = cap_off[s] + (hi - lo);
  }
  int64_t* d_ts = nullptr;
  double* d_vals = nullptr;
  uint64_t* d_boff = nullptr;
  uint32_t* d_sbs = nullptr;
  uint64_t* d_ooff = nullptr;
  int64_t* d_ots = nullptr;
  double* d_ovals = nullptr;
  uint64_t* d_ocnt = nullptr;
  int* d_err = nullptr;
  DHIP_TRY(vm_dev_malloc(&d_ts, total * 8), "alloc ts");
  DHIP_TRY(vm_dev_malloc(&d_vals, total * 8), "alloc vals");
  DHIP_TRY(vm_dev_malloc(&d_boff, (size_t)(n_blocks + 1) * 8), "alloc boff");
  DHIP_TRY(vm_dev_malloc(&d_sbs, (size_t)(n_series + 1) * 4), "alloc sbs");
  DHIP_TRY(vm_dev_malloc(&d_ooff, (size_t)(n_series + 1) * 8), "alloc ooff");
  /* +16 B slack for the pipe kernel's clamped tail pair load */
  DHIP_TRY(vm_dev_malloc(&d_ots, cap_off[n_series] * 8 + 16), "alloc out ts");
  DHIP_TRY(vm_dev_malloc(&d_ovals, cap_off[n_series] * 8 + 16), "alloc out vals");
  DHIP_TRY(vm_dev_malloc(&d_ocnt, (size_t)n_series * 8), "alloc out counts");
  DHIP_TRY(vm_dev_malloc(&d_err, 4), "alloc err");
  DHIP_TRY(hipMemcpyAsync(d_ts, ts, total * 8, hipMemcpyHostToDevice, st), "ul ts");
  DHIP_TRY(hipMemcpyAsync(d_vals, vals, total * 8, hipMemcpyHostToDevice, st), "ul vals");
  DHIP_TRY(hipMemcpyAsync(d_boff, block_offsets, (size_t)(n_blocks + 1) * 8, hipMemcpyHostToDevice, st), "ul boff");
  DHIP_TRY(hipMemcpyAsync(d_sbs, series_block_start, (size_t)(n_series + 1) * 4, hipMemcpyHostToDevice, st), "ul sbs");
  DHIP_TRY(hipMemcpyAsync(d_ooff, cap_off.data(), (size_t)(n_series + 1) * 8, hipMemcpyHostToDevice, st), "ul ooff");
  DHIP_TRY(hipMemsetAsync(d_err, 0, 4, st), "zero err");
  uint32_t grid = std::min<uint32_t>((n_series + 3) / 4, 2048);
  hipLaunchKernelGGL(merge_blocks_kernel, dim3(grid), dim3(DBLOCK), 0, st,
                     d_ts, d_vals, d_boff, d_sbs, d_ooff, n_series,
                     dedup_interval, d_ots, d_ovals, d_ocnt, d_err);
  std::vector<uint64_t> counts(n_series);
  int err_h = 0;
  DHIP_TRY(hipMemcpyAsync(counts.data(), d_ocnt, (size_t)n_series * 8, hipMemcpyDeviceToHost, st), "dl counts");
  DHIP_TRY(hipMemcpyAsync(&err_h, d_err, 4, hipMemcpyDeviceToHost, st), "dl err");
  /* compact download: per series, copy the merged prefix */
  DHIP_TRY(hipStreamSynchronize(st), "sync");
  hipError_t kerr = hipGetLastError();
  if (kerr == hipSuccess && err_h == 0) {
    uint64_t w = 0;
    out_offsets[0] = 0;
    for (uint32_t s = 0; s < n_series; s++) {
      uint64_t n = counts[s];
      DHIP_TRY(hipMemcpyAsync(out_ts + w, d_ots + cap_off[s], n * 8,
                              hipMemcpyDeviceToHost, st), "dl merged ts");
      DHIP_TRY(hipMemcpyAsync(out_vals + w, d_ovals + cap_off[s], n * 8,
                              hipMemcpyDeviceToHost, st), "dl merged vals");
      w += n;
      out_offsets[s + 1] = w;
      out_counts[s] = n;
    }
    DHIP_TRY(hipStreamSynchronize(st), "sync dl");
  }
  (void)vm_dev_free(d_ts); (void)vm_dev_free(d_vals); (void)vm_dev_free(d_boff);
  (void)vm_dev_free(d_sbs); (void)vm_dev_free(d_ooff); (void)vm_dev_free(d_ots);
  (void)vm_dev_free(d_ovals); (void)vm_dev_free(d_ocnt); (void)vm_dev_free(d_err);
  if (kerr != hipSuccess) return dhip_err(errbuf, errbuf_len, "merge kernel", kerr);
  if (err_h != 0) {
    char msg[64];
    snprintf(msg, sizeof(msg), "vmgpu: merge error %d", err_h);
    return dset_err(errbuf, errbuf_len, msg);
  }
  return 0;
}

}  /* extern "C" */

/* ------------------------------------------------------------------ */
/* fused decode -> merge -> resident batch (cold-cache query path)    */
/* ------------------------------------------------------------------ */

/* gather each series' merged prefix [cap_off[s], cap_off[s]+counts[s])
 * into the dense final CSR at final_off[s] — wave-parallel per series */
__global__ __launch_bounds__(DBLOCK) void compact_series_kernel(
    const int64_t* src_ts, const double* src_vals, const uint64_t* cap_off,
    const uint64_t* final_off, uint32_t n_series,
    int64_t* dst_ts, double* dst_vals) {
  const int lane = threadIdx.x % DWAVE;
  const int wv = threadIdx.x / DWAVE;
  uint32_t wid = blockIdx.x * (DBLOCK / DWAVE) + wv;
  uint32_t stride = gridDim.x * (DBLOCK / DWAVE);
  for (uint32_t s = wid; s < n_series; s += stride) {
    uint64_t src = cap_off[s];
    uint64_t dst = final_off[s];
    uint64_t n = final_off[s + 1] - dst;
    for (uint64_t k = lane; k < n; k += DWAVE) {
      dst_ts[dst + k] = src_ts[src + k];
      dst_vals[dst + k] = src_vals[src + k];
    }
  }
}

/* Internal (cross-TU) seam used by vmgpu_batch_create_from_blocks:
 * decode all blocks, merge+dedup per series, compact to a dense CSR —
 * everything device-resident.  On success *out_d_ts / *out_d_vals are
 * device buffers owned by the caller and h_final_offsets[n_series+1]
 * holds the CSR offsets. */
extern "C" int vmdec_decode_merge_device(
    const uint8_t* payload, uint64_t payload_len,
    const vmgpu_block_desc* blocks, uint32_t n_blocks, uint64_t total_rows,
    const uint32_t* series_block_start, uint32_t n_series,
    int64_t dedup_interval, hipStream_t st,
    int64_t** out_d_ts, double** out_d_vals, uint64_t* h_final_offsets,
    char* errbuf, size_t errbuf_len) {
  if (!payload || !blocks || !series_block_start || !out_d_ts || !out_d_vals ||
      !h_final_offsets || n_blocks == 0 || n_series == 0)
    return dset_err(errbuf, errbuf_len, "vmgpu: bad fused decode args");
  /* host-side offsets for the merge: block b rows at [out_off, out_off+rows) */
  std::vector<uint64_t> boff(n_blocks + 1);
  for (uint32_t b = 0; b < n_blocks; b++) boff[b] = blocks[b].out_off;
  boff[n_blocks] = total_rows;
  std::vector<uint64_t> cap_off(n_series + 1);
  cap_off[0] = 0;
  for (uint32_t s = 0; s < n_series; s++) {
    uint64_t lo = boff[series_block_start[s]];
    uint64_t hi = boff[series_block_start[s + 1]];
    cap_off[s + 1] = cap_off[s] + (hi - lo);
  }

  uint8_t* d_payload = nullptr;
  vmgpu_block_desc* d_blocks = nullptr;
  long long* d_scratch = nullptr;
  int64_t* d_raw_ts = nullptr;
  double* d_raw_vals = nullptr;
  uint64_t* d_boff = nullptr;
  uint32_t* d_sbs = nullptr;
  uint64_t* d_capoff = nullptr;
  int64_t* d_mts = nullptr;
  double* d_mvals = nullptr;
  uint64_t* d_cnt = nullptr;
  uint64_t* d_finaloff = nullptr;
  int* d_err = nullptr;
  int64_t* d_fts = nullptr;
  double* d_fvals = nullptr;
  int rc = 0;
  hipError_t kerr = hipSuccess;
  std::vector<uint64_t> counts(n_series);
  int err_h = 0;
  uint32_t grid = 0, mgrid = 0;

#define FDM_TRY(expr, what)                                                  \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) { rc = dhip_err(errbuf, errbuf_len, what, _e);     \
      goto cleanup; }                                                        \
  } while (0)

  FDM_TRY(vm_dev_malloc(&d_payload, payload_len ? payload_len : 1), "alloc payload");
  FDM_TRY(vm_dev_malloc(&d_blocks, (size_t)n_blocks * sizeof(vmgpu_block_desc)), "alloc descs");
  FDM_TRY(vm_dev_malloc(&d_scratch, (size_t)DGRID * DMAX_ROWS * 8), "alloc scratch");
  FDM_TRY(vm_dev_malloc(&d_raw_ts, (size_t)total_rows * 8), "alloc raw ts");
  FDM_TRY(vm_dev_malloc(&d_raw_vals, (size_t)total_rows * 8), "alloc raw vals");
  FDM_TRY(vm_dev_malloc(&d_err, 4), "alloc err");
  FDM_TRY(hipMemcpyAsync(d_payload, payload, payload_len ? payload_len : 1,
                         hipMemcpyHostToDevice, st), "ul payload");
  FDM_TRY(hipMemcpyAsync(d_blocks, blocks, (size_t)n_blocks * sizeof(vmgpu_block_desc),
                         hipMemcpyHostToDevice, st), "ul descs");
  FDM_TRY(hipMemsetAsync(d_err, 0, 4, st), "zero err");
  grid = std::min<uint32_t>(n_blocks, DGRID);
  hipLaunchKernelGGL(decode_blocks_kernel, dim3(grid), dim3(DBLOCK), 0, st,
                     d_payload, d_blocks, n_blocks, d_scratch,
                     d_raw_ts, d_raw_vals, d_err);

  FDM_TRY(vm_dev_malloc(&d_boff, (size_t)(n_blocks + 1) * 8), "alloc boff");
  FDM_TRY(vm_dev_malloc(&d_sbs, (size_t)(n_series + 1) * 4), "alloc sbs");
  FDM_TRY(vm_dev_malloc(&d_capoff, (size_t)(n_series + 1) * 8), "alloc capoff");
  FDM_TRY(vm_dev_malloc(&d_mts, (cap_off[n_series] ? cap_off[n_series] : 1) * 8), "alloc merged ts");
  FDM_TRY(vm_dev_malloc(&d_mvals, (cap_off[n_series] ? cap_off[n_series] : 1) * 8), "alloc merged vals");
  FDM_TRY(vm_dev_malloc(&d_cnt, (size_t)n_series * 8), "alloc counts");
  FDM_TRY(hipMemcpyAsync(d_boff, boff.data(), (size_t)(n_blocks + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul boff");
  FDM_TRY(hipMemcpyAsync(d_sbs, series_block_start, (size_t)(n_series + 1) * 4,
                         hipMemcpyHostToDevice, st), "ul sbs");
  FDM_TRY(hipMemcpyAsync(d_capoff, cap_off.data(), (size_t)(n_series + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul capoff");
  mgrid = std::min<uint32_t>((n_series + 3) / 4, 2048);
  hipLaunchKernelGGL(merge_blocks_kernel, dim3(mgrid), dim3(DBLOCK), 0, st,
                     d_raw_ts, d_raw_vals, d_boff, d_sbs, d_capoff, n_series,
                     dedup_interval, d_mts, d_mvals, d_cnt, d_err);
  FDM_TRY(hipMemcpyAsync(counts.data(), d_cnt, (size_t)n_series * 8,
                         hipMemcpyDeviceToHost, st), "dl counts");
  FDM_TRY(hipMemcpyAsync(&err_h, d_err, 4, hipMemcpyDeviceToHost, st), "dl err");
  FDM_TRY(hipStreamSynchronize(st), "sync fused");
  kerr = hipGetLastError();
  if (kerr != hipSuccess) { rc = dhip_err(errbuf, errbuf_len, "fused kernels", kerr); goto cleanup; }
  if (err_h != 0) {
    char msg[64];
    snprintf(msg, sizeof(msg), "vmgpu: fused decode/merge error %d", err_h);
    rc = dset_err(errbuf, errbuf_len, msg);
    goto cleanup;
  }

  h_final_offsets[0] = 0;
  for (uint32_t s = 0; s < n_series; s++)
    h_final_offsets[s + 1] = h_final_offsets[s] + counts[s];
  FDM_TRY(vm_dev_malloc(&d_finaloff, (size_t)(n_series + 1) * 8), "alloc finaloff");
  FDM_TRY(hipMemcpyAsync(d_finaloff, h_final_offsets, (size_t)(n_series + 1) * 8,
                         hipMemcpyHostToDevice, st), "ul finaloff");
  /* +16 B slack for the pipe kernel's clamped tail pair load */
  FDM_TRY(vm_dev_malloc(&d_fts, (h_final_offsets[n_series] ? h_final_offsets[n_series] : 1) * 8 + 16), "alloc final ts");
  FDM_TRY(vm_dev_malloc(&d_fvals, (h_final_offsets[n_series] ? h_final_offsets[n_series] : 1) * 8 + 16), "alloc final vals");
  hipLaunchKernelGGL(compact_series_kernel, dim3(mgrid), dim3(DBLOCK), 0, st,
                     d_mts, d_mvals, d_capoff, d_finaloff, n_series,
                     d_fts, d_fvals);
  FDM_TRY(hipStreamSynchronize(st), "sync compact");
  kerr = hipGetLastError();
  if (kerr != hipSuccess) { rc = dhip_err(errbuf, errbuf_len, "compact kernel", kerr); goto cleanup; }

  *out_d_ts = d_fts;
  *out_d_vals = d_fvals;
  d_fts = nullptr;
  d_fvals = nullptr;

cleanup:
  (void)vm_dev_free(d_payload); (void)vm_dev_free(d_blocks); (void)vm_dev_free(d_scratch);
  (void)vm_dev_free(d_raw_ts); (void)vm_dev_free(d_raw_vals); (void)vm_dev_free(d_boff);
  (void)vm_dev_free(d_sbs); (void)vm_dev_free(d_capoff); (void)vm_dev_free(d_mts);
  (void)vm_dev_free(d_mvals); (void)vm_dev_free(d_cnt); (void)vm_dev_free(d_finaloff);
  (void)vm_dev_free(d_err); (void)vm_dev_free(d_fts); (void)vm_dev_free(d_fvals);
#undef FDM_TRY
  return rc;
}
