/* vm_decimal.h — CPU oracle for lib/decimal + lib/encoding.
 * TEST INFRASTRUCTURE ONLY (see vm_oracle.h header note). */
#ifndef VM_DECIMAL_H
#define VM_DECIMAL_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- lib/decimal ---- */
int vm_decimal_is_special(int64_t v);
double vm_decimal_to_float(int64_t v, int16_t e);
void vm_decimal_append_to_float(double* dst, const int64_t* va, int64_t n, int16_t e);
void vm_decimal_positive_float_to_decimal(double f, int64_t* out_v, int16_t* out_e);
void vm_decimal_from_float(double f, int64_t* out_v, int16_t* out_e);
int16_t vm_decimal_max_up_exponent(int64_t v);
void vm_decimal_append_float_to_decimal(const double* src, int64_t n,
                                        int64_t* out_va, int16_t* out_e);
double vm_decimal_round_to_decimal_digits(double f, int digits);
double vm_decimal_round_to_significant_figures(double f, int digits);
int16_t vm_decimal_calibrate_scale(int64_t* a, int64_t na, int16_t ae,
                                   int64_t* b, int64_t nb, int16_t be);

/* ---- lib/encoding ---- */
/* Marshal types (encoding.go:20-43). */
enum {
  VM_MT_ZSTD_NEAREST_DELTA2 = 1,
  VM_MT_DELTA_CONST = 2,
  VM_MT_CONST = 3,
  VM_MT_ZSTD_NEAREST_DELTA = 4,
  VM_MT_NEAREST_DELTA2 = 5,
  VM_MT_NEAREST_DELTA = 6,
};

/* Variable-length ints (lib/encoding/int.go). Return bytes written/read;
 * read returns <=0 on error. */
size_t vm_marshal_varint64s(uint8_t* dst, const int64_t* vs, int64_t n);
int64_t vm_unmarshal_varint64s(int64_t* dst, int64_t n, const uint8_t* src, size_t src_len);

/* nearest delta codecs (lib/encoding/nearest_delta{,2}.go). out must hold
 * 10*(n-1) bytes. */
size_t vm_marshal_nearest_delta(uint8_t* dst, const int64_t* src, int64_t n,
                                uint8_t precision_bits, int64_t* out_first);
int vm_unmarshal_nearest_delta(int64_t* dst, const uint8_t* src, size_t src_len,
                               int64_t first_value, int64_t items);
size_t vm_marshal_nearest_delta2(uint8_t* dst, const int64_t* src, int64_t n,
                                 uint8_t precision_bits, int64_t* out_first);
int vm_unmarshal_nearest_delta2(int64_t* dst, const uint8_t* src, size_t src_len,
                                int64_t first_value, int64_t items);

/* marshalInt64Array / unmarshalInt64Array (encoding.go:119-250) with zstd via
 * the system libzstd (dlopen'd; returns -1 if zstd needed but unavailable).
 * dst must hold 10*n + 64 bytes.  Returns bytes written, sets *out_mt and
 * *out_first. */
int64_t vm_marshal_int64_array(uint8_t* dst, const int64_t* a, int64_t n,
                               uint8_t precision_bits, uint8_t* out_mt,
                               int64_t* out_first);
/* Returns 0 on success. */
int vm_unmarshal_int64_array(int64_t* dst, int64_t items, const uint8_t* src,
                             size_t src_len, uint8_t mt, int64_t first_value);

int64_t vm_zstd_decompress(uint8_t* dst, size_t cap, const uint8_t* src, size_t n);

/* detectors (encoding.go:288-366) */
int vm_is_const(const int64_t* a, int64_t n);
int vm_is_delta_const(const int64_t* a, int64_t n);
int vm_is_gauge(const int64_t* a, int64_t n);

/* EnsureNonDecreasingSequence (encoding.go:255-286) */
void vm_ensure_non_decreasing(int64_t* a, int64_t n, int64_t v_min, int64_t v_max);

/* DeduplicateSamples (lib/storage/dedup.go:29-92). In-place; returns new n. */
int64_t vm_deduplicate_samples(int64_t* ts, double* vals, int64_t n, int64_t dedup_interval);

/* mergeSortBlocks (netstorage.go:564-614): k-way merge of sorted blocks
 * (CSR via offsets, n_blocks entries) + DeduplicateSamples.  dst arrays must
 * hold the total sample count.  Returns merged length. */
int64_t vm_merge_sort_blocks(const int64_t* ts, const double* vals,
                             const uint64_t* offsets, int32_t n_blocks,
                             int64_t dedup_interval,
                             int64_t* dst_ts, double* dst_vals);

#ifdef __cplusplus
}
#endif
#endif
