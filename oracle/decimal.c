/* decimal.c — CPU oracle for lib/decimal (int64·10^e ↔ float64).
 *
 * TEST INFRASTRUCTURE ONLY (see vm_oracle.h header note).
 * Faithful C restatement of lib/decimal/decimal.go:
 *   CalibrateScale 12-70, AppendDecimalToFloat 100-167, AppendFloatToDecimal
 *   171-256, maxUpExponent 262-321, RoundToDecimalDigits 325-337,
 *   RoundToSignificantFigures 341-370, ToFloat 376-392, FromFloat 437-457,
 *   positiveFloatToDecimal 467-550, special values 403-431.
 */
#define _USE_MATH_DEFINES
#include "vm_decimal.h"
#include <math.h>
#ifndef M_LN2
#define M_LN2 0.693147180559945309417232121458176568
#endif
#ifndef M_LN10
#define M_LN10 2.30258509299404568401799145468436421
#endif
#include <string.h>

#define V_INF_POS ((int64_t)0x7fffffffffffffffLL)        /* 1<<63 - 1 */
#define V_INF_NEG ((int64_t)0x8000000000000000LL)        /* -1<<63 */
#define V_STALE_NAN ((int64_t)0x7ffffffffffffffeLL)      /* 1<<63 - 2 */
#define V_MAX ((int64_t)0x7ffffffffffffffdLL)            /* 1<<63 - 3 */
#define V_MIN (-(int64_t)0x7fffffffffffffffLL)           /* -1<<63 + 1 */
#define STALE_NAN_BITS 0x7ff0000000000002ULL
#define CONVERSION_PRECISION 1e12

static double stale_nan_f(void) {
  double v;
  uint64_t b = STALE_NAN_BITS;
  memcpy(&v, &b, 8);
  return v;
}

int vm_decimal_is_special(int64_t v) { return v > V_MAX || v < V_MIN; }

/* ToFloat (decimal.go:376-392) */
double vm_decimal_to_float(int64_t v, int16_t e) {
  if (vm_decimal_is_special(v)) {
    if (v == V_INF_POS) return INFINITY;
    if (v == V_INF_NEG) return -INFINITY;
    return stale_nan_f();
  }
  double f = (double)v;
  /* negative exponents divide for precision */
  if (e < 0) return f / pow(10.0, (double)(-e));
  return f * pow(10.0, (double)e);
}

/* AppendDecimalToFloat (decimal.go:100-167) */
void vm_decimal_append_to_float(double* dst, const int64_t* va, int64_t n, int16_t e) {
  if (e == 0) {
    for (int64_t i = 0; i < n; i++) {
      int64_t v = va[i];
      dst[i] = (double)v;
      if (!vm_decimal_is_special(v)) continue;
      if (v == V_INF_POS) dst[i] = INFINITY;
      else if (v == V_INF_NEG) dst[i] = -INFINITY;
      else dst[i] = stale_nan_f();
    }
    return;
  }
  if (e < 0) {
    double e10 = pow(10.0, (double)(-e));
    for (int64_t i = 0; i < n; i++) {
      int64_t v = va[i];
      dst[i] = (double)v / e10;
      if (!vm_decimal_is_special(v)) continue;
      if (v == V_INF_POS) dst[i] = INFINITY;
      else if (v == V_INF_NEG) dst[i] = -INFINITY;
      else dst[i] = stale_nan_f();
    }
    return;
  }
  double e10 = pow(10.0, (double)e);
  for (int64_t i = 0; i < n; i++) {
    int64_t v = va[i];
    dst[i] = (double)v * e10;
    if (!vm_decimal_is_special(v)) continue;
    if (v == V_INF_POS) dst[i] = INFINITY;
    else if (v == V_INF_NEG) dst[i] = -INFINITY;
    else dst[i] = stale_nan_f();
  }
}

/* getDecimalAndScale (decimal.go:480-500) */
static void get_decimal_and_scale(uint64_t u, int64_t* out_v, int16_t* out_scale) {
  int16_t scale = 0;
  while (u >= (1ULL << 55)) {
    u /= 10;
    scale++;
  }
  if (u % 10 != 0) {
    *out_v = (int64_t)u;
    *out_scale = scale;
    return;
  }
  u /= 10;
  scale++;
  while (u != 0 && u % 10 == 0) {
    u /= 10;
    scale++;
  }
  *out_v = (int64_t)u;
  *out_scale = scale;
}

/* positiveFloatToDecimalSlow (decimal.go:502-550) */
static void positive_float_to_decimal_slow(double f, int64_t* out_v, int16_t* out_scale) {
  int16_t scale = 0;
  double prec = CONVERSION_PRECISION;
  if (f > 1e6 || f < 1e-6) {
    if (f > 1e6) prec = 1e15;
    int exp;
    frexp(f, &exp);
    if (exp < -1022) exp = -1022;
    else if (exp > 1023) exp = 1023;
    scale = (int16_t)((double)exp * (M_LN2 / M_LN10));
    f *= pow(10.0, (double)(-(int)scale));
  }
  while (f < prec) {
    double x, frac;
    frac = modf(f, &x);
    if (frac * prec < x) {
      f = x;
      break;
    }
    if ((1 - frac) * prec < x) {
      f = x + 1;
      break;
    }
    f *= 100;
    scale -= 2;
  }
  uint64_t u = (uint64_t)f;
  if (u % 10 != 0) {
    *out_v = (int64_t)u;
    *out_scale = scale;
    return;
  }
  u /= 10;
  scale++;
  *out_v = (int64_t)u;
  *out_scale = scale;
}

/* positiveFloatToDecimal (decimal.go:467-478) */
void vm_decimal_positive_float_to_decimal(double f, int64_t* out_v, int16_t* out_e) {
  uint64_t u = (uint64_t)f;
  if ((double)u != f) {
    positive_float_to_decimal_slow(f, out_v, out_e);
    return;
  }
  if (u < (1ULL << 55) && u % 10 != 0) {
    *out_v = (int64_t)u;
    *out_e = 0;
    return;
  }
  get_decimal_and_scale(u, out_v, out_e);
}

/* FromFloat (decimal.go:437-457) */
void vm_decimal_from_float(double f, int64_t* out_v, int16_t* out_e) {
  if (f == 0) {
    *out_v = 0;
    *out_e = 0;
    return;
  }
  uint64_t bits;
  memcpy(&bits, &f, 8);
  if (bits == STALE_NAN_BITS) {
    *out_v = V_STALE_NAN;
    *out_e = 0;
    return;
  }
  if (isinf(f)) {
    *out_v = (f > 0) ? V_INF_POS : V_INF_NEG;
    *out_e = 0;
    return;
  }
  if (f > 0) {
    int64_t v;
    int16_t e;
    vm_decimal_positive_float_to_decimal(f, &v, &e);
    if (v > V_MAX) v = V_MAX;
    *out_v = v;
    *out_e = e;
    return;
  }
  int64_t v;
  int16_t e;
  vm_decimal_positive_float_to_decimal(-f, &v, &e);
  v = -v;
  if (v < V_MIN) v = V_MIN;
  *out_v = v;
  *out_e = e;
}

/* maxUpExponent (decimal.go:262-321) */
int16_t vm_decimal_max_up_exponent(int64_t v) {
  if (v == 0 || vm_decimal_is_special(v)) return 1024;
  if (v < 0) v = -v;
  if (v < 0) return 0; /* -1<<63 corner (cannot happen: special) */
  static const int64_t int64_max = 0x7fffffffffffffffLL;
  int16_t e = 18;
  int64_t lim = int64_max / 1000000000000000000LL;
  /* mirror the switch ladder: largest e with v <= int64Max/10^e */
  static const int64_t pows[19] = {
      1LL, 10LL, 100LL, 1000LL, 10000LL, 100000LL, 1000000LL, 10000000LL,
      100000000LL, 1000000000LL, 10000000000LL, 100000000000LL,
      1000000000000LL, 10000000000000LL, 100000000000000LL,
      1000000000000000LL, 10000000000000000LL, 100000000000000000LL,
      1000000000000000000LL};
  (void)lim;
  for (e = 18; e >= 1; e--) {
    if (v <= int64_max / pows[e]) return e;
  }
  return 0;
}

/* AppendFloatToDecimal (decimal.go:171-256): converts src floats into int64
 * decimals sharing one exponent. */
void vm_decimal_append_float_to_decimal(const double* src, int64_t n,
                                        int64_t* out_va, int16_t* out_e) {
  if (n == 0) {
    *out_e = 0;
    return;
  }
  /* per-item (v, e) */
  int16_t min_exp = 0x7fff;
  /* stack-free: two passes, store e in out_va temporarily? need both; use
   * heap */
  int16_t ea_static[256];
  int16_t* ea = ea_static;
  int64_t* va = out_va;
  int use_heap = n > 256;
  if (use_heap) ea = (int16_t*)__builtin_malloc((size_t)n * sizeof(int16_t));
  for (int64_t i = 0; i < n; i++) {
    int64_t v;
    int16_t e;
    vm_decimal_from_float(src[i], &v, &e);
    va[i] = v;
    ea[i] = e;
    if (e < min_exp && !vm_decimal_is_special(v)) min_exp = e;
  }
  int16_t down_exp = 0;
  for (int64_t i = 0; i < n; i++) {
    int16_t up_exp = (int16_t)(ea[i] - min_exp);
    int16_t max_up = vm_decimal_max_up_exponent(va[i]);
    if ((int16_t)(up_exp - max_up) > down_exp) down_exp = (int16_t)(up_exp - max_up);
  }
  min_exp = (int16_t)(min_exp + down_exp);
  for (int64_t i = 0; i < n; i++) {
    int64_t v = va[i];
    if (vm_decimal_is_special(v)) {
      va[i] = v;
      continue;
    }
    int16_t adj = (int16_t)(ea[i] - min_exp);
    while (adj > 0) {
      v *= 10;
      adj--;
    }
    while (adj < 0) {
      v /= 10;
      adj++;
    }
    va[i] = v;
  }
  if (use_heap) __builtin_free(ea);
  *out_e = min_exp;
}

/* RoundToDecimalDigits (decimal.go:325-337) */
/* math.Pow10 (math/pow10.go): the exact table-product construction the
 * reference multiplies/divides by — libm pow can differ by 1 ulp at
 * |n| >= 23. */
static double go_pow10(int n) {
  static const double tab[32] = {
      1e0,  1e1,  1e2,  1e3,  1e4,  1e5,  1e6,  1e7,  1e8,  1e9,  1e10,
      1e11, 1e12, 1e13, 1e14, 1e15, 1e16, 1e17, 1e18, 1e19, 1e20, 1e21,
      1e22, 1e23, 1e24, 1e25, 1e26, 1e27, 1e28, 1e29, 1e30, 1e31};
  static const double postab32[10] = {1e0,   1e32,  1e64,  1e96,  1e128,
                                      1e160, 1e192, 1e224, 1e256, 1e288};
  static const double negtab32[11] = {1e0,    1e-32,  1e-64,  1e-96,
                                      1e-128, 1e-160, 1e-192, 1e-224,
                                      1e-256, 1e-288, 1e-320};
  if (0 <= n && n <= 308) return postab32[n / 32] * tab[n % 32];
  if (-323 <= n && n < 0) return negtab32[(-n) / 32] / tab[(-n) % 32];
  if (n > 308) return INFINITY;
  return 0.0;
}

double vm_decimal_round_to_decimal_digits(double f, int digits) {
  uint64_t bits;
  memcpy(&bits, &f, 8);
  if (bits == STALE_NAN_BITS) return f;
  if (digits <= -100 || digits >= 100) return f;
  double m = go_pow10(digits);
  return round(f * m) / m;
}

/* RoundToSignificantFigures (decimal.go:341-370) */
double vm_decimal_round_to_significant_figures(double f, int digits) {
  uint64_t bits;
  memcpy(&bits, &f, 8);
  if (bits == STALE_NAN_BITS) return f;
  if (digits <= 0 || digits >= 18) return f;
  if (isnan(f) || isinf(f) || f == 0) return f;
  int64_t nlim = (int64_t)pow(10.0, (double)digits);
  int is_negative = f < 0;
  if (is_negative) f = -f;
  int64_t v;
  int16_t e;
  vm_decimal_positive_float_to_decimal(f, &v, &e);
  if (v > V_MAX) v = V_MAX;
  int64_t rem = 0;
  while (v > nlim) {
    rem = v % 10;
    v /= 10;
    e++;
  }
  if (rem >= 5) v++;
  if (is_negative) v = -v;
  return vm_decimal_to_float(v, e);
}

/* CalibrateScale (decimal.go:12-70) */
int16_t vm_decimal_calibrate_scale(int64_t* a, int64_t na, int16_t ae,
                                   int64_t* b, int64_t nb, int16_t be) {
  if (ae == be) return ae;
  if (na == 0) return be;
  if (nb == 0) return ae;
  if (ae < be) {
    int64_t* t = a; a = b; b = t;
    int64_t tn = na; na = nb; nb = tn;
    int16_t te = ae; ae = be; be = te;
  }
  int16_t up_exp = (int16_t)(ae - be);
  int16_t down_exp = 0;
  for (int64_t i = 0; i < na; i++) {
    int16_t max_up = vm_decimal_max_up_exponent(a[i]);
    if ((int16_t)(up_exp - max_up) > down_exp) down_exp = (int16_t)(up_exp - max_up);
  }
  up_exp = (int16_t)(up_exp - down_exp);
  static const int64_t pows[19] = {
      1LL, 10LL, 100LL, 1000LL, 10000LL, 100000LL, 1000000LL, 10000000LL,
      100000000LL, 1000000000LL, 10000000000LL, 100000000000LL,
      1000000000000LL, 10000000000000LL, 100000000000000LL,
      1000000000000000LL, 10000000000000000LL, 100000000000000000LL,
      1000000000000000000LL};
  if (up_exp > 0) {
    int64_t m = (up_exp < 19) ? pows[up_exp] : 1;
    for (int64_t i = 0; i < na; i++) {
      if (vm_decimal_is_special(a[i])) continue;
      a[i] *= m;
    }
  }
  if (down_exp > 0) {
    if (down_exp > 18) {
      for (int64_t i = 0; i < nb; i++) {
        if (vm_decimal_is_special(b[i])) continue;
        b[i] = 0;
      }
    } else {
      int64_t m = pows[down_exp];
      for (int64_t i = 0; i < nb; i++) {
        if (vm_decimal_is_special(b[i])) continue;
        b[i] /= m;
      }
    }
  }
  return (int16_t)(be + down_exp);
}
