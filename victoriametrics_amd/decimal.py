"""Product host mirror of the lib/decimal pieces used at the Exec surface:
`RoundToDecimalDigits` (decimal.go:325-335) is applied to every output
value after download when the request's round_digits < 100
(exec.go:93-101, prometheus.go:1133).  Kept OUT of the kernels so it never
affects kernel parity; the full decimal codec restatement lives in
oracle/decimal.c (test infrastructure) — this module is the small product
subset."""
import math

import numpy as np

STALE_NAN_BITS = np.uint64(0x7FF0000000000002)

_SIGN = np.uint64(1 << 63)
_FRAC = np.uint64((1 << 52) - 1)
_ONE_BITS = np.uint64(0x3FF0000000000000)


def go_round(values):
    """math.Round (round half away from zero) — bit-exact vectorized port
    of Go's bit-twiddling implementation (math/floor.go), correct on the
    0.49999999999999994 edge where trunc(x + copysign(0.5, x)) is not."""
    x = np.ascontiguousarray(values, np.float64)
    bits = x.view(np.uint64).copy()
    e = (bits >> np.uint64(52)) & np.uint64(0x7FF)
    out = bits.copy()
    small = e < 1023
    out[small] = bits[small] & _SIGN
    bump = small & (e == 1022)
    out[bump] |= _ONE_BITS
    mid = (e >= 1023) & (e < 1075)
    em = (e[mid] - np.uint64(1023))
    half = np.uint64(1 << 51) >> em
    b = bits[mid] + half
    b &= ~(_FRAC >> em)
    out[mid] = b
    return out.view(np.float64)


# Go math.Pow10 (math/pow10.go): exact table-product construction — can
# differ from libm pow(10, n) by 1 ulp at extreme n, and the reference's
# RoundToDecimalDigits divides by exactly this value.
_POW10TAB = [float(f"1e{i}") for i in range(32)]
_POW10POSTAB32 = [float(f"1e{32 * i}") for i in range(10)]
_POW10NEGTAB32 = [float(f"1e-{32 * i}") for i in range(11)]


def go_pow10(n):
    n = int(n)
    if 0 <= n <= 308:
        return _POW10POSTAB32[n // 32] * _POW10TAB[n % 32]
    if -323 <= n < 0:
        return _POW10NEGTAB32[(-n) // 32] / _POW10TAB[(-n) % 32]
    return math.inf if n > 308 else 0.0


def round_to_decimal_digits(values, digits):
    """RoundToDecimalDigits (decimal.go:325-335), elementwise over an
    array; stale-NaN marks pass through untouched.  The 10^digits factor
    uses the Go Pow10 table product, not libm pow (1-ulp parity at
    |digits| >= 23)."""
    v = np.ascontiguousarray(values, np.float64)
    if digits <= -100 or digits >= 100:
        return v
    stale = v.view(np.uint64) == STALE_NAN_BITS
    m = go_pow10(digits)
    with np.errstate(invalid="ignore", over="ignore"):
        r = go_round(v * m) / m
    return np.where(stale, v, r)
