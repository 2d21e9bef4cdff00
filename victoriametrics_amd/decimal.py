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


def round_to_decimal_digits(values, digits):
    """RoundToDecimalDigits (decimal.go:325-335), elementwise over an
    array; stale-NaN marks pass through untouched."""
    v = np.ascontiguousarray(values, np.float64)
    if digits <= -100 or digits >= 100:
        return v
    stale = v.view(np.uint64) == STALE_NAN_BITS
    m = math.pow(10.0, digits)
    with np.errstate(invalid="ignore", over="ignore"):
        r = go_round(v * m) / m
    return np.where(stale, v, r)
