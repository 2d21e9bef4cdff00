"""Multi-series rollup functions (rollup.go:1490-1560).

`count_values_over_time` and `histogram_over_time` emit a DATA-DEPENDENT
number of output series per input series — one per observed formatted value
/ non-zero vmrange bucket — via the timeseriesMap side channel
(newTimeseriesMap, rollup.go:629; DoTimeseriesMap, rollup.go:700).  They do
not fit the one-value-per-(series, grid point) kernel contract, so they run
host-side exactly where the reference computes them, over the decoded
columns; the per-point windows are the doInternal windows
(seekFirstTimestampIdxAfter semantics) for an explicit lookbehind window.

Neither function is in rollupFuncsCanAdjustWindow, so the window is the
range selector's literal duration (required: both take m[d]).  Neither is
in rollupFuncsKeepMetricName, so the output metric group resets unless the
query carries keep_metric_names.  Stale NaNs are dropped (neither is in the
keep set, eval.go:2108).
"""
import math

import numpy as np

from .aggregate import (_H_LOWER, _H_UPPER, _histogram_ranges,
                        _histogram_update)
from .binary_op import Series
from .decimal import STALE_NAN_BITS


def format_go_float_g(v):
    """strconv.FormatFloat(v, 'g', -1, 64): shortest digits; scientific
    form when the decimal exponent is < -4 or >= 6 (strconv/ftoa.go:
    `if shortest { eprec = 6 }`), with a 2+-digit exponent."""
    if v != v:
        return "NaN"
    if math.isinf(v):
        return "+Inf" if v > 0 else "-Inf"
    if v == 0:
        return "-0" if math.copysign(1.0, v) < 0 else "0"
    sci = np.format_float_scientific(v, unique=True, trim="-")
    mant, _, es = sci.partition("e")
    exp = int(es)
    neg = mant.startswith("-")
    digits = mant.lstrip("-").replace(".", "").rstrip("0") or "0"
    if exp < -4 or exp >= 6:
        m = digits[0] + ("." + digits[1:] if len(digits) > 1 else "")
        return f"{'-' if neg else ''}{m}e{'+' if exp >= 0 else '-'}{abs(exp):02d}"
    # positional form
    return np.format_float_positional(v, unique=True, trim="-")


def _drop_stale(ts_col, vals_col):
    bits = np.asarray(vals_col, dtype=np.float64).view(np.uint64)
    keep = bits != STALE_NAN_BITS
    if keep.all():
        return ts_col, vals_col
    return ts_col[keep], vals_col[keep]


def _windows(ts_col, start, end, step, window):
    """Per-grid-point [i, j) half-open sample ranges: i/j = first index with
    ts > (t_end - window) / t_end (seekFirstTimestampIdxAfter,
    rollup.go:825)."""
    if window <= 0:
        raise ValueError("multi-series rollups need an explicit window "
                         "(m[d] range selector)")
    n_grid = 1 + (int(end) - int(start)) // int(step)
    t_end = np.asarray(start, dtype=np.int64) + \
        np.arange(n_grid, dtype=np.int64) * int(step)
    j = np.searchsorted(ts_col, t_end, side="right")
    i = np.searchsorted(ts_col, t_end - int(window), side="right")
    return i, j, n_grid


def _base_mn(mn, keep_metric_names):
    out = mn.copy()
    if not keep_metric_names:
        out.reset_metric_group()
    return out


def count_values_over_time(label_name, ts_col, vals_col, mn, start, end,
                           step, window, keep_metric_names=False,
                           drop_stale=True):
    """newRollupCountValues (rollup.go:1490): per grid point, count each
    distinct window value, one output series per distinct
    strconv 'g'-formatted value under label_name.  Returns
    (list[Series], samples_scanned)."""
    ts_col = np.ascontiguousarray(ts_col, dtype=np.int64)
    vals_col = np.ascontiguousarray(vals_col, dtype=np.float64)
    if drop_stale:
        ts_col, vals_col = _drop_stale(ts_col, vals_col)
    i, j, n_grid = _windows(ts_col, start, end, step, window)
    base = _base_mn(mn, keep_metric_names)
    out = {}
    order = []
    scanned = 0
    for g in range(n_grid):
        lo, hi = int(i[g]), int(j[g])
        scanned += hi - lo
        for v in vals_col[lo:hi]:
            key = format_go_float_g(float(v))
            s = out.get(key)
            if s is None:
                m2 = base.copy()
                m2.remove_tag(label_name)
                m2.add_tag(label_name, key)
                s = Series(m2, np.full(n_grid, math.nan))
                out[key] = s
                order.append(key)
            cur = s.values[g]
            s.values[g] = 1.0 if math.isnan(cur) else cur + 1.0
    return [out[k] for k in order], scanned


def histogram_over_time(ts_col, vals_col, mn, start, end, step, window,
                        keep_metric_names=False, drop_stale=True):
    """rollupHistogram (rollup.go:1525): per grid point, a VictoriaMetrics
    metrics.Histogram over the window values; each non-zero vmrange bucket
    becomes an output series labeled vmrange=<range> with the count at that
    point (NaN elsewhere — the timeseriesMap origin is NaN-filled).
    Returns (list[Series], samples_scanned)."""
    ts_col = np.ascontiguousarray(ts_col, dtype=np.int64)
    vals_col = np.ascontiguousarray(vals_col, dtype=np.float64)
    if drop_stale:
        ts_col, vals_col = _drop_stale(ts_col, vals_col)
    i, j, n_grid = _windows(ts_col, start, end, step, window)
    base = _base_mn(mn, keep_metric_names)
    ranges = _histogram_ranges()
    out = {}
    order = []
    scanned = 0
    for g in range(n_grid):
        lo, hi = int(i[g]), int(j[g])
        scanned += hi - lo
        buckets = {}
        for v in vals_col[lo:hi]:
            _histogram_update(buckets, float(v))
        for key, count in buckets.items():
            if count == 0:
                continue  # VisitNonZeroBuckets
            if key == "lower":
                vmrange = _H_LOWER
            elif key == "upper":
                vmrange = _H_UPPER
            else:
                vmrange = ranges[key]
            s = out.get(vmrange)
            if s is None:
                m2 = base.copy()
                m2.remove_tag("vmrange")
                m2.add_tag("vmrange", vmrange)
                s = Series(m2, np.full(n_grid, math.nan))
                out[vmrange] = s
                order.append(vmrange)
            s.values[g] = float(count)
    return [out[k] for k in order], scanned
