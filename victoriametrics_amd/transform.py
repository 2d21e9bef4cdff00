"""Transform-function layer — host mirror of transform.go's dispatch.

Value math runs on the GPU (vmgpu_transform, csrc/transform.hip); label
functions (label_*, sort_by_label, drop_common_labels, limit_offset) are
host metadata work mirrored here on MetricName objects, exactly as the
reference keeps them on the Go side of the C-ABI seam.

Value funcs operate on a [n_series x n_grid] matrix in place and take
scalar args as per-grid rows (getScalar semantics, eval.go): `transform(
"clamp", values, args=[min_row, max_row])`.
"""
import ctypes
import math
import re

import numpy as np

from . import engine
from .metric_name import MetricName

# func name -> (device id, n scalar-arg rows, scalar_arg index or None)
_ELEMENTWISE = {
    "abs": 0, "ceil": 1, "floor": 2, "exp": 3, "ln": 4, "log2": 5,
    "log10": 6, "sqrt": 7, "sin": 8, "cos": 9, "tan": 10, "asin": 11,
    "acos": 12, "atan": 13, "sinh": 14, "cosh": 15, "tanh": 16,
    "asinh": 17, "acosh": 18, "atanh": 19, "deg": 20, "rad": 21, "sgn": 22,
}
_CLAMP = {"clamp": 23, "clamp_min": 24, "clamp_max": 25}
_ROUND = 26
_BITMAP = {"bitmap_and": 27, "bitmap_or": 28, "bitmap_xor": 29}
_DATETIME = {
    "day_of_month": 30, "day_of_week": 31, "day_of_year": 32,
    "days_in_month": 33, "hour": 34, "minute": 35, "month": 36, "year": 37,
}
_SERIES = {
    "keep_last_value": 100, "keep_next_value": 101, "interpolate": 102,
    "running_sum": 103, "running_min": 104, "running_max": 105,
    "running_avg": 106, "range_sum": 107, "range_min": 108,
    "range_max": 109, "range_avg": 110, "range_first": 111,
    "range_last": 112, "range_normalize": 113, "range_zscore": 114,
    "range_trim_zscore": 115, "range_stddev": 116, "range_stdvar": 117,
    "range_linear_regression": 118, "range_mad": 119,
    "range_trim_outliers": 120, "range_trim_spikes": 121,
    "range_quantile": 122, "smooth_exponential": 123, "remove_resets": 124,
}
# funcs whose MetricName group must be reset (transform.go:142
# transformFuncsKeepMetricName lists the keepers; everything else resets)
KEEP_METRIC_NAME_FUNCS = {
    "ceil", "clamp", "clamp_min", "clamp_max", "floor", "interpolate",
    "keep_last_value", "keep_next_value", "range_avg", "range_first",
    "range_last", "range_linear_regression", "range_max", "range_median",
    "range_min", "range_normalize", "range_quantile", "range_stddev",
    "range_stdvar", "range_sum", "range_trim_outliers", "range_trim_spikes",
    "range_trim_zscore", "range_zscore", "round", "running_avg",
    "running_max", "running_min", "running_sum", "smooth_exponential",
}


def decimal_from_float_exponent(f):
    """FromFloat's decimal exponent (lib/decimal/decimal.go:437 +
    positiveFloatToDecimal :467) — the host-side piece transformRound
    needs for its p10 (transform.go:2357-2361)."""
    if f == 0 or math.isnan(f) or math.isinf(f):
        return 0
    f = abs(f)
    u = int(f)
    if float(u) == f:
        if u < (1 << 55) and u % 10 != 0:
            return 0
        # getDecimalAndScale (decimal.go:480)
        scale = 0
        v = u
        if v < (1 << 55) and v % 10 != 0:
            return scale
        while v >= (1 << 55):
            v //= 10
            scale += 1
        if v % 10 != 0:
            return scale
        v //= 10
        scale += 1
        while v != 0 and v % 10 == 0:
            v //= 10
            scale += 1
        return scale
    # positiveFloatToDecimalSlow (decimal.go:502)
    scale = 0
    prec = 1e12
    if f > 1e6 or f < 1e-6:
        if f > 1e6:
            prec = 1e15
        _, exp = math.frexp(f)
        exp = max(-1022, min(1023, exp))
        scale = int(exp * (math.log(2) / math.log(10)))
        f *= math.pow(10.0, -scale)
    while f < prec:
        frac, x = math.modf(f)
        if frac * prec < x:
            f = x
            break
        if (1 - frac) * prec < x:
            f = x + 1
            break
        f *= 100
        scale -= 2
    u = int(f)
    if u % 10 != 0:
        return scale
    return scale + 1


def _call(func_id, values, ts=None, arg1=None, arg2=None, scalar_arg=0.0,
          want_keep=False):
    engine.init()
    lib = engine._load_lib()
    v = np.ascontiguousarray(values, dtype=np.float64)
    n_series, n_grid = v.shape
    a1 = np.ascontiguousarray(arg1, dtype=np.float64) if arg1 is not None else None
    a2 = np.ascontiguousarray(arg2, dtype=np.float64) if arg2 is not None else None
    t = np.ascontiguousarray(ts, dtype=np.int64) if ts is not None else None
    keep = np.ones(n_series, dtype=np.uint8) if want_keep else None
    errbuf = ctypes.create_string_buffer(256)
    dp = ctypes.POINTER(ctypes.c_double)
    rc = lib.vmgpu_transform(
        ctypes.c_int32(func_id), v.ctypes.data_as(dp),
        ctypes.c_uint32(n_series), ctypes.c_uint32(n_grid),
        t.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)) if t is not None else None,
        a1.ctypes.data_as(dp) if a1 is not None else None,
        a2.ctypes.data_as(dp) if a2 is not None else None,
        ctypes.c_double(scalar_arg),
        keep.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)) if keep is not None else None,
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise engine.VmGpuError(f"vmgpu_transform failed ({rc}): "
                                f"{errbuf.value.decode()}")
    values[:] = v
    return (values, keep) if want_keep else values


def transform(name, values, ts=None, args=(), scalar=0.0):
    """Apply one transform func to the [n_series x n_grid] matrix in place.

    args: scalar-arg rows [n_grid] in the reference's argument order
    (clamp: min,max; round: nearest; bitmap_*: mask; smooth_exponential:
    smoothing factors).  scalar: phi/k/z for the range_trim/quantile funcs.
    Returns values (and for range_normalize a keep mask)."""
    name = name.lower()
    if name in _ELEMENTWISE:
        return _call(_ELEMENTWISE[name], values)
    if name in _CLAMP:
        return _call(_CLAMP[name], values,
                     arg1=args[0] if len(args) > 0 else None,
                     arg2=args[1] if len(args) > 1 else None)
    if name == "round":
        nearest = np.ascontiguousarray(
            args[0] if args else np.ones(values.shape[1]), dtype=np.float64)
        # p10 per point, hoisting the reference's nPrev cache host-side
        p10 = np.empty_like(nearest)
        prev_n, prev_p = None, 1.0
        for i, nv in enumerate(nearest):
            if nv != prev_n:
                prev_n = nv
                # p10 = Pow10(-e) of FromFloat(nearest) (transform.go:2360)
                prev_p = math.pow(10.0, -decimal_from_float_exponent(nv))
            p10[i] = prev_p
        return _call(_ROUND, values, arg1=nearest, arg2=p10)
    if name in _BITMAP:
        return _call(_BITMAP[name], values, arg1=args[0])
    if name in _DATETIME:
        return _call(_DATETIME[name], values)
    if name in _SERIES:
        fid = _SERIES[name]
        if name == "range_normalize":
            return _call(fid, values, want_keep=True)
        if name == "smooth_exponential":
            return _call(fid, values, arg1=args[0] if args else None)
        if name == "range_linear_regression":
            return _call(fid, values, ts=ts)
        return _call(fid, values, scalar_arg=scalar)
    raise ValueError(f"unknown transform func {name!r}")


# ---------------------------------------------------------------------------
# label funcs — host metadata work (transform.go label_* family)
# ---------------------------------------------------------------------------

def reset_metric_group_if_needed(name, series_list):
    """doTransformValues resets MetricGroup unless the func keeps the
    physical meaning (transform.go:142-209) or keep_metric_names is set."""
    if name.lower() in KEEP_METRIC_NAME_FUNCS:
        return
    for s in series_list:
        s.mn.reset_metric_group()


def label_set(series_list, pairs):
    # transformLabelSet (transform.go): (label, value) literal pairs
    for s in series_list:
        for k, v in pairs:
            if k == "__name__":
                s.mn.metric_group = MetricName._b(v)
            elif v == "":
                s.mn.remove_tag(k)
            else:
                s.mn.set_tag(k, v)
    return series_list


def label_del(series_list, labels):
    for s in series_list:
        for k in labels:
            s.mn.remove_tag(k)
    return series_list


def label_keep(series_list, labels):
    keep = {MetricName._b(k) for k in labels}
    for s in series_list:
        if b"__name__" not in keep:
            s.mn.reset_metric_group()
        s.mn.tags = [(k, v) for k, v in s.mn.tags if k in keep]
    return series_list


def label_copy(series_list, pairs, remove_src=False):
    # transformLabelCopy / transformLabelMove (src, dst pairs)
    for s in series_list:
        for src, dst in pairs:
            v = s.mn.get_tag_value(src)
            if v is None or len(v) == 0:
                continue
            if dst == "__name__":
                s.mn.metric_group = v
            else:
                s.mn.set_tag(dst, v)
            if remove_src and src != dst:
                s.mn.remove_tag(src)
    return series_list


def label_move(series_list, pairs):
    return label_copy(series_list, pairs, remove_src=True)


def label_join(series_list, dst_label, separator, src_labels):
    # transformLabelJoin
    sep = MetricName._b(separator)
    for s in series_list:
        parts = []
        for k in src_labels:
            v = s.mn.get_tag_value(k)
            parts.append(v if v is not None else b"")
        joined = sep.join(parts)
        if joined == b"":
            # empty result removes the dst label (transform.go:2059)
            s.mn.remove_tag(dst_label)
        elif dst_label == "__name__":
            s.mn.metric_group = joined
        else:
            s.mn.set_tag(dst_label, joined)
    return series_list


def label_replace(series_list, dst_label, replacement, src_label, regex):
    # transformLabelReplace: RE2 full-anchored match semantics
    pat = re.compile("^(?:" + regex + ")$")
    for s in series_list:
        v = s.mn.get_tag_value(src_label)
        src = (v or b"").decode("utf-8", "surrogateescape")
        m = pat.match(src)
        if m is None:
            continue
        val = _go_expand(m, replacement)
        if dst_label == "__name__":
            s.mn.metric_group = val.encode()
        elif val == "":
            s.mn.remove_tag(dst_label)
        else:
            s.mn.set_tag(dst_label, val)
    return series_list


def label_uppercase(series_list, labels):
    for s in series_list:
        for k in labels:
            v = s.mn.get_tag_value(k)
            if v is not None:
                target_up = v.decode("utf-8", "surrogateescape").upper()
                if k == "__name__":
                    s.mn.metric_group = target_up.encode()
                else:
                    s.mn.set_tag(k, target_up)
    return series_list


def label_lowercase(series_list, labels):
    for s in series_list:
        for k in labels:
            v = s.mn.get_tag_value(k)
            if v is not None:
                low = v.decode("utf-8", "surrogateescape").lower()
                if k == "__name__":
                    s.mn.metric_group = low.encode()
                else:
                    s.mn.set_tag(k, low)
    return series_list


def label_match(series_list, label, regex, negate=False):
    # transformLabelMatch / transformLabelMismatch
    pat = re.compile("^(?:" + regex + ")$")
    out = []
    for s in series_list:
        v = (s.mn.get_tag_value(label) or b"").decode("utf-8",
                                                      "surrogateescape")
        hit = pat.match(v) is not None
        if hit != negate:
            out.append(s)
    return out


def drop_common_labels(series_list):
    # transformDropCommonLabels: remove (k,v) present on every series;
    # __name__ participates as a label
    if not series_list:
        return series_list
    from collections import Counter
    counts = Counter()
    for s in series_list:
        seen = {(b"__name__", s.mn.metric_group)} if s.mn.metric_group else set()
        for kv in s.mn.tags:
            seen.add(kv)
        for kv in seen:
            counts[kv] += 1
    n = len(series_list)
    common = {kv for kv, c in counts.items() if c == n}
    for s in series_list:
        if (b"__name__", s.mn.metric_group) in common:
            s.mn.reset_metric_group()
        s.mn.tags = [kv for kv in s.mn.tags if kv not in common]
    return series_list


def drop_empty_series(series_list):
    return [s for s in series_list if not np.all(np.isnan(s.values))]


def limit_offset(series_list, limit, offset):
    # transformLimitOffset (transform.go:2293): empty series are filtered
    # out BEFORE the offset is applied
    from .binary_op import remove_empty_series
    series_list = remove_empty_series(series_list)
    return series_list[offset:offset + limit]


def sort_by_label(series_list, labels, desc=False):
    # newTransformFuncSortByLabel: stable multi-key sort by tag values
    def key(s):
        return tuple((s.mn.get_tag_value(k) or b"") for k in labels)
    return sorted(series_list, key=key, reverse=desc)


def sort_series(series_list, desc=False):
    """sort/sort_desc (newTransformFuncSort, transform.go:2580): compare
    POSITIONALLY from the last grid index backwards — at each index a NaN
    on one side (only) makes that series sort first regardless of desc;
    equal-or-both-NaN advances to the previous index; the first differing
    pair of values decides (flipped for desc)."""
    import functools

    def cmp(sa, sb):
        a, b = sa.values, sb.values
        n = len(a) - 1
        while n >= 0:
            if not math.isnan(a[n]):
                if math.isnan(b[n]):
                    return 1  # b has NaN here -> b sorts first
                if a[n] != b[n]:
                    break
            elif not math.isnan(b[n]):
                return -1     # a has NaN here -> a sorts first
            n -= 1
        if n < 0:
            return 0
        if desc:
            return -1 if b[n] < a[n] else 1
        return -1 if a[n] < b[n] else 1

    return sorted(series_list, key=functools.cmp_to_key(cmp))


# ---------------------------------------------------------------------------
# histogram transforms (transform.go:512 vmrangeBucketsToLE + :992-1190)
# ---------------------------------------------------------------------------

def _is_zero_series(s):
    return not np.any(s.values > 0)


def _merge_non_overlapping(dst, src):
    from .binary_op import merge_non_overlapping
    return merge_non_overlapping(dst, src)


def vmrange_buckets_to_le(series_list):
    """vmrangeBucketsToLE (transform.go:512): convert VictoriaMetrics
    `vmrange=start...end` buckets into cumulative Prometheus `le` buckets;
    Prometheus-style `le` series pass through."""
    rvs = []
    m = {}
    for s in series_list:
        vmrange = s.mn.get_tag_value("vmrange")
        if not vmrange:
            if s.mn.get_tag_value("le"):
                rvs.append(s)
            continue
        txt = vmrange.decode("utf-8", "surrogateescape")
        if "..." not in txt:
            continue
        start_str, _, end_str = txt.partition("...")
        try:
            start = float(start_str)
            end = float(end_str)
        except ValueError:
            continue
        s.mn.remove_tag("le")
        s.mn.remove_tag("vmrange")
        k = s.mn.marshal_sorted()
        m.setdefault(k, []).append((start_str, end_str, start, end, s))

    for xss in m.values():
        xss.sort(key=lambda x: x[3])
        xss_new = []
        prev = None
        uniq = {}

        def copy_ts(src, le_str):
            ts = src.copy_shallow()
            ts.values[:] = 0.0
            ts.mn.remove_tag("le")
            ts.mn.add_tag("le", le_str)
            return ts

        prev_end = 0.0  # Go zero-value xsPrev.end: a first bucket starting
                        # at 0 produces no gap series (transform_test.go:89)
        for start_str, end_str, start, end, s in xss:
            if _is_zero_series(s):
                continue
            if start != prev_end:
                if uniq.get(start_str) is None:
                    uniq[start_str] = s
                    xss_new.append((None, start_str, None, start,
                                    copy_ts(s, start_str)))
            s.mn.add_tag("le", end_str)
            prev_ts = uniq.get(end_str)
            if prev_ts is not None:
                _merge_non_overlapping(prev_ts, s)
            else:
                xss_new.append((start_str, end_str, start, end, s))
                uniq[end_str] = s
            prev = (start_str, end_str, start, end, s)
            prev_end = end
        if prev is not None and not math.isinf(prev[3]) \
                and not _is_zero_series(prev[4]):
            xss_new.append((None, "+Inf", None, math.inf,
                            copy_ts(prev[4], "+Inf")))
        if not xss_new:
            continue
        # cumulative counts per grid point
        n_grid = len(xss_new[0][4].values)
        count = np.zeros(n_grid)
        for _, _, _, _, s in xss_new:
            v = s.values
            pos = ~np.isnan(v) & (v > 0)
            count = count + np.where(pos, v, 0.0)
            s.values = count.copy()
        rvs.extend(s for _, _, _, _, s in xss_new)
    return rvs


def group_le_timeseries(series_list):
    """groupLeTimeseries (transform.go): key = sorted tags minus le, with
    the metric group reset; value = [(le, Series)]."""
    m = {}
    for s in series_list:
        lev = s.mn.get_tag_value("le")
        if not lev:
            continue
        try:
            le = float(lev.decode("utf-8", "surrogateescape"))
        except ValueError:
            continue
        s.mn.reset_metric_group()
        s.mn.remove_tag("le")
        m.setdefault(s.mn.marshal_sorted(), []).append((le, s))
    return m


def _merge_same_le(xss):
    # mergeSameLE (transform.go): sum values of equal-le buckets
    dst = [xss[0]]
    for le, s in xss[1:]:
        if le != dst[-1][0]:
            dst.append((le, s))
        else:
            dst[-1][1].values = dst[-1][1].values + s.values
    return dst


def histogram_transform(name, series_list, arg=None, bounds_label=None):
    """histogram_quantile / histogram_avg / histogram_stddev /
    histogram_stdvar / histogram_share over raw bucket series (vmrange or
    le).  arg: phi (quantile) or le (share); for histogram_fraction the
    (lower, upper) pair — transformHistogramFraction (transform.go:751-828)
    is share(upper) - share(lower) on the same fixed buckets, evaluated
    here as two hshare kernel passes (fixBrokenBuckets is deterministic,
    so both passes see identical fixed values).  Returns result Series
    list (+ lower/upper series when bounds_label is set)."""
    from . import engine
    from .binary_op import Series
    if name == "histogram_fraction":
        lo_req = np.atleast_1d(np.asarray(arg[0], np.float64))
        hi_req = np.atleast_1d(np.asarray(arg[1], np.float64))
        if lo_req[0] >= hi_req[0]:
            raise ValueError(
                "lower le cannot be greater than upper le; got lower le: "
                "%f, upper le: %f" % (lo_req[0], hi_req[0]))
    tss = vmrange_buckets_to_le(series_list)
    m = group_le_timeseries(tss)
    if not m:
        return []
    group_keys, rows, les, goff = [], [], [], [0]
    dsts = []
    for k, xss in m.items():
        xss.sort(key=lambda x: x[0])
        if name in ("histogram_quantile", "histogram_share",
                    "histogram_fraction"):
            xss = _merge_same_le(xss)
        for le, s in xss:
            rows.append(s.values)
            les.append(le)
        goff.append(len(rows))
        dsts.append(xss[0][1])
        group_keys.append(k)
    bv = np.stack(rows)
    les = np.asarray(les)
    goff = np.asarray(goff, np.uint64)
    n_grid = bv.shape[1]
    lo = hi = None
    if name == "histogram_quantile":
        out, lo, hi = engine.histogram_quantile(
            float(arg), bv, les, goff, bounds=bounds_label is not None)
    elif name == "histogram_share":
        req = np.full(n_grid, float(arg))
        out, lo, hi = engine.histogram_share(
            req, bv, les, goff, bounds=bounds_label is not None)
    elif name == "histogram_fraction":
        def _full(r):
            if r.size == 1:
                return np.full(n_grid, r[0])
            if r.size != n_grid:
                raise ValueError("le series length mismatch")
            return np.ascontiguousarray(r)
        out_hi, _, _ = engine.histogram_share(_full(hi_req), bv, les, goff)
        out_lo, _, _ = engine.histogram_share(_full(lo_req), bv, les, goff)
        out = out_hi - out_lo
    else:
        mode = {"histogram_avg": 0, "histogram_stddev": 1,
                "histogram_stdvar": 2}[name]
        out = engine.histogram_stat(mode, bv, les, goff)
    rvs = []
    for gi, dst in enumerate(dsts):
        if bounds_label is not None and lo is not None:
            sl = Series(dst.mn.copy(), lo[gi])
            sl.mn.remove_tag(bounds_label)
            sl.mn.add_tag(bounds_label, "lower")
            su = Series(dst.mn.copy(), hi[gi])
            su.mn.remove_tag(bounds_label)
            su.mn.add_tag(bounds_label, "upper")
            rvs.append(sl)
            rvs.append(su)
        dst.values = out[gi]
        rvs.append(dst)
    return rvs


def absent(series_list, n_grid, base_mn=None):
    """transformAbsent (transform.go:215): a single series of 1s where NO
    input series has a value.  base_mn: the MetricName to copy constant
    selector tags from (getAbsentTimeseries — the caller extracts them from
    the AST's MetricExpr)."""
    from .binary_op import Series
    mn = base_mn.copy() if base_mn is not None else MetricName()
    vals = np.ones(n_grid)
    if series_list:
        present = np.zeros(n_grid, dtype=bool)
        for s in series_list:
            present |= ~np.isnan(s.values)
        vals[present] = math.nan
    return [Series(mn, vals)]


def union(args):
    """transformUnion (transform.go:1743): concatenate series lists,
    dropping later series whose metric name was already seen.  Special
    case (transform.go:1749): when EVERY arg is a single scalar series,
    all of them are returned without dedup — this feeds the
    `q == (v1,...,vN)` list comparisons."""
    from .binary_op import is_scalar
    if args and all(is_scalar(arg) for arg in args):
        return [arg[0] for arg in args]
    rvs, seen = [], set()
    for arg in args:
        for s in arg:
            k = s.mn.marshal_sorted()
            if k in seen:
                continue
            seen.add(k)
            rvs.append(s)
    return rvs


def prometheus_buckets(series_list):
    """transformPrometheusBuckets = vmrangeBucketsToLE."""
    return vmrange_buckets_to_le(series_list)


def buckets_limit(limit, series_list):
    """transformBucketsLimit (transform.go): merge lowest-hit buckets until
    each le-group holds at most `limit` buckets (min 3), trimming
    consecutively-empty edge buckets first."""
    limit = int(limit)
    if limit <= 0:
        raise ValueError("limit must be greater than 0")
    if limit < 3:
        limit = 3
    tss = vmrange_buckets_to_le(series_list)
    if not tss:
        return []
    m = {}
    for s in tss:
        le_str = s.mn.get_tag_value("le")
        if not le_str:
            continue
        try:
            le = float(le_str.decode("utf-8", "surrogateescape"))
        except ValueError:
            continue
        mn = s.mn.copy()
        mn.remove_tag("le")
        m.setdefault(mn.marshal_sorted(), []).append([le, 0.0, s])
    rvs = []
    for le_group in m.values():
        if len(le_group) <= limit:
            rvs.extend(x[2] for x in le_group)
            continue
        le_group.sort(key=lambda x: x[0])
        n_pts = len(le_group[0][2].values)
        for n in range(n_pts):
            prev = 0.0
            for x in le_group:
                v = x[2].values[n]
                x[1] += v - prev
                prev = v
        eps = 1e-9

        def empty(h):
            return not math.isnan(h) and abs(h) < eps

        l, r = 0, len(le_group) - 1
        while r - l + 1 > limit and empty(le_group[r][1]):
            r -= 1
        while r - l + 1 > limit and empty(le_group[l][1]):
            l += 1
        le_group = le_group[l:r + 1]
        while len(le_group) > limit:
            min_idx = 1
            min_hits = le_group[1][1] + le_group[2][1]
            for i in range(len(le_group) - 3):
                h = le_group[i + 1][1] + le_group[i + 2][1]
                if h < min_hits:
                    min_idx = i + 1
                    min_hits = h
            le_group[min_idx + 1][1] += le_group[min_idx][1]
            del le_group[min_idx]
        rvs.extend(x[2] for x in le_group)
    return rvs


def _go_expand(m, template):
    """Go regexp.Expand template semantics (regexp/regexp.go): `$name`
    takes the LONGEST run of [A-Za-z0-9_] (so `$1y` means group "1y", not
    group 1 followed by "y"), `${name}` is explicit, `$$` is a literal $,
    and a reference to an absent or unmatched group expands to "" instead
    of raising — Python's Match.expand errors on all of these."""
    out = []
    i, n = 0, len(template)
    while i < n:
        c = template[i]
        if c != "$":
            out.append(c)
            i += 1
            continue
        if i + 1 >= n:
            out.append("$")
            break
        nxt = template[i + 1]
        if nxt == "$":
            out.append("$")
            i += 2
            continue
        if nxt == "{":
            j = template.find("}", i + 2)
            if j < 0:
                out.append(template[i:])
                break
            name = template[i + 2:j]
            i = j + 1
        else:
            j = i + 1
            while j < n and (template[j].isalnum() or template[j] == "_"):
                j += 1
            name = template[i + 1:j]
            if not name:
                out.append("$")
                i += 1
                continue
            i = j
        val = ""
        if name.isdigit():
            gi = int(name)
            if 0 <= gi <= m.re.groups:
                val = m.group(gi) or ""
        elif name in m.re.groupindex:
            val = m.group(name) or ""
        out.append(val)
    return "".join(out)


def _go_replace_all(regex, replacement, s):
    # Go regexp.ReplaceAll semantics with Expand template handling
    pat = re.compile(regex)
    return pat.sub(lambda m: _go_expand(m, replacement), s)


def label_transform(series_list, label, regex, replacement):
    """transformLabelTransform (transform.go:2066): UNANCHORED ReplaceAll
    on the label (label_replace is the anchored Prometheus form)."""
    pat = re.compile(regex)
    for s in series_list:
        v = (s.mn.get_tag_value(label) or b"").decode("utf-8",
                                                      "surrogateescape")
        if not pat.search(v):
            continue
        out = _go_replace_all(regex, replacement, v)
        if label == "__name__":
            s.mn.metric_group = out.encode()
        elif out == "":
            s.mn.remove_tag(label)
        else:
            s.mn.set_tag(label, out)
    return series_list


def label_value(series_list, label):
    """transformLabelValue (transform.go:2175): replace values with the
    parsed float of the label (NaN when unparsable); metric group reset;
    all-NaN series kept (for `default`)."""
    for s in series_list:
        s.mn.reset_metric_group()
        raw = s.mn.get_tag_value(label)
        try:
            v = float((raw or b"").decode())
        except ValueError:
            v = math.nan
        mask = ~np.isnan(s.values)
        s.values[mask] = v
    return series_list


def labels_equal(series_list, labels):
    """transformLabelsEqual (transform.go:2138): keep series whose named
    labels all carry the same value."""
    out = []
    for s in series_list:
        if len(labels) < 2:
            out.append(s)
            continue
        first = s.mn.get_tag_value(labels[0])
        if all(s.mn.get_tag_value(l) == first for l in labels[1:]):
            out.append(s)
    return out


def label_graphite_group(series_list, group_ids):
    """transformLabelGraphiteGroup (transform.go:2260): rebuild the metric
    group from dot-separated components by index."""
    for s in series_list:
        groups = s.mn.metric_group.split(b".")
        parts = []
        for gid in group_ids:
            parts.append(groups[gid] if 0 <= gid < len(groups) else b"")
        s.mn.metric_group = b".".join(parts)
    return series_list


def _num_prefix(s):
    # getNumPrefix (transform.go:2510)
    i = 0
    if s and s[0] in "+-":
        i += 1
    has_num = has_dot = False
    while i < len(s):
        c = s[i]
        if not c.isdigit():
            if not has_dot and c == ".":
                has_dot = True
                i += 1
                continue
            return s[:i] if has_num else ""
        has_num = True
        i += 1
    return s if has_num else ""


def _nonnum_prefix(s):
    for i, c in enumerate(s):
        if c.isdigit():
            return s[:i]
    return s


def numeric_less(a, b):
    """numericLess (transform.go:2486): mixed numeric/string segments."""
    while True:
        if not b:
            return False
        if not a:
            return True
        ap, bp = _num_prefix(a), _num_prefix(b)
        a, b = a[len(ap):], b[len(bp):]
        if ap or bp:
            if not ap:
                return False
            if not bp:
                return True
            an, bn = float(ap), float(bp)
            if an != bn:
                return an < bn
        ap, bp = _nonnum_prefix(a), _nonnum_prefix(b)
        a, b = a[len(ap):], b[len(bp):]
        if ap != bp:
            return ap < bp


def sort_by_label_numeric(series_list, labels, desc=False):
    """sort_by_label_numeric[_desc] (newTransformFuncNumericSort): stable
    multi-key sort with numericLess per label."""
    import functools

    def cmp(x, y):
        for lb in labels:
            a = (x.mn.get_tag_value(lb) or b"").decode("utf-8",
                                                       "surrogateescape")
            bb = (y.mn.get_tag_value(lb) or b"").decode("utf-8",
                                                        "surrogateescape")
            if a == bb:
                continue
            if numeric_less(a, bb):
                return 1 if desc else -1
            return -1 if desc else 1
        return 0

    return sorted(series_list, key=functools.cmp_to_key(cmp))


def timezone_offset(tz_name, grid_timestamps_ms):
    """transformTimezoneOffset (transform.go:2769): UTC offset in seconds
    of the named IANA timezone at each grid timestamp."""
    from zoneinfo import ZoneInfo
    from datetime import datetime
    try:
        tz = ZoneInfo(tz_name)
    except Exception as e:
        raise ValueError(f"cannot load timezone {tz_name!r}: {e}")
    out = np.empty(len(grid_timestamps_ms))
    for i, t in enumerate(grid_timestamps_ms):
        dt = datetime.fromtimestamp(int(t) // 1000, tz)
        out[i] = dt.utcoffset().total_seconds()
    return out


# ---------------------------------------------------------------------------
# evaluation-context funcs (transform.go: time/now/pi/start/end/step/vector/
# scalar/rand*) — host generation, mirroring evalNumber/evalTime (eval.go)
# ---------------------------------------------------------------------------

def eval_number(grid_timestamps_ms, value):
    """evalNumber: one scalar series over the grid."""
    from .binary_op import Series
    return [Series(MetricName(), np.full(len(grid_timestamps_ms),
                                         float(value)))]


def eval_time(grid_timestamps_ms):
    """transformTime: grid timestamps in seconds."""
    from .binary_op import Series
    t = np.asarray(grid_timestamps_ms, dtype=np.float64) / 1e3
    return [Series(MetricName(), t)]


def eval_step(grid_timestamps_ms, step_ms):
    return eval_number(grid_timestamps_ms, step_ms / 1e3)


def eval_start(grid_timestamps_ms):
    return eval_number(grid_timestamps_ms, grid_timestamps_ms[0] / 1e3)


def eval_end(grid_timestamps_ms):
    return eval_number(grid_timestamps_ms, grid_timestamps_ms[-1] / 1e3)


def eval_pi(grid_timestamps_ms):
    return eval_number(grid_timestamps_ms, math.pi)


def eval_now(grid_timestamps_ms, now_s=None):
    """transformNow (transform.go:2725): wall-clock seconds as a scalar
    series; now_s injectable for tests."""
    if now_s is None:
        import time as _time
        now_s = _time.time()
    return eval_number(grid_timestamps_ms, now_s)


def scalar(series_list):
    """transformScalar: a single series passes through; otherwise NaN
    (Prometheus scalar() semantics)."""
    from .binary_op import Series, is_scalar
    if len(series_list) == 1:
        return series_list
    n = len(series_list[0].values) if series_list else 0
    return [Series(MetricName(), np.full(n, math.nan))]


def vector(series_list):
    """transformVector: identity over instant vectors."""
    return series_list


def rand_series(grid_timestamps_ms, kind="uniform", seed=None):
    """rand/rand_exponential/rand_normal (newTransformRand): a seeded
    random series on the grid.  The reference uses Go math/rand; this
    mirror uses numpy's PCG64 — the CONTRACT is 'deterministic for a given
    seed', not cross-runtime bit-equality (the reference's own stream is
    Go-runtime-specific)."""
    from .binary_op import Series
    rng = np.random.default_rng(None if seed is None else int(seed))
    n = len(grid_timestamps_ms)
    if kind == "uniform":
        v = rng.random(n)
    elif kind == "exponential":
        v = rng.exponential(1.0, n)
    else:
        v = rng.standard_normal(n)
    return [Series(MetricName(), v)]


def histogram_quantiles(phi_label, phis, series_list):
    """transformHistogramQuantiles (transform.go): one histogram_quantile
    result set per phi, labeled phi_label=phi (Go %g formatting)."""
    out = []
    for phi in phis:
        part = histogram_transform("histogram_quantile",
                                   [s.copy_shallow() for s in series_list],
                                   arg=phi)
        for s in part:
            s.mn.remove_tag(phi_label)
            s.mn.add_tag(phi_label, "%g" % phi)
        out.extend(part)
    return out


def label_map(series_list, label, mapping):
    """transformLabelMap (transform.go): rewrite label values through a
    literal src->dst map; an empty result removes the label."""
    for s in series_list:
        cur = s.mn.get_tag_value(label)
        cur_s = (cur or b"").decode("utf-8", "surrogateescape")
        if cur_s in mapping:
            new = mapping[cur_s]
            if label == "__name__":
                s.mn.metric_group = MetricName._b(new)
            elif new == "":
                s.mn.remove_tag(label)
            else:
                s.mn.set_tag(label, new)
        elif cur is not None and len(cur) == 0:
            s.mn.remove_tag(label)
    return series_list
