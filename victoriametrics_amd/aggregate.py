"""Series-level aggregate dispatch — host mirror of aggr.go's aggrFuncExt
(:110) + removeGroupTags (:96) over Series lists, with the per-point math
on the GPU:

- simple reducers (sum/min/max/avg/count/sum2/geomean/group) and the
  column statistics (median/quantile/mad/mode/distinct/stddev/stdvar) run
  through vmgpu_colagg;
- share/zscore rewrite member series in place (per-series outputs);
- outliers_iqr / outliers_mad / outliersk filter member series via the
  bounds + filter kernels;
- topk/bottomk (per-point) and the topk_*/bottomk_* range families plus
  outliersk run host-side over resident Series (newAggrFuncTopK /
  newAggrFuncRangeTopK / aggrFuncOutliersK, aggr.go:669-802); the
  batch-resident selection path stays on the device kernels
  (engine.topk_range / topk_pointwise);
- any / limitk are metadata selections (host).

This layer exists for aggregation over RESIDENT result series (e.g. after
binary ops or transforms).  The rollup->aggregate hot path keeps using the
fused incremental aggregation inside the rollup kernels (SeriesBatch with
group_ids) — this module is the general aggr.go surface, not a replacement
for that path.

The reference's per-group member order is Go map/slice order (unspecified);
this mirror aggregates members in input order, which both the oracle and
the device kernels follow exactly.
"""
import math

import numpy as np

from . import engine
from .binary_op import Series, remove_empty_series

REDUCERS = {"sum", "min", "max", "avg", "count", "sum2", "geomean", "group",
            "median", "mad", "mode", "distinct", "stddev", "stdvar"}
PER_SERIES = {"share", "zscore"}


def remove_group_tags(mn, modifier_op, modifier_args):
    """removeGroupTags (aggr.go:96)."""
    op = (modifier_op or "").lower()
    if op in ("", "by"):
        mn.remove_tags_on(modifier_args)
    elif op == "without":
        mn.remove_tags_ignoring(modifier_args)
        mn.reset_metric_group()
    else:
        raise ValueError(f"unknown group modifier {modifier_op!r}")


def prepare_series(series, modifier_op="", modifier_args=(), max_series=0,
                   keep_original=False):
    """aggrPrepareSeries (aggr.go:121): drop empty series, group by the
    modifier key.  Returns list of (group_mn, [member Series])."""
    series = remove_empty_series(series)
    m = {}
    order = []
    for s in series:
        mn = s.mn.copy()
        remove_group_tags(mn, modifier_op, modifier_args)
        k = mn.marshal_sorted()
        if k not in m:
            if max_series > 0 and len(m) >= max_series:
                continue
            m[k] = (mn, [])
            order.append(k)
        m[k][1].append(s)
    return [m[k] for k in order]


def _matrix(groups):
    rows, group_rows, goff = [], [], [0]
    for _, members in groups:
        for s in members:
            group_rows.append(len(rows))
            rows.append(s.values)
        goff.append(len(group_rows))
    return (np.stack(rows), np.asarray(group_rows, np.uint32),
            np.asarray(goff, np.uint64))


def aggregate(name, series, modifier_op="", modifier_args=(), limit=0,
              arg=None, remaining_sum_tag=""):
    """aggrFuncs dispatch (aggr.go:40) over resident Series.  arg: phi for
    quantile, k for limitk and the topk/bottomk families (scalar or
    per-point array); remaining_sum_tag: the optional 3rd topk_*/bottomk_*
    argument ("tag" or "tag=value", aggr.go:751-760)."""
    name = name.lower()
    if name == "quantile":
        groups = prepare_series(series, modifier_op, modifier_args, limit)
        return _reduce("quantile", groups, phi=float(arg))
    if name in REDUCERS:
        groups = prepare_series(series, modifier_op, modifier_args, limit)
        return _reduce(name, groups)
    if name in PER_SERIES:
        groups = prepare_series(series, modifier_op, modifier_args, limit,
                                keep_original=True)
        if not groups:
            return []
        v, gr, go = _matrix(groups)
        out = engine.colagg(name, v, gr, go)
        rvs = []
        i = 0
        for _, members in groups:
            for s in members:
                s.values = out[i]
                i += 1
            rvs.extend(members)
        return rvs
    if name == "any":
        # aggrFuncAny (aggr.go:156): first member per group with its
        # ORIGINAL metric name (keepOriginal=true in the reference)
        groups = prepare_series(series, modifier_op, modifier_args,
                                min(limit, 1) if limit else limit,
                                keep_original=True)
        return [members[0] for _, members in groups]
    if name == "limitk":
        # aggrFuncLimitK (aggr.go:1108): per group, members sorted by
        # xxhash64 of the original metric name (uniform, call-stable
        # selection), first k kept; k<0 -> 0, k=inf -> all
        import xxhash
        if arg is None or math.isnan(float(arg)) or float(arg) < 0:
            k = 0
        elif math.isinf(float(arg)):
            k = None
        else:
            k = int(arg)
        groups = prepare_series(series, modifier_op, modifier_args, limit,
                                keep_original=True)
        rvs = []
        for _, members in groups:
            def _h(s):
                d = xxhash.xxh64()
                d.update(s.mn.metric_group)
                for tk, tv in sorted(s.mn.tags):
                    d.update(tk)
                    d.update(tv)
                return d.intdigest()
            ordered = sorted(members, key=_h)
            rvs.extend(ordered if k is None else ordered[:k])
        return rvs
    if name in ("outliers_iqr", "outliers_mad"):
        groups = prepare_series(series, modifier_op, modifier_args, limit,
                                keep_original=True)
        if not groups:
            return []
        v, gr, go = _matrix(groups)
        group_of = np.empty(v.shape[0], np.int32)
        for gi in range(len(groups)):
            group_of[int(go[gi]):int(go[gi + 1])] = gi
        if name == "outliers_iqr":
            lower, upper = engine.colagg("iqr_bounds", v, gr, go)
            flags = engine.colagg_filter("iqr", v, group_of, lower, upper)
        else:
            med = engine.colagg("median", v, gr, go)
            mad = engine.colagg("mad", v, gr, go)
            tol = float(arg)
            flags = engine.colagg_filter("mad", v, group_of, med, mad * tol)
        rvs = []
        i = 0
        for _, members in groups:
            for s in members:
                if flags[i]:
                    rvs.append(s)
                i += 1
        return rvs
    if name in ("topk", "bottomk") or name in _RANGE_TOPK \
            or name == "outliersk":
        groups = prepare_series(series, modifier_op, modifier_args, limit,
                                keep_original=True)
        if not groups:
            return []
        n_grid = len(groups[0][1][0].values)
        ks = np.broadcast_to(np.atleast_1d(
            np.asarray(arg, np.float64)), (n_grid,))
        if name == "topk":
            return _pointwise_topk(groups, ks, False)
        if name == "bottomk":
            return _pointwise_topk(groups, ks, True)
        if name == "outliersk":
            return _outliersk(groups, ks, modifier_op, modifier_args)
        summary, is_reverse = _RANGE_TOPK[name]
        return _range_topk(groups, ks, summary, is_reverse,
                           remaining_sum_tag, modifier_op, modifier_args)
    raise ValueError(f"unsupported aggregate {name!r} "
                     "(count_values/histogram: host metadata layer)")


def _reduce(op, groups, phi=0.0):
    if not groups:
        return []
    v, gr, go = _matrix(groups)
    out = engine.colagg(op, v, gr, go, phi=phi)
    rvs = []
    for gi, (gmn, members) in enumerate(groups):
        rvs.append(Series(gmn, out[gi]))
    return rvs


# ---------------------------------------------------------------------------
# topk/bottomk family over resident Series (newAggrFuncTopK /
# newAggrFuncRangeTopK / aggrFuncOutliersK, aggr.go:669-802, 1259-1290).
# Batch-resident selection stays on the device kernels
# (engine.topk_range / topk_pointwise); this is the aggr.go host surface.
# ---------------------------------------------------------------------------

def quantile_sorted(phi, values):
    """quantileSorted (aggr.go): values already sorted, NaN-free."""
    if len(values) == 0 or math.isnan(phi):
        return math.nan
    if phi < 0:
        return -math.inf
    if phi > 1:
        return math.inf
    n = float(len(values))
    rank = phi * (n - 1)
    lower = max(0.0, math.floor(rank))
    upper = min(n - 1, lower + 1)
    weight = rank - math.floor(rank)
    return values[int(lower)] * (1 - weight) + values[int(upper)] * weight


def go_quantile(phi, values):
    """quantile (aggr.go): prepareForQuantileFloat64 = drop NaNs + sort."""
    vs = sorted(float(v) for v in values if not math.isnan(v))
    return quantile_sorted(phi, vs)


def _min_value(values):
    m = math.nan
    for v in values:
        v = float(v)
        if not math.isnan(v) and not (v >= m):  # NaN m compares False
            m = v
    return m


def _max_value(values):
    m = math.nan
    for v in values:
        v = float(v)
        if not math.isnan(v) and not (v <= m):
            m = v
    return m


def _avg_value(values):
    s, count = 0.0, 0
    for v in values:
        v = float(v)
        if math.isnan(v):
            continue
        count += 1
        s += v
    return s / count if count else math.nan


def _median_value(values):
    return go_quantile(0.5, values)


def _last_value(values):
    # lastValue: last value before the trailing-NaN tail
    i = len(values)
    while i > 0 and math.isnan(float(values[i - 1])):
        i -= 1
    return float(values[i - 1]) if i else math.nan


_RANGE_TOPK = {
    "topk_min": (_min_value, False), "bottomk_min": (_min_value, True),
    "topk_max": (_max_value, False), "bottomk_max": (_max_value, True),
    "topk_avg": (_avg_value, False), "bottomk_avg": (_avg_value, True),
    "topk_median": (_median_value, False),
    "bottomk_median": (_median_value, True),
    "topk_last": (_last_value, False), "bottomk_last": (_last_value, True),
}


def _less_with_nans_key(v):
    # lessWithNaNs: NaNs sort below every number
    return (0, 0.0) if math.isnan(v) else (1, v)


def _greater_with_nans_key(v):
    # greaterWithNaNs ordering realized as an ascending key: NaNs first,
    # then numbers descending
    return (0, 0.0) if math.isnan(v) else (1, -v)


def _get_int_k(k, max_v):
    """getIntK + floatToIntBounded (aggr.go:1281): Go float->int with
    saturation; NaN and negatives mean 0."""
    if math.isnan(k):
        return 0
    if k >= 2.0 ** 63:
        kn = 1 << 63
    elif k <= -2.0 ** 63:
        kn = -(1 << 63)
    else:
        kn = int(k)  # truncation toward zero, as Go int(f)
    if kn < 0:
        return 0
    return min(kn, max_v)


def _sorted_by(tss, keyvals):
    # Go uses unstable sort.Slice; tie order there is unspecified, so the
    # stable sort here is one valid realization
    order = sorted(range(len(tss)), key=lambda i: keyvals[i])
    return [tss[i] for i in order]


def _remaining_sum_series(tss, modifier_op, modifier_args, ks, tag_name):
    """getRemainingSumTimeseries (aggr.go:751): per point, the sum of the
    non-selected (sorted-prefix) series' non-NaN values."""
    if not tag_name or not tss:
        return None
    mn = tss[0].mn.copy()
    remove_group_tags(mn, modifier_op, modifier_args)
    tag_value = tag_name
    if "=" in tag_name:
        tag_name, tag_value = tag_name.split("=", 1)
    mn.remove_tag(tag_name)
    mn.add_tag(tag_name, tag_value)
    n = len(tss)
    vals = np.empty(len(ks))
    for i, k in enumerate(ks):
        kn = _get_int_k(float(k), n)
        s, count = 0.0, 0
        for ts in tss[:n - kn]:
            v = float(ts.values[i])
            if math.isnan(v):
                continue
            s += v
            count += 1
        vals[i] = s if count else math.nan
    return Series(mn, vals)


def _range_topk(groups, ks, summary_f, is_reverse, remaining_sum_tag,
                modifier_op, modifier_args):
    """getRangeTopKTimeseries (aggr.go:704): rank whole series by a
    summary of their values, NaN-out all but the per-point top k."""
    keyf = _greater_with_nans_key if is_reverse else _less_with_nans_key
    rvs = []
    for _, tss in groups:
        tss = _sorted_by(tss, [keyf(summary_f(s.values)) for s in tss])
        rem = _remaining_sum_series(tss, modifier_op, modifier_args, ks,
                                    remaining_sum_tag)
        n = len(tss)
        for i, k in enumerate(ks):
            kn = _get_int_k(float(k), n)
            for s in tss[:n - kn]:
                s.values[i] = math.nan
        if rem is not None:
            tss.append(rem)
        tss = remove_empty_series(tss)
        tss.reverse()
        rvs.extend(tss)
    return rvs


def _pointwise_topk(groups, ks, is_reverse):
    """newAggrFuncTopK (aggr.go:669): re-rank per grid point."""
    keyf = _greater_with_nans_key if is_reverse else _less_with_nans_key
    rvs = []
    for _, tss in groups:
        tss = list(tss)
        n_grid = len(tss[0].values)
        for i in range(n_grid):
            tss = _sorted_by(tss, [keyf(float(s.values[i])) for s in tss])
            kn = _get_int_k(float(ks[i]), len(tss))
            for s in tss[:len(tss) - kn]:
                s.values[i] = math.nan
        tss = remove_empty_series(tss)
        tss.reverse()
        rvs.extend(tss)
    return rvs


def _per_point_medians(tss):
    """getPerPointMedians (aggr.go:730)."""
    n_grid = len(tss[0].values)
    out = np.empty(n_grid)
    for i in range(n_grid):
        out[i] = go_quantile(
            0.5, [float(s.values[i]) for s in tss
                  if not math.isnan(float(s.values[i]))])
    return out


def _outliersk(groups, ks, modifier_op, modifier_args):
    """aggrFuncOutliersK (aggr.go:686): range-topk by squared deviation
    from the per-point medians (sequential sum — a NaN value or median
    poisons that series' score, as in the reference)."""
    rvs = []
    for g in groups:
        medians = _per_point_medians(g[1])

        def f(values, medians=medians):
            s = 0.0
            for i, v in enumerate(values):
                d = float(v) - medians[i]
                s += d * d
            return s

        rvs.extend(_range_topk([g], ks, f, False, "", modifier_op,
                               modifier_args))
    return rvs


def format_go_float(v):
    """strconv.FormatFloat(v, 'f', -1, 64): shortest 'f'-format string that
    round-trips."""
    if v != v or math.isinf(v):
        return "NaN" if v != v else ("+Inf" if v > 0 else "-Inf")
    s = np.format_float_positional(v, unique=True, trim="-")
    return s


def count_values(dst_label, series, modifier_op="", modifier_args=(),
                 limit=0, max_series_per_aggr=1000):
    """aggrFuncCountValues (aggr.go:566): per distinct value, a series with
    dst_label=value counting occurrences per point.  dst_label is removed
    from the grouping as in Prometheus."""
    modifier_args = list(modifier_args)
    op = (modifier_op or "").lower()
    if op == "without":
        modifier_args = modifier_args + [dst_label]
    elif op == "by":
        modifier_args = [a for a in modifier_args if a != dst_label]
    groups = prepare_series(series, modifier_op, modifier_args, limit)
    rvs = []
    for gmn, members in groups:
        m = {}
        order = []
        for s in members:
            for i, v in enumerate(s.values):
                if math.isnan(v):
                    continue
                key = v
                dst = m.get(key)
                if dst is None:
                    if len(m) >= max_series_per_aggr:
                        raise ValueError(
                            f"more than {max_series_per_aggr} series "
                            "generated by count_values()")
                    # tss[0] inside the reference's afe carries the GROUP
                    # name (aggrPrepareSeries rewrites member names in
                    # place, aggr.go:121-146) — use gmn, not the member
                    mn = gmn.copy()
                    mn.remove_tag(dst_label)
                    mn.add_tag(dst_label, format_go_float(v))
                    dst = Series(mn, np.full(len(s.values), math.nan))
                    m[key] = dst
                    order.append(key)
                if math.isnan(dst.values[i]):
                    dst.values[i] = 1.0
                else:
                    dst.values[i] += 1.0
        rvs.extend(m[k] for k in order)
    return rvs


# ---------------------------------------------------------------------------
# aggrFuncHistogram (aggr.go:316) + the VictoriaMetrics histogram buckets
# (vendor/github.com/VictoriaMetrics/metrics/histogram.go: e10 range
# [-9,18], 18 buckets/decimal, vmrange = "%.3e...%.3e" with bounds built
# by repeated multiplication)
# ---------------------------------------------------------------------------

_H_E10_MIN, _H_E10_MAX, _H_BPD = -9, 18, 18
_H_BUCKETS = (_H_E10_MAX - _H_E10_MIN) * _H_BPD
_H_MULT = math.pow(10, 1.0 / _H_BPD)
_bucket_ranges = None


def _histogram_ranges():
    global _bucket_ranges
    if _bucket_ranges is None:
        v = math.pow(10, _H_E10_MIN)
        start = "%.3e" % v
        rs = []
        for _ in range(_H_BUCKETS):
            v *= _H_MULT
            end = "%.3e" % v
            rs.append(start + "..." + end)
            start = end
        _bucket_ranges = rs
    return _bucket_ranges


_H_LOWER = "0...%.3e" % math.pow(10, _H_E10_MIN)
_H_UPPER = "%.3e...+Inf" % math.pow(10, _H_E10_MAX)


def _histogram_update(buckets, v):
    # Histogram.Update (histogram.go:88): NaN and negatives skipped;
    # exact 10^n values drop to the lower bucket (Prometheus le logic)
    if math.isnan(v) or v < 0:
        return
    bucket_idx = (math.log10(v) - _H_E10_MIN) * _H_BPD if v > 0 else -1.0
    if bucket_idx < 0:
        buckets["lower"] = buckets.get("lower", 0) + 1
    elif bucket_idx >= _H_BUCKETS:
        buckets["upper"] = buckets.get("upper", 0) + 1
    else:
        idx = int(bucket_idx)
        if bucket_idx == float(idx) and idx > 0:
            idx -= 1
        buckets[idx] = buckets.get(idx, 0) + 1


def histogram_aggregate(series, modifier_op="", modifier_args=(), limit=0):
    """aggrFuncHistogram: per grid point, a VictoriaMetrics histogram over
    member values; non-zero buckets become vmrange series, converted to
    cumulative le buckets (vmrangeBucketsToLE)."""
    from .transform import vmrange_buckets_to_le
    ranges = _histogram_ranges()
    groups = prepare_series(series, modifier_op, modifier_args, limit)
    rvs = []
    for gmn, members in groups:
        n_grid = len(members[0].values)
        m = {}
        for i in range(n_grid):
            buckets = {}
            for s in members:
                _histogram_update(buckets, float(s.values[i]))
            for key, count in buckets.items():
                if key == "lower":
                    vmrange = _H_LOWER
                elif key == "upper":
                    vmrange = _H_UPPER
                else:
                    vmrange = ranges[key]
                ts = m.get(vmrange)
                if ts is None:
                    mn = gmn.copy()  # group name, as in aggr.go afe
                    mn.remove_tag("vmrange")
                    mn.add_tag("vmrange", vmrange)
                    ts = Series(mn, np.zeros(n_grid))
                    m[vmrange] = ts
                ts.values[i] = float(count)
        rvs.extend(m.values())
    return vmrange_buckets_to_le(rvs)


def quantiles(dst_label, phis, series, modifier_op="", modifier_args=(),
              limit=0):
    """aggrFuncQuantiles (aggr.go:1162): one quantile result set per phi,
    labeled dst_label=phi (Go %g)."""
    rvs = []
    for phi in phis:
        part = aggregate("quantile",
                         [s.copy_shallow() for s in series],
                         modifier_op, modifier_args, limit, arg=phi)
        for s in part:
            s.mn.remove_tag(dst_label)
            s.mn.add_tag(dst_label, "%g" % phi)
        rvs.extend(part)
    return rvs
