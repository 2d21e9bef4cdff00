"""Series-level aggregate dispatch — host mirror of aggr.go's aggrFuncExt
(:110) + removeGroupTags (:96) over Series lists, with the per-point math
on the GPU:

- simple reducers (sum/min/max/avg/count/sum2/geomean/group) and the
  column statistics (median/quantile/mad/mode/distinct/stddev/stdvar) run
  through vmgpu_colagg;
- share/zscore rewrite member series in place (per-series outputs);
- outliers_iqr / outliers_mad / outliersk filter member series via the
  bounds + filter kernels;
- topk/bottomk families route to the existing selection kernels
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
              arg=None):
    """aggrFuncs dispatch (aggr.go:40) over resident Series."""
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
        # aggrFuncAny: one series per group (deterministic: first member)
        groups = prepare_series(series, modifier_op, modifier_args, limit)
        rvs = []
        for gmn, members in groups:
            dst = members[0]
            dst.mn = gmn
            rvs.append(dst)
        return rvs
    if name == "limitk":
        # aggrFuncLimitK: first k member series per group, original names
        groups = prepare_series(series, modifier_op, modifier_args, 0,
                                keep_original=True)
        k = int(arg)
        rvs = []
        for _, members in groups:
            rvs.extend(members[:k])
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
    raise ValueError(f"unsupported aggregate {name!r} "
                     "(topk/bottomk: engine.topk_*; count_values/histogram: "
                     "host metadata layer)")


def _reduce(op, groups, phi=0.0):
    if not groups:
        return []
    v, gr, go = _matrix(groups)
    out = engine.colagg(op, v, gr, go, phi=phi)
    rvs = []
    for gi, (gmn, members) in enumerate(groups):
        rvs.append(Series(gmn, out[gi]))
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
                    mn = members[0].mn.copy()
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
                    mn = members[0].mn.copy()
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
