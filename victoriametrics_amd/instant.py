"""Instant-rollup incremental optimization — host mirror of
evalInstantRollup (eval.go:1176-1536), the §8f(2) instant-query fast path:
a repeated instant query `rf(m[window])` at time T is answered from a
cached instant value at T-offset plus two small `[offset]`-window
evaluations instead of re-scanning the whole window.

This layer sits ABOVE the device seam: `eval_at(func_name, timestamp_ms,
window_ms) -> list[Series]` is the storage+kernel callback (the
reference's evalRollupFuncNoCache at a single grid point — here a
SeriesBatch.exec with start == end and the named rollup function); the
evaluator only composes results and talks to RollupResultCache's
instant-values store.

Supported compositions (same function set as the reference):
  - sum-decomposable (count_*_over_time, count_over_time, increase,
    increase_pure, sum_over_time):  cached + start - end
  - max_over_time / min_over_time:  max/min(cached, start) with the
    consistency check on the end window (fall back + cache invalidation
    when the extremum may have left the window)
  - rate:      increase / (window seconds)        (eval.go:1325-1346)
  - avg_over_time:  sum_over_time / count_over_time  (eval.go:1267-1291)
When an incremental-aggregate name is present (`iafc_name`), the
optimization applies only when it matches the function's aggregation
(sum for the sum-decomposable set and rate, max/min for max/min), as in
the reference; eval_at is then expected to return already-aggregated
series, for which the same compositions remain exact.
"""
import math

import numpy as np

from .metric_name import MetricName  # noqa: F401  (re-export convenience)

# -search.minWindowForInstantRollupOptimization default (eval.go:50)
MIN_WINDOW_MS = 3 * 3600 * 1000
# -search.cacheTimestampOffset default (rollup_result_cache.go:29)
from .cache import CACHE_TIMESTAMP_OFFSET_MS

_SUM_FUNCS = {"count_eq_over_time", "count_gt_over_time",
              "count_le_over_time", "count_ne_over_time", "count_over_time",
              "increase", "increase_pure", "sum_over_time"}


def _name_key(s):
    return s.mn.marshal_sorted()


def has_duplicate_series(tss):
    """hasDuplicateSeries (eval.go:1537)."""
    seen = set()
    for s in tss:
        k = _name_key(s)
        if k in seen:
            return True
        seen.add(k)
    return False


def _to_instant(series_list):
    """Force 1-point values (assertInstantValues shape)."""
    for s in series_list:
        v = np.asarray(s.values, np.float64).reshape(-1)
        if v.size != 1:
            raise ValueError("instant series must have exactly one value")
        s.values = v
    return series_list


def get_sum_instant_values(cached, start, end):
    """getSumInstantValues (eval.go:1644): cached + start - end, keyed by
    sorted metric name; series missing from `cached` are adopted from
    `start`."""
    m = {}
    order = []
    for s in cached:
        k = _name_key(s)
        if k in m:
            raise ValueError("duplicate cached series")
        m[k] = s
        order.append(k)
    for s in start:
        k = _name_key(s)
        t = m.get(k)
        if t is not None and not math.isnan(t.values[0]):
            if not math.isnan(s.values[0]):
                t.values[0] += s.values[0]
        else:
            if k not in m:
                order.append(k)
            m[k] = s
    for s in end:
        t = m.get(_name_key(s))
        if t is not None and not math.isnan(t.values[0]):
            if not math.isnan(s.values[0]):
                t.values[0] -= s.values[0]
    return [m[k] for k in order]


def get_minmax_instant_values(cached, start, end, f):
    """getMinMaxInstantValues (eval.go:1589): combine cached/start with f;
    returns (series, ok) — ok=False when a value in `end` equals the
    winning extremum while `start` can't reproduce it (the extremum may
    have left the lookbehind window)."""
    m = {}
    order = []
    for s in cached:
        k = _name_key(s)
        if k in m:
            raise ValueError("duplicate cached series")
        m[k] = s
        order.append(k)
    m_start = {}
    for s in start:
        k = _name_key(s)
        if k in m_start:
            raise ValueError("duplicate start series")
        m_start[k] = s
        t = m.get(k)
        if t is not None and not math.isnan(t.values[0]):
            if not math.isnan(s.values[0]):
                t.values[0] = f(s.values[0], t.values[0])
        else:
            if k not in m:
                order.append(k)
            m[k] = s
    for s in end:
        k = _name_key(s)
        t = m.get(k)
        if t is not None and not math.isnan(t.values[0]) \
                and not math.isnan(s.values[0]):
            if s.values[0] == f(s.values[0], t.values[0]):
                ts_start = m_start.get(k)
                if ts_start is None or math.isnan(ts_start.values[0]) \
                        or ts_start.values[0] != f(s.values[0],
                                                   ts_start.values[0]):
                    return None, False
    return [m[k] for k in order], True


class InstantRollupEvaluator:
    def __init__(self, cache, eval_at, step, now_ms, filters=b"",
                 may_cache=True, min_window_ms=MIN_WINDOW_MS):
        """cache: RollupResultCache; eval_at(func_name, timestamp_ms,
        window_ms) -> list[Series] (one value each, at that timestamp);
        now_ms: the evaluation wall clock (injectable; the reference uses
        fasttime)."""
        self.cache = cache
        self._eval_at = eval_at
        self.step = int(step)
        self.now_ms = int(now_ms)
        self.filters = filters
        self.may_cache = may_cache
        self.min_window_ms = min_window_ms

    def eval_at(self, func_name, timestamp, window):
        return _to_instant(self._eval_at(func_name, int(timestamp),
                                         int(window)))

    def _too_big_offset(self, offset, window):
        # eval.go:1196: offset must stay below min(window/2, 30min)
        return offset >= min(window // 2, 1800 * 1000)

    def _get_cached_series(self, func_name, expr, timestamp, window):
        """getCachedSeries (eval.go:1202): returns (series, offset) with
        series evaluated at timestamp-offset; offset==0 means the result
        is already exact for `timestamp`."""
        while True:
            got_n, got_v, got_ts = self.cache.get_instant_values(
                expr, window, self.step, self.filters)
            if got_n is None:
                start = self.now_ms - CACHE_TIMESTAMP_OFFSET_MS
                offset = timestamp - start
                if offset < 0:
                    start = timestamp
                    offset = 0
                if self._too_big_offset(offset, window):
                    return self.eval_at(func_name, timestamp, window), 0
                tss = self.eval_at(func_name, start, window)
                if has_duplicate_series(tss):
                    return self.eval_at(func_name, timestamp, window), 0
                self.cache.put_instant_values(
                    expr, window, self.step,
                    [(s.mn.metric_group, tuple(s.mn.tags)) for s in tss],
                    np.asarray([s.values for s in tss], np.float64)
                    .reshape(len(tss), 1) if tss else np.empty((0, 1)),
                    [start], filters=self.filters)
                return tss, offset
            offset = timestamp - got_ts
            if offset < 0 or self._too_big_offset(offset, window):
                self.cache.delete_instant_values(expr, window, self.step,
                                                 self.filters)
                continue
            from .binary_op import Series
            tss = [Series(MetricName(g, list(t)), got_v[i].copy())
                   for i, (g, t) in enumerate(got_n)]
            return tss, offset

    def eval(self, func_name, expr, timestamp, window, iafc_name=None):
        """evalInstantRollup dispatch (eval.go:1258-1536)."""
        timestamp, window = int(timestamp), int(window)
        if not self.may_cache or window < self.min_window_ms:
            return self.eval_at(func_name, timestamp, window)

        if func_name == "avg_over_time":
            if iafc_name is not None:
                return self.eval_at(func_name, timestamp, window)
            s = self.eval("sum_over_time", ("sum_over_time", expr),
                          timestamp, window)
            c = self.eval("count_over_time", ("count_over_time", expr),
                          timestamp, window)
            cm = {_name_key(x): x for x in c}
            out = []
            for x in s:
                y = cm.get(_name_key(x))
                if y is not None:
                    with np.errstate(divide="ignore", invalid="ignore"):
                        x.values = x.values / y.values
                    out.append(x)
            return out

        if func_name == "rate":
            if iafc_name is not None and iafc_name.lower() != "sum":
                return self.eval_at(func_name, timestamp, window)
            d = window if window != 0 else self.step
            tss = self.eval("increase", ("increase", expr), timestamp,
                            window, iafc_name=iafc_name)
            for s in tss:
                s.values = s.values / (d / 1000.0)
            return tss

        if func_name in ("max_over_time", "min_over_time"):
            want = "max" if func_name == "max_over_time" else "min"
            if iafc_name is not None and iafc_name.lower() != want:
                return self.eval_at(func_name, timestamp, window)
            cached, offset = self._get_cached_series(func_name, expr,
                                                     timestamp, window)
            if offset == 0:
                return cached
            start = self.eval_at(func_name, timestamp, offset)
            if has_duplicate_series(start):
                return self.eval_at(func_name, timestamp, window)
            end = self.eval_at(func_name, timestamp - window, offset)
            if has_duplicate_series(end):
                return self.eval_at(func_name, timestamp, window)
            f = max if want == "max" else min
            tss, ok = get_minmax_instant_values(cached, start, end, f)
            if not ok:
                self.cache.delete_instant_values(expr, window, self.step,
                                                 self.filters)
                return self.eval_at(func_name, timestamp, window)
            return tss

        if func_name in _SUM_FUNCS:
            if iafc_name is not None and iafc_name.lower() != "sum":
                return self.eval_at(func_name, timestamp, window)
            cached, offset = self._get_cached_series(func_name, expr,
                                                     timestamp, window)
            if offset == 0:
                return cached
            start = self.eval_at(func_name, timestamp, offset)
            if has_duplicate_series(start):
                return self.eval_at(func_name, timestamp, window)
            end = self.eval_at(func_name, timestamp - window, offset)
            if has_duplicate_series(end):
                return self.eval_at(func_name, timestamp, window)
            return get_sum_instant_values(cached, start, end)

        return self.eval_at(func_name, timestamp, window)
