"""Instant-rollup incremental optimization (evalInstantRollup,
eval.go:1176-1536): composition correctness against a transparent
sample-store eval_at, cache population/invalidation behavior, and the
fallback gates."""
import math

import numpy as np
import pytest

from victoriametrics_amd import instant
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.cache import RollupResultCache
from victoriametrics_amd.metric_name import MetricName

H = 3600 * 1000
NAN = math.nan


class SampleStore:
    """eval_at over explicit (ts, value) samples per series; a window is
    (t-w, t] — the premise the decompositions rely on.  Records every
    (func, timestamp, window) call for assertions."""

    def __init__(self, series):
        # series: {name_str: [(ts_ms, value), ...]}
        self.series = series
        self.calls = []

    def eval_at(self, func, timestamp, window):
        self.calls.append((func, timestamp, window))
        out = []
        for name, samples in self.series.items():
            vals = [v for (t, v) in samples
                    if timestamp - window < t <= timestamp]
            if func in ("sum_over_time", "increase"):
                # increase under a cumulative-counter premise = the sum of
                # per-sample deltas in the window: sum-decomposable
                r = sum(vals) if vals else NAN
            elif func == "count_over_time":
                r = float(len(vals)) if vals else NAN
            elif func == "max_over_time":
                r = max(vals) if vals else NAN
            elif func == "min_over_time":
                r = min(vals) if vals else NAN
            else:
                raise AssertionError(func)
            if not math.isnan(r):
                out.append(Series(MetricName(b"m", [(b"s", name.encode())]),
                                  np.asarray([r])))
        return out

    def direct(self, func, timestamp, window):
        calls = self.calls
        self.calls = []
        try:
            return self.eval_at(func, timestamp, window)
        finally:
            self.calls = calls


def make_store(rng, n_series=4, t0=0, t1=10 * H, step=60_000):
    series = {}
    for i in range(n_series):
        ts = np.arange(t0, t1, step) + int(rng.integers(0, 1000))
        vs = rng.random(len(ts)) * 10
        series[f"s{i}"] = list(zip(ts.tolist(), vs.tolist()))
    return SampleStore(series)


def as_map(tss):
    return {s.mn.get_tag_value("s"): float(s.values[0]) for s in tss}


def make_ev(store, now, cache=None, step=300_000, **kw):
    return instant.InstantRollupEvaluator(
        cache if cache is not None else RollupResultCache(),
        store.eval_at, step=step, now_ms=now, **kw)


@pytest.mark.parametrize("func", ["sum_over_time", "count_over_time",
                                  "increase"])
def test_sum_decomposition_exact(func):
    rng = np.random.default_rng(3)
    store = make_store(rng)
    window, now = 4 * H, 9 * H
    ev = make_ev(store, now)
    # first call populates the cache at now - 5min
    t1 = now - 3 * 60_000
    got1 = ev.eval(func, "q", t1, window)
    assert as_map(got1) == pytest.approx(
        as_map(store.direct(func, t1, window)), rel=1e-12)
    # second call: only offset-sized windows hit the store
    store.calls.clear()
    t2 = now - 60_000
    got2 = ev.eval(func, "q", t2, window)
    assert as_map(got2) == pytest.approx(
        as_map(store.direct(func, t2, window)), rel=1e-12)
    assert store.calls and all(w < window for (_, _, w) in store.calls), \
        store.calls


def test_minmax_decomposition():
    rng = np.random.default_rng(4)
    store = make_store(rng)
    window, now = 4 * H, 9 * H
    for func in ("max_over_time", "min_over_time"):
        ev = make_ev(store, now)
        for t in (now - 4 * 60_000, now - 60_000, now):
            got = ev.eval(func, "q", t, window)
            assert as_map(got) == pytest.approx(
                as_map(store.direct(func, t, window)), rel=1e-12), (func, t)


def test_max_extremum_leaving_window_falls_back():
    # the global max sits at the oldest edge; once it slides out of the
    # window the end-check must refuse the cached composition
    samples = [(i * 60_000, 1.0) for i in range(200)]
    samples[0] = (0, 100.0)  # spike at t=0
    store = SampleStore({"s0": samples})
    now = 196 * 60_000
    window = 190 * 60_000
    ev = make_ev(store, now, step=60_000)
    t1 = now - 4 * 60_000  # t1-w = 2min: spike (t=0) already outside
    got = ev.eval("max_over_time", "q", t1, window)
    assert as_map(got) == pytest.approx(
        as_map(store.direct("max_over_time", t1, window)), rel=1e-12)
    t2 = now
    got = ev.eval("max_over_time", "q", t2, window)
    assert as_map(got) == pytest.approx(
        as_map(store.direct("max_over_time", t2, window)), rel=1e-12)


def test_max_spike_inside_window_stays_exact():
    samples = [(i * 60_000, 1.0) for i in range(200)]
    samples[100] = (100 * 60_000, 100.0)
    store = SampleStore({"s0": samples})
    now = 196 * 60_000
    ev = make_ev(store, now, step=60_000)
    for t in (now - 4 * 60_000, now - 60_000, now):
        got = ev.eval("max_over_time", "q", t, 190 * 60_000)
        assert as_map(got)[b"s0"] == 100.0


def test_rate_is_increase_over_seconds():
    rng = np.random.default_rng(5)
    store = make_store(rng, n_series=2)
    window, now = 4 * H, 9 * H
    ev = make_ev(store, now)
    t = now - 2 * 60_000
    got = ev.eval("rate", "q", t, window)
    inc = as_map(store.direct("increase", t, window))
    for k, v in as_map(got).items():
        assert v == pytest.approx(inc[k] / (window / 1000), rel=1e-12)


def test_avg_is_sum_over_count():
    rng = np.random.default_rng(6)
    store = make_store(rng, n_series=3)
    window, now = 4 * H, 9 * H
    ev = make_ev(store, now)
    t = now - 2 * 60_000
    got = ev.eval("avg_over_time", "q", t, window)
    s_direct = as_map(store.direct("sum_over_time", t, window))
    c_direct = as_map(store.direct("count_over_time", t, window))
    for k, v in as_map(got).items():
        assert v == pytest.approx(s_direct[k] / c_direct[k], rel=1e-12)
    # no full-window avg_over_time call went to the store
    assert all(f in ("sum_over_time", "count_over_time")
               for (f, _, _) in store.calls)


def test_small_window_and_no_cache_fall_back():
    store = SampleStore({"s0": [(0, 1.0), (60_000, 2.0)]})
    ev = make_ev(store, 10 * H, step=60_000)
    ev.eval("sum_over_time", "q", 100_000, 2 * H)  # < 3h window
    assert store.calls == [("sum_over_time", 100_000, 2 * H)]
    store.calls.clear()
    ev2 = make_ev(store, 10 * H, step=60_000, may_cache=False)
    ev2.eval("sum_over_time", "q", 100_000, 4 * H)
    assert store.calls == [("sum_over_time", 100_000, 4 * H)]


def test_too_big_offset_no_caching():
    store = SampleStore({"s0": [(i * 60_000, 1.0) for i in range(600)]})
    cache = RollupResultCache()
    now = 600 * 60_000
    ev = make_ev(store, now, cache=cache, step=60_000)
    # timestamp 30min past the cacheable clock (now - 5min): offset 35min
    # >= the min(window/2, 30min) cap -> direct eval, nothing cached
    t = now + 30 * 60_000
    ev.eval("sum_over_time", "q", t, 8 * H)
    assert store.calls == [("sum_over_time", t, 8 * H)]
    assert cache.get_instant_values("q", 8 * H, 60_000)[0] is None
    # a PAST timestamp is cacheable at that very timestamp (offset 0)
    store.calls.clear()
    t2 = now - 2 * H
    ev.eval("sum_over_time", "q", t2, 8 * H)
    assert store.calls == [("sum_over_time", t2, 8 * H)]
    assert cache.get_instant_values("q", 8 * H, 60_000)[2] == t2


def test_stale_cache_newer_than_timestamp_is_deleted():
    store = SampleStore({"s0": [(i * 60_000, 1.0) for i in range(600)]})
    cache = RollupResultCache()
    now = 600 * 60_000
    window = 8 * H
    # poison the cache with a FUTURE timestamp
    cache.put_instant_values("q", window, 60_000, [(b"m", ())],
                             np.asarray([[42.0]]), [now + H])
    ev = make_ev(store, now, cache=cache, step=60_000)
    t = now - 60_000
    got = ev.eval("sum_over_time", "q", t, window)
    assert as_map(got)[b"s0"] == pytest.approx(
        as_map(store.direct("sum_over_time", t, window))[b"s0"])
    n, v, ts = cache.get_instant_values("q", window, 60_000)
    assert ts is not None and ts <= t  # repopulated at now-5min


def test_iafc_gating():
    store = SampleStore({"s0": [(i * 60_000, 1.0) for i in range(600)]})
    now = 600 * 60_000
    ev = make_ev(store, now, step=60_000)
    t, w = now - 60_000, 8 * H
    # non-sum aggregate: no optimization, single full-window call
    store.calls.clear()
    ev.eval("sum_over_time", "q", t, w, iafc_name="avg")
    assert store.calls == [("sum_over_time", t, w)]
    # sum aggregate: the optimization applies (offset-sized windows)
    store.calls.clear()
    ev.eval("sum_over_time", "q2", t, w, iafc_name="sum")
    assert any(wi < w for (_, _, wi) in store.calls)


def test_new_series_appears_between_cache_and_now():
    # a series with samples only inside the offset window is adopted from
    # tssStart (getSumInstantValues else-branch)
    base = {"s0": [(i * 60_000, 2.0) for i in range(600)]}
    store = SampleStore(base)
    now = 600 * 60_000
    ev = make_ev(store, now, step=60_000)
    w = 8 * H
    ev.eval("sum_over_time", "q", now - 4 * 60_000, w)  # warm cache
    # new series appears after the cached timestamp
    store.series["s1"] = [(now - 60_000, 7.0)]
    got = ev.eval("sum_over_time", "q", now, w)
    m = as_map(got)
    assert m[b"s1"] == 7.0
    assert m[b"s0"] == pytest.approx(
        as_map(store.direct("sum_over_time", now, w))[b"s0"], rel=1e-12)


def test_randomized_composition_soak():
    """Randomized scenarios: sparse series, gaps, series appearing and
    disappearing, repeated queries at drifting timestamps — composed
    results must match direct evaluation for every supported func."""
    rng = np.random.default_rng(12)
    for trial in range(25):
        n_series = int(rng.integers(1, 6))
        series = {}
        t_end = 10 * H
        for i in range(n_series):
            # each series lives on a random sub-interval with gaps
            lo = int(rng.integers(0, 5 * H))
            hi = int(rng.integers(lo + H, t_end))
            ts = np.arange(lo, hi, 60_000)
            keep = rng.random(len(ts)) > 0.2
            ts = ts[keep]
            vs = rng.random(len(ts)) * 100
            series[f"s{i}"] = list(zip(ts.tolist(), vs.tolist()))
        store = SampleStore(series)
        now = int(rng.integers(8 * H, 10 * H))
        window = int(rng.integers(instant.MIN_WINDOW_MS, 6 * H))
        func = ["sum_over_time", "count_over_time", "max_over_time",
                "min_over_time"][int(rng.integers(0, 4))]
        ev = make_ev(store, now, step=60_000)
        # three queries at drifting timestamps reusing the same cache
        for q in range(3):
            t = now - int(rng.integers(0, 4 * 60_000))
            got = as_map(ev.eval(func, "q", t, window))
            want = as_map(store.direct(func, t, window))
            # accepted reference divergence (getSumInstantValues): a series
            # whose samples all slid out between the cached timestamp and t
            # composes to an explicit 0 (cached - end), while the direct
            # evaluation omits it entirely
            extra = set(got) - set(want)
            assert all(got[k] == pytest.approx(0.0, abs=1e-7)
                       for k in extra), (trial, q, func, extra)
            assert set(want) <= set(got), (trial, q, func)
            for k in want:
                assert got[k] == pytest.approx(want[k], rel=1e-9), \
                    (trial, q, func, k)


def test_get_sum_instant_values_reference_vectors():
    # TestGetSumInstantValues (eval_test.go:100): the six cached/start/end
    # combinations, expected sums verbatim
    from victoriametrics_amd.instant import get_sum_instant_values
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName

    def ts(name, value):
        return Series(MetricName(name), np.asarray([value], np.float64))

    def run(cached, start, end):
        out = get_sum_instant_values(cached, start, end)
        return {s.mn.metric_group: s.values[0] for s in out}

    # start only -> adopted
    assert run([], [ts("foo", 1)], []) == {b"foo": 1.0}
    # start - end (no cache) -> 0
    assert run([], [ts("foo", 1)], [ts("foo", 1)]) == {b"foo": 0.0}
    # cached + start -> 2
    assert run([ts("foo", 1)], [ts("foo", 1)], []) == {b"foo": 2.0}
    # cached + start - end -> 1
    assert run([ts("foo", 1)], [ts("foo", 1)], [ts("foo", 1)]) == \
        {b"foo": 1.0}
    # cached - end -> 0
    assert run([ts("foo", 1)], [], [ts("foo", 1)]) == {b"foo": 0.0}
    # cached only -> unchanged
    assert run([ts("foo", 1)], [], []) == {b"foo": 1.0}
