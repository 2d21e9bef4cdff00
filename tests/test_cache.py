"""rollupResultCache layer: merge vectors transcribed from the reference's
TestMergeSeries (rollup_result_cache_test.go:322-480) + binary-layout pins
for marshalTimeseriesFast (timeseries.go:81)."""
import math
import struct

import numpy as np
import pytest

from victoriametrics_amd import cache

NAN = math.nan
EC = dict(start=1000, end=2000, step=200)
NONAME = (b"", ())


def nan_eq(a, b):
    a, b = np.asarray(a, dtype=float), np.asarray(b, dtype=float)
    return (np.isnan(a) == np.isnan(b)).all() and \
        (a[~np.isnan(a)] == b[~np.isnan(b)]).all()


def test_merge_bstart_equals_start():
    r = cache.merge_series([], np.empty((0, 0)), [NONAME],
                           [[1, 2, 3, 4, 5, 6]], 1000, **EC)
    names, vals = r
    assert nan_eq(vals[0], [1, 2, 3, 4, 5, 6])


def test_merge_a_empty():
    names, vals = cache.merge_series([], np.empty((0, 2)), [NONAME],
                                     [[3, 4, 5, 6]], 1400, **EC)
    assert nan_eq(vals[0], [NAN, NAN, 3, 4, 5, 6])


def test_merge_b_empty():
    names, vals = cache.merge_series([NONAME], [[2, 1]], [], np.empty((0, 4)),
                                     1400, **EC)
    assert nan_eq(vals[0], [2, 1, NAN, NAN, NAN, NAN])


def test_merge_non_empty():
    names, vals = cache.merge_series([NONAME], [[2, 1]], [NONAME],
                                     [[3, 4, 5, 6]], 1400, **EC)
    assert nan_eq(vals[0], [2, 1, 3, 4, 5, 6])


def test_merge_distinct_names():
    names, vals = cache.merge_series([(b"bar", ())], [[2, 1]], [(b"foo", ())],
                                     [[3, 4, 5, 6]], 1400, **EC)
    # b series first, then leftover a series (reference order)
    assert names[0][0] == b"foo" and names[1][0] == b"bar"
    assert nan_eq(vals[0], [NAN, NAN, 3, 4, 5, 6])
    assert nan_eq(vals[1], [2, 1, NAN, NAN, NAN, NAN])


def test_merge_duplicate_series_fails():
    assert cache.merge_series([NONAME, NONAME], [[2, 1], [3, 3]], [NONAME],
                              [[3, 4, 5, 6]], 1400, **EC) is None
    assert cache.merge_series([NONAME], [[1, 2]], [NONAME, NONAME],
                              [[3, 4, 5, 6], [7, 8, 9, 10]], 1400,
                              **EC) is None


def test_merge_tag_order_insensitive_keys():
    a_name = (b"m", ((b"a", b"1"), (b"b", b"2")))
    b_name = (b"m", ((b"b", b"2"), (b"a", b"1")))  # same key, sorted form
    names, vals = cache.merge_series([a_name], [[2, 1]], [b_name],
                                     [[3, 4, 5, 6]], 1400, **EC)
    assert len(names) == 1
    assert nan_eq(vals[0], [2, 1, 3, 4, 5, 6])


def test_marshal_layout_pins():
    # empty: two big-endian zero u64s (timeseries.go:83-87)
    assert cache.marshal_timeseries_fast([], np.empty((0, 0)),
                                         np.empty(0, np.int64)) == b"\x00" * 16
    ts = np.array([1000, 1200], dtype=np.int64)
    vals = np.array([[1.5, 2.5]])
    names = [(b"foo", ((b"k", b"v"),))]
    data = cache.marshal_timeseries_fast(names, vals, ts)
    # header: big-endian counts
    assert data[:16] == struct.pack(">QQ", 1, 2)
    # timestamps then values as raw native 8-byte slices
    assert data[16:32] == ts.tobytes()
    assert data[32:48] == vals.tobytes()
    # metric name: u16 group len + group, u16 ntags, u16-framed key/value
    exp = struct.pack(">H", 3) + b"foo" + struct.pack(">H", 1) + \
        struct.pack(">H", 1) + b"k" + struct.pack(">H", 1) + b"v"
    assert data[48:] == exp
    back_names, back_vals, back_ts = cache.unmarshal_timeseries_fast(data)
    assert back_names == names
    assert np.array_equal(back_vals, vals)
    assert np.array_equal(back_ts, ts)


def test_marshal_roundtrip_random():
    rng = np.random.default_rng(1)
    for n, npts in ((1, 1), (5, 30), (100, 7)):
        vals = rng.standard_normal((n, npts))
        vals[rng.random((n, npts)) < 0.1] = np.nan
        ts = np.arange(npts, dtype=np.int64) * 200 + 10**12
        names = [(f"m{i}".encode(),
                  ((b"pod", f"p{i % 7}".encode()), (b"ns", b"x")))
                 for i in range(n)]
        data = cache.marshal_timeseries_fast(names, vals, ts)
        bn, bv, bt = cache.unmarshal_timeseries_fast(data)
        assert bn == names
        assert nan_eq(bv, vals)
        assert np.array_equal(bt, ts)


def test_marshal_max_size_guard():
    ts = np.arange(100, dtype=np.int64)
    vals = np.zeros((10, 100))
    names = [(b"m", ())] * 10
    assert cache.marshal_timeseries_fast(names, vals, ts, max_size=64) == b""


FUTURE = 1 << 60  # now_ms far past every test timestamp (no deadline trim)


def _grid(start, end, step):
    return np.arange(start, end + 1, step, dtype=np.int64)


def test_cache_roundtrip_partial_hit():
    c = cache.RollupResultCache()
    names = [(b"m", ())]
    full = np.array([[1.0, 2.0, 3.0, 4.0, 5.0, 6.0]])
    # cache [1000, 1400]
    c.put_series("rate(m[5m])", 300_000, 200, names, full[:, :3],
                 _grid(1000, 1400, 200), now_ms=FUTURE)
    got_n, got_v, _, new_start = c.get_series("rate(m[5m])", 300_000, 200,
                                              1000, 2000)
    assert new_start == 1600
    assert nan_eq(got_v, full[:, :3])
    # compute the suffix, merge, and store the full window
    merged_n, merged_v = cache.merge_series(got_n, got_v, names,
                                            full[:, 3:], new_start,
                                            1000, 2000, 200)
    assert nan_eq(merged_v, full)
    c.put_series("rate(m[5m])", 300_000, 200, merged_n, merged_v,
                 _grid(1000, 2000, 200), now_ms=FUTURE)
    _, v2, _, ns2 = c.get_series("rate(m[5m])", 300_000, 200, 1000, 2000)
    assert ns2 > 2000 and nan_eq(v2, full)
    # different key misses
    assert c.get_series("rate(m[1m])", 60_000, 200, 1000, 2000)[0] is None


class TestRollupResultCacheGolden:
    """TestRollupResultCache (rollup_result_cache_test.go:30-320) — the
    eleven subcases, with ec = {Start:1000, End:2000, Step:200},
    window=456.  `fe`/`ae` from the reference are distinct cache keys."""
    W, STEP, START, END = 456, 200, 1000, 2000

    def _get(self, c, expr="foo"):
        return c.get_series(expr, self.W, self.STEP, self.START, self.END)

    def _put(self, c, ts, vals, expr="foo", names=None):
        c.put_series(expr, self.W, self.STEP,
                     names or [(b"", ())],
                     np.asarray(vals, np.float64).reshape(len(names or [0]), -1),
                     np.asarray(ts, np.int64), now_ms=FUTURE)

    def test_empty(self):
        c = cache.RollupResultCache()
        n, v, t, ns = self._get(c)
        assert n is None and ns == self.START

    @pytest.mark.parametrize("expr", ["foo", "foobar"])
    def test_start_overlap(self, expr):
        # start-overlap-no-ae / -with-ae: {800,1000,1200} -> {1000,1200},
        # newStart 1400
        c = cache.RollupResultCache()
        self._put(c, [800, 1000, 1200], [0, 1, 2], expr=expr)
        n, v, t, ns = self._get(c, expr=expr)
        assert ns == 1400
        assert list(t) == [1000, 1200] and list(v[0]) == [1, 2]

    def test_end_overlap(self):
        # {1800,2000,2200,2400} doesn't contain Start -> miss
        c = cache.RollupResultCache()
        self._put(c, [1800, 2000, 2200, 2400], [333, 0, 1, 2])
        n, v, t, ns = self._get(c)
        assert n is None and ns == 1000

    def test_full_cover(self):
        # {1200,1400,1600} starts after Start -> miss
        c = cache.RollupResultCache()
        self._put(c, [1200, 1400, 1600], [0, 1, 2])
        n, v, t, ns = self._get(c)
        assert n is None and ns == 1000

    def test_before_start(self):
        c = cache.RollupResultCache()
        self._put(c, [200, 400, 600], [0, 1, 2])
        n, v, t, ns = self._get(c)
        assert n is None and ns == 1000

    def test_after_end(self):
        c = cache.RollupResultCache()
        self._put(c, [2200, 2400, 2600], [0, 1, 2])
        n, v, t, ns = self._get(c)
        assert n is None and ns == 1000

    def test_bigger_than_start_end(self):
        c = cache.RollupResultCache()
        self._put(c, [800, 1000, 1200, 1400, 1600, 1800, 2000, 2200],
                  [0, 1, 2, 3, 4, 5, 6, 7])
        n, v, t, ns = self._get(c)
        assert ns == 2200
        assert list(t) == [1000, 1200, 1400, 1600, 1800, 2000]
        assert list(v[0]) == [1, 2, 3, 4, 5, 6]

    def test_start_end_match(self):
        c = cache.RollupResultCache()
        self._put(c, [1000, 1200, 1400, 1600, 1800, 2000],
                  [1, 2, 3, 4, 5, 6])
        n, v, t, ns = self._get(c)
        assert ns == 2200
        assert list(v[0]) == [1, 2, 3, 4, 5, 6]

    def test_big_timeseries(self):
        # 1000 series -> marshaled size > 64Kb in the reference (tests the
        # GetBig/SetBig split there; here it must simply round-trip)
        c = cache.RollupResultCache()
        names = [(b"metric %d" % i, ()) for i in range(1000)]
        vals = np.tile([1.0, 2, 3, 4, 5, 6], (1000, 1))
        self._put(c, [1000, 1200, 1400, 1600, 1800, 2000], vals, names=names)
        n, v, t, ns = self._get(c)
        assert ns == 2200
        assert n == names and nan_eq(v, vals)

    def test_duplicate_series_not_stored(self):
        c = cache.RollupResultCache()
        names = [(b"", ()), (b"", ())]
        self._put(c, [800, 1000, 1200], np.tile([0.0, 1, 2], (2, 1)),
                  names=names)
        n, v, t, ns = self._get(c)
        assert n is None and ns == self.START

    def test_multi_timeseries(self):
        # three entries under one key; GetBestKey picks the one covering
        # Start with the longest usable span -> tss1, newStart 1400
        c = cache.RollupResultCache()
        self._put(c, [800, 1000, 1200], [0, 1, 2])
        self._put(c, [1800, 2000, 2200, 2400], [333, 0, 1, 2])
        self._put(c, [1200, 1400, 1600], [0, 1, 2])
        n, v, t, ns = self._get(c)
        assert ns == 1400
        assert list(t) == [1000, 1200] and list(v[0]) == [1, 2]

    def test_deadline_trims_fresh_tail(self):
        # PutSeries drops points newer than now - step - 5m
        # (rollup_result_cache.go:392-415)
        c = cache.RollupResultCache()
        now = 2_000_000
        ts = [1000, 1200, 1400, 1_800_000]
        c.put_series("foo", self.W, self.STEP, [(b"", ())],
                     np.asarray([[1.0, 2, 3, 4]]), np.asarray(ts, np.int64),
                     now_ms=now)
        n, v, t, ns = self._get(c)
        assert ns == 1600 and list(t) == [1000, 1200, 1400]

    def test_instant_values_roundtrip(self):
        # GetInstantValues/PutInstantValues/DeleteInstantValues
        # (rollup_result_cache.go:220-281): flat key, one point per series
        c = cache.RollupResultCache()
        names = [(b"a", ()), (b"b", ((b"x", b"y"),))]
        vals = np.asarray([[1.5], [2.5]])
        c.put_instant_values("up", 0, 300, names, vals, [5000])
        n, v, t = c.get_instant_values("up", 0, 300)
        assert n == [(b"a", ()), (b"b", ((b"x", b"y"),))]
        assert nan_eq(v, vals) and t == 5000
        # different (window, step) miss; delete clears
        assert c.get_instant_values("up", 1, 300)[0] is None
        c.delete_instant_values("up", 0, 300)
        assert c.get_instant_values("up", 0, 300)[0] is None
        # multi-point input is rejected (assertInstantValues)
        with pytest.raises(ValueError):
            c.put_instant_values("up", 0, 300, names,
                                 np.ones((2, 2)), [5000, 5300])

    def test_metainfo_entry_cap(self):
        # AddKey keeps at most 10 entries, dropping the oldest 5 past that
        # (rollup_result_cache.go:595-607)
        c = cache.RollupResultCache()
        for k in range(12):  # disjoint ranges far above [START, END]
            base = 10_000 + k * 1000
            self._put(c, [base, base + 200], [k, k])
        entries = c._meta[c._key("foo", self.W, self.STEP)]
        assert len(entries) <= 10
        # the latest entry is still retrievable
        n, v, t, ns = c.get_series("foo", self.W, self.STEP, 21_000, 21_200)
        assert ns == 21_400 and list(v[0]) == [11, 11]


def test_align_start_end():
    """alignStartEnd (eval.go:103-112): floor start, ceil end to step."""
    from victoriametrics_amd.engine import align_start_end
    assert align_start_end(1000, 2000, 200) == (1000, 2000)
    assert align_start_end(1050, 1950, 200) == (1000, 2000)
    assert align_start_end(999, 2001, 200) == (800, 2200)
    assert align_start_end(0, 0, 200) == (0, 0)
    # negative starts: Go's truncated % (unlike Python's floored %):
    # -150 % 200 == -150 in Go -> start stays -150 - (-150) = 0;
    # -50 % 200 == -50 -> adjust <= 0 -> end unchanged
    assert align_start_end(-150, 50, 200) == (0, 200)
    assert align_start_end(-350, -50, 200) == (-200, -50)


def test_subquery_grid_parameters():
    """evalRollupFuncWithSubquery's grid derivation (eval.go:1033-1060):
    sq grid extends the outer window + maxSilenceInterval and aligns."""
    from victoriametrics_amd import engine
    captured = {}

    def inner(sq_start, sq_end, sq_step):
        captured["grid"] = (sq_start, sq_end, sq_step)
        n = 1 + (sq_end - sq_start) // sq_step
        raise RuntimeError("stop")  # grid captured; no GPU needed

    start, end, step, window, sq_step = 1_000_000, 2_000_000, 200_000, \
        300_000, 100_000
    try:
        engine.rollup_subquery("max_over_time", start, end, step, window,
                               sq_step, inner)
    except RuntimeError:
        pass
    sq_start, sq_end, got_step = captured["grid"]
    assert got_step == sq_step
    exp_start, exp_end = engine.align_start_end(
        start - (window + sq_step + engine.MAX_SILENCE_INTERVAL_MS),
        end + sq_step, sq_step)
    assert (sq_start, sq_end) == (exp_start, exp_end)
    assert sq_start <= start - window
    assert sq_end >= end
    # sq_step defaults to the outer step (eval.go:1040)
    captured.clear()
    try:
        engine.rollup_subquery("max_over_time", start, end, step, window,
                               0, inner)
    except RuntimeError:
        pass
    assert captured["grid"][2] == step


def test_cache_get_best_key_fuzz_vs_model():
    """Model-based check of GetSeries entry selection: random cached
    windows on a shared step grid; the returned slice must match a brute
    force over all stored entries using GetBestKey's rule (max usable
    span d = min(end, e.end) - start among entries with e.start <= start,
    later entries winning ties)."""
    rng = np.random.default_rng(21)
    step = 100
    for trial in range(50):
        c = cache.RollupResultCache()
        entries = []
        for _ in range(int(rng.integers(1, 7))):
            s = int(rng.integers(0, 40)) * step
            e = s + int(rng.integers(1, 30)) * step
            ts = np.arange(s, e + 1, step, dtype=np.int64)
            vals = rng.random((1, len(ts)))
            before = len(c._meta.get(c._key("q", 0, step), []) or [])
            c.put_series("q", 0, step, [(b"", ())], vals, ts,
                         now_ms=FUTURE)
            after = len(c._meta.get(c._key("q", 0, step), []) or [])
            covered = any(s >= es and e <= ee for es, ee, _ in entries)
            if after > before:
                entries.append((s, e, (ts, vals)))
            else:
                assert covered, (trial, s, e, entries)
        start = int(rng.integers(0, 60)) * step
        end = start + int(rng.integers(1, 30)) * step
        n, v, t, ns = c.get_series("q", 0, step, start, end)
        # brute-force GetBestKey
        best, d_max = None, 0
        for es, ee, data in entries:
            if start < es:
                continue
            d = (end if end <= ee else ee) - start
            if d >= d_max:
                d_max, best = d, (es, ee, data)
        if best is None:
            assert n is None and ns == start, trial
            continue
        es, ee, (ts_b, vals_b) = best
        i = int(np.searchsorted(ts_b, start))
        if i == len(ts_b) or ts_b[i] != start:
            assert n is None and ns == start, trial
            continue
        j = int(np.searchsorted(ts_b, end, side="right"))
        if j <= i:
            assert n is None and ns == start, trial
            continue
        assert n is not None, (trial, best, start, end)
        np.testing.assert_array_equal(t, ts_b[i:j])
        np.testing.assert_array_equal(v.view(np.int64),
                                      vals_b[:, i:j].view(np.int64))
        assert ns == int(ts_b[j - 1]) + step


def test_cache_persistence_roundtrip(tmp_path):
    """InitRollupResultCache/StopRollupResultCache disk round trip
    (rollup_result_cache.go:119-199): series entries, multi-range metainfo
    and instant values survive a save/load; blobs stay in the
    marshalTimeseriesFast layout."""
    from victoriametrics_amd.cache import (RollupResultCache,
                                           load_rollup_result_cache,
                                           save_rollup_result_cache)
    c = RollupResultCache()
    names = [(b"m", ((b"a", b"b"),))]
    ts1 = np.arange(1_000_000, 1_000_000 + 10 * 15_000, 15_000,
                    dtype=np.int64)
    v1 = np.arange(10, dtype=np.float64).reshape(1, 10)
    now = int(ts1[-1]) + 10_000_000
    c.put_series("rate(m)", 300_000, 15_000, names, v1, ts1, now_ms=now)
    ts2 = ts1 + 20 * 15_000
    c.put_series("rate(m)", 300_000, 15_000, names, v1 * 2.0, ts2,
                 now_ms=now)
    c.put_instant_values("sum(m)", 0, 15_000, names,
                         np.array([[42.0]]), np.array([123_000], np.int64))

    path = str(tmp_path / "rollupResult")
    save_rollup_result_cache(c, path)
    c2 = load_rollup_result_cache(path)

    got_n, got_v, got_t, new_start = c2.get_series(
        "rate(m)", 300_000, 15_000, int(ts1[0]), int(ts1[-1]))
    assert got_n is not None
    np.testing.assert_array_equal(got_t, ts1)
    np.testing.assert_array_equal(got_v, v1)
    got_n2, got_v2, got_t2, _ = c2.get_series(
        "rate(m)", 300_000, 15_000, int(ts2[0]), int(ts2[-1]))
    np.testing.assert_array_equal(got_v2, v1 * 2.0)
    inames, ivals, its = c2.get_instant_values("sum(m)", 0, 15_000)
    assert its == 123_000 and ivals[0, 0] == 42.0
    assert c2._size == c._size


def test_cache_persistence_missing_or_corrupt(tmp_path):
    from victoriametrics_amd.cache import load_rollup_result_cache
    c = load_rollup_result_cache(str(tmp_path / "nope"))
    assert c.get_series("x", 0, 15_000, 0, 15_000)[0] is None
    bad = tmp_path / "bad"
    bad.write_bytes(b"garbage")
    c2 = load_rollup_result_cache(str(bad))
    assert c2.get_series("x", 0, 15_000, 0, 15_000)[0] is None


def test_cache_persistence_fuzz_equivalence(tmp_path):
    """Random put/get/save/load interleavings: a reloaded cache must answer
    every get_series/get_instant_values exactly like the live cache."""
    from victoriametrics_amd.cache import (RollupResultCache,
                                           load_rollup_result_cache,
                                           save_rollup_result_cache)
    rng = np.random.default_rng(12)
    c = RollupResultCache()
    step = 15_000
    exprs = ["rate(a)", "sum(b)", "max(c)"]
    now = 2_000_000_000_000
    for i in range(120):
        expr = exprs[int(rng.integers(len(exprs)))]
        window = int(rng.choice([0, 300_000]))
        n = int(rng.integers(2, 30))
        t0 = 1_000_000_000_000 + int(rng.integers(0, 50)) * step
        ts = t0 + np.arange(n, dtype=np.int64) * step
        vals = rng.standard_normal((1, n))
        names = [(b"m", ((b"i", str(i % 7).encode()),))]
        if rng.random() < 0.8:
            c.put_series(expr, window, step, names, vals, ts, now_ms=now)
        else:
            c.put_instant_values(expr, window, step, names,
                                 vals[:, :1], ts[:1])
    path = str(tmp_path / "cache")
    save_rollup_result_cache(c, path)
    c2 = load_rollup_result_cache(path)
    for expr in exprs:
        for window in (0, 300_000):
            for probe in range(12):
                start = 1_000_000_000_000 + int(rng.integers(0, 80)) * step
                end = start + int(rng.integers(1, 40)) * step
                a = c.get_series(expr, window, step, start, end)
                b = c2.get_series(expr, window, step, start, end)
                assert (a[0] is None) == (b[0] is None), (expr, window, start)
                assert a[3] == b[3]
                if a[0] is not None:
                    assert a[0] == b[0]
                    np.testing.assert_array_equal(a[1], b[1])
                    np.testing.assert_array_equal(a[2], b[2])
            ia = c.get_instant_values(expr, window, step)
            ib = c2.get_instant_values(expr, window, step)
            assert (ia[0] is None) == (ib[0] is None)
            if ia[0] is not None:
                assert ia[0] == ib[0] and ia[2] == ib[2]
                np.testing.assert_array_equal(ia[1], ib[1])
