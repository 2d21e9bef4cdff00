"""rollupResultCache layer: merge vectors transcribed from the reference's
TestMergeSeries (rollup_result_cache_test.go:322-480) + binary-layout pins
for marshalTimeseriesFast (timeseries.go:81)."""
import math
import struct

import numpy as np

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


def test_cache_roundtrip_partial_hit():
    c = cache.RollupResultCache()
    names = [(b"m", ())]
    full = np.array([[1.0, 2.0, 3.0, 4.0, 5.0, 6.0]])
    # cache [1000, 1400]
    c.put("rate(m[5m])", 300_000, 200, 1000, 1400, names, full[:, :3])
    got_n, got_v, new_start = c.get("rate(m[5m])", 300_000, 200, 1000, 2000)
    assert new_start == 1600
    assert nan_eq(got_v, full[:, :3])
    # compute the suffix, merge, and store the full window
    merged_n, merged_v = cache.merge_series(got_n, got_v, names,
                                            full[:, 3:], new_start,
                                            1000, 2000, 200)
    assert nan_eq(merged_v, full)
    c.put("rate(m[5m])", 300_000, 200, 1000, 2000, merged_n, merged_v)
    _, v2, ns2 = c.get("rate(m[5m])", 300_000, 200, 1000, 2000)
    assert ns2 > 2000 and nan_eq(v2, full)
    # different key misses
    assert c.get("rate(m[1m])", 60_000, 200, 1000, 2000)[0] is None


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
