"""Transform layer, CPU side: oracle value-semantics pins (hand-derived
from transform.go) + the host label/metadata funcs + the decimal exponent
port transformRound needs."""
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd import transform as tf
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

NAN = math.nan


def row(*vals):
    return np.asarray([list(vals)], dtype=np.float64)


def S(name, tags, values):
    return Series(MetricName(name, tags), np.asarray(values, np.float64))


# ---------------------------------------------------------------------------
# oracle value semantics
# ---------------------------------------------------------------------------

def test_keep_last_value():
    out, _ = oracle.tf_apply(100, row(NAN, 1, NAN, NAN, 4, NAN))
    # leading NaN keeps values[0] (= NaN seed), gaps carry last value
    assert math.isnan(out[0, 0])
    assert list(out[0, 1:]) == [1, 1, 1, 4, 4]


def test_keep_next_value():
    out, _ = oracle.tf_apply(101, row(NAN, 1, NAN, 4, NAN))
    assert out[0, 0] == 1 and out[0, 2] == 4 and math.isnan(out[0, 4])


def test_interpolate():
    out, _ = oracle.tf_apply(102, row(NAN, 1.0, NAN, NAN, 4.0, NAN))
    # leading/trailing NaNs untouched; interior linear
    assert math.isnan(out[0, 0]) and math.isnan(out[0, 5])
    np.testing.assert_allclose(out[0, 1:5], [1, 2, 3, 4])


def test_running_and_range_sum():
    out, _ = oracle.tf_apply(103, row(1, 2, NAN, 3))
    assert list(out[0]) == [1, 3, 3, 6]
    out2, _ = oracle.tf_apply(107, row(1, 2, NAN, 3))
    assert list(out2[0]) == [6, 6, 6, 6]


def test_running_avg():
    out, _ = oracle.tf_apply(106, row(2, 4, 6))
    np.testing.assert_allclose(out[0], [2, 3, 4])


def test_range_first_last():
    out, _ = oracle.tf_apply(111, row(NAN, 5, 7, NAN))
    assert list(out[0]) == [5, 5, 5, 5]
    out2, _ = oracle.tf_apply(112, row(NAN, 5, 7, NAN))
    assert list(out2[0]) == [7, 7, 7, 7]


def test_range_normalize_keep_flag():
    v = np.asarray([[1.0, 3.0, 2.0], [NAN, NAN, NAN]])
    out, keep = oracle.tf_apply(113, v)
    np.testing.assert_allclose(out[0], [0, 1, 0.5])
    assert keep[0] == 1 and keep[1] == 0


def test_range_quantile_and_median():
    out, _ = oracle.tf_apply(122, row(4, 1, 3, 2), scalar=0.5)
    np.testing.assert_allclose(out[0], [2.5] * 4)


def test_range_trim_spikes():
    vals = list(range(10)) + [1000.0, -1000.0]
    out, _ = oracle.tf_apply(121, row(*vals), scalar=0.2)
    assert math.isnan(out[0, 10]) and math.isnan(out[0, 11])
    assert not math.isnan(out[0, 5])


def test_range_zscore_properties():
    rng = np.random.default_rng(5)
    v = rng.standard_normal((1, 50)) * 7 + 3
    out, _ = oracle.tf_apply(114, v.copy())
    assert abs(float(np.mean(out))) < 1e-12
    assert abs(float(np.std(out)) - 1.0) < 1e-12


def test_remove_resets_matches_rollup_semantics():
    # removeCounterResetsMaybeNaNs: NaNs skipped, partial-reset heuristic
    out, _ = oracle.tf_apply(124, row(10, 12, NAN, 2, 15))
    assert list(out[0])[:2] == [10, 12]
    assert math.isnan(out[0, 2])
    assert list(out[0])[3:] == [14, 27]


def test_smooth_exponential():
    sfs = [0.5] * 4
    out, _ = oracle.tf_apply(123, row(10.0, 20.0, 30.0, NAN), arg1=sfs)
    assert out[0, 0] == 10 and out[0, 1] == 15 and out[0, 2] == 22.5
    assert math.isnan(out[0, 3])


def test_elementwise_specials():
    out, _ = oracle.tf_apply(22, row(-3.5, 0.0, 2.0, NAN))  # sgn
    # transformSgn has no NaN check: NaN compares false both ways -> 0
    assert list(out[0]) == [-1, 0, 1, 0]
    out2, _ = oracle.tf_apply(23, row(1.0, 5.0, 9.0), arg1=[2, 2, 2],
                              arg2=[8, 8, 8])  # clamp
    assert list(out2[0]) == [2, 5, 8]


def test_bitmap_go_semantics():
    out, _ = oracle.tf_apply(27, row(6.0, 255.0, NAN), arg1=[3.0, 3.0, 3.0])
    assert list(out[0])[:2] == [2.0, 3.0] and math.isnan(out[0, 2])
    out2, _ = oracle.tf_apply(29, row(6.0), arg1=[3.0])  # xor
    assert out2[0, 0] == 5.0


def test_datetime_funcs():
    # 2021-03-14 15:09:26 UTC = 1615734566
    t = 1615734566.0
    cases = {30: 14, 31: 0, 32: 73, 33: 31, 34: 15, 35: 9, 36: 3, 37: 2021}
    for fid, exp in cases.items():
        out, _ = oracle.tf_apply(fid, row(t))
        assert out[0, 0] == exp, (fid, out[0, 0], exp)
    # leap year Feb
    t2 = 1582934400.0  # 2020-02-29
    out, _ = oracle.tf_apply(33, row(t2))
    assert out[0, 0] == 29


def test_range_linear_regression():
    ts = np.arange(4, dtype=np.int64) * 1000
    out, _ = oracle.tf_apply(118, row(1.0, 3.0, 5.0, 7.0), ts=ts)
    np.testing.assert_allclose(out[0], [1, 3, 5, 7], rtol=1e-12)
    # const fast path
    out2, _ = oracle.tf_apply(118, row(4.0, 4.0, 4.0, 4.0), ts=ts)
    assert list(out2[0]) == [4, 4, 4, 4]


# ---------------------------------------------------------------------------
# transformRound's decimal exponent port
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("nearest,expected_p10", [
    (1.0, 1.0), (0.1, 10.0), (0.01, 100.0), (0.5, 10.0), (10.0, 0.1),
    (100.0, 0.01), (0.25, 100.0), (2.0, 1.0), (1000.0, 0.001),
])
def test_decimal_exponent(nearest, expected_p10):
    e = tf.decimal_from_float_exponent(nearest)
    assert math.pow(10.0, -e) == expected_p10, (nearest, e)


def test_round_host_path_matches_oracle():
    vals = np.asarray([[1.234, 5.678, -2.345, 0.05]])
    nearest = np.asarray([0.1, 0.1, 0.1, 0.1])
    p10 = np.asarray([10.0] * 4)
    exp, _ = oracle.tf_apply(26, vals.copy(), arg1=nearest, arg2=p10)
    np.testing.assert_allclose(exp[0], [1.2, 5.7, -2.3, 0.1])


# ---------------------------------------------------------------------------
# host label funcs
# ---------------------------------------------------------------------------

def test_label_set_del_keep():
    s = S("m", [("a", "1"), ("b", "2")], [1])
    tf.label_set([s], [("c", "3"), ("a", "9")])
    assert s.mn.get_tag_value("c") == b"3" and s.mn.get_tag_value("a") == b"9"
    tf.label_del([s], ["b"])
    assert s.mn.get_tag_value("b") is None
    tf.label_keep([s], ["a"])
    assert s.mn.get_tag_value("c") is None
    assert s.mn.metric_group == b""  # __name__ not kept


def test_label_copy_move():
    s = S("m", [("src", "x")], [1])
    tf.label_copy([s], [("src", "dst")])
    assert s.mn.get_tag_value("dst") == b"x"
    assert s.mn.get_tag_value("src") == b"x"
    s2 = S("m", [("src", "y")], [1])
    tf.label_move([s2], [("src", "dst")])
    assert s2.mn.get_tag_value("dst") == b"y"
    assert s2.mn.get_tag_value("src") is None


def test_label_join_replace():
    s = S("m", [("a", "x"), ("b", "y")], [1])
    tf.label_join([s], "joined", "-", ["a", "b"])
    assert s.mn.get_tag_value("joined") == b"x-y"
    tf.label_replace([s], "r", "$1!", "joined", "(x)-y")
    assert s.mn.get_tag_value("r") == b"x!"
    # non-matching regex leaves things alone
    tf.label_replace([s], "r2", "$1", "joined", "zzz")
    assert s.mn.get_tag_value("r2") is None


def test_label_match_mismatch():
    a = S("m", [("env", "prod")], [1])
    b = S("m", [("env", "dev")], [1])
    assert tf.label_match([a, b], "env", "prod") == [a]
    assert tf.label_match([a, b], "env", "prod", negate=True) == [b]


def test_drop_common_labels():
    a = S("m", [("common", "c"), ("x", "1")], [1])
    b = S("m", [("common", "c"), ("x", "2")], [1])
    tf.drop_common_labels([a, b])
    assert a.mn.get_tag_value("common") is None
    assert a.mn.get_tag_value("x") == b"1"
    assert a.mn.metric_group == b""  # __name__ common -> dropped


def test_sort_series():
    # positional comparison from the last index (transform.go:2580):
    # b's NaN at index 1 sorts it before a regardless of desc; c (all
    # NaN) sorts before b at index 0.  The NaN rule is NOT flipped by
    # desc — only the value comparison is.
    a = S("a", [], [1, 5])
    b = S("b", [], [9, NAN])
    c = S("c", [], [NAN, NAN])
    out = tf.sort_series([a, b, c])
    assert out == [c, b, a]
    out_desc = tf.sort_series([a, b, c], desc=True)
    assert out_desc == [c, b, a]


def test_sort_by_label():
    a = S("m", [("k", "2")], [1])
    b = S("m", [("k", "1")], [1])
    assert tf.sort_by_label([a, b], ["k"]) == [b, a]


def test_limit_offset():
    ss = [S("m", [("i", str(i))], [1]) for i in range(5)]
    assert tf.limit_offset(ss, 2, 1) == ss[1:3]


# ---------------------------------------------------------------------------
# vmrangeBucketsToLE pins (transcribed from transform_test.go:70-210)
# ---------------------------------------------------------------------------

def _vmrange(buckets):
    out = []
    for rng, v in buckets:
        out.append(S("foo", [("vmrange", rng)], [float(v)]))
    return out


def _le_result(series):
    return [(s.mn.get_tag_value("le").decode(), float(s.values[0]))
            for s in series]


@pytest.mark.parametrize("buckets,expected", [
    # single non-empty bucket
    ([("4.084e+02...4.642e+02", 2)],
     [("4.084e+02", 0), ("4.642e+02", 2), ("+Inf", 2)]),
    # 0...+Inf: no gap series (Go zero-value xsPrev.end == 0)
    ([("0...+Inf", 5)], [("+Inf", 5)]),
    ([("-Inf...0", 4)], [("-Inf", 0), ("0", 4), ("+Inf", 4)]),
    ([("-Inf...+Inf", 1.23)], [("-Inf", 0), ("+Inf", 1.23)]),
    ([("0...0", 5.3)], [("0", 5.3), ("+Inf", 5.3)]),
    # adjacent empty bucket merged away
    ([("7.743e+05...8.799e+05", 5), ("6.813e+05...7.743e+05", 0)],
     [("7.743e+05", 0), ("8.799e+05", 5), ("+Inf", 5)]),
    # multiple adjacent empty buckets
    ([("7.743e+05...8.799e+05", 5), ("6.813e+05...7.743e+05", 0),
      ("5.813e+05...6.813e+05", 0)],
     [("7.743e+05", 0), ("8.799e+05", 5), ("+Inf", 5)]),
    ([("8.799e+05...9.813e+05", 0), ("7.743e+05...8.799e+05", 5),
      ("6.813e+05...7.743e+05", 0), ("5.813e+05...6.813e+05", 0)],
     [("7.743e+05", 0), ("8.799e+05", 5), ("+Inf", 5)]),
    # multiple non-empty buckets
    ([("4.084e+02...4.642e+02", 2), ("1.234e+02...4.084e+02", 3)],
     [("1.234e+02", 0), ("4.084e+02", 3), ("4.642e+02", 5), ("+Inf", 5)]),
    # disjoint buckets
    ([("1...2", 2), ("4...6", 3)],
     [("1", 0), ("2", 2), ("4", 2), ("6", 5), ("+Inf", 5)]),
    # intersected buckets
    ([("1...5", 2), ("4...6", 3)],
     [("1", 0), ("5", 2), ("4", 2), ("6", 5), ("+Inf", 5)]),
    # same end range: short series refuse the merge, dropped duplicate
    ([("1...5", 2), ("0...5", 3)],
     [("1", 0), ("5", 2), ("0", 2), ("+Inf", 2)]),
    # single/multiple empty buckets vanish
    ([("0...1", 0)], []),
    ([("0...+Inf", 0)], []),
    ([("-Inf...0", 0)], []),
    ([("0...0", 0)], []),
    ([("-Inf...+Inf", 0)], []),
    ([("2...3", 0), ("1...2", 0)], []),
])
def test_vmrange_buckets_to_le(buckets, expected):
    got = tf.vmrange_buckets_to_le(_vmrange(buckets))
    assert _le_result(got) == [(le, float(v)) for le, v in expected]


def test_vmrange_le_passthrough():
    le_series = [S("foo", [("le", "10")], [3.0])]
    out = tf.vmrange_buckets_to_le(le_series)
    assert out == le_series


def test_group_le_and_merge_same_le():
    xs = [S("m", [("le", "1"), ("pod", "a")], [1.0]),
          S("m", [("le", "2"), ("pod", "a")], [2.0]),
          S("m", [("le", "2"), ("pod", "a")], [3.0]),
          S("m", [("le", "5"), ("pod", "b")], [4.0]),
          S("m", [("le", "bad"), ("pod", "b")], [9.0])]
    m = tf.group_le_timeseries(xs)
    assert len(m) == 2
    groups = sorted(m.values(), key=len)
    assert [le for le, _ in groups[0]] == [5.0]
    big = sorted(groups[1], key=lambda x: x[0])
    merged = tf._merge_same_le(big)
    assert [le for le, _ in merged] == [1.0, 2.0]
    assert merged[1][1].values[0] == 5.0  # 2+3 summed


def test_absent_union():
    a = S("m", [], [1.0, NAN, NAN])
    b = S("m2", [], [NAN, 2.0, NAN])
    out = tf.absent([a, b], 3)
    v = out[0].values
    assert math.isnan(v[0]) and math.isnan(v[1]) and v[2] == 1.0
    out2 = tf.absent([], 2)
    assert list(out2[0].values) == [1.0, 1.0]
    # union keeps first occurrence per name
    u = tf.union([[S("m", [("x", "1")], [1])],
                  [S("m", [("x", "1")], [2]), S("m", [("x", "2")], [3])]])
    assert len(u) == 2
    assert u[0].values[0] == 1


def test_buckets_limit():
    # 6 buckets with concentrated hits; limit to 3 keeps first/last
    series = []
    cum = [1.0, 50.0, 51.0, 52.0, 99.0, 100.0]
    for le, v in zip(["1", "2", "3", "4", "5", "+Inf"], cum):
        series.append(S("m", [("le", le)], [v]))
    out = tf.buckets_limit(3, series)
    les = sorted(float(s.mn.get_tag_value("le")) for s in out)
    assert len(out) == 3
    assert les[0] == 1.0 and math.isinf(les[-1])
    # under the limit: untouched
    out2 = tf.buckets_limit(10, [s.copy_shallow() for s in series])
    assert len(out2) == 6


def test_any_representative_group_ids():
    from victoriametrics_amd.engine import any_representative_group_ids
    gids = any_representative_group_ids([2, 0, 2, 1, 0, -1, 1])
    assert list(gids) == [2, 0, -1, 1, -1, -1, -1]


def test_label_transform_unanchored():
    s = S("m", [("path", "a.b.c")], [1])
    tf.label_transform([s], "path", r"\.", "-")
    assert s.mn.get_tag_value("path") == b"a-b-c"
    # non-matching: untouched
    s2 = S("m", [("path", "abc")], [1])
    tf.label_transform([s2], "path", r"\d+", "X")
    assert s2.mn.get_tag_value("path") == b"abc"


def test_label_value():
    a = S("m", [("v", "2.5")], [1.0, NAN])
    b = S("m", [("v", "junk")], [3.0])
    tf.label_value([a, b], "v")
    assert a.mn.metric_group == b""
    assert a.values[0] == 2.5 and math.isnan(a.values[1])
    assert math.isnan(b.values[0])


def test_labels_equal():
    a = S("m", [("x", "1"), ("y", "1")], [1])
    b = S("m", [("x", "1"), ("y", "2")], [1])
    assert tf.labels_equal([a, b], ["x", "y"]) == [a]


def test_label_graphite_group():
    s = S("a.b.c.d", [], [1])
    tf.label_graphite_group([s], [0, 2])
    assert s.mn.metric_group == b"a.c"
    s2 = S("a.b", [], [1])
    tf.label_graphite_group([s2], [5, 1])
    assert s2.mn.metric_group == b".b"


def test_numeric_less_and_sort():
    assert tf.numeric_less("2", "10")
    assert not tf.numeric_less("10", "2")
    assert tf.numeric_less("v2", "v10")
    assert tf.numeric_less("abc", "abd")
    assert tf.numeric_less("1.5x", "1.5y")
    a = S("m", [("n", "pod10")], [1])
    b = S("m", [("n", "pod2")], [1])
    assert tf.sort_by_label_numeric([a, b], ["n"]) == [b, a]
    assert tf.sort_by_label_numeric([a, b], ["n"], desc=True) == [a, b]


def test_timezone_offset():
    # 2021-03-14 is the US DST transition day: 06:00 UTC is EST (-5h),
    # 15:09 UTC is EDT (-4h)
    ts = [1615698000000, 1615734566000]
    out = tf.timezone_offset("America/New_York", ts)
    assert list(out) == [-5 * 3600.0, -4 * 3600.0]
    utc = tf.timezone_offset("UTC", ts)
    assert list(utc) == [0.0, 0.0]


def test_eval_context_funcs():
    ts = [1000, 2000, 3000]
    assert list(tf.eval_time(ts)[0].values) == [1.0, 2.0, 3.0]
    assert list(tf.eval_number(ts, 7)[0].values) == [7.0, 7.0, 7.0]
    assert tf.eval_start(ts)[0].values[0] == 1.0
    assert tf.eval_end(ts)[0].values[0] == 3.0
    assert tf.eval_step(ts, 1000)[0].values[0] == 1.0
    assert abs(tf.eval_pi(ts)[0].values[0] - math.pi) < 1e-15
    s = S("m", [("a", "1")], [1, 2, 3])
    assert tf.scalar([s]) == [s]
    out = tf.scalar([s, s])
    assert all(math.isnan(v) for v in out[0].values)
    r1 = tf.rand_series(ts, seed=5)[0].values
    r2 = tf.rand_series(ts, seed=5)[0].values
    assert list(r1) == list(r2)


def test_label_map():
    a = S("m", [("env", "dev")], [1])
    b = S("m", [("env", "prd")], [1])
    c = S("m", [("env", "x")], [1])
    tf.label_map([a, b, c], "env", {"dev": "development",
                                    "prd": "production", "x": ""})
    assert a.mn.get_tag_value("env") == b"development"
    assert b.mn.get_tag_value("env") == b"production"
    assert c.mn.get_tag_value("env") is None
    # unmapped value untouched
    d = S("m", [("env", "qa")], [1])
    tf.label_map([d], "env", {"dev": "x"})
    assert d.mn.get_tag_value("env") == b"qa"


# ---------------------------------------------------------------------------
# histogram_fraction = share(upper) - share(lower) (transform.go:751-828),
# pinned against the TestExecSuccess expected arrays via the CPU oracle
# (the GPU path is pinned in test_exec_golden.py)
# ---------------------------------------------------------------------------

def _oracle_fraction(lower, upper, bucket_vals, les):
    bv = np.asarray(bucket_vals, np.float64)
    le = np.asarray(les, np.float64)
    off = np.asarray([0, len(le)], np.uint64)
    n = bv.shape[1]
    hi, _, _ = oracle.histogram_share(np.full(n, float(upper)), bv, le, off)
    lo, _, _ = oracle.histogram_share(np.full(n, float(lower)), bv, le, off)
    return (hi - lo)[0]


def test_histogram_fraction_exec_pins():
    # exec_test.go histogram_fraction(single-value-valid-le*) cases
    # (grid collapsed to one point; values are per-point constants there)
    assert _oracle_fraction(0, 100, [[100.0]], [200.0]) == [0.5]
    assert _oracle_fraction(200, 300, [[100.0]], [200.0]) == [0.0]
    # max-le: buckets {10: 0, 50: 40, 100: 100} -> fraction(0,100) = 1
    assert _oracle_fraction(0, 100, [[0.0], [40.0], [100.0]],
                            [10.0, 50.0, 100.0]) == [1.0]
    # min-le -> 0
    assert _oracle_fraction(0, 10, [[0.0], [40.0], [100.0]],
                            [10.0, 50.0, 100.0]) == [0.0]
    # mid-le: buckets {55: 0, 200: 100}, fraction(55, 105)
    got = _oracle_fraction(55, 105, [[0.0], [100.0]], [55.0, 200.0])
    assert got.view(np.int64) == np.float64(0.3448275862068966).view(np.int64)
    # NaN le propagates
    assert math.isnan(_oracle_fraction(math.nan, 105,
                                       [[0.0], [100.0]], [55.0, 200.0])[0])


def test_histogram_share_exec_pins():
    # histogram_share(single-value-valid-le / -mid-le) expected arrays
    def share(req, bv, les):
        bv = np.asarray(bv, np.float64)
        le = np.asarray(les, np.float64)
        off = np.asarray([0, len(le)], np.uint64)
        out, _, _ = oracle.histogram_share(
            np.full(bv.shape[1], float(req)), bv, le, off)
        return out[0]

    assert share(80, [[100.0]], [200.0]) == [0.4]
    assert share(200, [[100.0]], [200.0]) == [1.0]
    assert share(300, [[100.0]], [200.0]) == [1.0]
    got = share(105, [[0.0], [100.0]], [55.0, 200.0])
    assert got.view(np.int64) == np.float64(0.3448275862068966).view(np.int64)
    assert share(55, [[0.0], [100.0]], [55.0, 200.0]) == [0.0]
    assert share(0, [[0.0], [100.0]], [55.0, 200.0]) == [0.0]


def test_histogram_fraction_host_validation():
    # lower >= upper is a host-side error (transform.go:767-770)
    with pytest.raises(ValueError):
        tf.histogram_transform("histogram_fraction",
                               [S("m", [("le", "200")], [100.0])],
                               arg=(456, 123))
    # buckets without le / invalid le produce an empty result
    assert tf.histogram_transform("histogram_fraction",
                                  [S("m", [("foo", "bar")], [100.0])],
                                  arg=(123, 456)) == []
    assert tf.histogram_transform("histogram_fraction",
                                  [S("m", [("le", "foobar")], [100.0])],
                                  arg=(50, 60)) == []


def test_transform_dispatch_covers_reference_map():
    # the 111 keys of transformFuncs (transform.go:31-145) each have an
    # entry point: the device kernel dispatch tables or a named host func
    reference_names = {
        "", "abs", "absent", "acos", "acosh", "asin", "asinh", "atan",
        "atanh", "bitmap_and", "bitmap_or", "bitmap_xor", "buckets_limit",
        "ceil", "clamp", "clamp_max", "clamp_min", "cos", "cosh",
        "day_of_month", "day_of_week", "day_of_year", "days_in_month",
        "deg", "drop_common_labels", "drop_empty_series", "end", "exp",
        "floor", "histogram_avg", "histogram_fraction", "histogram_quantile",
        "histogram_quantiles", "histogram_share", "histogram_stddev",
        "histogram_stdvar", "hour", "interpolate", "keep_last_value",
        "keep_next_value", "label_copy", "label_del", "label_graphite_group",
        "label_join", "label_keep", "label_lowercase", "label_map",
        "label_match", "label_mismatch", "label_move", "label_replace",
        "label_set", "label_transform", "label_uppercase", "label_value",
        "labels_equal", "limit_offset", "ln", "log2", "log10", "minute",
        "month", "now", "pi", "prometheus_buckets", "rad", "rand",
        "rand_exponential", "rand_normal", "range_avg",
        "range_first", "range_last", "range_linear_regression", "range_mad",
        "range_max", "range_min", "range_normalize",
        "range_quantile", "range_stddev", "range_stdvar", "range_sum",
        "range_trim_outliers", "range_trim_spikes", "range_trim_zscore",
        "range_zscore", "remove_resets", "round", "running_avg",
        "running_max", "running_min", "running_sum", "scalar", "sgn", "sin",
        "sinh", "smooth_exponential", "sort", "sort_by_label",
        "sort_by_label_desc", "sort_by_label_numeric",
        "sort_by_label_numeric_desc", "sort_desc", "sqrt", "start", "step",
        "tan", "tanh", "time", "timezone_offset", "union", "vector",
        "year"}
    kernel = (set(tf._ELEMENTWISE) | set(tf._CLAMP) | {"round"} |
              set(tf._BITMAP) | set(tf._DATETIME) | set(tf._SERIES))
    host = {
        "absent": tf.absent, "buckets_limit": tf.buckets_limit,
        "drop_common_labels": tf.drop_common_labels,
        "drop_empty_series": None,  # remove_empty_series (binary_op)
        "end": tf.eval_end, "start": tf.eval_start, "step": tf.eval_step,
        "time": tf.eval_time, "pi": tf.eval_pi, "now": tf.eval_now,
        "histogram_avg": tf.histogram_transform,
        "histogram_fraction": tf.histogram_transform,
        "histogram_quantile": tf.histogram_transform,
        "histogram_share": tf.histogram_transform,
        "histogram_stddev": tf.histogram_transform,
        "histogram_stdvar": tf.histogram_transform,
        "histogram_quantiles": tf.histogram_quantiles,
        "label_copy": tf.label_copy, "label_del": tf.label_del,
        "label_graphite_group": tf.label_graphite_group,
        "label_join": tf.label_join, "label_keep": tf.label_keep,
        "label_lowercase": tf.label_lowercase, "label_map": tf.label_map,
        "label_match": tf.label_match,
        "label_mismatch": tf.label_match,   # negate=True
        "label_move": tf.label_move, "label_replace": tf.label_replace,
        "label_set": tf.label_set, "label_transform": tf.label_transform,
        "label_uppercase": tf.label_uppercase,
        "label_value": tf.label_value, "labels_equal": tf.labels_equal,
        "limit_offset": tf.limit_offset,
        "prometheus_buckets": tf.prometheus_buckets,
        "rand": tf.rand_series, "rand_exponential": tf.rand_series,
        "rand_normal": tf.rand_series,
        "scalar": tf.scalar, "vector": tf.vector, "union": tf.union,
        "sort": tf.sort_series, "sort_desc": tf.sort_series,  # desc=True
        "sort_by_label": tf.sort_by_label,
        "sort_by_label_desc": tf.sort_by_label,
        "sort_by_label_numeric": tf.sort_by_label_numeric,
        "sort_by_label_numeric_desc": tf.sort_by_label_numeric,
        "timezone_offset": tf.timezone_offset,
        "": None,  # the identity entry (metricsql union sugar)
    }
    missing = reference_names - kernel - set(host)
    assert not missing, f"transformFuncs without an entry point: {missing}"


def test_eval_now():
    out = tf.eval_now([0, 1000], now_s=1234.5)
    assert list(out[0].values) == [1234.5, 1234.5]


# ---------------------------------------------------------------------------
# union / drop_common_labels / limit_offset TestExecSuccess pins
# (exec_test.go:2113-2180, 2547-2600, 9311-9500) — host-side funcs
# ---------------------------------------------------------------------------

TIME6 = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def test_union_exec_pins():
    assert tf.union([]) == []
    # identical labels: later duplicates dropped
    out = tf.union([[S("", [("foo", "bar")], [1] * 6)],
                    [S("", [("foo", "bar")], [2] * 6)]])
    assert len(out) == 1 and list(out[0].values) == [1] * 6
    # metric group participates in identity
    out = tf.union([[S("xx", [("foo", "bar")], [1] * 6)],
                    [S("yy", [("foo", "bar")], [2] * 6)]])
    assert len(out) == 2
    by_g = {s.mn.metric_group: list(s.values) for s in out}
    assert by_g == {b"xx": [1] * 6, b"yy": [2] * 6}
    # tag order doesn't matter for identity
    out = tf.union([[S("xx", [("a", "1"), ("b", "2")], [1] * 6)],
                    [S("xx", [("b", "2"), ("a", "1")], [2] * 6)]])
    assert len(out) == 1


def test_drop_common_labels_exec_pins():
    # single series: everything common -> all labels dropped
    s = S("xxx", [("foo", "bar"), ("q", "we")], TIME6)
    tf.drop_common_labels([s])
    assert s.mn.metric_group == b"" and s.mn.tags == []
    # multi series: only (foo, bar) common; names differ and are kept
    s1 = S("xxx", [("foo", "bar"), ("q", "we")], TIME6)
    s2 = S("yyy", [("foo", "bar")], TIME6 / 10)
    tf.drop_common_labels([s1, s2])
    assert s1.mn.metric_group == b"xxx"
    assert s1.mn.tags == [(b"q", b"we")]
    assert s2.mn.metric_group == b"yyy" and s2.mn.tags == []
    # multi_args case: same __name__ on both -> name dropped too
    s1 = S("xxx", [("foo", "bar"), ("q", "we")], TIME6)
    s2 = S("xxx", [("foo", "bar")], TIME6 / 10)
    tf.drop_common_labels([s1, s2])
    assert s1.mn.metric_group == b"" and s1.mn.tags == [(b"q", b"we")]
    assert s2.mn.metric_group == b"" and s2.mn.tags == []


def test_limit_offset_exec_pins():
    def mk():
        return [S("", [("foo", "a")], TIME6 * 2),
                S("", [("foo", "x")], TIME6 * 3),
                S("", [("foo", "y")], TIME6 * 1)]  # sorted by foo

    out = tf.limit_offset(mk(), 1, 1)
    assert len(out) == 1 and out[0].mn.get_tag_value("foo") == b"x"
    np.testing.assert_array_equal(out[0].values, TIME6 * 3)
    assert tf.limit_offset(mk(), 1, 10) == []
    # empty series are filtered out BEFORE the offset (limit_offset NaN
    # case): desc order [3:all-NaN, 2:partial, 1:full] -> offset 1 skips
    # the partial series, not the NaN one
    vals3 = np.where(TIME6 * 3 < 3000, TIME6 * 3, math.nan)
    vals2 = np.where(TIME6 * 2 < 3000, TIME6 * 2, math.nan)
    tss = [S("", [("foo", "3")], vals3),
           S("", [("foo", "2")], vals2),
           S("", [("foo", "1")], TIME6)]
    out = tf.limit_offset(tss, 1, 1)
    assert len(out) == 1 and out[0].mn.get_tag_value("foo") == b"1"
    np.testing.assert_array_equal(out[0].values, TIME6)


def test_buckets_limit_exec_pins():
    # TestExecSuccess buckets_limit cases (exec_test.go:5236-5470)
    def bucket(le, v, extra=()):
        return S("metric", [("le", le)] + list(extra), np.full(6, float(v)))

    def les(out):
        return [(s.mn.get_tag_value("le").decode(), s.values[0]) for s in
                sorted(out, key=lambda s: float(s.mn.get_tag_value("le")))]

    # trim_zero_preserve_empty_when_limit_not_reached (limit 3): zero edge
    # buckets trimmed right-first, then left, down to the limit
    out = tf.buckets_limit(3, [
        bucket("+Inf", 36), bucket("25", 36), bucket("21", 36),
        bucket("19", 36), bucket("18", 36), bucket("17", 36),
        bucket("16", 36), bucket("12", 27), bucket("9", 14),
        bucket("6", 0), bucket("1", 0)])
    assert les(out) == [("9", 14.0), ("12", 27.0), ("16", 36.0)]
    # trim_zero (limit 5): only the right zero-delta edge is trimmed; the
    # left zero buckets survive because the limit is already met
    out = tf.buckets_limit(5, [
        bucket("18", 36), bucket("17", 36), bucket("16", 36),
        bucket("12", 27), bucket("9", 14), bucket("6", 0), bucket("1", 0)])
    assert les(out) == [("1", 0.0), ("6", 0.0), ("9", 14.0),
                        ("12", 27.0), ("16", 36.0)]
    # unused (limit 5 >= buckets): untouched
    out = tf.buckets_limit(5, [bucket("inf", 100, [("x", "y")]),
                               bucket("120", 50, [("x", "y")])])
    assert les(out) == [("120", 50.0), ("inf", 100.0)]
    # used (limit 2 -> min 3): smallest adjacent-delta pairs merged away
    out = tf.buckets_limit(2, [
        bucket("inf", 100, [("x", "y")]), bucket("300", 98, [("x", "y")]),
        bucket("200", 52, [("x", "y")]), bucket("120", 50, [("x", "y")]),
        bucket("70", 20, [("x", "y")]), bucket("30", 10, [("x", "y")]),
        bucket("10", 9, [("x", "y")])])
    assert les(out) == [("10", 9.0), ("300", 98.0), ("inf", 100.0)]


def test_sort_positional_nan_semantics():
    # newTransformFuncSort (transform.go:2580) compares positionally from
    # the LAST index backwards; a NaN opposite a value sorts first
    # REGARDLESS of desc; ties advance to the previous index
    a = S("a", [], [5.0, NAN])
    b = S("b", [], [1.0, 3.0])
    out = tf.sort_series([b, a])
    assert [s.mn.metric_group for s in out] == [b"a", b"b"]
    # desc flips only the value comparison, not the NaN rule
    out = tf.sort_series([b, a], desc=True)
    assert [s.mn.metric_group for s in out] == [b"a", b"b"]
    # tie at the last index -> decided by the previous one
    c = S("c", [], [2.0, 7.0])
    d = S("d", [], [9.0, 7.0])
    out = tf.sort_series([d, c])
    assert [s.mn.metric_group for s in out] == [b"c", b"d"]
    # fully equal -> input order preserved (stable realization of Go's
    # unspecified tie order)
    e = S("e", [], [2.0, 7.0])
    out = tf.sort_series([c, e])
    assert [s.mn.metric_group for s in out] == [b"c", b"e"]
    # exec pins: sort(2 or label_set(1,...)) / sort_desc(1 or label_set(2,..))
    two = S("", [], [2.0] * 6)
    one = S("", [("xx", "foo")], [1.0] * 6)
    assert tf.sort_series([two, one])[0] is one
    two2 = S("", [("xx", "foo")], [2.0] * 6)
    one2 = S("", [], [1.0] * 6)
    assert tf.sort_series([one2, two2], desc=True)[0] is two2


def test_sort_by_label_name():
    # "__name__" sorts by the metric group (exec_test.go sort_by_label)
    foo = S("foo", [], [1.0] * 6)
    bar = S("bar", [], [2.0] * 6)
    out = tf.sort_by_label([foo, bar], ["__name__"])
    assert [s.mn.metric_group for s in out] == [b"bar", b"foo"]
    out = tf.sort_by_label([foo, bar], ["__name__"], desc=True)
    assert [s.mn.metric_group for s in out] == [b"foo", b"bar"]


def test_label_join_empty_removes_dst():
    # transform.go:2059: an empty joined value removes the dst label —
    # a SINGLE missing source joins to "" (no separator inserted)
    s = S("m", [("keep", "x"), ("j", "old")], [1])
    tf.label_join([s], "j", "-", ["missing"])
    assert s.mn.get_tag_value("j") is None
    # two missing sources still produce the separator ("-"), which is
    # non-empty and therefore kept
    s2 = S("m", [], [1])
    tf.label_join([s2], "j", "-", ["a", "b"])
    assert s2.mn.get_tag_value("j") == b"-"


def test_round_to_decimal_digits_vs_oracle():
    # product decimal.round_to_decimal_digits vs the oracle C restatement
    from victoriametrics_amd import decimal as vmd
    import ctypes
    l = oracle.lib()
    l.vm_decimal_round_to_decimal_digits.restype = ctypes.c_double
    l.vm_decimal_round_to_decimal_digits.argtypes = [ctypes.c_double,
                                                     ctypes.c_int]
    rng = np.random.default_rng(11)
    vals = np.concatenate([
        rng.standard_normal(200) * 10.0 ** rng.integers(-12, 12, 200),
        [0.0, -0.0, 0.5, -0.5, 1.5, 2.5, -2.5, 0.49999999999999994,
         -0.49999999999999994, 4503599627370495.5, math.inf, -math.inf,
         math.nan,
         np.uint64(0x7FF0000000000002).view if False else
         np.frombuffer(np.uint64(0x7FF0000000000002).tobytes(),
                       np.float64)[0]],
    ])
    for digits in (-101, -5, -1, 0, 1, 2, 5, 12, 50, 99, 100):
        got = vmd.round_to_decimal_digits(vals, digits)
        for i, v in enumerate(vals):
            want = l.vm_decimal_round_to_decimal_digits(
                ctypes.c_double(v), digits)
            g, w = float(got[i]), float(want)
            assert (np.float64(g).view(np.int64) ==
                    np.float64(w).view(np.int64)), (v, digits, g, w)


def test_go_round_edges():
    from victoriametrics_amd import decimal as vmd
    # the classic trunc(x+0.5) trap: 0.49999999999999994 rounds to 0
    got = vmd.go_round([0.49999999999999994, -0.49999999999999994,
                        0.5, -0.5, 1.5, 2.5, -2.5, 0.0, -0.0])
    assert list(got[:2]) == [0.0, -0.0] or (got[0] == 0 and got[1] == 0)
    assert list(got[2:7]) == [1.0, -1.0, 2.0, 3.0, -3.0]
    # signed zero preserved
    assert math.copysign(1, got[7]) == 1 and math.copysign(1, got[8]) == -1


def test_sort_by_label_numeric_exec_pins():
    # exec_test.go:9797-10010 numericLess cases
    def mk(vals_by_tag, tag="foo"):
        return [S("", [(tag, v)], [i]) for i, v in enumerate(vals_by_tag)]

    # string-only labels: lexicographic within equal first key
    a = S("", [("x", "b"), ("y", "aa")], [1])
    b = S("", [("x", "a"), ("y", "aa")], [2])
    out = tf.sort_by_label_numeric([a, b], ["y", "x"])
    assert out == [b, a]
    # numeric segments: "1:0:2" < "1:0:15"
    a = S("", [("x", "1:0:2"), ("y", "1:0:1")], [1])
    b = S("", [("x", "1:0:15"), ("y", "1:0:1")], [2])
    assert tf.sort_by_label_numeric([b, a], ["x", "y"]) == [a, b]
    assert tf.sort_by_label_numeric([a, b], ["x", "y"],
                                    desc=True) == [b, a]
    # alias numbers with special chars
    tss = mk(["DS50:1/0/15", "DS50:1/0/0", "DS50:1/0/1", "DS50:1/0/2"],
             tag="a")
    out = tf.sort_by_label_numeric(tss, ["a"])
    assert [s.mn.get_tag_value("a") for s in out] == [
        b"DS50:1/0/0", b"DS50:1/0/1", b"DS50:1/0/2", b"DS50:1/0/15"]
    # the desc multi-value case feeding limit_offset (exec_test.go:9914)
    tss = mk(["1:0:3", "5:0:15", "1:0:2", "7:0:15", "3:0:1", "1:0:2",
              "9:0:15"])
    out = tf.sort_by_label_numeric(tss, ["foo"], desc=True)
    assert [s.mn.get_tag_value("foo") for s in out] == [
        b"9:0:15", b"7:0:15", b"5:0:15", b"3:0:1", b"1:0:3", b"1:0:2",
        b"1:0:2"]


def test_prometheus_buckets_overlapped_ranges_exec_pin():
    # exec_test.go:5661: overlapping vmranges accumulate cumulatively in
    # le order, with the zero-width "0...0" bucket first
    t20 = TIME6 / 20
    t100 = TIME6 / 100
    t10 = TIME6 / 10
    series = [
        S("xxx", [("foo", "bar"), ("vmrange", "0...0")], np.full(6, 90.0)),
        S("xxx", [("foo", "bar"), ("vmrange", "0...0.2")], t20),
        S("xxx", [("foo", "bar"), ("vmrange", "0.2...0.25")], t20),
        S("xxx", [("foo", "bar"), ("vmrange", "0...0.26")], t20),
        S("xxx", [("foo", "bar"), ("vmrange", "0.2...40")], t100),
        S("xxx", [("foo", "bar"), ("vmrange", "40...Inf")], t10),
    ]
    out = tf.prometheus_buckets(series)
    got = {s.mn.get_tag_value("le").decode(): list(s.values) for s in out}
    assert got["0"] == [90.0] * 6
    assert got["0.2"] == [140, 150, 160, 170, 180, 190]
    assert got["0.25"] == [190, 210, 230, 250, 270, 290]
    assert got["0.26"] == [240, 270, 300, 330, 360, 390]
    assert got["40"] == [250, 282, 314, 346, 378, 410]
    # the end string is kept VERBATIM ("Inf", not "+Inf") — the +Inf
    # synthetic bucket is only added when the last range is finite
    assert got["Inf"] == [350, 402, 454, 506, 558, 610]
    assert all(s.mn.metric_group == b"xxx" for s in out)


def test_absent_exec_pins():
    # absent(time() > 1500) -> [1,1,1,nan,nan,nan] (exec_test.go:1099)
    masked = TIME6.copy()
    masked[TIME6 <= 1500] = NAN
    out = tf.absent([S("", [], masked)], 6)
    got = out[0].values
    assert list(got[:3]) == [1.0, 1.0, 1.0]
    assert all(math.isnan(x) for x in got[3:])
    # absent over a fully-present series -> all NaN (removed at Exec)
    out = tf.absent([S("", [], TIME6.copy())], 6)
    assert all(math.isnan(x) for x in out[0].values)
    # absent of nothing -> all 1s
    out = tf.absent([], 6)
    assert list(out[0].values) == [1.0] * 6


def test_absent_over_time_multi_ts_combination():
    # absent_over_time(multi-ts) (exec_test.go:1084): the per-series
    # absent rows combine to 1 only where EVERY series was absent
    from victoriametrics_amd import engine
    one = [NAN, NAN, 1.0, 1, 1, 1]     # present early -> absent later
    two = [1.0, 1, 1, 1, NAN, NAN]     # present late  -> absent early
    out = engine.aggregate_absent_over_time([one, two], 6)
    got = out[0].values
    assert math.isnan(got[0]) and math.isnan(got[1])
    assert got[2] == 1.0 and got[3] == 1.0
    assert math.isnan(got[4]) and math.isnan(got[5])


def test_prometheus_buckets_missing_vmrange_exec_pin():
    # exec_test.go:5471: series with an existing `le` pass through
    # untouched; malformed vmranges drop; valid vmranges convert with a
    # zero lower-bound bucket inserted
    t20, t100, t80, t40 = TIME6 / 20, TIME6 / 100, TIME6 / 80, TIME6 / 40
    series = [
        S("xyz", [("foo", "bar"), ("le", "0.2")], t20),
        S("xxx", [("foo", "bar"), ("vmrange", "foobar")], t100),
        S("xxx", [("foo", "bar"), ("vmrange", "30...foobar")], t100),
        S("xxx", [("foo", "bar"), ("vmrange", "30...40")], t100),
        S("yyy", [("foo", "bar"), ("vmrange", "0...900"), ("le", "54")],
          t80),
        S("yyy", [("foo", "bar"), ("vmrange", "900...+Inf"),
                  ("le", "2343")], t40),
    ]
    out = tf.prometheus_buckets(series)
    got = {(s.mn.metric_group.decode(),
            s.mn.get_tag_value("le").decode()): list(s.values)
           for s in out}
    assert got[("xxx", "30")] == [0.0] * 6
    assert got[("xxx", "40")] == [10, 12, 14, 16, 18, 20]
    assert got[("xxx", "+Inf")] == [10, 12, 14, 16, 18, 20]
    assert got[("yyy", "900")] == [12.5, 15, 17.5, 20, 22.5, 25]
    assert got[("yyy", "+Inf")] == [37.5, 45, 52.5, 60, 67.5, 75]
    assert got[("xyz", "0.2")] == [50, 60, 70, 80, 90, 100]
    assert len(got) == 6


def test_prometheus_buckets_zero_and_valid_exec_pins():
    # exec_test.go:5580: an all-zero 0...0 bucket alone yields nothing
    out = tf.prometheus_buckets([S("", [("vmrange", "0...0")],
                                   np.zeros(6))])
    assert out == []
    # exec_test.go:5586 `prometheus_buckets(valid)`
    series = [
        S("xxx", [("foo", "bar"), ("vmrange", "0...0")], np.full(6, 90.0)),
        S("xxx", [("foo", "bar"), ("vmrange", "0...0.2")], TIME6 / 20),
        S("xxx", [("foo", "bar"), ("vmrange", "0.2...40")], TIME6 / 100),
        S("xxx", [("foo", "bar"), ("vmrange", "40...Inf")], TIME6 / 10),
    ]
    out = tf.prometheus_buckets(series)
    got = {s.mn.get_tag_value("le").decode(): list(s.values) for s in out}
    assert got["0"] == [90.0] * 6
    assert got["0.2"] == [140, 150, 160, 170, 180, 190]
    assert got["40"] == [150, 162, 174, 186, 198, 210]
    assert got["Inf"] == [250, 282, 314, 346, 378, 410]
    assert len(got) == 4


def test_interpolate_keep_value_exec_pins():
    # exec_test.go keep_last_value()/keep_next_value()/interpolate() over
    # `time() < 1300 default time() > 1700` = [1000,1200,nan,nan,1800,2000]
    base = TIME6.copy()
    base[(base >= 1300) & (base <= 1700)] = NAN
    out, _ = oracle.tf_apply(100, base.reshape(1, -1).copy())
    assert list(out[0]) == [1000, 1200, 1200, 1200, 1800, 2000]
    out, _ = oracle.tf_apply(101, base.reshape(1, -1).copy())
    assert list(out[0]) == [1000, 1200, 1800, 1800, 1800, 2000]
    out, _ = oracle.tf_apply(102, base.reshape(1, -1).copy())
    assert list(out[0]) == [1000, 1200, 1400, 1600, 1800, 2000]
    # interpolate(tail): trailing NaNs stay NaN
    v = np.where(TIME6 < 1300, TIME6, NAN).reshape(1, -1)
    out, _ = oracle.tf_apply(102, v.copy())
    assert list(out[0][:2]) == [1000, 1200] and np.isnan(out[0][2:]).all()
    # interpolate(head): leading NaNs stay NaN
    v = np.where(TIME6 > 1500, TIME6, NAN).reshape(1, -1)
    out, _ = oracle.tf_apply(102, v.copy())
    assert np.isnan(out[0][:3]).all()
    assert list(out[0][3:]) == [1600, 1800, 2000]
    # interpolate(tail_head_and_middle):
    # [nan,1200,nan,nan,1800,nan] -> [nan,1200,1400,1600,1800,nan]
    v = np.full(6, NAN)
    v[1], v[4] = 1200.0, 1800.0
    out, _ = oracle.tf_apply(102, v.reshape(1, -1))
    assert np.isnan(out[0][0]) and np.isnan(out[0][5])
    assert list(out[0][1:5]) == [1200, 1400, 1600, 1800]


def test_histogram_quantiles_exec_pin(monkeypatch):
    # `histogram_quantiles("phi", 0.2, 0.3, buckets)` exec_test.go — one
    # result per phi labeled phi=%g; values via the oracle quantile
    # (the device kernel is pinned against it in the GPU suite)
    bv = np.stack([np.zeros(6), np.full(6, 100.0), np.full(6, 300.0)])
    les = np.asarray([10.0, 30.0, np.inf])
    off = np.asarray([0, 3], np.uint64)
    for phi, want in ((0.2, 22.0), (0.3, 28.0)):
        out, _, _ = oracle.histogram_quantile(phi, bv, les, off)
        assert list(out[0]) == [want] * 6

    def fake_ht(name, series, arg=None, bounds_label=None):
        assert name == "histogram_quantile"
        out, _, _ = oracle.histogram_quantile(float(arg), bv, les, off)
        return [S("", [("foo", "bar")], out[0])]

    monkeypatch.setattr(tf, "histogram_transform", fake_ht)
    out = tf.histogram_quantiles("phi", [0.2, 0.3], [])
    got = {s.mn.get_tag_value("phi").decode(): list(s.values) for s in out}
    assert got["0.2"] == [22.0] * 6
    assert got["0.3"] == [28.0] * 6
    assert all(s.mn.get_tag_value("foo") == b"bar" for s in out)


def test_start_end_step_exec_pins():
    # `time() - start()` -> [0..1000]; `end() - time()` -> [1000..0];
    # `time() / step()` -> [5..10] on the fixed grid
    ts = (TIME6 * 1000).astype(np.int64)
    start = tf.eval_start(ts)[0].values
    end = tf.eval_end(ts)[0].values
    step = tf.eval_step(ts, 200_000)[0].values
    assert list(TIME6 - start) == [0, 200, 400, 600, 800, 1000]
    assert list(end - TIME6) == [1000, 800, 600, 400, 200, 0]
    assert list(TIME6 / step) == [5, 6, 7, 8, 9, 10]


def test_scalar_multi_timeseries_exec_pin():
    # `scalar(1 or label_set(2, "xx", "foo"))` -> NaN series (two members)
    out = tf.scalar([S("", [], np.ones(6)),
                     S("", [("xx", "foo")], np.full(6, 2.0))])
    assert len(out) == 1
    assert np.isnan(out[0].values).all()


def test_smooth_exponential_exec_pins():
    # exec_test.go: sf=1 -> identity; sf=0 -> frozen at the first value
    # (clamped to [0,1] semantics); sf=0.5 -> EMA
    for sf, want in ((1.0, [1000, 1200, 1400, 1600, 1800, 2000]),
                     (0.0, [1000.0] * 6),
                     (0.5, [1000, 1100, 1250, 1425, 1612.5, 1806.25])):
        out, _ = oracle.tf_apply(123, TIME6.reshape(1, -1).copy(),
                                 arg1=np.full(6, sf))
        assert list(out[0]) == want, sf


def test_range_normalize_exec_pins():
    # `range_normalize(time(), -time())` -> [0..1] and [1..0]
    out, _ = oracle.tf_apply(113, TIME6.reshape(1, -1).copy())
    assert list(out[0]) == [0, 0.2, 0.4, 0.6, 0.8, 1.0]
    out, _ = oracle.tf_apply(113, (-TIME6).reshape(1, -1).copy())
    assert list(out[0]) == [1.0, 0.8, 0.6, 0.4, 0.2, 0]
    # filtered variants keep NaN holes and normalize over the present
    # points only: time() > 1200 < 1800 -> [nan,nan,0,1,nan,nan]
    v = np.where((TIME6 > 1200) & (TIME6 < 1800), TIME6, NAN)
    out, _ = oracle.tf_apply(113, v.reshape(1, -1).copy())
    g = out[0]
    assert np.isnan(g[[0, 1, 4, 5]]).all()
    assert list(g[2:4]) == [0.0, 1.0]
    # -(time() > 1200 < 2000) -> [nan,nan,1,0.5,0,nan]
    v = np.where((TIME6 > 1200) & (TIME6 < 2000), -TIME6, NAN)
    out, _ = oracle.tf_apply(113, v.reshape(1, -1).copy())
    g = out[0]
    assert np.isnan(g[[0, 1, 5]]).all()
    assert list(g[2:5]) == [1.0, 0.5, 0.0]


def test_range_mad_exec_pins():
    # `range_mad(time())` -> 300; filtered -> 100
    out, _ = oracle.tf_apply(119, TIME6.reshape(1, -1).copy())
    assert list(out[0]) == [300.0] * 6
    v = np.where((TIME6 > 1200) & (TIME6 < 1800), TIME6, NAN)
    out, _ = oracle.tf_apply(119, v.reshape(1, -1).copy())
    # the constant MAD is written to EVERY grid point, NaN holes included
    # (the reference's expected array is all-100)
    assert list(out[0]) == [100.0] * 6


def test_range_trim_filtered_exec_pins():
    # `range_trim_outliers(0.5, time() > 1200)` -> [nan,nan,nan,1600,1800,nan]
    v = np.where(TIME6 > 1200, TIME6, NAN)
    out, _ = oracle.tf_apply(120, v.reshape(1, -1).copy(), scalar=0.5)
    g = out[0]
    assert np.isnan(g[[0, 1, 2, 5]]).all()
    assert list(g[3:5]) == [1600.0, 1800.0]
    # `range_trim_spikes(0.2, time() > 1200 <= 1800)` -> only 1600 left
    v = np.where((TIME6 > 1200) & (TIME6 <= 1800), TIME6, NAN)
    out, _ = oracle.tf_apply(121, v.reshape(1, -1).copy(), scalar=0.2)
    g = out[0]
    assert list(g[3:4]) == [1600.0]
    assert np.isnan(np.delete(g, 3)).all()


def test_remove_resets_exec_pins():
    # `remove_resets(abs(1500-time()))` -> [500, 800, 900, 900, 1100, 1300]
    v = np.abs(1500.0 - TIME6).reshape(1, -1)
    out, _ = oracle.tf_apply(124, v.copy())
    assert list(out[0]) == [500, 800, 900, 900, 1100, 1300]
    # `remove_resets(sum(time(), time()/5 < 300))`: the summed series
    # [1200,1440,1680,1600,1800,2000] has one reset at idx 3
    partial = np.where(TIME6 / 5 < 300, TIME6 / 5, NAN)
    summed = TIME6 + np.nan_to_num(partial)
    out, _ = oracle.tf_apply(124, summed.reshape(1, -1).copy())
    assert list(out[0]) == [1200, 1440, 1680, 1680, 1880, 2080]


def test_running_avg_exec_pins():
    # `running_avg(time())` -> prefix means [1000..1500]
    out, _ = oracle.tf_apply(106, TIME6.reshape(1, -1).copy())
    assert list(out[0]) == [1000, 1100, 1200, 1300, 1400, 1500]
    # `running_avg(time() > 1200 < 1800)` -> NaNs skipped, mean over the
    # seen present points only
    v = np.where((TIME6 > 1200) & (TIME6 < 1800), TIME6, NAN)
    out, _ = oracle.tf_apply(106, v.reshape(1, -1).copy())
    g = out[0]
    assert np.isnan(g[:2]).all()
    assert list(g[2:]) == [1400, 1500, 1500, 1500]


def test_get_num_prefix_reference_vectors():
    # TestGetNumPrefix (transform_test.go:283): every vector verbatim;
    # non-empty prefixes must parse as floats
    cases = [
        ("", ""), ("foo", ""), ("-", ""), (".", ""), ("-.", ""),
        ("+..", ""), ("1", "1"), ("12", "12"), ("1foo", "1"),
        ("-123", "-123"), ("-123bar", "-123"), ("+123", "+123"),
        ("+123.", "+123."), ("+123..", "+123."), ("+123.-", "+123."),
        ("12.34..", "12.34"), ("-12.34..", "-12.34"), ("-12.-34..", "-12."),
    ]
    for s_in, want in cases:
        got = tf._num_prefix(s_in)
        assert got == want, (s_in, got, want)
        if got:
            float(got)  # must parse


def test_numeric_less_reference_vectors():
    # TestNumericLess (transform_test.go:317): every vector verbatim,
    # including the 309-digit no-panic regression (GHSA-9g98-8jgr-x2vv)
    big = "9" * 309
    cases = [
        ("", "", False), ("", "321", True), ("321", "", False),
        ("", "abc", True), ("abc", "", False), ("foo", "123", False),
        ("123", "foo", True), ("123", "321", True), ("321", "123", False),
        ("123", "123", False), ("a", "b", True), ("b", "a", False),
        ("a", "a", False), ("foo123", "foo", False), ("foo", "foo123", True),
        ("foo", "foo", False), ("123foo", "123bar", False),
        ("123bar", "123foo", True), ("123bar", "123bar", False),
        ("1:0:0", "1:0:2", True), ("1:0:15", "1:0:2", False),
        ("0", "00", False), ("aa", "ab", True), ("ab", "abc", True),
        ("a0001", "a0000001", False), ("a10", "abcdefgh2", True),
        ("a1b", "a01b", False), ("a001b01", "a01b001", False),
        ("a01b001", "a001b01", False), ("a1", "a1x", True),
        ("1b", "1ax", False), ("082", "83", True), ("083a", "9a", False),
        ("083a", "94a", True), ("-123", "123", True),
        ("-123", "+123", True), ("-123", "-123", False),
        ("123", "-123", False), ("12.9", "12.56", False),
        ("12.56", "12.9", True), ("12.9", "12.9", False),
        (big, "1", False), ("1", big, True), (big, big, False),
        ("-" + big, big, True), (big, "-" + big, False),
    ]
    for a, b, want in cases:
        assert tf.numeric_less(a, b) == want, (a, b, want)


def test_go_expand_template_semantics():
    # Go regexp.Expand rules (regexp/regexp.go): longest-name refs,
    # unknown groups -> "", $$ literal, ${name} explicit
    import re as _re
    m = _re.match(r"^(?:(a+)(b+))$", "aabbb")
    ge = tf._go_expand
    assert ge(m, "x${1}y") == "xaay"
    assert ge(m, "x$1y") == "x"          # $1y reads group "1y" -> absent
    assert ge(m, "$1-$2") == "aa-bbb"
    assert ge(m, "$$1") == "$1"
    assert ge(m, "$9") == ""             # out-of-range -> empty
    assert ge(m, "${9}z") == "z"
    assert ge(m, "lone$") == "lone$"
    assert ge(m, "${unclosed") == "${unclosed"
    m2 = _re.match(r"^(?:(?P<word>\w+))$", "hello")
    assert ge(m2, "<$word>") == "<hello>"
    assert ge(m2, "<$wordx>") == "<>"    # longest-run name "wordx" absent
    assert ge(m2, "<${word}x>") == "<hellox>"


def test_histogram_normal_bucket_count_exec_pins():
    # exec_test.go:5018-5165 normal-bucket-count family via the oracle
    # walk (the device kernel is pinned against it in the GPU suite):
    # quantile(0.2) -> 22 with bounds [10, 30]; share(35) -> 1/3;
    # share(22) -> 0.2 with bounds [0, 1/3]; fraction(22,35) -> 2/15
    bv = np.stack([np.zeros(6), np.full(6, 100.0), np.full(6, 300.0)])
    les = np.asarray([10.0, 30.0, np.inf])
    off = np.asarray([0, 3], np.uint64)
    out, lo, hi = oracle.histogram_quantile(0.2, bv, les, off, bounds=True)
    assert list(out[0]) == [22.0] * 6
    assert list(lo[0]) == [10.0] * 6
    assert list(hi[0]) == [30.0] * 6
    share35 = oracle.histogram_share(np.full(6, 35.0), bv, les, off)
    out, lo, hi = share35[0], share35[1], share35[2]
    np.testing.assert_allclose(out[0], [1.0 / 3] * 6, rtol=1e-15)
    share22 = oracle.histogram_share(np.full(6, 22.0), bv, les, off)
    out, lo, hi = share22[0], share22[1], share22[2]
    np.testing.assert_allclose(out[0], [0.2] * 6, rtol=1e-15)
    assert list(lo[0]) == [0.0] * 6
    np.testing.assert_allclose(hi[0], [1.0 / 3] * 6, rtol=1e-15)
    frac = share35[0][0] - share22[0][0]
    np.testing.assert_allclose(frac, [0.1333333333333333] * 6, rtol=1e-14)


def test_histogram_quantile_duplicate_le_exec_pin():
    # `histogram_quantile(0.6, ... le=5 or le=5.0 or le=6.0 or +Inf)`
    # exec_test.go — equal-le buckets merge by SUM (mergeSameLE) before
    # the walk; round(_, 0.1) -> 4.7
    from victoriametrics_amd.decimal import go_round
    series = [
        S("", [("foo", "bar"), ("le", "5")], np.full(6, 90.0)),
        S("", [("foo", "bar"), ("le", "5.0")], np.full(6, 100.0)),
        S("", [("foo", "bar"), ("le", "6.0")], np.full(6, 200.0)),
        S("", [("foo", "bar"), ("le", "+Inf")], np.full(6, 300.0)),
    ]
    groups = tf.group_le_timeseries(series)
    assert len(groups) == 1
    xss = sorted(next(iter(groups.values())), key=lambda p: p[0])
    merged = tf._merge_same_le(xss)
    les = np.asarray([le for le, _ in merged])
    bv = np.stack([s.values for _, s in merged])
    np.testing.assert_array_equal(les, [5.0, 6.0, np.inf])
    np.testing.assert_array_equal(bv[:, 0], [190.0, 200.0, 300.0])
    off = np.asarray([0, len(merged)], np.uint64)
    out, _, _ = oracle.histogram_quantile(0.6, bv, les, off)
    got = go_round(out[0] * 10.0) / 10.0
    np.testing.assert_array_equal(got, [4.7] * 6)


def test_round_to_decimal_digits_reference_vectors():
    # TestRoundToDecimalDigits (lib/decimal/decimal_test.go:10-41)
    # verbatim, including the StaleNaN pass-through
    from victoriametrics_amd import decimal as vmd
    stale = np.frombuffer(np.int64(0x7FF0000000000002).tobytes(),
                          np.float64)[0]
    cases = [
        (12.34, 0, 12.0), (12.57, 0, 13.0), (-1.578, 2, -1.58),
        (-1.578, 3, -1.578), (1234.0, -2, 1200.0), (1235.0, -1, 1240.0),
        (1234.0, 0, 1234.0), (1234.6, 0, 1235.0),
        (123.4e-99, 99, 123e-99),
    ]
    for v, digits, want in cases:
        got = vmd.round_to_decimal_digits(np.asarray([v]), digits)[0]
        assert got == want, (v, digits, got, want)
    got = vmd.round_to_decimal_digits(np.asarray([math.nan]), 10)[0]
    assert math.isnan(got) and np.isnan(got)
    got = vmd.round_to_decimal_digits(np.asarray([stale]), 10)[0]
    assert np.asarray([got]).view(np.int64)[0] == 0x7FF0000000000002


def test_trig_exec_pins():
    # exec_test.go:1334-1444 — trig family over pi()*(2000-time())/1000
    # or (2000-time())/1000.  The expected arrays encode Go's math
    # package (its OWN soft-float sin/cos implementations, not libm) —
    # near sin(pi) they differ from glibc by an ulp, so compare at the
    # same 1e-14 relative tolerance the GPU exec pins use
    x = (2000.0 - TIME6) / 1000.0
    pix = np.pi * x

    def tfm(fid, v):
        out, _ = oracle.tf_apply(fid, np.asarray(v).reshape(1, -1).copy())
        return out.ravel()

    def close(got, want):
        np.testing.assert_allclose(got, want, rtol=1e-14, atol=1e-300)

    close(tfm(8, pix),                      # sin
          [1.2246467991473515e-16, 0.5877852522924732, 0.9510565162951536,
           0.9510565162951535, 0.5877852522924731, 0])
    close(tfm(14, pix),                     # sinh
          [11.548739357257748, 6.132140673514712, 3.217113080357038,
           1.6144880404748523, 0.6704839982471175, 0])
    close(tfm(11, x),                       # asin
          [1.5707963267948966, 0.9272952180016123, 0.6435011087932843,
           0.41151684606748806, 0.20135792079033082, 0])
    close(tfm(17, tfm(14, x)),              # asinh(sinh)
          [1, 0.8000000000000002, 0.6, 0.4000000000000001, 0.2, 0])
    close(tfm(13, x),                       # atan
          [0.7853981633974483, 0.6747409422235526, 0.5404195002705842,
           0.3805063771123649, 0.19739555984988078, 0])
    close(tfm(19, tfm(16, x)),              # atanh(tanh)
          [1, 0.8000000000000002, 0.6, 0.4000000000000001, 0.2, 0])
    close(tfm(9, pix),                      # cos
          [-1, -0.8090169943749475, -0.30901699437494734,
           0.30901699437494745, 0.8090169943749473, 1])
    close(tfm(12, x),                       # acos
          [0, 0.6435011087932843, 0.9272952180016123, 1.1592794807274085,
           1.3694384060045657, 1.5707963267948966])
    # `time() atan2 time()/10` :1389 — atan2 shares the */ precedence
    # level (left-assoc), so this is (time() atan2 time()) / 10
    got = oracle.binop_apply("atan2", TIME6.copy(), TIME6.copy())
    close(np.asarray(got).ravel() / 10.0, [0.07853981633974483] * 6)
    # pi() :1334
    assert tf.eval_pi((TIME6 * 1000).astype(np.int64))[0].values[0] == \
        3.141592653589793
