"""Label-function pins transcribed from TestExecSuccess
(app/vmselect/promql/exec_test.go:1565-2520): each case hand-composes the
query's series construction (label_set literals over the fixed 6-point
grid) and checks the reference's expected MetricName + Values verbatim.
Tags are compared in canonical sorted order — the order the reference's
output marshaling produces.  Pure host logic: runs in the CPU suite.
"""
import math

import numpy as np

from victoriametrics_amd import transform as tf
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def S(name=b"", tags=(), values=None):
    v = TIME.copy() if values is None else np.asarray(values, np.float64)
    if v.ndim == 0 or v.size == 1:
        v = np.full(6, float(v))
    return Series(MetricName(name, list(tags)), v.copy())


def chk(s, name, tags, values=None):
    assert s.mn.metric_group == MetricName._b(name), s.mn.metric_group
    got = sorted(s.mn.tags)
    want = sorted((MetricName._b(k), MetricName._b(v)) for k, v in tags)
    assert got == want, (got, want)
    if values is not None:
        g = np.asarray(s.values, np.float64)
        w = np.asarray(values, np.float64)
        assert (np.isnan(g) == np.isnan(w)).all(), (g, w)
        np.testing.assert_array_equal(g[~np.isnan(w)], w[~np.isnan(w)])


def test_label_set_variants():
    # :1565 label_set(time(), "tagname", "tagvalue")
    out = tf.label_set([S()], [("tagname", "tagvalue")])
    chk(out[0], "", [("tagname", "tagvalue")], TIME)
    # :1580 label_set(time(), "__name__", "foobar")
    out = tf.label_set([S()], [("__name__", "foobar")])
    chk(out[0], "foobar", [], TIME)
    # :1592 nested: name then tag
    out = tf.label_set(tf.label_set([S()], [("__name__", "foobar")]),
                       [("tagname", "tagvalue")])
    chk(out[0], "foobar", [("tagname", "tagvalue")], TIME)
    # :1611 del_metricname — setting "" clears the group
    out = tf.label_set(tf.label_set([S()], [("__name__", "foobar")]),
                       [("__name__", "")])
    chk(out[0], "", [], TIME)
    # :1625 del_tag
    out = tf.label_set(tf.label_set([S()], [("tagname", "foobar")]),
                       [("tagname", "")])
    chk(out[0], "", [], TIME)
    # :1639 multi
    out = tf.label_set([S(values=TIME + 100)],
                       [("t1", "v1"), ("t2", "v2"), ("__name__", "v3")])
    chk(out[0], "v3", [("t1", "v1"), ("t2", "v2")], TIME + 100)


def test_label_map_match():
    # :1661 label_map(5 series, "label", v1->foo, v2->bar, ""->qwe, v4->"")
    xs = [
        S(tags=[("label", "v1")]),
        S(tags=[("label", "v2")], values=TIME + 100),
        S(tags=[("label", "v3")], values=TIME + 200),
        S(tags=[("x", "y")], values=TIME + 300),
        S(tags=[("label", "v4")], values=TIME + 400),
    ]
    out = tf.label_map(xs, "label",
                       {"v1": "foo", "v2": "bar", "": "qwe", "v4": ""})
    out = tf.sort_series(out)
    chk(out[0], "", [("label", "foo")], TIME)
    chk(out[1], "", [("label", "bar")], TIME + 100)
    chk(out[2], "", [("label", "v3")], TIME + 200)
    chk(out[3], "", [("label", "qwe"), ("x", "y")], TIME + 300)
    chk(out[4], "", [], TIME + 400)


def test_label_case_folding():
    # :1720 / :1748 — only the named labels fold; absent labels ignored
    out = tf.label_uppercase(
        [S(tags=[("foo", "bAr"), ("XXx", "yyy"), ("zzz", "abc")])],
        ["foo", "XXx", "aaa"])
    chk(out[0], "", [("XXx", "YYY"), ("foo", "BAR"), ("zzz", "abc")], TIME)
    out = tf.label_lowercase(
        [S(tags=[("foo", "bAr"), ("XXx", "yyy"), ("zzz", "aBc")])],
        ["foo", "XXx", "aaa"])
    chk(out[0], "", [("XXx", "yyy"), ("foo", "bar"), ("zzz", "aBc")], TIME)


def test_label_copy_move_variants():
    # :1776-:2040 — every copy/move aliasing case
    out = tf.label_copy([S(tags=[("tagname", "foobar")])],
                        [("tagname", "xxx")])
    chk(out[0], "", [("tagname", "foobar"), ("xxx", "foobar")])
    out = tf.label_move([S(tags=[("tagname", "foobar")])],
                        [("tagname", "xxx")])
    chk(out[0], "", [("xxx", "foobar")])
    # same_tag: no-op for both
    for fn in (tf.label_copy, tf.label_move):
        out = fn([S(tags=[("tagname", "foobar")])],
                 [("tagname", "tagname")])
        chk(out[0], "", [("tagname", "foobar")])
    # nonexisting src: no-op
    for fn in (tf.label_copy, tf.label_move):
        out = fn([S(tags=[("tagname", "foobar")])],
                 [("non-existing-tag", "tagname")])
        chk(out[0], "", [("tagname", "foobar")])
    # existing dst gets overwritten
    out = tf.label_copy([S(tags=[("tagname", "foobar"), ("xx", "yy")])],
                        [("xx", "tagname")])
    chk(out[0], "", [("tagname", "yy"), ("xx", "yy")])
    out = tf.label_move([S(tags=[("tagname", "foobar"), ("xx", "yy")])],
                        [("xx", "tagname")])
    chk(out[0], "", [("tagname", "yy")])
    # from metric group
    out = tf.label_copy([S("yy", [("tagname", "foobar")])],
                        [("__name__", "aa")])
    chk(out[0], "yy", [("aa", "yy"), ("tagname", "foobar")])
    out = tf.label_move([S("yy", [("tagname", "foobar")])],
                        [("__name__", "aa")])
    chk(out[0], "", [("aa", "yy"), ("tagname", "foobar")])
    # to metric group
    out = tf.label_copy([S(tags=[("tagname", "foobar")])],
                        [("tagname", "__name__")])
    chk(out[0], "foobar", [("tagname", "foobar")])
    out = tf.label_move([S(tags=[("tagname", "foobar")])],
                        [("tagname", "__name__")])
    chk(out[0], "foobar", [])


def test_labels_equal_exec():
    # :2040 labels_equal((3 series), "instance", "host")
    xs = [
        S(tags=[("instance", "qwe"), ("host", "rty")], values=10),
        S(tags=[("instance", "qwe"), ("host", "qwe")], values=20),
        S(tags=[("aaa", "bbb"), ("instance", "foo"), ("host", "foo")],
          values=30),
    ]
    out = tf.sort_series(tf.labels_equal(xs, ["instance", "host"]))
    assert len(out) == 2
    chk(out[0], "", [("host", "qwe"), ("instance", "qwe")], [20.0] * 6)
    chk(out[1], "",
        [("aaa", "bbb"), ("host", "foo"), ("instance", "foo")], [30.0] * 6)


def test_label_keep_del_exec():
    # :2172-:2256
    out = tf.label_keep([S()], ["foo", "bar"])
    chk(out[0], "", [])
    mk = lambda: S("xxx", [("foo", "bar"), ("q", "we")])
    out = tf.label_keep([mk()], ["foo", "nonexisting-label"])
    chk(out[0], "", [("foo", "bar")])
    out = tf.label_keep([mk()], ["nonexisting-label", "__name__"])
    chk(out[0], "xxx", [])
    out = tf.label_del([S()], ["foo", "bar"])
    chk(out[0], "", [])
    out = tf.label_del([mk()], ["foo", "nonexisting-label"])
    chk(out[0], "xxx", [("q", "we")])
    out = tf.label_del([mk()], ["nonexisting-label", "__name__"])
    chk(out[0], "", [("foo", "bar"), ("q", "we")])


def test_label_join_exec():
    # :2258 empty: all srcs missing -> empty join -> tag removed
    out = tf.label_join([S()], "tt", "(sep)", ["BAR"])
    chk(out[0], "", [])
    # :2269 two missing srcs joined by "(sep)" -> the separator itself
    out = tf.label_join([S()], "tt", "(sep)", ["foo", "BAR"])
    chk(out[0], "", [("tt", "(sep)")])
    # :2284 into __name__, three srcs -> "(sep)(sep)"
    out = tf.label_join([S()], "__name__", "(sep)", ["foo", "BAR", ""])
    chk(out[0], "(sep)(sep)", [])
    # :2296 nested join reading the produced __name__
    out = tf.label_join(
        tf.label_join([S()], "__name__", "(sep)", ["foo", "BAR"]),
        "xxx", ",", ["foobar", "__name__"])
    chk(out[0], "(sep)", [("xxx", ",(sep)")])
    # :2312 dst == one of the srcs: reads the OLD value
    out = tf.label_join(
        tf.label_join([S()], "bar", "sep1", ["a", "b"]),
        "bar", "sep2", ["a", "bar"])
    chk(out[0], "", [("bar", "sep2sep1")])


def test_label_transform_exec():
    # :2367 mismatch -> unchanged; :2378 unanchored ReplaceAll
    out = tf.label_transform([S()], "__name__", "foobar", "xx")
    chk(out[0], "", [])
    out = tf.label_transform([S(tags=[("foo", "a.bar.baz")])],
                             "foo", r"\.", "-")
    chk(out[0], "", [("foo", "a-bar-baz")])


def test_label_replace_exec():
    # :2395 anchored ".+" vs missing src -> no change
    out = tf.label_replace([S()], "__name__", "x${1}y", "foo", ".+")
    chk(out[0], "", [])
    # :2406 "" matches the missing src
    out = tf.label_replace([S()], "foo", "x", "bar", "")
    chk(out[0], "", [("foo", "x")])
    # :2423 mismatching regex vs missing src -> no change
    out = tf.label_replace([S()], "foo", "x", "bar", "y")
    chk(out[0], "", [])
    # :2434 anchored "bar(.+)" does NOT match "foobar"
    out = tf.label_replace([S(tags=[("foo", "foobar")])],
                           "__name__", "x${1}y", "foo", "bar(.+)")
    chk(out[0], "", [("foo", "foobar")])
    # :2449 ".*" matches empty -> __name__ = "xy"
    out = tf.label_replace([S()], "__name__", "x${1}y", "foo", ".*")
    chk(out[0], "xy", [])
    # :2460 nested three-deep replace
    out = tf.label_replace([S()], "__name__", "x${1}y", "foo", ".*")
    out = tf.label_replace(out, "xxx", "foo${1}bar(${1})", "__name__",
                           "(.+)")
    out = tf.label_replace(out, "xxx", "AA$1", "xxx", "foox(.+)")
    chk(out[0], "xy", [("xxx", "AAybar(xy)")], TIME)


def test_label_value_exec():
    # :2327 sort(x + label_value(x, "foo")) — the label parses to the
    # series value (NaN if unparsable), name reset, then x + that
    def x():
        return [
            S("aaa", [("foo", "123.456")],
              np.where(TIME > 1500, TIME, np.nan)),
            S("bbb", [("foo", "bar")], -TIME),
            S("bxs", [], -TIME),
            S("", [("foo", "45"), ("bar", "xs")], -TIME),
        ]
    lv = tf.label_value(x(), "foo")
    # parsed constants land only on non-NaN points
    want = [123.456, math.nan, math.nan, 45.0]
    for s, w in zip(lv, want):
        assert s.mn.metric_group == b""
        g = s.values
        if math.isnan(w):
            assert np.isnan(g).all()
        else:
            assert (g[~np.isnan(g)] == w).all()
    # x + label_value(x): pairwise on identical tag sets; all-NaN dropped
    from victoriametrics_amd.binary_op import BinOpSpec
    from test_binary_op import _eval  # oracle-backed CPU apply (test infra)
    out = _eval(BinOpSpec("+"), x(), lv)
    out = [s for s in out if not np.isnan(s.values).all()]
    out = tf.sort_series(out)
    assert len(out) == 2
    chk(out[0], "", [("bar", "xs"), ("foo", "45")],
        [-955, -1155, -1355, -1555, -1755, -1955])
    chk(out[1], "", [("foo", "123.456")],
        [np.nan, np.nan, np.nan, 1723.456, 1923.456, 2123.456])


def _le(val, le, extra=()):
    return S("metric", [("le", le)] + list(extra), float(val))


def test_buckets_limit_exec():
    # exec_test.go:5236-5470 — the four buckets_limit cases, expected
    # surviving buckets verbatim (sorted ascending by values, like sort())
    # trim_zero_preserve_empty_when_limit_not_reached: zero buckets trim
    # first, then inner buckets with the least delta
    out = tf.sort_series(tf.buckets_limit(3, [
        _le(36, "+Inf"), _le(36, "25"), _le(36, "21"), _le(36, "19"),
        _le(36, "18"), _le(36, "17"), _le(36, "16"), _le(27, "12"),
        _le(14, "9"), _le(0, "6"), _le(0, "1")]))
    assert [(s.mn.get_tag_value(b"le"), s.values[0]) for s in out] == [
        (b"9", 14.0), (b"12", 27.0), (b"16", 36.0)]
    # trim_zero: limit not reached after zero-trim -> keep left zeros
    out = tf.sort_series(tf.buckets_limit(5, [
        _le(36, "18"), _le(36, "17"), _le(36, "16"), _le(27, "12"),
        _le(14, "9"), _le(0, "6"), _le(0, "1")]))
    assert [(s.mn.get_tag_value(b"le"), s.values[0]) for s in out] == [
        (b"1", 0.0), (b"6", 0.0), (b"9", 14.0), (b"12", 27.0),
        (b"16", 36.0)]
    # unused: under the limit, everything survives (tags preserved)
    out = tf.sort_series(tf.buckets_limit(5, [
        _le(100, "inf", [("x", "y")]), _le(50, "120", [("x", "y")])]))
    assert len(out) == 2
    chk(out[0], "metric", [("le", "120"), ("x", "y")], [50.0] * 6)
    chk(out[1], "metric", [("le", "inf"), ("x", "y")], [100.0] * 6)
    # used: limit 2 still keeps 3 (first/last protected + biggest deltas)
    out = tf.sort_series(tf.buckets_limit(2, [
        _le(100, "inf", [("x", "y")]), _le(98, "300", [("x", "y")]),
        _le(52, "200", [("x", "y")]), _le(50, "120", [("x", "y")]),
        _le(20, "70", [("x", "y")]), _le(10, "30", [("x", "y")]),
        _le(9, "10", [("x", "y")])]))
    assert [(s.mn.get_tag_value(b"le"), s.values[0]) for s in out] == [
        (b"10", 9.0), (b"300", 98.0), (b"inf", 100.0)]


def test_sort_family_exec():
    # :2691-2811 sort()/sort_desc()/sort_by_label{,_desc}/multiple_labels
    out = tf.sort_series([S(values=2.0), S(tags=[("xx", "foo")],
                                           values=1.0)])
    chk(out[0], "", [("xx", "foo")], [1.0] * 6)
    chk(out[1], "", [], [2.0] * 6)
    out = tf.sort_series([S(values=1.0), S(tags=[("xx", "foo")],
                                           values=2.0)], desc=True)
    chk(out[0], "", [("xx", "foo")], [2.0] * 6)
    chk(out[1], "", [], [1.0] * 6)
    # `two_timeseries` :2621 — sort_desc over time() and the constant 2
    out = tf.sort_series([S(), S(tags=[("xx", "foo")], values=2.0)],
                         desc=True)
    chk(out[0], "", [], TIME)
    chk(out[1], "", [("xx", "foo")], [2.0] * 6)
    # sort_by_label on __name__
    out = tf.sort_by_label([S("foo", values=1.0), S("bar", values=2.0)],
                           ["__name__"])
    chk(out[0], "bar", [], [2.0] * 6)
    chk(out[1], "foo", [], [1.0] * 6)
    out = tf.sort_by_label([S("foo", values=1.0), S("bar", values=2.0)],
                           ["__name__"], desc=True)
    chk(out[0], "foo", [], [1.0] * 6)
    chk(out[1], "bar", [], [2.0] * 6)
    # multiple labels: equal y, then x decides
    out = tf.sort_by_label(
        [S(tags=[("x", "b"), ("y", "aa")], values=1.0),
         S(tags=[("x", "a"), ("y", "aa")], values=2.0)], ["y", "x"])
    chk(out[0], "", [("x", "a"), ("y", "aa")], [2.0] * 6)
    chk(out[1], "", [("x", "b"), ("y", "aa")], [1.0] * 6)


def test_label_match_mismatch_name_exec():
    # :2483/:2499 — regex filter on __name__ keeps/drops whole series
    xs = [S("foo", [], TIME), S("bar", [], TIME * 2)]
    out = tf.label_match([s for s in xs], "__name__", "f.+")
    assert len(out) == 1
    chk(out[0], "foo", [], TIME)
    out = tf.label_match([S("foo", [], TIME), S("bar", [], TIME * 2)],
                         "__name__", "f.+", negate=True)
    assert len(out) == 1
    chk(out[0], "bar", [], TIME * 2)


def test_label_graphite_group_multi_ids_exec():
    # :2515 — group ids 1 and 3: out-of-range components become "",
    # joined with "."
    xs = [S("foo.bar.baz", [], np.ones(6)),
          S("abc", [], np.full(6, 2.0)),
          S("a.xx.zz.asd", [("qwe", "rty")], np.full(6, 3.0))]
    out = tf.sort_series(tf.label_graphite_group(xs, [1, 3]))
    chk(out[0], "bar.", [], [1.0] * 6)
    chk(out[1], ".", [], [2.0] * 6)
    chk(out[2], "xx.asd", [("qwe", "rty")], [3.0] * 6)
