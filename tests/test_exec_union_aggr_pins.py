"""union / limitk / any / group pins transcribed from TestExecSuccess
(app/vmselect/promql/exec_test.go:6940-7070, :9311-9525): expected
MetricNames and value arrays verbatim.  Host-side selection logic; the
group() reduction array is pinned through the oracle reducer (test infra).
"""
import math

import numpy as np

import oracle
from victoriametrics_amd import aggregate as agg
from victoriametrics_amd import transform as tf
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

NAN = math.nan
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
        np.testing.assert_array_equal(s.values,
                                      np.asarray(values, np.float64))


def test_union_exec():
    # :9311 union() -> empty; :9317 union(1) -> the scalar series
    assert tf.union([]) == []
    out = tf.union([[S(values=1.0)]])
    chk(out[0], "", [], [1.0] * 6)
    # :9339 identical labels dedup to the FIRST occurrence
    out = tf.union([[S(tags=[("foo", "bar")], values=1.0)],
                    [S(tags=[("foo", "bar")], values=2.0)]])
    assert len(out) == 1
    chk(out[0], "", [("foo", "bar")], [1.0] * 6)
    # :9369 names participate in identity
    out = tf.union([[S("xx", [("foo", "bar")], 1.0)],
                    [S("xx", [("foo", "bar")], 2.0)]])
    assert len(out) == 1
    chk(out[0], "xx", [("foo", "bar")], [1.0] * 6)
    # :9401 different names -> both kept
    out = tf.union([[S("xx", [("foo", "bar")], 1.0)],
                    [S("yy", [("foo", "bar")], 2.0)]])
    assert len(out) == 2
    chk(out[0], "xx", [("foo", "bar")], [1.0] * 6)
    chk(out[1], "yy", [("foo", "bar")], [2.0] * 6)
    # :9482 more than two args, later lists appended after dedup
    out = tf.union([
        [S("xx", [("foo", "bar")], 1.0)],
        [S("yy", [("foo", "bar")], 2.0)],
        [S(tags=[("qwe", "123")]), S("rt", [], 3.0)],
    ])
    assert len(out) == 4
    chk(out[2], "", [("qwe", "123")], TIME)
    chk(out[3], "rt", [], [3.0] * 6)


def _limitk_input():
    return [S(tags=[("foo", "bar")], values=10.0),
            S(tags=[("baz", "sss")], values=TIME / 150)]


def test_limitk_exec():
    # :6940 limitk(-1) -> empty
    assert agg.aggregate("limitk", _limitk_input(), arg=-1) == []
    # :6946 limitk(1): the xxhash-smaller name wins — the reference's own
    # expected output says that is {foo="bar"}
    out = agg.aggregate("limitk",
                        [S(tags=[("foo", "bar")], values=10.0),
                         S(tags=[("xbaz", "sss")], values=TIME / 150)],
                        arg=1)
    assert len(out) == 1
    chk(out[0], "", [("foo", "bar")], [10.0] * 6)
    # :6961 limitk(10) and :6985 limitk(inf): both keep everything
    for k in (10, math.inf):
        out = tf.sort_series(agg.aggregate("limitk", _limitk_input(),
                                           arg=k))
        assert len(out) == 2
        # sort() compares from the LAST point backwards: 10 < 13.33
        chk(out[0], "", [("foo", "bar")], [10.0] * 6)
        chk(out[1], "", [("baz", "sss")],
            [6.666666666666667, 8, 9.333333333333334, 10.666666666666666,
             12, 13.333333333333334])


def test_any_exec():
    # :7009 any(): one series, ORIGINAL full name kept
    out = agg.aggregate("any",
                        [S("x", [("foo", "bar")], 10.0),
                         S("y", [("baz", "sss")], TIME / 150)])
    assert len(out) == 1
    chk(out[0], "x", [("foo", "bar")], [10.0] * 6)
    # :7025 any(empty-series) -> empty (removeEmptySeries runs first)
    out = agg.aggregate("any", [S(tags=[("foo", "bar")],
                                  values=np.full(6, NAN))])
    assert out == []


def test_group_exec():
    # :7031/:7051 group() by (test) / without (point): value 1 at every
    # point with any sample; group name carries only the `test` tag
    members = [
        S("data", [("test", "three samples"), ("point", "a")], 5.0),
        S("data", [("test", "three samples"), ("point", "b")], 6.0),
        S("data", [("test", "three samples"), ("point", "c")], 7.0),
    ]
    for op, args in (("by", ["test"]), ("without", ["point"])):
        groups = agg.prepare_series([s.copy_shallow() for s in members], op, args)
        assert len(groups) == 1
        gmn, mem = groups[0]
        assert gmn.metric_group == b""
        assert sorted(gmn.tags) == [(b"test", b"three samples")], op
        assert len(mem) == 3
    # the reduction array itself, via the oracle reducer (test infra)
    v = np.stack([s.values for s in members])
    gr = np.arange(3, dtype=np.uint32)
    go = np.asarray([0, 3], np.uint64)
    out = oracle.colagg("group", v, gr, go)
    np.testing.assert_array_equal(out[0], np.ones(6))


def test_sum_over_union_exec():
    # `sum((1, 2, 3))` :5882 -> 6: the all-scalar union keeps every
    # member; `sum((alias(1,"foo"), alias(2,"foo"), alias(3,"foo")))`
    # :5893 -> 1: named duplicates dedup to the FIRST in the union
    scalars = tf.union([[S(values=1.0)], [S(values=2.0)],
                        [S(values=3.0)]])
    assert len(scalars) == 3
    v = np.stack([s.values for s in scalars])
    gr = np.arange(3, dtype=np.uint32)
    go = np.asarray([0, 3], np.uint64)
    out = oracle.colagg("sum", v, gr, go)
    np.testing.assert_array_equal(out[0], [6.0] * 6)
    named = tf.union([[S("foo", values=1.0)], [S("foo", values=2.0)],
                      [S("foo", values=3.0)]])
    assert len(named) == 1
    np.testing.assert_array_equal(named[0].values, [1.0] * 6)
