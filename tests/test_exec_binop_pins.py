"""Binary-operator pins transcribed from the reference's TestExecSuccess
(app/vmselect/promql/exec_test.go; the t.Run names are kept in each test's
comment with the exec_test.go line).  Grid start=1000e3 end=2000e3
step=200e3; `time()` = [1000..2000].  Expected arrays/labels are the
reference's own outputs, copied verbatim; the per-point math is delegated
to the oracle (test infra) exactly as in test_binary_op.py, so these pin
the HOST matching layers (createTimeseriesMapByTagSet, adjustBinaryOpTags,
group_left/right joins, the `or` merge walk, removeEmptySeries at the exec
tail)."""
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd.binary_op import (BinOpSpec, Series,
                                           binary_op_eval,
                                           remove_empty_series)
from victoriametrics_amd.metric_name import MetricName

from test_binary_op import _eval  # oracle-backed apply/mask/or fns

NAN = math.nan
TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def S(name, tags, values):
    return Series(MetricName(name, tags),
                  np.asarray(values, np.float64).copy())


def t_series(extra=0.0):
    return S("", [], TIME + extra)


def scalar(v):
    return S("", [], np.full(6, float(v)))


def by_tag(out, key):
    got = {}
    for s in out:
        got[s.mn.get_tag_value(key) or b""] = s
    return got


def eq(values, expected):
    a = np.asarray(values, np.float64)
    b = np.asarray(expected, np.float64)
    an, bn = np.isnan(a), np.isnan(b)
    assert (an == bn).all(), (a, b)
    np.testing.assert_array_equal(a[~an], b[~bn])


# ---------------------------------------------------------------------
# comparisons (exec_test.go:2834-2980)
# ---------------------------------------------------------------------

def test_time_gt_bool_scalar():
    # `time() >bool 1234` :2834
    out = _eval(BinOpSpec(">", bool_modifier=True), [t_series()],
                [scalar(1234)])
    eq(out[0].values, [0, 0, 1, 1, 1, 1])


def test_nan_gt_bool_scalar():
    # `(time() > 1234) >bool 1450` :2845 — NaNs from the inner filter stay
    inner = np.where(TIME > 1234, TIME, NAN)
    out = _eval(BinOpSpec(">", bool_modifier=True), [S("", [], inner)],
                [scalar(1450)])
    eq(out[0].values, [NAN, NAN, 0, 1, 1, 1])


def test_nan_ne_bool_scalar():
    # `(time() > 1234) !=bool 1400` :2856.  (The reference's companion case
    # `1400 !=bool (time() > 1234)` :2867 relies on metricsql canonicalizing
    # scalar-op-vector into vector-op-scalar BEFORE evaluation — parser-side
    # work that stays in the host Go layer per the §8b seam, so the engine
    # is handed the canonical operand order.)
    inner = np.where(TIME > 1234, TIME, NAN)
    out = _eval(BinOpSpec("!=", bool_modifier=True), [S("", [], inner)],
                [scalar(1400)])
    eq(out[0].values, [NAN, NAN, 0, 1, 1, 1])


def test_scalar_gt_time_empty():
    # `123 > time()` :2878 — all filtered; removeEmptySeries at the tail
    out = _eval(BinOpSpec(">"), [scalar(123)], [t_series()])
    assert remove_empty_series(out) == []


def test_cmp_bool_drops_metric_group():
    # `a cmp bool scalar (drop MetricGroup)` :2930
    left = [S("foo", [("a", "x")], TIME),
            S("bar", [("a", "y")], TIME + 200)]
    out = _eval(BinOpSpec(">=", bool_modifier=True), left, [scalar(1200)])
    got = by_tag(out, b"a")
    assert got[b"x"].mn.metric_group == b""
    assert got[b"y"].mn.metric_group == b""
    eq(got[b"x"].values, [0, 1, 1, 1, 1, 1])
    eq(got[b"y"].values, [1, 1, 1, 1, 1, 1])


# ---------------------------------------------------------------------
# and / unless (exec_test.go:3160-3205)
# ---------------------------------------------------------------------

def test_one_and_empty():
    # `1 and (0 > 1)` :3160 (issue 6637): empty right -> empty
    out = _eval(BinOpSpec("and"), [scalar(1)], [])
    assert remove_empty_series(out) == []


def test_time_and_scalar():
    # `time() and 2` :3166
    out = _eval(BinOpSpec("and"), [t_series()], [scalar(2)])
    assert len(out) == 1
    eq(out[0].values, [1000, 1200, 1400, 1600, 1800, 2000])


def test_time_and_filtered():
    # `time() and time() > 1300` :3177
    right = np.where(TIME > 1300, TIME, NAN)
    out = _eval(BinOpSpec("and"), [t_series()], [S("", [], right)])
    eq(out[0].values, [NAN, NAN, 1400, 1600, 1800, 2000])


def test_time_unless_scalar_empty():
    # `time() unless 2` :3188
    out = _eval(BinOpSpec("unless"), [t_series()], [scalar(2)])
    assert remove_empty_series(out) == []


def test_time_unless_filtered():
    # `time() unless time() > 1500` :3194
    right = np.where(TIME > 1500, TIME, NAN)
    out = _eval(BinOpSpec("unless"), [t_series()], [S("", [], right)])
    eq(out[0].values, [1000, 1200, 1400, NAN, NAN, NAN])


# ---------------------------------------------------------------------
# or / default (exec_test.go:3205-3360)
# ---------------------------------------------------------------------

def test_series_or_series():
    # `series or series` :3205 — same-key left wins whole, disjoint added
    left = [S("", [("x", "foo")], TIME), S("", [("x", "bar")], TIME + 1)]
    right = [S("", [("x", "foo")], TIME + 2), S("", [("x", "baz")], TIME + 3)]
    out = _eval(BinOpSpec("or"), left, right)
    got = by_tag(out, b"x")
    assert set(got) == {b"foo", b"bar", b"baz"}
    eq(got[b"foo"].values, TIME)
    eq(got[b"bar"].values, TIME + 1)
    eq(got[b"baz"].values, TIME + 3)


def test_scalar_or_scalar():
    # `time() > 1400 or 123` :3250
    left = np.where(TIME > 1400, TIME, NAN)
    out = _eval(BinOpSpec("or"), [S("", [], left)], [scalar(123)])
    assert len(out) == 1
    eq(out[0].values, [123, 123, 123, 1600, 1800, 2000])


def test_scalar_default_scalar():
    # `time() > 1400 default 123` :3276
    left = np.where(TIME > 1400, TIME, NAN)
    out = _eval(BinOpSpec("default"), [S("", [], left)], [scalar(123)])
    eq(out[0].values, [123, 123, 123, 1600, 1800, 2000])


def test_scalar_default_unmatched_vector():
    # `time() > 1400 default label_set(123, "foo", "bar")` :3298 —
    # differing tag sets do not match; left stays filtered
    left = np.where(TIME > 1400, TIME, NAN)
    out = _eval(BinOpSpec("default"), [S("", [], left)],
                [S("", [("foo", "bar")], np.full(6, 123.0))])
    assert len(out) == 1
    eq(out[0].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_scalar_default_vector_matching_by_tags():
    # `time() > 1400 default (label_set(123,"foo","bar"),
    #  label_set(456,"__name__","xxx"))` :3309 — grouping keys use TAGS
    # only (metric group excluded), so the tagless xxx series matches
    left = np.where(TIME > 1400, TIME, NAN)
    right = [S("", [("foo", "bar")], np.full(6, 123.0)),
             S("xxx", [], np.full(6, 456.0))]
    out = _eval(BinOpSpec("default"), [S("", [], left)], right)
    assert len(out) == 1
    eq(out[0].values, [456, 456, 456, 1600, 1800, 2000])


def test_scalar_default_nan():
    # `time() > 1400 default (time() < -100)` :3323 — NaN defaults change
    # nothing
    left = np.where(TIME > 1400, TIME, NAN)
    out = _eval(BinOpSpec("default"), [S("", [], left)],
                [S("", [], np.full(6, NAN))])
    eq(out[0].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_vector_default_scalar():
    # `vector default scalar` :3334 — per-series fill, names kept
    lx = np.where(TIME > 1400, TIME, NAN)
    ly = np.where(TIME < 1700, TIME, NAN)
    out = _eval(BinOpSpec("default"),
                [S("x", [("foo", "bar")], lx), S("y", [("foo", "baz")], ly)],
                [scalar(123)])
    got = by_tag(out, b"foo")
    assert got[b"bar"].mn.metric_group == b"x"
    assert got[b"baz"].mn.metric_group == b"y"
    eq(got[b"bar"].values, [123, 123, 123, 1600, 1800, 2000])
    eq(got[b"baz"].values, [1000, 1200, 1400, 1600, 123, 123])


# ---------------------------------------------------------------------
# group_left / group_right (exec_test.go:3473-3760)
# ---------------------------------------------------------------------

def test_scalar_mul_on_group_right():
    # `2 * on() group_right() (...)` :3473
    right = [S("", [("foo", "bar")], TIME),
             S("", [("foo", "qwert")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=[],
                          join_op="group_right"), [scalar(2)], right)
    got = by_tag(out, b"foo")
    eq(got[b"bar"].values, [2000, 2400, 2800, 3200, 3600, 4000])
    eq(got[b"qwert"].values, [20] * 6)
    assert got[b"bar"].mn.metric_group == b""


def test_scalar_mul_on_group_right_keep_metric_names():
    # :3497 — keep_metric_names keeps the many-side (right) names
    right = [S("q1", [("foo", "bar")], TIME),
             S("q2", [("foo", "qwert")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=[],
                          join_op="group_right", keep_metric_names=True),
                [scalar(2)], right)
    got = by_tag(out, b"foo")
    assert got[b"bar"].mn.metric_group == b"q1"
    assert got[b"qwert"].mn.metric_group == b"q2"
    eq(got[b"bar"].values, [2000, 2400, 2800, 3200, 3600, 4000])
    eq(got[b"qwert"].values, [20] * 6)


def test_scalar_mul_ignoring_group_right_join_tag():
    # `label_set(2,"a","2") * ignoring(foo,a) group_right(a) (...)` :3523 —
    # the join tag a is copied from the ONE (left) side onto every result
    left = [S("", [("a", "2")], np.full(6, 2.0))]
    right = [S("", [("foo", "bar"), ("a", "1")], TIME),
             S("", [("foo", "qwert")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("*", group_op="ignoring", group_tags=["foo", "a"],
                          join_op="group_right", join_tags=["a"]),
                left, right)
    got = by_tag(out, b"foo")
    assert got[b"bar"].mn.get_tag_value(b"a") == b"2"
    assert got[b"qwert"].mn.get_tag_value(b"a") == b"2"
    eq(got[b"bar"].values, [2000, 2400, 2800, 3200, 3600, 4000])
    eq(got[b"qwert"].values, [20] * 6)


def test_group_left_duplicate_differ_by_join_tag():
    # `... + on(foo) group_left(op) (le/ge split)` :3629 — two many-side
    # matches per one-side series are legal when the join tag disambiguates
    left = [S("qwert", [("foo", "bar"), ("xx", "yy")], TIME / 10)]
    r_le = np.where(TIME < 1400, TIME, NAN)
    r_ge = np.where(TIME >= 1400, TIME, NAN)
    right = [S("", [("foo", "bar"), ("op", "le")], r_le),
             S("", [("foo", "bar"), ("op", "ge")], r_ge)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_left", join_tags=["op"]),
                left, right)
    got = by_tag(out, b"op")
    eq(got[b"le"].values, [1100, 1320, NAN, NAN, NAN, NAN])
    eq(got[b"ge"].values, [NAN, NAN, 1540, 1760, 1980, 2200])
    # group_left keeps the left's extra tags
    assert got[b"le"].mn.get_tag_value(b"xx") == b"yy"
    assert got[b"ge"].mn.get_tag_value(b"xx") == b"yy"


def test_on_duplicate_nonoverlapping_merge():
    # `... + on(foo) (le/ge split)` :3676 — WITHOUT group_left the two
    # non-overlapping right series merge into one result (on() labels only)
    left = [S("qwert", [("foo", "bar"), ("xx", "yy")], TIME / 10)]
    r_le = np.where(TIME < 1400, TIME, NAN)
    r_ge = np.where(TIME >= 1400, TIME, NAN)
    right = [S("", [("foo", "bar"), ("op", "le")], r_le),
             S("", [("foo", "bar"), ("op", "ge")], r_ge)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"]),
                left, right)
    assert len(out) == 1
    eq(out[0].values, [1100, 1320, 1540, 1760, 1980, 2200])
    assert out[0].mn.tags == [(b"foo", b"bar")]


def test_on_group_left_empty_join_nonoverlapping_merge():
    # `... + on(foo) group_left() (le/ge split)` :3696 — group_left()
    # without join tags also merges, but keeps the left's extra tags
    left = [S("qwert", [("foo", "bar"), ("xx", "yy")], TIME / 10)]
    r_le = np.where(TIME < 1400, TIME, NAN)
    r_ge = np.where(TIME >= 1400, TIME, NAN)
    right = [S("", [("foo", "bar"), ("op", "le")], r_le),
             S("", [("foo", "bar"), ("op", "ge")], r_ge)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_left"), left, right)
    assert len(out) == 1
    eq(out[0].values, [1100, 1320, 1540, 1760, 1980, 2200])
    assert out[0].mn.get_tag_value(b"xx") == b"yy"


def test_group_left_name_join_tag():
    # `... + on(foo) group_left(__name__) label_set(..., "__name__","aaa")`
    # :3720 — __name__ as a join tag copies the metric group from the right
    left = [S("qwert", [("foo", "bar"), ("xx", "yy")], TIME / 10)]
    right = [S("aaa", [("foo", "bar")], TIME)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_left", join_tags=["__name__"]),
                left, right)
    assert len(out) == 1
    assert out[0].mn.metric_group == b"aaa"
    eq(out[0].values, [1100, 1320, 1540, 1760, 1980, 2200])
    assert out[0].mn.get_tag_value(b"xx") == b"yy"


def test_vector_plus_vector_partial_matching_keep_metric_names():
    # `vector + vector partial matching keep_metric_names` :3857 —
    # with keep_metric_names the metric group joins the matching key, so
    # only the q1/t1=v1 pair matches; the result keeps the name
    left = [S("q1", [("t1", "v1")], TIME),
            S("q2", [("t2", "v2")], np.full(6, 10.0))]
    right = [S("q1", [("t1", "v1")], np.full(6, 100.0)),
             S("", [("t2", "v3")], TIME)]
    out = remove_empty_series(
        _eval(BinOpSpec("+", keep_metric_names=True), left, right))
    assert len(out) == 1
    assert out[0].mn.metric_group == b"q1"
    assert out[0].mn.tags == [(b"t1", b"v1")]
    eq(out[0].values, [1100, 1300, 1500, 1700, 1900, 2100])


def test_vector_plus_vector_no_matching():
    # `vector + vector no matching` :3877 -> empty
    left = [S("", [("t2", "v1")], TIME),
            S("", [("t2", "v2")], np.full(6, 10.0))]
    right = [S("", [("t1", "v1")], np.full(6, 100.0)),
             S("", [("t2", "v3")], TIME)]
    out = remove_empty_series(_eval(BinOpSpec("+"), left, right))
    assert out == []


def test_vector_plus_vector_on_matching():
    # `vector + vector on (foo, t2) matching` :3887 — only the t2=v3 pair
    # matches; on() keeps only the on-tags present
    left = [S("", [("t1", "v123"), ("t2", "v3")], TIME),
            S("", [("t2", "v2")], np.full(6, 10.0))]
    right = [S("", [("t1", "v1")], np.full(6, 100.0)),
             S("", [("t2", "v3")], TIME)]
    out = remove_empty_series(
        _eval(BinOpSpec("+", group_op="on", group_tags=["foo", "t2"]),
              left, right))
    assert len(out) == 1
    assert out[0].mn.tags == [(b"t2", b"v3")]
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])


# ---------------------------------------------------------------------
# group_left joins with tag copying and the fill() family
# (exec_test.go:3908-4300)
# ---------------------------------------------------------------------

def _tags(s):
    return sorted(s.mn.tags)


def test_group_left_tag_copy_matching():
    # `(... ) + on (foo, t2) group_left (t1, noxxx) (...)` :3908 — the
    # join copies t1/noxxx from the one side; an absent join tag REMOVES
    # the tag from the result
    from victoriametrics_amd import transform as tfm
    left = [S("", [("t1", "v123"), ("t2", "v3")], TIME),
            S("", [("t2", "v3"), ("xxx", "yy")], np.full(6, 10.0))]
    right = [S("", [("t1", "v1")], np.full(6, 100.0)),
             S("", [("t2", "v3"), ("noxxx", "aa")], TIME)]
    spec = BinOpSpec("+", group_op="on", group_tags=["foo", "t2"],
                     join_op="group_left", join_tags=["t1", "noxxx"])
    out = tfm.sort_series(_eval(spec, left, right), desc=True)
    assert len(out) == 2
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert _tags(out[0]) == [(b"noxxx", b"aa"), (b"t2", b"v3")]
    eq(out[1].values, [1010, 1210, 1410, 1610, 1810, 2010])
    assert _tags(out[1]) == [(b"noxxx", b"aa"), (b"t2", b"v3"),
                             (b"xxx", b"yy")]


def test_group_left_star_and_prefix():
    # :3952 group_left(*): copy ALL right tags except the on() set;
    # :4000 adds `prefix "abc_"` to the copied names
    from victoriametrics_amd import transform as tfm

    def run(prefix):
        left = [S("", [("t1", "v123"), ("t2", "v3")], TIME),
                S("", [("t2", "v3"), ("xxx", "yy")], np.full(6, 10.0))]
        right = [S("", [("t1", "v1")], np.full(6, 100.0)),
                 S("", [("t2", "v3"), ("noxxx", "aa")], TIME)]
        spec = BinOpSpec("+", group_op="on", group_tags=["foo", "t2"],
                         join_op="group_left", join_tags=["*"],
                         join_prefix=prefix)
        return tfm.sort_series(_eval(spec, left, right), desc=True)

    out = run("")
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert _tags(out[0]) == [(b"noxxx", b"aa"), (b"t1", b"v123"),
                             (b"t2", b"v3")]
    eq(out[1].values, [1010, 1210, 1410, 1610, 1810, 2010])
    assert _tags(out[1]) == [(b"noxxx", b"aa"), (b"t2", b"v3"),
                             (b"xxx", b"yy")]
    out = run("abc_")
    assert _tags(out[0]) == [(b"abc_noxxx", b"aa"), (b"t1", b"v123"),
                             (b"t2", b"v3")]
    assert _tags(out[1]) == [(b"abc_noxxx", b"aa"), (b"t2", b"v3"),
                             (b"xxx", b"yy")]


def test_group_left_copies_name():
    # `... + on (t2, dfdf) group_left (__name__, xxx) ...` :4048 — the
    # result takes the one side's __name__; absent xxx copies nothing
    from victoriametrics_amd import transform as tfm
    left = [S("vv3", [("t2", "v3"), ("x", "y")], TIME),
            S("yy", [("t2", "v3")], np.full(6, 10.0))]
    right = [S("", [("t1", "v1")], np.full(6, 100.0)),
             S("abc", [("t2", "v3")], TIME)]
    spec = BinOpSpec("+", group_op="on", group_tags=["t2", "dfdf"],
                     join_op="group_left", join_tags=["__name__", "xxx"])
    out = tfm.sort_series(_eval(spec, left, right), desc=True)
    assert len(out) == 2
    for s in out:
        assert s.mn.metric_group == b"abc"
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert _tags(out[0]) == [(b"t2", b"v3"), (b"x", b"y")]
    eq(out[1].values, [1010, 1210, 1410, 1610, 1810, 2010])
    assert _tags(out[1]) == [(b"t2", b"v3")]


def test_ignoring_matching():
    # `... + ignoring (foo, t1, bar) ...` :4084 — one matching pair
    left = [S("", [("t1", "v123"), ("t2", "v3")], TIME),
            S("", [("t2", "v2")], np.full(6, 10.0))]
    right = [S("", [("t1", "v1")], np.full(6, 100.0)),
             S("", [("t2", "v3")], TIME)]
    spec = BinOpSpec("+", group_op="ignoring",
                     group_tags=["foo", "t1", "bar"])
    out = _eval(spec, left, right)
    assert len(out) == 1
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert _tags(out[0]) == [(b"t2", b"v3")]


def test_ignoring_group_right_matching():
    # `... + ignoring (foo, t2) group_right () ...` :4105 — one left
    # joins two right series; results keep the RIGHT side's tags
    from victoriametrics_amd import transform as tfm
    left = [S("", [("t1", "v123"), ("t2", "v3")], TIME),
            S("", [("t2", "v321"), ("t1", "v123"), ("t32", "v32")],
              np.full(6, 10.0))]
    right = [S("", [("t1", "v123")], np.full(6, 100.0)),
             S("", [("t1", "v123"), ("t2", "v3")], TIME)]
    spec = BinOpSpec("+", group_op="ignoring", group_tags=["foo", "t2"],
                     join_op="group_right", join_tags=[])
    out = tfm.sort_series(_eval(spec, left, right), desc=True)
    assert len(out) == 2
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert _tags(out[0]) == [(b"t1", b"v123"), (b"t2", b"v3")]
    eq(out[1].values, [1100, 1300, 1500, 1700, 1900, 2100])
    assert _tags(out[1]) == [(b"t1", b"v123")]


def _fill_inputs():
    left = [S("", [("foo", "common")], np.full(6, 1.0)),
            S("", [("foo", "left_only")], np.full(6, 2.0))]
    right = [S("", [("foo", "common")], np.full(6, 3.0)),
             S("", [("foo", "right_only")], np.full(6, 4.0))]
    return left, right


def _by_foo(out):
    return {s.mn.get_tag_value(b"foo"): s for s in out}


def test_fill_both_sides():
    # `(...) + fill(0) (...)` :4139 — unmatched series on either side
    # join against the fill constant
    left, right = _fill_inputs()
    out = _by_foo(_eval(BinOpSpec("+", fill_left=0.0, fill_right=0.0),
                        left, right))
    assert set(out) == {b"common", b"left_only", b"right_only"}
    eq(out[b"common"].values, [4] * 6)
    eq(out[b"left_only"].values, [2] * 6)
    eq(out[b"right_only"].values, [4] * 6)


def test_fill_per_point_nan():
    # `(time()<=1200 ...) + fill(10) (time()>=1600 ...)` :4178 — the fill
    # applies per POINT: a lone NaN side is filled, both-NaN stays NaN
    left = [S("", [("foo", "common")],
              np.where(TIME <= 1200, TIME, NAN))]
    right = [S("", [("foo", "common")],
               np.where(TIME >= 1600, TIME, NAN))]
    out = _eval(BinOpSpec("+", fill_left=10.0, fill_right=10.0),
                left, right)
    assert len(out) == 1
    eq(out[0].values, [1010, 1210, NAN, 1610, 1810, 2010])


def test_fill_left_right_distinct():
    # `... + fill_left(10) fill_right(20) ...` :4197 and
    # `... + fill_right(20) ...` :4236 (right_only dropped without
    # fill_left)
    left, right = _fill_inputs()
    out = _by_foo(_eval(BinOpSpec("+", fill_left=10.0, fill_right=20.0),
                        left, right))
    assert set(out) == {b"common", b"left_only", b"right_only"}
    eq(out[b"common"].values, [4] * 6)
    eq(out[b"left_only"].values, [22] * 6)
    eq(out[b"right_only"].values, [14] * 6)
    left, right = _fill_inputs()
    out = _by_foo(_eval(BinOpSpec("+", fill_right=20.0), left, right))
    assert set(out) == {b"common", b"left_only"}
    eq(out[b"common"].values, [4] * 6)
    eq(out[b"left_only"].values, [22] * 6)


def test_fill_with_on_matching():
    # `... + on(foo) fill(0) ...` :4266 — on() reduces the result name to
    # the foo tag; fills still apply per key group
    left = [S("", [("foo", "common"), ("extra", "l")], np.full(6, 1.0)),
            S("", [("foo", "left_only"), ("extra", "l")], np.full(6, 2.0))]
    right = [S("", [("foo", "common"), ("extra", "r")], np.full(6, 3.0)),
             S("", [("foo", "right_only"), ("extra", "r")],
               np.full(6, 4.0))]
    out = _by_foo(_eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                                  fill_left=0.0, fill_right=0.0),
                        left, right))
    assert set(out) == {b"common", b"left_only", b"right_only"}
    eq(out[b"common"].values, [4] * 6)
    assert _tags(out[b"common"]) == [(b"foo", b"common")]
    eq(out[b"left_only"].values, [2] * 6)
    eq(out[b"right_only"].values, [4] * 6)


# ---------------------------------------------------------------------
# if / ifnot set ops (exec_test.go:7539-7660) and group_left fill_right
# (:4305)
# ---------------------------------------------------------------------

def test_vector_if_vector_partial():
    # `(x=y, foo=bar:x) if (foo=bar filtered)` :7539 — only the series
    # with a matching right key survives, masked by the right's NaNs;
    # the x=y series has no match and is dropped entirely
    left = [S("", [("x", "y")], TIME / 10),
            S("x", [("foo", "bar")], TIME)]
    right = [S("", [("foo", "bar")], np.where(TIME > 1400, TIME, NAN))]
    out = _eval(BinOpSpec("if"), left, right)
    out = remove_empty_series(out)
    assert len(out) == 1
    assert out[0].mn.metric_group == b"x"
    eq(out[0].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_vector_if_vector_both_match():
    # :7560 — both left series find their mask series
    left = [S("", [("x", "y")], TIME / 10),
            S("x", [("foo", "bar")], TIME)]
    right = [S("", [("foo", "bar")], np.where(TIME > 1400, TIME, NAN)),
             S("", [("x", "y")], np.where(TIME < 1400, TIME, NAN))]
    out = remove_empty_series(_eval(BinOpSpec("if"), left, right))
    got = by_tag(out, "x")
    eq(got[b"y"].values, [100, 120, NAN, NAN, NAN, NAN])
    got = by_tag(out, "foo")
    eq(got[b"bar"].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_scalar_if_vector():
    # :7591 `time() if (labeled vector)` -> empty (no key match);
    # :7599 with a nameless-matching series -> masked time()
    left = [t_series()]
    right = [S("", [("foo", "bar")], np.full(6, 123.0))]
    out = remove_empty_series(_eval(BinOpSpec("if"), left, right))
    assert out == []
    left = [t_series()]
    right = [S("", [("foo", "bar")], np.full(6, 123.0)),
             S("xxx", [], np.where(TIME > 1400, TIME, NAN))]
    out = remove_empty_series(_eval(BinOpSpec("if"), left, right))
    assert len(out) == 1
    eq(out[0].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_if_ifnot_default_chain():
    # `time() if time() > 1400 default -time()` :7613 -> [-1000..2000];
    # `time() ifnot time() > 1400 default -time()` :7624;
    # `time() ifnot time() > 1400` :7635
    cond = [S("", [], np.where(TIME > 1400, TIME, NAN))]
    masked = _eval(BinOpSpec("if"), [t_series()], cond)
    out = _eval(BinOpSpec("default"), masked, [S("", [], -TIME)])
    eq(out[0].values, [-1000, -1200, -1400, 1600, 1800, 2000])
    cond = [S("", [], np.where(TIME > 1400, TIME, NAN))]
    masked = _eval(BinOpSpec("ifnot"), [t_series()], cond)
    eq(masked[0].values, [1000, 1200, 1400, NAN, NAN, NAN])
    out = _eval(BinOpSpec("default"), masked, [S("", [], -TIME)])
    eq(out[0].values, [1000, 1200, 1400, -1600, -1800, -2000])


def test_ifnot_no_matching_series():
    # :7646 — different keys: ifnot keeps the left side untouched
    left = [S("", [("foo", "bar")], TIME)]
    right = [S("", [("x", "y")], np.where(TIME > 1400, TIME, NAN))]
    out = _eval(BinOpSpec("ifnot"), left, right)
    assert len(out) == 1
    eq(out[0].values, [1000, 1200, 1400, 1600, 1800, 2000])
    assert out[0].mn.get_tag_value("foo") == b"bar"


def test_group_left_fill_right():
    # `(codes) + on(method) group_left() fill_right(0) (10 method=get)`
    # :4305 — many-side codes join the one side; the put series joins the
    # fill constant
    left = [S("", [("method", "get"), ("code", "500")], np.full(6, 1.0)),
            S("", [("method", "get"), ("code", "404")], np.full(6, 2.0)),
            S("", [("method", "put"), ("code", "501")], np.full(6, 3.0))]
    right = [S("", [("method", "get")], np.full(6, 10.0))]
    spec = BinOpSpec("+", group_op="on", group_tags=["method"],
                     join_op="group_left", join_tags=[], fill_right=0.0)
    out = _eval(spec, left, right)
    got = {s.mn.get_tag_value(b"code"): s for s in out}
    assert set(got) == {b"500", b"404", b"501"}
    eq(got[b"500"].values, [11] * 6)
    assert got[b"500"].mn.get_tag_value(b"method") == b"get"
    eq(got[b"404"].values, [12] * 6)
    eq(got[b"501"].values, [3] * 6)
    assert got[b"501"].mn.get_tag_value(b"method") == b"put"


# ---------------------------------------------------------------------
# `or on(...)` merge walk (exec_test.go:10049-10430): the right side
# fills only the points no left series of the same on() group covers
# ---------------------------------------------------------------------

def _or_on(left, right, tags):
    return _eval(BinOpSpec("or", group_op="on", group_tags=tags),
                 left, right)


def test_nan_or_on_series():
    # `(label_set(1,a,b1) == 0) or on(a) label_set(2,a,b2)` :10049 — the
    # all-NaN comparison result drops, the right side survives whole
    left = [S("", [("a", "a"), ("b", "b1")], np.full(6, NAN))]
    right = [S("", [("a", "a"), ("b", "b2")], np.full(6, 2.0))]
    out = remove_empty_series(_or_on(left, right, ["a"]))
    assert len(out) == 1
    eq(out[0].values, [2] * 6)
    assert out[0].mn.get_tag_value("b") == b"b2"


def test_series_with_nans_or_scalar():
    # `(time()>=1600 tagged) or 1` :10069 — plain or (whole-series key
    # matching): disjoint keys, both kept as-is
    left = [S("", [("a", "a"), ("b", "b1")],
              np.where(TIME >= 1600, TIME, NAN))]
    right = [S("", [], np.ones(6))]
    out = _eval(BinOpSpec("or"), left, right)
    assert len(out) == 2
    m = by_tag(out, "b")
    eq(m[b"b1"].values, [NAN, NAN, NAN, 1600, 1800, 2000])
    eq(m[b""].values, [1] * 6)


def test_series_or_on_empty_scalar():
    # `(time()>1200 tagged) or on() vector(0)` :10093 — on() makes ONE
    # group; the scalar fills exactly the left's NaN points
    left = [S("", [("a", "a"), ("b", "b1")],
              np.where(TIME > 1200, TIME, NAN))]
    right = [S("", [], np.zeros(6))]
    out = _or_on(left, right, [])
    assert len(out) == 2
    m = by_tag(out, "b")
    eq(m[b"b1"].values, [NAN, NAN, 1400, 1600, 1800, 2000])
    eq(m[b""].values, [0, 0, NAN, NAN, NAN, NAN])


def test_series_or_on_series_disjoint_and_overlap():
    # :10118 disjoint halves -> right keeps only the uncovered points;
    # :10150 left full -> right dropped; :10170 overlap -> right loses
    # the overlapped points
    left = [S("", [("a", "a"), ("b", "b1")],
              np.where(TIME <= 1200, TIME, NAN))]
    right = [S("", [("a", "a"), ("b", "b2")],
               np.where(TIME > 1200, TIME, NAN))]
    out = _or_on(left, right, ["a"])
    m = by_tag(remove_empty_series(out), "b")
    eq(m[b"b1"].values, [1000, 1200, NAN, NAN, NAN, NAN])
    eq(m[b"b2"].values, [NAN, NAN, 1400, 1600, 1800, 2000])
    left = [S("", [("a", "a"), ("b", "b1")], TIME)]
    right = [S("", [("a", "a"), ("b", "b2")], np.full(6, NAN))]
    out = remove_empty_series(_or_on(left, right, ["a"]))
    assert len(out) == 1
    eq(out[0].values, [1000, 1200, 1400, 1600, 1800, 2000])
    left = [S("", [("a", "a"), ("b", "b1")],
              np.where(TIME <= 1500, TIME, NAN))]
    right = [S("", [("a", "a"), ("b", "b2")],
               np.where(TIME > 1100, TIME, NAN))]
    m = by_tag(remove_empty_series(_or_on(left, right, ["a"])), "b")
    eq(m[b"b1"].values, [1000, 1200, 1400, NAN, NAN, NAN])
    eq(m[b"b2"].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_series_or_on_series_merge_same_key():
    # :10203 — identical full keys merge into ONE series
    left = [S("", [("a", "a"), ("b", "b1")],
              np.where(TIME <= 1200, TIME, NAN))]
    right = [S("", [("a", "a"), ("b", "b1")],
               np.where(TIME > 1400, TIME, NAN))]
    out = remove_empty_series(_or_on(left, right, ["a"]))
    assert len(out) == 1
    eq(out[0].values, [1000, 1200, NAN, 1600, 1800, 2000])


def test_series_or_many_series():
    # :10243 — one left with a hole at 1200; BOTH right series fill only
    # that hole, keeping their own tags
    left = [S("", [("x", "foo")], np.where(TIME != 1200, TIME, NAN))]
    right = [S("", [("x", "foo"), ("y", "bar")], TIME + 1),
             S("", [("y", "baz"), ("x", "foo")], TIME + 2)]
    out = remove_empty_series(_or_on(left, right, ["x"]))
    assert len(out) == 3
    m = by_tag(out, "y")
    eq(m[b""].values, [1000, NAN, 1400, 1600, 1800, 2000])
    eq(m[b"bar"].values, [NAN, 1201, NAN, NAN, NAN, NAN])
    eq(m[b"baz"].values, [NAN, 1202, NAN, NAN, NAN, NAN])


def test_many_series_or_series():
    # :10290 — the left GROUP's union covers every point (the y=baz
    # series is complete), so the right side contributes nothing
    left = [S("", [("x", "foo")], np.where(TIME != 1200, TIME, NAN)),
            S("", [("x", "foo"), ("y", "baz")], TIME + 1)]
    right = [S("", [("x", "foo"), ("y", "bar")], TIME + 2)]
    out = remove_empty_series(_or_on(left, right, ["x"]))
    assert len(out) == 2
    m = by_tag(out, "y")
    eq(m[b""].values, [1000, NAN, 1400, 1600, 1800, 2000])
    eq(m[b"baz"].values, TIME + 1)


def test_many_series_or_series_merge_matrix():
    # :10327 no-merge: the group union covers all points -> right dropped;
    # :10367 merge: 1400 uncovered -> BOTH right series fill it
    def lhs(a2_from):
        return [S("", [("job", "a1"), ("a", "a")],
                  np.where(TIME != 1400, TIME, NAN)),
                S("", [("job", "a2"), ("a", "a")],
                  np.where(TIME >= a2_from, TIME, NAN))]
    right_pair = lambda: [S("", [("job", "a3"), ("a", "a")], TIME.copy()),
                          S("", [("job", "a4"), ("a", "a")], TIME.copy())]
    out = remove_empty_series(_or_on(lhs(1400), right_pair(), ["a"]))
    m = by_tag(out, "job")
    assert set(m) == {b"a1", b"a2"}
    eq(m[b"a1"].values, [1000, 1200, NAN, 1600, 1800, 2000])
    eq(m[b"a2"].values, [NAN, NAN, 1400, 1600, 1800, 2000])
    out = remove_empty_series(_or_on(lhs(1600), right_pair(), ["a"]))
    m = by_tag(out, "job")
    assert set(m) == {b"a1", b"a2", b"a3", b"a4"}
    eq(m[b"a1"].values, [1000, 1200, NAN, 1600, 1800, 2000])
    eq(m[b"a2"].values, [NAN, NAN, NAN, 1600, 1800, 2000])
    eq(m[b"a3"].values, [NAN, NAN, 1400, NAN, NAN, NAN])
    eq(m[b"a4"].values, [NAN, NAN, 1400, NAN, NAN, NAN])


def test_union_list_comparisons():
    # `time() == (100, 1000, 1400, 600)` :6468 and the reversed form
    # :6479 -> [1000, nan, 1400, nan, nan, nan]; `!=` :6490 keeps the
    # complement.  The union side is the scalar list (binary_op.go:55-113)
    from victoriametrics_amd.binary_op import union_list_cmp
    from victoriametrics_amd import transform as tfm

    def lst():
        return tfm.union([[scalar(100)], [scalar(1000)], [scalar(1400)],
                          [scalar(600)]])
    assert len(lst()) == 4  # all-scalar union keeps every member
    out = union_list_cmp("==", [t_series()], lst())
    assert len(out) == 1
    eq(out[0].values, [1000, NAN, 1400, NAN, NAN, NAN])
    # reversed operands: the union side still masks the data side
    out = union_list_cmp("==", lst(), [t_series()], union_on_left=True)
    assert len(out) == 1
    eq(out[0].values, [1000, NAN, 1400, NAN, NAN, NAN])
    # `alias(time(),"foobar") != union(100, 1000, 1400, 600)` :6490
    out = union_list_cmp("!=", [S("foobar", [], TIME)], lst())
    assert len(out) == 1
    assert out[0].mn.metric_group == b"foobar"
    eq(out[0].values, [NAN, 1200, NAN, 1600, 1800, 2000])
    # empty sides (binary_op.go:69/95): == -> empty, != -> left as-is
    assert union_list_cmp("==", [t_series()], []) == []
    out = union_list_cmp("!=", [t_series()], [])
    eq(out[0].values, TIME)


def test_vector_scalar_arith_and_keep_metric_names():
    # `vector / scalar` :3362 and the keep_metric_names variants
    # :3386/:3447 — plain arith resets __name__, keep_metric_names keeps it
    left = [S("q1", [("foo", "bar")], TIME),
            S("q2", [("foo", "qwert")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("/"), left, [scalar(2)])
    m = by_tag(out, "foo")
    eq(m[b"bar"].values, [500, 600, 700, 800, 900, 1000])
    eq(m[b"qwert"].values, [5] * 6)
    for s in out:
        assert s.mn.metric_group == b""
    left = [S("q1", [("foo", "bar")], TIME),
            S("q2", [("foo", "qwert")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("/", keep_metric_names=True), left, [scalar(2)])
    names = {s.mn.get_tag_value("foo"): s.mn.metric_group for s in out}
    assert names == {b"bar": b"q1", b"qwert": b"q2"}
    left = [S("q1", [("foo", "bar")], TIME),
            S("q2", [("foo", "qwert")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("*", keep_metric_names=True), [scalar(2)], left)
    m = by_tag(out, "foo")
    eq(m[b"bar"].values, [2000, 2400, 2800, 3200, 3600, 4000])
    eq(m[b"qwert"].values, [20] * 6)


def test_vector_mul_on_foo_scalar():
    # `(...) * on(foo) label_set(2, foo=bar, aa=bb)` :3591 — only the
    # foo=bar left matches; on(foo) reduces the result name to foo
    left = [S("", [("foo", "bar"), ("xx", "yy")], TIME),
            S("", [("foo", "qwert")], np.full(6, 10.0))]
    right = [S("", [("foo", "bar"), ("aa", "bb")], np.full(6, 2.0))]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=["foo"]),
                left, right)
    assert len(out) == 1
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert sorted(out[0].mn.tags) == [(b"foo", b"bar")]
    # :3608 keep_metric_names: __name__ joins the grouping and survives
    left = [S("q1", [("foo", "bar"), ("xx", "yy")], TIME),
            S("q2", [("foo", "qwert")], np.full(6, 10.0))]
    right = [S("q2", [("foo", "bar"), ("aa", "bb")], np.full(6, 2.0))]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=["foo"],
                          keep_metric_names=True), left, right)
    assert len(out) == 1
    eq(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])
    assert out[0].mn.metric_group == b"q1"
    assert sorted(out[0].mn.tags) == [(b"foo", b"bar")]


def test_vector_on_empty_group_left_scalar():
    # `(...) * on() group_left 2` :3786 — every left series joins the
    # lone scalar; left names survive whole
    left = [S("", [("foo", "bar")], TIME),
            S("", [("foo", "qwert")], np.full(6, 10.0))]
    spec = BinOpSpec("*", group_op="on", group_tags=[],
                     join_op="group_left", join_tags=[])
    out = _eval(spec, left, [scalar(2)])
    m = by_tag(out, "foo")
    eq(m[b"bar"].values, [2000, 2400, 2800, 3200, 3600, 4000])
    eq(m[b"qwert"].values, [20] * 6)


# ---------------------------------------------------------------------
# NaN comparison semantics (exec_test.go:2980-3126): issue 150 (explicit
# NaN is a value: `x != NaN` is true), issue 10018 (NaN from a vector
# comparison is a FILTERED sample: drop it on the right), PR 11100
# (per-point drop for partially-filtered right sides)
# ---------------------------------------------------------------------

def test_compare_to_nan_scalar_sides():
    # `1 != nan` :2980 -> 1s; `nan != 1` :2991 -> empty
    out = remove_empty_series(_eval(BinOpSpec("!="), [scalar(1)],
                                    [S("", [], np.full(6, NAN))]))
    assert len(out) == 1
    eq(out[0].values, [1] * 6)
    out = remove_empty_series(_eval(BinOpSpec("!="),
                                    [S("", [], np.full(6, NAN))],
                                    [scalar(1)]))
    assert out == []
    # :2997 vector != NaN-scalar -> kept whole
    out = remove_empty_series(_eval(BinOpSpec("!="),
                                    [S("", [("foo", "bar")], TIME)],
                                    [S("", [], np.full(6, NAN))]))
    eq(out[0].values, TIME)
    # :3012 vector != 1200 filters the equal point
    out = _eval(BinOpSpec("!="), [S("", [("foo", "bar")], TIME)],
                [scalar(1200)])
    eq(out[0].values, [1000, NAN, 1400, 1600, 1800, 2000])


def test_compare_to_explicit_nan_vector_right():
    # :3027 the right carries an EXPLICIT NaN sample (not produced by a
    # comparison): `x != NaN` is true, series kept whole
    left = [S("", [("foo", "bar")], TIME)]
    right = [S("", [("foo", "bar")], np.full(6, NAN))]
    out = remove_empty_series(_eval(BinOpSpec("!="), left, right,
                                    drop_nan_right=False))
    assert len(out) == 1
    eq(out[0].values, TIME)
    # :3042 `x != (1 > 2)` — a scalar-only comparison also counts as an
    # explicit NaN (isVectorComparisonExpr is false)
    right = [S("", [], np.full(6, NAN))]
    out = remove_empty_series(_eval(BinOpSpec("!="),
                                    [S("", [("foo", "bar")], TIME)],
                                    right, drop_nan_right=False))
    eq(out[0].values, TIME)


def test_compare_to_filtered_vector_right():
    # :3057 the right's NaNs come from a VECTOR comparison -> dropped
    # per point (drop_nan_right=True) -> empty result
    left = [S("", [("foo", "bar")], TIME)]
    right = [S("", [("foo", "bar")], np.full(6, NAN))]
    out = remove_empty_series(_eval(BinOpSpec("!="), left, right,
                                    drop_nan_right=True))
    assert out == []
    # :3069 all-NaN LEFT needs no special flag
    out = remove_empty_series(_eval(
        BinOpSpec("!="), [S("", [("foo", "bar")], np.full(6, NAN))],
        [S("", [("foo", "bar")], TIME)]))
    assert out == []
    # :3077 bool form drops the same way
    out = remove_empty_series(_eval(
        BinOpSpec("==", bool_modifier=True),
        [S("", [("foo", "bar")], TIME)],
        [S("", [("foo", "bar")], np.full(6, NAN))], drop_nan_right=True))
    assert out == []
    # :3083 partially filtered right: time()*2 > 2800 -> only the tail
    # survives the per-point drop
    right = [S("", [("foo", "bar")],
               np.where(TIME * 2 > 2800, TIME * 2, NAN))]
    out = _eval(BinOpSpec("!="), [S("", [("foo", "bar")], TIME)], right,
                drop_nan_right=True)
    eq(out[0].values, [NAN, NAN, NAN, 1600, 1800, 2000])


def test_compare_to_filtered_right_with_fills():
    # :3108 fill_left does NOT rescue the dropped right side -> empty;
    # :3117 fill_right replaces the dropped NaN -> full series survives
    left = [S("", [("foo", "bar")], TIME)]
    right = [S("", [("foo", "bar")], np.full(6, NAN))]
    out = remove_empty_series(_eval(
        BinOpSpec("!=", fill_left=0.0), left, right, drop_nan_right=True))
    assert out == []
    left = [S("", [("foo", "bar")], TIME)]
    right = [S("", [("foo", "bar")], np.full(6, NAN))]
    out = remove_empty_series(_eval(
        BinOpSpec("!=", fill_right=0.0), left, right, drop_nan_right=True))
    assert len(out) == 1
    eq(out[0].values, TIME)


def test_vector_div_vector_ignoring_fill():
    # `(... by code) / ignoring(code) fill(0) (... by method)` :4362 —
    # the post series exists only on the right: fill_left gives 0/5 = 0
    left = [S("", [("method", "get"), ("code", "500")], np.full(6, 6.0)),
            S("", [("method", "put"), ("code", "500")], np.full(6, 1.0))]
    right = [S("", [("method", "get")], np.full(6, 12.0)),
             S("", [("method", "post")], np.full(6, 5.0)),
             S("", [("method", "put")], np.full(6, 10.0))]
    out = _eval(BinOpSpec("/", group_op="ignoring", group_tags=["code"],
                          fill_left=0.0, fill_right=0.0), left, right)
    m = by_tag(out, "method")
    assert set(m) == {b"get", b"post", b"put"}
    eq(m[b"get"].values, [0.5] * 6)
    eq(m[b"post"].values, [0] * 6)
    eq(m[b"put"].values, [0.1] * 6)
    for s in out:
        assert s.mn.get_tag_value("code") is None


def test_result_sorting_canonical():
    # `result sorting` :9731 — the exec tail sorts by metric name then
    # tag pairs (sortSeriesByMetricName/metricNameLess)
    from victoriametrics_amd.binary_op import sort_series_by_metric_name
    xs = [
        S("", [("instance", "localhost:1001"), ("type", "free")], 1.0 * np.ones(6)),
        S("", [("instance", "localhost:1001"), ("type", "buffers")], np.ones(6)),
        S("", [("instance", "localhost:1000"), ("type", "buffers")], np.ones(6)),
        S("", [("instance", "localhost:1000"), ("type", "free")], np.ones(6)),
    ]
    sort_series_by_metric_name(xs)  # sorts in place, like the reference
    got = [(s.mn.get_tag_value("instance"), s.mn.get_tag_value("type"))
           for s in xs]
    assert got == [(b"localhost:1000", b"buffers"),
                   (b"localhost:1000", b"free"),
                   (b"localhost:1001", b"buffers"),
                   (b"localhost:1001", b"free")]


def test_sort_by_label_numeric_special_chars():
    # :9836 / :9983 — numeric ranks inside special-char strings
    from victoriametrics_amd import transform as tfm
    xs = [S("", [("x", "1:0:2"), ("y", "1:0:1")], np.ones(6)),
          S("", [("x", "1:0:15"), ("y", "1:0:1")], np.full(6, 2.0))]
    out = tfm.sort_by_label_numeric(xs, ["x", "y"])
    assert [s.mn.get_tag_value("x") for s in out] == [b"1:0:2", b"1:0:15"]
    xs = [S("", [("a", "DS50:1/0/15")], np.full(6, 4.0)),
          S("", [("a", "DS50:1/0/0")], np.ones(6)),
          S("", [("a", "DS50:1/0/1")], np.full(6, 2.0)),
          S("", [("a", "DS50:1/0/2")], np.full(6, 3.0))]
    out = tfm.sort_by_label_numeric(xs, ["a"])
    assert [s.mn.get_tag_value("a") for s in out] == [
        b"DS50:1/0/0", b"DS50:1/0/1", b"DS50:1/0/2", b"DS50:1/0/15"]


def test_scalar_cmp_edges():
    # `vector(1) == bool time()` :2963 -> all 0 (1 never equals time);
    # `vector(1) == time()` :2974 -> empty after removeEmptySeries;
    # `1 > 2` :2957 -> empty; `-1 < 2` :3127 -> the left scalar survives;
    # `time() >= bool 2` :3149 -> all 1
    out = _eval(BinOpSpec("==", bool_modifier=True),
                [scalar(1)], [t_series()])
    assert len(out) == 1
    eq(out[0].values, [0] * 6)
    out = remove_empty_series(_eval(BinOpSpec("=="), [scalar(1)],
                                    [t_series()]))
    assert out == []
    out = remove_empty_series(_eval(BinOpSpec(">"), [scalar(1)],
                                    [scalar(2)]))
    assert out == []
    out = _eval(BinOpSpec("<"), [scalar(-1)], [scalar(2)])
    eq(out[0].values, [-1] * 6)
    out = _eval(BinOpSpec(">=", bool_modifier=True), [t_series()],
                [scalar(2)])
    eq(out[0].values, [1] * 6)


def test_nan_pow_any():
    # `nan^any` exec_test.go:10038 / binaryop.Pow (funcs.go:78, issue
    # 7359): NaN^0 is NaN, overriding IEEE pow(NaN, 0) = 1
    left = [S("", [], np.full(6, NAN))]
    out = remove_empty_series(_eval(BinOpSpec("^"), left, [scalar(0)]))
    assert out == []
    # and a present base stays IEEE: x^0 = 1
    out = _eval(BinOpSpec("^"), [t_series()], [scalar(0)])
    eq(out[0].values, [1] * 6)
