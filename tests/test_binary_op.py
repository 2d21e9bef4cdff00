"""Binary-operator host logic (victoriametrics_amd/binary_op.py) vs the
reference semantics (binary_op.go + metricsql/binaryop), CPU-only: the
per-point math is delegated to the oracle (test infra), so these tests pin
the label-matching machinery — adjustBinaryOpTags fast/slow paths,
group_left/right joins, set-op key matching, metric-group reset rules —
against hand-derived expectations from the reference code."""
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd.binary_op import (
    BinOpSpec, Series, DuplicateSeriesError, binary_op_eval,
    adjust_binary_op_tags, merge_non_overlapping, is_scalar,
    MASK_AND, MASK_UNLESS, MASK_DEFAULT,
)
from victoriametrics_amd.metric_name import MetricName

NAN = math.nan


def _oracle_apply(spec, left, right, dst, drop_nan_right):
    from victoriametrics_amd.binary_op import OP_IDS
    for tl, tr, td in zip(left, right, dst):
        td.values = oracle.binop_apply(
            OP_IDS[spec.op], tl.values, tr.values,
            is_bool=spec.bool_modifier, drop_nan_right=drop_nan_right,
            fill_left=spec.fill_left, fill_right=spec.fill_right)
    return dst


def _oracle_mask(mode, lrows, lgroup, grows, goff):
    # reference loops: addRightNaNsToLeft (:549), addLeftNaNsIfNoRightNaNs
    # (:729), fillLeftNaNsWithRightValues (:622)
    for t, gi in zip(lrows, lgroup):
        rights = grows[goff[gi]:goff[gi + 1]]
        for i in range(len(t.values)):
            has = any(not math.isnan(r.values[i]) for r in rights)
            if mode == MASK_AND and not has:
                t.values[i] = NAN
            elif mode == MASK_UNLESS and has:
                t.values[i] = NAN
            elif mode == MASK_DEFAULT and math.isnan(t.values[i]):
                for r in rights:
                    if not math.isnan(r.values[i]):
                        t.values[i] = r.values[i]
                        break


def _oracle_or(groups):
    # fillLeftNaNsWithRightValuesOrMerge (:645), exact Go loop order
    for tss_left, tss_right, cm in groups:
        for li, tl in enumerate(tss_left):
            for i in range(len(tl.values)):
                left_nan = math.isnan(tl.values[i])
                for ri, tr in enumerate(tss_right):
                    mergeable = bool(cm[li, ri])
                    if left_nan and mergeable:
                        tl.values[i] = tr.values[i]
                    if not left_nan or mergeable:
                        tr.values[i] = NAN


def _eval(spec, left, right, **kw):
    return binary_op_eval(spec, left, right, apply_fn=_oracle_apply,
                          mask_fn=_oracle_mask, or_fn=_oracle_or, **kw)


def S(name, tags, values):
    return Series(MetricName(name, tags), np.asarray(values, np.float64))


def names(tss):
    return sorted(repr(t.mn) for t in tss)


# ---------------------------------------------------------------------------
# arith/cmp matching
# ---------------------------------------------------------------------------

def test_vector_op_scalar_fast_path():
    left = [S("m", [("a", "1")], [1, 2, 3]), S("m", [("a", "2")], [4, 5, 6])]
    right = [S("", [], [10, 10, 10])]
    out = _eval(BinOpSpec("+"), left, right)
    assert len(out) == 2
    # resetMetricGroupIfRequired: arith op resets the metric group
    assert all(t.mn.metric_group == b"" for t in out)
    np.testing.assert_array_equal(out[0].values, [11, 12, 13])
    np.testing.assert_array_equal(out[1].values, [14, 15, 16])


def test_scalar_op_vector_fast_path():
    left = [S("", [], [100, 100, 100])]
    right = [S("m", [("a", "1")], [1, 2, 3])]
    out = _eval(BinOpSpec("-"), left, right)
    np.testing.assert_array_equal(out[0].values, [99, 98, 97])


def test_vector_vector_default_matching():
    # ignoring nothing: full label sets must match; metric group reset
    left = [S("a", [("x", "1")], [1, 2]), S("a", [("x", "2")], [3, 4])]
    right = [S("b", [("x", "2")], [10, 10]), S("b", [("x", "1")], [5, 5])]
    out = _eval(BinOpSpec("*"), left, right)
    assert len(out) == 2
    got = {tuple(t.mn.tags): list(t.values) for t in out}
    assert got[((b"x", b"1"),)] == [5, 10]
    assert got[((b"x", b"2"),)] == [30, 40]


def test_on_matching():
    left = [S("a", [("x", "1"), ("y", "p")], [6, 8])]
    right = [S("b", [("x", "1"), ("z", "q")], [2, 4])]
    out = _eval(BinOpSpec("/", group_op="on", group_tags=["x"]), left, right)
    assert len(out) == 1
    # RemoveTagsOn keeps only x
    assert out[0].mn.tags == [(b"x", b"1")]
    assert out[0].mn.metric_group == b""
    np.testing.assert_array_equal(out[0].values, [3, 2])


def test_ignoring_matching():
    left = [S("a", [("x", "1"), ("y", "p")], [6, 8])]
    right = [S("b", [("x", "1"), ("y", "r")], [2, 4])]
    out = _eval(BinOpSpec("/", group_op="ignoring", group_tags=["y"]),
                left, right)
    assert len(out) == 1
    assert out[0].mn.tags == [(b"x", b"1")]


def test_unmatched_series_dropped():
    left = [S("a", [("x", "1")], [1, 1]), S("a", [("x", "2")], [2, 2])]
    right = [S("b", [("x", "1")], [1, 1])]
    out = _eval(BinOpSpec("+"), left, right)
    assert len(out) == 1
    assert out[0].mn.tags == [(b"x", b"1")]


def test_cmp_keeps_metric_group_without_bool():
    left = [S("m", [("a", "1")], [1, 5])]
    right = [S("", [], [3, 3])]
    out = _eval(BinOpSpec(">"), left, right)
    # non-bool cmp keeps MetricGroup (binary_op.go:509-512)
    assert out[0].mn.metric_group == b"m"
    v = out[0].values
    assert math.isnan(v[0]) and v[1] == 5


def test_cmp_bool_modifier():
    left = [S("m", [], [1, 5, NAN])]
    right = [S("", [], [3, 3, 3])]
    out = _eval(BinOpSpec(">", bool_modifier=True), left, right)
    assert out[0].mn.metric_group == b""
    v = out[0].values
    assert v[0] == 0 and v[1] == 1 and math.isnan(v[2])


def test_cmp_does_not_drop_all_nan_series():
    # binary_op.go:168-172: comparisons keep all-NaN series
    left = [S("m", [], [NAN, NAN])]
    right = [S("", [], [0, 0])]
    out = _eval(BinOpSpec("=="), left, right)
    assert len(out) == 1
    out2 = _eval(BinOpSpec("+"), left, right)
    assert out2 == []


def test_keep_metric_names():
    # keep_metric_names keeps MetricGroup in the grouping key
    # (createTimeseriesMapByTagSet:771), so differing names do not match...
    left = [S("m", [("x", "1")], [1, 2])]
    right = [S("n", [("x", "1")], [1, 2])]
    assert _eval(BinOpSpec("+", keep_metric_names=True), left, right) == []
    # ...and equal names match with the group preserved
    left = [S("m", [("x", "1")], [1, 2])]
    right = [S("m", [("x", "1")], [1, 2])]
    out = _eval(BinOpSpec("+", keep_metric_names=True), left, right)
    assert out[0].mn.metric_group == b"m"


def test_duplicate_series_error():
    left = [S("a", [("x", "1")], [1] * 5), S("b", [("x", "1")], [2] * 5)]
    right = [S("c", [("x", "1")], [3] * 5)]
    with pytest.raises(DuplicateSeriesError):
        _eval(BinOpSpec("+"), left, right)


def test_duplicate_series_merged_when_non_overlapping():
    # ensureSingleTimeseries merges series with <=2 overlapping points
    left = [S("a", [("x", "1")], [1, 1, NAN, NAN]),
            S("b", [("x", "1")], [NAN, NAN, 2, 2])]
    right = [S("c", [("x", "1")], [10, 10, 10, 10])]
    out = _eval(BinOpSpec("+"), left, right)
    assert len(out) == 1
    np.testing.assert_array_equal(out[0].values, [11, 11, 12, 12])


def test_group_left_join():
    # one right series joined to many lefts
    left = [S("req", [("pod", "a"), ("node", "n1")], [1, 2]),
            S("req", [("pod", "b"), ("node", "n1")], [3, 4])]
    right = [S("info", [("node", "n1")], [10, 10])]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=["node"],
                          join_op="group_left"), left, right)
    assert len(out) == 2
    got = {t.mn.get_tag_value("pod"): list(t.values) for t in out}
    assert got[b"a"] == [10, 20] and got[b"b"] == [30, 40]


def test_group_left_copy_tags():
    left = [S("req", [("pod", "a"), ("node", "n1")], [1, 2])]
    right = [S("info", [("node", "n1"), ("rack", "r9")], [10, 10])]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=["node"],
                          join_op="group_left", join_tags=["rack"]),
                left, right)
    assert out[0].mn.get_tag_value("rack") == b"r9"


def test_group_right_join():
    left = [S("info", [("node", "n1")], [10, 10])]
    right = [S("req", [("pod", "a"), ("node", "n1")], [1, 2]),
             S("req", [("pod", "b"), ("node", "n1")], [3, 4])]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=["node"],
                          join_op="group_right"), left, right)
    assert len(out) == 2
    got = {t.mn.get_tag_value("pod"): list(t.values) for t in out}
    assert got[b"a"] == [10, 20] and got[b"b"] == [30, 40]


def test_fill_right():
    # fill_right: unmatched left gets a NaN right series, then the fill
    # value applies pointwise
    left = [S("a", [("x", "1")], [1, 2]), S("a", [("x", "2")], [3, NAN])]
    right = [S("b", [("x", "1")], [10, 10])]
    out = _eval(BinOpSpec("+", fill_right=100.0), left, right)
    assert len(out) == 2
    got = {tuple(t.mn.tags): list(map(str, t.values)) for t in out}
    assert got[((b"x", b"1"),)] == ["11.0", "12.0"]
    assert got[((b"x", b"2"),)] == ["103.0", "nan"]


# ---------------------------------------------------------------------------
# set ops
# ---------------------------------------------------------------------------

def test_and_op():
    left = [S("a", [("x", "1")], [1, 2, 3]), S("a", [("x", "2")], [4, 5, 6])]
    right = [S("b", [("x", "1")], [NAN, 7, NAN])]
    out = _eval(BinOpSpec("and"), left, right)
    assert len(out) == 1
    v = out[0].values
    assert math.isnan(v[0]) and v[1] == 2 and math.isnan(v[2])


def test_or_op_disjoint():
    left = [S("a", [("x", "1")], [1, 1])]
    right = [S("b", [("x", "2")], [2, 2])]
    out = _eval(BinOpSpec("or"), left, right)
    assert len(out) == 2


def test_or_op_fills_gaps():
    # https://github.com/VictoriaMetrics/VictoriaMetrics/issues/7759 shape:
    # same key group, mergeable names -> left gaps filled, right consumed
    left = [S("a", [("x", "1")], [1, NAN, 3])]
    right = [S("a", [("x", "1")], [9, 2, 9])]
    out = _eval(BinOpSpec("or"), left, right)
    assert len(out) == 1
    np.testing.assert_array_equal(out[0].values, [1, 2, 3])


def test_or_scalar_fastpath_not_merged_into_vector():
    # metric_selector or on() vector(0): scalar right cannot merge into a
    # labeled left; right survives where left has no values... and the
    # right is NaN-ed where left has values
    left = [S("a", [("x", "1")], [1, NAN])]
    right = [S("", [], [5, 5])]
    out = _eval(BinOpSpec("or", group_op="on", group_tags=[]), left, right)
    assert len(out) == 2
    got = {t.mn.metric_group: t.values for t in out}
    np.testing.assert_array_equal(got[b"a"], [1, NAN])
    v = got[b""]
    assert math.isnan(v[0]) and v[1] == 5


def test_unless_op():
    left = [S("a", [("x", "1")], [1, 2, 3])]
    right = [S("b", [("x", "1")], [NAN, 9, NAN])]
    out = _eval(BinOpSpec("unless"), left, right)
    assert len(out) == 1
    v = out[0].values
    assert v[0] == 1 and math.isnan(v[1]) and v[2] == 3


def test_unless_no_match_passthrough():
    left = [S("a", [("x", "1")], [1, 2])]
    right = [S("b", [("x", "2")], [9, 9])]
    out = _eval(BinOpSpec("unless"), left, right)
    assert len(out) == 1
    np.testing.assert_array_equal(out[0].values, [1, 2])


def test_if_op_with_scalar_right():
    # seriesByKey scalar fallback (binary_op.go:741-755)
    left = [S("a", [("x", "1")], [1, 2])]
    right = [S("", [], [NAN, 1])]
    out = _eval(BinOpSpec("if"), left, right)
    assert len(out) == 1
    v = out[0].values
    assert math.isnan(v[0]) and v[1] == 2


def test_ifnot_op():
    left = [S("a", [("x", "1")], [1, 2])]
    right = [S("b", [("x", "1")], [NAN, 1])]
    out = _eval(BinOpSpec("ifnot"), left, right)
    v = out[0].values
    assert v[0] == 1 and math.isnan(v[1])


def test_default_op():
    left = [S("a", [("x", "1")], [1, NAN, 3])]
    right = [S("b", [("x", "1")], [9, 2, 9])]
    out = _eval(BinOpSpec("default"), left, right)
    assert len(out) == 1
    np.testing.assert_array_equal(out[0].values, [1, 2, 3])


def test_default_empty_left_returns_right():
    out = _eval(BinOpSpec("default"), [], [S("b", [], [1, 2])])
    assert len(out) == 1


# ---------------------------------------------------------------------------
# oracle elementwise semantics pins (metricsql/binaryop/funcs.go)
# ---------------------------------------------------------------------------

def test_binop_scalar_semantics():
    # bool-modifier wrapper (binary_op.go:137-151): NaN left -> NaN output
    # BEFORE the comparison runs, even for !=
    assert math.isnan(oracle.binop_scalar("!=", NAN, 1.0, is_bool=True))
    assert math.isnan(oracle.binop_scalar("==", NAN, NAN, is_bool=True))
    assert oracle.binop_scalar("!=", 1.0, NAN, is_bool=True) == 1.0
    assert oracle.binop_scalar(">", 2.0, 1.0) == 2.0
    assert math.isnan(oracle.binop_scalar(">", 1.0, 2.0))
    assert math.isnan(oracle.binop_scalar("^", NAN, 0.0))  # Go: NaN^0 = NaN
    assert oracle.binop_scalar("%", -7.5, 2.0) == math.fmod(-7.5, 2.0)
    assert oracle.binop_scalar("atan2", 1.0, 1.0) == math.atan2(1.0, 1.0)
    assert oracle.binop_scalar("default", NAN, 5.0) == 5.0
    assert oracle.binop_scalar("or", 3.0, 5.0) == 3.0
    assert oracle.binop_scalar("or", NAN, 5.0) == 5.0
    assert math.isnan(oracle.binop_scalar("and", 3.0, NAN))
    assert oracle.binop_scalar("and", 3.0, 5.0) == 3.0


def test_drop_nan_right():
    # binary_op.go:199-205: cmp with vector right drops NaN-right points
    out = oracle.binop_apply(">", [5.0, 5.0], [1.0, NAN], drop_nan_right=True)
    assert out[0] == 5.0 and math.isnan(out[1])
    out2 = oracle.binop_apply(">", [5.0, 5.0], [1.0, NAN])
    assert out2[0] == 5.0 and math.isnan(out2[1])  # NaN cmp false anyway


def test_merge_non_overlapping_limits():
    a = Series(MetricName("a"), np.asarray([1.0, NAN, NAN, NAN]))
    b = Series(MetricName("a"), np.asarray([NAN, 2.0, 3.0, NAN]))
    assert merge_non_overlapping(a, b)
    np.testing.assert_array_equal(a.values, [1, 2, 3, NAN])
    # >2 overlaps refused
    c = Series(MetricName("a"), np.asarray([1.0, 1, 1, 1]))
    d = Series(MetricName("a"), np.asarray([2.0, 2, 2, NAN]))
    assert not merge_non_overlapping(c, d)
    # short series (<=2 points both sides) refused
    e = Series(MetricName("a"), np.asarray([1.0, NAN]))
    f = Series(MetricName("a"), np.asarray([NAN, 2.0]))
    assert not merge_non_overlapping(e, f)


# ---------------------------------------------------------------------------
# TestExecSuccess vector-matching pins (exec_test.go:3629-3810), CPU via
# the oracle apply harness.  TIME = [1000..2000]; left = time()/10 with
# {foo=bar, xx=yy, __name__=qwert}.
# ---------------------------------------------------------------------------

TIME_V = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def _left_qwert():
    return S("qwert", [("foo", "bar"), ("xx", "yy")], TIME_V / 10)


def _tagmap(tss):
    out = {}
    for t in tss:
        key = (t.mn.metric_group,
               tuple(sorted((k, v) for k, v in t.mn.tags)))
        assert key not in out
        out[key] = t.values
    return out


def test_exec_group_left_additional_tag():
    # + on(foo) group_left(op): two right series differ by op and do not
    # overlap in time -> two outputs, op copied, left tags kept
    lo = np.where(TIME_V < 1400, TIME_V, NAN)
    hi = np.where(TIME_V >= 1400, TIME_V, NAN)
    right = [S("", [("foo", "bar"), ("op", "le")], lo),
             S("", [("foo", "bar"), ("op", "ge")], hi)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_left", join_tags=["op"]),
                [_left_qwert()], right)
    m = _tagmap(out)
    k_le = (b"", ((b"foo", b"bar"), (b"op", b"le"), (b"xx", b"yy")))
    k_ge = (b"", ((b"foo", b"bar"), (b"op", b"ge"), (b"xx", b"yy")))
    assert set(m) == {k_le, k_ge}
    np.testing.assert_array_equal(
        m[k_le].view(np.int64),
        np.asarray([1100.0, 1320, NAN, NAN, NAN, NAN]).view(np.int64))
    np.testing.assert_array_equal(
        m[k_ge].view(np.int64),
        np.asarray([NAN, NAN, 1540.0, 1760, 1980, 2200]).view(np.int64))


def test_exec_on_duplicate_nonoverlapping_merged():
    # + on(foo) with NO join modifier: the two non-overlapping right
    # series merge into one (mergeNonOverlappingTimeseries); the output
    # name is the on() projection {foo}
    lo = np.where(TIME_V < 1400, TIME_V, NAN)
    hi = np.where(TIME_V >= 1400, TIME_V, NAN)
    right = [S("", [("foo", "bar"), ("op", "le")], lo),
             S("", [("foo", "bar"), ("op", "ge")], hi)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"]),
                [_left_qwert()], right)
    assert len(out) == 1
    assert out[0].mn.metric_group == b""
    assert sorted(out[0].mn.tags) == [(b"foo", b"bar")]
    np.testing.assert_allclose(out[0].values,
                               [1100, 1320, 1540, 1760, 1980, 2200])


def test_exec_group_left_empty_duplicate_nonoverlapping():
    # group_left() keeps the left extra tags (xx=yy) while merging
    lo = np.where(TIME_V < 1400, TIME_V, NAN)
    hi = np.where(TIME_V >= 1400, TIME_V, NAN)
    right = [S("", [("foo", "bar"), ("op", "le")], lo),
             S("", [("foo", "bar"), ("op", "ge")], hi)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_left"),
                [_left_qwert()], right)
    assert len(out) == 1
    assert sorted(out[0].mn.tags) == [(b"foo", b"bar"), (b"xx", b"yy")]
    np.testing.assert_allclose(out[0].values,
                               [1100, 1320, 1540, 1760, 1980, 2200])


def test_exec_group_left_copies_name():
    # group_left(__name__) copies the right metric group onto the result
    right = [S("aaa", [("foo", "bar")], TIME_V.copy())]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_left", join_tags=["__name__"]),
                [_left_qwert()], right)
    assert len(out) == 1
    assert out[0].mn.metric_group == b"aaa"
    assert sorted(out[0].mn.tags) == [(b"foo", b"bar"), (b"xx", b"yy")]
    np.testing.assert_allclose(out[0].values,
                               [1100, 1320, 1540, 1760, 1980, 2200])


def test_exec_group_right_copies_left_tag():
    # + on(foo) group_right(xx): many right series, xx copied from left
    right = [S("aaa", [("foo", "bar")], TIME_V.copy()),
             S("yyy", [("foo", "bar"), ("ppp", "123")], TIME_V + 3)]
    out = _eval(BinOpSpec("+", group_op="on", group_tags=["foo"],
                          join_op="group_right", join_tags=["xx"]),
                [_left_qwert()], right)
    m = _tagmap(out)
    k1 = (b"", ((b"foo", b"bar"), (b"xx", b"yy")))
    k2 = (b"", ((b"foo", b"bar"), (b"ppp", b"123"), (b"xx", b"yy")))
    assert set(m) == {k1, k2}
    np.testing.assert_allclose(m[k1], [1100, 1320, 1540, 1760, 1980, 2200])
    np.testing.assert_allclose(m[k2], [1103, 1323, 1543, 1763, 1983, 2203])


def test_exec_on_empty_group_left_scalar_ish():
    # (a or b) * on() group_left 2: on() makes one join key; the scalar
    # side is a plain series with no tags
    left = [S("", [("foo", "bar")], TIME_V.copy()),
            S("", [("foo", "qwert")], np.full(6, 10.0))]
    right = [S("", [], np.full(6, 2.0))]
    out = _eval(BinOpSpec("*", group_op="on", group_tags=[],
                          join_op="group_left"), left, right)
    m = _tagmap(out)
    k1 = (b"", ((b"foo", b"bar"),))
    k2 = (b"", ((b"foo", b"qwert"),))
    assert set(m) == {k1, k2}
    np.testing.assert_allclose(m[k1], TIME_V * 2)
    np.testing.assert_allclose(m[k2], [20.0] * 6)
