"""Exec tail (exec.go:37-131): result ordering, duplicate detection,
first-point truncation, rounding, the sort-exemption table."""
import math

import numpy as np
import pytest

from victoriametrics_amd import exec_surface as xs
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

NAN = math.nan


def S(name, tags, values):
    return Series(MetricName(name, tags), np.asarray(values, np.float64))


def test_may_sort_results():
    assert not xs.may_sort_results("func", "sort")
    assert not xs.may_sort_results("func", "SORT_DESC")
    assert not xs.may_sort_results("func", "limit_offset")
    assert xs.may_sort_results("func", "abs")
    assert not xs.may_sort_results("aggr", "topk_median")
    assert xs.may_sort_results("aggr", "sum")
    assert not xs.may_sort_results("binop", "or")
    assert xs.may_sort_results("binop", "+")
    assert xs.may_sort_results("rollup", "rate")


def test_sort_by_metric_name():
    # group first, then sorted tags (key then value), shorter prefix first
    a = S("b", [], [1])
    b = S("a", [("z", "1")], [1])
    c = S("a", [("a", "2")], [1])
    d = S("a", [("a", "2"), ("b", "1")], [1])
    e = S("a", [], [1])
    out = xs.sort_series_by_metric_name([a, b, c, d, e])
    assert out == [e, c, d, b, a]


def test_timeseries_to_result_pipeline():
    a = S("m", [("i", "2")], [1.234567, 2.0])
    b = S("m", [("i", "1")], [5.678999, NAN])
    empty = S("m", [("i", "3")], [NAN, NAN])
    out = xs.timeseries_to_result([a, b, empty], may_sort=True,
                                  round_digits=2)
    assert out == [b, a]  # empty removed, sorted by tags
    assert list(out[0].values) == [5.68] or out[0].values[0] == 5.68
    assert out[1].values[0] == 1.23
    # round_digits=100 (default) leaves values alone
    c = S("m", [], [1.23456789])
    out = xs.timeseries_to_result([c], may_sort=False)
    assert out[0].values[0] == 1.23456789


def test_first_point_only():
    a = S("m", [], [7.0, 8.0, 9.0])
    out = xs.timeseries_to_result([a], may_sort=True, first_point_only=True)
    assert list(out[0].values) == [7.0]
    # an all-NaN first point empties the series
    b = S("m", [], [NAN, 8.0])
    assert xs.timeseries_to_result([b], may_sort=True,
                                   first_point_only=True) == []


def test_duplicate_output_series():
    a = S("m", [("x", "1")], [1.0])
    b = S("m", [("x", "1")], [2.0])
    with pytest.raises(xs.DuplicateOutputSeriesError):
        xs.timeseries_to_result([a, b], may_sort=True)
    # tag-order-insensitive duplicate detection
    c = S("m", [("a", "1"), ("b", "2")], [1.0])
    d = S("m", [("b", "2"), ("a", "1")], [2.0])
    with pytest.raises(xs.DuplicateOutputSeriesError):
        xs.timeseries_to_result([c, d], may_sort=True)


def test_max_response_series_guard():
    tss = [S("m", [("i", str(i))], [1.0]) for i in range(5)]
    with pytest.raises(ValueError, match="maxResponseSeries"):
        xs.timeseries_to_result(tss, may_sort=True, max_response_series=4)
    assert len(xs.timeseries_to_result(tss, may_sort=True,
                                       max_response_series=5)) == 5
    assert len(xs.timeseries_to_result(tss, may_sort=True,
                                       max_response_series=0)) == 5


def test_plan_with_at_modifier():
    """evalRollupFunc `@` handling (eval.go:903-950): single-point grid at
    the first non-NaN `@` value (seconds -> ms), broadcast to the grid."""
    import numpy as np
    from victoriametrics_amd import engine
    plan, ts, bc = engine.plan_with_at(
        "rate", 1000_000, 2000_000, 200_000,
        [float("nan"), 1600.0, 1700.0], window=300_000)
    assert plan._c.start == plan._c.end == 1_600_000
    np.testing.assert_array_equal(
        ts, np.arange(1000_000, 2000_001, 200_000))
    out = bc(np.asarray([[2.5], [3.5]]))
    assert out.shape == (2, 6)
    assert (out[0] == 2.5).all() and (out[1] == 3.5).all()
    import pytest
    with pytest.raises(engine.VmGpuError):
        engine.plan_with_at("rate", 0, 100, 10, [[1.0], [2.0]])
    with pytest.raises(engine.VmGpuError):
        engine.plan_with_at("rate", 0, 100, 10, [float("nan")])
