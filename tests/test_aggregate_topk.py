"""topk/bottomk family + outliersk over resident Series — host mirror of
newAggrFuncTopK / newAggrFuncRangeTopK / aggrFuncOutliersK (aggr.go), pinned
against the TestExecSuccess expected arrays (exec_test.go:7071-7430,
7826-7878).  The common fixture is the reference's own:
`label_set(10, "foo", "bar") or label_set(time()/150, "baz", "sss")` on the
grid start=1000e3 end=2000e3 step=200e3 (seconds: 1000..2000)."""
import math

import numpy as np

from victoriametrics_amd import aggregate as agg
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

NAN = math.nan
TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def A():
    return Series(MetricName(b"", [(b"foo", b"bar")]), np.full(6, 10.0))


def B(div=150.0):
    return Series(MetricName(b"", [(b"baz", b"sss")]), TIME / div)


def by_tag(rvs):
    """{first_tag_pair: values}"""
    out = {}
    for s in rvs:
        tag = tuple((k.decode(), v.decode()) for k, v in s.mn.tags)
        assert tag not in out
        out[tag] = s.values
    return out


def eq(got, want):
    np.testing.assert_array_equal(np.asarray(got, np.float64).view(np.int64),
                                  np.asarray(want, np.float64).view(np.int64))


B_VALS = [6.666666666666667, 8, 9.333333333333334,
          10.666666666666666, 12, 13.333333333333334]


def test_topk_negative_and_nan_k():
    assert agg.aggregate("topk", [A(), B()], arg=-1) == []
    assert agg.aggregate("topk", [A(), B()], arg=NAN) == []


def test_topk_1():
    m = by_tag(agg.aggregate("topk", [A(), B()], arg=1))
    eq(m[(("baz", "sss"),)], [NAN, NAN, NAN] + B_VALS[3:])
    eq(m[(("foo", "bar"),)], [10, 10, 10, NAN, NAN, NAN])


def test_topk_all():
    for k in (2, 100500):
        m = by_tag(agg.aggregate("topk", [A(), B()], arg=k))
        eq(m[(("foo", "bar"),)], [10.0] * 6)
        eq(m[(("baz", "sss"),)], B_VALS)


def test_topk_nan_timeseries():
    # topk(1, label_set(NaN,...) or B): the NaN series is dropped whole
    a = A()
    a.values[:] = NAN
    m = by_tag(agg.aggregate("topk", [a, B()], arg=1))
    assert list(m) == [(("baz", "sss"),)]
    eq(m[(("baz", "sss"),)], B_VALS)


def test_bottomk_1():
    m = by_tag(agg.aggregate("bottomk", [A(), B()], arg=1))
    eq(m[(("foo", "bar"),)], [NAN, NAN, NAN, 10, 10, 10])
    eq(m[(("baz", "sss"),)], B_VALS[:3] + [NAN, NAN, NAN])


def test_range_topk_single_winner():
    # exec_test.go topk_min/bottomk_min/topk_max/bottomk_max/topk_avg/
    # bottomk_avg/topk_median/topk_last (1): whole-series selection
    # NOTE bottomk_avg/topk_avg: avg(A) == avg(B) == 10.0 exactly — the tie
    # resolves by input order under Go's 2-element insertion sort, which the
    # stable sort here reproduces; the reference expects B (baz) for both.
    for name, tag in [("topk_min", "foo"), ("bottomk_min", "baz"),
                      ("topk_max", "baz"), ("bottomk_max", "foo"),
                      ("topk_avg", "baz"), ("bottomk_avg", "baz"),
                      ("topk_median", "baz"), ("topk_last", "baz")]:
        rvs = agg.aggregate(name, [A(), B()], arg=1)
        assert len(rvs) == 1, name
        assert rvs[0].mn.tags[0][0].decode() == tag, name
        eq(rvs[0].values, [10.0] * 6 if tag == "foo" else B_VALS)


def test_range_bottomk_median_last():
    # with time()/15 the B series' median/last exceed 10 -> A selected
    for name in ("bottomk_median", "bottomk_last"):
        rvs = agg.aggregate(name, [A(), B(div=15.0)], arg=1)
        assert len(rvs) == 1
        eq(rvs[0].values, [10.0] * 6)


def test_topk_max_remaining_sum():
    m = by_tag(agg.aggregate("topk_max", [A(), B()], arg=1,
                             remaining_sum_tag="remaining_sum=foo"))
    eq(m[(("baz", "sss"),)], B_VALS)
    eq(m[(("remaining_sum", "foo"),)], [10.0] * 6)
    assert (("foo", "bar"),) not in m


def test_topk_max_remaining_sum_all_selected():
    # k >= len(series): the remaining-sum series is all-NaN and removed
    for k in (2, 3):
        m = by_tag(agg.aggregate("topk_max", [A(), B()], arg=k,
                                 remaining_sum_tag="remaining_sum"))
        assert set(m) == {(("foo", "bar"),), (("baz", "sss"),)}
        eq(m[(("baz", "sss"),)], B_VALS)


def test_outliersk():
    def fix(v):
        return [Series(MetricName(b"", [(b"foo", b"bar")]), np.full(6, v)),
                Series(MetricName(b"", [(b"baz", b"sss")]), TIME.copy())]

    assert agg.aggregate("outliersk", fix(1300.0), arg=0) == []
    rvs = agg.aggregate("outliersk", fix(2000.0), arg=1)
    assert len(rvs) == 1 and rvs[0].mn.tags[0][0] == b"baz"
    eq(rvs[0].values, TIME)
    m = by_tag(agg.aggregate("outliersk", fix(1300.0), arg=3))
    eq(m[(("foo", "bar"),)], [1300.0] * 6)
    eq(m[(("baz", "sss"),)], TIME)


def test_pointwise_k_array():
    # per-point k (ks is a scalar series in the reference)
    ks = np.asarray([0.0, 1, 2, 1, 0, 2])
    m = by_tag(agg.aggregate("topk", [A(), B()], arg=ks))
    eq(m[(("foo", "bar"),)], [NAN, 10, 10, NAN, NAN, 10])
    eq(m[(("baz", "sss"),)], [NAN, NAN, 9.333333333333334,
                              10.666666666666666, NAN, 13.333333333333334])


def test_grouped_range_topk():
    # by(job): selection happens within each group independently
    def S(job, inst, vals):
        return Series(MetricName(b"m", [(b"job", job.encode()),
                                        (b"inst", inst.encode())]),
                      np.asarray(vals, np.float64))

    series = [S("a", "1", [1] * 6), S("a", "2", [5] * 6),
              S("b", "1", [9] * 6), S("b", "2", [2] * 6)]
    rvs = agg.aggregate("topk_max", series, modifier_op="by",
                        modifier_args=["job"], arg=1)
    got = {(s.mn.get_tag_value("job"), s.mn.get_tag_value("inst"))
           for s in rvs}
    assert got == {(b"a", b"2"), (b"b", b"1")}


def test_summary_helpers_against_go_semantics():
    # minValue/maxValue/avgValue/medianValue/lastValue (aggr.go:804-860)
    v = [NAN, 3.0, 1.0, NAN, 2.0, NAN]
    assert agg._min_value(v) == 1.0
    assert agg._max_value(v) == 3.0
    assert agg._avg_value(v) == 2.0
    assert agg._median_value(v) == 2.0
    assert agg._last_value(v) == 2.0
    assert math.isnan(agg._min_value([NAN, NAN]))
    assert math.isnan(agg._last_value([]))
    # quantileSorted edge semantics
    assert agg.quantile_sorted(-0.5, [1.0, 2.0]) == -math.inf
    assert agg.quantile_sorted(1.5, [1.0, 2.0]) == math.inf
    assert math.isnan(agg.quantile_sorted(NAN, [1.0]))
    assert math.isnan(agg.quantile_sorted(0.5, []))


def test_get_int_k():
    assert agg._get_int_k(NAN, 5) == 0
    assert agg._get_int_k(-3.0, 5) == 0
    assert agg._get_int_k(2.9, 5) == 2
    assert agg._get_int_k(math.inf, 5) == 5
    assert agg._get_int_k(1e300, 5) == 5


def test_aggr_dispatch_covers_reference_map():
    # the 37 keys of aggrFuncs (aggr.go:17-56) each have a host entry point:
    # aggregate() for 34, count_values()/quantiles()/histogram_aggregate()
    # for the three multi-output metadata funcs
    reference_names = {
        "any", "avg", "bottomk", "bottomk_avg", "bottomk_last",
        "bottomk_max", "bottomk_median", "bottomk_min", "count",
        "count_values", "distinct", "geomean", "group", "histogram",
        "limitk", "mad", "max", "median", "min", "mode", "outliers_iqr",
        "outliers_mad", "outliersk", "quantile", "quantiles", "share",
        "stddev", "stdvar", "sum", "sum2", "topk", "topk_avg", "topk_last",
        "topk_max", "topk_median", "topk_min", "zscore"}
    dispatched = (agg.REDUCERS | agg.PER_SERIES | set(agg._RANGE_TOPK) |
                  {"quantile", "any", "limitk", "outliers_iqr",
                   "outliers_mad", "topk", "bottomk", "outliersk"})
    separate = {"count_values": agg.count_values,
                "quantiles": agg.quantiles,
                "histogram": agg.histogram_aggregate}
    missing = reference_names - dispatched - set(separate)
    assert not missing, f"aggrFuncs without a host entry point: {missing}"
    for fn in separate.values():
        assert callable(fn)


def test_pointwise_topk_randomized_vs_numpy():
    # with distinct finite values the per-point selection is exactly
    # "keep the k largest (smallest for bottomk) at each grid index"
    rng = np.random.default_rng(7)
    for trial in range(20):
        n_series, n_grid = int(rng.integers(1, 8)), int(rng.integers(1, 12))
        vals = rng.permutation(n_series * n_grid).astype(np.float64)
        vals = vals.reshape(n_series, n_grid)
        k = int(rng.integers(0, n_series + 2))
        series = [Series(MetricName(b"", [(b"i", str(i).encode())]),
                         vals[i].copy()) for i in range(n_series)]
        for name, keep_largest in (("topk", True), ("bottomk", False)):
            got = agg.aggregate(name, [s.copy_shallow() for s in series],
                                arg=k)
            out = np.full((n_series, n_grid), math.nan)
            for s in got:
                out[int(s.mn.get_tag_value("i"))] = s.values
            expect = np.full((n_series, n_grid), math.nan)
            kn = min(k, n_series)
            for j in range(n_grid):
                order = np.argsort(vals[:, j])
                # note: order[-kn:] would be wrong for kn == 0
                sel = order[n_series - kn:] if keep_largest else order[:kn]
                expect[sel, j] = vals[sel, j]
            # series that end up all-NaN are removed, already reflected
            np.testing.assert_array_equal(
                out.view(np.int64), expect.view(np.int64),
                err_msg=f"{name} trial {trial} k={k}")


def test_bottomk_exec_pin():
    """`bottomk(1, ...)` exec_test.go:7398: per-point bottom-1 across a
    constant-10 series, time()/150 and a NaN-only comparison series —
    NaN rows never win, and the per-point winner flips when time()/150
    crosses 10."""
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    def S(name, tags, values):
        return Series(MetricName(name.encode(), [(k.encode(), v.encode())
                                                 for k, v in tags]), values)
    s1 = S("", [("foo", "bar")], np.full(6, 10.0))
    s2 = S("", [("baz", "sss")], TIME / 150)
    s3 = S("", [("a", "b")], np.where(TIME < 100, TIME, NAN))
    out = agg.aggregate("bottomk", [s1, s2, s3], arg=1)
    got = {tuple(t.mn.tags): t.values for t in out
           if not np.isnan(t.values).all()}
    assert set(got) == {((b"foo", b"bar"),), ((b"baz", b"sss"),)}
    v1 = got[((b"foo", b"bar"),)]
    v2 = got[((b"baz", b"sss"),)]
    np.testing.assert_array_equal(v1[3:], [10, 10, 10])
    assert np.isnan(v1[:3]).all()
    np.testing.assert_allclose(
        v2[:3], [6.666666666666667, 8, 9.333333333333334], rtol=0, atol=0)
    assert np.isnan(v2[3:]).all()
