"""Non-incremental cross-series aggregates: oracle pins (mode vectors
transcribed from aggr_test.go TestModeNoNaNs; hand-derived quantile/mad/
share/zscore cases from aggr.go) — CPU side."""
import math

import numpy as np
import pytest

import oracle

NAN = math.nan


def _one_group(columns):
    """columns: list of per-series single values -> [n x 1] matrix + CSR."""
    v = np.asarray(columns, dtype=np.float64).reshape(-1, 1)
    gr = np.arange(v.shape[0], dtype=np.uint32)
    go = np.asarray([0, v.shape[0]], np.uint64)
    return v, gr, go


# modeNoNaNs vectors with the NaN seed (aggr_test.go:70-77); the aggregate
# path always seeds prevValue=NaN (aggrFuncMode :446)
@pytest.mark.parametrize("vals,expected", [
    ([123], 123),
    ([1, 2, 3], 1),
    ([1, 2, 2], 2),
    ([1, 1, 2], 1),
    ([1, 1, 1], 1),
    ([1, 2, 2, 3], 2),
    ([1, 1, 2, 2, 3, 3, 3], 3),
])
def test_mode_vectors(vals, expected):
    v, gr, go = _one_group(vals)
    out = oracle.colagg("mode", v, gr, go)
    assert out[0, 0] == expected


def test_mode_empty():
    v, gr, go = _one_group([NAN, NAN])
    out = oracle.colagg("mode", v, gr, go)
    assert math.isnan(out[0, 0])


def test_median_quantile():
    v, gr, go = _one_group([4.0, 1.0, 3.0, 2.0])
    assert oracle.colagg("median", v, gr, go)[0, 0] == 2.5
    assert oracle.colagg("quantile", v, gr, go, phi=0.0)[0, 0] == 1.0
    assert oracle.colagg("quantile", v, gr, go, phi=1.0)[0, 0] == 4.0
    # NaNs are filtered, not counted (quantile(), aggr.go)
    v2, gr2, go2 = _one_group([4.0, NAN, 2.0])
    assert oracle.colagg("median", v2, gr2, go2)[0, 0] == 3.0


def test_mad():
    v, gr, go = _one_group([1.0, 2.0, 3.0, 100.0])
    # median 2.5; |v-2.5| = [1.5, .5, .5, 97.5]; median of sorted = 1.0
    assert oracle.colagg("mad", v, gr, go)[0, 0] == 1.0


def test_stddev_stdvar():
    v, gr, go = _one_group([2.0, 4.0, 4.0, 4.0, 5.0, 5.0, 7.0, 9.0])
    out = oracle.colagg("stdvar", v, gr, go)
    assert abs(out[0, 0] - 4.0) < 1e-12
    out2 = oracle.colagg("stddev", v, gr, go)
    assert abs(out2[0, 0] - 2.0) < 1e-12
    # single member: exactly zero (aggrFuncStdvar fast path semantics)
    v1, gr1, go1 = _one_group([5.0])
    assert oracle.colagg("stdvar", v1, gr1, go1)[0, 0] == 0.0
    # all-NaN: NaN
    vn, grn, gon = _one_group([NAN])
    assert math.isnan(oracle.colagg("stdvar", vn, grn, gon)[0, 0])


def test_distinct():
    v, gr, go = _one_group([1.0, 1.0, 2.0, NAN, 3.0, 2.0])
    assert oracle.colagg("distinct", v, gr, go)[0, 0] == 3.0
    vn, grn, gon = _one_group([NAN])
    assert math.isnan(oracle.colagg("distinct", vn, grn, gon)[0, 0])


def test_share():
    # negative and NaN values -> NaN; others divided by non-negative sum
    v, gr, go = _one_group([2.0, 6.0, -1.0, NAN])
    out = oracle.colagg("share", v, gr, go)
    assert out[0, 0] == 0.25 and out[1, 0] == 0.75
    assert math.isnan(out[2, 0]) and math.isnan(out[3, 0])


def test_zscore():
    v, gr, go = _one_group([1.0, 2.0, 3.0])
    out = oracle.colagg("zscore", v, gr, go)
    got = out[:, 0]
    assert abs(got[1]) < 1e-12
    assert abs(got[0] + got[2]) < 1e-12
    # all-NaN column: values pass through
    vn, grn, gon = _one_group([NAN, NAN])
    outn = oracle.colagg("zscore", vn, grn, gon)
    assert math.isnan(outn[0, 0])


def test_iqr_bounds_and_filter():
    vals = [1.0, 2.0, 3.0, 4.0, 100.0]
    v, gr, go = _one_group(vals)
    lower, upper = oracle.colagg("iqr_bounds", v, gr, go)
    # q25 = 2, q75 = 4 -> iqr*1.5 = 3 -> bounds [-1, 7]
    assert lower[0, 0] == -1.0 and upper[0, 0] == 7.0
    flags = oracle.colagg_filter("iqr", v, np.zeros(5, np.int32),
                                 lower, upper)
    assert list(flags) == [0, 0, 0, 0, 1]


def test_mad_filter():
    vals = [1.0, 2.0, 3.0, 100.0]
    v, gr, go = _one_group(vals)
    med = oracle.colagg("median", v, gr, go)
    mad = oracle.colagg("mad", v, gr, go)
    tol = 3.0
    flags = oracle.colagg_filter("mad", v, np.zeros(4, np.int32),
                                 med, mad * tol)
    # |v - 2.5| > 3*1.0 -> only 100.0
    assert list(flags) == [0, 0, 0, 1]


def test_multi_group_multi_point():
    rng = np.random.default_rng(9)
    v = rng.standard_normal((10, 7))
    v[rng.random((10, 7)) < 0.2] = NAN
    gr = np.asarray([0, 1, 2, 3, 4, 5, 6, 7, 8, 9], np.uint32)
    go = np.asarray([0, 4, 10], np.uint64)
    out = oracle.colagg("median", v, gr, go)
    for grp, (lo, hi) in enumerate([(0, 4), (4, 10)]):
        for g in range(7):
            col = v[lo:hi, g]
            col = col[~np.isnan(col)]
            if len(col) == 0:
                assert math.isnan(out[grp, g])
            else:
                exp = np.quantile(np.sort(col), 0.5) if len(col) else NAN
                assert abs(out[grp, g] - float(np.median(col))) < 1e-12


# ---------------------------------------------------------------------------
# GPU parity
# ---------------------------------------------------------------------------

@pytest.mark.gpu
@pytest.mark.parametrize("op", ["median", "quantile", "mad", "stddev",
                                "stdvar", "mode", "distinct"])
def test_gpu_colagg_reduce(op):
    from victoriametrics_amd import engine
    rng = np.random.default_rng(abs(hash(op)) % 2**31)
    n_series, n_grid, n_groups = 300, 120, 23
    v = rng.standard_normal((n_series, n_grid)) * 50
    v[rng.random((n_series, n_grid)) < 0.2] = NAN
    # quantize some values so mode/distinct see duplicates
    v[::3] = np.round(v[::3])
    gids = rng.integers(0, n_groups, n_series)
    order = np.argsort(gids, kind="stable").astype(np.uint32)
    go = np.zeros(n_groups + 1, np.uint64)
    for g in gids:
        go[g + 1] += 1
    go = np.cumsum(go).astype(np.uint64)
    phi = 0.75
    got = engine.colagg(op, v, order, go, phi=phi)
    exp = oracle.colagg(op, v, order, go, phi=phi)
    np.testing.assert_array_equal(got.view(np.int64), exp.view(np.int64),
                                  err_msg=op)


@pytest.mark.gpu
@pytest.mark.parametrize("op", ["share", "zscore"])
def test_gpu_colagg_per_series(op):
    from victoriametrics_amd import engine
    rng = np.random.default_rng(abs(hash(op)) % 2**31)
    n_series, n_grid = 200, 80
    v = rng.standard_normal((n_series, n_grid)) * 10
    v[rng.random((n_series, n_grid)) < 0.25] = NAN
    gids = rng.integers(0, 11, n_series)
    order = np.argsort(gids, kind="stable").astype(np.uint32)
    go = np.zeros(12, np.uint64)
    for g in gids:
        go[g + 1] += 1
    go = np.cumsum(go).astype(np.uint64)
    got = engine.colagg(op, v, order, go)
    exp = oracle.colagg(op, v, order, go)
    np.testing.assert_array_equal(got.view(np.int64), exp.view(np.int64),
                                  err_msg=op)


@pytest.mark.gpu
def test_gpu_iqr_bounds_and_filters():
    from victoriametrics_amd import engine
    rng = np.random.default_rng(77)
    n_series, n_grid = 150, 60
    v = rng.standard_normal((n_series, n_grid)) * 20
    v[rng.random((n_series, n_grid)) < 0.1] = NAN
    gids = rng.integers(0, 7, n_series).astype(np.int32)
    order = np.argsort(gids, kind="stable").astype(np.uint32)
    go = np.zeros(8, np.uint64)
    for g in gids:
        go[g + 1] += 1
    go = np.cumsum(go).astype(np.uint64)
    glo, gup = engine.colagg("iqr_bounds", v, order, go)
    elo, eup = oracle.colagg("iqr_bounds", v, order, go)
    np.testing.assert_array_equal(glo.view(np.int64), elo.view(np.int64))
    np.testing.assert_array_equal(gup.view(np.int64), eup.view(np.int64))
    gf = engine.colagg_filter("iqr", v, gids, glo, gup)
    ef = oracle.colagg_filter("iqr", v, gids, elo, eup)
    np.testing.assert_array_equal(gf, ef)
    med = engine.colagg("median", v, order, go)
    mad = engine.colagg("mad", v, order, go)
    gf2 = engine.colagg_filter("mad", v, gids, med, mad * 3.0)
    ef2 = oracle.colagg_filter("mad", v, gids, med, mad * 3.0)
    np.testing.assert_array_equal(gf2, ef2)


# ---------------------------------------------------------------------------
# series-level aggregate dispatch (aggregate.py)
# ---------------------------------------------------------------------------

def _series_set():
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    out = []
    for pod, node, vals in [("a", "n1", [1, 2]), ("b", "n1", [3, 4]),
                            ("c", "n2", [5, NAN]), ("d", "n2", [NAN, NAN])]:
        out.append(Series(MetricName("m", [("pod", pod), ("node", node)]),
                          np.asarray(vals, np.float64)))
    return out


def test_prepare_series_by_without():
    from victoriametrics_amd import aggregate as agg
    groups = agg.prepare_series(_series_set(), "by", ["node"])
    # the all-NaN series is dropped first; two node groups remain
    assert len(groups) == 2
    sizes = sorted(len(m) for _, m in groups)
    assert sizes == [1, 2]
    gmn = groups[0][0]
    assert gmn.tags == [(b"node", b"n1")] and gmn.metric_group == b""
    groups_w = agg.prepare_series(_series_set(), "without", ["pod"])
    assert len(groups_w) == 2
    assert groups_w[0][0].get_tag_value("node") == b"n1"


@pytest.mark.gpu
def test_aggregate_sum_by_node():
    from victoriametrics_amd import aggregate as agg
    out = agg.aggregate("sum", _series_set(), "by", ["node"])
    got = {s.mn.get_tag_value("node"): list(s.values) for s in out}
    assert got[b"n1"] == [4.0, 6.0]
    assert got[b"n2"][0] == 5.0 and math.isnan(got[b"n2"][1])


@pytest.mark.gpu
def test_aggregate_median_and_quantile():
    from victoriametrics_amd import aggregate as agg
    out = agg.aggregate("median", _series_set(), "by", ["node"])
    got = {s.mn.get_tag_value("node"): list(s.values) for s in out}
    assert got[b"n1"] == [2.0, 3.0]
    out2 = agg.aggregate("quantile", _series_set(), "by", ["node"], arg=1.0)
    got2 = {s.mn.get_tag_value("node"): s.values for s in out2}
    assert got2[b"n1"][0] == 3.0


@pytest.mark.gpu
def test_aggregate_share_keeps_names():
    from victoriametrics_amd import aggregate as agg
    out = agg.aggregate("share", _series_set(), "by", ["node"])
    pods = sorted(s.mn.get_tag_value("pod") for s in out)
    assert pods == [b"a", b"b", b"c"]
    got = {s.mn.get_tag_value("pod"): list(s.values) for s in out}
    assert got[b"a"] == [0.25, 1 / 3]


@pytest.mark.gpu
def test_aggregate_outliers():
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    series = [Series(MetricName("m", [("i", str(i))]),
                     np.full(4, float(i))) for i in range(9)]
    series.append(Series(MetricName("m", [("i", "out")]),
                         np.full(4, 1000.0)))
    out = agg.aggregate("outliers_iqr", series)
    assert [s.mn.get_tag_value("i") for s in out] == [b"out"]


def test_aggregate_any_limitk():
    from victoriametrics_amd import aggregate as agg
    out = agg.aggregate("any", _series_set(), "by", ["node"])
    assert len(out) == 2
    # keepOriginal=true (aggr.go:166): the winner keeps its FULL name
    assert all(s.mn.get_tag_value("pod") is not None for s in out)
    out2 = agg.aggregate("limitk", _series_set(), "by", ["node"], arg=1)
    assert len(out2) == 2
    assert all(s.mn.get_tag_value("pod") is not None for s in out2)


def test_count_values():
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    series = [
        Series(MetricName("m", [("pod", "a")]), np.asarray([1.0, 2.0, NAN])),
        Series(MetricName("m", [("pod", "b")]), np.asarray([1.0, 1.0, 2.0])),
    ]
    out = agg.count_values("v", series)
    got = {s.mn.get_tag_value("v"): list(s.values) for s in out}
    assert got[b"1"][0] == 2.0 and got[b"1"][1] == 1.0
    assert math.isnan(got[b"1"][2])
    assert math.isnan(got[b"2"][0]) and got[b"2"][1] == 1.0
    assert got[b"2"][2] == 1.0
    # Go shortest-format labels
    out2 = agg.count_values("v", [Series(MetricName("m"),
                                         np.asarray([0.5, 0.25]))])
    labels = sorted(s.mn.get_tag_value("v") for s in out2)
    assert labels == [b"0.25", b"0.5"]


def test_histogram_aggregate():
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    series = [Series(MetricName("m", [("pod", p)]),
                     np.asarray([v], np.float64))
              for p, v in [("a", 1.0), ("b", 1.0), ("c", 5.0),
                           ("d", -3.0), ("e", NAN)]]
    out = agg.histogram_aggregate(series)
    # cumulative le buckets; last bucket (le=+Inf) counts all 3 valid values
    les = [(float(s.mn.get_tag_value("le")), s.values[0]) for s in out]
    les.sort()
    assert les[-1][0] == math.inf and les[-1][1] == 3.0
    # exact 10^0 value lands in the bucket ENDING at 1.000e+00
    one_buckets = [le for le, _ in les if abs(le - 1.0) < 1e-9]
    assert one_buckets, les
    # monotone cumulative counts
    counts = [c for _, c in les]
    assert counts == sorted(counts)


@pytest.mark.gpu
def test_quantiles_plural():
    from victoriametrics_amd import aggregate as agg
    out = agg.quantiles("phi", [0.25, 0.75], _series_set(), "by", ["node"])
    assert len(out) == 4
    labels = sorted(set(s.mn.get_tag_value("phi") for s in out))
    assert labels == [b"0.25", b"0.75"]


def test_format_go_float_pins():
    # count_values label formatting: strconv.FormatFloat(v, 'f', -1, 64)
    # shortest round-trip positional form (aggr.go count_values)
    from victoriametrics_amd.aggregate import format_go_float as f
    assert f(1.0) == "1"
    assert f(0.1) == "0.1"
    assert f(-2.5) == "-2.5"
    assert f(1e-9) == "0.000000001"
    assert f(123456789.0) == "123456789"
    assert f(1e21) == "1000000000000000000000"
    assert f(0.30000000000000004) == "0.30000000000000004"
    assert f(float("inf")) == "+Inf"
    assert f(float("-inf")) == "-Inf"
    assert f(float("nan")) == "NaN"
    assert f(0.0) == "0"
    # round-trip property on random values
    import numpy as np
    rng = np.random.default_rng(2)
    for v in rng.standard_normal(200) * 10.0 ** rng.integers(-15, 15, 200):
        assert float(f(float(v))) == float(v), v


# ---------------------------------------------------------------------------
# count_values exec pins (exec_test.go:9526-9737): expected MetricNames and
# per-point count arrays verbatim.  These pin that the produced series carry
# the GROUP metric name (aggrPrepareSeries rewrites members in place before
# the afe runs), not the first member's original labels.
# ---------------------------------------------------------------------------

def _cv_series(name, tags, values):
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    v = np.asarray(values, np.float64)
    if v.size == 1:
        v = np.full(6, float(v))
    return Series(MetricName(name, tags), v.copy())


def _cv_check(out, want):
    """want: list of (sorted-tags-tuple, values)."""
    got = {}
    for s in out:
        assert s.mn.metric_group == b""
        got[tuple(sorted(s.mn.tags))] = s.values
    assert set(got) == {k for k, _ in want}, (sorted(got), want)
    for k, w in want:
        g, w = got[k], np.asarray(w, np.float64)
        assert (np.isnan(g) == np.isnan(w)).all(), (k, g, w)
        np.testing.assert_array_equal(g[~np.isnan(w)], w[~np.isnan(w)])


def test_count_values_exec():
    # :9557 count_values("xxx", label_set(10,foo=bar) or
    #                           label_set(time()/100, foo=bar, baz=xx))
    from victoriametrics_amd import aggregate as agg
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    series = [
        _cv_series("", [("foo", "bar")], 10.0),
        _cv_series("", [("foo", "bar"), ("baz", "xx")], TIME / 100),
    ]
    out = agg.count_values("xxx", series)
    _cv_check(out, [
        (((b"xxx", b"10"),), [2, 1, 1, 1, 1, 1]),
        (((b"xxx", b"12"),), [NAN, 1, NAN, NAN, NAN, NAN]),
        (((b"xxx", b"14"),), [NAN, NAN, 1, NAN, NAN, NAN]),
        (((b"xxx", b"16"),), [NAN, NAN, NAN, 1, NAN, NAN]),
        (((b"xxx", b"18"),), [NAN, NAN, NAN, NAN, 1, NAN]),
        (((b"xxx", b"20"),), [NAN, NAN, NAN, NAN, NAN, 1]),
    ])


def test_count_values_by_xxx_exec():
    # :9629 — `by (xxx)` with dst label xxx removed from the grouping, so
    # everything lands in ONE empty-name group; the input xxx=aaa tag must
    # NOT leak into the outputs
    from victoriametrics_amd import aggregate as agg
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    series = [
        _cv_series("", [("foo", "bar"), ("xxx", "aaa")], 10.0),
        _cv_series("", [("foo", "bar"), ("baz", "xx")],
                   np.floor(TIME / 600)),
    ]
    out = agg.count_values("xxx", series, "by", ["xxx"])
    _cv_check(out, [
        (((b"xxx", b"1"),), [1, NAN, NAN, NAN, NAN, NAN]),
        (((b"xxx", b"2"),), [NAN, 1, 1, 1, NAN, NAN]),
        (((b"xxx", b"3"),), [NAN, NAN, NAN, NAN, 1, 1]),
        (((b"xxx", b"10"),), [1, 1, 1, 1, 1, 1]),
    ])


def test_count_values_without_baz_exec():
    # :9680 — `without (baz)` keeps foo=bar on the outputs
    from victoriametrics_amd import aggregate as agg
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    series = [_cv_series("m", [("foo", "bar")], np.floor(TIME / 600))]
    out = agg.count_values("xxx", series, "without", ["baz"])
    _cv_check(out, [
        (((b"foo", b"bar"), (b"xxx", b"1")), [1, NAN, NAN, NAN, NAN, NAN]),
        (((b"foo", b"bar"), (b"xxx", b"2")), [NAN, 1, 1, 1, NAN, NAN]),
        (((b"foo", b"bar"), (b"xxx", b"3")), [NAN, NAN, NAN, NAN, 1, 1]),
    ])


def test_count_values_big_numbers_exec():
    # :9526 — FormatFloat(v, 'f', -1, 64): big integers stay positional
    from victoriametrics_amd import aggregate as agg
    series = [
        _cv_series("first", [], 772424014.0),
        _cv_series("second", [], 772424230.0),
    ]
    out = agg.count_values("xxx", series)
    _cv_check(out, [
        (((b"xxx", b"772424014"),), [1] * 6),
        (((b"xxx", b"772424230"),), [1] * 6),
    ])


def test_outliers_exec_pins():
    # exec_test.go outliers_iqr() / outliers_mad(1) / outliers_mad(5):
    # expected survivor sets verified through the oracle bounds+filter
    # kernels (the host selection wiring is covered in the GPU suite)
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    v = np.stack([TIME, TIME * 1.5, TIME * 10, TIME * 1.2, TIME * 0.1])
    gof = np.zeros(5, np.int32)
    go = np.asarray([0, 5], np.uint64)
    gr = np.arange(5, dtype=np.uint32)
    lo, hi = oracle.colagg("iqr_bounds", v, gr, go)
    flags = oracle.colagg_filter("iqr", v, gof, lo, hi)
    # m5 (0.1x) and m3 (10x) are the outliers, in sort() order
    assert list(flags) == [0, 0, 1, 0, 1]
    # outliers_mad(1, (t, 1.5t, 0.9t)): only the 1.5x series deviates
    # beyond 1 MAD; at k=5 nothing does
    v = np.stack([TIME, TIME * 1.5, TIME * 0.9])
    gof3 = np.zeros(3, np.int32)
    go3 = np.asarray([0, 3], np.uint64)
    gr3 = np.arange(3, dtype=np.uint32)
    med = oracle.colagg("median", v, gr3, go3)
    mad = oracle.colagg("mad", v, gr3, go3)
    flags = oracle.colagg_filter("mad", v, gof3, med, mad * 1.0)
    assert list(flags) == [0, 1, 0]
    flags = oracle.colagg_filter("mad", v, gof3, med, mad * 5.0)
    assert list(flags) == [0, 0, 0]


def test_histogram_aggregate_exec_pins():
    # exec_test.go `histogram(scalar)` / `histogram(vector)` — the
    # VictoriaMetrics-histogram vmrange boundaries and the cumulative le
    # conversion, before the exec queries' `+` merge (pinned separately
    # in the binop suite)
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName

    def const(v, name="", tags=()):
        return Series(MetricName(name, list(tags)), np.full(6, float(v)))

    out = agg.histogram_aggregate([const(123.0)])
    got = {s.mn.get_tag_value("le").decode(): s.values[0] for s in out}
    assert got == {"1.136e+02": 0.0, "1.292e+02": 1.0, "+Inf": 1.0}
    out = agg.histogram_aggregate([
        const(1.0, tags=[("foo", "bar")]),
        const(1.1, tags=[("xx", "yy")]),
        const(1.15, "foobar"),
    ])
    got = {s.mn.get_tag_value("le").decode(): s.values[0] for s in out}
    assert got == {"8.799e-01": 0.0, "1.000e+00": 1.0, "1.136e+00": 2.0,
                   "1.292e+00": 3.0, "+Inf": 3.0}


def test_quantile_out_of_range_and_mad_exec_pins():
    # exec_test.go quantile(-2) -> -Inf, quantile(3) -> +Inf,
    # quantile(NaN) -> NaN; mad(3 series) -> [100..200]
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    v = np.stack([np.full(6, 10.0), TIME / 150])
    gr = np.arange(2, dtype=np.uint32)
    go = np.asarray([0, 2], np.uint64)
    out = oracle.colagg("quantile", v, gr, go, phi=-2.0)
    assert np.isneginf(out[0]).all()
    out = oracle.colagg("quantile", v, gr, go, phi=3.0)
    assert np.isposinf(out[0]).all()
    out = oracle.colagg("quantile", v, gr, go, phi=NAN)
    assert np.isnan(out[0]).all()
    v = np.stack([TIME, TIME * 1.5, TIME * 0.9])
    gr = np.arange(3, dtype=np.uint32)
    go = np.asarray([0, 3], np.uint64)
    out = oracle.colagg("mad", v, gr, go)
    assert list(out[0]) == [100, 120, 140, 160, 180, 200]


def test_distinct_exec_pin():
    # `distinct(union(1+time() > 1100, (time() > 1700 tagged)))` ->
    # [nan, 1, 1, 1, 2, 2]
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    # `1+time() > 1100` filters on the SUM: 1001 <= 1100 -> NaN at idx 0
    a = np.where(1 + TIME > 1100, 1 + TIME, NAN)
    b = np.where(TIME > 1700, TIME, NAN)
    v = np.stack([a, b])
    gr = np.arange(2, dtype=np.uint32)
    go = np.asarray([0, 2], np.uint64)
    out = oracle.colagg("distinct", v, gr, go)
    g = out[0]
    assert np.isnan(g[0])
    assert list(g[1:]) == [1, 1, 1, 2, 2]


def test_sum_by_name_exec_pin():
    # `sum(...) by (__name__)` exec_test.go:10428 — grouping on __name__
    # keeps the metric group and drops all tags
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    series = [
        Series(MetricName("bar", [("baz", "sss"), ("x", "y")]),
               np.full(6, 10.0)),
        Series(MetricName("aaa", [("baz", "sss")]), TIME / 100),
    ]
    groups = agg.prepare_series(series, "by", ["__name__"])
    assert len(groups) == 2
    got = {}
    for gmn, members in groups:
        assert gmn.tags == []
        v = np.stack([s.values for s in members])
        gr = np.arange(len(members), dtype=np.uint32)
        go = np.asarray([0, len(members)], np.uint64)
        got[gmn.metric_group] = oracle.colagg("sum", v, gr, go)[0]
    np.testing.assert_array_equal(got[b"bar"], [10.0] * 6)
    np.testing.assert_array_equal(got[b"aaa"], [10, 12, 14, 16, 18, 20])


def test_sum_by_known_tag_limit_exec_pin():
    # `sum(...) by (foo) limit 1` exec_test.go:6374 — the limit caps the
    # number of GROUPS, keeping the first-seen one
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    series = [
        Series(MetricName("", [("foo", "bar")]), np.full(6, 10.0)),
        Series(MetricName("", [("baz", "sss")]), TIME / 100),
    ]
    groups = agg.prepare_series(series, "by", ["foo"], max_series=1)
    assert len(groups) == 1
    gmn, members = groups[0]
    assert sorted(gmn.tags) == [(b"foo", b"bar")]
    np.testing.assert_array_equal(members[0].values, [10.0] * 6)
    # without the limit both groups exist (:6354)
    groups = agg.prepare_series(
        [s.copy_shallow() for s in series], "by", ["foo"])
    assert len(groups) == 2


def test_sum_label_graphite_group_exec_pin():
    # `sort(sum by (__name__) (label_graphite_group((...), 1)))`
    # exec_test.go:2597 — graphite component 1 becomes the metric group,
    # then by (__name__) folds the two "bar" series
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd import transform as tfm
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    series = [
        Series(MetricName("foo.bar.baz"), np.full(6, 1.0)),
        Series(MetricName("x.y.z"), np.full(6, 2.0)),
        Series(MetricName("qe.bar.qqq"), np.full(6, 3.0)),
    ]
    out = tfm.label_graphite_group(series, [1])
    assert sorted(s.mn.metric_group for s in out) == [b"bar", b"bar", b"y"]
    groups = agg.prepare_series(out, "by", ["__name__"])
    got = {}
    for gmn, members in groups:
        v = np.stack([s.values for s in members])
        gr = np.arange(len(members), dtype=np.uint32)
        go = np.asarray([0, len(members)], np.uint64)
        got[gmn.metric_group] = oracle.colagg("sum", v, gr, go)[0]
    np.testing.assert_array_equal(got[b"y"], [2.0] * 6)
    np.testing.assert_array_equal(got[b"bar"], [4.0] * 6)


def test_single_and_multi_vector_aggregate_exec_pins():
    # exec_test.go:6210-6409 — sum/geomean/sum2 over time()/100 (single
    # member: identity); geomean over two members (rounded 0.1); by with
    # a DUPLICATE tag in the list; avg without () over a scalar
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    from victoriametrics_amd.decimal import go_round
    TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])
    t100 = TIME / 100

    def one(op, vals):
        v = np.stack([np.asarray(r, np.float64) for r in vals])
        gr = np.arange(v.shape[0], dtype=np.uint32)
        go = np.asarray([0, v.shape[0]], np.uint64)
        return oracle.colagg(op, v, gr, go)[0]

    np.testing.assert_array_equal(one("sum", [t100]), t100)
    np.testing.assert_array_equal(one("geomean", [t100]), t100)
    np.testing.assert_array_equal(one("sum2", [t100]), t100 * t100)
    got = one("geomean", [np.full(6, 10.0), t100])
    np.testing.assert_array_equal(go_round(got * 10.0) / 10.0,
                                  [10, 11, 11.8, 12.6, 13.4, 14.1])
    # `sum(...) by (foo, baz, foo)` — duplicate by-tag is harmless; both
    # members share foo=bar/baz=sss and fold into one group
    series = [
        Series(MetricName("", [("foo", "bar"), ("baz", "sss"),
                               ("x", "y")]), np.full(6, 10.0)),
        Series(MetricName("", [("baz", "sss"), ("foo", "bar")]), t100),
    ]
    groups = agg.prepare_series(series, "by", ["foo", "baz", "foo"])
    assert len(groups) == 1
    gmn, members = groups[0]
    assert sorted(gmn.tags) == [(b"baz", b"sss"), (b"foo", b"bar")]
    got = one("sum", [s.values for s in members])
    np.testing.assert_array_equal(got, [20, 22, 24, 26, 28, 30])
    # `avg without (xx, yy) (123)` — scalar input, all tags removed
    groups = agg.prepare_series(
        [Series(MetricName(""), np.full(6, 123.0))], "without",
        ["xx", "yy"])
    assert len(groups) == 1 and groups[0][0].tags == []
    got = one("avg", [groups[0][1][0].values])
    np.testing.assert_array_equal(got, [123.0] * 6)
