"""End-to-end pins transcribed from the reference's own TestExecSuccess
(app/vmselect/promql/exec_test.go, fixed grid start=1000e3 end=2000e3
step=200e3): each case hand-composes the same expression from this
engine's pieces (eval-context series + transform/binop kernels) and
compares against exec_test.go's expected float64 arrays — the reference's
outputs, not the oracle's.  Exact for arithmetic-only expressions; 1-ulp
tolerance for the transcendental ones (the expected arrays encode Go's
math library, the engine computes with device libm)."""
import math

import numpy as np
import pytest

from victoriametrics_amd import transform as tfm
from victoriametrics_amd import binary_op as bop
from victoriametrics_amd.binary_op import BinOpSpec, Series
from victoriametrics_amd.metric_name import MetricName

pytestmark = pytest.mark.gpu

GRID_MS = np.arange(1000_000, 2000_001, 200_000, dtype=np.int64)
TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def tmat():
    return TIME.reshape(1, -1).copy()


def _exact(got, expected):
    np.testing.assert_array_equal(
        np.asarray(got, np.float64).ravel().view(np.int64),
        np.asarray(expected, np.float64).view(np.int64))


def _ulp(got, expected):
    np.testing.assert_allclose(np.asarray(got).ravel(), expected,
                               rtol=1e-14, atol=0)


def test_abs_time():
    # abs(1500-time()) -> 500, 300, 100, 100, 300, 500
    v = (1500.0 - tmat())
    _exact(tfm.transform("abs", v), [500, 300, 100, 100, 300, 500])


def test_ceil_floor():
    _exact(tfm.transform("ceil", tmat() / 500), [2, 3, 3, 4, 4, 4])
    _exact(tfm.transform("floor", tmat() / 500), [2, 2, 2, 3, 3, 4])


def test_clamp_family():
    lo = np.full(6, 1400.0)
    hi = np.full(6, 1800.0)
    _exact(tfm.transform("clamp", tmat(), args=[lo, hi]),
           [1400, 1400, 1400, 1600, 1800, 1800])
    _exact(tfm.transform("clamp_max", tmat(), args=[np.full(6, 1400.0)]),
           [1000, 1200, 1400, 1400, 1400, 1400])
    # clamp_min(1500, time()): scalar arg clamped from below by time()
    _exact(tfm.transform("clamp_min", np.full((1, 6), 1500.0),
                         args=[TIME]),
           [1500, 1500, 1500, 1600, 1800, 2000])


def test_bitmap_time():
    mask = np.full(6, 17.0)  # 0x11
    _exact(tfm.transform("bitmap_and", tmat(), args=[mask]),
           [0, 16, 16, 0, 0, 16])
    _exact(tfm.transform("bitmap_or", tmat(), args=[mask]),
           [1017, 1201, 1401, 1617, 1817, 2001])
    _exact(tfm.transform("bitmap_xor", tmat(), args=[mask]),
           [1017, 1185, 1385, 1617, 1817, 1985])


def test_round_nearest():
    # round(time()/1e3, 0.5) -> 1, 1, 1.5, 1.5, 2, 2
    _exact(tfm.transform("round", tmat() / 1e3, args=[np.full(6, 0.5)]),
           [1, 1, 1.5, 1.5, 2, 2])


def test_sgn():
    _exact(tfm.transform("sgn", tmat() - 1400), [-1, -1, 0, 1, 1, 1])


def test_transcendental():
    _ulp(tfm.transform("exp", tmat() / 1e3),
         [2.718281828459045, 3.3201169227365472, 4.0551999668446745,
          4.953032424395115, 6.0496474644129465, 7.38905609893065])
    _ulp(tfm.transform("sqrt", tmat()),
         [31.622776601683793, 34.64101615137755, 37.416573867739416, 40,
          42.42640687119285, 44.721359549995796])
    _ulp(tfm.transform("ln", tmat()),
         [6.907755278982137, 7.090076835776092, 7.24422751560335,
          7.3777589082278725, 7.495541943884256, 7.600902459542082])
    _ulp(tfm.transform("log2", tmat()),
         [9.965784284662087, 10.228818690495881, 10.451211111832329,
          10.643856189774725, 10.813781191217037, 10.965784284662087])
    _ulp(tfm.transform("log10", tmat()),
         [3, 3.0791812460476247, 3.1461280356782377, 3.2041199826559246,
          3.255272505103306, 3.3010299956639813])


def test_rad_deg_roundtrip():
    v = tfm.transform("deg", tmat() / 500)
    _ulp(tfm.transform("rad", v),
         [2, 2.3999999999999995, 2.8, 3.2, 3.6, 4])


def test_running_range_funcs():
    _exact(tfm.transform("running_sum", np.ones((1, 6))),
           [1, 2, 3, 4, 5, 6])
    _exact(tfm.transform("range_sum", tmat()), [9000.0] * 6)
    _exact(tfm.transform("range_max", tmat()), [2000.0] * 6)
    _exact(tfm.transform("range_first", tmat()), [1000.0] * 6)
    _exact(tfm.transform("range_quantile", tmat(), scalar=0.5),
           [1500.0] * 6)
    v = tfm.transform("abs", 1300.0 - tmat())
    _exact(tfm.transform("running_max", v), [300, 300, 300, 300, 500, 700])


def _masked_default_chain():
    """time() < 1300 default time() > 1700 (the exec_test gap fixture)"""
    t = Series(MetricName(), TIME.copy())
    lt = bop.binary_op_eval(BinOpSpec("<"),
                            [Series(MetricName(), TIME.copy())],
                            [Series(MetricName(), np.full(6, 1300.0))])
    gt = bop.binary_op_eval(BinOpSpec(">"),
                            [Series(MetricName(), TIME.copy())],
                            [Series(MetricName(), np.full(6, 1700.0))])
    out = bop.binary_op_eval(BinOpSpec("default"), lt, gt)
    assert len(out) == 1
    return out[0].values.reshape(1, -1)


def test_cmp_default_interpolate_keep():
    chain = _masked_default_chain()
    # fixture: 1000, 1200, NaN, NaN, 1800, 2000
    assert not np.isnan(chain[0, 0]) and np.isnan(chain[0, 2])
    _exact(tfm.transform("interpolate", chain.copy()),
           [1000, 1200, 1400, 1600, 1800, 2000])
    _exact(tfm.transform("keep_last_value", chain.copy()),
           [1000, 1200, 1200, 1200, 1800, 2000])
    _exact(tfm.transform("keep_next_value", chain.copy()),
           [1000, 1200, 1800, 1800, 1800, 2000])


def test_time_plus_time():
    out = bop.binary_op_eval(BinOpSpec("+"),
                             [Series(MetricName(), TIME.copy())],
                             [Series(MetricName(), TIME.copy())])
    _exact(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])


def test_timezone_offset_newyork():
    # exec_test computes the offset dynamically; the fixed grid (1970-01-01)
    # falls in EST: -5h
    out = tfm.timezone_offset("America/New_York", GRID_MS)
    _exact(np.asarray(out), [-18000.0] * 6)


# ---------------------------------------------------------------------------
# vector-matching pins (exec_test.go:3561-3620): on/ignoring matching +
# keep_metric_names through the real binop dispatch
# ---------------------------------------------------------------------------

def _ls(value, *tags, name=""):
    """label_set(value, k, v, ...) fixture"""
    mn = MetricName(name, list(zip(tags[::2], tags[1::2])))
    v = TIME.copy() if value == "time" else np.full(6, float(value))
    return Series(mn, v)


def test_scalarish_mul_ignoring():
    # label_set(2,"foo","bar") * ignoring(a) (label_set(time(),foo=bar) or
    # label_set(10,foo=qwert)) -> {foo=bar} 2000..4000
    left = [_ls(2, "foo", "bar")]
    right = bop.binary_op_eval(BinOpSpec("or"),
                               [_ls("time", "foo", "bar")],
                               [_ls(10, "foo", "qwert")])
    out = bop.binary_op_eval(BinOpSpec("*", group_op="ignoring",
                                       group_tags=["a"]), left, right)
    assert len(out) == 1
    assert out[0].mn.tags == [(b"foo", b"bar")]
    assert out[0].mn.metric_group == b""
    _exact(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])


def test_scalarish_mul_on_foo():
    left = [_ls(2, "foo", "bar", "aa", "bb")]
    right = bop.binary_op_eval(BinOpSpec("or"),
                               [_ls("time", "foo", "bar", "xx", "yy")],
                               [_ls(10, "foo", "qwert")])
    out = bop.binary_op_eval(BinOpSpec("*", group_op="on",
                                       group_tags=["foo"]), left, right)
    assert len(out) == 1
    assert out[0].mn.tags == [(b"foo", b"bar")]
    _exact(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])


def test_vector_mul_on_scalarish():
    left = [_ls("time", "foo", "bar", "xx", "yy"), _ls(10, "foo", "qwert")]
    right = [_ls(2, "foo", "bar", "aa", "bb")]
    out = bop.binary_op_eval(BinOpSpec("*", group_op="on",
                                       group_tags=["foo"]), left, right)
    assert len(out) == 1
    assert out[0].mn.tags == [(b"foo", b"bar")]
    _exact(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])


def test_vector_mul_on_keep_metric_names():
    left = [_ls("time", "foo", "bar", "xx", "yy", name="q1"),
            _ls(10, "foo", "qwert", name="q2")]
    right = [_ls(2, "foo", "bar", "aa", "bb", name="q2")]
    out = bop.binary_op_eval(BinOpSpec("*", group_op="on",
                                       group_tags=["foo"],
                                       keep_metric_names=True), left, right)
    assert len(out) == 1
    assert out[0].mn.metric_group == b"q1"
    assert out[0].mn.tags == [(b"foo", b"bar")]
    _exact(out[0].values, [2000, 2400, 2800, 3200, 3600, 4000])


# ---------------------------------------------------------------------------
# aggregate pins (exec_test.go): the aggregate dispatch against the
# reference's expected arrays
# ---------------------------------------------------------------------------

def _agg(name, series, **kw):
    from victoriametrics_amd import aggregate as agg
    return agg.aggregate(name, series, **kw)


def test_median_union():
    # median(union(label_set(10,foo=bar), label_set(time()/150,baz=sss),
    #              time()/200))
    series = [_ls(10, "foo", "bar"),
              Series(MetricName("", [("baz", "sss")]), TIME / 150),
              Series(MetricName(), TIME / 200)]
    out = _agg("median", series)
    assert len(out) == 1
    _exact(out[0].values,
           [6.666666666666667, 8, 9.333333333333334, 10, 10, 10])


def test_stddev_avg_or():
    left = [_ls(10, "foo", "bar")]
    right = [Series(MetricName("", [("baz", "sss")]), TIME / 100)]
    both = bop.binary_op_eval(BinOpSpec("or"),
                              [s.copy_shallow() for s in left],
                              [s.copy_shallow() for s in right])
    out = _agg("stddev", [s.copy_shallow() for s in both])
    _exact(out[0].values, [0, 1, 2, 3, 4, 5])
    out2 = _agg("avg", [s.copy_shallow() for s in both])
    _exact(out2[0].values, [10, 11, 12, 13, 14, 15])


def test_count_with_nan_tails():
    # count(label_set(time()<1500,..) or label_set(time()<1800,..))
    a = bop.binary_op_eval(BinOpSpec("<"), [_ls("time", "foo", "bar")],
                           [_ls(1500)])
    b = bop.binary_op_eval(BinOpSpec("<"), [_ls("time", "baz", "sss")],
                           [_ls(1800)])
    both = a + b
    out = _agg("count", both)
    v = out[0].values
    assert list(v[:4]) == [2, 2, 2, 1]
    assert math.isnan(v[4]) and math.isnan(v[5])


def test_geomean_sum2():
    out = _agg("geomean", [Series(MetricName(), TIME / 100)])
    np.testing.assert_allclose(out[0].values, [10, 12, 14, 16, 18, 20],
                               rtol=1e-14)
    out2 = _agg("sum2", [Series(MetricName(), TIME / 100)])
    _exact(out2[0].values, [100, 144, 196, 256, 324, 400])


def test_mode_aliases():
    series = [Series(MetricName(f"m{i}"), np.full(6, float(v)))
              for i, v in enumerate([3, 2, 3, 4, 3, 2])]
    out = _agg("mode", series)
    _exact(out[0].values, [3, 3, 3, 3, 3, 3])


def test_min_by_unknown_tag():
    left = [_ls(10, "foo", "bar")]
    right = [Series(MetricName("", [("baz", "sss")]), TIME / 100 / 1.5)]
    both = bop.binary_op_eval(BinOpSpec("or"), left, right)
    out = _agg("min", both, modifier_op="by", modifier_args=["unknowntag"])
    assert len(out) == 1
    _exact(out[0].values,
           [6.666666666666667, 8, 9.333333333333334, 10, 10, 10])


def test_distinct_union():
    # distinct(union(1+time() > 1100, label_set(time() > 1700, foo=bar)))
    a = bop.binary_op_eval(BinOpSpec(">"),
                           [Series(MetricName(), 1 + TIME)], [_ls(1100)])
    b = bop.binary_op_eval(BinOpSpec(">"), [_ls("time", "foo", "bar")],
                           [_ls(1700)])
    out = _agg("distinct", a + b)
    v = out[0].values
    assert math.isnan(v[0])
    assert list(v[1:]) == [1, 1, 1, 2, 2]


# ---------------------------------------------------------------------------
# histogram family (exec_test.go histogram_quantile / histogram_share /
# histogram_fraction cases) through the device hq/hshare kernels
# ---------------------------------------------------------------------------

def _buckets(*pairs):
    """pairs: (le_string, constant_value) -> bucket Series on the 6-pt grid"""
    return [Series(MetricName(b"", [(b"le", le.encode())]),
                   np.full(6, float(v)))
            for le, v in pairs]


def _hist(name, series, arg, bounds_label=None):
    return tfm.histogram_transform(name, series, arg=arg,
                                   bounds_label=bounds_label)


class TestHistogramExecPins:
    def test_quantile_no_le(self):
        # histogram_quantile(single-value-no-le / -invalid-le) -> empty
        s = [Series(MetricName(b"", [(b"foo", b"bar")]), np.full(6, 100.0))]
        assert _hist("histogram_quantile", s, 0.6) == []
        assert _hist("histogram_quantile",
                     _buckets(("foobar", 100)), 0.6) == []

    def test_quantile_inf_le_only(self):
        # the walk breaks on +Inf and lastNonInf finds nothing -> all-NaN
        # series; the exec layer's removeEmptySeries yields the reference's
        # empty result (transform.go:1062-1072)
        from victoriametrics_amd.binary_op import remove_empty_series
        got = _hist("histogram_quantile", _buckets(("+Inf", 100)), 0.6)
        assert len(got) == 1 and np.isnan(got[0].values).all()
        assert remove_empty_series(got) == []

    def test_quantile_zero_value_inf_le(self):
        got = _hist("histogram_quantile",
                    _buckets(("+Inf", 100), ("42", 0)), 0.6)
        _exact(got[0].values, [42.0] * 6)

    def test_quantile_single_valid_le(self):
        got = _hist("histogram_quantile", _buckets(("200", 100)), 0.6)
        _exact(got[0].values, [120.0] * 6)

    def test_quantile_max_min_phi(self):
        bl = lambda: _buckets(("200", 100), ("55", 0))
        _exact(_hist("histogram_quantile", bl(), 1.0)[0].values, [200.0] * 6)
        _exact(_hist("histogram_quantile", bl(), 0.0)[0].values, [55.0] * 6)

    def test_quantile_bounds_label(self):
        got = _hist("histogram_quantile", _buckets(("200", 100)), 0.6,
                    bounds_label="foobar")
        by = {s.mn.get_tag_value("foobar"): s.values for s in got}
        _exact(by[b"lower"], [0.0] * 6)
        _exact(by[b"upper"], [200.0] * 6)
        _exact(by[None], [120.0] * 6)

    def test_share_single_valid_le(self):
        for req, want in [(80, 0.4), (200, 1.0), (300, 1.0)]:
            got = _hist("histogram_share", _buckets(("200", 100)), req)
            _exact(got[0].values, [want] * 6)

    def test_share_mid_le(self):
        bl = lambda: _buckets(("200", 100), ("55", 0))
        _exact(_hist("histogram_share", bl(), 105)[0].values,
               [0.3448275862068966] * 6)
        _exact(_hist("histogram_share", bl(), 55)[0].values, [0.0] * 6)
        _exact(_hist("histogram_share", bl(), 0)[0].values, [0.0] * 6)

    def test_share_bounds_label(self):
        got = _hist("histogram_share", _buckets(("200", 100)), 120,
                    bounds_label="foobar")
        by = {s.mn.get_tag_value("foobar"): s.values for s in got}
        _exact(by[b"lower"], [0.0] * 6)
        _exact(by[b"upper"], [1.0] * 6)
        _exact(by[None], [0.6] * 6)

    def test_fraction_empty_and_invalid(self):
        s = [Series(MetricName(b"", [(b"foo", b"bar")]), np.full(6, 100.0))]
        assert _hist("histogram_fraction", s, (123, 456)) == []
        assert _hist("histogram_fraction",
                     _buckets(("foobar", 100)), (50, 60)) == []

    def test_fraction_valid_le(self):
        _exact(_hist("histogram_fraction", _buckets(("200", 100)),
                     (0, 100))[0].values, [0.5] * 6)
        _exact(_hist("histogram_fraction", _buckets(("200", 100)),
                     (200, 300))[0].values, [0.0] * 6)

    def test_fraction_three_buckets(self):
        bl = lambda: _buckets(("100", 100), ("50", 40), ("10", 0))
        _exact(_hist("histogram_fraction", bl(), (0, 100))[0].values,
               [1.0] * 6)
        _exact(_hist("histogram_fraction", bl(), (0, 10))[0].values,
               [0.0] * 6)

    def test_fraction_mid_le(self):
        got = _hist("histogram_fraction", _buckets(("200", 100), ("55", 0)),
                    (55, 105))
        _exact(got[0].values, [0.3448275862068966] * 6)


def test_running_range_funcs_with_gaps():
    # exec_test.go masked running/range cases: NaN gating inside the
    # running scans and range reductions
    nan = math.nan
    # running_min(abs(1500-time()) < 400 > 100) -> [nan,300,300,300,300,300]
    v = np.abs(1500.0 - tmat())
    v[(v >= 400) | (v <= 100)] = nan
    _exact(tfm.transform("running_min", v), [nan, 300, 300, 300, 300, 300])
    # running_max(abs(1300-time()) > 300 < 700) -> [nan]*4 + [500, 500]
    v = np.abs(1300.0 - tmat())
    v[(v <= 300) | (v >= 700)] = nan
    _exact(tfm.transform("running_max", v), [nan, nan, nan, nan, 500, 500])
    # running_sum(time()/1e3 > 1.2 < 1.8) -> [nan, nan, 1.4, 3, 3, 3]
    v = tmat() / 1e3
    v[(v <= 1.2) | (v >= 1.8)] = nan
    _exact(tfm.transform("running_sum", v), [nan, nan, 1.4, 3, 3, 3])
    # range_* over time() > 1200 < 1800 = [nan,nan,1400,1600,nan,nan]
    def gated():
        v = tmat()
        v[(v <= 1200) | (v >= 1800)] = nan
        return v
    _exact(tfm.transform("range_max", gated()), [1600.0] * 6)
    _exact(tfm.transform("range_sum", gated()), [3000.0] * 6)
    _exact(tfm.transform("range_last", gated()), [1600.0] * 6)
    _exact(tfm.transform("range_first", gated()), [1400.0] * 6)


def test_range_linear_regression_exec():
    ts = GRID_MS.copy()
    _exact(tfm.transform("range_linear_regression", tmat(), ts=ts),
           [1000.0, 1200, 1400, 1600, 1800, 2000])
    _exact(tfm.transform("range_linear_regression", -tmat(), ts=ts),
           [-1000.0, -1200, -1400, -1600, -1800, -2000])


def test_range_stddev_stdvar_exec():
    # round(range_stddev(time()), 0.01) etc.
    def rounded(name, v):
        out = tfm.transform(name, v)
        return tfm.transform("round", out, args=[np.full(6, 0.01)])
    _exact(rounded("range_stddev", tmat()), [341.57] * 6)
    _exact(rounded("range_stdvar", tmat()), [116666.67] * 6)
    v = tmat()
    v[(v <= 1200) | (v >= 1800)] = math.nan
    _exact(rounded("range_stddev", v), [100.0] * 6)
    v = tmat()
    v[(v <= 1200) | (v >= 1800)] = math.nan
    _exact(rounded("range_stdvar", v), [10000.0] * 6)


# ---------------------------------------------------------------------
# cross-series aggregate pins (exec_test.go:5941-6060): mode/share/zscore
# through the colagg device kernels + the aggregate.py host layer,
# rounded exactly as the reference query does
# ---------------------------------------------------------------------

def _kser(div, off):
    from victoriametrics_amd.binary_op import Series
    return Series(MetricName(b"", [(b"k", b"v")]), TIME / div + off)


def _share_inputs():
    from victoriametrics_amd.binary_op import Series
    return [
        Series(MetricName(b"", [(b"k", b"v1")]), TIME / 100 + 10),
        Series(MetricName(b"", [(b"k", b"v2")]), TIME / 200 + 5),
        Series(MetricName(b"", [(b"k", b"v3")]), TIME / 110 - 10),
        Series(MetricName(b"", [(b"k", b"v4")]), TIME / 90 - 5),
    ]


def _round_milli(rows):
    out = tfm.transform("round", np.vstack([s.values for s in rows]),
                        args=[np.full(6, 0.001)])
    return {s.mn.get_tag_value(b"k"): out[i] for i, s in enumerate(rows)}


def test_mode_aggregate():
    # `mode()` exec_test.go:5941 -> constant 3
    from victoriametrics_amd import aggregate as agg
    from victoriametrics_amd.binary_op import Series
    series = [Series(MetricName(name.encode(), []), np.full(6, float(v)))
              for name, v in [("m1", 3), ("m2", 2), ("m3", 3), ("m4", 4),
                              ("m5", 3), ("m6", 2)]]
    out = agg.aggregate("mode", series)
    assert len(out) == 1
    _exact(out[0].values, [3, 3, 3, 3, 3, 3])


def test_share_aggregate():
    # `share()` exec_test.go:5959 (round(share(...), 0.001) verbatim)
    from victoriametrics_amd import aggregate as agg
    out = agg.aggregate("share", _share_inputs())
    got = _round_milli(out)
    _exact(got[b"v1"], [0.554, 0.521, 0.487, 0.462, 0.442, 0.426])
    _exact(got[b"v2"], [0.277, 0.26, 0.243, 0.231, 0.221, 0.213])
    v3 = got[b"v3"]
    assert math.isnan(v3[0])
    _exact(v3[1:], [0.022, 0.055, 0.081, 0.1, 0.116])
    _exact(got[b"v4"], [0.169, 0.197, 0.214, 0.227, 0.237, 0.245])


def test_zscore_aggregate():
    # `zscore()` exec_test.go:6038 (first two rows of the pin)
    from victoriametrics_amd import aggregate as agg
    out = agg.aggregate("zscore", _share_inputs())
    got = _round_milli(out)
    _exact(got[b"v1"], [1.482, 1.511, 1.535, 1.552, 1.564, 1.57])
    _exact(got[b"v2"], [0.159, 0.058, -0.042, -0.141, -0.237, -0.329])


def test_range_trim_family_exec():
    # `range_trim_outliers(0.5, time())` -> [nan,nan,1400,1600,nan,nan]
    # `range_trim_spikes(0.2, time())`   -> [nan,1200,1400,1600,1800,nan]
    # `range_trim_zscore(0.9, time())`   -> [nan,1200,1400,1600,1800,nan]
    def nan_eq(got, want):
        g = np.asarray(got, np.float64).ravel()
        w = np.asarray(want, np.float64)
        assert (np.isnan(g) == np.isnan(w)).all(), (g, w)
        np.testing.assert_array_equal(g[~np.isnan(g)], w[~np.isnan(w)])
    nan_eq(tfm.transform("range_trim_outliers", tmat(), scalar=0.5),
           [math.nan, math.nan, 1400, 1600, math.nan, math.nan])
    nan_eq(tfm.transform("range_trim_spikes", tmat(), scalar=0.2),
           [math.nan, 1200, 1400, 1600, 1800, math.nan])
    nan_eq(tfm.transform("range_trim_zscore", tmat(), scalar=0.9),
           [math.nan, 1200, 1400, 1600, 1800, math.nan])


def _ser(name, tags, values):
    from victoriametrics_amd.binary_op import Series
    return Series(MetricName(name, [(k, v) for k, v in tags]),
                  np.asarray(values, np.float64))


def test_median_quantile_aggregates_exec():
    # `median()` :7442, `median(3-timeseries)` :7457, `quantile(3)` :7472
    # and `quantile(NaN)` :7486 — colagg device kernels + aggregate.py
    from victoriametrics_amd import aggregate as agg
    two = [_ser(b"", [(b"foo", b"bar")], np.full(6, 10.0)),
           _ser(b"", [(b"baz", b"sss")], TIME / 150)]
    out = agg.aggregate("median", [s.copy_shallow() for s in two])
    assert len(out) == 1
    _ulp(out[0].values,
         [8.333333333333334, 9, 9.666666666666668, 10.333333333333332,
          11, 11.666666666666668])
    three = two + [_ser(b"", [], TIME / 200)]
    out = agg.aggregate("median", [s.copy_shallow() for s in three])
    _ulp(out[0].values,
         [6.666666666666667, 8, 9.333333333333334, 10, 10, 10])
    out = agg.aggregate("quantile", [s.copy_shallow() for s in two], arg=3.0)
    assert np.isposinf(out[0].values).all()
    out = agg.aggregate("quantile", [s.copy_shallow() for s in two],
                        arg=math.nan)
    from victoriametrics_amd.binary_op import remove_empty_series
    assert remove_empty_series(out) == []


def test_quantiles_aggregate_exec():
    # `quantiles("phi", 0.2, 0.5, ...)` exec_test.go:7697
    from victoriametrics_amd import aggregate as agg
    two = [_ser(b"", [(b"foo", b"bar")], np.full(6, 10.0)),
           _ser(b"", [(b"baz", b"sss")], TIME / 150)]
    out = agg.quantiles("phi", [0.2, 0.5], two)
    got = {s.mn.get_tag_value(b"phi"): s.values for s in out}
    assert set(got) == {b"0.2", b"0.5"}
    _ulp(got[b"0.2"],
         [7.333333333333334, 8.4, 9.466666666666669, 10.133333333333333,
          10.4, 10.666666666666668])
    _ulp(got[b"0.5"],
         [8.333333333333334, 9, 9.666666666666668, 10.333333333333332,
          11, 11.666666666666668])


def test_histogram_share_and_fraction_valid_exec():
    # `histogram_share(25, ...)` :4920 and `histogram_fraction(0, 25, ...)`
    # :4950 over two le-groups -> 0.325 / 0.9166666666666666
    from victoriametrics_amd.transform import histogram_transform
    def buckets():
        return [
            _ser(b"", [(b"foo", b"bar"), (b"le", b"10")], np.full(6, 90.0)),
            _ser(b"", [(b"foo", b"bar"), (b"le", b"30")], np.full(6, 100.0)),
            _ser(b"", [(b"foo", b"bar"), (b"le", b"+Inf")], np.full(6, 300.0)),
            _ser(b"", [(b"tag", b"xx"), (b"le", b"10")], np.full(6, 200.0)),
            _ser(b"", [(b"tag", b"xx"), (b"le", b"30")], np.full(6, 300.0)),
        ]
    out = histogram_transform("histogram_share", buckets(), arg=25.0)
    got = {}
    for s in out:
        key = s.mn.get_tag_value(b"foo") or s.mn.get_tag_value(b"tag")
        got[key] = s.values
    _ulp(got[b"bar"], [0.325] * 6)
    _ulp(got[b"xx"], [0.9166666666666666] * 6)
    out = histogram_transform("histogram_fraction", buckets(),
                              arg=(0.0, 25.0))
    got = {}
    for s in out:
        key = s.mn.get_tag_value(b"foo") or s.mn.get_tag_value(b"tag")
        got[key] = s.values
    _ulp(got[b"bar"], [0.325] * 6)
    _ulp(got[b"xx"], [0.9166666666666666] * 6)


def test_histogram_quantile_negative_and_nan_bucket_exec():
    # `histogram_quantile(negative-bucket-count)` :4981 -> 30 (fixBroken)
    from victoriametrics_amd.transform import histogram_transform
    series = [
        _ser(b"", [(b"foo", b"bar"), (b"le", b"10")], np.full(6, 90.0)),
        _ser(b"", [(b"foo", b"bar"), (b"le", b"30")], np.full(6, -100.0)),
        _ser(b"", [(b"foo", b"bar"), (b"le", b"+Inf")], np.full(6, 300.0)),
    ]
    out = histogram_transform("histogram_quantile", series, arg=0.6)
    assert len(out) == 1
    _exact(out[0].values, [30.0] * 6)


def test_datetime_funcs_exec():
    # exec_test.go:836-913 — the date/time family over time()-derived
    # arguments (Unix seconds in UTC, Go time package semantics):
    # expected arrays verbatim.
    _exact(tfm.transform("minute", tmat()), [16, 20, 23, 26, 30, 33])
    _exact(tfm.transform("day_of_month", tmat() * 1e4),
           [26, 19, 12, 5, 28, 20])
    _exact(tfm.transform("day_of_week", tmat() * 1e4), [0, 2, 5, 0, 2, 4])
    _exact(tfm.transform("day_of_year", tmat() * 1e4),
           [116, 139, 163, 186, 209, 232])
    _exact(tfm.transform("days_in_month", tmat() * 2e4),
           [31, 31, 30, 31, 28, 30])
    _exact(tfm.transform("hour", tmat() * 1e4), [17, 21, 0, 4, 8, 11])
    _exact(tfm.transform("month", tmat() * 1e4), [4, 5, 6, 7, 7, 8])
    _exact(tfm.transform("year", tmat() * 1e5),
           [1973, 1973, 1974, 1975, 1975, 1976])


def test_minute_shifted_and_nan_exec():
    # `minute(30*60+time())` :915 -> [46, 50, 53, 56, 0, 3];
    # `minute(time() <= 1200 or time() > 1600)` :926 keeps NaN holes
    _exact(tfm.transform("minute", 30.0 * 60.0 + tmat()),
           [46, 50, 53, 56, 0, 3])
    v = np.where((TIME <= 1200) | (TIME > 1600), TIME,
                 np.nan).reshape(1, -1)
    got = np.asarray(tfm.transform("minute", v)).ravel()
    want = np.asarray([16, 20, np.nan, np.nan, 30, 33])
    assert (np.isnan(got) == np.isnan(want)).all()
    np.testing.assert_array_equal(got[~np.isnan(want)],
                                  want[~np.isnan(want)])
