"""GPU transform kernels vs the oracle over randomized matrices — bit-exact
for arithmetic-only funcs; transcendental one-arg funcs (device OCML vs
host glibc software implementations) compared at 1e-12 rtol, matching the
tolerance the reference's own tests use for these (exec_test.go)."""
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd import transform as tf

pytestmark = pytest.mark.gpu

NAN = math.nan

# device-libm funcs: ulp-level differences vs glibc are expected
ULP_FUNCS = {"exp", "ln", "log2", "log10", "sin", "cos", "tan", "asin",
             "acos", "atan", "sinh", "cosh", "tanh", "asinh", "acosh",
             "atanh"}

EXACT_ELEMENTWISE = ["abs", "ceil", "floor", "sqrt", "deg", "rad", "sgn"]
SERIES_FUNCS = ["keep_last_value", "keep_next_value", "interpolate",
                "running_sum", "running_min", "running_max", "running_avg",
                "range_sum", "range_min", "range_max", "range_avg",
                "range_first", "range_last", "range_zscore",
                "range_stddev", "range_stdvar", "range_mad",
                "range_linear_regression", "remove_resets"]


def _mat(rng, n_series=64, n_grid=240, nan_frac=0.25, scale=100.0):
    v = rng.standard_normal((n_series, n_grid)) * scale
    v[rng.random((n_series, n_grid)) < nan_frac] = NAN
    return v


def _ids(name):
    return {**tf._ELEMENTWISE, **tf._CLAMP, **tf._BITMAP, **tf._DATETIME,
            "round": tf._ROUND, **tf._SERIES}[name]


def _bitwise(got, exp, msg=""):
    np.testing.assert_array_equal(got.view(np.int64), exp.view(np.int64),
                                  err_msg=msg)


@pytest.mark.parametrize("name", EXACT_ELEMENTWISE)
def test_elementwise_exact(name):
    rng = np.random.default_rng(abs(hash(name)) % 2**31)
    v = _mat(rng)
    exp, _ = oracle.tf_apply(_ids(name), v.copy())
    got = tf.transform(name, v.copy())
    _bitwise(got, exp, name)


@pytest.mark.parametrize("name", sorted(ULP_FUNCS))
def test_elementwise_ulp(name):
    rng = np.random.default_rng(abs(hash(name)) % 2**31)
    # domain-restricted inputs so most values are finite
    v = rng.uniform(-0.99, 0.99, (32, 120))
    if name in ("acosh",):
        v = 1.0 + np.abs(v) * 3
    if name in ("exp", "sinh", "cosh", "tanh", "asinh", "sin", "cos", "tan",
                "atan"):
        v = v * 20
    if name in ("ln", "log2", "log10"):
        v = np.abs(v) * 1e3 + 1e-6
    exp, _ = oracle.tf_apply(_ids(name), v.copy())
    got = tf.transform(name, v.copy())
    fin = np.isfinite(exp)
    np.testing.assert_allclose(got[fin], exp[fin], rtol=1e-12, atol=1e-300,
                               err_msg=name)
    assert np.array_equal(np.isnan(got), np.isnan(exp)), name


def test_clamp_family():
    rng = np.random.default_rng(3)
    v = _mat(rng)
    lo = rng.standard_normal(240) * 10
    hi = lo + np.abs(rng.standard_normal(240)) * 50
    exp, _ = oracle.tf_apply(23, v.copy(), arg1=lo, arg2=hi)
    got = tf.transform("clamp", v.copy(), args=[lo, hi])
    _bitwise(got, exp)
    for name, fid in (("clamp_min", 24), ("clamp_max", 25)):
        exp, _ = oracle.tf_apply(fid, v.copy(), arg1=lo)
        got = tf.transform(name, v.copy(), args=[lo])
        _bitwise(got, exp, name)


def test_round():
    rng = np.random.default_rng(4)
    v = _mat(rng, scale=1000.0)
    for nearest in (1.0, 0.1, 0.5, 10.0, 0.25):
        nrow = np.full(240, nearest)
        p10 = np.full(240, math.pow(
            10.0, -tf.decimal_from_float_exponent(nearest)))
        exp, _ = oracle.tf_apply(26, v.copy(), arg1=nrow, arg2=p10)
        got = tf.transform("round", v.copy(), args=[nrow])
        _bitwise(got, exp, f"round nearest={nearest}")


def test_bitmap():
    rng = np.random.default_rng(5)
    v = np.floor(np.abs(_mat(rng, scale=1e6)))
    mask = np.floor(np.abs(rng.standard_normal(240)) * 255)
    for name, fid in (("bitmap_and", 27), ("bitmap_or", 28),
                      ("bitmap_xor", 29)):
        exp, _ = oracle.tf_apply(fid, v.copy(), arg1=mask)
        got = tf.transform(name, v.copy(), args=[mask])
        _bitwise(got, exp, name)


def test_datetime_funcs():
    rng = np.random.default_rng(6)
    # unix seconds across decades incl. pre-1970
    v = rng.uniform(-5e8, 2.2e9, (16, 200))
    v[rng.random((16, 200)) < 0.1] = NAN
    for name in tf._DATETIME:
        exp, _ = oracle.tf_apply(_ids(name), v.copy())
        got = tf.transform(name, v.copy())
        _bitwise(got, exp, name)


@pytest.mark.parametrize("name", SERIES_FUNCS)
def test_series_funcs(name):
    rng = np.random.default_rng(abs(hash(name)) % 2**31)
    v = _mat(rng, n_series=96)
    ts = (1_000_000_000_000 + np.arange(240) * 15_000).astype(np.int64)
    exp, _ = oracle.tf_apply(_ids(name), v.copy(), ts=ts)
    got = tf.transform(name, v.copy(), ts=ts)
    if name in ("range_zscore", "range_stddev", "range_stdvar",
                "range_linear_regression"):
        # sqrt/div chains: identical operations, but the oracle runs with
        # gcc -ffp-contract=off as does hipcc; should still be bit-equal
        _bitwise(got, exp, name)
    else:
        _bitwise(got, exp, name)


def test_range_normalize_keep():
    rng = np.random.default_rng(17)
    v = _mat(rng, n_series=32)
    v[5, :] = NAN
    exp, ekeep = oracle.tf_apply(113, v.copy())
    got, gkeep = tf.transform("range_normalize", v.copy())
    np.testing.assert_array_equal(gkeep, ekeep)
    keep = gkeep.astype(bool)
    _bitwise(got[keep], exp[keep])


@pytest.mark.parametrize("scalar,fid,name", [
    (0.5, 122, "range_quantile"), (0.9, 122, "range_quantile"),
    (0.2, 121, "range_trim_spikes"), (3.0, 120, "range_trim_outliers"),
    (2.0, 115, "range_trim_zscore"),
])
def test_scalar_arg_funcs(scalar, fid, name):
    rng = np.random.default_rng(fid * 7 + int(scalar * 10))
    v = _mat(rng, n_series=48)
    exp, _ = oracle.tf_apply(fid, v.copy(), scalar=scalar)
    got = tf.transform(name, v.copy(), scalar=scalar)
    _bitwise(got, exp, f"{name} {scalar}")


def test_smooth_exponential():
    rng = np.random.default_rng(23)
    v = _mat(rng, n_series=48)
    sfs = rng.uniform(-0.5, 1.5, 240)  # out-of-range sfs exercise clamping
    sfs[rng.random(240) < 0.1] = NAN
    exp, _ = oracle.tf_apply(123, v.copy(), arg1=sfs)
    got = tf.transform("smooth_exponential", v.copy(), args=[sfs])
    _bitwise(got, exp)


# ---------------------------------------------------------------------------
# histogram stat/share kernels + the full bucket pipeline
# ---------------------------------------------------------------------------

def _bucket_fixture(rng, n_groups=40, n_les=12, n_grid=96):
    les, goff = [], [0]
    rows = []
    for g in range(n_groups):
        ls = np.sort(rng.uniform(0.1, 100, n_les - 1))
        ls = np.concatenate([ls, [np.inf]])
        base = np.cumsum(np.abs(rng.standard_normal((n_les, n_grid))), axis=0)
        base[rng.random((n_les, n_grid)) < 0.05] = math.nan  # broken buckets
        rows.append(base)
        les.extend(ls)
        goff.append(goff[-1] + n_les)
    return (np.concatenate(rows, axis=0), np.asarray(les),
            np.asarray(goff, np.uint64))


@pytest.mark.parametrize("mode", ["avg", "stddev", "stdvar"])
def test_histogram_stat_kernel(mode):
    import oracle
    from victoriametrics_amd import engine
    rng = np.random.default_rng(61)
    bv, les, goff = _bucket_fixture(rng)
    got = engine.histogram_stat(mode, bv, les, goff)
    exp = oracle.histogram_stat({"avg": 0, "stddev": 1, "stdvar": 2}[mode],
                                bv, les, goff)
    np.testing.assert_array_equal(got.view(np.int64), exp.view(np.int64))


@pytest.mark.parametrize("le_req", [0.5, 20.0, 99.5, -1.0, math.inf,
                                    math.nan])
def test_histogram_share_kernel(le_req):
    import oracle
    from victoriametrics_amd import engine
    rng = np.random.default_rng(62)
    bv, les, goff = _bucket_fixture(rng)
    req = np.full(bv.shape[1], le_req)
    got, glo, ghi = engine.histogram_share(req, bv, les, goff, bounds=True)
    exp, elo, ehi = oracle.histogram_share(req, bv, les, goff)
    np.testing.assert_array_equal(got.view(np.int64), exp.view(np.int64))
    np.testing.assert_array_equal(glo.view(np.int64), elo.view(np.int64))
    np.testing.assert_array_equal(ghi.view(np.int64), ehi.view(np.int64))


def test_histogram_transform_pipeline():
    """vmrange buckets -> le conversion -> grouped stat on device, whole
    host pipeline."""
    import oracle
    from victoriametrics_amd import transform as tfm
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    rng = np.random.default_rng(63)
    series = []
    edges = [0.5, 1.0, 2.0, 4.0, 8.0]
    for pod in ("a", "b", "c"):
        for lo, hi in zip(edges[:-1], edges[1:]):
            v = np.abs(rng.standard_normal(32)) * 3
            series.append(Series(
                MetricName("req_duration",
                           [("pod", pod), ("vmrange", f"{lo}...{hi}")]), v))
    out = tfm.histogram_transform("histogram_quantile",
                                  [s.copy_shallow() for s in series], arg=0.9)
    assert len(out) == 3
    # compare against the oracle walk on the same converted buckets
    conv = tfm.vmrange_buckets_to_le([s.copy_shallow() for s in series])
    m = tfm.group_le_timeseries(conv)
    for k, xss in m.items():
        xss.sort(key=lambda x: x[0])
        xss = tfm._merge_same_le(xss)
        bv = np.stack([s.values for _, s in xss])
        les = np.asarray([le for le, _ in xss])
        goff = np.asarray([0, len(xss)], np.uint64)
        exp, _, _ = oracle.histogram_quantile(0.9, bv, les, goff)
        got = [s for s in out
               if s.mn.marshal_sorted() == xss[0][1].mn.marshal_sorted()]
        assert len(got) == 1
        np.testing.assert_array_equal(got[0].values.view(np.int64),
                                      exp[0].view(np.int64))


def test_histogram_avg_stddev_pipeline():
    from victoriametrics_amd import transform as tfm
    from victoriametrics_amd.binary_op import Series
    from victoriametrics_amd.metric_name import MetricName
    rng = np.random.default_rng(64)
    series = []
    for pod in ("x", "y"):
        for le in (1.0, 2.0, 4.0, math.inf):
            v = np.cumsum(np.abs(rng.standard_normal(16)))
            series.append(Series(
                MetricName("lat", [("pod", pod), ("le", str(le))]), v))
    for name in ("histogram_avg", "histogram_stddev", "histogram_stdvar"):
        out = tfm.histogram_transform(name,
                                      [s.copy_shallow() for s in series])
        assert len(out) == 2
        for s in out:
            assert s.values.shape == (16,)
