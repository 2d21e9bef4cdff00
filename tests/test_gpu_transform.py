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
