"""Pins the topk/bottomk + histogram_quantile oracle against expected values
from the reference's exec tests (app/vmselect/promql/exec_test.go topk cases
7077-7400 on the fixed 6-point grid, histogram_quantile cases 4482-4696)."""
import math

import numpy as np
import pytest

import oracle

# exec_test fixture: time() on the 6-point grid [1000e3..2000e3:200e3], in s
TIME = np.array([1000.0, 1200.0, 1400.0, 1600.0, 1800.0, 2000.0])
S_CONST10 = np.full(6, 10.0)
S_TIME150 = TIME / 150.0  # [6.667, 8, 9.333, 10.667, 12, 13.333]


def nan_eq(a, b):
    a, b = np.asarray(a), np.asarray(b)
    return (np.isnan(a) == np.isnan(b)).all() and np.allclose(
        a[~np.isnan(a)], b[~np.isnan(b)], rtol=1e-12, atol=0)


def test_topk_1_pointwise():
    # exec_test.go:7077 `topk(1, ...)`
    m = np.vstack([S_CONST10, S_TIME150])
    out = oracle.topk_pointwise(m, 1)
    assert nan_eq(out[0], [10, 10, 10, np.nan, np.nan, np.nan])
    assert nan_eq(out[1], [np.nan, np.nan, np.nan, 10.666666666666666, 12,
                           13.333333333333334])


def test_topk_2_and_100500_keep_all():
    # exec_test.go:7344 topk(2), 7374 topk(100500): both series kept intact
    m = np.vstack([S_CONST10, S_TIME150])
    for k in (2, 100500):
        out = oracle.topk_pointwise(m, k)
        assert nan_eq(out[0], S_CONST10)
        assert nan_eq(out[1], S_TIME150)


def test_topk_nan_k_drops_all():
    # exec_test.go:7368 topk(NaN) -> empty result (k=0 keeps nothing)
    m = np.vstack([S_CONST10, S_TIME150])
    out = oracle.topk_pointwise(m, math.nan)
    assert np.isnan(out).all()


def test_bottomk_1():
    # bottomk keeps the smallest per point
    m = np.vstack([S_CONST10, S_TIME150])
    out = oracle.topk_pointwise(m, 1, reverse=True)
    # time/150 < 10 for first 3 points; 10 smaller afterwards
    assert nan_eq(out[0], [np.nan, np.nan, np.nan, 10, 10, 10])
    assert nan_eq(out[1], [6.666666666666667, 8, 9.333333333333334,
                           np.nan, np.nan, np.nan])


def test_topk_min_range():
    # exec_test.go:7101 topk_min(1): min(s0)=10 beats min(s1)=6.67
    m = np.vstack([S_CONST10, S_TIME150])
    sel, _ = oracle.topk_range(m, 1, "min")
    assert list(sel) == [0]
    # bottomk_min(1) keeps the series with the smallest min
    sel, _ = oracle.topk_range(m, 1, "min", reverse=True)
    assert list(sel) == [1]
    # topk_avg / topk_max / topk_last orderings
    sel, _ = oracle.topk_range(m, 1, "max")
    assert list(sel) == [1]  # max(s1)=13.33 > 10
    sel, _ = oracle.topk_range(m, 1, "last")
    assert list(sel) == [1]  # last(s1)=13.33


def test_topk_range_remaining_sum():
    m = np.vstack([S_CONST10, S_TIME150, S_TIME150 * 2])
    sel, rem = oracle.topk_range(m, 1, "avg", remaining=True)
    assert list(sel) == [2]
    assert nan_eq(rem, S_CONST10 + S_TIME150)


def test_topk_range_nan_series_ranks_last():
    m = np.vstack([np.full(6, np.nan), S_CONST10])
    sel, _ = oracle.topk_range(m, 1, "avg")
    assert list(sel) == [1]


def test_hq_single_bucket():
    # exec_test.go:4482: histogram_quantile(0.6, {le="200"}=100) -> 120
    bv = np.full((1, 6), 100.0)
    out, _, _ = oracle.histogram_quantile(0.6, bv, [200.0], [0, 1])
    assert nan_eq(out[0], np.full(6, 120.0))


def test_hq_max_phi_zero_lower_bucket():
    # exec_test.go:4683: phi=1 over {le=55: 0, le=200: 100} -> 200
    bv = np.vstack([np.zeros(6), np.full(6, 100.0)])
    out, _, _ = oracle.histogram_quantile(1.0, bv, [55.0, 200.0], [0, 2])
    assert nan_eq(out[0], np.full(6, 200.0))


def test_hq_zero_last_bucket_is_nan():
    # vLast == 0 -> NaN (transform.go:1038-1040)
    bv = np.zeros((2, 6))
    out, _, _ = oracle.histogram_quantile(0.5, bv, [1.0, math.inf], [0, 2])
    assert np.isnan(out).all()


def test_hq_phi_out_of_range():
    bv = np.vstack([np.full(6, 40.0), np.full(6, 100.0)])
    out, lo, hi = oracle.histogram_quantile(-0.5, bv, [1.0, 2.0], [0, 2],
                                            bounds=True)
    assert (out == -math.inf).all() and (lo == -math.inf).all()
    assert np.allclose(hi, 40.0)
    out, lo, hi = oracle.histogram_quantile(1.5, bv, [1.0, 2.0], [0, 2],
                                            bounds=True)
    assert (out == math.inf).all() and (hi == math.inf).all()
    assert np.allclose(lo, 100.0)


def test_hq_broken_buckets_fixed():
    # decreasing upper bucket values are substituted (fixBrokenBuckets)
    bv = np.vstack([np.full(6, 50.0), np.full(6, 30.0), np.full(6, 100.0)])
    out, _, _ = oracle.histogram_quantile(0.5, bv, [1.0, 2.0, 4.0], [0, 3])
    # after fix: buckets 50,50,100; vReq=50; le=1: v=50>=50 -> interp from 0
    assert np.allclose(out[0], 1.0)


def test_hq_inf_le_tail():
    # quantile falling in +Inf bucket returns last non-inf le
    bv = np.vstack([np.full(6, 10.0), np.full(6, 100.0)])
    out, lo, hi = oracle.histogram_quantile(0.99, bv, [5.0, math.inf], [0, 2],
                                            bounds=True)
    assert np.allclose(out[0], 5.0)
    assert (hi == math.inf).all()


def test_hq_multiple_groups():
    bv = np.vstack([np.full(6, 100.0), np.full(6, 40.0), np.full(6, 100.0)])
    out, _, _ = oracle.histogram_quantile(0.6, bv, [200.0, 1.0, 2.0], [0, 1, 3])
    assert np.allclose(out[0], 120.0)
    # group 2: buckets 40,100 @ le 1,2: vReq=60: le=2: 40<60<=100:
    # q = 1 + (2-1)*(60-40)/(100-40) = 1.3333...
    assert np.allclose(out[1], 1 + 20.0 / 60.0)
