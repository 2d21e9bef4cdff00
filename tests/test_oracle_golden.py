"""Pins the CPU oracle against the reference's own in-repo golden vectors
(transcribed into tests/golden/rollup_golden.json from
app/vmselect/promql/rollup_test.go). This is the trust anchor for all parity
claims: GPU results are compared against the oracle, and the oracle is pinned
here against the reference's expected values."""
import json
import math
import os

import numpy as np
import pytest

from conftest import decode_float, decode_floats, assert_values_equal

import oracle

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden", "rollup_golden.json")))
TEST_VALUES = GOLDEN["test_values"]
TEST_TIMESTAMPS = GOLDEN["test_timestamps"]


def _case_values(case):
    v = case.get("values")
    if v is None:
        return list(TEST_VALUES), list(TEST_TIMESTAMPS)
    if v == "iota_1e4":
        n = 10000
        return [float(i) for i in range(n)], [i // 2 for i in range(n)]
    return decode_floats(v), case["timestamps"]


@pytest.mark.parametrize("case", GOLDEN["remove_counter_resets"], ids=lambda c: c["name"])
def test_remove_counter_resets(case):
    got = oracle.remove_counter_resets(decode_floats(case["values"]), case["timestamps"], case["msi"])
    assert_values_equal(got, decode_floats(case["expected"]), context=case["name"])


def test_remove_counter_resets_monotonic():
    # rollup_test.go:159-169: output must be monotonically non-decreasing
    values = [34.094223, 2.7518, 2.140669, 0.044878, 1.887095, 2.546569,
              2.490149, 0.045, 0.035684, 0.062454, 0.058296]
    got = oracle.remove_counter_resets(values, list(range(len(values))), 0)
    assert all(b >= a for a, b in zip(got, got[1:]))


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["delta_values"]))
def test_delta_values(i, case):
    v = decode_floats(case["values"])
    if case.get("rcr_first"):
        v = oracle.remove_counter_resets(v, TEST_TIMESTAMPS, 0)
    got = oracle.delta_values(v)
    assert_values_equal(got, decode_floats(case["expected"]), context=f"delta_values[{i}]")


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["deriv_values"]))
def test_deriv_values(i, case):
    v = decode_floats(case["values"])
    ts = case["timestamps"]
    if case.get("rcr_first"):
        v = oracle.remove_counter_resets(v, ts, 0)
    got = oracle.deriv_values(v, ts)
    assert_values_equal(got, decode_floats(case["expected"]), context=f"deriv_values[{i}]")


@pytest.mark.parametrize("case", GOLDEN["func_scalar"],
                         ids=lambda c: f"{c['func']}_{c.get('arg', '')}")
def test_func_scalar(case):
    """testRollupFunc harness (rollup_test.go:223-259): rfa over the whole
    fixture, prevValue=NaN, prevTimestamp=0, realPrev=NaN, realNext=0 (Go zero
    value), currTimestamp=0, window = ts[last]-ts[0]."""
    func = case["func"]
    values = list(map(float, TEST_VALUES))
    ts = list(TEST_TIMESTAMPS)
    if func in oracle.REMOVE_COUNTER_RESETS_FUNCS:
        values = list(oracle.remove_counter_resets(values, ts, 0))
    got = oracle.call_rollup_fn(
        func, values, ts,
        prev_value=math.nan, prev_timestamp=0,
        real_prev_value=math.nan, real_next_value=0.0,
        curr_timestamp=0, idx=0, window=ts[-1] - ts[0],
        arg=float(case.get("arg", 0.0)), arg2=float(case.get("arg2", 0.0)))
    expected = decode_float(case["expected"])
    assert_values_equal([got], [expected], rel=1e-13, context=func)


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["rollup_delta_cases"]))
def test_rollup_delta_direct(i, case):
    got = oracle.call_rollup_fn(
        "delta", decode_floats(case["values"]), [0] * len(case["values"]),
        prev_value=decode_float(case["prev"]),
        real_prev_value=decode_float(case["real_prev"]),
        real_next_value=decode_float(case["real_next"]))
    e = decode_float(case["expected"])
    if math.isnan(e):
        assert math.isnan(got)
    else:
        assert got == e, f"case {i}: got {got}, want {e}"


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["ideriv_cases"]))
def test_ideriv_direct(i, case):
    got = oracle.call_rollup_fn(
        "ideriv", decode_floats(case["values"]), case["timestamps"],
        prev_value=decode_float(case.get("prev", "nan")),
        prev_timestamp=case.get("prev_ts", 0))
    e = decode_float(case["expected"])
    if math.isnan(e):
        assert math.isnan(got)
    else:
        assert got == e, f"case {i}: got {got}, want {e}"


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["deriv_fast_prometheus_cases"]))
def test_deriv_fast_prometheus_direct(i, case):
    got = oracle.call_rollup_fn(
        "rate_prometheus", decode_floats(case["values"]),
        [0] * len(case["values"]), window=case["window"])
    e = decode_float(case["expected"])
    if math.isnan(e):
        assert math.isnan(got)
    else:
        assert got == e


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["outlier_iqr_cases"]))
def test_outlier_iqr_direct(i, case):
    got = oracle.call_rollup_fn("outlier_iqr_over_time",
                                decode_floats(case["values"]),
                                [0] * len(case["values"]))
    e = decode_float(case["expected"])
    if math.isnan(e):
        assert math.isnan(got)
    else:
        assert got == e


@pytest.mark.parametrize("case", GOLDEN["rollup_do"], ids=lambda c: c["name"])
def test_rollup_do_grids(case):
    """rollupConfig.Do grid tests. The reference's literals construct
    rollupConfig{} directly: MayAdjustWindow=false, samplesScannedPerCall=0."""
    values, ts = _case_values(case)
    rc = oracle.make_config(case["func"], case["start"], case["end"], case["step"],
                            window=case.get("window", 0),
                            lookback_delta=case.get("lookback_delta", 0),
                            may_adjust_window=0, samples_scanned_per_call=0)
    got, scanned = oracle.rollup_do(rc, values, ts)
    assert_values_equal(got, decode_floats(case["expected"]), context=case["name"])
    if "samples_scanned" in case:
        assert scanned == case["samples_scanned"], \
            f"{case['name']}: samplesScanned {scanned} != {case['samples_scanned']}"


def test_quantile_pins():
    # quantileSorted pins via quantile_over_time scalars are in func_scalar;
    # basic invariants here.
    assert math.isinf(oracle.quantile(-1, [1.0, 2.0]))
    assert oracle.quantile(0, [3.0, 1.0, 2.0]) == 1.0
    assert oracle.quantile(1, [3.0, 1.0, 2.0]) == 3.0
    assert math.isnan(oracle.quantile(0.5, []))
    assert math.isnan(oracle.quantile(math.nan, [1.0]))


def test_linear_regression_via_deriv():
    # TestLinearRegression (rollup_test.go:540-555) via deriv/predict_linear:
    # interceptTime = timestamps[0] + 100.
    for case in GOLDEN["linear_regression_cases"]:
        v_exp = decode_float(case["exp_v"])
        k_exp = decode_float(case["exp_k"])
        ts = case["timestamps"]
        k = oracle.call_rollup_fn("deriv", decode_floats(case["values"]), ts,
                                  curr_timestamp=ts[0] + 100)
        v = oracle.call_rollup_fn("predict_linear", decode_floats(case["values"]), ts,
                                  curr_timestamp=ts[0] + 100, arg=0.0)
        if math.isnan(k_exp):
            assert math.isnan(k)
        else:
            assert_values_equal([k], [k_exp], context="lr_k")
        if math.isnan(v_exp):
            assert math.isnan(v)
        else:
            assert_values_equal([v], [v_exp], context="lr_v")


def test_stale_nan_bits():
    sn = oracle.stale_nan()
    assert math.isnan(sn)
    assert oracle.lib().vm_is_stale_nan(sn) == 1
    assert oracle.lib().vm_is_stale_nan(math.nan) == 0
    v, t = oracle.drop_stale_nans([1.0, sn, 2.0], [1, 2, 3])
    assert list(v) == [1.0, 2.0] and list(t) == [1, 3]


def test_scrape_interval_tiers():
    # getMaxPrevInterval tiers (rollup.go:899-919)
    f = oracle.lib().vm_get_max_prev_interval
    assert f(1000) == 5000
    assert f(3000) == 9000
    assert f(8000) == 16000
    assert f(16000) == 24000
    assert f(30000) == 37500
    assert f(60000) == 67500


def test_incremental_aggr_vectors():
    """TestIncrementalAggr (aggr_incremental_test.go:15-100): the exact
    per-point incremental aggregate expectations over the 7-series fixture,
    via the oracle's update/merge/finalize path (a last_over_time rollup at
    the fixture's own timestamps reproduces the raw column)."""
    nan = math.nan
    values = [
        [1, nan, 2, nan],
        [3, nan, nan, 4],
        [nan, nan, 5, 6],
        [7, nan, 8, 9],
        [4, nan, nan, nan],
        [2, nan, 3, 2],
        [0, nan, 1, 1],
    ]
    ts_grid = [100_000, 200_000, 300_000, 400_000]
    # CSR with the NaN samples PRESENT (the reference feeds NaN values to
    # updateTimeseries, which skips them)
    ts = np.asarray(ts_grid * len(values), np.int64)
    vals = np.asarray([v for row in values for v in row], np.float64)
    offsets = np.arange(len(values) + 1, dtype=np.uint64) * 4
    gids = np.zeros(len(values), np.int32)
    expected = {
        "sum": [17, nan, 19, 22],
        "min": [0, nan, 1, 1],
        "max": [7, nan, 8, 9],
        "avg": [2.8333333333333335, nan, 3.8, 4.4],
        "count": [6, nan, 5, 5],
        "sum2": [79, nan, 103, 138],
        "geomean": [0, nan, 2.9925557394776896, 3.365865436338599],
    }
    # last_over_time with window=step reproduces each raw sample at its own
    # grid point (NaN values count as present samples but NaN results)
    rc = oracle.make_config("last_over_time", ts_grid[0], ts_grid[-1],
                            100_000, window=100_000,
                            may_adjust_window=0)
    for name, exp in expected.items():
        out, counts, _ = oracle.rollup_eval_batch(
            rc, ts, vals, offsets, group_ids=gids, n_groups=1, aggr=name)
        got = out[0]
        for g, e in zip(got, exp):
            if math.isnan(e):
                assert math.isnan(g), (name, got, exp)
            else:
                assert abs(g - e) < 1e-12, (name, got, exp)
