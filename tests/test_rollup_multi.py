"""Multi-series rollup functions (rollup.go:1490-1560): host
implementations over decoded columns — count_values_over_time
(newRollupCountValues) and histogram_over_time (rollupHistogram).
The vmrange strings are pinned verbatim against exec_test.go:6643's
expected tag values ("1.000e+00...1.136e+00" etc.)."""
import math

import numpy as np

from victoriametrics_amd.decimal import STALE_NAN_BITS
from victoriametrics_amd.metric_name import MetricName
from victoriametrics_amd.rollup_multi import (count_values_over_time,
                                              format_go_float_g,
                                              histogram_over_time)

START = 1_600_000_000_000
STEP = 15_000


def _mn():
    return MetricName(b"m", [(b"x", b"y")])


def test_format_go_float_g_matches_strconv():
    # hand-verified against Go strconv.FormatFloat(v, 'g', -1, 64)
    cases = [
        (1.5, "1.5"), (0.0001, "0.0001"), (0.00001, "1e-05"),
        (1e6, "1e+06"), (1234567.0, "1.234567e+06"), (123456.0, "123456"),
        (1e21, "1e+21"), (-2.5e-07, "-2.5e-07"), (3.0, "3"), (0.0, "0"),
        (0.4, "0.4"), (999999.9, "999999.9"), (1e100, "1e+100"),
        (float("nan"), "NaN"), (float("inf"), "+Inf"),
        (float("-inf"), "-Inf"), (-0.0, "-0"), (100.0, "100"),
        (0.30000000000000004, "0.30000000000000004"),
    ]
    for v, want in cases:
        assert format_go_float_g(v) == want, (v, format_go_float_g(v), want)


def test_count_values_over_time_basic():
    # 3 points, window=step: values fall in (t_end-15s, t_end]
    ts = np.array([START - 5_000, START, START + 10_000, START + 15_000,
                   START + 25_000, START + 30_000], dtype=np.int64)
    vals = np.array([0.4, 0.4, 0.0, 0.4, 1e6, 1e6])
    series, scanned = count_values_over_time(
        "foo", ts, vals, _mn(), START, START + 2 * STEP, STEP, STEP)
    assert scanned == 6
    by_label = {s.mn.get_tag_value(b"foo"): s for s in series}
    assert set(by_label) == {b"0.4", b"0", b"1e+06"}
    s04 = by_label[b"0.4"].values
    assert s04[0] == 2.0 and s04[1] == 1.0 and math.isnan(s04[2])
    s0 = by_label[b"0"].values
    assert math.isnan(s0[0]) and s0[1] == 1.0 and math.isnan(s0[2])
    s1m = by_label[b"1e+06"].values
    assert math.isnan(s1m[0]) and math.isnan(s1m[1]) and s1m[2] == 2.0
    # metric group reset (not in rollupFuncsKeepMetricName), x=y preserved
    for s in series:
        assert s.mn.metric_group == b""
        assert s.mn.get_tag_value(b"x") == b"y"


def test_count_values_replaces_existing_label_and_keeps_names():
    ts = np.array([START], dtype=np.int64)
    vals = np.array([7.0])
    mn = MetricName(b"m", [(b"foo", b"old")])
    series, _ = count_values_over_time(
        "foo", ts, vals, mn, START, START, STEP, STEP,
        keep_metric_names=True)
    assert len(series) == 1
    assert series[0].mn.metric_group == b"m"
    tags = dict(series[0].mn.tags)
    assert tags == {b"foo": b"7"}


def test_count_values_drops_stale_nans():
    stale = np.uint64(STALE_NAN_BITS).view(np.float64)
    ts = np.array([START - 1000, START], dtype=np.int64)
    vals = np.array([stale, 2.0])
    series, scanned = count_values_over_time(
        "v", ts, vals, _mn(), START, START, STEP, STEP)
    assert scanned == 1
    assert len(series) == 1
    assert series[0].mn.get_tag_value(b"v") == b"2"


def test_histogram_over_time_vmranges():
    # exec_test.go:6643 pins the bucket string for values in [1.0, 1.136):
    # "1.000e+00...1.136e+00"
    ts = np.array([START - 5_000, START, START + 15_000], dtype=np.int64)
    vals = np.array([1.05, 1.10, 2.0])
    series, scanned = histogram_over_time(
        ts, vals, _mn(), START, START + STEP, STEP, STEP)
    assert scanned == 3
    by_range = {s.mn.get_tag_value(b"vmrange"): s for s in series}
    b1 = by_range[b"1.000e+00...1.136e+00"].values
    assert b1[0] == 2.0 and math.isnan(b1[1])
    # 2.0 lands in the bucket ending at 2.154e+00 (18 buckets/decade)
    (k2,) = [k for k in by_range if k != b"1.000e+00...1.136e+00"]
    assert k2.endswith(b"...2.154e+00")
    b2 = by_range[k2].values
    assert math.isnan(b2[0]) and b2[1] == 1.0
    for s in series:
        assert s.mn.metric_group == b""


def test_histogram_over_time_zero_and_upper():
    ts = np.array([START + 1, START + 2, START + 3], dtype=np.int64)
    vals = np.array([0.0, 1e19, -5.0])  # zero -> lower; 1e19 -> upper;
    series, _ = histogram_over_time(     # negatives skipped entirely
        ts, vals, _mn(), START + STEP, START + STEP, STEP, STEP)
    by_range = {s.mn.get_tag_value(b"vmrange"): s for s in series}
    assert set(by_range) == {b"0...1.000e-09", b"1.000e+18...+Inf"}
    assert by_range[b"0...1.000e-09"].values[0] == 1.0
    assert by_range[b"1.000e+18...+Inf"].values[0] == 1.0


def test_window_required():
    ts = np.array([START], dtype=np.int64)
    vals = np.array([1.0])
    import pytest
    with pytest.raises(ValueError):
        count_values_over_time("foo", ts, vals, _mn(), START, START, STEP, 0)


def test_count_values_fuzz_vs_brute_force():
    """Random ragged inputs vs an independent brute-force model: per grid
    point the multiset of formatted window values must match, and summing
    every output series per point must equal count_over_time."""
    rng = np.random.default_rng(99)
    for trial in range(40):
        n = int(rng.integers(1, 60))
        ts = np.sort(rng.integers(START - 400_000, START + 400_000, n)
                     ).astype(np.int64)
        vals = np.round(rng.standard_normal(n) * rng.choice([1, 10, 1e6]), 2)
        vals[rng.random(n) < 0.2] = np.round(vals[0], 2)  # duplicates
        window = int(rng.choice([15_000, 60_000, 250_000]))
        start = START + int(rng.integers(-5, 5)) * STEP
        end = start + int(rng.integers(0, 6)) * STEP
        series, scanned = count_values_over_time(
            "v", ts, vals, _mn(), start, end, STEP, window)
        n_grid = 1 + (end - start) // STEP
        exp_scanned = 0
        for g in range(n_grid):
            t_end = start + g * STEP
            in_win = (ts > t_end - window) & (ts <= t_end)
            exp_scanned += int(in_win.sum())
            counts = {}
            for v in vals[in_win]:
                k = format_go_float_g(float(v))
                counts[k] = counts.get(k, 0) + 1
            got = {}
            for s in series:
                x = s.values[g]
                if not math.isnan(x):
                    got[s.mn.get_tag_value(b"v").decode()] = int(x)
            assert got == counts, (trial, g, got, counts)
        assert scanned == exp_scanned


def test_histogram_over_time_total_counts_property():
    """Sum over all vmrange series at each point == number of COUNTABLE
    window values (NaN and negatives are skipped by Histogram.Update)."""
    rng = np.random.default_rng(7)
    for trial in range(25):
        n = int(rng.integers(1, 80))
        ts = np.sort(rng.integers(START - 300_000, START + 300_000, n)
                     ).astype(np.int64)
        vals = rng.standard_normal(n) * 10.0 ** rng.integers(-5, 6, n)
        vals[rng.random(n) < 0.1] = np.nan
        window = int(rng.choice([30_000, 120_000]))
        start, end = START, START + 4 * STEP
        series, _ = histogram_over_time(ts, vals, _mn(), start, end, STEP,
                                        window)
        n_grid = 1 + (end - start) // STEP
        for g in range(n_grid):
            t_end = start + g * STEP
            in_win = (ts > t_end - window) & (ts <= t_end)
            w = vals[in_win]
            countable = int((~np.isnan(w) & (w >= 0)).sum())
            total = sum(int(s.values[g]) for s in series
                        if not math.isnan(s.values[g]))
            assert total == countable, (trial, g, total, countable)
