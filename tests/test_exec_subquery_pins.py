"""Subquery pins transcribed from TestExecSuccess (exec_test.go:8279-8323):
rate() over subquery grids, including subquery step defaulting, the
alignStartEnd extension and offset-shifted evaluation — the engine side is
engine.rollup_subquery (evalRollupFuncWithSubquery, eval.go:1033) feeding
the device rollup, so these run on GPU.  Expected arrays are the
reference's own outputs, verbatim."""
import numpy as np
import pytest

from victoriametrics_amd import engine

pytestmark = pytest.mark.gpu

START_MS = 1000_000
END_MS = 2000_000
STEP_MS = 200_000


@pytest.fixture(scope="module", autouse=True)
def _init():
    engine.init()


def inner_2000_minus_time(sq_start, sq_end, sq_step):
    ts = np.arange(sq_start, sq_end + 1, sq_step, dtype=np.int64)
    return (2000.0 - ts / 1e3).reshape(1, -1)


def _run(start, end, window, sq_step):
    out, _, _ = engine.rollup_subquery(
        "rate", start, end, STEP_MS, window, sq_step,
        inner_2000_minus_time)
    return np.asarray(out).ravel()


def test_rate_subquery_default_step():
    # `rate((2000-time())[100s:])` :8279 -> [5, 4, 3, 2, 1, 0]
    # (empty step defaults to the outer step, eval.go:1041-1043)
    got = _run(START_MS, END_MS, 100_000, 0)
    np.testing.assert_allclose(got, [5, 4, 3, 2, 1, 0], rtol=1e-12, atol=0)


def test_rate_subquery_100s_step():
    # `rate((2000-time())[100s:100s])` :8290 -> [0, 0, 6, 4, 2, 0]
    got = _run(START_MS, END_MS, 100_000, 100_000)
    np.testing.assert_allclose(got, [0, 0, 6, 4, 2, 0], rtol=1e-12, atol=0)


def test_rate_subquery_offset():
    # `rate((2000-time())[100s:100s] offset 100s)` :8301 ->
    # [0, 0, 7, 5, 3, 1]: the grid shifts back by the offset and results
    # are reported on the original timestamps (eval.go:954-1008)
    off = 100_000
    got = _run(START_MS - off, END_MS - off, 100_000, 100_000)
    np.testing.assert_allclose(got, [0, 0, 7, 5, 3, 1], rtol=1e-12, atol=0)


def test_rate_subquery_double_offset():
    # `rate((2000-time())[100s:100s] offset 100s)[:] offset 100s` :8312 ->
    # [0, 0, 0, 7, 5, 3]: the outer subquery shifts the whole inner
    # evaluation back another 100s; with default inner step the outer
    # [:] rollup is default_rollup (last value) over the shifted grid.
    off = 100_000

    def inner(sq_start, sq_end, sq_step):
        out, _, _ = engine.rollup_subquery(
            "rate", sq_start - off, sq_end - off, sq_step, 100_000, 100_000,
            inner_2000_minus_time)
        return np.asarray(out)

    out, _, _ = engine.rollup_subquery(
        "default_rollup", START_MS - off, END_MS - off, STEP_MS, 0, 0, inner)
    got = np.asarray(out).ravel()
    np.testing.assert_allclose(got, [0, 0, 0, 7, 5, 3], rtol=1e-12, atol=0)


def _inner_fn(fn):
    def inner(sq_start, sq_end, sq_step):
        ts = np.arange(sq_start, sq_end + 1, sq_step, dtype=np.int64)
        return fn(ts / 1e3).reshape(1, -1)
    return inner


def _sq(func, window, sq_step, fn, arg=0.0):
    out, _, _ = engine.rollup_subquery(
        func, START_MS, END_MS, STEP_MS, window, sq_step, _inner_fn(fn),
        arg=arg)
    return np.asarray(out).ravel()


def _eq(got, want):
    g = np.asarray(got, np.float64)
    w = np.asarray(want, np.float64)
    gn, wn = np.isnan(g), np.isnan(w)
    assert (gn == wn).all(), (g, w)
    np.testing.assert_allclose(g[~gn], w[~wn], rtol=1e-12, atol=0)


def test_duration_over_time():
    # `duration_over_time((time()<1200)[600s:10s], 20s)` exec_test.go ->
    # [590, 580, 380, 180, nan, nan]
    got = _sq("duration_over_time", 600_000, 10_000,
              lambda t: np.where(t < 1200, t, np.nan), arg=20.0)
    _eq(got, [590, 580, 380, 180, np.nan, np.nan])


def test_mode_over_time():
    # `mode_over_time(round(time()/500)[100s:1s])` -> [2, 2, 3, 3, 4, 4]
    got = _sq("mode_over_time", 100_000, 1_000,
              lambda t: np.round(t / 500))
    _eq(got, [2, 2, 3, 3, 4, 4])


def test_rate_over_sum():
    # `rate_over_sum(round(time()/500)[100s:5s])` -> [.4,.4,.6,.6,.71,.8]
    # (compared after round(_, 0.01) as in the reference query)
    got = _sq("rate_over_sum", 100_000, 5_000, lambda t: np.round(t / 500))
    from victoriametrics_amd.decimal import go_round
    got = go_round(got * 100.0) / 100.0
    _eq(got, [0.4, 0.4, 0.6, 0.6, 0.71, 0.8])


def test_integrate_time():
    # `integrate(time()/1e3)` -> [160, 200, 240, 280, 320, 360]
    got = _sq("integrate", 0, 0, lambda t: t / 1e3)
    _eq(got, [160, 200, 240, 280, 320, 360])


def test_rate_of_time_is_one():
    # `rate(label_set(alias(time(), "foo"), "x", "y"))` -> [1]*6
    got = _sq("rate", 0, 0, lambda t: t)
    _eq(got, [1, 1, 1, 1, 1, 1])


def test_distinct_over_time_window():
    # `distinct_over_time((time() < 1700)[500s])` -> [3, 3, 3, 3, 2, 1]
    got = _sq("distinct_over_time", 500_000, 0,
              lambda t: np.where(t < 1700, t, np.nan))
    _eq(got, [3, 3, 3, 3, 2, 1])


def test_quantiles_over_time_single_sample():
    # `quantiles_over_time("phi", 0.5, 0.9, time()[100s:100s])`
    # exec_test.go:6513 — one sample per window, so every phi returns the
    # sample itself; the expansion tags each output phi=%g
    plans = engine.quantiles_over_time_plans(
        "phi", [0.5, 0.9], START_MS, END_MS, STEP_MS)
    assert [lbl for lbl, _ in plans] == ["0.5", "0.9"]
    for lbl, plan in plans:
        out, _, _ = engine.rollup_subquery(
            "quantile_over_time", START_MS, END_MS, STEP_MS, 100_000,
            100_000, _inner_fn(lambda t: t), arg=plan._c.arg)
        np.testing.assert_array_equal(
            np.asarray(out).ravel(), [1000, 1200, 1400, 1600, 1800, 2000])


def test_deriv_cases():
    # `deriv(1000)` -> zeros; `deriv(2*time())` -> 2 (exec_test.go:8784)
    got = _sq("deriv", 0, 0, lambda t: np.full_like(t, 1000.0))
    _eq(got, [0, 0, 0, 0, 0, 0])
    got = _sq("deriv", 0, 0, lambda t: 2.0 * t)
    _eq(got, [2, 2, 2, 2, 2, 2])


def test_lag_subquery_nondivisor_step():
    # `lag(time()[60s:17s])` exec_test.go:9294 -> [14, 10, 6, 2, 15, 11]
    # (17s inner step does not divide the 200s outer step — pins the
    # alignStartEnd + window walk at a misaligned cadence)
    got = _sq("lag", 60_000, 17_000, lambda t: t)
    _eq(got, [14, 10, 6, 2, 15, 11])


def test_tlast_change_over_time_cases():
    # exec_test.go:757-790: hit_last / hit_middle / miss over [1h] windows
    got = _sq("tlast_change_over_time", 3_600_000, 0, lambda t: t)
    _eq(got, [1000, 1200, 1400, 1600, 1800, 2000])
    got = _sq("tlast_change_over_time", 3_600_000, 0,
              lambda t: (t >= 1600).astype(np.float64))
    _eq(got, [np.nan, np.nan, np.nan, 1600, 1600, 1600])
    got = _sq("tlast_change_over_time", 3_600_000, 0,
              lambda t: np.ones_like(t, dtype=np.float64))
    assert np.isnan(got).all()  # constant: no change -> removeEmptySeries


def test_timestamp_with_name_filtered():
    # `timestamp_with_name(alias(time()>=1600,"foo"))` :791 — timestamp of
    # each point's last sample; filtered points carry no sample
    got = _sq("timestamp", 0, 0,
              lambda t: np.where(t >= 1600, t, np.nan))
    _eq(got, [np.nan, np.nan, np.nan, 1600, 1800, 2000])
