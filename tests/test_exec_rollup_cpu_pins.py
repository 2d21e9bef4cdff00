"""Rollup exec pins transcribed from TestExecSuccess (exec_test.go:5865,
:8434-8900) with deterministic inputs, run through the ORACLE rollup
backend on CPU (test infra) via the shared subquery-grid helper — the GPU
parity suite separately pins the device kernels against the same oracle,
so these close the loop to the reference's own expected arrays."""
import numpy as np

from test_exec_ru_ttf_pins import _sq_cpu
from victoriametrics_amd.decimal import go_round

TIME = np.asarray([1000.0, 1200, 1400, 1600, 1800, 2000])


def _eq(got, want):
    np.testing.assert_allclose(got, want, rtol=1e-12, atol=0)


def test_sum2_geomean_range_over_time_3i():
    # `sum2_over_time(alias(time()/100, "foobar")[3i])` -> 200..980
    got = _sq_cpu("sum2_over_time", lambda t: t / 100.0, window=600_000)
    _eq(got, [200, 308, 440, 596, 776, 980])
    # `round(geomean_over_time(alias(time()/100, "foobar")[3i]), 0.1)`
    got = _sq_cpu("geomean_over_time", lambda t: t / 100.0, window=600_000)
    _eq(go_round(got * 10.0) / 10.0, [7.8, 9.9, 11.9, 13.9, 15.9, 17.9])
    # `range_over_time(alias(time()/100, "foobar")[3i])` -> max-min = 4
    got = _sq_cpu("range_over_time", lambda t: t / 100.0, window=600_000)
    _eq(got, [4, 4, 4, 4, 4, 4])


def test_increase_delta_of_time():
    # :8324/:8345 increase/increase_pure(time()) and :8806 delta(time())
    # are the 200s/step slope
    for fn in ("increase", "increase_pure", "delta"):
        got = _sq_cpu(fn, lambda t: t)
        _eq(got, [200] * 6)
    # `increase(2000-time())` :8334 -> counter-reset clamping per window
    got = _sq_cpu("increase", lambda t: 2000.0 - t)
    _eq(got, [1000, 800, 600, 400, 200, 0])
    # `delta(-time())` :8828 keeps the sign; `delta(1)` :8839 -> 0
    _eq(_sq_cpu("delta", lambda t: -t), [-200] * 6)
    _eq(_sq_cpu("delta", lambda t: np.ones_like(t)), [0] * 6)


def test_prometheus_variants_window_strictness():
    # :8356/:8850 `*_prometheus(time())` with the default window: the
    # left-open (tEnd-window, tEnd] interval holds ONE sample, and the
    # Prometheus variants use no lookbehind sample -> empty result
    for fn in ("delta_prometheus", "increase_prometheus"):
        assert np.isnan(_sq_cpu(fn, lambda t: t)).all()
    # :8361/:8855 — at [201s] the window spans exactly two samples
    for fn in ("delta_prometheus", "increase_prometheus"):
        got = _sq_cpu(fn, lambda t: t, window=201_000)
        _eq(got, [200] * 6)


def test_zscore_over_time_const():
    # `zscore_over_time(1[100s:10s])` :8920 -> all zero (no variance)
    got = _sq_cpu("zscore_over_time", lambda t: np.ones_like(t),
                  window=100_000, sq_step=10_000)
    _eq(got, [0, 0, 0, 0, 0, 0])


def test_median_over_time_scalar():
    # `median_over_time(12)` :8872 -> 12 at every point
    got = _sq_cpu("median_over_time", lambda t: np.full_like(t, 12.0))
    _eq(got, [12] * 6)


def _fake(parent, fn, window=0, sq_step=0, tag=""):
    """rollup_* pseudo-function expansion (rollup.go:436-516) over the
    subquery grid, oracle backend; returns {rollup_tag: values}."""
    from victoriametrics_amd import engine
    plans = engine.rollup_fake_plans(parent, 1000_000, 2000_000, 200_000,
                                     tag=tag, window=window,
                                     keep_stale_nans=True)
    return {t: _sq_cpu(None, fn, window=window, sq_step=sq_step, plan=p)
            for t, p in plans}


def test_rollup_exec():
    # `sort(rollup(time()[:50s]))` exec_test.go — min/avg/max of the four
    # 50s-substep samples per 200s window
    out = _fake("rollup", lambda t: t, sq_step=50_000)
    _eq(out["min"], [850, 1050, 1250, 1450, 1650, 1850])
    _eq(out["avg"], [925, 1125, 1325, 1525, 1725, 1925])
    _eq(out["max"], [1000, 1200, 1400, 1600, 1800, 2000])


def test_rollup_increase_exec():
    # `sort(rollup_increase(time()))` — all three tags are the 200 slope
    out = _fake("rollup_increase", lambda t: t)
    for tag in ("min", "max", "avg"):
        _eq(out[tag], [200] * 6)


def test_rollup_rate_exec():
    # `rollup_rate((2200-time())[600s])` — per-window pairwise rates of
    # the counter-reset-adjusted decreasing series
    out = _fake("rollup_rate", lambda t: 2200.0 - t, window=600_000)
    _eq(out["avg"], [6, 5, 4, 3, 2, 1])
    _eq(out["max"], [7, 6, 5, 4, 3, 2])
    _eq(out["min"], [5, 4, 3, 2, 1, 0])
    # the optional second arg narrows to ONE labeled series
    out = _fake("rollup_rate", lambda t: 2200.0 - t, window=600_000,
                tag="max")
    assert list(out) == ["max"]
    _eq(out["max"], [7, 6, 5, 4, 3, 2])
    out = _fake("rollup_rate", lambda t: 2200.0 - t, window=600_000,
                tag="avg")
    _eq(out["avg"], [6, 5, 4, 3, 2, 1])


def test_rollup_deriv_and_scrape_interval_exec():
    # `sort(rollup_deriv(time()[100s:50s]))` — time() slope is 1
    out = _fake("rollup_deriv", lambda t: t, window=100_000,
                sq_step=50_000)
    for tag in ("min", "max", "avg"):
        _eq(out[tag], [1] * 6)
    # `rollup_scrape_interval(1[5m:10S])` — constant 10s sub-step
    out = _fake("rollup_scrape_interval", lambda t: np.ones_like(t),
                window=300_000, sq_step=10_000)
    for tag in ("min", "max", "avg"):
        _eq(out[tag], [10] * 6)


def test_at_modifier_exec():
    # exec_test.go:1202-1256 `time() @ X`: the rollup evaluates on the
    # single-point grid [at, at] and broadcasts to the report grid
    # (eval.go:903-950).  @1h -> 3600, @start() -> 1000, @end() -> 2000,
    # @end() offset 10m and @(end()-10m) -> 1400.
    from victoriametrics_amd import engine
    for at_s, want in ((3600.0, 3600.0), (1000.0, 1000.0),
                       (2000.0, 2000.0), (1400.0, 1400.0)):
        plan, report_ts, bc = engine.plan_with_at(
            "default_rollup", 1000_000, 2000_000, 200_000, [[at_s]])
        out = _sq_cpu(None, lambda t: t, start=plan._c.start,
                      end=plan._c.end, plan=plan)
        assert out.shape == (1,)
        full = bc(out.reshape(1, 1)).ravel()
        assert list(report_ts) == [1000_000, 1200_000, 1400_000,
                                   1600_000, 1800_000, 2000_000]
        _eq(full, [want] * 6)


def test_subquery_offset_exec():
    # `time()[:100] offset 0` :452 -> identity (100s sub-step divides the
    # grid); `time()[300:100] offset 100` :647 and the `i`-duration form
    # `time()[1.5i:0.5i] offset 0.5i` :658 -> one 100s sub-step back
    got = _sq_cpu("default_rollup", lambda t: t, sq_step=100_000)
    _eq(got, [1000, 1200, 1400, 1600, 1800, 2000])
    got = _sq_cpu("default_rollup", lambda t: t, window=300_000,
                  sq_step=100_000, start=1000_000 - 100_000,
                  end=2000_000 - 100_000)
    _eq(got, [900, 1100, 1300, 1500, 1700, 1900])
