"""GPU parity tests: the HIP engine against the CPU oracle on identical
seeded inputs.  The oracle is pinned against the reference's own golden
vectors in test_oracle_golden.py, so green here means parity with the
reference Go path.

Bar: bit-exact for arithmetic-only functions (each grid point
is one lane running the same left-to-right loops), rtol<=1e-12 for functions
with non-correctly-rounded transcendentals (pow), rtol<=1e-9 for cross-series
float sums whose atomic ordering differs.
"""
import math

import numpy as np
import pytest

import oracle
from seriesgen import ragged_batch, assert_parity

pytestmark = pytest.mark.gpu

START = 1_000_000_000_000
STEP = 15_000

# funcs whose device implementation uses pow (not correctly rounded): compare
# with rtol instead of bitwise.
POW_FUNCS = {"geomean_over_time", "hoeffding_bound_lower",
             "hoeffding_bound_upper"}  # pow/log: device-libm ulp

ALL_FUNCS = sorted(set(oracle.FUNC_IDS) - {"increase_prometheus", "timestamp",
                                           "timestamp_with_name"})
ARG_FUNCS = {
    "quantile_over_time": 0.9, "count_le_over_time": 5.0,
    "count_gt_over_time": 5.0, "count_eq_over_time": 1.0,
    "count_ne_over_time": 1.0, "share_le_over_time": 5.0,
    "share_gt_over_time": 5.0, "share_eq_over_time": 1.0,
    "sum_le_over_time": 5.0, "sum_gt_over_time": 5.0,
    "sum_eq_over_time": 1.0, "predict_linear": 120.0,
    "duration_over_time": 60.0, "quantile_over_time": 0.9,
    "hoeffding_bound_lower": 0.9, "hoeffding_bound_upper": 0.9,
    "holt_winters": 0.3,
}
ARG2_FUNCS = {"holt_winters": 0.4}


@pytest.fixture(scope="module")
def engine():
    from victoriametrics_amd import engine as e
    e.init()
    return e


@pytest.fixture(scope="module")
def counter_small():
    from victoriametrics_amd import synth
    return synth.counter_batch(2048, 240, START, seed=8428)


def _oracle_batch(engine_plan, ts, vals, offsets, group_ids=None, n_groups=0,
                  aggr="none", n_threads=4):
    rc = oracle.RollupConfigC(
        func=engine_plan._c.func,
        may_adjust_window=engine_plan._c.may_adjust_window,
        start=engine_plan._c.start, end=engine_plan._c.end,
        step=engine_plan._c.step, window=engine_plan._c.window,
        lookback_delta=engine_plan._c.lookback_delta,
        min_staleness_interval=engine_plan._c.min_staleness_interval,
        is_default_rollup=engine_plan._c.is_default_rollup,
        samples_scanned_per_call=engine_plan._c.samples_scanned_per_call,
        arg=engine_plan._c.arg, arg2=engine_plan._c.arg2)
    return oracle.rollup_eval_batch(
        rc, ts, vals, offsets, group_ids=group_ids, n_groups=n_groups,
        aggr=aggr,
        remove_counter_resets=bool(engine_plan._c.remove_counter_resets),
        max_staleness_interval=engine_plan._c.max_staleness_interval,
        drop_stale_nans=bool(engine_plan._c.drop_stale_nans),
        n_threads=n_threads, pre_func=engine_plan._c.pre_func)


def test_rate_counter_batch_bitexact(engine, counter_small):
    ts, vals, offsets = counter_small
    end = START + 239 * STEP
    plan = engine.RollupPlan("rate", START, end, STEP, window=300_000)
    out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
    ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
    assert scanned == ref_scanned
    assert_parity(out, ref, exact=True, context="rate")


@pytest.mark.parametrize("func", ALL_FUNCS)
def test_all_funcs_ragged(engine, func):
    """Every rollup function over a ragged batch with empty series, duplicate
    timestamps and varying lengths, window explicitly set."""
    ts, vals, offsets = ragged_batch(300, 260, START, seed=99, dup_p=0.02)
    end = START + 100 * STEP
    plan = engine.RollupPlan(func, START, end, STEP, window=200_000,
                             arg=ARG_FUNCS.get(func, 0.0),
                             arg2=ARG2_FUNCS.get(func, 0.0))
    out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
    ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
    assert scanned == ref_scanned, f"{func}: scanned {scanned} != {ref_scanned}"
    assert_parity(out, ref, exact=func not in POW_FUNCS, context=func)


@pytest.mark.parametrize("func", ["rate", "delta", "avg_over_time",
                                  "default_rollup", "deriv_fast"])
def test_window_autoadjust(engine, func):
    """window=0 paths: per-series scrape-interval estimate + window adjust
    (rollup.go:719-756)."""
    ts, vals, offsets = ragged_batch(200, 120, START, seed=7)
    end = START + 50 * STEP
    for lbd in (0, 40_000):
        plan = engine.RollupPlan(func, START, end, STEP, window=0,
                                 lookback_delta=lbd)
        out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
        ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
        assert scanned == ref_scanned, f"{func} lbd={lbd}"
        assert_parity(out, ref, exact=True, context=f"{func} lbd={lbd}")


def test_instant_query(engine):
    """start == end (instant query): maxPrevInterval = step directly."""
    ts, vals, offsets = ragged_batch(100, 100, START, seed=3)
    plan = engine.RollupPlan("rate", START, START, STEP, window=300_000)
    out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
    ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
    assert scanned == ref_scanned
    assert_parity(out, ref, exact=True, context="instant rate")


def test_stale_nans_dropped(engine):
    ts, vals, offsets = ragged_batch(150, 150, START, seed=11, stale_p=0.05)
    end = START + 60 * STEP
    for func in ("rate", "avg_over_time", "default_rollup",
                 "stale_samples_over_time"):
        plan = engine.RollupPlan(func, START, end, STEP, window=120_000)
        out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
        ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
        assert scanned == ref_scanned, func
        assert_parity(out, ref, exact=True, context=f"stale {func}")


def test_staleness_gap_resets(engine):
    """removeCounterResets with max_staleness_interval (issue 8072 path):
    engine sets it when lookback_delta != 0."""
    ts, vals, offsets = ragged_batch(150, 150, START, seed=13)
    end = START + 60 * STEP
    plan = engine.RollupPlan("increase", START, end, STEP, window=120_000,
                             lookback_delta=20_000)
    out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
    ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
    assert scanned == ref_scanned
    assert_parity(out, ref, exact=True, context="staleness gap")


@pytest.mark.parametrize("max_len,name", [(520, "wave_edge"),
                                          (3000, "block"),
                                          (4100, "block_edge"),
                                          (9000, "huge")])
def test_long_series_kernels(engine, max_len, name):
    """Exercises the block (512 < n <= 4064) and huge (n > 4064) kernels,
    including series lengths straddling each boundary."""
    ts, vals, offsets = ragged_batch(48, max_len, START, seed=max_len)
    end = START + 100 * STEP
    plan = engine.RollupPlan("rate", START, end, STEP, window=400_000)
    out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
    ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
    assert scanned == ref_scanned, name
    assert_parity(out, ref, exact=True, context=name)
    plan2 = engine.RollupPlan("avg_over_time", START, end, STEP, window=400_000)
    out2, _, _ = engine.rollup_eval(plan2, ts, vals, offsets)
    ref2, _, _ = _oracle_batch(plan2, ts, vals, offsets)
    assert_parity(out2, ref2, exact=True, context=name + "_avg")


@pytest.mark.parametrize("aggr", ["sum", "min", "max", "avg", "count",
                                  "sum2", "group", "geomean"])
def test_grouped_aggregation(engine, counter_small, aggr):
    """sum/min/max/avg/... by(pod) — grouped matrices vs the oracle.
    min/max/count/group bit-exact; sums/avg at 1e-9 (atomic order)."""
    ts, vals, offsets = counter_small
    n_series = len(offsets) - 1
    n_groups = 37
    rng = np.random.default_rng(5)
    gids = rng.integers(0, n_groups, n_series).astype(np.int32)
    end = START + 239 * STEP
    plan = engine.RollupPlan("rate", START, end, STEP, window=300_000,
                             aggr=aggr)
    out, counts, scanned = engine.rollup_eval(plan, ts, vals, offsets,
                                              group_ids=gids,
                                              n_groups=n_groups)
    ref, ref_counts, ref_scanned = _oracle_batch(plan, ts, vals, offsets,
                                                 group_ids=gids,
                                                 n_groups=n_groups, aggr=aggr)
    assert scanned == ref_scanned
    exact = aggr in ("min", "max", "count", "group")
    assert_parity(out, ref, exact=exact, rtol=1e-9, context=f"aggr {aggr}")


def test_grouped_skip_finalize_allreduce_cut(engine, counter_small):
    """skip_finalize exposes the pre-all-reduce matrices; finalizing on the
    host must match the fused finalize."""
    ts, vals, offsets = counter_small
    n_series = len(offsets) - 1
    n_groups = 16
    gids = (np.arange(n_series) % n_groups).astype(np.int32)
    end = START + 239 * STEP
    plan_raw = engine.RollupPlan("rate", START, end, STEP, window=300_000,
                                 aggr="avg", skip_finalize=True)
    raw, counts, _ = engine.rollup_eval(plan_raw, ts, vals, offsets,
                                        group_ids=gids, n_groups=n_groups)
    host_fin = engine.aggr_finalize("avg", raw.copy(), counts)
    plan_fin = engine.RollupPlan("rate", START, end, STEP, window=300_000,
                                 aggr="avg")
    fin, _, _ = engine.rollup_eval(plan_fin, ts, vals, offsets,
                                   group_ids=gids, n_groups=n_groups)
    # two independent execs accumulate f64 atomics in different orders, so
    # the comparison bar is the float-sum bar (1e-9), not bit equality
    assert_parity(host_fin.reshape(fin.shape), fin, exact=False, rtol=1e-9,
                  context="skip_finalize")


def test_negative_group_id_skipped(engine):
    """group_id == -1 means 'not grouped' (series skipped), mirroring the
    aggregate limit skip (aggr_incremental.go:117-121)."""
    ts, vals, offsets = ragged_batch(64, 100, START, seed=21)
    n_series = len(offsets) - 1
    gids = np.full(n_series, -1, dtype=np.int32)
    gids[::2] = 0
    end = START + 30 * STEP
    plan = engine.RollupPlan("rate", START, end, STEP, window=120_000,
                             aggr="sum")
    out, _, _ = engine.rollup_eval(plan, ts, vals, offsets, group_ids=gids,
                                   n_groups=1)
    ref, _, _ = _oracle_batch(plan, ts, vals, offsets, group_ids=gids,
                              n_groups=1, aggr="sum")
    assert_parity(out, ref, exact=False, rtol=1e-9, context="gid -1")


def test_full_config2_checksums(engine):
    """Full BASELINE config-2 size (1M x 240) — size-independent properties
    plus full comparison against the multithreaded oracle (count/min/max of
    the rate grid; bit-exact everywhere since rollup is per-series)."""
    from victoriametrics_amd import synth
    n_series, n_samples = 1_000_000, 240
    ts, vals, offsets = synth.counter_batch(n_series, n_samples, START)
    end = START + 239 * STEP
    plan = engine.RollupPlan("rate", START, end, STEP, window=300_000)
    with engine.SeriesBatch(ts, vals, offsets) as b:
        out, _, scanned = b.exec(plan)
    # oracle on a deterministic 1% subsample of series, bit-exact
    sel = np.arange(0, n_series, 97)
    sub_off = [0]
    sub_ts, sub_vals = [], []
    for s in sel:
        lo, hi = int(offsets[s]), int(offsets[s + 1])
        sub_ts.append(ts[lo:hi])
        sub_vals.append(vals[lo:hi])
        sub_off.append(sub_off[-1] + hi - lo)
    ref, _, _ = _oracle_batch(
        plan, np.concatenate(sub_ts), np.concatenate(sub_vals),
        np.asarray(sub_off, dtype=np.uint64))
    assert_parity(out[sel], ref, exact=True, context="config2 subsample")
    # whole-grid properties
    assert scanned == n_series * n_samples + n_series * 2 * plan.n_grid
    nan_frac = np.isnan(out).mean()
    assert nan_frac < 0.02, f"unexpected NaN fraction {nan_frac}"
    assert np.nanmin(out) >= 0.0  # rates of counters are non-negative


# ---- topk family + histogram_quantile (configs 4-5 rows) ----

def test_topk_range_gpu(engine, counter_small):
    ts, vals, offsets = counter_small
    end = START + 239 * STEP
    plan = engine.RollupPlan("avg_over_time", START, end, STEP, window=300_000)
    with engine.SeriesBatch(ts, vals, offsets) as b:
        host_out, _, _ = b.exec(plan)
        for summary in ("avg", "min", "max", "median", "last"):
            for rev in (False, True):
                sel, rem = engine.topk_range(b, 100, summary=summary,
                                             reverse=rev, remaining=True)
                ref_sel, ref_rem = oracle.topk_range(host_out, 100, summary,
                                                     reverse=rev,
                                                     remaining=True)
                # the selected summary-value sequences must be identical;
                # WHICH rows carry a boundary-tied value is unspecified in
                # the reference too (unstable sort), so compare ids only
                # away from the boundary value.
                sv = [oracle.topk_summary(summary, host_out[i]) for i in sel]
                rv = [oracle.topk_summary(summary, host_out[i]) for i in ref_sel]
                assert sv == rv, f"{summary} rev={rev}: value order differs"
                boundary = sv[-1] if sv else None
                ids_g = {i for i, v in zip(sel, sv) if v != boundary}
                ids_r = {i for i, v in zip(ref_sel, rv) if v != boundary}
                assert ids_g == ids_r, f"{summary} rev={rev}"
                tied = sv.count(boundary) if sv else 0
                all_vals = [oracle.topk_summary(summary, host_out[i])
                            for i in range(host_out.shape[0])]
                if tied == sum(1 for v in all_vals if v == boundary):
                    # no boundary tie ambiguity: remaining sums comparable
                    gn, rn = np.isnan(rem), np.isnan(ref_rem)
                    assert (gn == rn).all()
                    assert np.allclose(rem[~gn], ref_rem[~rn], rtol=1e-9,
                                       atol=0), f"{summary} rev={rev} remaining"


def test_topk_range_k_edge_cases(engine, counter_small):
    ts, vals, offsets = counter_small
    end = START + 239 * STEP
    plan = engine.RollupPlan("avg_over_time", START, end, STEP, window=300_000)
    with engine.SeriesBatch(ts, vals, offsets) as b:
        b.exec(plan, download=False)
        sel, _ = engine.topk_range(b, 0)
        assert len(sel) == 0
        sel, _ = engine.topk_range(b, math.nan)
        assert len(sel) == 0
        sel, _ = engine.topk_range(b, 10**9)
        assert len(sel) == len(offsets) - 1


def test_topk_pointwise_gpu(engine):
    from victoriametrics_amd import synth
    ts, vals, offsets = synth.gauge_batch(3000, 120, START, seed=77)
    end = START + 119 * STEP
    plan = engine.RollupPlan("avg_over_time", START, end, STEP, window=45_000)
    with engine.SeriesBatch(ts, vals, offsets) as b:
        host_out, _, _ = b.exec(plan)
        got = engine.topk_pointwise(b, 10)
    ref = oracle.topk_pointwise(host_out, 10)
    gn, rn = np.isnan(got), np.isnan(ref)
    assert (gn == rn).all(), \
        f"NaN placement differs at {np.argwhere(gn != rn)[:5]}"
    assert np.array_equal(got[~gn], ref[~rn])


def test_bottomk_pointwise_gpu(engine):
    from victoriametrics_amd import synth
    ts, vals, offsets = synth.gauge_batch(1000, 60, START, seed=78)
    end = START + 59 * STEP
    plan = engine.RollupPlan("last_over_time", START, end, STEP, window=45_000)
    with engine.SeriesBatch(ts, vals, offsets) as b:
        host_out, _, _ = b.exec(plan)
        got = engine.topk_pointwise(b, 7, reverse=True)
    ref = oracle.topk_pointwise(host_out, 7, reverse=True)
    gn, rn = np.isnan(got), np.isnan(ref)
    assert (gn == rn).all()
    assert np.array_equal(got[~gn], ref[~rn])


def test_histogram_quantile_gpu(engine):
    rng = np.random.default_rng(12)
    n_groups, n_les, n_grid = 50, 20, 100
    les = np.concatenate([np.sort(rng.random(n_les - 1) * 100),
                          [np.inf]])
    rows = []
    le_col = []
    offs = [0]
    for gid in range(n_groups):
        counts = np.cumsum(rng.random((n_les, n_grid)) * 10, axis=0)
        # inject broken buckets + NaNs + zero columns
        counts[rng.integers(0, n_les), rng.integers(0, n_grid, 5)] = np.nan
        if gid % 7 == 0:
            counts[:, 0] = 0.0
        rows.append(counts)
        le_col.extend(les)
        offs.append(offs[-1] + n_les)
    bv = np.vstack(rows)
    for phi in (0.5, 0.99, 0.0, 1.0, -0.5, 1.5, math.nan):
        got, glo, ghi = engine.histogram_quantile(phi, bv, le_col, offs,
                                                  bounds=True)
        ref, rlo, rhi = oracle.histogram_quantile(phi, bv, le_col, offs,
                                                  bounds=True)
        for g, r, name in ((got, ref, "q"), (glo, rlo, "lo"), (ghi, rhi, "hi")):
            gn, rn = np.isnan(g), np.isnan(r)
            assert (gn == rn).all(), f"phi={phi} {name}"
            assert np.array_equal(g[~gn], r[~rn]), f"phi={phi} {name}"


def test_subquery_rollup(engine):
    """rate(m[5m])[1h:15s]-style subquery (§3e): the same kernels applied to
    a resident inner grid, vs the oracle doing the identical orchestration."""
    rng = np.random.default_rng(31)
    n_series = 300

    def inner_eval(sq_start, sq_end, sq_step):
        m = (sq_end - sq_start) // sq_step + 1
        vals = np.cumsum(rng.random((n_series, m)) * 10, axis=1)
        vals[rng.random((n_series, m)) < 0.05] = np.nan  # holes
        inner_eval.cache = vals
        return vals

    start = START
    end = START + 60 * STEP
    out, _, scanned = engine.rollup_subquery(
        "rate", start, end, STEP, window=300_000, sq_step=STEP,
        inner_eval=inner_eval)

    # oracle side: same removeNanValues + preFunc + doInternal
    vals = inner_eval.cache
    sq_start = start - (300_000 + STEP + engine.MAX_SILENCE_INTERVAL_MS)
    sq_start, sq_end = engine.align_start_end(sq_start, end + STEP, STEP)
    sq_ts = np.arange(sq_start, sq_end + 1, STEP, dtype=np.int64)
    keep = ~np.isnan(vals)
    offsets = np.zeros(n_series + 1, dtype=np.uint64)
    np.cumsum(keep.sum(axis=1), out=offsets[1:])
    cvals = vals[keep]
    cts = np.broadcast_to(sq_ts, vals.shape)[keep].astype(np.int64)
    rc = oracle.make_config("rate", start, end, STEP, window=300_000)
    ref, _, ref_scanned = oracle.rollup_eval_batch(
        rc, cts, cvals, offsets, remove_counter_resets=True,
        drop_stale_nans=False)
    assert scanned == ref_scanned
    assert_parity(out, ref, exact=True, context="subquery")


def test_relayout_per_series_output_order():
    """Grouped batches are physically relayouted by group id; per-series
    outputs must still come back in the ORIGINAL series order (perm
    unapplied on download)."""
    from victoriametrics_amd import synth
    from victoriametrics_amd.engine import RollupPlan, SeriesBatch
    n_series, n_samples = 300, 100
    ts, vals, offsets = synth.counter_batch(n_series, n_samples,
                                            1_000_000_000_000)
    start = int(ts[0]) + 120_000
    plan = RollupPlan("rate", start, start + 20 * 15_000, 15_000,
                      window=300_000)
    # interleaved group ids force a non-identity relayout
    gids = (np.arange(n_series) % 7).astype(np.int32)
    grouped = SeriesBatch(ts, vals, offsets, group_ids=gids, n_groups=7)
    assert grouped._perm is not None
    out_g, _, _ = grouped.exec(plan)
    grouped.close()
    plain = SeriesBatch(ts, vals, offsets)
    out_p, _, _ = plain.exec(plan)
    plain.close()
    np.testing.assert_array_equal(out_g.view(np.int64), out_p.view(np.int64))


def test_relayout_grouped_matches_unrelayouted_semantics():
    """sum by group on an interleaved-gid batch equals the host-computed
    group sums of the per-series results."""
    from victoriametrics_amd import synth
    from victoriametrics_amd.engine import RollupPlan, SeriesBatch
    n_series, n_samples = 256, 120
    ts, vals, offsets = synth.counter_batch(n_series, n_samples,
                                            1_000_000_000_000)
    start = int(ts[0]) + 120_000
    plan_g = RollupPlan("rate", start, start + 30 * 15_000, 15_000,
                        window=300_000, aggr="sum")
    plan_p = RollupPlan("rate", start, start + 30 * 15_000, 15_000,
                        window=300_000)
    gids = ((np.arange(n_series) * 13) % 9).astype(np.int32)
    grouped = SeriesBatch(ts, vals, offsets, group_ids=gids, n_groups=9)
    out_g, counts, _ = grouped.exec(plan_g)
    grouped.close()
    plain = SeriesBatch(ts, vals, offsets)
    per, _, _ = plain.exec(plan_p)
    plain.close()
    for g in range(9):
        member = per[gids == g]
        exp = np.where(np.all(np.isnan(member), axis=0), np.nan,
                       np.nansum(member, axis=0))
        np.testing.assert_allclose(out_g[g], exp, rtol=1e-9, equal_nan=True)


@pytest.mark.parametrize("parent", ["rollup", "rollup_rate", "rollup_deriv",
                                    "rollup_increase", "rollup_delta",
                                    "rollup_scrape_interval",
                                    "rollup_candlestick"])
def test_rollup_fake_expansions(engine, parent):
    """getRollupConfigs' rollup_* pseudo-function expansions
    (rollup.go:436-516): preFunc value transform + min/max/avg (or
    candlestick) sub-configs, each compared to the oracle with the same
    preFunc."""
    ts, vals, offsets = ragged_batch(40, 300, START, dup_p=0.03)
    start = START + 120_000
    end = start + 30 * STEP
    plans = engine.rollup_fake_plans(parent, start, end, STEP,
                                     window=300_000)
    assert len(plans) == (4 if parent == "rollup_candlestick" else 3)
    for tag, plan in plans:
        out, _, scanned = engine.rollup_eval(plan, ts, vals, offsets)
        ref, _, ref_scanned = _oracle_batch(plan, ts, vals, offsets)
        assert scanned == ref_scanned, f"{parent}/{tag} samplesScanned"
        assert_parity(out, ref, exact=True, context=f"{parent}/{tag}")


def test_aggr_over_time_expansion(engine):
    ts, vals, offsets = ragged_batch(24, 200, START)
    start = START + 120_000
    end = start + 20 * STEP
    plans = engine.aggr_over_time_plans(
        ["min_over_time", "max_over_time", "sum_over_time",
         "count_over_time"], start, end, STEP, window=300_000)
    for tag, plan in plans:
        out, _, _ = engine.rollup_eval(plan, ts, vals, offsets)
        ref, _, _ = _oracle_batch(plan, ts, vals, offsets)
        assert_parity(out, ref, exact=True, context=f"aggr_over_time/{tag}")


def test_topk_median_device_vs_oracle(engine):
    """Device median selection (8-pass byte-histogram rank selection in
    topk_summary_kernel) vs the oracle's quantile(0.5): exact equality of
    the selected summary values, including NaN rows, duplicates and
    even/odd non-NaN counts."""
    rng = np.random.default_rng(42)
    n_series, n = 500, 97
    vals = rng.standard_normal((n_series, n)) * 100
    # duplicates + NaN holes + all-NaN rows + constant rows
    vals[rng.random((n_series, n)) < 0.15] = np.nan
    vals[7, :] = np.nan
    vals[11, :] = 3.25
    vals[13, : n // 2] = -1.5
    ts = np.tile(START + np.arange(n, dtype=np.int64) * STEP, n_series)
    offsets = np.arange(n_series + 1, dtype=np.uint64) * n
    plan = engine.RollupPlan("last_over_time", START,
                             START + (n - 1) * STEP, STEP, window=STEP)
    with engine.SeriesBatch(ts.copy(), vals.reshape(-1).copy(), offsets) as b:
        host_out, _, _ = b.exec(plan)
        for rev in (False, True):
            sel, _ = engine.topk_range(b, 50, summary="median", reverse=rev)
            ref_sel, _ = oracle.topk_range(host_out, 50, "median",
                                           reverse=rev)
            sv = [oracle.topk_summary("median", host_out[i]) for i in sel]
            rv = [oracle.topk_summary("median", host_out[i])
                  for i in ref_sel]
            assert [repr(x) for x in sv] == [repr(x) for x in rv], \
                f"rev={rev}"
