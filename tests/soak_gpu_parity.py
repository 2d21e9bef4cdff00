"""Randomized GPU-vs-oracle parity soak — not part of the default suite;
run directly on a GPU box with a time budget:

    python tests/soak_gpu_parity.py [seconds] [seed0]

Each iteration draws a random plan (function, window/step/grid shape,
lookback, args, aggregation) and a random ragged batch (lengths spanning
the wave/block kernel classes, stale NaNs, duplicate timestamps, empty
series) and compares the engine against the oracle at the standard parity
bar (bit-exact; pow-based funcs and grouped float reductions at rtol).
Any mismatch prints a REPRO line with the seed."""
import sys
import time

import numpy as np

sys.path.insert(0, ".")
sys.path.insert(0, "tests")

import oracle  # noqa: E402
from victoriametrics_amd import engine  # noqa: E402
from victoriametrics_amd.engine import FUNC_IDS, RollupPlan  # noqa: E402
from seriesgen import ragged_batch, assert_parity  # noqa: E402
from test_gpu_parity import _oracle_batch  # noqa: E402

START = 1_000_000_000_000
POW_FUNCS = {"geomean_over_time", "hoeffding_bound_lower",
             "hoeffding_bound_upper"}  # pow/log: device-libm ulp
ARG_RANGES = {
    "quantile_over_time": (0.0, 1.0), "predict_linear": (-600.0, 600.0),
    "share_le_over_time": (0.0, 100.0), "share_gt_over_time": (0.0, 100.0),
    "count_le_over_time": (0.0, 100.0), "count_gt_over_time": (0.0, 100.0),
    "count_eq_over_time": (0.0, 100.0), "count_ne_over_time": (0.0, 100.0),
    "sum_le_over_time": (0.0, 100.0), "sum_gt_over_time": (0.0, 100.0),
    "sum_eq_over_time": (0.0, 100.0), "duration_over_time": (0.0, 120_000.0),
    "hoeffding_bound_lower": (0.1, 0.99), "hoeffding_bound_upper": (0.1, 0.99),
    "holt_winters": (0.05, 0.95),
}
ARG2_RANGES = {"holt_winters": (0.05, 0.95)}
FUNCS = sorted(set(FUNC_IDS) - {"increase_prometheus", "timestamp",
                                "timestamp_with_name"})
AGGRS = ["none"] * 4 + ["sum", "min", "max", "avg", "count", "sum2",
                        "geomean", "group"]


def one_iter(seed):
    rng = np.random.default_rng(seed)
    func = FUNCS[int(rng.integers(0, len(FUNCS)))]
    n_series = int(rng.integers(1, 80))
    # lengths spanning wave (<=256/512), block (<=4064) AND huge (>4064)
    max_len = int(rng.choice([40, 200, 520, 700, 700, 4500]))
    ts, vals, offsets = ragged_batch(
        n_series, max_len, START, seed=seed,
        stale_p=float(rng.choice([0.0, 0.02, 0.3])),
        dup_p=float(rng.choice([0.0, 0.05])))
    step = int(rng.choice([5_000, 15_000, 60_000]))
    n_grid = int(rng.integers(1, 100))
    start = START + int(rng.integers(-2, 60)) * step
    end = start + (n_grid - 1) * step
    window = int(rng.choice([0, step, 2 * step, 20 * step, 300_000,
                             7 * step, 100 * step]))  # 100*step: dg >= 64,
    # the pipe kernel's jbuf path instead of the fused shuffle-seek
    lookback = int(rng.choice([0, 0, 5 * 60 * 1000, 2 * step]))
    aggr = AGGRS[int(rng.integers(0, len(AGGRS)))]
    lo, hi = ARG_RANGES.get(func, (0.0, 0.0))
    arg = float(rng.uniform(lo, hi))
    lo2, hi2 = ARG2_RANGES.get(func, (0.0, 0.0))
    arg2 = float(rng.uniform(lo2, hi2))
    n_groups, gids = 0, None
    if aggr != "none":
        n_groups = int(rng.integers(1, max(2, n_series)))
        gids = rng.integers(-1, n_groups, n_series).astype(np.int32)
    # 1-in-8 iterations exercise the rollup_* preFunc expansions instead
    if rng.integers(0, 8) == 0 and aggr == "none":
        parent = ["rollup", "rollup_rate", "rollup_increase",
                  "rollup_scrape_interval",
                  "rollup_candlestick"][int(rng.integers(0, 5))]
        plans = engine.rollup_fake_plans(parent, start, end, step,
                                         window=window,
                                         lookback_delta=lookback)
        tag, plan = plans[int(rng.integers(0, len(plans)))]
        func = f"{parent}/{tag}"
    else:
        plan = RollupPlan(func, start, end, step, window=window,
                          lookback_delta=lookback, arg=arg, arg2=arg2,
                          aggr=aggr, skip_finalize=False)
    out, counts, scanned = engine.rollup_eval(plan, ts, vals, offsets,
                                              group_ids=gids,
                                              n_groups=n_groups)
    ref, ref_counts, ref_scanned = _oracle_batch(
        plan, ts, vals, offsets, group_ids=gids, n_groups=n_groups,
        aggr=aggr, n_threads=2)
    ctx = f"seed={seed} func={func} aggr={aggr} grid={n_grid} win={window}"
    assert scanned == ref_scanned, \
        f"{ctx}: samplesScanned {scanned} != {ref_scanned}"
    if aggr == "none":
        assert_parity(out, ref, exact=func not in POW_FUNCS, context=ctx)
    else:
        assert_parity(out, ref, exact=False, rtol=1e-9, atol=1e-11,
                      context=ctx)


def main():
    budget_s = float(sys.argv[1]) if len(sys.argv) > 1 else 120.0
    seed0 = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    engine.init()
    t_end = time.time() + budget_s
    i = 0
    fails = 0
    while time.time() < t_end:
        seed = seed0 + i
        try:
            one_iter(seed)
        except AssertionError as e:
            fails += 1
            print(f"REPRO: python tests/soak_gpu_parity.py 1 {seed}")
            print(f"  {str(e)[:400]}")
            if fails >= 5:
                break
        except Exception as e:  # noqa: BLE001
            fails += 1
            print(f"ERROR at seed {seed}: {type(e).__name__}: {str(e)[:300]}")
            if fails >= 5:
                break
        i += 1
    print(f"soak: {i} iterations, {fails} failures")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
