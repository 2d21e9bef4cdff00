"""Ad-hoc single-GPU measurements of the BASELINE config-4 and config-5
pipeline shapes (the driver's bench contract covers config 2; these are
evidence lines for DESIGN.md §3b).

  config 4: histogram_quantile(0.99, sum by(le)(rate(bucket[5m]))),
            100k bucket series, 24h@30s grid (2880 points) — full chain on
            one GPU (the real config shards 4-way by seriesID).
  config 5: avg_over_time(m[5m]) + topk(100) over the 1-GPU shard of the
            8-way 10M-series config: 1.25M series x 1440 samples (6h@15s).

Run: python tests/bench_configs.py [config4|config5]
"""
import sys
import time

import numpy as np

sys.path.insert(0, ".")
from victoriametrics_amd import engine  # noqa: E402
from victoriametrics_amd.engine import RollupPlan, SeriesBatch  # noqa: E402

START = 1_000_000_000_000


def synth_counters(n_series, n_samples, step_ms, seed=8428):
    rng = np.random.default_rng(seed)
    total = n_series * n_samples
    base = np.arange(n_samples, dtype=np.int64) * step_ms
    ts = np.empty(total, np.int64)
    vals = np.empty(total, np.float64)
    CH = 4096
    for s0 in range(0, n_series, CH):
        s1 = min(s0 + CH, n_series)
        k = s1 - s0
        jit = rng.integers(-step_ms // 30, step_ms // 30, (k, n_samples))
        t = START + base[None, :] + jit
        t.sort(axis=1)
        v = np.cumsum(rng.poisson(10.0 * step_ms / 1000.0,
                                  (k, n_samples)).astype(np.float64), axis=1)
        ts[s0 * n_samples:s1 * n_samples] = t.ravel()
        vals[s0 * n_samples:s1 * n_samples] = v.ravel()
    offsets = np.arange(n_series + 1, dtype=np.uint64) * n_samples
    return ts, vals, offsets


def config4():
    n_series, n_samples, step = 100_000, 2880, 30_000
    n_hist, n_le = 2_500, 40  # 2500 histograms x 40 le buckets
    print(f"config4: {n_series} bucket series x {n_samples} samples "
          f"(24h@30s), {n_hist} histograms x {n_le} le buckets")
    ts, vals, offsets = synth_counters(n_series, n_samples, step)
    # group id = (histogram, le) pair: sum by(le) within each histogram
    gids = np.arange(n_series, dtype=np.int32) % (n_hist * n_le)
    t0 = time.time()
    batch = SeriesBatch(ts, vals, offsets, group_ids=gids,
                        n_groups=n_hist * n_le)
    print(f"batch upload: {time.time() - t0:.2f}s "
          f"({n_series * n_samples * 16 / 1e9:.1f} GB)")
    start = START + 600_000
    end = START + (n_samples - 1) * step
    plan = RollupPlan("rate", start, end, step, window=300_000, aggr="sum")
    for _ in range(2):
        out, counts, _ = batch.exec(plan)
    t0 = time.time()
    out, counts, _ = batch.exec(plan)
    rollup_wall = time.time() - t0
    rollup_ms = engine.last_kernel_ms()
    n_grid = plan.n_grid
    print(f"rate+sum by(le): kernel {rollup_ms:.2f} ms, wall "
          f"{rollup_wall * 1e3:.0f} ms, out {n_hist * n_le}x{n_grid}")
    # isolation: ungrouped and coarse-group variants of the same rollup
    plain = SeriesBatch(ts, vals, offsets)
    p2 = RollupPlan("rate", start, end, step, window=300_000)
    for _ in range(2):
        plain.exec(p2, download=False)
    plain.exec(p2, download=False)
    print(f"rate ungrouped (block-class): kernel "
          f"{engine.last_kernel_ms():.2f} ms")
    plain.close()
    # hq over le groups: les ascending per histogram
    les = np.tile(np.geomspace(0.001, 10.0, n_le), n_hist)
    goff = np.arange(n_hist + 1, dtype=np.uint64) * n_le
    t0 = time.time()
    q, _, _ = engine.histogram_quantile(0.99, out, les, goff)
    hq_wall = time.time() - t0
    samples = n_series * n_samples
    print(f"histogram_quantile(0.99): wall {hq_wall * 1e3:.0f} ms, "
          f"out {n_hist}x{n_grid}")
    print(f"pipeline samples/s (kernel): "
          f"{samples / (rollup_ms / 1e3) / 1e9:.1f} G")


def config5():
    n_series, n_samples, step = 1_250_000, 1440, 15_000
    print(f"config5 1-GPU shard: {n_series} series x {n_samples} samples "
          f"(6h@15s) = {n_series * n_samples * 16 / 1e9:.0f} GB")
    ts, vals, offsets = synth_counters(n_series, n_samples, step)
    t0 = time.time()
    batch = SeriesBatch(ts, vals, offsets)
    print(f"batch upload: {time.time() - t0:.2f}s")
    start = START + 600_000
    end = START + (n_samples - 1) * step
    plan = RollupPlan("avg_over_time", start, end, step, window=300_000)
    for _ in range(2):
        batch.exec(plan, download=False)
    t0 = time.time()
    _, _, scanned = batch.exec(plan, download=False)
    wall = time.time() - t0
    k_ms = engine.last_kernel_ms()
    samples = n_series * n_samples
    print(f"avg_over_time: kernel {k_ms:.2f} ms -> "
          f"{samples / (k_ms / 1e3) / 1e9:.1f} Gsamples/s, "
          f"wall {wall * 1e3:.0f} ms")
    t0 = time.time()
    res = engine.topk_range(batch, 100, summary="avg")
    topk_wall = time.time() - t0
    sel = res[0] if isinstance(res, tuple) else res
    print(f"topk(100, avg): wall {topk_wall * 1e3:.0f} ms, "
          f"{len(sel)} selected")


def config_huge():
    """Huge-kernel class isolation: 50k series x 20k samples (3.5 days
    @15s merged into one logical series) = 16 GB."""
    n_series, n_samples, step = 50_000, 20_000, 15_000
    print(f"huge-class: {n_series} series x {n_samples} samples")
    ts, vals, offsets = synth_counters(n_series, n_samples, step)
    batch = SeriesBatch(ts, vals, offsets)
    start = START + 600_000
    end = START + (n_samples - 1) * step
    plan = RollupPlan("rate", start, end, step, window=300_000)
    for _ in range(2):
        batch.exec(plan, download=False)
    batch.exec(plan, download=False)
    k_ms = engine.last_kernel_ms()
    samples = n_series * n_samples
    print(f"rate huge-class: kernel {k_ms:.2f} ms -> "
          f"{samples / (k_ms / 1e3) / 1e9:.1f} Gsamples/s")
    batch.close()


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "config4"
    if which == "config4":
        config4()
    elif which == "huge":
        config_huge()
    else:
        config5()
