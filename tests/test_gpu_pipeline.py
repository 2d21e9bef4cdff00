"""Full-stack integration: one flow through every layer — compressed
blocks -> fused resident batch -> grouped rollup -> histogram pipeline ->
binary op -> aggregates -> topk -> cache fill — with querytracer spans,
asserting cross-layer consistency (each stage against its oracle or an
independently computed expectation)."""
import json
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd import cache, engine, tracer
from victoriametrics_amd import transform as tfm
from victoriametrics_amd import aggregate as agg
from victoriametrics_amd import binary_op as bop
from victoriametrics_amd.binary_op import BinOpSpec, Series
from victoriametrics_amd.engine import RollupPlan, SeriesBatch
from victoriametrics_amd.metric_name import MetricName

from test_gpu_decode import _make_block

pytestmark = pytest.mark.gpu

START = 1_000_000_000_000


def test_full_query_pipeline():
    qt = tracer.Tracer.new(True, "histogram_quantile(0.9, sum by(le)"
                                 "(rate(bucket[5m]))) pipeline")
    rng = np.random.default_rng(99)
    n_hist, n_le, rows = 24, 8, 120
    n_series = n_hist * n_le

    # 1. encoded storage blocks (oracle encoder = the write path)
    blocks, sbs = [], [0]
    names = []
    for h in range(n_hist):
        for b in range(n_le):
            blk, rt, rv = _make_block(rng, "counter", rows)
            blocks.append(blk)
            sbs.append(len(blocks))
            names.append((f"h{h}", b))
    sbs = np.asarray(sbs, np.uint32)

    # 2. fused cold-cache fetch -> resident batch, grouped by (hist, le)
    gids = np.asarray([h * n_le + le for h, le in
                       [(i // n_le, i % n_le) for i in range(n_series)]],
                      np.int32)
    c = tracer.new_child(qt, "fetch %d blocks", len(blocks))
    batch = SeriesBatch.from_blocks(blocks, sbs, group_ids=gids,
                                    n_groups=n_series)
    tracer.donef(c, "resident, %d samples", int(batch.offsets[-1]))

    # 3. grouped rollup: rate + sum by (hist, le)
    t0 = START  # blocks all start at the same base
    lo_ts = int(batch.offsets[0])
    start = 1_000_000_000_000 + 600_000
    plan = RollupPlan("rate", start, start + 40 * 15_000, 15_000,
                      window=300_000, aggr="sum")
    out, counts, scanned = batch.exec(plan, tracer=qt)
    n_grid = plan.n_grid
    assert out.shape == (n_series, n_grid)
    assert scanned > 0

    # 4. histogram_quantile over the le-grouped matrix (host pipeline)
    series = []
    for i, (hname, le_idx) in enumerate(names):
        series.append(Series(
            MetricName("req", [("hist", hname), ("le", str(2.0 ** le_idx))]),
            out[i]))
    hq = tfm.histogram_transform("histogram_quantile",
                                 [s.copy_shallow() for s in series], arg=0.9)
    assert len(hq) == n_hist
    # verify one group against the oracle walk
    g0 = [s for s in series if s.mn.get_tag_value("hist") == b"h0"]
    g0.sort(key=lambda s: float(s.mn.get_tag_value("le")))
    bv = np.stack([s.values for s in g0])
    les = np.asarray([float(s.mn.get_tag_value("le")) for s in g0])
    exp, _, _ = oracle.histogram_quantile(
        0.9, bv, les, np.asarray([0, n_le], np.uint64))
    got0 = [s for s in hq if s.mn.get_tag_value("hist") == b"h0"][0]
    np.testing.assert_array_equal(got0.values.view(np.int64),
                                  exp[0].view(np.int64))

    # 5. binary op: share of each histogram's p90 vs the fleet max
    fleet_max = agg.aggregate("max", [s.copy_shallow() for s in hq])
    assert len(fleet_max) == 1
    ratio = bop.binary_op_eval(
        BinOpSpec("/", group_op="on", group_tags=[], join_op="group_left"),
        [s.copy_shallow() for s in hq], fleet_max)
    assert len(ratio) == n_hist
    finite = np.concatenate([r.values for r in ratio])
    finite = finite[~np.isnan(finite)]
    assert (finite <= 1.0 + 1e-12).all()

    # 6. aggregates over the quantile series
    med = agg.aggregate("median", [s.copy_shallow() for s in hq])
    assert len(med) == 1 and med[0].values.shape == (n_grid,)

    # 7. topk over the original grouped rollup
    sel, _ = engine.topk_range(batch, 5, summary="avg")
    assert len(sel) == 5

    # 8. cache fill straight off the device + roundtrip
    grid_ts = plan.timestamps()
    cnames = [(b"req", ((b"g", str(i).encode()),)) for i in range(n_series)]
    blob = cache.marshal_from_batch(batch, cnames, grid_ts)
    rnames, rvals, rts = cache.unmarshal_timeseries_fast(blob)[:3]
    np.testing.assert_array_equal(np.asarray(rts), grid_ts)
    np.testing.assert_array_equal(np.asarray(rvals).view(np.int64),
                                  out.view(np.int64))

    batch.close()
    qt.done()
    tree = json.loads(qt.to_json())
    assert tree["children"], tree


def test_instant_rollup_optimization_on_device():
    """InstantRollupEvaluator over the device path: eval_at = one-point
    SeriesBatch.exec; the composed cached+offset result must equal a
    direct full-window device evaluation bit-for-bit for the
    sum-decomposable funcs and max/min."""
    from victoriametrics_amd import instant, synth
    from victoriametrics_amd.cache import RollupResultCache

    n_series, n_samples = 64, 720  # 1h of 5s samples
    step_ms = 5_000
    t0 = 1_600_000_000_000
    # reset-free counters: the increase composition identity
    # inc(w@t-off) + inc(off@t) - inc(off@t-w) = inc(w@t) holds exactly
    # only when no reset falls near a sub-window baseline (the reference
    # accepts the same edge inaccuracy for cached increase, issue 10098)
    ts, vals, offsets = synth.counter_batch(n_series, n_samples, t0,
                                            step=step_ms, seed=42,
                                            reset_p=0.0)
    batch = SeriesBatch(ts, vals, offsets)
    names = [MetricName(b"m", [(b"s", str(i).encode())])
             for i in range(n_series)]
    window = 3 * 3600 * 1000  # below the batch span? span = 720*5s = 1h
    # use min_window_ms = 30min so the optimization engages on this batch
    min_window = 30 * 60 * 1000
    window = 45 * 60 * 1000

    calls = []

    def eval_at(func, timestamp, w):
        calls.append((func, timestamp, w))
        plan = RollupPlan(func, timestamp, timestamp, step_ms, window=w)
        out, counts, _ = batch.exec(plan)
        from victoriametrics_amd.binary_op import Series
        return [Series(names[i].copy(), out[i].copy())
                for i in range(n_series)
                if not math.isnan(out[i, 0])]

    now = int(ts[-1])
    for func in ("sum_over_time", "count_over_time", "increase",
                 "max_over_time", "min_over_time"):
        ev = instant.InstantRollupEvaluator(
            RollupResultCache(), eval_at, step=step_ms, now_ms=now,
            min_window_ms=min_window)
        t1 = now - 5 * 60 * 1000 - 2_500
        got1 = ev.eval(func, "q", t1, window)
        direct1 = {s.mn.get_tag_value("s"): s.values[0]
                   for s in eval_at(func, t1, window)}
        for s in got1:
            assert s.values[0] == direct1[s.mn.get_tag_value("s")], func
        # second query at a later timestamp: composed from cache + two
        # offset windows, still bit-exact vs the direct device result
        calls.clear()
        t2 = now - 2_500
        got2 = ev.eval(func, "q", t2, window)
        offset_calls = [c for c in calls if c[2] < window]
        assert offset_calls, (func, calls)
        direct2 = {s.mn.get_tag_value("s"): s.values[0]
                   for s in eval_at(func, t2, window)}
        assert len(got2) == len(direct2)
        for s in got2:
            want = direct2[s.mn.get_tag_value("s")]
            g = float(s.values[0])
            if func in ("sum_over_time", "increase"):
                # cached + start - end is algebraically exact but
                # reassociated; allow 1-ulp-scale tolerance
                assert g == pytest.approx(float(want), rel=1e-12), func
            else:
                assert g == float(want), func
    batch.close()
