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
