"""Fused cold-cache fetch path (§8f(1)): compressed blocks -> device decode
-> merge+dedup -> resident batch -> rollup, with the decoded columns never
crossing PCIe.  Parity: the fused batch's rollup output must be bit-equal
to the staged path (decode to host + oracle merge + plain batch upload)."""
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd import engine
from victoriametrics_amd.engine import RollupPlan, SeriesBatch

from test_gpu_decode import _make_block

pytestmark = pytest.mark.gpu

START = 1_000_000_000_000


def _series_blocks(rng, n_series, blocks_per_series, rows, kind="counter"):
    """Build per-series disjoint block chains + the oracle-decoded truth."""
    blocks, sbs, truth = [], [0], []
    for s in range(n_series):
        ts_all, va_all = [], []
        t0 = START + int(rng.integers(0, 10_000))
        for b in range(blocks_per_series):
            blk, rt, rv = _make_block(rng, kind, rows)
            # shift this block's timestamps after the previous one
            shift = t0 - int(rt[0])
            rt = rt + shift
            blk = dict(blk)
            blk["min_timestamp"] = int(rt[0])
            blk["max_timestamp"] = int(rt[-1])
            # recompute ts stream with the shifted start: re-marshal
            tdata, tmt, tfirst = oracle.marshal_int64_array(rt, 64)
            if tmt in (oracle.MT_ZSTD_NEAREST_DELTA,
                       oracle.MT_ZSTD_NEAREST_DELTA2):
                tdata = oracle.zstd_decompress(tdata)
                tmt = oracle.MT_NEAREST_DELTA if \
                    tmt == oracle.MT_ZSTD_NEAREST_DELTA else \
                    oracle.MT_NEAREST_DELTA2
            blk["ts_data"], blk["ts_mt"] = tdata, tmt
            blocks.append(blk)
            ts_all.append(rt)
            va_all.append(rv)
            t0 = int(rt[-1]) + int(rng.integers(1_000, 30_000))
        sbs.append(len(blocks))
        truth.append((np.concatenate(ts_all), np.concatenate(va_all)))
    return blocks, np.asarray(sbs, np.uint32), truth


def test_fused_batch_matches_staged_path():
    rng = np.random.default_rng(31)
    n_series, bps, rows = 128, 3, 80
    blocks, sbs, truth = _series_blocks(rng, n_series, bps, rows)

    batch = SeriesBatch.from_blocks(blocks, sbs)
    # merged offsets = concatenation (disjoint chains, no dedup)
    exp_offsets = np.zeros(n_series + 1, np.uint64)
    for i, (t, _) in enumerate(truth):
        exp_offsets[i + 1] = exp_offsets[i] + len(t)
    np.testing.assert_array_equal(batch.offsets, exp_offsets)

    start = START + 120_000
    plan = RollupPlan("rate", start, start + 40 * 15_000, 15_000,
                      window=300_000)
    got, _, scanned = batch.exec(plan)
    batch.close()

    ts_cat = np.concatenate([t for t, _ in truth])
    vals_cat = np.concatenate([v for _, v in truth])
    staged = SeriesBatch(ts_cat, vals_cat, exp_offsets)
    exp, _, exp_scanned = staged.exec(plan)
    staged.close()

    np.testing.assert_array_equal(got.view(np.int64), exp.view(np.int64))
    assert scanned == exp_scanned


def test_fused_batch_dedup_replicated_blocks():
    # replicated blocks (two storage nodes) collapse under dedup
    rng = np.random.default_rng(32)
    blocks, sbs, truth = _series_blocks(rng, 32, 2, 60)
    # duplicate every series' first block at the end of its chain
    dup_blocks, dup_sbs = [], [0]
    for s in range(32):
        lo, hi = sbs[s], sbs[s + 1]
        chain = list(blocks[lo:hi]) + [blocks[lo]]
        dup_blocks.extend(chain)
        dup_sbs.append(len(dup_blocks))
    batch = SeriesBatch.from_blocks(dup_blocks, np.asarray(dup_sbs, np.uint32),
                                    dedup_interval=1)
    # expected: oracle merge over the duplicated chains
    exp_counts = []
    for s in range(32):
        t, v = truth[s]
        lo = sbs[s]
        blk0 = blocks[lo]
        t0 = oracle.unmarshal_int64_array(blk0["ts_data"], blk0["rows"],
                                          blk0["ts_mt"], 0)
        # merge original chain + replica of first block
        n0 = blk0["rows"]
        mt, mv = oracle.merge_sort_blocks(
            [(t[:n0], v[:n0]), (t[n0:], v[n0:]), (t[:n0], v[:n0])],
            dedup_interval=1)
        exp_counts.append(len(mt))
    got_counts = np.diff(batch.offsets.astype(np.int64))
    np.testing.assert_array_equal(got_counts, exp_counts)
    batch.close()


def test_fused_batch_grouped_rollup():
    rng = np.random.default_rng(33)
    n_series = 64
    blocks, sbs, truth = _series_blocks(rng, n_series, 2, 50)
    gids = (np.arange(n_series) % 8).astype(np.int32)
    batch = SeriesBatch.from_blocks(blocks, sbs, group_ids=gids, n_groups=8)
    start = START + 100_000
    plan = RollupPlan("rate", start, start + 20 * 15_000, 15_000,
                      window=300_000, aggr="sum")
    got, counts, _ = batch.exec(plan)
    batch.close()

    exp_offsets = np.zeros(n_series + 1, np.uint64)
    for i, (t, _) in enumerate(truth):
        exp_offsets[i + 1] = exp_offsets[i] + len(t)
    staged = SeriesBatch(np.concatenate([t for t, _ in truth]),
                         np.concatenate([v for _, v in truth]),
                         exp_offsets, group_ids=gids, n_groups=8)
    exp, exp_counts, _ = staged.exec(plan)
    staged.close()
    np.testing.assert_allclose(got, exp, rtol=1e-9)
    np.testing.assert_array_equal(counts, exp_counts)


def test_marshal_from_batch_matches_host_marshal():
    """§8f(2): the cache fill downloaded straight off the device must be
    byte-identical to the host-side marshalTimeseriesFast."""
    from victoriametrics_amd import cache, synth
    rng = np.random.default_rng(41)
    n_series = 96
    ts, vals, offsets = synth.counter_batch(n_series, 100, START)
    start = START + 120_000
    plan = RollupPlan("rate", start, start + 30 * 15_000, 15_000,
                      window=300_000)
    batch = SeriesBatch(ts, vals, offsets)
    out, _, _ = batch.exec(plan)
    names = [(b"m", ((b"pod", b"p%d" % i),)) for i in range(n_series)]
    grid_ts = plan.timestamps()
    expected = cache.marshal_timeseries_fast(names, out, grid_ts)
    got = cache.marshal_from_batch(batch, names, grid_ts)
    batch.close()
    assert got == expected


def test_packed_stream_matches_direct_batch():
    """Native descriptor path (vmgpu_batch_create_packed): oracle-packed
    stream -> C parse -> fused decode/merge -> rollup must be bit-equal to
    a batch built from the raw decoded columns."""
    rng = np.random.default_rng(77)
    n_series, rows = 257, 240
    step = 15_000
    ts_parts, vi_parts, offs = [], [], [0]
    for s in range(n_series):
        n = rows if s % 7 else int(rng.integers(1, 50))
        t = START + np.cumsum(rng.integers(step - 500, step + 501, n)).astype(np.int64)
        v = np.cumsum(rng.integers(0, 500, n)).astype(np.int64)
        ts_parts.append(t)
        vi_parts.append(v)
        offs.append(offs[-1] + n)
    ts = np.concatenate(ts_parts)
    vi = np.concatenate(vi_parts)
    offsets = np.asarray(offs, dtype=np.uint64)

    packed, n_blocks, sbs = oracle.pack_blocks(ts, vi, offsets)
    engine.init()
    b_packed = SeriesBatch.from_packed(packed, n_blocks, sbs)
    np.testing.assert_array_equal(b_packed.offsets, offsets)
    b_direct = SeriesBatch(ts, vi.astype(np.float64), offsets)

    start = START + 600_000
    plan = RollupPlan("rate", start, start + 100 * step, step, window=300_000)
    out_p, _, sc_p = b_packed.exec(plan)
    out_d, _, sc_d = b_direct.exec(plan)
    assert sc_p == sc_d
    np.testing.assert_array_equal(out_p.view(np.int64), out_d.view(np.int64))
    b_packed.close()
    b_direct.close()
