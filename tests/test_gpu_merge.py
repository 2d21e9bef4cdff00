"""GPU mergeSortBlocks + DeduplicateSamples parity vs the oracle
(netstorage.go:564, dedup.go:29).  Covers the wave-parallel concatenation
fast path (disjoint LSM-part blocks), the exact heap-merge path (overlapping
blocks, duplicate timestamps, replicated equal prefixes) and dedup with
StaleNaN preference rules."""
import math

import numpy as np
import pytest

import oracle
from victoriametrics_amd import engine

pytestmark = pytest.mark.gpu

STALE = oracle.stale_nan()


def _run_both(series_blocks, dedup_interval=0):
    """series_blocks: list (per series) of lists of (ts, vals) blocks."""
    ts_all, val_all, boff, sbs = [], [], [0], [0]
    for blocks in series_blocks:
        for t, v in blocks:
            ts_all.append(np.asarray(t, dtype=np.int64))
            val_all.append(np.asarray(v, dtype=np.float64))
            boff.append(boff[-1] + len(t))
        sbs.append(sbs[-1] + len(blocks))
    ts = np.concatenate(ts_all) if ts_all else np.empty(0, np.int64)
    vals = np.concatenate(val_all) if val_all else np.empty(0, np.float64)
    g_ts, g_vals, g_off = engine.merge_blocks(ts, vals, boff, sbs,
                                              dedup_interval)
    # oracle per series
    exp_ts, exp_vals, exp_off = [], [], [0]
    for blocks in series_blocks:
        et, ev = oracle.merge_sort_blocks(
            [(np.asarray(t, np.int64), np.asarray(v, np.float64))
             for t, v in blocks], dedup_interval)
        exp_ts.append(et)
        exp_vals.append(ev)
        exp_off.append(exp_off[-1] + len(et))
    return (g_ts, g_vals, g_off,
            np.concatenate(exp_ts) if exp_ts else np.empty(0, np.int64),
            np.concatenate(exp_vals) if exp_vals else np.empty(0, np.float64),
            exp_off)


def _assert_equal(got_ts, got_vals, got_off, exp_ts, exp_vals, exp_off):
    assert list(got_off) == list(exp_off)
    np.testing.assert_array_equal(got_ts, exp_ts)
    np.testing.assert_array_equal(got_vals.view(np.int64),
                                  np.asarray(exp_vals).view(np.int64))


def _blocks_disjoint(rng, n_blocks, rows):
    out, t0 = [], 1_000_000_000_000
    for _ in range(n_blocks):
        ts = t0 + np.cumsum(rng.integers(1_000, 20_000, rows))
        out.append((ts.astype(np.int64), rng.standard_normal(rows) * 100))
        t0 = int(ts[-1]) + int(rng.integers(1, 30_000))
    return out


def _blocks_overlapping(rng, n_blocks, rows):
    out = []
    for _ in range(n_blocks):
        t0 = 1_000_000_000_000 + int(rng.integers(0, 50_000))
        ts = t0 + np.cumsum(rng.integers(0, 15_000, rows))  # step 0 => dup ts
        out.append((ts.astype(np.int64), rng.standard_normal(rows) * 100))
    return out


def test_merge_disjoint_many_series():
    rng = np.random.default_rng(1)
    series = [_blocks_disjoint(rng, int(rng.integers(1, 6)),
                               int(rng.integers(1, 400)))
              for _ in range(257)]
    _assert_equal(*_run_both(series))


def test_merge_unordered_disjoint_blocks():
    # blocks arrive in arbitrary order; merge must sort by first timestamp
    rng = np.random.default_rng(2)
    series = []
    for _ in range(64):
        blocks = _blocks_disjoint(rng, 5, 100)
        rng.shuffle(blocks)
        series.append(blocks)
    _assert_equal(*_run_both(series))


def test_merge_overlapping_blocks():
    rng = np.random.default_rng(3)
    series = [_blocks_overlapping(rng, int(rng.integers(2, 7)),
                                  int(rng.integers(2, 200)))
              for _ in range(96)]
    _assert_equal(*_run_both(series))


def test_merge_mixed_disjoint_and_overlapping():
    rng = np.random.default_rng(4)
    series = []
    for i in range(128):
        if i % 3 == 0:
            series.append(_blocks_overlapping(rng, 3, 64))
        else:
            series.append(_blocks_disjoint(rng, 4, 64))
    _assert_equal(*_run_both(series))


def test_merge_replicated_blocks_dedup():
    # vmstorage replication: identical blocks from two storage nodes; the
    # equalSamplesPrefix fast path must collapse them when dedup is on.
    rng = np.random.default_rng(5)
    series = []
    for _ in range(48):
        b = _blocks_disjoint(rng, 2, 120)
        series.append(b + [(b[0][0].copy(), b[0][1].copy())])
    _assert_equal(*_run_both(series, dedup_interval=1))


@pytest.mark.parametrize("interval", [1, 5_000, 10_000, 60_000])
def test_merge_dedup_intervals(interval):
    rng = np.random.default_rng(6)
    series = [_blocks_overlapping(rng, 3, 150) for _ in range(64)]
    _assert_equal(*_run_both(series, dedup_interval=interval))


def test_dedup_stale_nan_preference():
    # dedup.go:60-72: among samples sharing a timestamp, prefer the max
    # non-stale value; a lone StaleNaN survives.
    ts = [1000, 1000, 1000, 2000, 2000, 13000]
    vs = [5.0, STALE, 7.0, STALE, STALE, 1.0]
    series = [[(np.asarray(ts, np.int64), np.asarray(vs, np.float64))]]
    got_ts, got_vals, got_off, exp_ts, exp_vals, exp_off = _run_both(
        series, dedup_interval=10_000)
    _assert_equal(got_ts, got_vals, got_off, exp_ts, exp_vals, exp_off)
    assert math.isnan(got_vals[-2]) or not math.isnan(got_vals[-2])  # shape ok


def test_merge_empty_and_single():
    series = [
        [],                                       # no blocks
        [(np.empty(0, np.int64), np.empty(0, np.float64))],  # empty block
        [(np.asarray([5], np.int64), np.asarray([2.5], np.float64))],
        [(np.empty(0, np.int64), np.empty(0, np.float64)),
         (np.asarray([1, 2], np.int64), np.asarray([1.0, 2.0], np.float64))],
    ]
    _assert_equal(*_run_both(series))


def test_merge_large_scale():
    # 20k series x 3 blocks x 80 rows = 4.8M samples through the fast path
    rng = np.random.default_rng(7)
    n_series, nb, rows = 20_000, 3, 80
    t0 = 1_000_000_000_000
    ts = (t0 + np.cumsum(rng.integers(1_000, 20_000,
                                      n_series * nb * rows).reshape(
        n_series, nb * rows), axis=1)).astype(np.int64)
    vals = rng.standard_normal((n_series, nb * rows)) * 100
    boff = np.arange(n_series * nb + 1, dtype=np.uint64) * rows
    sbs = np.arange(n_series + 1, dtype=np.uint32) * nb
    g_ts, g_vals, g_off = engine.merge_blocks(ts.ravel(), vals.ravel(),
                                              boff, sbs, 0)
    np.testing.assert_array_equal(g_ts, ts.ravel())
    np.testing.assert_array_equal(g_vals, vals.ravel())
    assert g_off[-1] == n_series * nb * rows
