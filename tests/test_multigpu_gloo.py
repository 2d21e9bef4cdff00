"""Multi-process merge-path test on CPU (gloo, world_size=2).

Validates the SURVEY.md §8e exchange: series sharded by id across ranks,
each rank produces identity-filled partial [groups x grid] value/count
matrices (skip_finalize cut), ONE all-reduce merges them, the host finalize
(product code: vmgpu_aggr_finalize_host) produces the final series.  The
partials here come from the CPU oracle (no GPU in CI); the GPU path of the
same cut is covered by test_gpu_parity.py::test_grouped_skip_finalize_*.
"""
import os
import sys

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from conftest import REPO_ROOT

N_SERIES = 400
N_SAMPLES = 120
N_GROUPS = 13
START = 1_000_000_000_000
STEP = 15_000
END = START + 60 * STEP

REDUCE_OPS = {"sum": dist.ReduceOp.SUM, "avg": dist.ReduceOp.SUM,
              "min": dist.ReduceOp.MIN, "max": dist.ReduceOp.MAX,
              "count": dist.ReduceOp.SUM}
IDENTITY = {"sum": 0.0, "avg": 0.0, "min": float("inf"),
            "max": float("-inf"), "count": 0.0}


def _partials(aggr, ts, vals, offsets, gids, series_sel):
    """Per-rank partial matrices from the oracle rollup + update callbacks
    (identity-filled, matching the GPU skip_finalize layout)."""
    import oracle
    n_grid = oracle.grid_points(START, END, STEP)
    v = np.full((N_GROUPS, n_grid), IDENTITY[aggr], dtype=np.float64)
    c = np.zeros((N_GROUPS, n_grid), dtype=np.float64)
    rc = oracle.make_config("rate", START, END, STEP, window=300_000)
    for s in series_sel:
        lo, hi = int(offsets[s]), int(offsets[s + 1])
        sv = oracle.remove_counter_resets(vals[lo:hi], ts[lo:hi], 0)
        row, _ = oracle.rollup_do(rc, sv, ts[lo:hi])
        g = gids[s]
        keep = ~np.isnan(row)
        if aggr == "sum" or aggr == "avg":
            v[g, keep] += row[keep]
            c[g, keep] += 1
        elif aggr == "min":
            v[g, keep] = np.minimum(v[g, keep], row[keep])
            c[g, keep] += 1
        elif aggr == "max":
            v[g, keep] = np.maximum(v[g, keep], row[keep])
            c[g, keep] += 1
        elif aggr == "count":
            v[g, keep] += 1
            c[g, keep] += 1
    return v, c


def _worker(rank, world, aggr, port, ret):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    sys.path.insert(0, REPO_ROOT)
    sys.path.insert(0, os.path.join(REPO_ROOT, "tests"))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from victoriametrics_amd import synth
        ts, vals, offsets = synth.counter_batch(N_SERIES, N_SAMPLES, START,
                                                seed=4242)
        gids = (np.arange(N_SERIES) * 7 % N_GROUPS).astype(np.int32)
        mine = np.arange(rank, N_SERIES, world)  # shard by seriesID
        v, c = _partials(aggr, ts, vals, offsets, gids, mine)
        tv = torch.from_numpy(v)
        tc = torch.from_numpy(c)
        dist.all_reduce(tv, op=REDUCE_OPS[aggr])
        dist.all_reduce(tc, op=dist.ReduceOp.SUM)
        if rank == 0:
            from victoriametrics_amd import engine
            fin = engine.aggr_finalize(aggr, tv.numpy().copy(),
                                       tc.numpy().copy())
            ret.put((aggr, fin))
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("aggr,world", [
    ("sum", 2), ("min", 2), ("max", 2), ("avg", 2), ("count", 2),
    # the 8-GPU shape's merge semantics at a wider world size
    ("sum", 4), ("avg", 4),
])
def test_sharded_allreduce_merge(aggr, world, tmp_path):
    import oracle
    from victoriametrics_amd import synth
    ctx = mp.get_context("spawn")
    ret = ctx.Queue()
    port = 29531 + (hash(aggr) + world * 37) % 200
    procs = [ctx.Process(target=_worker, args=(r, world, aggr, port, ret))
             for r in range(world)]
    for p in procs:
        p.start()
    got_aggr, fin = ret.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
    assert got_aggr == aggr

    # single-process oracle over ALL series = the unsharded truth
    ts, vals, offsets = synth.counter_batch(N_SERIES, N_SAMPLES, START,
                                            seed=4242)
    gids = (np.arange(N_SERIES) * 7 % N_GROUPS).astype(np.int32)
    rc = oracle.make_config("rate", START, END, STEP, window=300_000)
    ref, _, _ = oracle.rollup_eval_batch(
        rc, ts, vals, offsets, group_ids=gids, n_groups=N_GROUPS, aggr=aggr,
        remove_counter_resets=True, n_threads=2)
    fin = fin.reshape(ref.shape)
    gn, rn = np.isnan(fin), np.isnan(ref)
    assert (gn == rn).all()
    assert np.allclose(fin[~gn], ref[~rn], rtol=1e-9, atol=0)


def test_topk_shard_merge_matches_global():
    """per-shard local top-k candidates + host merge == global
    top-k over the union (candidate completeness argument)."""
    import numpy as np
    from victoriametrics_amd.engine import topk_merge_shards
    rng = np.random.default_rng(3)
    k = 7
    n_per_shard = 40
    # distinct summaries -> tie-free
    sums = rng.permutation(2 * n_per_shard).astype(np.float64)
    shard_sums = [sums[:n_per_shard], sums[n_per_shard:]]
    shard_ids, shard_vals = [], []
    for sv in shard_sums:
        local_top = np.argsort(-sv)[:k]
        shard_ids.append(local_top)
        shard_vals.append(sv[local_top])
    got = topk_merge_shards(shard_ids, shard_vals, k)
    got_sums = sorted(shard_sums[s][i] for s, i in got)
    exp = sorted(np.sort(sums)[-k:])
    assert got_sums == [float(x) for x in exp]
    # bottomk
    shard_ids_b, shard_vals_b = [], []
    for sv in shard_sums:
        local_bot = np.argsort(sv)[:k]
        shard_ids_b.append(local_bot)
        shard_vals_b.append(sv[local_bot])
    got_b = topk_merge_shards(shard_ids_b, shard_vals_b, k, reverse=True)
    got_sums_b = sorted(shard_sums[s][i] for s, i in got_b)
    exp_b = sorted(np.sort(sums)[:k])
    assert got_sums_b == [float(x) for x in exp_b]
    # NaN summaries sort last
    got_n = topk_merge_shards([[0, 1], [0]],
                              [np.asarray([float("nan"), 5.0]),
                               np.asarray([7.0])], 2)
    assert got_n == [(1, 0), (0, 1)]
