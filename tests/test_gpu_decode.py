"""GPU block-decode parity: device varint/delta decode + decimal->float vs
the oracle codec (both sides pinned by the reference's test vectors).  Test
blocks are produced by the ORACLE encoder (input generation — the write path
is vminsert's job and out of scope); zstd frames are
decompressed on the host before upload, exactly as the production host side
would (zstd is a CPU format)."""
import math

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

START = 1_000_000_000_000


def _make_block(rng, kind, rows, pb=64, scale=0):
    """Build one (ts_ints, val_ints) block + its marshaled streams."""
    ts = START + np.cumsum(rng.integers(5_000, 25_000, rows)).astype(np.int64)
    if kind == "counter":
        va = np.cumsum(rng.integers(0, 1000, rows)).astype(np.int64)
    elif kind == "gauge":
        va = (np.cumsum(rng.standard_normal(rows)) * 1e6).astype(np.int64)
    elif kind == "const":
        va = np.full(rows, int(rng.integers(-5, 100)), dtype=np.int64)
    elif kind == "delta_const":
        va = (np.arange(rows) * int(rng.integers(1, 50)) + 7).astype(np.int64)
    elif kind == "specials":
        va = np.cumsum(rng.integers(0, 100, rows)).astype(np.int64)
        va[rng.integers(0, rows)] = 2**63 - 1      # vInfPos
        va[rng.integers(0, rows)] = -(2**63)       # vInfNeg
        va[rng.integers(0, rows)] = 2**63 - 2      # vStaleNaN
    else:
        raise ValueError(kind)

    tdata, tmt, tfirst = oracle.marshal_int64_array(ts, pb)
    vdata, vmt, vfirst = oracle.marshal_int64_array(va, pb)
    if tmt in (oracle.MT_ZSTD_NEAREST_DELTA, oracle.MT_ZSTD_NEAREST_DELTA2):
        tdata = oracle.zstd_decompress(tdata)
        tmt = oracle.MT_NEAREST_DELTA if tmt == oracle.MT_ZSTD_NEAREST_DELTA \
            else oracle.MT_NEAREST_DELTA2
    if vmt in (oracle.MT_ZSTD_NEAREST_DELTA, oracle.MT_ZSTD_NEAREST_DELTA2):
        vdata = oracle.zstd_decompress(vdata)
        vmt = oracle.MT_NEAREST_DELTA if vmt == oracle.MT_ZSTD_NEAREST_DELTA \
            else oracle.MT_NEAREST_DELTA2

    # reference decode (the oracle side of parity)
    ts_ref = oracle.unmarshal_int64_array(tdata, rows, tmt, tfirst)
    if pb < 64:
        ts_ref = oracle.ensure_non_decreasing(ts_ref, int(ts[0]), int(ts[-1]))
    va_ref = oracle.unmarshal_int64_array(vdata, rows, vmt, vfirst)
    vals_ref = oracle.decimal_append_to_float(va_ref, scale)

    block = {
        "ts_data": tdata, "ts_mt": tmt,
        "min_timestamp": int(ts[0]), "max_timestamp": int(ts[-1]),
        "val_data": vdata, "val_mt": vmt, "first_value": vfirst,
        "scale": scale, "precision_bits": pb, "rows": rows,
    }
    return block, np.asarray(ts_ref), np.asarray(vals_ref)


def _check(blocks, refs):
    from victoriametrics_amd import engine
    ts, vals, offsets = engine.decode_blocks(blocks)
    for i, (rt, rv) in enumerate(refs):
        lo, hi = int(offsets[i]), int(offsets[i + 1])
        assert np.array_equal(ts[lo:hi], rt), f"block {i}: timestamps differ"
        got = vals[lo:hi]
        gb = got.view(np.uint64)
        rb = np.asarray(rv).view(np.uint64)
        assert np.array_equal(gb, rb), \
            f"block {i}: values differ bitwise at " \
            f"{np.argwhere(gb != rb)[:5].ravel()}"


@pytest.mark.parametrize("kind", ["counter", "gauge", "const", "delta_const",
                                  "specials"])
def test_decode_kinds(kind):
    rng = np.random.default_rng(hash(kind) % 2**31)
    blocks, refs = [], []
    for rows in (1, 2, 3, 64, 65, 1000, 8192):
        if kind == "delta_const" and rows < 2:
            continue
        b, rt, rv = _make_block(rng, kind, rows)
        blocks.append(b)
        refs.append((rt, rv))
    _check(blocks, refs)


@pytest.mark.parametrize("pb", [4, 8, 16, 32, 63])
def test_decode_lossy_precision(pb):
    """precisionBits < 64: EnsureNonDecreasingSequence applies to timestamps
    and lossy deltas round-trip with the same values as the CPU decode."""
    rng = np.random.default_rng(pb)
    blocks, refs = [], []
    for rows in (3, 100, 2048):
        b, rt, rv = _make_block(rng, "counter", rows, pb=pb)
        blocks.append(b)
        refs.append((rt, rv))
        b, rt, rv = _make_block(rng, "gauge", rows, pb=pb)
        blocks.append(b)
        refs.append((rt, rv))
    _check(blocks, refs)


@pytest.mark.parametrize("scale", [-9, -3, -1, 0, 1, 3, 18])
def test_decode_scales(scale):
    rng = np.random.default_rng(scale + 100)
    blocks, refs = [], []
    for kind in ("counter", "specials"):
        b, rt, rv = _make_block(rng, kind, 500, scale=scale)
        blocks.append(b)
        refs.append((rt, rv))
    _check(blocks, refs)


def test_decode_many_blocks():
    """More blocks than decode workgroups (grid-stride path)."""
    rng = np.random.default_rng(42)
    blocks, refs = [], []
    for i in range(1500):
        rows = int(rng.integers(1, 120))
        kind = ["counter", "gauge", "const"][i % 3]
        b, rt, rv = _make_block(rng, kind, max(rows, 1))
        blocks.append(b)
        refs.append((rt, rv))
    _check(blocks, refs)


def test_decode_then_rollup_end_to_end():
    """Compressed blocks -> GPU decode -> GPU rollup, against the full CPU
    pipeline (oracle decode + oracle rollup)."""
    from victoriametrics_amd import engine
    rng = np.random.default_rng(7)
    blocks, refs = [], []
    for s in range(200):
        b, rt, rv = _make_block(rng, "counter", 240)
        blocks.append(b)
        refs.append((rt, rv))
    ts, vals, offsets = engine.decode_blocks(blocks)
    start = START + 300_000
    end = start + 100 * 15_000
    plan = engine.RollupPlan("rate", start, end, 15_000, window=120_000)
    out, _, _ = engine.rollup_eval(plan, ts, vals, offsets)
    rc = oracle.make_config("rate", start, end, 15_000, window=120_000)
    rts = np.concatenate([r[0] for r in refs])
    rvs = np.concatenate([r[1] for r in refs])
    ref, _, _ = oracle.rollup_eval_batch(rc, rts, rvs, offsets,
                                         remove_counter_resets=True,
                                         drop_stale_nans=True)
    gn, rn = np.isnan(out), np.isnan(ref)
    assert (gn == rn).all()
    assert np.array_equal(out[~gn], ref[~rn])
