"""Packed block stream (vmgpu_batch_create_packed wire form): the oracle
packer's output parsed back on the CPU and decoded with the oracle codec —
pins the header layout and the post-zstd marshal-type contract that the
native C parser (victoriametrics_amd/csrc/vmgpu.hip vmgpu_batch_create_packed)
relies on.  Reference analog: the per-series sortedBlock handoff of
netstorage.go:423-614 re-shaped as one contiguous buffer.
"""
import struct

import numpy as np
import pytest

import oracle

HDR = struct.Struct("<qqqIiIIBBB5x")  # vmgpu_packed_block_hdr, 48 bytes


def _parse(packed, n_blocks):
    """Parse the packed stream exactly as the C parser does."""
    out = []
    off = 0
    for _ in range(n_blocks):
        (min_ts, max_ts, first_value, rows, scale, ts_len, val_len,
         ts_mt, val_mt, pbits) = HDR.unpack_from(packed, off)
        off += HDR.size
        ts_data = packed[off:off + ts_len]
        off += ts_len
        val_data = packed[off:off + val_len]
        off += val_len
        out.append(dict(min_timestamp=min_ts, max_timestamp=max_ts,
                        first_value=first_value, rows=rows, scale=scale,
                        ts_mt=ts_mt, val_mt=val_mt, precision_bits=pbits,
                        ts_data=ts_data, val_data=val_data))
    assert off == len(packed), "stream has trailing bytes"
    return out


def _mixed_batch(rng, n_series=64):
    """Ragged CSR with const / delta-const / gauge / counter series and a
    single-sample series."""
    ts_parts, val_parts, offs = [], [], [0]
    for s in range(n_series):
        kind = s % 5
        n = int(rng.integers(2, 300)) if kind != 4 else 1
        t = 1_600_000_000_000 + np.cumsum(
            rng.integers(14_500, 15_501, n)).astype(np.int64)
        if kind == 0:
            v = np.full(n, 42, dtype=np.int64)                 # const
        elif kind == 1:
            v = 7 + 13 * np.arange(n, dtype=np.int64)          # delta const
        elif kind == 2:
            v = rng.integers(-1000, 1000, n).astype(np.int64)  # gauge
        else:
            v = np.cumsum(rng.integers(0, 500, n)).astype(np.int64)  # counter
        ts_parts.append(t)
        val_parts.append(v)
        offs.append(offs[-1] + n)
    return (np.concatenate(ts_parts), np.concatenate(val_parts),
            np.asarray(offs, dtype=np.uint64))


def test_pack_parse_decode_roundtrip():
    rng = np.random.default_rng(8428)
    ts, vals, offsets = _mixed_batch(rng)
    packed, n_blocks, sbs = oracle.pack_blocks(ts, vals, offsets)
    assert n_blocks == len(offsets) - 1
    assert list(sbs) == list(range(n_blocks + 1))
    blocks = _parse(packed, n_blocks)
    for s, b in enumerate(blocks):
        lo, hi = int(offsets[s]), int(offsets[s + 1])
        want_ts, want_vals = ts[lo:hi], vals[lo:hi]
        assert b["rows"] == hi - lo
        assert b["min_timestamp"] == want_ts[0]
        assert b["max_timestamp"] == want_ts[-1]
        assert b["scale"] == 0
        # post-zstd contract: no zstd marshal types in the stream
        assert b["ts_mt"] not in (oracle.MT_ZSTD_NEAREST_DELTA,
                                  oracle.MT_ZSTD_NEAREST_DELTA2)
        assert b["val_mt"] not in (oracle.MT_ZSTD_NEAREST_DELTA,
                                   oracle.MT_ZSTD_NEAREST_DELTA2)
        got_ts = oracle.unmarshal_int64_array(
            b["ts_data"], b["rows"], b["ts_mt"], b["min_timestamp"])
        got_vals = oracle.unmarshal_int64_array(
            b["val_data"], b["rows"], b["val_mt"], b["first_value"])
        np.testing.assert_array_equal(got_ts, want_ts)
        np.testing.assert_array_equal(got_vals, want_vals)


def test_pack_rejects_empty_and_huge_series():
    ts = np.arange(10, dtype=np.int64)
    vals = np.arange(10, dtype=np.int64)
    with pytest.raises(ValueError):
        oracle.pack_blocks(ts, vals, np.asarray([0, 0, 10], dtype=np.uint64))
    big_n = 9000
    tb = np.arange(big_n, dtype=np.int64)
    with pytest.raises(ValueError):
        oracle.pack_blocks(tb, tb, np.asarray([0, big_n], dtype=np.uint64))
