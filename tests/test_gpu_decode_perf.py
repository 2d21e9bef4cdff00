"""Decode-throughput measurement (evidence for the §8f(1) row; numbers land
in profiles/).  Not a parity test — prints one JSON line with decoded
samples/s and the compressed:decoded byte ratio (the PCIe win of shipping
compressed blocks instead of decoded columns)."""
import json
import time

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

START = 1_000_000_000_000


def test_decode_throughput(capsys):
    from victoriametrics_amd import engine
    rng = np.random.default_rng(8428)
    n_blocks, rows = 4096, 8192  # 33.5M samples
    blocks = []
    payload_bytes = 0
    for i in range(n_blocks):
        ts = START + np.cumsum(np.full(rows, 15_000, dtype=np.int64)) \
            + rng.integers(-500, 500, rows)
        ts = np.sort(ts)
        va = np.cumsum(rng.integers(100, 200, rows)).astype(np.int64)
        tdata, tmt, tfirst = oracle.marshal_int64_array(ts, 64)
        vdata, vmt, vfirst = oracle.marshal_int64_array(va, 64)
        for which, data, mt in (("t", tdata, tmt), ("v", vdata, vmt)):
            if mt in (oracle.MT_ZSTD_NEAREST_DELTA,
                      oracle.MT_ZSTD_NEAREST_DELTA2):
                payload_bytes += len(data)  # compressed size (PCIe side)
                data = oracle.zstd_decompress(data)
                mt = oracle.MT_NEAREST_DELTA \
                    if mt == oracle.MT_ZSTD_NEAREST_DELTA \
                    else oracle.MT_NEAREST_DELTA2
            else:
                payload_bytes += len(data)
            if which == "t":
                tdata, tmt = data, mt
            else:
                vdata, vmt = data, mt
        blocks.append({
            "ts_data": tdata, "ts_mt": tmt,
            "min_timestamp": int(ts[0]), "max_timestamp": int(ts[-1]),
            "val_data": vdata, "val_mt": vmt, "first_value": vfirst,
            "scale": 0, "precision_bits": 64, "rows": rows,
        })
    total_rows = n_blocks * rows
    # warmup + timed runs (includes PCIe upload of payload + download of
    # decoded columns; kernel-only time would need the resident-batch API)
    engine.decode_blocks(blocks[:64])
    t0 = time.perf_counter()
    ts_out, vals_out, _ = engine.decode_blocks(blocks)
    dt = time.perf_counter() - t0
    assert len(ts_out) == total_rows
    assert bool(np.all(np.diff(ts_out[:rows]) >= 0))
    line = {
        "metric": "block decode samples/s (end-to-end incl. PCIe)",
        "value": total_rows / dt,
        "total_samples": total_rows,
        "wall_s": dt,
        "decoded_bytes": total_rows * 16,
        "payload_bytes_uncompressed_varint": sum(
            len(b["ts_data"]) + len(b["val_data"]) for b in blocks),
        "compression_vs_decoded": (total_rows * 16) / max(payload_bytes, 1),
    }
    with capsys.disabled():
        print("\nDECODE_BENCH " + json.dumps(line), flush=True)
