"""Ad-hoc measurement: the fused cold-cache fetch path (compressed blocks ->
device decode -> merge -> resident batch -> rollup) vs the staged host-decode
path.  Run on a GPU box: python tests/bench_cold_fetch.py [n_series].
Numbers land in DESIGN.md §3c."""
import sys
import time

import numpy as np

sys.path.insert(0, ".")
import oracle  # noqa: E402
from victoriametrics_amd import engine  # noqa: E402
from victoriametrics_amd.engine import RollupPlan, SeriesBatch  # noqa: E402

START = 1_000_000_000_000


def make_blocks(n_series, rows):
    rng = np.random.default_rng(8428)
    blocks, sbs = [], [0]
    t0 = time.time()
    # one block per series, counter-shaped (varint-friendly)
    step_ms = 15_000
    for s in range(n_series):
        ts = START + np.cumsum(
            rng.integers(step_ms - 500, step_ms + 500, rows)).astype(np.int64)
        va = np.cumsum(rng.integers(0, 1000, rows)).astype(np.int64)
        tdata, tmt, tfirst = oracle.marshal_int64_array(ts, 64)
        vdata, vmt, vfirst = oracle.marshal_int64_array(va, 64)
        if tmt in (oracle.MT_ZSTD_NEAREST_DELTA, oracle.MT_ZSTD_NEAREST_DELTA2):
            tdata = oracle.zstd_decompress(tdata)
            tmt = oracle.MT_NEAREST_DELTA if tmt == oracle.MT_ZSTD_NEAREST_DELTA \
                else oracle.MT_NEAREST_DELTA2
        if vmt in (oracle.MT_ZSTD_NEAREST_DELTA, oracle.MT_ZSTD_NEAREST_DELTA2):
            vdata = oracle.zstd_decompress(vdata)
            vmt = oracle.MT_NEAREST_DELTA if vmt == oracle.MT_ZSTD_NEAREST_DELTA \
                else oracle.MT_NEAREST_DELTA2
        blocks.append({
            "ts_data": tdata, "ts_mt": tmt, "min_timestamp": int(ts[0]),
            "max_timestamp": int(ts[-1]), "val_data": vdata, "val_mt": vmt,
            "first_value": vfirst, "scale": 0, "precision_bits": 64,
            "rows": rows,
        })
        sbs.append(len(blocks))
    gen_s = time.time() - t0
    payload = sum(len(b["ts_data"]) + len(b["val_data"]) for b in blocks)
    print(f"generated {n_series}x{rows} blocks in {gen_s:.1f}s, "
          f"payload {payload / 1e6:.1f} MB "
          f"({payload / (n_series * rows):.2f} B/sample)")
    return blocks, np.asarray(sbs, np.uint32)


def main():
    n_series = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000
    rows = 240
    blocks, sbs = make_blocks(n_series, rows)
    engine.init()

    # fused path: payload -> resident batch (decode+merge on device)
    t0 = time.time()
    batch = SeriesBatch.from_blocks(blocks, sbs)
    fused_s = time.time() - t0
    n_samples = n_series * rows
    print(f"fused decode->batch: {fused_s * 1e3:.1f} ms "
          f"= {n_samples / fused_s / 1e6:.0f} Msamples/s (incl PCIe payload)")

    start = START + 600_000
    plan = RollupPlan("rate", start, start + 100 * 15_000, 15_000,
                      window=300_000)
    t0 = time.time()
    out, _, _ = batch.exec(plan)
    print(f"rollup on fused batch: {(time.time() - t0) * 1e3:.1f} ms wall, "
          f"{engine.last_kernel_ms():.2f} ms kernel")
    batch.close()

    # staged path: device decode -> host columns -> re-upload
    t0 = time.time()
    ts, vals, offsets = engine.decode_blocks(blocks)
    sbs64 = np.zeros(n_series + 1, np.uint64)
    for s in range(n_series):
        sbs64[s + 1] = sbs64[s] + rows
    staged = SeriesBatch(ts, vals, sbs64)
    staged_s = time.time() - t0
    print(f"staged decode->host->batch: {staged_s * 1e3:.1f} ms "
          f"= {n_samples / staged_s / 1e6:.0f} Msamples/s")
    out2, _, _ = staged.exec(plan)
    staged.close()
    same = np.array_equal(out.view(np.int64), out2.view(np.int64))
    print(f"fused == staged results: {same}")


if __name__ == "__main__":
    main()
