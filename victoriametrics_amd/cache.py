"""rollupResultCache host layer: the binary series layout and the partial-hit
merge (SURVEY.md §8a "rollupResultCache" row).

Product host code (this layer is host-side in the reference too).  Mirrors:
  - marshalTimeseriesFast / unmarshalTimeseriesFast (timeseries.go:81-180):
    big-endian u64 counts, shared timestamps once, raw 8-byte values
    (native little-endian memory, as the reference's unsafe slice cast),
    then metric names (u16-framed MetricGroup + tag key/values,
    timeseries.go:226-310).
  - mergeSeries (rollup_result_cache.go:618-720): concatenate the cached
    prefix window with freshly computed suffix series on the shared grid,
    NaN-filling series missing on either side; refuses duplicate keys.
  - a window cache in the spirit of rollupResultCache.GetSeries/PutSeries
    (rollup_result_cache.go): key = (expr, window, step, filters); a hit
    whose [start, end'] prefixes the request shrinks the fetch to
    (end' + step, end].

Series identity: callers pass metric names as
(metric_group: bytes, tags: tuple[(key: bytes, value: bytes), ...]);
grouping keys use the sorted-tags form (marshalMetricNameSorted semantics).
"""
import struct

import numpy as np


def _name_key(name):
    group, tags = name
    return (bytes(group), tuple(sorted((bytes(k), bytes(v)) for k, v in tags)))


def marshal_timeseries_fast(names, values, timestamps, max_size=1 << 62, step=0):
    """values: [n_series x n_points] f64; timestamps: [n_points] i64."""
    values = np.ascontiguousarray(values, dtype=np.float64)
    timestamps = np.ascontiguousarray(timestamps, dtype=np.int64)
    n = len(names)
    if n == 0:
        return struct.pack(">QQ", 0, 0)
    assert values.shape == (n, len(timestamps))
    out = bytearray()
    out += struct.pack(">QQ", n, len(timestamps))
    out += timestamps.tobytes()
    out += values.tobytes()
    for group, tags in names:
        out += struct.pack(">H", len(group)) + bytes(group)
        out += struct.pack(">H", len(tags))
        for k, v in tags:
            out += struct.pack(">H", len(k)) + bytes(k)
            out += struct.pack(">H", len(v)) + bytes(v)
    if len(out) > max_size:
        return b""
    return bytes(out)


def unmarshal_timeseries_fast(data):
    if len(data) < 16:
        raise ValueError("need at least 16 bytes")
    n, npts = struct.unpack(">QQ", data[:16])
    off = 16
    timestamps = np.frombuffer(data, dtype=np.int64, count=npts, offset=off)
    off += 8 * npts
    values = np.frombuffer(data, dtype=np.float64, count=n * npts,
                           offset=off).reshape(n, npts) if n else \
        np.empty((0, npts))
    off += 8 * n * npts
    names = []
    for _ in range(n):
        glen = struct.unpack(">H", data[off:off + 2])[0]
        off += 2
        group = data[off:off + glen]
        off += glen
        ntags = struct.unpack(">H", data[off:off + 2])[0]
        off += 2
        tags = []
        for _ in range(ntags):
            klen = struct.unpack(">H", data[off:off + 2])[0]
            off += 2
            k = data[off:off + klen]
            off += klen
            vlen = struct.unpack(">H", data[off:off + 2])[0]
            off += 2
            v = data[off:off + vlen]
            off += vlen
            tags.append((k, v))
        names.append((group, tuple(tags)))
    if off != len(data):
        raise ValueError(f"unexpected tail of {len(data) - off} bytes")
    return names, values, timestamps


def marshal_from_batch(batch, names, timestamps, rows=None, max_size=1 << 62):
    """rollupResultCache binary fill straight off the device:
    the values section of marshalTimeseriesFast IS the engine's row-major
    [rows x n_points] f64 output, so the device buffer is downloaded
    directly into the marshal buffer at its final offset — no host repack
    pass.  batch: an engine.SeriesBatch whose last exec produced `rows`
    series rows on the same grid as `timestamps`."""
    timestamps = np.ascontiguousarray(timestamps, dtype=np.int64)
    n = len(names)
    npts = len(timestamps)
    if n == 0:
        return struct.pack(">QQ", 0, 0)
    if rows is None:
        rows = n
    assert rows == n
    head = struct.pack(">QQ", n, npts) + timestamps.tobytes()
    tail = bytearray()
    for group, tags in names:
        tail += struct.pack(">H", len(group)) + bytes(group)
        tail += struct.pack(">H", len(tags))
        for k, v in tags:
            tail += struct.pack(">H", len(k)) + bytes(k)
            tail += struct.pack(">H", len(v)) + bytes(v)
    vbytes = 8 * n * npts
    out = bytearray(len(head) + vbytes + len(tail))
    out[:len(head)] = head
    out[len(head) + vbytes:] = tail
    dst = np.frombuffer(out, dtype=np.float64, count=n * npts,
                        offset=len(head)).reshape(n, npts)
    batch.fetch_out_into(dst, npts)
    del dst
    if len(out) > max_size:
        return b""
    return bytes(out)


def merge_series(a_names, a_values, b_names, b_values, b_start, start, end, step):
    """mergeSeries (rollup_result_cache.go:618-720).  a covers
    [start, b_start), b covers [b_start, end].  Returns (names, values) on
    the shared [start:end:step] grid, or None when duplicate keys prevent
    merging."""
    shared = np.arange(start, end + 1, step, dtype=np.int64)
    n_prefix = int(np.searchsorted(shared, b_start, side="left"))
    n_total = len(shared)
    if n_prefix == 0:
        return list(b_names), np.asarray(b_values, dtype=np.float64)

    m_a = {}
    for name, row in zip(a_names, np.asarray(a_values, dtype=np.float64)):
        k = _name_key(name)
        if k in m_a:
            return None
        m_a[k] = (name, row)
    seen_b = set()
    out_names, out_rows = [], []
    for name, row in zip(b_names, np.asarray(b_values, dtype=np.float64)):
        k = _name_key(name)
        if k in seen_b:
            return None
        seen_b.add(k)
        merged = np.full(n_total, np.nan)
        if k in m_a:
            merged[:n_prefix] = m_a.pop(k)[1]
        merged[n_prefix:] = row
        out_names.append(name)
        out_rows.append(merged)
    for k, (name, row) in m_a.items():
        merged = np.full(n_total, np.nan)
        merged[:n_prefix] = row
        out_names.append(name)
        out_rows.append(merged)
    vals = np.vstack(out_rows) if out_rows else np.empty((0, n_total))
    return out_names, vals


class RollupResultCache:
    """Prefix-window cache in the spirit of rollupResultCache
    (rollup_result_cache.go): stores marshaled series per
    (expr, window, step, filters) key; get() returns (series, new_start)
    where new_start > start means only (new_start..end] must be computed,
    after which merge_series + put() complete the round trip."""

    def __init__(self, max_bytes=256 << 20):
        self.max_bytes = max_bytes
        self._store = {}
        self._size = 0

    @staticmethod
    def _key(expr, window, step, filters=b""):
        return (str(expr), int(window), int(step), bytes(filters))

    def put(self, expr, window, step, start, end, names, values, filters=b""):
        timestamps = np.arange(start, end + 1, step, dtype=np.int64)
        data = marshal_timeseries_fast(names, values, timestamps)
        k = self._key(expr, window, step, filters)
        old = self._store.pop(k, None)
        if old is not None:
            self._size -= len(old[2])
        self._store[k] = (start, end, data)
        self._size += len(data)
        while self._size > self.max_bytes and self._store:
            _, (s, e, d) = self._store.popitem()
            self._size -= len(d)

    def get(self, expr, window, step, start, end, filters=b""):
        """Returns (names, values, new_start).  A miss returns
        (None, None, start)."""
        k = self._key(expr, window, step, filters)
        hit = self._store.get(k)
        if hit is None:
            return None, None, start
        c_start, c_end, data = hit
        if c_start != start or c_end < start:
            return None, None, start
        names, values, timestamps = unmarshal_timeseries_fast(data)
        usable_end = min(c_end, end)
        npts = (usable_end - start) // step + 1
        return names, values[:, :npts], start + npts * step
