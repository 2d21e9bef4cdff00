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
import time

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


# -search.cacheTimestampOffset default (rollup_result_cache.go:29): points
# newer than now - step - offset are not cached (they may still be rewritten
# by late inserts).
CACHE_TIMESTAMP_OFFSET_MS = 5 * 60 * 1000


class RollupResultCache:
    """rollupResultCache mirror (rollup_result_cache.go:283-465): stores
    marshaled series under per-(expr, window, step, filters) metainfo with
    up to 10 time-range entries (AddKey drops the oldest 5 past that,
    rollup_result_cache.go:595-607); get_series() picks the best entry
    (GetBestKey: latest-start entry containing `start`, maximal usable
    span, rollup_result_cache.go:575-593), slices it to [start, end], and
    returns new_start = last returned timestamp + step, so only
    (new_start..end] must be computed; merge_series + put_series complete
    the round trip."""

    def __init__(self, max_bytes=256 << 20):
        self.max_bytes = max_bytes
        self._meta = {}    # key -> [[start, end, skey], ...]  (AddKey order)
        self._blobs = {}   # skey -> marshaled series (dict preserves order)
        self._size = 0
        self._suffix = 0

    def reset(self):
        self._meta.clear()
        self._blobs.clear()
        self._size = 0

    @staticmethod
    def _key(expr, window, step, filters=b""):
        return (str(expr), int(window), int(step), bytes(filters))

    def _evict(self):
        while self._size > self.max_bytes and self._blobs:
            skey = next(iter(self._blobs))
            self._size -= len(self._blobs.pop(skey))
            # dangling metainfo entries are dropped lazily on get_series,
            # as the reference's RemoveKey-on-cache-miss path does
            # (rollup_result_cache.go:313-320).

    def put_series(self, expr, window, step, names, values, timestamps,
                   filters=b"", now_ms=None):
        """PutSeries (rollup_result_cache.go:364-465).  `timestamps` is the
        actual grid the series were computed on (it may extend beyond the
        request range on either side)."""
        if len(names) == 0:
            return
        if len(names) > 1:
            # series with duplicate naming cannot be merged later; skip
            # (rollup_result_cache.go:376-390)
            seen = set()
            for name in names:
                k = _name_key(name)
                if k in seen:
                    return
                seen.add(k)
        timestamps = np.ascontiguousarray(timestamps, np.int64)
        values = np.ascontiguousarray(values, np.float64)
        # drop trailing points newer than now - step - cacheTimestampOffset
        # (rollup_result_cache.go:392-415)
        if now_ms is None:
            now_ms = int(time.time() * 1000)
        deadline = int(now_ms) - int(step) - CACHE_TIMESTAMP_OFFSET_MS
        i = len(timestamps)
        while i > 0 and timestamps[i - 1] > deadline:
            i -= 1
        if i == 0:
            return
        if i < len(timestamps):
            timestamps = timestamps[:i]
            values = values[:, :i]
        start, end = int(timestamps[0]), int(timestamps[-1])
        key = self._key(expr, window, step, filters)
        entries = self._meta.setdefault(key, [])
        # CoversTimeRange (rollup_result_cache.go:563-573)
        for e_start, e_end, _ in entries:
            if start >= e_start and end <= e_end:
                return
        data = marshal_timeseries_fast(names, values, timestamps)
        if not data:
            return
        self._suffix += 1
        skey = self._suffix
        self._blobs[skey] = data
        self._size += len(data)
        self._evict()
        entries.append([start, end, skey])
        if len(entries) > 10:
            del entries[:5]

    # -- instant values (rollup_result_cache.go:220-281): a flat keyed
    # store of single-point series, used by the instant-rollup
    # optimization above this layer.  Keys are disjoint from the series
    # keys (the reference prefixes rollupResultCacheTypePrefix).

    @staticmethod
    def _ikey(expr, window, step, filters=b""):
        return ("instant", str(expr), int(window), int(step), bytes(filters))

    @staticmethod
    def _assert_instant(values, timestamps):
        # assertInstantValues: one point per series, shared timestamp
        if np.asarray(timestamps).size != 1:
            raise ValueError("instant series must have exactly one point")
        if np.asarray(values).ndim == 2 and np.asarray(values).shape[1] != 1:
            raise ValueError("instant series must have exactly one value")

    def put_instant_values(self, expr, window, step, names, values,
                           timestamps, filters=b""):
        if len(names) == 0:
            return
        self._assert_instant(values, timestamps)
        values = np.asarray(values, np.float64).reshape(len(names), 1)
        data = marshal_timeseries_fast(
            names, values, np.asarray(timestamps, np.int64).reshape(1))
        k = self._ikey(expr, window, step, filters)
        old = self._blobs.pop(k, None)
        if old is not None:
            self._size -= len(old)
        self._blobs[k] = data
        self._size += len(data)
        self._evict()

    def get_instant_values(self, expr, window, step, filters=b""):
        """Returns (names, values[n,1], timestamp) or (None, None, None)."""
        data = self._blobs.get(self._ikey(expr, window, step, filters))
        if data is None:
            return None, None, None
        names, values, timestamps = unmarshal_timeseries_fast(data)
        if len(names) == 0:
            return None, None, None
        return names, values, int(np.asarray(timestamps)[0])

    def delete_instant_values(self, expr, window, step, filters=b""):
        old = self._blobs.pop(self._ikey(expr, window, step, filters), None)
        if old is not None:
            self._size -= len(old)

    def get_series(self, expr, window, step, start, end, filters=b""):
        """GetSeries (rollup_result_cache.go:283-361).  Returns
        (names, values, timestamps, new_start); a miss returns
        (None, None, None, start)."""
        start, end, step = int(start), int(end), int(step)
        key = self._key(expr, window, step, filters)
        entries = self._meta.get(key)
        if not entries:
            return None, None, None, start
        # GetBestKey: among entries starting at or before `start`, the one
        # with the largest usable span d = min(end, e.end) - start (>= 0)
        best = None
        d_max = 0
        for e in entries:
            e_start, e_end, _ = e
            if start < e_start:
                continue
            d = (end if end <= e_end else e_end) - start
            if d >= d_max:
                d_max = d
                best = e
        if best is None:
            return None, None, None, start
        data = self._blobs.get(best[2])
        if data is None:  # evicted under the metainfo: RemoveKey + miss
            entries.remove(best)
            return None, None, None, start
        names, values, timestamps = unmarshal_timeseries_fast(data)
        timestamps = np.asarray(timestamps, np.int64)
        i = int(np.searchsorted(timestamps, start, side="left"))
        if i == len(timestamps) or timestamps[i] != start:
            # cached series don't cover `start` on the exact grid
            return None, None, None, start
        j = int(np.searchsorted(timestamps, end, side="right"))
        if j <= i:
            return None, None, None, start
        out_ts = timestamps[i:j]
        new_start = int(out_ts[-1]) + step
        return names, values[:, i:j], out_ts, new_start


# -- disk persistence (InitRollupResultCache / StopRollupResultCache,
# rollup_result_cache.go:119-199: with a cachePath the fastcache working
# set is loaded at startup and saved at shutdown).  The blob payloads stay
# in the marshalTimeseriesFast binary layout; the index around them is this
# engine's own compact framing (the reference's fastcache file format is an
# implementation detail of its cache library, not of the query path).

_CACHE_MAGIC = b"vmgpu-rollupcache-v1\n"


def _w_bytes(f, b):
    f.write(struct.pack(">I", len(b)))
    f.write(b)


def _r_bytes(f):
    (n,) = struct.unpack(">I", f.read(4))
    return f.read(n)


def save_rollup_result_cache(cache, path):
    """StopRollupResultCache's save half (rollup_result_cache.go:190-199):
    persist every live series entry and instant entry."""
    import os
    tmp = path + ".tmp"
    with open(tmp, "wb") as f:
        f.write(_CACHE_MAGIC)
        # series entries (only those whose blob is still resident)
        live = []
        for key, entries in cache._meta.items():
            for e_start, e_end, skey in entries:
                blob = cache._blobs.get(skey)
                if blob is not None:
                    live.append((key, e_start, e_end, blob))
        f.write(struct.pack(">I", len(live)))
        for (expr, window, step, filters), e_start, e_end, blob in live:
            _w_bytes(f, expr.encode("utf-8", "surrogateescape"))
            f.write(struct.pack(">qq", window, step))
            _w_bytes(f, filters)
            f.write(struct.pack(">qq", e_start, e_end))
            _w_bytes(f, blob)
        # instant entries
        instants = [(k, b) for k, b in cache._blobs.items()
                    if isinstance(k, tuple)]
        f.write(struct.pack(">I", len(instants)))
        for (_, expr, window, step, filters), blob in instants:
            _w_bytes(f, expr.encode("utf-8", "surrogateescape"))
            f.write(struct.pack(">qq", window, step))
            _w_bytes(f, filters)
            _w_bytes(f, blob)
    os.replace(tmp, path)


def load_rollup_result_cache(path, max_bytes=256 << 20):
    """InitRollupResultCache's load half: returns a RollupResultCache
    seeded from `path`, or an empty one when the file is missing or
    corrupt (fastcache.Load semantics — a bad cache never fails startup)."""
    cache = RollupResultCache(max_bytes=max_bytes)
    try:
        with open(path, "rb") as f:
            if f.read(len(_CACHE_MAGIC)) != _CACHE_MAGIC:
                return cache
            (n_series,) = struct.unpack(">I", f.read(4))
            for _ in range(n_series):
                expr = _r_bytes(f).decode("utf-8", "surrogateescape")
                window, step = struct.unpack(">qq", f.read(16))
                filters = _r_bytes(f)
                e_start, e_end = struct.unpack(">qq", f.read(16))
                blob = _r_bytes(f)
                key = cache._key(expr, window, step, filters)
                cache._suffix += 1
                skey = cache._suffix
                cache._blobs[skey] = blob
                cache._size += len(blob)
                cache._meta.setdefault(key, []).append(
                    [e_start, e_end, skey])
            (n_instant,) = struct.unpack(">I", f.read(4))
            for _ in range(n_instant):
                expr = _r_bytes(f).decode("utf-8", "surrogateescape")
                window, step = struct.unpack(">qq", f.read(16))
                filters = _r_bytes(f)
                blob = _r_bytes(f)
                k = cache._ikey(expr, window, step, filters)
                cache._blobs[k] = blob
                cache._size += len(blob)
            cache._evict()
    except (OSError, struct.error):
        return RollupResultCache(max_bytes=max_bytes)
    return cache
