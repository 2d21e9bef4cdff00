"""Synthetic decoded-series generator for benchmarks and tests.

Mirrors the BASELINE.md input spec (seed 8428): counter series whose rate is
~Poisson(10/s) sampled every 15 s with +/-500 ms jitter and counter-reset
probability 0.01/sample.  Produces the CSR batch shape the engine consumes
(the decoded RunParallel callback input, netstorage.go:45 Result).
"""
import numpy as np


def counter_batch(n_series, n_samples, start, step=15_000, jitter_ms=500,
                  rate_per_sec=10.0, reset_p=0.01, seed=8428, chunk=65536):
    """Returns (ts, vals, offsets) CSR arrays; samples are on the grid
    [start, start+(n_samples-1)*step] with jitter."""
    rng = np.random.default_rng(seed)
    total = n_series * n_samples
    ts = np.empty(total, dtype=np.int64)
    vals = np.empty(total, dtype=np.float64)
    offsets = np.arange(n_series + 1, dtype=np.uint64) * n_samples
    lam = rate_per_sec * step / 1000.0
    base_t = start + np.arange(n_samples, dtype=np.int64) * step
    for s0 in range(0, n_series, chunk):
        s1 = min(s0 + chunk, n_series)
        m = s1 - s0
        inc = rng.poisson(lam, size=(m, n_samples)).astype(np.float64)
        c = np.cumsum(inc, axis=1)
        resets = rng.random((m, n_samples)) < reset_p
        resets[:, 0] = False
        # counter restarts at each reset: value = c - c[last reset]
        base = np.where(resets, c, -np.inf)
        base = np.maximum.accumulate(base, axis=1)
        base = np.where(np.isfinite(base), base, 0.0)
        v = c - base
        jit = rng.integers(-jitter_ms, jitter_ms + 1, size=(m, n_samples))
        t = base_t[None, :] + jit
        t = np.sort(t, axis=1)
        ts[s0 * n_samples:s1 * n_samples] = t.reshape(-1)
        vals[s0 * n_samples:s1 * n_samples] = v.reshape(-1)
    return ts, vals, offsets


def gauge_batch(n_series, n_samples, start, step=15_000, seed=8428, chunk=65536):
    """Random-walk gauges (config 5 shape)."""
    rng = np.random.default_rng(seed)
    total = n_series * n_samples
    ts = np.empty(total, dtype=np.int64)
    vals = np.empty(total, dtype=np.float64)
    offsets = np.arange(n_series + 1, dtype=np.uint64) * n_samples
    base_t = start + np.arange(n_samples, dtype=np.int64) * step
    for s0 in range(0, n_series, chunk):
        s1 = min(s0 + chunk, n_series)
        m = s1 - s0
        v = np.cumsum(rng.standard_normal((m, n_samples)), axis=1)
        ts[s0 * n_samples:s1 * n_samples] = np.broadcast_to(
            base_t, (m, n_samples)).reshape(-1)
        vals[s0 * n_samples:s1 * n_samples] = v.reshape(-1)
    return ts, vals, offsets


