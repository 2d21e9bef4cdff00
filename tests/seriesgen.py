"""Irregular series batches for parity tests (test infrastructure)."""
import numpy as np

import oracle


def ragged_batch(n_series, max_samples, start, step=15_000, seed=1234,
                 stale_p=0.0, dup_p=0.0, include_empty=True):
    """Varying lengths (including empty series), optional Prometheus stale
    NaNs and duplicate timestamps — the edge cases the reference tests."""
    rng = np.random.default_rng(seed)
    ts_parts, val_parts = [], []
    offsets = [0]
    for s in range(n_series):
        if include_empty and s % 17 == 0:
            offsets.append(offsets[-1])
            continue
        n = int(rng.integers(1, max_samples + 1))
        t = start + np.cumsum(rng.integers(1, 2 * step, n)).astype(np.int64)
        if dup_p > 0:
            dup = rng.random(n) < dup_p
            t[dup] = np.roll(t, 1)[dup]
            t = np.sort(t)
        if rng.random() < 0.5:
            # counter-like with resets
            v = np.cumsum(np.abs(rng.standard_normal(n)))
            r = rng.random(n) < 0.05
            base = np.maximum.accumulate(np.where(r, v, -np.inf))
            v = v - np.where(np.isfinite(base), base, 0.0)
        else:
            v = np.abs(np.cumsum(rng.standard_normal(n))) * 10
        if stale_p > 0:
            sm = rng.random(n) < stale_p
            v = np.where(sm, oracle.stale_nan(), v)
        ts_parts.append(t)
        val_parts.append(v)
        offsets.append(offsets[-1] + n)
    ts = np.concatenate(ts_parts) if ts_parts else np.empty(0, np.int64)
    vals = np.concatenate(val_parts) if val_parts else np.empty(0, np.float64)
    return (ts.astype(np.int64), vals.astype(np.float64),
            np.asarray(offsets, dtype=np.uint64))


def assert_parity(got, ref, exact=True, rtol=1e-12, atol=0.0, context=""):
    """NaN==NaN; exact bit equality for arithmetic-only funcs, else rtol
    (+ optional atol floor: cross-series float sums of near-cancelling
    columns — e.g. sum(zscore_over_time) ~ 0 — carry addition-order noise
    at machine epsilon that no relative tolerance can bound; the reference
    itself sums in nondeterministic worker order)."""
    assert got.shape == ref.shape, f"{context}: shape {got.shape} vs {ref.shape}"
    gn, rn = np.isnan(got), np.isnan(ref)
    assert (gn == rn).all(), \
        f"{context}: NaN placement differs at {np.argwhere(gn != rn)[:5]}"
    g, r = got[~gn], ref[~rn]
    if exact:
        bad = g != r
        assert not bad.any(), \
            f"{context}: {bad.sum()} exact mismatches; first diffs " \
            f"{g[bad][:3]} vs {r[bad][:3]}"
    else:
        ok = np.isclose(g, r, rtol=rtol, atol=atol)
        assert ok.all(), \
            f"{context}: {np.sum(~ok)} mismatches beyond rtol={rtol}; " \
            f"first {g[~ok][:3]} vs {r[~ok][:3]}"
