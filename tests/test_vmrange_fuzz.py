"""Model-based fuzz of vmrangeBucketsToLE (transform.go:512-650) — the
vmrange→le conversion feeding prometheus_buckets / histogram_quantile:
le passthrough, malformed vmranges, zero-bucket trimming, gap filling
with a zero le=start bucket, duplicate-end merging, the +Inf cap, and
the per-point cumulative sum.  300 random bucket sets vs a literal
restatement.  Duplicate-end buckets are generated NON-overlapping so the
reference's unstable sort order cannot change the outcome."""
import math

import numpy as np
import pytest

from victoriametrics_amd import transform as tf
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

from test_binop_setop_fuzz import _marshal

NAN = math.nan
N = 6


def _copy_zero(src, le_str):
    ts = src.copy_shallow()
    ts.values[:] = 0.0
    ts.mn.remove_tag("le")
    ts.mn.add_tag("le", le_str)
    return ts


def _is_zero(ts):
    return not np.any(ts.values > 0)


def _model(series):
    rvs = []
    groups, order = {}, []
    for s in series:
        vr = s.mn.get_tag_value("vmrange")
        if not vr:
            if s.mn.get_tag_value("le"):
                rvs.append(s)
            continue
        txt = vr.decode()
        n = txt.find("...")
        if n < 0:
            continue
        ss, es = txt[:n], txt[n + 3:]
        try:
            start, end = float(ss), float(es)
        except ValueError:
            continue
        s.mn.remove_tag("le")
        s.mn.remove_tag("vmrange")
        k = _marshal(s.mn)
        if k not in groups:
            groups[k] = []
            order.append(k)
        groups[k].append([ss, es, start, end, s])
    for k in order:
        xss = sorted(groups[k], key=lambda x: x[3])
        xss_new = []
        # Go's zero-value xsPrev has end == 0.0: a first bucket starting
        # at exactly 0 gets NO le=0 fill (transform.go:585)
        prev = None
        prev_end = 0.0
        uniq = {}
        for xs in xss:
            ss, es, start, end, ts = xs
            if _is_zero(ts):
                continue
            if start != prev_end:
                if uniq.get(ss) is None:
                    uniq[ss] = ts
                    xss_new.append([None, ss, None, start,
                                    _copy_zero(ts, ss)])
            ts.mn.add_tag("le", es)
            prev_ts = uniq.get(es)
            if prev_ts is not None:
                # mergeNonOverlapping (tolerates <=2 overlaps)
                overlaps = int(np.sum(~np.isnan(ts.values) &
                                      ~np.isnan(prev_ts.values)))
                ok = not (overlaps > 2 or
                          (len(ts.values) <= 2 and
                           len(prev_ts.values) <= 2))
                if ok:
                    fill = ~np.isnan(ts.values)
                    prev_ts.values[fill] = ts.values[fill]
            else:
                xss_new.append(xs)
                uniq[es] = ts
            prev = xs
            prev_end = end
        if prev is not None and not math.isinf(prev[3]) and \
                not _is_zero(prev[4]):
            xss_new.append([None, "+Inf", None, math.inf,
                            _copy_zero(prev[4], "+Inf")])
        if not xss_new:
            continue
        for i in range(N):
            count = 0.0
            for xs in xss_new:
                ts = xs[4]
                v = ts.values[i]
                if v == v and v > 0:
                    count += v
                ts.values[i] = count
        rvs.extend(xs[4] for xs in xss_new)
    return rvs


def _fp(tss):
    out = []
    for s in tss:
        vals = tuple(-0.0 if v != v else float(v) for v in s.values)
        out.append((_marshal(s.mn), vals))
    return sorted(out)


def _rand_buckets(rng):
    series = []
    n_groups = int(rng.integers(1, 3))
    for g in range(n_groups):
        tags = [("g", str(g))]
        bounds = sorted(rng.choice(
            [0.0, 0.1, 0.5, 1.0, 2.0, 5.0, 10.0, 50.0],
            size=int(rng.integers(2, 6)), replace=False))
        segs = list(zip(bounds[:-1], bounds[1:]))
        used_ends = set()
        for lo, hi in segs:
            if rng.random() < 0.25:
                continue  # gap
            v = rng.uniform(0, 20, N)
            v[rng.random(N) < 0.2] = 0.0
            if rng.random() < 0.2:
                v[:] = 0.0  # zero bucket
            vr = f"{lo:g}...{hi:g}"
            series.append(Series(
                MetricName("m", tags + [("vmrange", vr)]), v))
            if hi in used_ends:
                pass
            used_ends.add(hi)
        # occasional duplicate-end bucket, NON-overlapping values
        if segs and rng.random() < 0.3:
            lo, hi = segs[-1]
            v = np.full(N, NAN)
            series.append(Series(
                MetricName("m", tags + [("vmrange", f"{lo:g}...{hi:g}")]),
                v))
        # occasional +Inf terminal bucket
        if rng.random() < 0.4:
            v = rng.uniform(0, 5, N)
            series.append(Series(
                MetricName("m", tags + [("vmrange",
                                         f"{bounds[-1]:g}...+Inf")]), v))
        # malformed / le-passthrough noise
        if rng.random() < 0.3:
            series.append(Series(
                MetricName("m", tags + [("vmrange", "foobar")]),
                np.ones(N)))
        if rng.random() < 0.3:
            series.append(Series(
                MetricName("m", tags + [("le", "0.25")]),
                np.full(N, 7.0)))
    return series


@pytest.mark.parametrize("seed", range(3))
def test_vmrange_buckets_to_le_matches_model(seed):
    rng = np.random.default_rng(8800 + seed)
    for it in range(100):
        series = _rand_buckets(rng)
        s2 = [s.copy_shallow() for s in series]
        want = _fp(_model(s2))
        got = _fp(tf.vmrange_buckets_to_le(series))
        assert got == want, f"seed={seed} it={it}\n got={got}\nwant={want}"
