"""Model-based fuzz of the set-op walks (and/or/unless/if/ifnot/default)
against a literal restatement of the reference's per-point algorithms
(binary_op.go:521-800: createTimeseriesMapByTagSet, seriesByKey,
addRightNaNsToLeft, addLeftNaNsIfNoRightNaNs, fillLeftNaNsWithRightValues,
fillLeftNaNsWithRightValuesOrMerge).  The model is naive loops; the
implementation under test batches the same semantics through the oracle
mask/or kernels — 400 random (series set × op × modifier) scenarios must
agree as multisets of (metric name, values)."""
import math

import numpy as np
import pytest

from victoriametrics_amd.binary_op import (BinOpSpec, Series,
                                           remove_empty_series)
from victoriametrics_amd.metric_name import MetricName

from test_binary_op import _eval

NAN = math.nan
N = 6


def _marshal(mn):
    return (bytes(mn.metric_group), tuple(sorted(mn.tags)))


def _is_scalar(tss):
    return (len(tss) == 1 and not tss[0].mn.metric_group and
            not tss[0].mn.tags)


def _group_key(spec, s):
    mn = s.mn.copy()
    if not spec.keep_metric_names:
        mn.reset_metric_group()
    if spec.group_op == "on":
        mn.remove_tags_on(list(spec.group_tags))
    else:
        mn.remove_tags_ignoring(list(spec.group_tags))
    return _marshal(mn)


def _tag_map(spec, tss):
    m = {}
    for s in tss:
        m.setdefault(_group_key(spec, s), []).append(s)
    return m


def _series_by_key(m, k):
    tss = m.get(k)
    if tss is not None:
        return tss
    if len(m) != 1:
        return None
    only = next(iter(m.values()))
    return only if _is_scalar(only) else None


def _add_right_nans_to_left(tss_left, tss_right):
    for sl in tss_left:
        for i in range(N):
            if not any(not math.isnan(sr.values[i]) for sr in tss_right):
                sl.values[i] = NAN
    return remove_empty_series(tss_left)


def _add_left_nans_if_no_right_nans(tss_left, tss_right):
    for sl in tss_left:
        for i in range(N):
            if any(not math.isnan(sr.values[i]) for sr in tss_right):
                sl.values[i] = NAN
    return remove_empty_series(tss_left)


def _fill_left_nans(tss_left, tss_right):
    for sl in tss_left:
        for i in range(N):
            if math.isnan(sl.values[i]):
                for sr in tss_right:
                    if not math.isnan(sr.values[i]):
                        sl.values[i] = sr.values[i]
                        break


def _fill_or_merge(tss_left, tss_right):
    # fillLeftNaNsWithRightValuesOrMerge (binary_op.go:647)
    if _is_scalar(tss_right):
        can = _is_scalar(tss_left)
        vr = tss_right[0].values
        for sl in tss_left:
            for i in range(N):
                left_nan = math.isnan(sl.values[i])
                if left_nan and can:
                    sl.values[i] = vr[i]
                if not left_nan or can:
                    vr[i] = NAN
        return
    for sl in tss_left:
        nl = _marshal(sl.mn)
        for i in range(N):
            left_nan = math.isnan(sl.values[i])
            for sr in tss_right:
                can = _marshal(sr.mn) == nl
                if left_nan and can:
                    sl.values[i] = sr.values[i]
                if not left_nan or can:
                    sr.values[i] = NAN


def _model(spec, left, right):
    m_left = _tag_map(spec, left)
    m_right = _tag_map(spec, right)
    rvs = []
    op = spec.op
    if op == "and":
        for k, tss_right in m_right.items():
            tss_left = m_left.get(k)
            if tss_left is None:
                continue
            rvs.extend(_add_right_nans_to_left(tss_left, tss_right))
    elif op == "if":
        for k, tss_left in m_left.items():
            tss_right = _series_by_key(m_right, k)
            if tss_right is None:
                continue
            rvs.extend(_add_right_nans_to_left(tss_left, tss_right))
    elif op == "ifnot":
        for k, tss_left in m_left.items():
            tss_right = _series_by_key(m_right, k)
            if tss_right is None:
                rvs.extend(tss_left)
                continue
            rvs.extend(_add_left_nans_if_no_right_nans(tss_left,
                                                       tss_right))
    elif op == "unless":
        for k, tss_left in m_left.items():
            tss_right = m_right.get(k)
            if tss_right is None:
                rvs.extend(tss_left)
                continue
            rvs.extend(_add_left_nans_if_no_right_nans(tss_left,
                                                       tss_right))
    elif op == "default":
        if not m_left:
            for tss in m_right.values():
                rvs.extend(tss)
            return rvs
        for k, tss_left in m_left.items():
            rvs.extend(tss_left)
            tss_right = _series_by_key(m_right, k)
            if tss_right is None:
                continue
            _fill_left_nans(tss_left, tss_right)
    elif op == "or":
        from victoriametrics_amd.binary_op import \
            sort_series_by_metric_name
        for k in list(m_left):
            m_left[k] = remove_empty_series(m_left[k])
            rvs.extend(m_left[k])
        # issue 5393: left block sorted, appended right block sorted
        sort_series_by_metric_name(rvs)
        n_before = len(rvs)
        for k, tss_right in m_right.items():
            tss_left = m_left.get(k)
            if not tss_left:
                rvs.extend(tss_right)
                continue
            _fill_or_merge(tss_left, tss_right)
            rvs.extend(remove_empty_series(tss_right))
        tail = rvs[n_before:]
        sort_series_by_metric_name(tail)
        rvs[n_before:] = tail
    else:
        raise AssertionError(op)
    return rvs


def _fingerprint(tss):
    out = []
    for s in tss:
        vals = tuple(-0.0 if v != v else float(v) for v in s.values)
        out.append((_marshal(s.mn), vals))
    return sorted(out)


def _rand_series_set(rng, n):
    tss = []
    for _ in range(n):
        name = rng.choice(["", "m1"])
        tags = []
        for key in ("a", "b", "c"):
            r = rng.random()
            if r < 0.5:
                tags.append((key, rng.choice(["x", "y"])))
        v = rng.uniform(1, 100, N)
        v[rng.random(N) < 0.35] = NAN
        tss.append(Series(MetricName(name, tags), v))
    return tss


@pytest.mark.parametrize("seed", range(6))
def test_setop_walks_match_reference_model(seed):
    rng = np.random.default_rng(1000 + seed)
    ops = ["and", "or", "unless", "if", "ifnot", "default"]
    for it in range(100):
        op = ops[it % len(ops)]
        group_op = rng.choice(["", "on", "ignoring"])
        group_tags = list(rng.choice(["a", "b", "c"],
                                     size=rng.integers(0, 3),
                                     replace=False))
        kmn = bool(rng.random() < 0.3)
        spec = BinOpSpec(op, group_op=group_op, group_tags=group_tags,
                         keep_metric_names=kmn)
        left = _rand_series_set(rng, int(rng.integers(0, 5)))
        right = _rand_series_set(rng, int(rng.integers(0, 5)))
        left2 = [s.copy_shallow() for s in left]
        right2 = [s.copy_shallow() for s in right]
        model_out = _model(spec, left2, right2)
        spec2 = BinOpSpec(op, group_op=group_op, group_tags=group_tags,
                          keep_metric_names=kmn)
        impl_out = _eval(spec2, left, right)
        if op == "or":
            # `or` output ORDER is part of the contract (exec excludes it
            # from the final sort): compare as sequences
            want = [f for f in map(lambda s: _fingerprint([s])[0],
                                   model_out)]
            got = [f for f in map(lambda s: _fingerprint([s])[0],
                                  impl_out)]
        else:
            want = _fingerprint(model_out)
            got = _fingerprint(impl_out)
        assert got == want, (
            f"seed={seed} it={it} op={op} group_op={group_op} "
            f"tags={group_tags} kmn={kmn}\n got={got}\nwant={want}")
