"""Model-based fuzz of aggrPrepareSeries (aggr.go:96-150) — the grouping
layer every aggregate rides on: removeGroupTags (by => RemoveTagsOn +
keep group only when __name__ listed; without => RemoveTagsIgnoring +
ResetMetricGroup), empty-series removal first, group order by first
appearance, the max_series group cap, and keep_original member naming.
300 random scenarios vs a naive restatement."""
import math

import numpy as np
import pytest

from victoriametrics_amd import aggregate as agg
from victoriametrics_amd.binary_op import Series
from victoriametrics_amd.metric_name import MetricName

from test_binop_setop_fuzz import _marshal, _rand_series_set

NAN = math.nan


def _model_key(s, op, args):
    mn = s.mn.copy()
    if op in ("", "by"):
        # RemoveTagsOn: keep only the named tags; the group survives only
        # when __name__ is in the list (metric_name.go RemoveTagsOn)
        keep = {MetricName._b(a) for a in args}
        if b"__name__" not in keep:
            mn.reset_metric_group()
        mn.tags = [(k, v) for k, v in mn.tags if k in keep]
    elif op == "without":
        drop = {MetricName._b(a) for a in args}
        mn.tags = [(k, v) for k, v in mn.tags if k not in drop]
        mn.reset_metric_group()
    else:
        raise AssertionError(op)
    return mn


def _model(series, op, args, max_series=0, keep_original=False):
    series = [s for s in series if not np.isnan(s.values).all()]
    order, groups = [], {}
    for s in series:
        mn = _model_key(s, op, args)
        k = _marshal(mn)
        if k not in groups:
            if max_series > 0 and len(groups) >= max_series:
                continue
            groups[k] = (mn, [])
            order.append(k)
        groups[k][1].append(s)
    return [groups[k] for k in order]


@pytest.mark.parametrize("seed", range(3))
def test_prepare_series_matches_model(seed):
    rng = np.random.default_rng(4200 + seed)
    for it in range(100):
        op = str(rng.choice(["", "by", "without"]))
        args = list(rng.choice(["a", "b", "c", "__name__"],
                               size=int(rng.integers(0, 3)),
                               replace=False))
        max_series = int(rng.integers(0, 3))
        keep_original = bool(rng.random() < 0.4)
        series = _rand_series_set(rng, int(rng.integers(0, 7)))
        # make some all-NaN series so removeEmptySeries matters
        for s in series:
            if rng.random() < 0.15:
                s.values[:] = NAN
        s2 = [s.copy_shallow() for s in series]
        want = _model(s2, op, args, max_series, keep_original)
        got = agg.prepare_series(series, op, args, max_series,
                                 keep_original=keep_original)
        ctx = f"seed={seed} it={it} op={op!r} args={args} max={max_series}"
        assert len(got) == len(want), ctx
        for (gmn_g, mem_g), (gmn_w, mem_w) in zip(got, want):
            assert _marshal(gmn_g) == _marshal(gmn_w), ctx
            assert len(mem_g) == len(mem_w), ctx
            for a, b in zip(mem_g, mem_w):
                np.testing.assert_array_equal(
                    np.nan_to_num(a.values, nan=-1),
                    np.nan_to_num(b.values, nan=-1), err_msg=ctx)
                if keep_original:
                    # members keep their ORIGINAL names
                    assert _marshal(a.mn) == _marshal(b.mn), ctx
