"""Model-based fuzz of the arithmetic/comparison matching layer
(adjustBinaryOpTags + groupJoin + ensureSingleTimeseries +
newBinaryOpFunc, binary_op.go:162-470) against a literal per-point
restatement.  Covers the scalar fast paths, on/ignoring keys,
group_left/group_right joins (tag copy, `*`, prefix, right-duplicate
merge), fill()/fill_left/fill_right, keep_metric_names, bool cmp and
drop_nan_right — 400 random scenarios must agree as multisets, including
which scenarios raise the duplicate-series error."""
import math

import numpy as np
import pytest

from victoriametrics_amd.binary_op import (BinOpSpec, DuplicateSeriesError,
                                           Series, remove_empty_series)
from victoriametrics_amd.metric_name import MetricName

from test_binary_op import _eval
from test_binop_setop_fuzz import (_fingerprint, _group_key, _is_scalar,
                                   _marshal, _rand_series_set)

NAN = math.nan
N = 6
CMP = {"==", "!=", ">", "<", ">=", "<="}


def _go_op(op, a, b):
    # IEEE semantics via numpy scalars (Go float64 ops == C == IEEE-754)
    with np.errstate(all="ignore"):
        if op == "+":
            return float(np.float64(a) + np.float64(b))
        if op == "-":
            return float(np.float64(a) - np.float64(b))
        if op == "*":
            return float(np.float64(a) * np.float64(b))
        if op == "/":
            return float(np.float64(a) / np.float64(b))
        if op == "%":
            return float(np.fmod(np.float64(a), np.float64(b)))
        if op == "^":
            # binaryop.Pow (funcs.go:78): NaN^any = NaN (issue 7359,
            # overriding IEEE pow(NaN,0)=1), then C libm pow; math.pow IS
            # libm — fall back on Python's domain errors
            if a != a:
                return NAN
            try:
                return math.pow(a, b)
            except (ValueError, OverflowError):
                if b != b:
                    return NAN
                return NAN if a < 0 else math.inf
        if op == "atan2":
            return math.atan2(a, b)  # libm, matches the oracle bitwise
    raise AssertionError(op)


def _cmp(op, a, b):
    return {"==": a == b, "!=": a != b, ">": a > b, "<": a < b,
            ">=": a >= b, "<=": a <= b}[op]


def _bf(spec, a, b):
    if spec.op in CMP:
        if not spec.bool_modifier:
            return a if _cmp(spec.op, a, b) else NAN
        if a != a:
            return NAN
        return 1.0 if _cmp(spec.op, a, b) else 0.0
    return _go_op(spec.op, a, b)


def _reset_group_if_required(spec, s):
    if spec.op in CMP and not spec.bool_modifier:
        return
    if spec.keep_metric_names:
        return
    s.mn.reset_metric_group()


def _new_fill(spec, src):
    ts = src.copy_shallow()
    if not spec.keep_metric_names:
        ts.mn.reset_metric_group()
    if spec.group_op == "on":
        ts.mn.remove_tags_on(list(spec.group_tags))
    else:
        ts.mn.remove_tags_ignoring(list(spec.group_tags))
    ts.values[:] = NAN
    return ts


def _merge_non_overlapping(dst, src):
    overlaps = int(np.sum(~np.isnan(src.values) & ~np.isnan(dst.values)))
    if overlaps > 2:
        return False
    if len(src.values) <= 2 and len(dst.values) <= 2:
        return False
    fill = ~np.isnan(src.values)
    dst.values[fill] = src.values[fill]
    return True


def _ensure_single(tss):
    while len(tss) > 1:
        if not _merge_non_overlapping(tss[0], tss[-1]):
            raise DuplicateSeriesError("model duplicate")
        tss = tss[:-1]
    return tss


def _group_join(spec, rvs_one, rvs_many, tss_many, tss_one):
    """groupJoin (binary_op.go:409): tss_many is the many-side whose names
    take the join tags from tss_one members."""
    skip = list(spec.group_tags) if spec.group_op == "on" else []
    for ts_m in tss_many:
        _reset_group_if_required(spec, ts_m)
        if len(tss_one) == 1:
            ts_m.mn.set_tags(list(spec.join_tags), spec.join_prefix, skip,
                             tss_one[0].mn)
            rvs_many.append(ts_m)
            rvs_one.append(tss_one[0])
            continue
        m = {}
        order = []
        for ts_o in tss_one:
            ts_copy = ts_m.copy_shallow()
            ts_copy.mn.set_tags(list(spec.join_tags), spec.join_prefix,
                                skip, ts_o.mn)
            k = _marshal(ts_copy.mn)
            pair = m.get(k)
            if pair is None:
                m[k] = [ts_copy, ts_o]
                order.append(k)
                continue
            tmp = pair[1].copy_shallow()
            if not _merge_non_overlapping(tmp, ts_o):
                raise DuplicateSeriesError("model duplicate join")
            pair[1] = tmp
        for k in order:
            rvs_many.append(m[k][0])
            rvs_one.append(m[k][1])


def _adjust(spec, left, right):
    if not spec.group_op and not spec.join_op:
        if _is_scalar(left):
            for s in right:
                _reset_group_if_required(spec, s)
            return [left[0]] * len(right), right, right
        if _is_scalar(right):
            for s in left:
                _reset_group_if_required(spec, s)
            return left, [right[0]] * len(left), left
    m_left, m_right = {}, {}
    group_tags = list(spec.group_tags)
    group_op = spec.group_op or "ignoring"
    if spec.keep_metric_names and group_op == "on":
        group_tags = group_tags + ["__name__"]
    for s in left:
        m_left.setdefault(_group_key(spec, s), []).append(s)
    for s in right:
        m_right.setdefault(_group_key(spec, s), []).append(s)
    if spec.fill_left is not None:
        for k in m_right:
            m_left.setdefault(k, [])
    rvs_left, rvs_right = [], []
    for k, tss_left in m_left.items():
        tss_right = m_right.get(k, [])
        if not tss_left:
            tss_left = [_new_fill(spec, tss_right[0])]
        if not tss_right:
            if spec.fill_right is None:
                continue
            tss_right = [_new_fill(spec, tss_left[0])]
        if spec.join_op == "group_left":
            _group_join(spec, rvs_right, rvs_left, tss_left, tss_right)
        elif spec.join_op == "group_right":
            _group_join(spec, rvs_left, rvs_right, tss_right, tss_left)
        else:
            tss_left = _ensure_single(tss_left)
            tss_right = _ensure_single(tss_right)
            ts_left = tss_left[0]
            _reset_group_if_required(spec, ts_left)
            if group_op == "on":
                ts_left.mn.remove_tags_on(group_tags)
            else:
                ts_left.mn.remove_tags_ignoring(group_tags)
            rvs_left.append(ts_left)
            rvs_right.append(tss_right[0])
    dst = rvs_left if spec.join_op != "group_right" else rvs_right
    return rvs_left, rvs_right, dst


def _model(spec, left, right, drop_nan_right):
    if spec.op not in CMP:
        left = remove_empty_series(left)
        right = remove_empty_series(right)
    if not left and not right:
        return []
    if not left and spec.fill_left is None:
        return []
    if not right and spec.fill_right is None:
        return []
    L, R, D = _adjust(spec, left, right)
    out = []
    for ts_l, ts_r, ts_d in zip(L, R, D):
        vals = np.empty(N)
        for j in range(N):
            a, b = float(ts_l.values[j]), float(ts_r.values[j])
            lnan, rnan = a != a, b != b
            if lnan and rnan:
                vals[j] = _bf(spec, a, b)
                continue
            if drop_nan_right and rnan and spec.fill_right is None:
                vals[j] = NAN
                continue
            if lnan and spec.fill_left is not None:
                a = spec.fill_left
            if rnan and spec.fill_right is not None:
                b = spec.fill_right
            vals[j] = _bf(spec, a, b)
        out.append(Series(ts_d.mn.copy(), vals))
    return out


def _rand_spec(rng):
    ops = ["+", "-", "*", "/", "%", "^", "atan2",
           "==", "!=", ">", "<", ">=", "<="]
    op = ops[int(rng.integers(0, len(ops)))]
    group_op = str(rng.choice(["", "on", "ignoring"]))
    group_tags = list(rng.choice(["a", "b", "c"],
                                 size=int(rng.integers(0, 3)),
                                 replace=False))
    join_op, join_tags, prefix = "", [], ""
    if group_op and rng.random() < 0.5:
        join_op = str(rng.choice(["group_left", "group_right"]))
        r = rng.random()
        if r < 0.3:
            join_tags = ["*"]
            if rng.random() < 0.5:
                prefix = "p_"
        elif r < 0.7:
            join_tags = list(rng.choice(["a", "b", "c", "__name__"],
                                        size=int(rng.integers(1, 3)),
                                        replace=False))
    fl = 3.0 if rng.random() < 0.25 else None
    fr = 5.0 if rng.random() < 0.25 else None
    return BinOpSpec(op, bool_modifier=bool(rng.random() < 0.4),
                     group_op=group_op, group_tags=group_tags,
                     join_op=join_op, join_tags=join_tags,
                     join_prefix=prefix,
                     keep_metric_names=bool(rng.random() < 0.25),
                     fill_left=fl, fill_right=fr)


@pytest.mark.parametrize("seed", range(6))
def test_adjust_and_apply_match_reference_model(seed):
    rng = np.random.default_rng(7000 + seed)
    agree = raised = 0
    for it in range(100):
        spec = _rand_spec(rng)
        dnr = bool(rng.random() < 0.3) and spec.op in CMP
        left = _rand_series_set(rng, int(rng.integers(0, 5)))
        right = _rand_series_set(rng, int(rng.integers(0, 5)))
        # quantize some values so == / != hit real equalities
        for s in left + right:
            mask = rng.random(N) < 0.5
            s.values[mask] = np.round(s.values[mask] / 10) * 10
        l2 = [s.copy_shallow() for s in left]
        r2 = [s.copy_shallow() for s in right]
        spec2 = BinOpSpec(spec.op, bool_modifier=spec.bool_modifier,
                          group_op=spec.group_op,
                          group_tags=list(spec.group_tags),
                          join_op=spec.join_op,
                          join_tags=list(spec.join_tags),
                          join_prefix=spec.join_prefix,
                          keep_metric_names=spec.keep_metric_names,
                          fill_left=spec.fill_left,
                          fill_right=spec.fill_right)
        try:
            want = _fingerprint(_model(spec, l2, r2, dnr))
            want_err = None
        except DuplicateSeriesError:
            want, want_err = None, True
        try:
            got = _fingerprint(_eval(spec2, left, right,
                                     drop_nan_right=dnr))
            got_err = None
        except DuplicateSeriesError:
            got, got_err = None, True
        ctx = (f"seed={seed} it={it} op={spec.op} "
               f"bool={spec.bool_modifier} {spec.group_op}"
               f"({spec.group_tags}) {spec.join_op}({spec.join_tags}) "
               f"prefix={spec.join_prefix!r} kmn={spec.keep_metric_names} "
               f"fl={spec.fill_left} fr={spec.fill_right} dnr={dnr}")
        assert want_err == got_err, ctx
        if want_err:
            raised += 1
            continue
        assert got == want, f"{ctx}\n got={got}\nwant={want}"
        agree += 1
    assert agree >= 50  # the bulk of scenarios must be non-error
