"""Binary operators on the result grid — host mirror of
app/vmselect/promql/binary_op.go.

Label matching (createTimeseriesMapByTagSet :759, adjustBinaryOpTags :271,
groupJoin :404, ensureSingleTimeseries :392) is host metadata work, mirrored
here; the per-point arithmetic runs on the GPU through the C-ABI
(vmgpu_binop_eval / vmgpu_binop_mask / vmgpu_binop_or, csrc/binop.hip).

Series are (MetricName, values) pairs on a shared grid; timestamps are
carried by the evaluation context as in the reference (timeseries share one
timestamps slice, timeseries.go:25).

Divergences from the reference, by design:
- Output order: Go map iteration order is randomized, so the reference's
  output order is unspecified; this mirror iterates keys in first-seen
  order (deterministic superset of the allowed behaviours).
- The `q == (union)` / `q != (union)` special cases (binary_op.go:58-120)
  need the expression AST and live in the (host Go) caller; callers pass
  plain series lists here.

Tests pass apply_fn/mask_fn/or_fn backed by the CPU oracle; the default is
the GPU path.
"""
import math

import numpy as np

from . import engine
from .metric_name import MetricName


OP_IDS = {
    "+": 0, "-": 1, "*": 2, "/": 3, "%": 4, "^": 5, "atan2": 6,
    "==": 7, "!=": 8, ">": 9, "<": 10, ">=": 11, "<=": 12,
    "default": 13, "if": 14, "ifnot": 15, "and": 16, "or": 17,
}
CMP_OPS = {"==", "!=", ">", "<", ">=", "<="}
SET_OPS = {"and", "or", "unless", "if", "ifnot", "default"}


class Series:
    """One result series: MetricName + values on the shared grid."""
    __slots__ = ("mn", "values")

    def __init__(self, mn, values):
        self.mn = mn if isinstance(mn, MetricName) else MetricName(mn)
        self.values = np.ascontiguousarray(values, dtype=np.float64)

    def copy_shallow(self):
        # CopyFromShallowTimestamps (timeseries.go:51): values are copied,
        # the name is deep-copied
        return Series(self.mn.copy(), self.values.copy())


class BinOpSpec:
    """The BinaryOpExpr fields the evaluator consumes (metricsql AST)."""

    def __init__(self, op, bool_modifier=False, group_op="", group_tags=(),
                 join_op="", join_tags=(), join_prefix="",
                 keep_metric_names=False, fill_left=None, fill_right=None):
        self.op = op.lower()
        self.bool_modifier = bool_modifier
        self.group_op = group_op.lower()
        self.group_tags = list(group_tags)
        self.join_op = join_op.lower()
        self.join_tags = list(join_tags)
        self.join_prefix = join_prefix
        self.keep_metric_names = keep_metric_names
        self.fill_left = fill_left
        self.fill_right = fill_right


class DuplicateSeriesError(ValueError):
    pass


def remove_empty_series(tss):
    return [t for t in tss if not np.all(np.isnan(t.values))]


def is_scalar(tss):
    # isScalar (binary_op.go:799)
    return (len(tss) == 1 and not tss[0].mn.metric_group and
            not tss[0].mn.tags)


def sort_series_by_metric_name(tss):
    # sortSeriesByMetricName/metricNameLess (exec.go:162-192); tags sorted
    # first as the reference requires of its callers
    def key(t):
        return (t.mn.metric_group, sorted(t.mn.tags))
    tss.sort(key=key)


def _reset_metric_group_if_required(spec, ts):
    # resetMetricGroupIfRequired (binary_op.go:508)
    if spec.op in CMP_OPS and not spec.bool_modifier:
        return
    if spec.keep_metric_names:
        return
    ts.mn.reset_metric_group()


def _group_key(spec, ts):
    mn = ts.mn.copy()
    if not spec.keep_metric_names:
        mn.reset_metric_group()
    if spec.group_op == "on":
        mn.remove_tags_on(spec.group_tags)
    else:
        mn.remove_tags_ignoring(spec.group_tags)
    return mn.marshal_sorted()


def create_map_by_tag_set(spec, left, right):
    # createTimeseriesMapByTagSet (binary_op.go:759)
    def tags_map(arg):
        m = {}
        for ts in arg:
            m.setdefault(_group_key(spec, ts), []).append(ts)
        return m
    return tags_map(left), tags_map(right)


def merge_non_overlapping(dst, src):
    # mergeNonOverlappingTimeseries (binary_op.go:473)
    sv, dv = src.values, dst.values
    overlaps = int(np.sum(~np.isnan(sv) & ~np.isnan(dv)))
    if overlaps > 2:
        return False
    if len(sv) <= 2 and len(dv) <= 2:
        return False
    fill = ~np.isnan(sv)
    dv[fill] = sv[fill]
    return True


def _ensure_single(side, spec, tss):
    # ensureSingleTimeseries (binary_op.go:392)
    while len(tss) > 1:
        if not merge_non_overlapping(tss[0], tss[-1]):
            raise DuplicateSeriesError(
                f"duplicate time series on the {side} side of "
                f"{spec.op} {spec.group_op}({','.join(spec.group_tags)}): "
                f"{tss[0].mn} and {tss[-1].mn}")
        tss = tss[:-1]
    return tss


def _new_fill_series(spec, src):
    # newFillTimeseries (binary_op.go:376)
    ts = src.copy_shallow()
    if not spec.keep_metric_names:
        ts.mn.reset_metric_group()
    if spec.group_op == "on":
        ts.mn.remove_tags_on(spec.group_tags)
    else:
        ts.mn.remove_tags_ignoring(spec.group_tags)
    ts.values[:] = math.nan
    return ts


def _group_join(single_side, spec, rvs_left, rvs_right, tss_left, tss_right):
    # groupJoin (binary_op.go:404)
    join_tags = spec.join_tags
    skip_tags = spec.group_tags if spec.group_op == "on" else []
    prefix = spec.join_prefix
    for ts_left in tss_left:
        _reset_metric_group_if_required(spec, ts_left)
        if len(tss_right) == 1:
            ts_left.mn.set_tags(join_tags, prefix, skip_tags, tss_right[0].mn)
            rvs_left.append(ts_left)
            rvs_right.append(tss_right[0])
            continue
        m = {}
        for ts_right in tss_right:
            ts_copy = ts_left.copy_shallow()
            ts_copy.mn.set_tags(join_tags, prefix, skip_tags, ts_right.mn)
            k = ts_copy.mn.marshal_sorted()
            pair = m.get(k)
            if pair is None:
                m[k] = [ts_copy, ts_right]
                continue
            tmp = pair[1].copy_shallow()
            if not merge_non_overlapping(tmp, ts_right):
                raise DuplicateSeriesError(
                    f"duplicate time series on the {single_side} side of "
                    f"`{spec.op} {spec.group_op}({','.join(spec.group_tags)}) "
                    f"{spec.join_op}({','.join(join_tags)})`: "
                    f"{tmp.mn} and {ts_right.mn}")
            pair[1] = tmp
        for pair in m.values():
            rvs_left.append(pair[0])
            rvs_right.append(pair[1])
    return rvs_left, rvs_right


def adjust_binary_op_tags(spec, left, right):
    # adjustBinaryOpTags (binary_op.go:271): returns (left, right, dst)
    if not spec.group_op and not spec.join_op:
        if is_scalar(left):
            ts_left = left[0]
            for ts_right in right:
                _reset_metric_group_if_required(spec, ts_right)
            return [ts_left] * len(right), right, right
        if is_scalar(right):
            ts_right = right[0]
            for ts_left in left:
                _reset_metric_group_if_required(spec, ts_left)
            return left, [ts_right] * len(left), left

    rvs_left, rvs_right = [], []
    m_left, m_right = create_map_by_tag_set(spec, left, right)
    group_op = spec.group_op or "ignoring"
    group_tags = spec.group_tags
    if spec.keep_metric_names and group_op == "on":
        group_tags = group_tags + ["__name__"]
    if spec.fill_left is not None:
        for k in m_right:
            m_left.setdefault(k, [])
    for k, tss_left in m_left.items():
        tss_right = m_right.get(k, [])
        if not tss_left:
            tss_left = [_new_fill_series(spec, tss_right[0])]
        if not tss_right:
            if spec.fill_right is None:
                continue
            tss_right = [_new_fill_series(spec, tss_left[0])]
        if spec.join_op == "group_left":
            _group_join("right", spec, rvs_left, rvs_right, tss_left, tss_right)
        elif spec.join_op == "group_right":
            _group_join("left", spec, rvs_right, rvs_left, tss_right, tss_left)
        else:
            tss_left = _ensure_single("left", spec, tss_left)
            tss_right = _ensure_single("right", spec, tss_right)
            ts_left = tss_left[0]
            _reset_metric_group_if_required(spec, ts_left)
            if group_op == "on":
                ts_left.mn.remove_tags_on(group_tags)
            else:
                ts_left.mn.remove_tags_ignoring(group_tags)
            rvs_left.append(ts_left)
            rvs_right.append(tss_right[0])
    dst = rvs_left if spec.join_op != "group_right" else rvs_right
    return rvs_left, rvs_right, dst


def _series_by_key(m, key):
    # seriesByKey (binary_op.go:741): exact key, else the lone scalar
    tss = m.get(key)
    if tss is not None:
        return tss
    if len(m) != 1:
        return None
    only = next(iter(m.values()))
    return only if is_scalar(only) else None


def _gpu_apply(spec, left, right, dst, drop_nan_right):
    rows_l, idx_of_l = [], {}
    rows_r, idx_of_r = [], {}
    li = np.empty(len(left), dtype=np.uint32)
    ri = np.empty(len(left), dtype=np.uint32)
    for p, (tl, tr) in enumerate(zip(left, right)):
        if id(tl) not in idx_of_l:
            idx_of_l[id(tl)] = len(rows_l)
            rows_l.append(tl.values)
        if id(tr) not in idx_of_r:
            idx_of_r[id(tr)] = len(rows_r)
            rows_r.append(tr.values)
        li[p] = idx_of_l[id(tl)]
        ri[p] = idx_of_r[id(tr)]
    out = engine.binop_eval(
        OP_IDS[spec.op], spec.bool_modifier, drop_nan_right,
        np.stack(rows_l), li, np.stack(rows_r), ri,
        spec.fill_left, spec.fill_right)
    for p, ts in enumerate(dst):
        ts.values = out[p]
    return dst


def union_list_cmp(op, left, right, union_on_left=False):
    """binaryOpEqFunc / binaryOpNeqFunc's union-list special case
    (binary_op.go:55-113): `q == (v1,...,vN)` keeps each left point only
    when SOME union member holds that exact value there; `!=` keeps it
    when NO member does.  The union side may be either operand
    (union_on_left swaps).  Host-side mask: the list is tiny (parsed
    literals), the data side stays resident."""
    if union_on_left:
        left, right = right, left
    if op == "==":
        if not left or not right:
            return []
    else:
        if not left:
            return []
        if not right:
            return left
    rv = np.stack([np.asarray(s.values, np.float64) for s in right])
    for s in left:
        hit = (rv == np.asarray(s.values, np.float64)).any(axis=0)
        if op == "==":
            s.values[~hit] = math.nan
        else:
            s.values[hit] = math.nan
    return left


def binary_op_eval(spec, left, right, drop_nan_right=False, apply_fn=None,
                   mask_fn=None, or_fn=None, union_list=None):
    """binaryOpFuncs dispatch (binary_op.go:15).  apply_fn/mask_fn/or_fn
    override the GPU kernels (tests only — oracle-backed).  union_list:
    "left"/"right" marks that operand as a scalar-list union expression,
    routing == / != through union_list_cmp (binary_op.go:55-113)."""
    if union_list and spec.op in ("==", "!="):
        return union_list_cmp(spec.op, left, right,
                              union_on_left=(union_list == "left"))
    if spec.op in SET_OPS:
        return _set_op(spec, left, right, mask_fn, or_fn)

    # newBinaryOpFunc (binary_op.go:162)
    if spec.op not in CMP_OPS:
        left = remove_empty_series(left)
        right = remove_empty_series(right)
    if not left and not right:
        return []
    if not left and spec.fill_left is None:
        return []
    if not right and spec.fill_right is None:
        return []
    left, right, dst = adjust_binary_op_tags(spec, left, right)
    if not left:
        return []
    if apply_fn is not None:
        return apply_fn(spec, left, right, dst, drop_nan_right)
    return _gpu_apply(spec, left, right, dst, drop_nan_right)


def _batch_mask(mode, pairs, n_grid, mask_fn):
    """pairs: list of (tss_left, tss_right) key groups; applies the mask
    kernel over all groups in one call, mutating left values in place."""
    if not pairs:
        return
    lrows, lgroup, grows, goff = [], [], [], [0]
    for gi, (tl, tr) in enumerate(pairs):
        for t in tl:
            lrows.append(t)
            lgroup.append(gi)
        grows.extend(tr)
        goff.append(len(grows))
    if mask_fn is not None:
        mask_fn(mode, lrows, lgroup, grows, goff)
        return
    lmat = np.stack([t.values for t in lrows])
    rmat = np.stack([t.values for t in grows]) if grows else \
        np.zeros((1, n_grid))
    engine.binop_mask(mode, lmat, np.asarray(lgroup, np.uint32), rmat,
                      np.asarray(goff, np.uint32))
    for i, t in enumerate(lrows):
        t.values = lmat[i]


MASK_AND = 0
MASK_UNLESS = 1
MASK_DEFAULT = 2


def _set_op(spec, left, right, mask_fn, or_fn):
    m_left, m_right = create_map_by_tag_set(spec, left, right)
    op = spec.op
    n_grid = left[0].values.shape[0] if left else (
        right[0].values.shape[0] if right else 0)
    rvs = []
    if op == "and":
        # binaryOpAnd (:535): iterate right keys, direct left lookup
        pairs = []
        for k, tss_right in m_right.items():
            tss_left = m_left.get(k)
            if tss_left:
                pairs.append((tss_left, tss_right))
        _batch_mask(MASK_AND, pairs, n_grid, mask_fn)
        for tl, _ in pairs:
            rvs.extend(remove_empty_series(tl))
        return rvs
    if op == "if":
        # binaryOpIf (:521): seriesByKey lookup
        pairs = []
        for k, tss_left in m_left.items():
            tss_right = _series_by_key(m_right, k)
            if tss_right is not None:
                pairs.append((tss_left, tss_right))
        _batch_mask(MASK_AND, pairs, n_grid, mask_fn)
        for tl, _ in pairs:
            rvs.extend(remove_empty_series(tl))
        return rvs
    if op in ("unless", "ifnot"):
        # binaryOpUnless (:715) / binaryOpIfnot (:700)
        pairs, passthrough = [], []
        for k, tss_left in m_left.items():
            tss_right = (m_right.get(k) if op == "unless"
                         else _series_by_key(m_right, k))
            if tss_right is None:
                passthrough.append(tss_left)
            else:
                pairs.append((tss_left, tss_right))
        _batch_mask(MASK_UNLESS, pairs, n_grid, mask_fn)
        out_by_group = {}
        for tl, _ in pairs:
            out_by_group[id(tl)] = remove_empty_series(tl)
        for k, tss_left in m_left.items():
            tss_right = (m_right.get(k) if op == "unless"
                         else _series_by_key(m_right, k))
            if tss_right is None:
                rvs.extend(tss_left)
            else:
                rvs.extend(out_by_group[id(tss_left)])
        return rvs
    if op == "default":
        # binaryOpDefault (:568)
        if not m_left:
            for tss in m_right.values():
                rvs.extend(tss)
            return rvs
        pairs = []
        for k, tss_left in m_left.items():
            rvs.extend(tss_left)
            tss_right = _series_by_key(m_right, k)
            if tss_right is not None:
                pairs.append((tss_left, tss_right))
        _batch_mask(MASK_DEFAULT, pairs, n_grid, mask_fn)
        return rvs
    if op == "or":
        return _or_op(spec, m_left, m_right, or_fn)
    raise ValueError(f"unknown set op {op!r}")


def _or_op(spec, m_left, m_right, or_fn):
    # binaryOpOr (:588)
    rvs = []
    for k in list(m_left):
        tss_left = remove_empty_series(m_left[k])
        m_left[k] = tss_left
        rvs.extend(tss_left)
    sort_series_by_metric_name(rvs)
    n_before = len(rvs)

    groups = []          # (tss_left, tss_right, can_merge matrix)
    appended_right = []
    for k, tss_right in m_right.items():
        tss_left = m_left.get(k)
        if not tss_left:
            appended_right.append((None, tss_right))
            continue
        # canBeMerged per (l, r): scalar fast path (binary_op.go:652-658)
        # or exact marshaled-name equality (:684)
        if is_scalar(tss_right):
            scalar_l = is_scalar(tss_left)
            cm = np.full((len(tss_left), len(tss_right)),
                         1 if scalar_l else 0, dtype=np.uint8)
        else:
            cm = np.zeros((len(tss_left), len(tss_right)), dtype=np.uint8)
            lkeys = [t.mn.marshal_sorted() for t in tss_left]
            rkeys = [t.mn.marshal_sorted() for t in tss_right]
            for i, lk in enumerate(lkeys):
                for j, rk in enumerate(rkeys):
                    cm[i, j] = 1 if lk == rk else 0
        groups.append((tss_left, tss_right, cm))
        appended_right.append((tss_left, tss_right))

    if groups:
        if or_fn is not None:
            or_fn(groups)
        else:
            lrows, rrows = [], []
            loff, roff = [0], [0]
            lrow_ids, rrow_ids = {}, {}
            lidx, ridx, cms, moff = [], [], [], []
            base = 0
            for tl, tr, cm in groups:
                for t in tl:
                    if id(t) not in lrow_ids:
                        lrow_ids[id(t)] = len(lrows)
                        lrows.append(t)
                    lidx.append(lrow_ids[id(t)])
                for t in tr:
                    if id(t) not in rrow_ids:
                        rrow_ids[id(t)] = len(rrows)
                        rrows.append(t)
                    ridx.append(rrow_ids[id(t)])
                loff.append(len(lidx))
                roff.append(len(ridx))
                moff.append(base)
                cms.append(cm.ravel())
                base += cm.size
            lmat = np.stack([t.values for t in lrows])
            rmat = np.stack([t.values for t in rrows])
            engine.binop_or(lmat, rmat,
                            np.asarray(loff, np.uint32),
                            np.asarray(lidx, np.uint32),
                            np.asarray(roff, np.uint32),
                            np.asarray(ridx, np.uint32),
                            np.concatenate(cms) if cms else
                            np.zeros(0, np.uint8),
                            np.asarray(moff, np.uint64))
            for i, t in enumerate(lrows):
                t.values = lmat[i]
            for i, t in enumerate(rrows):
                t.values = rmat[i]

    for tss_left, tss_right in appended_right:
        if tss_left is None:
            rvs.extend(tss_right)
        else:
            rvs.extend(remove_empty_series(tss_right))
    tail = rvs[n_before:]
    sort_series_by_metric_name(tail)
    rvs[n_before:] = tail
    return rvs
