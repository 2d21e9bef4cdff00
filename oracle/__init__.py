"""CPU oracle bindings — TEST INFRASTRUCTURE ONLY.

ctypes bindings to liboracle.so (the C restatement of the reference's
vmselect rollup path; see vm_oracle.h). Only tests/, __graft_entry__.smoke()
and bench.py's cpu_baseline leg may import this package. The product path
(victoriametrics_amd/) must never import it.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "liboracle.so")

# Func-name → enum mapping, mirroring the reference's rollupFuncs map keys
# (app/vmselect/promql/rollup.go:24-108).  Aliases point at the shared
# implementation exactly as the reference map does.
FUNC_IDS = {
    "rate": 0,
    "increase": 1,
    "increase_pure": 2,
    "delta": 3,
    "delta_prometheus": 4,
    "rate_prometheus": 5,
    "irate": 6,
    "ideriv": 7,
    "idelta": 8,
    "deriv_fast": 9,
    "avg_over_time": 10,
    "min_over_time": 11,
    "max_over_time": 12,
    "sum_over_time": 13,
    "sum2_over_time": 14,
    "count_over_time": 15,
    "first_over_time": 16,
    "last_over_time": 17,
    "quantile_over_time": 18,
    "median_over_time": 19,
    "stddev_over_time": 20,
    "stdvar_over_time": 21,
    "changes": 22,
    "changes_prometheus": 23,
    "resets": 24,
    "lag": 25,
    "lifetime": 26,
    "scrape_interval": 27,
    "rate_over_sum": 28,
    "range_over_time": 29,
    "tfirst_over_time": 30,
    "tlast_over_time": 31,
    "tmin_over_time": 32,
    "tmax_over_time": 33,
    "tlast_change_over_time": 34,
    "geomean_over_time": 35,
    "present_over_time": 36,
    "absent_over_time": 37,
    "stale_samples_over_time": 38,
    "count_le_over_time": 39,
    "count_gt_over_time": 40,
    "count_eq_over_time": 41,
    "count_ne_over_time": 42,
    "share_le_over_time": 43,
    "share_gt_over_time": 44,
    "share_eq_over_time": 45,
    "sum_le_over_time": 46,
    "sum_gt_over_time": 47,
    "sum_eq_over_time": 48,
    "deriv": 49,
    "predict_linear": 50,
    "ascent_over_time": 51,
    "descent_over_time": 52,
    "zscore_over_time": 53,
    "integrate": 54,
    "distinct_over_time": 55,
    "increases_over_time": 56,
    "decreases_over_time": 57,
    "mad_over_time": 58,
    "default_rollup": 59,
    "mode_over_time": 60,
    "duration_over_time": 61,
    "outlier_iqr_over_time": 62,
    "rollup_open": 63,
    "rollup_close": 64,
    "rollup_low": 65,
    "rollup_high": 66,
    "holt_winters": 67,
    "hoeffding_bound_lower": 68,
    "hoeffding_bound_upper": 69,
    # aliases (same implementations as in the reference's map)
    "increase_prometheus": 4,   # rollupDeltaPrometheus
    "timestamp": 31,            # rollupTlast
    "timestamp_with_name": 31,  # rollupTlast
    "iqr_over_time": 62,        # rollupAggrFuncs alias (rollup.go:169)
}

# rollupFuncsRemoveCounterResets (rollup.go:223-232)
REMOVE_COUNTER_RESETS_FUNCS = {
    "increase", "increase_prometheus", "increase_pure", "irate", "rate",
    "rate_prometheus", "rollup_increase", "rollup_rate",
}

# rollupFuncsSamplesScannedPerCall (rollup.go:238-263)
SAMPLES_SCANNED_PER_CALL = {
    "absent_over_time": 1, "count_over_time": 1, "default_rollup": 1,
    "delta": 2, "delta_prometheus": 2, "deriv_fast": 2, "first_over_time": 1,
    "idelta": 2, "ideriv": 2, "increase": 2, "increase_prometheus": 2,
    "increase_pure": 2, "irate": 2, "lag": 1, "last_over_time": 1,
    "lifetime": 2, "present_over_time": 1, "rate": 2, "rate_prometheus": 2,
    "scrape_interval": 2, "tfirst_over_time": 1, "timestamp": 1,
    "timestamp_with_name": 1, "tlast_over_time": 1,
}

# rollupFuncsCanAdjustWindow (rollup.go:204-219)
CAN_ADJUST_WINDOW_FUNCS = {
    "default_rollup", "deriv", "deriv_fast", "ideriv", "irate", "rate",
    "rate_over_sum", "rollup", "rollup_candlestick", "rollup_deriv",
    "rollup_rate", "rollup_scrape_interval", "scrape_interval", "timestamp",
    # the four per-config funcs getRollupConfigs derives from
    # rollup_candlestick (rollup.go:454-475) inherit its MayAdjustWindow
    "rollup_open", "rollup_close", "rollup_low", "rollup_high",
}

AGGR_IDS = {
    "none": 0, "sum": 1, "min": 2, "max": 3, "avg": 4,
    "count": 5, "sum2": 6, "geomean": 7, "group": 8,
}


class RollupConfigC(ctypes.Structure):
    _fields_ = [
        ("func", ctypes.c_int32),
        ("may_adjust_window", ctypes.c_int32),
        ("start", ctypes.c_int64),
        ("end", ctypes.c_int64),
        ("step", ctypes.c_int64),
        ("window", ctypes.c_int64),
        ("lookback_delta", ctypes.c_int64),
        ("min_staleness_interval", ctypes.c_int64),
        ("is_default_rollup", ctypes.c_int32),
        ("samples_scanned_per_call", ctypes.c_int32),
        ("arg", ctypes.c_double),
        ("arg2", ctypes.c_double),
    ]


def build(force=False):
    """Compile liboracle.so with the committed Makefile (gcc)."""
    if force or not os.path.exists(_LIB_PATH):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)
    return _LIB_PATH


_lib = None


def lib():
    global _lib
    if _lib is None:
        build()
        _lib = ctypes.CDLL(_LIB_PATH)
        _lib.vm_grid_points.restype = ctypes.c_int64
        _lib.vm_grid_points.argtypes = [ctypes.c_int64] * 3
        _lib.vm_quantile.restype = ctypes.c_double
        _lib.vm_quantile_sorted.restype = ctypes.c_double
        _lib.vm_call_rollup_fn.restype = ctypes.c_double
        _lib.vm_rollup_do.restype = ctypes.c_uint64
        _lib.vm_stale_nan.restype = ctypes.c_double
        _lib.vm_is_stale_nan.restype = ctypes.c_int
        _lib.vm_is_stale_nan.argtypes = [ctypes.c_double]
        _lib.vm_get_scrape_interval.restype = ctypes.c_int64
        _lib.vm_get_max_prev_interval.restype = ctypes.c_int64
        _lib.vm_get_max_prev_interval.argtypes = [ctypes.c_int64]
        _lib.vm_drop_stale_nans.restype = ctypes.c_int64
        _lib.vm_rollup_eval_batch.restype = ctypes.c_int
    return _lib


def _f64(a):
    return np.ascontiguousarray(a, dtype=np.float64)


def _i64(a):
    return np.ascontiguousarray(a, dtype=np.int64)


def _ptr(a, typ):
    return a.ctypes.data_as(ctypes.POINTER(typ))


def stale_nan():
    return lib().vm_stale_nan()


def grid_points(start, end, step):
    return lib().vm_grid_points(start, end, step)


def get_timestamps(start, end, step):
    n = grid_points(start, end, step)
    out = np.empty(n, dtype=np.int64)
    lib().vm_get_timestamps(ctypes.c_int64(start), ctypes.c_int64(end),
                            ctypes.c_int64(step), _ptr(out, ctypes.c_int64))
    return out


def remove_counter_resets(values, timestamps, msi=0):
    v = _f64(values).copy()
    t = _i64(timestamps)
    lib().vm_remove_counter_resets(_ptr(v, ctypes.c_double), _ptr(t, ctypes.c_int64),
                                   ctypes.c_int64(len(v)), ctypes.c_int64(msi))
    return v


def delta_values(values):
    v = _f64(values).copy()
    lib().vm_delta_values(_ptr(v, ctypes.c_double), ctypes.c_int64(len(v)))
    return v


def deriv_values(values, timestamps):
    v = _f64(values).copy()
    t = _i64(timestamps)
    lib().vm_deriv_values(_ptr(v, ctypes.c_double), _ptr(t, ctypes.c_int64),
                          ctypes.c_int64(len(v)))
    return v


def drop_stale_nans(values, timestamps):
    v = _f64(values).copy()
    t = _i64(timestamps).copy()
    n = lib().vm_drop_stale_nans(_ptr(v, ctypes.c_double), _ptr(t, ctypes.c_int64),
                                 ctypes.c_int64(len(v)))
    return v[:n], t[:n]


def quantile(phi, values):
    v = _f64(values)
    return lib().vm_quantile(ctypes.c_double(phi), _ptr(v, ctypes.c_double),
                             ctypes.c_int64(len(v)))


def call_rollup_fn(func_name, values, timestamps, prev_value=float("nan"),
                   prev_timestamp=0, real_prev_value=float("nan"),
                   real_next_value=float("nan"), curr_timestamp=0, idx=0,
                   window=0, arg=0.0, arg2=0.0):
    v = _f64(values)
    t = _i64(timestamps)
    return lib().vm_call_rollup_fn(
        ctypes.c_int32(FUNC_IDS[func_name]), ctypes.c_double(prev_value),
        ctypes.c_int64(prev_timestamp), _ptr(v, ctypes.c_double),
        _ptr(t, ctypes.c_int64), ctypes.c_int64(len(v)),
        ctypes.c_double(real_prev_value), ctypes.c_double(real_next_value),
        ctypes.c_int64(curr_timestamp), ctypes.c_int64(idx),
        ctypes.c_int64(window), ctypes.c_double(arg), ctypes.c_double(arg2))


def make_config(func_name, start, end, step, window=0, lookback_delta=0,
                min_staleness_interval=0, arg=0.0, arg2=0.0,
                may_adjust_window=None, samples_scanned_per_call=None):
    """Build a RollupConfigC the way getRollupConfigs does (rollup.go:374-516)
    when may_adjust_window/samples_scanned_per_call are None; pass explicit
    values (e.g. 0) to mirror the reference's direct rollupConfig{} literals in
    unit tests."""
    if may_adjust_window is None:
        may_adjust_window = func_name in CAN_ADJUST_WINDOW_FUNCS
    if samples_scanned_per_call is None:
        samples_scanned_per_call = SAMPLES_SCANNED_PER_CALL.get(func_name, 0)
    return RollupConfigC(
        func=FUNC_IDS[func_name],
        may_adjust_window=1 if may_adjust_window else 0,
        start=start, end=end, step=step, window=window,
        lookback_delta=lookback_delta,
        min_staleness_interval=min_staleness_interval,
        is_default_rollup=1 if func_name == "default_rollup" else 0,
        samples_scanned_per_call=samples_scanned_per_call,
        arg=arg, arg2=arg2)


def rollup_do(rc, values, timestamps):
    v = _f64(values)
    t = _i64(timestamps)
    n_grid = grid_points(rc.start, rc.end, rc.step)
    dst = np.empty(n_grid, dtype=np.float64)
    scanned = lib().vm_rollup_do(ctypes.byref(rc), _ptr(v, ctypes.c_double),
                                 _ptr(t, ctypes.c_int64), ctypes.c_int64(len(v)),
                                 _ptr(dst, ctypes.c_double))
    return dst, scanned


def rollup_eval_batch(rc, ts, vals, offsets, group_ids=None, n_groups=0,
                      aggr="none", remove_counter_resets=False,
                      max_staleness_interval=0, drop_stale_nans=False,
                      n_threads=1, pre_func=0):
    """Batch CSR evaluation — mirrors the product C-ABI shape."""
    t = _i64(ts)
    v = _f64(vals)
    off = np.ascontiguousarray(offsets, dtype=np.uint64)
    n_series = len(off) - 1
    n_grid = grid_points(rc.start, rc.end, rc.step)
    aggr_id = AGGR_IDS[aggr]
    if group_ids is not None and aggr_id != 0:
        gids = np.ascontiguousarray(group_ids, dtype=np.int32)
        out = np.empty((n_groups, n_grid), dtype=np.float64)
        counts = np.empty((n_groups, n_grid), dtype=np.float64)
        gptr = _ptr(gids, ctypes.c_int32)
        cptr = _ptr(counts, ctypes.c_double)
    else:
        out = np.empty((n_series, n_grid), dtype=np.float64)
        counts = None
        gptr = None
        cptr = None
    scanned = ctypes.c_uint64(0)
    rcode = lib().vm_rollup_eval_batch(
        ctypes.byref(rc), ctypes.c_int32(1 if remove_counter_resets else 0),
        ctypes.c_int64(max_staleness_interval),
        ctypes.c_int32(1 if drop_stale_nans else 0),
        ctypes.c_int32(int(pre_func)),
        _ptr(t, ctypes.c_int64), _ptr(v, ctypes.c_double),
        _ptr(off, ctypes.c_uint64), ctypes.c_uint32(n_series),
        gptr, ctypes.c_uint32(n_groups), ctypes.c_int32(aggr_id),
        _ptr(out, ctypes.c_double), cptr, ctypes.byref(scanned),
        ctypes.c_int(n_threads))
    if rcode != 0:
        raise RuntimeError(f"vm_rollup_eval_batch failed: {rcode}")
    return out, counts, scanned.value


# ---------------------------------------------------------------------------
# lib/decimal + lib/encoding + merge/dedup oracle (vm_decimal.h)
# ---------------------------------------------------------------------------

MT_ZSTD_NEAREST_DELTA2 = 1
MT_DELTA_CONST = 2
MT_CONST = 3
MT_ZSTD_NEAREST_DELTA = 4
MT_NEAREST_DELTA2 = 5
MT_NEAREST_DELTA = 6


def _codec_lib():
    l = lib()
    if not hasattr(l.vm_decimal_to_float, "_typed"):
        l.vm_decimal_to_float.restype = ctypes.c_double
        l.vm_decimal_to_float.argtypes = [ctypes.c_int64, ctypes.c_int16]
        l.vm_decimal_round_to_decimal_digits.restype = ctypes.c_double
        l.vm_decimal_round_to_decimal_digits.argtypes = [ctypes.c_double, ctypes.c_int]
        l.vm_decimal_round_to_significant_figures.restype = ctypes.c_double
        l.vm_decimal_round_to_significant_figures.argtypes = [ctypes.c_double, ctypes.c_int]
        l.vm_decimal_max_up_exponent.restype = ctypes.c_int16
        l.vm_decimal_max_up_exponent.argtypes = [ctypes.c_int64]
        l.vm_decimal_calibrate_scale.restype = ctypes.c_int16
        l.vm_marshal_varint64s.restype = ctypes.c_size_t
        l.vm_unmarshal_varint64s.restype = ctypes.c_int64
        l.vm_marshal_nearest_delta.restype = ctypes.c_size_t
        l.vm_marshal_nearest_delta2.restype = ctypes.c_size_t
        l.vm_unmarshal_nearest_delta.restype = ctypes.c_int
        l.vm_unmarshal_nearest_delta2.restype = ctypes.c_int
        l.vm_marshal_int64_array.restype = ctypes.c_int64
        l.vm_pack_blocks.restype = ctypes.c_int64
        l.vm_unmarshal_int64_array.restype = ctypes.c_int
        l.vm_deduplicate_samples.restype = ctypes.c_int64
        l.vm_merge_sort_blocks.restype = ctypes.c_int64
        l.vm_decimal_to_float._typed = True
    return l


def decimal_from_float(f):
    l = _codec_lib()
    v = ctypes.c_int64(0)
    e = ctypes.c_int16(0)
    l.vm_decimal_from_float(ctypes.c_double(f), ctypes.byref(v), ctypes.byref(e))
    return v.value, e.value


def positive_float_to_decimal(f):
    l = _codec_lib()
    v = ctypes.c_int64(0)
    e = ctypes.c_int16(0)
    l.vm_decimal_positive_float_to_decimal(ctypes.c_double(f), ctypes.byref(v),
                                           ctypes.byref(e))
    return v.value, e.value


def decimal_to_float(v, e):
    return _codec_lib().vm_decimal_to_float(ctypes.c_int64(v), ctypes.c_int16(e))


def decimal_append_to_float(va, e):
    l = _codec_lib()
    a = np.ascontiguousarray(va, dtype=np.int64)
    out = np.empty(len(a), dtype=np.float64)
    l.vm_decimal_append_to_float(_ptr(out, ctypes.c_double), _ptr(a, ctypes.c_int64),
                                 ctypes.c_int64(len(a)), ctypes.c_int16(e))
    return out


def float_to_decimal(src):
    l = _codec_lib()
    s = np.ascontiguousarray(src, dtype=np.float64)
    va = np.empty(len(s), dtype=np.int64)
    e = ctypes.c_int16(0)
    l.vm_decimal_append_float_to_decimal(_ptr(s, ctypes.c_double),
                                         ctypes.c_int64(len(s)),
                                         _ptr(va, ctypes.c_int64), ctypes.byref(e))
    return va, e.value


def marshal_varint64s(vs):
    l = _codec_lib()
    a = np.ascontiguousarray(vs, dtype=np.int64)
    dst = np.empty(len(a) * 10 + 16, dtype=np.uint8)
    n = l.vm_marshal_varint64s(_ptr(dst, ctypes.c_uint8), _ptr(a, ctypes.c_int64),
                               ctypes.c_int64(len(a)))
    return bytes(dst[:n])


def unmarshal_varint64s(data, n):
    l = _codec_lib()
    src = np.frombuffer(bytes(data), dtype=np.uint8)
    dst = np.empty(n, dtype=np.int64)
    used = l.vm_unmarshal_varint64s(_ptr(dst, ctypes.c_int64), ctypes.c_int64(n),
                                    _ptr(src, ctypes.c_uint8),
                                    ctypes.c_size_t(len(src)))
    if used < 0:
        raise ValueError(f"varint decode error {used}")
    return dst, used


def marshal_nearest_delta(src, precision_bits, delta2=False):
    l = _codec_lib()
    a = np.ascontiguousarray(src, dtype=np.int64)
    dst = np.empty(len(a) * 10 + 16, dtype=np.uint8)
    first = ctypes.c_int64(0)
    fn = l.vm_marshal_nearest_delta2 if delta2 else l.vm_marshal_nearest_delta
    n = fn(_ptr(dst, ctypes.c_uint8), _ptr(a, ctypes.c_int64),
           ctypes.c_int64(len(a)), ctypes.c_uint8(precision_bits),
           ctypes.byref(first))
    return bytes(dst[:n]), first.value


def unmarshal_nearest_delta(data, first_value, items, delta2=False):
    l = _codec_lib()
    src = np.frombuffer(bytes(data), dtype=np.uint8)
    dst = np.empty(items, dtype=np.int64)
    fn = l.vm_unmarshal_nearest_delta2 if delta2 else l.vm_unmarshal_nearest_delta
    rc = fn(_ptr(dst, ctypes.c_int64),
            _ptr(src, ctypes.c_uint8) if len(src) else None,
            ctypes.c_size_t(len(src)), ctypes.c_int64(first_value),
            ctypes.c_int64(items))
    if rc != 0:
        raise ValueError(f"nearest-delta decode error {rc}")
    return dst


def pack_blocks(ts, vals_int, offsets, precision_bits=64):
    """Synthetic packed block stream for the cold-query bench/tests: CSR
    int64 columns (scale 0) -> the wire form vmgpu_batch_create_packed
    parses (one block per series, post-zstd marshal types).  Returns
    (packed_bytes, n_blocks, series_block_start)."""
    l = _codec_lib()
    t = np.ascontiguousarray(ts, dtype=np.int64)
    v = np.ascontiguousarray(vals_int, dtype=np.int64)
    off = np.ascontiguousarray(offsets, dtype=np.uint64)
    n_series = len(off) - 1
    cap = int(len(t)) * 22 + n_series * 192 + 64
    dst = np.empty(cap, dtype=np.uint8)
    sbs = np.zeros(n_series + 1, dtype=np.uint32)
    n = l.vm_pack_blocks(_ptr(t, ctypes.c_int64), _ptr(v, ctypes.c_int64),
                         _ptr(off, ctypes.c_uint64), ctypes.c_uint32(n_series),
                         ctypes.c_uint8(precision_bits),
                         _ptr(dst, ctypes.c_uint8), ctypes.c_int64(cap),
                         _ptr(sbs, ctypes.c_uint32))
    if n < 0:
        raise ValueError(f"pack_blocks error {n}")
    return bytes(dst[:n].data), n_series, sbs


def marshal_int64_array(a, precision_bits=64):
    l = _codec_lib()
    arr = np.ascontiguousarray(a, dtype=np.int64)
    dst = np.empty(len(arr) * 10 + 64, dtype=np.uint8)
    mt = ctypes.c_uint8(0)
    first = ctypes.c_int64(0)
    n = l.vm_marshal_int64_array(_ptr(dst, ctypes.c_uint8), _ptr(arr, ctypes.c_int64),
                                 ctypes.c_int64(len(arr)),
                                 ctypes.c_uint8(precision_bits),
                                 ctypes.byref(mt), ctypes.byref(first))
    if n < 0:
        raise ValueError(f"marshal error {n}")
    return bytes(dst[:n]), mt.value, first.value


def unmarshal_int64_array(data, items, mt, first_value):
    l = _codec_lib()
    src = np.frombuffer(bytes(data), dtype=np.uint8)
    dst = np.empty(items, dtype=np.int64)
    rc = l.vm_unmarshal_int64_array(_ptr(dst, ctypes.c_int64), ctypes.c_int64(items),
                                    _ptr(src, ctypes.c_uint8) if len(src) else None,
                                    ctypes.c_size_t(len(src)),
                                    ctypes.c_uint8(mt), ctypes.c_int64(first_value))
    if rc != 0:
        raise ValueError(f"unmarshal error {rc}")
    return dst


def deduplicate_samples(ts, vals, dedup_interval):
    l = _codec_lib()
    t = np.ascontiguousarray(ts, dtype=np.int64).copy()
    v = np.ascontiguousarray(vals, dtype=np.float64).copy()
    n = l.vm_deduplicate_samples(_ptr(t, ctypes.c_int64), _ptr(v, ctypes.c_double),
                                 ctypes.c_int64(len(t)),
                                 ctypes.c_int64(dedup_interval))
    return t[:n], v[:n]


def merge_sort_blocks(blocks, dedup_interval=0):
    """blocks: list of (ts array, vals array)."""
    l = _codec_lib()
    offs = [0]
    for t, v in blocks:
        offs.append(offs[-1] + len(t))
    ts = np.concatenate([np.asarray(t, dtype=np.int64) for t, _ in blocks]) \
        if blocks and offs[-1] else np.empty(0, np.int64)
    vals = np.concatenate([np.asarray(v, dtype=np.float64) for _, v in blocks]) \
        if blocks and offs[-1] else np.empty(0, np.float64)
    offsets = np.asarray(offs, dtype=np.uint64)
    dst_t = np.empty(max(offs[-1], 1), dtype=np.int64)
    dst_v = np.empty(max(offs[-1], 1), dtype=np.float64)
    n = l.vm_merge_sort_blocks(_ptr(ts, ctypes.c_int64) if len(ts) else None,
                               _ptr(vals, ctypes.c_double) if len(vals) else None,
                               _ptr(offsets, ctypes.c_uint64),
                               ctypes.c_int32(len(blocks)),
                               ctypes.c_int64(dedup_interval),
                               _ptr(dst_t, ctypes.c_int64),
                               _ptr(dst_v, ctypes.c_double))
    return dst_t[:n], dst_v[:n]


# ---------------------------------------------------------------------------
# topk family + histogram_quantile oracle (oracle/topk.c)
# ---------------------------------------------------------------------------

TOPK_SUMMARY_OPS = {"avg": 0, "min": 1, "max": 2, "median": 3, "last": 4}


def topk_pointwise(values, k, reverse=False):
    """newAggrFuncTopK: per-point top-k NaN-fill; returns a new matrix."""
    l = lib()
    v = np.ascontiguousarray(values, dtype=np.float64).copy()
    n_series, n_grid = v.shape
    l.vm_topk_pointwise(_ptr(v, ctypes.c_double), ctypes.c_int64(n_series),
                        ctypes.c_int64(n_grid), ctypes.c_double(k),
                        ctypes.c_int32(1 if reverse else 0))
    return v


def topk_range(values, k, summary="avg", reverse=False, remaining=False):
    l = lib()
    l.vm_topk_range.restype = ctypes.c_int64
    v = np.ascontiguousarray(values, dtype=np.float64)
    n_series, n_grid = v.shape
    sel = np.empty(max(int(k), 1), dtype=np.int64)
    rem = np.empty(n_grid, dtype=np.float64) if remaining else None
    m = l.vm_topk_range(_ptr(v, ctypes.c_double), ctypes.c_int64(n_series),
                        ctypes.c_int64(n_grid), ctypes.c_double(k),
                        ctypes.c_int32(TOPK_SUMMARY_OPS[summary]),
                        ctypes.c_int32(1 if reverse else 0),
                        _ptr(sel, ctypes.c_int64),
                        _ptr(rem, ctypes.c_double) if remaining else None)
    return sel[:m], rem


def histogram_quantile(phi, bucket_values, les, group_offsets, bounds=False):
    l = lib()
    bv = np.ascontiguousarray(bucket_values, dtype=np.float64)
    le = np.ascontiguousarray(les, dtype=np.float64)
    off = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    n_groups = len(off) - 1
    n_grid = bv.shape[1]
    out = np.empty((n_groups, n_grid), dtype=np.float64)
    lo = np.empty((n_groups, n_grid), dtype=np.float64) if bounds else None
    hi = np.empty((n_groups, n_grid), dtype=np.float64) if bounds else None
    l.vm_histogram_quantile(ctypes.c_double(phi), _ptr(bv, ctypes.c_double),
                            _ptr(le, ctypes.c_double), _ptr(off, ctypes.c_uint64),
                            ctypes.c_int64(n_groups), ctypes.c_int64(n_grid),
                            _ptr(out, ctypes.c_double),
                            _ptr(lo, ctypes.c_double) if bounds else None,
                            _ptr(hi, ctypes.c_double) if bounds else None)
    return (out, lo, hi) if bounds else (out, None, None)


def topk_summary(op, row):
    l = lib()
    l.vm_topk_summary.restype = ctypes.c_double
    r = _f64(row)
    return l.vm_topk_summary(ctypes.c_int32(TOPK_SUMMARY_OPS[op]),
                             _ptr(r, ctypes.c_double), ctypes.c_int64(len(r)))


def zstd_decompress(data, cap=None):
    l = _codec_lib()
    l.vm_zstd_decompress.restype = ctypes.c_int64
    src = np.frombuffer(bytes(data), dtype=np.uint8)
    if cap is None:
        cap = max(len(src) * 64, 1 << 20)
    dst = np.empty(cap, dtype=np.uint8)
    n = l.vm_zstd_decompress(_ptr(dst, ctypes.c_uint8), ctypes.c_size_t(cap),
                             _ptr(src, ctypes.c_uint8), ctypes.c_size_t(len(src)))
    if n < 0:
        raise ValueError(f"zstd decompress error {n}")
    return bytes(dst[:n])


def ensure_non_decreasing(a, v_min, v_max):
    l = _codec_lib()
    arr = np.ascontiguousarray(a, dtype=np.int64).copy()
    l.vm_ensure_non_decreasing(_ptr(arr, ctypes.c_int64), ctypes.c_int64(len(arr)),
                               ctypes.c_int64(v_min), ctypes.c_int64(v_max))
    return arr


# ---------------------------------------------------------------------------
# binary operator oracle (oracle/binop.c)
# ---------------------------------------------------------------------------

BINOP_IDS = {
    "+": 0, "-": 1, "*": 2, "/": 3, "%": 4, "^": 5, "atan2": 6,
    "==": 7, "!=": 8, ">": 9, "<": 10, ">=": 11, "<=": 12,
    "default": 13, "if": 14, "ifnot": 15, "and": 16, "or": 17,
}


def _binop_lib():
    l = lib()
    if not hasattr(l.vm_binop_scalar, "_typed"):
        l.vm_binop_scalar.restype = ctypes.c_double
        l.vm_binop_scalar.argtypes = [ctypes.c_int32, ctypes.c_int32,
                                      ctypes.c_double, ctypes.c_double]
        l.vm_binop_apply.restype = None
        l.vm_binop_scalar._typed = True
    return l


def binop_scalar(op, a, b, is_bool=False):
    return _binop_lib().vm_binop_scalar(
        ctypes.c_int32(BINOP_IDS[op] if isinstance(op, str) else op),
        ctypes.c_int32(1 if is_bool else 0),
        ctypes.c_double(a), ctypes.c_double(b))


def binop_apply(op, a, b, is_bool=False, drop_nan_right=False,
                fill_left=None, fill_right=None):
    """newBinaryOpFunc value loop (binary_op.go:162-236) over one pair."""
    l = _binop_lib()
    a = np.ascontiguousarray(a, dtype=np.float64)
    b = np.ascontiguousarray(b, dtype=np.float64)
    out = np.empty(len(a), dtype=np.float64)
    l.vm_binop_apply(
        ctypes.c_int32(BINOP_IDS[op] if isinstance(op, str) else op),
        ctypes.c_int32(1 if is_bool else 0),
        ctypes.c_int32(1 if drop_nan_right else 0),
        _ptr(a, ctypes.c_double), _ptr(b, ctypes.c_double),
        ctypes.c_int64(len(a)),
        ctypes.c_int32(0 if fill_left is None else 1),
        ctypes.c_double(fill_left if fill_left is not None else 0.0),
        ctypes.c_int32(0 if fill_right is None else 1),
        ctypes.c_double(fill_right if fill_right is not None else 0.0),
        _ptr(out, ctypes.c_double))
    return out


# ---------------------------------------------------------------------------
# transform oracle (oracle/transform.c); func ids = include/vmgpu.h VMGPU_TF_*
# ---------------------------------------------------------------------------

def tf_apply(func_id, values, ts=None, arg1=None, arg2=None, scalar=0.0):
    """Apply a transform func to [n_series x n_grid] values in place;
    returns (values, keep_flags)."""
    l = lib()
    l.vm_tf_apply.restype = None
    v = np.ascontiguousarray(values, dtype=np.float64)
    ns, ng = v.shape
    t = np.ascontiguousarray(ts, dtype=np.int64) if ts is not None else None
    a1 = np.ascontiguousarray(arg1, dtype=np.float64) if arg1 is not None else None
    a2 = np.ascontiguousarray(arg2, dtype=np.float64) if arg2 is not None else None
    keep = np.ones(ns, dtype=np.uint8)
    l.vm_tf_apply(ctypes.c_int32(func_id), _ptr(v, ctypes.c_double),
                  ctypes.c_int64(ns), ctypes.c_int64(ng),
                  _ptr(t, ctypes.c_int64) if t is not None else None,
                  _ptr(a1, ctypes.c_double) if a1 is not None else None,
                  _ptr(a2, ctypes.c_double) if a2 is not None else None,
                  ctypes.c_double(scalar), _ptr(keep, ctypes.c_uint8))
    return v, keep


def histogram_stat(mode, bucket_values, les, group_offsets):
    """histogram_avg/stddev/stdvar oracle (mode 0/1/2)."""
    l = lib()
    l.vm_histogram_stat.restype = None
    bv = np.ascontiguousarray(bucket_values, dtype=np.float64)
    le = np.ascontiguousarray(les, dtype=np.float64)
    off = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    ng = len(off) - 1
    grid = bv.shape[1]
    out = np.empty((ng, grid), dtype=np.float64)
    l.vm_histogram_stat(ctypes.c_int32(mode), _ptr(bv, ctypes.c_double),
                        _ptr(le, ctypes.c_double), _ptr(off, ctypes.c_uint64),
                        ctypes.c_int64(ng), ctypes.c_int64(grid),
                        _ptr(out, ctypes.c_double))
    return out


def histogram_share(le_req, bucket_values, les, group_offsets):
    l = lib()
    l.vm_histogram_share.restype = None
    req = np.ascontiguousarray(le_req, dtype=np.float64)
    bv = np.ascontiguousarray(bucket_values, dtype=np.float64)
    le = np.ascontiguousarray(les, dtype=np.float64)
    off = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    ng = len(off) - 1
    grid = bv.shape[1]
    out = np.empty((ng, grid), dtype=np.float64)
    lo = np.empty((ng, grid), dtype=np.float64)
    hi = np.empty((ng, grid), dtype=np.float64)
    l.vm_histogram_share(_ptr(req, ctypes.c_double), _ptr(bv, ctypes.c_double),
                         _ptr(le, ctypes.c_double), _ptr(off, ctypes.c_uint64),
                         ctypes.c_int64(ng), ctypes.c_int64(grid),
                         _ptr(out, ctypes.c_double), _ptr(lo, ctypes.c_double),
                         _ptr(hi, ctypes.c_double))
    return out, lo, hi


def colagg(op, values, group_rows, group_offsets, phi=0.0):
    """Non-incremental cross-series aggregate oracle (aggr.go long tail);
    op ids/names match engine.COLAGG_OPS."""
    _OPS = {"median": 0, "quantile": 1, "mad": 2, "stddev": 3, "stdvar": 4,
            "mode": 5, "distinct": 6, "share": 7, "zscore": 8,
            "iqr_bounds": 9, "sum": 10, "min": 11, "max": 12, "avg": 13,
            "count": 14, "sum2": 15, "geomean": 16, "group": 17}
    opid = _OPS[op] if isinstance(op, str) else int(op)
    l = lib()
    l.vm_colagg.restype = None
    v = np.ascontiguousarray(values, dtype=np.float64)
    gr = np.ascontiguousarray(group_rows, dtype=np.uint32)
    go = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    ns, ng = v.shape
    n_groups = len(go) - 1
    per_series = opid in (7, 8)
    bounds = opid == 9
    out = None if per_series else np.empty((n_groups, ng), np.float64)
    out2 = np.empty((n_groups, ng), np.float64) if bounds else None
    vout = np.empty_like(v) if per_series else None
    l.vm_colagg(ctypes.c_int32(opid), _ptr(v, ctypes.c_double),
                ctypes.c_uint32(ns), ctypes.c_uint32(ng),
                _ptr(gr, ctypes.c_uint32), _ptr(go, ctypes.c_uint64),
                ctypes.c_uint32(n_groups), ctypes.c_double(phi),
                _ptr(out, ctypes.c_double) if out is not None else None,
                _ptr(out2, ctypes.c_double) if out2 is not None else None,
                _ptr(vout, ctypes.c_double) if vout is not None else None)
    if bounds:
        return out, out2
    if per_series:
        return vout
    return out


def colagg_filter(mode, values, group_of, b1, b2):
    l = lib()
    l.vm_colagg_filter.restype = None
    v = np.ascontiguousarray(values, dtype=np.float64)
    gof = np.ascontiguousarray(group_of, dtype=np.int32)
    b1 = np.ascontiguousarray(b1, dtype=np.float64)
    b2 = np.ascontiguousarray(b2, dtype=np.float64)
    ns, ng = v.shape
    flags = np.zeros(ns, dtype=np.uint8)
    l.vm_colagg_filter(ctypes.c_int32(0 if mode == "iqr" else 1),
                       _ptr(v, ctypes.c_double), _ptr(gof, ctypes.c_int32),
                       ctypes.c_uint32(ns), ctypes.c_uint32(ng),
                       _ptr(b1, ctypes.c_double), _ptr(b2, ctypes.c_double),
                       _ptr(flags, ctypes.c_uint8))
    return flags
