"""Host-side mirror of the reference's rollup operator interface.

This module is the product path: it mirrors the seam of
app/vmselect/promql/eval.go:1899-1904 (getRollupConfigs + the RunParallel
worker fan-out) over the C-ABI in include/vmgpu.h.  Function names, argument
meaning and error behavior follow the reference's promql package; the labels →
dense-group-id assignment that replaces marshalMetricNameSorted map keys
(aggr_incremental.go:98-139) is the caller's (see group_ids argument).

The HIP extension is REQUIRED: importing this module on a machine with a GPU
and calling any evaluation entry point without victoriametrics_amd/libvmgpu.so
raises immediately — there is no CPU fallback in the product path (the CPU
restatement under oracle/ is test infrastructure only and is never imported
here).
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.environ.get("VMGPU_LIB", os.path.join(_DIR, "libvmgpu.so"))

# rollupFuncs map keys -> func ids (rollup.go:24-108).  Ids are shared with
# include/vmgpu.h / oracle/vm_oracle.h.
FUNC_IDS = {
    "rate": 0, "increase": 1, "increase_pure": 2, "delta": 3,
    "delta_prometheus": 4, "rate_prometheus": 5, "irate": 6, "ideriv": 7,
    "idelta": 8, "deriv_fast": 9, "avg_over_time": 10, "min_over_time": 11,
    "max_over_time": 12, "sum_over_time": 13, "sum2_over_time": 14,
    "count_over_time": 15, "first_over_time": 16, "last_over_time": 17,
    "quantile_over_time": 18, "median_over_time": 19, "stddev_over_time": 20,
    "stdvar_over_time": 21, "changes": 22, "changes_prometheus": 23,
    "resets": 24, "lag": 25, "lifetime": 26, "scrape_interval": 27,
    "rate_over_sum": 28, "range_over_time": 29, "tfirst_over_time": 30,
    "tlast_over_time": 31, "tmin_over_time": 32, "tmax_over_time": 33,
    "tlast_change_over_time": 34, "geomean_over_time": 35,
    "present_over_time": 36, "absent_over_time": 37,
    "stale_samples_over_time": 38, "count_le_over_time": 39,
    "count_gt_over_time": 40, "count_eq_over_time": 41,
    "count_ne_over_time": 42, "share_le_over_time": 43,
    "share_gt_over_time": 44, "share_eq_over_time": 45,
    "sum_le_over_time": 46, "sum_gt_over_time": 47, "sum_eq_over_time": 48,
    "deriv": 49, "predict_linear": 50, "ascent_over_time": 51,
    "descent_over_time": 52, "zscore_over_time": 53, "integrate": 54,
    "distinct_over_time": 55, "increases_over_time": 56,
    "decreases_over_time": 57, "mad_over_time": 58, "default_rollup": 59,
    "mode_over_time": 60, "duration_over_time": 61,
    "outlier_iqr_over_time": 62,
    "rollup_open": 63, "rollup_close": 64, "rollup_low": 65,
    "rollup_high": 66, "holt_winters": 67, "hoeffding_bound_lower": 68,
    "hoeffding_bound_upper": 69,
    # aliases to shared implementations, as in the reference map
    "increase_prometheus": 4, "timestamp": 31, "timestamp_with_name": 31,
    "iqr_over_time": 62,  # rollupAggrFuncs alias (rollup.go:169)
}

AGGR_IDS = {
    "none": 0, "sum": 1, "min": 2, "max": 3, "avg": 4,
    "count": 5, "sum2": 6, "geomean": 7, "group": 8,
}


def any_representative_group_ids(group_ids):
    """`any` incremental aggregate (aggr_incremental.go:54 updateAggrAny:
    the first series of each group wins; worker order makes the reference's
    pick scheduling-dependent — this engine picks the LOWEST series id,
    deterministically).  Returns group_ids with non-representative members
    masked to -1; run the plan with aggr="sum" (a single-member sum is the
    identity) to get `any` semantics."""
    gids = np.ascontiguousarray(group_ids, dtype=np.int32).copy()
    seen = {}
    for s, g in enumerate(gids):
        if g < 0:
            continue
        if g in seen:
            gids[s] = -1
        else:
            seen[g] = s
    return gids

# rollupFuncsRemoveCounterResets (rollup.go:223-232)
REMOVE_COUNTER_RESETS_FUNCS = {
    "increase", "increase_prometheus", "increase_pure", "irate", "rate",
    "rate_prometheus", "rollup_increase", "rollup_rate",
}

# rollupFuncsCanAdjustWindow (rollup.go:204-219)
CAN_ADJUST_WINDOW_FUNCS = {
    "default_rollup", "deriv", "deriv_fast", "ideriv", "irate", "rate",
    "rate_over_sum", "rollup", "rollup_candlestick", "rollup_deriv",
    "rollup_rate", "rollup_scrape_interval", "scrape_interval", "timestamp",
    "rollup_open", "rollup_close", "rollup_low", "rollup_high",
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

# funcs whose stale NaNs must be kept (dropStaleNaNs, eval.go:2108-2115)
KEEP_STALE_NANS_FUNCS = {"default_rollup", "stale_samples_over_time"}

# preFunc value transforms for the rollup_* pseudo-functions
# (getRollupConfigs, rollup.go:436-516)
PRE_FUNC_IDS = {"none": 0, "delta": 1, "deriv": 2, "scrape_interval": 3}

# rollup_* pseudo-function -> (pre_func, candlestick?) expansion
ROLLUP_FAKE_FUNCS = {
    "rollup": ("none", False),
    "rollup_rate": ("deriv", False),
    "rollup_deriv": ("deriv", False),
    "rollup_increase": ("delta", False),
    "rollup_delta": ("delta", False),
    "rollup_scrape_interval": ("scrape_interval", False),
    "rollup_candlestick": ("none", True),
}

# rollupFuncsKeepMetricName (rollup.go:267-287): these keep the metric
# group on the output series; everything else resets it unless the query
# carries the keep_metric_names modifier (doRollupForTimeseries,
# eval.go:2016-2018)
ROLLUP_KEEP_METRIC_NAME_FUNCS = {
    "avg_over_time", "default_rollup", "first_over_time",
    "geomean_over_time", "hoeffding_bound_lower", "hoeffding_bound_upper",
    "holt_winters", "iqr_over_time", "last_over_time", "max_over_time",
    "median_over_time", "min_over_time", "mode_over_time",
    "predict_linear", "quantile_over_time", "quantiles_over_time",
    "rollup", "rollup_candlestick", "timestamp_with_name",
}


def plan_with_at(func, start, end, step, at_values, **kwargs):
    """evalRollupFunc's `@` modifier handling (eval.go:903-950): the `@`
    expression must yield ONE series; its first non-NaN value (seconds) is
    the evaluation timestamp; the rollup runs on the single-point grid
    [at, at] and the one-point result broadcasts to the original grid.
    Returns (plan, report_timestamps, broadcast) where broadcast(out)
    expands the [rows x 1] output to [rows x n_grid]."""
    at_values = np.atleast_2d(np.asarray(at_values, np.float64))
    if at_values.shape[0] != 1:
        raise VmGpuError(
            "`@` modifier must return a single series; it returns "
            f"{at_values.shape[0]} series instead")
    finite = at_values[0][~np.isnan(at_values[0])]
    if finite.size == 0:
        raise VmGpuError("`@` modifier must return a non-NaN value")
    at_ms = int(finite[0] * 1000)
    plan = RollupPlan(func, at_ms, at_ms, int(step), **kwargs)
    n_grid = 1 + (int(end) - int(start)) // int(step)
    report_ts = np.asarray(start, np.int64) +         np.arange(n_grid, dtype=np.int64) * int(step)

    def broadcast(out):
        out = np.asarray(out, np.float64).reshape(-1, 1)
        return np.repeat(out, n_grid, axis=1)

    return plan, report_ts, broadcast


def plan_with_offset(func, start, end, step, offset_ms=0, **kwargs):
    """evalRollupFuncWithoutAt's offset handling (eval.go:954-1008):
    `rf(m[w] offset o)` evaluates on the grid shifted BACK by the offset
    and reports the original timestamps; rollup_candlestick additionally
    auto-applies `offset -step` so each point covers (t-step, t].
    Returns (plan, report_timestamps)."""
    offset = int(offset_ms)
    start, end = int(start) - offset, int(end) - offset
    if func == "rollup_candlestick" or kwargs.get(
            "parent_func") == "rollup_candlestick":
        start += int(step)
        end += int(step)
        offset -= int(step)
    plan = RollupPlan(func, start, end, int(step), **kwargs)
    return plan, plan.timestamps() + offset


def aggregate_absent_over_time(values_rows, n_grid, base_mn=None):
    """aggregateAbsentOverTime (eval.go:1012-1031): collapse the
    per-series absent_over_time rollup rows (1 where the series had no
    samples in the window, NaN where it had) into ONE series that is NaN
    wherever ANY input series was present."""
    from .binary_op import Series
    from .metric_name import MetricName
    mn = base_mn.copy() if base_mn is not None else MetricName()
    vals = np.ones(n_grid)
    for row in values_rows:
        vals[np.isnan(np.asarray(row, np.float64))] = np.nan
    return [Series(mn, vals)]


def finalize_rollup_metric_name(mn, func_name, keep_metric_names=False,
                                rollup_tag=""):
    """The naming step of doRollupForTimeseries (eval.go:2009-2018),
    applied to a copy of the source MetricName: attach the rollup=<tag>
    label for multi-result expansions and reset the metric group unless
    the function (or the keep_metric_names modifier) keeps it."""
    mn = mn.copy()
    if rollup_tag:
        mn.add_tag("rollup", rollup_tag)
    if not keep_metric_names and \
            func_name not in ROLLUP_KEEP_METRIC_NAME_FUNCS:
        mn.reset_metric_group()
    return mn


def rollup_fake_plans(parent, start, end, step, tag="", **kwargs):
    """getRollupConfigs' expansion of the rollup_* pseudo-functions
    (rollup.go:436-516): returns [(rollup_tag, RollupPlan)].  tag narrows
    to one sub-config (the optional second arg of rollup*(q, "tag"))."""
    if parent not in ROLLUP_FAKE_FUNCS:
        raise VmGpuError(f"not a rollup_* pseudo-function: {parent!r}")
    pre, candle = ROLLUP_FAKE_FUNCS[parent]
    if candle:
        subs = {"open": "rollup_open", "close": "rollup_close",
                "low": "rollup_low", "high": "rollup_high"}
    else:
        subs = {"min": "min_over_time", "max": "max_over_time",
                "avg": "avg_over_time"}
    names = [tag] if tag else list(subs)
    out = []
    for t in names:
        if t not in subs:
            raise VmGpuError(f"unexpected rollup tag {t!r} for {parent}")
        out.append((t, RollupPlan(subs[t], start, end, step, pre_func=pre,
                                  parent_func=parent, **kwargs)))
    return out


def aggr_over_time_plans(func_names, start, end, step, **kwargs):
    """aggr_over_time(q, "fn1", ...) expansion (rollup.go:496-509): one
    plan per listed rollup function, tagged with its name."""
    return [(name, RollupPlan(name, start, end, step, parent_func=name,
                              **kwargs)) for name in func_names]


class VmGpuError(RuntimeError):
    pass


class _PlanC(ctypes.Structure):
    _fields_ = [
        ("func", ctypes.c_int32),
        ("aggr", ctypes.c_int32),
        ("start", ctypes.c_int64),
        ("end", ctypes.c_int64),
        ("step", ctypes.c_int64),
        ("window", ctypes.c_int64),
        ("lookback_delta", ctypes.c_int64),
        ("min_staleness_interval", ctypes.c_int64),
        ("max_staleness_interval", ctypes.c_int64),
        ("may_adjust_window", ctypes.c_int32),
        ("is_default_rollup", ctypes.c_int32),
        ("remove_counter_resets", ctypes.c_int32),
        ("drop_stale_nans", ctypes.c_int32),
        ("samples_scanned_per_call", ctypes.c_int32),
        ("skip_finalize", ctypes.c_int32),
        ("pre_func", ctypes.c_int32),
        ("arg", ctypes.c_double),
        ("arg2", ctypes.c_double),
    ]


_lib = None


def _load_lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise VmGpuError(
                f"HIP extension not built: {_LIB_PATH} is missing. "
                "Run __graft_entry__.build() (hipcc --offload-arch=gfx950). "
                "There is no CPU fallback in the product path.")
        _lib = ctypes.CDLL(_LIB_PATH)
        _lib.vmgpu_init.argtypes = [ctypes.POINTER(ctypes.c_int), ctypes.c_int]
        _lib.vmgpu_last_kernel_ms.argtypes = [ctypes.POINTER(ctypes.c_double)]
    return _lib


_inited = False


def init(device=0):
    """Select the GPU (one process per GPU).  Raises if no usable device."""
    global _inited
    lib = _load_lib()
    if _inited:
        return
    dev = ctypes.c_int(int(device))
    rc = lib.vmgpu_init(ctypes.byref(dev), 1)
    if rc != 0:
        raise VmGpuError(
            f"vmgpu_init(device={device}) failed with code {rc}: no usable "
            "HIP device. The engine requires an MI355X-class GPU; there is "
            "no CPU fallback.")
    _inited = True


def shutdown():
    global _inited
    if _inited:
        _load_lib().vmgpu_shutdown()
        _inited = False


class RollupPlan:
    """Mirror of rollupConfig construction in getRollupConfigs
    (rollup.go:374-516) + the EvalConfig fields that reach it."""

    def __init__(self, func, start, end, step, window=0, lookback_delta=0,
                 min_staleness_interval=0, arg=0.0, arg2=0.0, aggr="none",
                 skip_finalize=False, keep_stale_nans=False, pre_func="none",
                 parent_func=None):
        if func not in FUNC_IDS:
            raise VmGpuError(f"unsupported rollup function {func!r}")
        if step <= 0:
            raise VmGpuError(f"step must be positive; got {step}")
        if start > end:
            raise VmGpuError(f"start {start} exceeds end {end}")
        from . import limits as _limits
        _limits.validate_max_points(1 + (int(end) - int(start)) // int(step))
        self.func = func
        self.start = int(start)
        self.end = int(end)
        self.step = int(step)
        self.window = int(window)
        self.lookback_delta = int(lookback_delta)
        self.min_staleness_interval = int(min_staleness_interval)
        self.arg = float(arg)
        self.arg2 = float(arg2)
        self.aggr = aggr
        self.skip_finalize = skip_finalize
        self.keep_stale_nans = keep_stale_nans

        # rollup_* expansions (getRollupConfigs, rollup.go:436-516): flags
        # come from the PARENT pseudo-function, the evaluated func is the
        # min/max/avg (or candlestick) sub-config
        flags_func = parent_func or func
        self.pre_func = pre_func
        rcr = flags_func in REMOVE_COUNTER_RESETS_FUNCS
        # stalenessInterval = lookbackDelta (+window when set), rollup.go:380-387
        staleness = self.lookback_delta
        if staleness != 0:
            staleness += self.window
        self._c = _PlanC(
            func=FUNC_IDS[func],
            aggr=AGGR_IDS[aggr],
            start=self.start, end=self.end, step=self.step,
            window=self.window,
            lookback_delta=self.lookback_delta,
            min_staleness_interval=self.min_staleness_interval,
            max_staleness_interval=staleness if rcr else 0,
            may_adjust_window=1 if flags_func in CAN_ADJUST_WINDOW_FUNCS else 0,
            is_default_rollup=1 if func == "default_rollup" else 0,
            remove_counter_resets=1 if rcr else 0,
            drop_stale_nans=0 if (keep_stale_nans or func in KEEP_STALE_NANS_FUNCS) else 1,
            samples_scanned_per_call=SAMPLES_SCANNED_PER_CALL.get(flags_func, 0),
            skip_finalize=1 if skip_finalize else 0,
            pre_func=PRE_FUNC_IDS[pre_func],
            arg=self.arg, arg2=self.arg2)

    @property
    def n_grid(self):
        return 1 + (self.end - self.start) // self.step

    def timestamps(self):
        """getTimestamps (eval.go:234-254)."""
        return np.arange(self.start, self.end + 1, self.step, dtype=np.int64)


class SeriesBatch:
    """A decoded-series batch resident on the GPU (the RunParallel callback
    input, batched: CSR (timestamps[], values[]) + optional dense group ids)."""

    def __init__(self, ts, vals, offsets, group_ids=None, n_groups=0):
        init()
        lib = _load_lib()
        self.ts = np.ascontiguousarray(ts, dtype=np.int64)
        self.vals = np.ascontiguousarray(vals, dtype=np.float64)
        self.offsets = np.ascontiguousarray(offsets, dtype=np.uint64)
        self.n_series = len(self.offsets) - 1
        self.n_groups = int(n_groups)
        if group_ids is not None:
            self.group_ids = np.ascontiguousarray(group_ids, dtype=np.int32)
            gptr = self.group_ids.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))
        else:
            self.group_ids = None
            gptr = None
        handle = ctypes.c_uint64(0)
        errbuf = ctypes.create_string_buffer(256)
        rc = lib.vmgpu_batch_create(
            self.ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            self.vals.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            self.offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            ctypes.c_uint32(self.n_series), gptr,
            ctypes.c_uint32(self.n_groups), ctypes.byref(handle),
            errbuf, ctypes.c_size_t(len(errbuf)))
        if rc != 0:
            raise VmGpuError(f"vmgpu_batch_create failed ({rc}): "
                             f"{errbuf.value.decode()}")
        self.handle = handle.value
        self._perm = self._fetch_perm() if group_ids is not None else None

    def _fetch_perm(self):
        """Physical-row -> original-series map of a group-relayouted
        batch (vmgpu_batch_perm); None when identity."""
        lib = _load_lib()
        perm = np.empty(self.n_series, dtype=np.uint32)
        rc = lib.vmgpu_batch_perm(
            ctypes.c_uint64(self.handle),
            perm.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)))
        if rc != 0:
            raise VmGpuError(f"vmgpu_batch_perm failed ({rc})")
        if np.array_equal(perm, np.arange(self.n_series, dtype=np.uint32)):
            return None
        return perm

    @classmethod
    def from_blocks(cls, blocks, series_block_start, dedup_interval=0,
                    group_ids=None, n_groups=0):
        """Fused cold-cache fetch (§8f(1)): compressed-block payload ->
        device decode -> per-series merge+dedup -> resident batch.  The
        decoded columns never cross PCIe.

        blocks: list of dicts as for decode_blocks(); series i owns blocks
        [series_block_start[i], series_block_start[i+1]).
        """
        import math as _math
        init()
        lib = _load_lib()
        payload = bytearray()
        descs = (_BlockDescC * len(blocks))()
        total_rows = 0
        for i, b in enumerate(blocks):
            d = descs[i]
            d.ts_data_off = len(payload)
            payload.extend(b["ts_data"])
            d.ts_data_len = len(b["ts_data"])
            d.val_data_off = len(payload)
            payload.extend(b["val_data"])
            d.val_data_len = len(b["val_data"])
            d.out_off = total_rows
            d.min_timestamp = int(b["min_timestamp"])
            d.max_timestamp = int(b["max_timestamp"])
            d.first_value = int(b["first_value"])
            d.scale = int(b["scale"])
            d.e10 = _math.pow(10.0, abs(int(b["scale"])))
            d.rows = int(b["rows"])
            d.ts_mt = int(b["ts_mt"])
            d.val_mt = int(b["val_mt"])
            d.precision_bits = int(b["precision_bits"])
            total_rows += int(b["rows"])
        pl = np.frombuffer(bytes(payload), dtype=np.uint8) if payload else \
            np.zeros(1, dtype=np.uint8)
        sbs = np.ascontiguousarray(series_block_start, dtype=np.uint32)
        n_series = len(sbs) - 1
        out_offsets = np.zeros(n_series + 1, dtype=np.uint64)
        if group_ids is not None:
            gids = np.ascontiguousarray(group_ids, dtype=np.int32)
            gptr = gids.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))
        else:
            gids, gptr = None, None
        handle = ctypes.c_uint64(0)
        errbuf = ctypes.create_string_buffer(256)
        rc = lib.vmgpu_batch_create_from_blocks(
            pl.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            ctypes.c_uint64(len(payload)), descs,
            ctypes.c_uint32(len(blocks)), ctypes.c_uint64(total_rows),
            sbs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
            ctypes.c_uint32(n_series), ctypes.c_int64(int(dedup_interval)),
            gptr, ctypes.c_uint32(int(n_groups)),
            ctypes.byref(handle),
            out_offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            errbuf, ctypes.c_size_t(256))
        if rc != 0:
            raise VmGpuError(f"vmgpu_batch_create_from_blocks failed ({rc}):"
                             f" {errbuf.value.decode()}")
        self = cls.__new__(cls)
        self.ts = None
        self.vals = None
        self.offsets = out_offsets
        self.n_series = n_series
        self.n_groups = int(n_groups)
        self.group_ids = gids
        self.handle = handle.value
        self._last_rows = 0
        self._last_n_grid = 0
        self._perm = self._fetch_perm() if group_ids is not None else None
        return self

    @classmethod
    def from_packed(cls, packed, n_blocks, series_block_start,
                    dedup_interval=0, group_ids=None, n_groups=0):
        """Cold fetch through the NATIVE descriptor path
        (vmgpu_batch_create_packed): one contiguous packed block stream in,
        resident batch out.  The per-block walk happens in C — no
        host-language marshaling per block (the cgo production shape)."""
        init()
        lib = _load_lib()
        pl = np.frombuffer(packed, dtype=np.uint8) \
            if isinstance(packed, (bytes, bytearray, memoryview)) \
            else np.ascontiguousarray(packed, dtype=np.uint8)
        sbs = np.ascontiguousarray(series_block_start, dtype=np.uint32)
        n_series = len(sbs) - 1
        out_offsets = np.zeros(n_series + 1, dtype=np.uint64)
        if group_ids is not None:
            gids = np.ascontiguousarray(group_ids, dtype=np.int32)
            gptr = gids.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))
        else:
            gids, gptr = None, None
        handle = ctypes.c_uint64(0)
        errbuf = ctypes.create_string_buffer(256)
        rc = lib.vmgpu_batch_create_packed(
            pl.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            ctypes.c_uint64(len(pl)), ctypes.c_uint64(int(n_blocks)),
            sbs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
            ctypes.c_uint32(n_series), ctypes.c_int64(int(dedup_interval)),
            gptr, ctypes.c_uint32(int(n_groups)),
            ctypes.byref(handle),
            out_offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            errbuf, ctypes.c_size_t(256))
        if rc != 0:
            raise VmGpuError(f"vmgpu_batch_create_packed failed ({rc}):"
                             f" {errbuf.value.decode()}")
        self = cls.__new__(cls)
        self.ts = None
        self.vals = None
        self.offsets = out_offsets
        self.n_series = n_series
        self.n_groups = int(n_groups)
        self.group_ids = gids
        self.handle = handle.value
        self._last_rows = 0
        self._last_n_grid = 0
        self._perm = self._fetch_perm() if group_ids is not None else None
        return self

    def close(self):
        if self.handle:
            _load_lib().vmgpu_batch_destroy(ctypes.c_uint64(self.handle))
            self.handle = 0

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    def exec(self, plan, download=True, tracer=None, deadline=None,
             out_buf=None, counts_buf=None):
        """Evaluate plan over this batch.  Returns (out, counts,
        samples_scanned); out is [n_series x n_grid] or [n_groups x n_grid].
        With download=False the results stay on the device (use fetch_out).

        tracer: optional victoriametrics_amd.tracer.Tracer span — a child
        span records the kernel wall time (hipEvents), mirroring the
        reference's per-stage querytracer children.  deadline: optional
        limits.Deadline checked before the launch."""
        lib = _load_lib()
        n_grid = plan.n_grid
        grouped = plan.aggr != "none"
        if grouped and self.group_ids is None:
            raise VmGpuError("aggregate plan requires a batch with group_ids")
        out = counts = None
        optr = cptr = None
        if download:
            rows = self.n_groups if grouped else self.n_series
            out = out_buf if out_buf is not None \
                else np.empty((rows, n_grid), dtype=np.float64)
            optr = out.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
            if grouped:
                counts = counts_buf if counts_buf is not None \
                    else np.empty((rows, n_grid), dtype=np.float64)
                cptr = counts.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
        from . import limits as _limits
        if deadline is not None:
            deadline.check("rollup evaluation")
        _limits.check_rollup_memory(
            self.n_series, n_grid,
            grouped_rows=self.n_groups if grouped else None)
        scanned = ctypes.c_uint64(0)
        errbuf = ctypes.create_string_buffer(256)
        rc = lib.vmgpu_rollup_exec(ctypes.byref(plan._c),
                                   ctypes.c_uint64(self.handle), optr, cptr,
                                   ctypes.byref(scanned), errbuf,
                                   ctypes.c_size_t(len(errbuf)))
        if rc != 0:
            raise VmGpuError(f"vmgpu_rollup_exec failed ({rc}): "
                             f"{errbuf.value.decode()}")
        if tracer is not None:
            c = tracer.new_child(
                "vmgpu rollup %s over %d series -> %d rows x %d points",
                plan.func, self.n_series,
                self.n_groups if grouped else self.n_series, n_grid)
            c.donef("kernel %.3f ms, %d samples scanned",
                    last_kernel_ms(), scanned.value)
        self._last_rows = self.n_groups if grouped else self.n_series
        self._last_n_grid = n_grid
        if out is not None and not grouped and \
                getattr(self, "_perm", None) is not None:
            # physical row i holds original series _perm[i]
            orig = np.empty_like(out)
            orig[self._perm] = out
            out = orig
        return out, counts, scanned.value

    def fetch_out_into(self, dst, n_grid):
        """Download the last exec's output rows directly into a caller
        buffer (e.g. a marshal buffer's values section) — no intermediate
        copy.  dst: writable f64 array view [rows x n_grid]."""
        lib = _load_lib()
        assert dst.dtype == np.float64
        rows = dst.shape[0]
        rc = lib.vmgpu_batch_fetch_out(
            ctypes.c_uint64(self.handle),
            dst.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.c_size_t(dst.size), None, ctypes.c_size_t(0))
        if rc != 0:
            raise VmGpuError(f"vmgpu_batch_fetch_out failed ({rc})")
        return dst, None

    def fetch_out(self, rows, n_grid, with_counts=False):
        lib = _load_lib()
        out = np.empty((rows, n_grid), dtype=np.float64)
        counts = np.empty((rows, n_grid), dtype=np.float64) if with_counts else None
        rc = lib.vmgpu_batch_fetch_out(
            ctypes.c_uint64(self.handle),
            out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.c_size_t(out.size),
            counts.ctypes.data_as(ctypes.POINTER(ctypes.c_double)) if with_counts else None,
            ctypes.c_size_t(counts.size if with_counts else 0))
        if rc != 0:
            raise VmGpuError(f"vmgpu_batch_fetch_out failed ({rc})")
        return out, counts


def host_alloc(nbytes):
    """Pinned (hipHostMalloc) host buffer exposed as a uint8 numpy array —
    PCIe-rate H2D/D2H staging.  Returns (array, raw_ptr); free with
    host_free(raw_ptr).  The cgo production layer would pool these the way
    netstorage pools its result buffers."""
    init()
    lib = _load_lib()
    ptr = ctypes.c_void_p(0)
    rc = lib.vmgpu_host_alloc(ctypes.c_uint64(int(nbytes)), ctypes.byref(ptr))
    if rc != 0 or not ptr.value:
        raise VmGpuError(f"vmgpu_host_alloc({nbytes}) failed ({rc})")
    buf = np.ctypeslib.as_array(
        ctypes.cast(ptr, ctypes.POINTER(ctypes.c_uint8)), shape=(int(nbytes),))
    return buf, ptr.value


def host_free(raw_ptr):
    if raw_ptr:
        _load_lib().vmgpu_host_free(ctypes.c_void_p(raw_ptr))


def rollup_eval(plan, ts, vals, offsets, group_ids=None, n_groups=0):
    """One-shot evaluation (upload + exec + free) — the cgo-shim shape."""
    with SeriesBatch(ts, vals, offsets, group_ids, n_groups) as b:
        return b.exec(plan)


def aggr_finalize(aggr, values, counts):
    """finalizeTimeseries tail (aggr_incremental.go:141-168): applied on the
    host after the cross-shard all-reduce of skip_finalize outputs."""
    lib = _load_lib()
    v = np.ascontiguousarray(values, dtype=np.float64)
    c = np.ascontiguousarray(counts, dtype=np.float64) if counts is not None else None
    lib.vmgpu_aggr_finalize_host(
        ctypes.c_int32(AGGR_IDS[aggr]),
        v.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        c.ctypes.data_as(ctypes.POINTER(ctypes.c_double)) if c is not None else None,
        ctypes.c_uint64(v.size))
    return v


TOPK_SUMMARY_OPS = {"avg": 0, "min": 1, "max": 2, "median": 3, "last": 4}


def topk_range(batch, k, summary="avg", reverse=False, remaining=False):
    """getRangeTopKTimeseries over the batch's last evaluated output:
    returns (selected row ids in output order, remaining-sum row or None)."""
    lib = _load_lib()
    rows = batch._last_rows or batch.n_series
    # topk(inf, q) selects everything (lessWithNaNs tolerates any float k);
    # clamp BEFORE int() so non-finite / huge k cannot overflow
    kf = float(k)
    if kf != kf or kf <= 0:
        kk = 0
    else:
        kk = rows if kf >= rows else int(kf)
    sel = np.empty(max(kk, 1), dtype=np.int64)
    n_sel = ctypes.c_int64(0)
    rem = None
    rptr = None
    if remaining:
        n_grid = batch._last_n_grid
        rem = np.empty(n_grid, dtype=np.float64)
        rptr = rem.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_topk_range(
        ctypes.c_uint64(batch.handle), ctypes.c_double(k),
        ctypes.c_int32(TOPK_SUMMARY_OPS[summary]),
        ctypes.c_int32(1 if reverse else 0),
        sel.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        ctypes.byref(n_sel), rptr, errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_topk_range failed ({rc}): {errbuf.value.decode()}")
    return sel[:n_sel.value], rem


def topk_pointwise(batch, k, reverse=False):
    """newAggrFuncTopK over the batch's last evaluated output: returns the
    [rows x n_grid] matrix with all but the per-point top k NaN-filled.
    NOTE: mutates the device-resident output."""
    lib = _load_lib()
    rows = batch._last_rows
    n_grid = batch._last_n_grid
    out = np.empty((rows, n_grid), dtype=np.float64)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_topk_pointwise(
        ctypes.c_uint64(batch.handle), ctypes.c_double(k),
        ctypes.c_int32(1 if reverse else 0),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_topk_pointwise failed ({rc}): {errbuf.value.decode()}")
    return out


def histogram_quantile(phi, bucket_values, les, group_offsets, bounds=False):
    """transformHistogramQuantile over grouped le-bucket rows (rows sorted by
    (group, le), same-le pre-merged; the host label plumbing — vmrange->le
    conversion and group-key assignment — happens above this call)."""
    init()
    lib = _load_lib()
    bv = np.ascontiguousarray(bucket_values, dtype=np.float64)
    le = np.ascontiguousarray(les, dtype=np.float64)
    off = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    n_groups = len(off) - 1
    n_grid = bv.shape[1]
    out = np.empty((n_groups, n_grid), dtype=np.float64)
    lo = np.empty((n_groups, n_grid), dtype=np.float64) if bounds else None
    hi = np.empty((n_groups, n_grid), dtype=np.float64) if bounds else None
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_histogram_quantile(
        ctypes.c_double(phi),
        bv.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        le.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        off.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ctypes.c_uint32(n_groups), ctypes.c_int32(n_grid),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        lo.ctypes.data_as(ctypes.POINTER(ctypes.c_double)) if bounds else None,
        hi.ctypes.data_as(ctypes.POINTER(ctypes.c_double)) if bounds else None,
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_histogram_quantile failed ({rc}): "
                         f"{errbuf.value.decode()}")
    return (out, lo, hi) if bounds else (out, None, None)


HSTAT_MODES = {"avg": 0, "stddev": 1, "stdvar": 2}


def histogram_stat(mode, bucket_values, les, group_offsets):
    """histogram_avg/stddev/stdvar (transformHistogramAvg/Stddev/Stdvar)
    over the same CSR bucket layout as histogram_quantile."""
    init()
    lib = _load_lib()
    bv = np.ascontiguousarray(bucket_values, dtype=np.float64)
    le = np.ascontiguousarray(les, dtype=np.float64)
    off = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    n_groups = len(off) - 1
    n_grid = bv.shape[1]
    out = np.empty((n_groups, n_grid), dtype=np.float64)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_histogram_stat(
        ctypes.c_int32(HSTAT_MODES[mode] if isinstance(mode, str) else mode),
        bv.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        le.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        off.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ctypes.c_uint32(n_groups), ctypes.c_int32(n_grid),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_histogram_stat failed ({rc}): "
                         f"{errbuf.value.decode()}")
    return out


def histogram_share(le_req, bucket_values, les, group_offsets, bounds=False):
    """histogram_share (transformHistogramShare); le_req is the per-grid
    scalar row."""
    init()
    lib = _load_lib()
    req = np.ascontiguousarray(le_req, dtype=np.float64)
    bv = np.ascontiguousarray(bucket_values, dtype=np.float64)
    le = np.ascontiguousarray(les, dtype=np.float64)
    off = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    n_groups = len(off) - 1
    n_grid = bv.shape[1]
    out = np.empty((n_groups, n_grid), dtype=np.float64)
    lo = np.empty((n_groups, n_grid), dtype=np.float64) if bounds else None
    hi = np.empty((n_groups, n_grid), dtype=np.float64) if bounds else None
    errbuf = ctypes.create_string_buffer(256)
    dp = ctypes.POINTER(ctypes.c_double)
    rc = lib.vmgpu_histogram_share(
        req.ctypes.data_as(dp), bv.ctypes.data_as(dp),
        le.ctypes.data_as(dp),
        off.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ctypes.c_uint32(n_groups), ctypes.c_int32(n_grid),
        out.ctypes.data_as(dp),
        lo.ctypes.data_as(dp) if bounds else None,
        hi.ctypes.data_as(dp) if bounds else None,
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_histogram_share failed ({rc}): "
                         f"{errbuf.value.decode()}")
    return (out, lo, hi) if bounds else (out, None, None)


def last_kernel_ms():
    lib = _load_lib()
    ms = ctypes.c_double(0)
    rc = lib.vmgpu_last_kernel_ms(ctypes.byref(ms))
    if rc != 0:
        raise VmGpuError("vmgpu_last_kernel_ms failed")
    return ms.value


def device_info():
    lib = _load_lib()
    name = ctypes.create_string_buffer(128)
    hbm = ctypes.c_double(0)
    cus = ctypes.c_int(0)
    rc = lib.vmgpu_device_info(name, ctypes.c_size_t(len(name)),
                               ctypes.byref(hbm), ctypes.byref(cus))
    if rc != 0:
        raise VmGpuError("vmgpu_device_info failed")
    return {"name": name.value.decode(), "hbm_gib": hbm.value, "cus": cus.value}


class _BlockDescC(ctypes.Structure):
    _fields_ = [
        ("ts_data_off", ctypes.c_uint64),
        ("val_data_off", ctypes.c_uint64),
        ("out_off", ctypes.c_uint64),
        ("min_timestamp", ctypes.c_int64),
        ("max_timestamp", ctypes.c_int64),
        ("first_value", ctypes.c_int64),
        ("e10", ctypes.c_double),
        ("ts_data_len", ctypes.c_uint32),
        ("val_data_len", ctypes.c_uint32),
        ("rows", ctypes.c_uint32),
        ("scale", ctypes.c_int32),
        ("ts_mt", ctypes.c_uint8),
        ("val_mt", ctypes.c_uint8),
        ("precision_bits", ctypes.c_uint8),
        ("_pad", ctypes.c_uint8),
    ]


def decode_blocks(blocks):
    """GPU block decode (Block.UnmarshalData + codecs + decimal->float).

    blocks: list of dicts with keys
      ts_data (bytes, post-zstd), ts_mt, min_timestamp, max_timestamp,
      val_data (bytes, post-zstd), val_mt, first_value, scale,
      precision_bits, rows
    Returns (ts int64 array, vals f64 array, offsets) — the decoded CSR.
    """
    import math as _math
    init()
    lib = _load_lib()
    payload = bytearray()
    descs = (_BlockDescC * len(blocks))()
    total_rows = 0
    for i, b in enumerate(blocks):
        d = descs[i]
        d.ts_data_off = len(payload)
        payload.extend(b["ts_data"])
        d.ts_data_len = len(b["ts_data"])
        d.val_data_off = len(payload)
        payload.extend(b["val_data"])
        d.val_data_len = len(b["val_data"])
        d.out_off = total_rows
        d.min_timestamp = int(b["min_timestamp"])
        d.max_timestamp = int(b["max_timestamp"])
        d.first_value = int(b["first_value"])
        d.scale = int(b["scale"])
        d.e10 = _math.pow(10.0, abs(int(b["scale"])))
        d.rows = int(b["rows"])
        d.ts_mt = int(b["ts_mt"])
        d.val_mt = int(b["val_mt"])
        d.precision_bits = int(b["precision_bits"])
        total_rows += int(b["rows"])
    pl = np.frombuffer(bytes(payload), dtype=np.uint8) if payload else \
        np.zeros(1, dtype=np.uint8)
    out_ts = np.empty(total_rows, dtype=np.int64)
    out_vals = np.empty(total_rows, dtype=np.float64)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_decode_blocks(
        pl.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        ctypes.c_uint64(len(payload)), descs, ctypes.c_uint32(len(blocks)),
        ctypes.c_uint64(total_rows),
        out_ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        out_vals.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_decode_blocks failed ({rc}): "
                         f"{errbuf.value.decode()}")
    offsets = np.zeros(len(blocks) + 1, dtype=np.uint64)
    for i, b in enumerate(blocks):
        offsets[i + 1] = offsets[i] + int(b["rows"])
    return out_ts, out_vals, offsets


def binop_eval(op, is_bool, drop_nan_right, left_mat, left_idx, right_mat,
               right_idx, fill_left=None, fill_right=None):
    """Pairwise binary-op values loop (binary_op.go:162-236) on device.

    left_mat/right_mat: [rows x n_grid] f64; pair p = (left_idx[p],
    right_idx[p]).  Returns [n_pairs x n_grid]."""
    init()
    lib = _load_lib()
    left_mat = np.ascontiguousarray(left_mat, dtype=np.float64)
    right_mat = np.ascontiguousarray(right_mat, dtype=np.float64)
    li = np.ascontiguousarray(left_idx, dtype=np.uint32)
    ri = np.ascontiguousarray(right_idx, dtype=np.uint32)
    n_pairs = len(li)
    n_grid = left_mat.shape[1]
    out = np.empty((n_pairs, n_grid), dtype=np.float64)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_binop_eval(
        ctypes.c_int32(int(op)), ctypes.c_int32(1 if is_bool else 0),
        ctypes.c_int32(1 if drop_nan_right else 0),
        left_mat.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.c_uint32(left_mat.shape[0]),
        li.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        right_mat.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.c_uint32(right_mat.shape[0]),
        ri.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        ctypes.c_uint32(n_pairs), ctypes.c_uint32(n_grid),
        ctypes.c_int32(0 if fill_left is None else 1),
        ctypes.c_double(fill_left if fill_left is not None else 0.0),
        ctypes.c_int32(0 if fill_right is None else 1),
        ctypes.c_double(fill_right if fill_right is not None else 0.0),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_binop_eval failed ({rc}): "
                         f"{errbuf.value.decode()}")
    return out


def binop_mask(mode, left_mat, left_group, right_mat, group_offsets):
    """Set-op masking (and/if, unless/ifnot, default fill) on device;
    left_mat modified in place."""
    init()
    lib = _load_lib()
    left_mat_c = np.ascontiguousarray(left_mat, dtype=np.float64)
    right_mat = np.ascontiguousarray(right_mat, dtype=np.float64)
    lg = np.ascontiguousarray(left_group, dtype=np.uint32)
    go = np.ascontiguousarray(group_offsets, dtype=np.uint32)
    rows = np.arange(right_mat.shape[0], dtype=np.uint32)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_binop_mask(
        ctypes.c_int32(int(mode)),
        left_mat_c.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.c_uint32(left_mat_c.shape[0]),
        lg.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        right_mat.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.c_uint32(right_mat.shape[0]),
        go.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        ctypes.c_uint32(len(go) - 1),
        rows.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        ctypes.c_uint32(left_mat_c.shape[1]),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_binop_mask failed ({rc}): "
                         f"{errbuf.value.decode()}")
    left_mat[:] = left_mat_c
    return left_mat


def binop_or(left_mat, right_mat, lgroup_offsets, lgroup_rows,
             rgroup_offsets, rgroup_rows, can_merge, merge_offsets):
    """`or` merge walk (binary_op.go:645) on device; both matrices
    modified in place."""
    init()
    lib = _load_lib()
    lm = np.ascontiguousarray(left_mat, dtype=np.float64)
    rm = np.ascontiguousarray(right_mat, dtype=np.float64)
    lo = np.ascontiguousarray(lgroup_offsets, dtype=np.uint32)
    lr = np.ascontiguousarray(lgroup_rows, dtype=np.uint32)
    ro = np.ascontiguousarray(rgroup_offsets, dtype=np.uint32)
    rr = np.ascontiguousarray(rgroup_rows, dtype=np.uint32)
    cm = np.ascontiguousarray(can_merge, dtype=np.uint8)
    mo = np.ascontiguousarray(merge_offsets, dtype=np.uint64)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_binop_or(
        lm.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.c_uint32(lm.shape[0]),
        rm.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.c_uint32(rm.shape[0]),
        lo.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        lr.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        ro.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        rr.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        cm.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        mo.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ctypes.c_uint64(cm.size), ctypes.c_uint32(len(mo)),
        ctypes.c_uint32(lm.shape[1]),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_binop_or failed ({rc}): "
                         f"{errbuf.value.decode()}")
    left_mat[:] = lm
    right_mat[:] = rm
    return left_mat, right_mat


def merge_blocks(ts, vals, block_offsets, series_block_start,
                 dedup_interval=0):
    """GPU per-series k-way merge of decoded blocks + dedup
    (netstorage.go:564 mergeSortBlocks + dedup.go:29 DeduplicateSamples).

    ts/vals: all decoded samples, block b = [block_offsets[b],
    block_offsets[b+1]); series s owns blocks [series_block_start[s],
    series_block_start[s+1]).  Returns (out_ts, out_vals, out_offsets):
    merged samples packed densely per series with n_series+1 offsets.
    """
    init()
    lib = _load_lib()
    ts = np.ascontiguousarray(ts, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    boff = np.ascontiguousarray(block_offsets, dtype=np.uint64)
    sbs = np.ascontiguousarray(series_block_start, dtype=np.uint32)
    n_blocks = len(boff) - 1
    n_series = len(sbs) - 1
    cap = int(boff[-1])
    out_ts = np.empty(max(cap, 1), dtype=np.int64)
    out_vals = np.empty(max(cap, 1), dtype=np.float64)
    out_off = np.zeros(n_series + 1, dtype=np.uint64)
    out_cnt = np.zeros(max(n_series, 1), dtype=np.uint64)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_merge_blocks(
        ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        vals.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        boff.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ctypes.c_uint32(n_blocks),
        sbs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        ctypes.c_uint32(n_series), ctypes.c_int64(int(dedup_interval)),
        out_ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        out_vals.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        out_off.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        out_cnt.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_merge_blocks failed ({rc}): "
                         f"{errbuf.value.decode()}")
    total = int(out_off[n_series])
    return out_ts[:total], out_vals[:total], out_off


def _go_mod(a, b):
    """Go's % (truncated toward zero), vs Python's floored %."""
    r = abs(a) % abs(b)
    return -r if a < 0 else r


def align_start_end(start, end, step):
    """alignStartEnd (eval.go:103-112): round start down and end up to
    step multiples — with Go's truncated %, which differs from Python's
    floored % for negative timestamps."""
    start = start - _go_mod(start, step)
    adjust = _go_mod(end, step)
    if adjust > 0:
        end += step - adjust
    return start, end


MAX_SILENCE_INTERVAL_MS = 5 * 60 * 1000  # maxSilenceInterval (eval.go:1920)


def rollup_subquery(outer_func, start, end, step, window, sq_step,
                    inner_eval, lookback_delta=0, arg=0.0, arg2=0.0,
                    aggr="none", group_ids=None, n_groups=0):
    """evalRollupFuncWithSubquery (eval.go:1033-1100): evaluate the inner
    expression on a finer, extended grid via `inner_eval(sq_start, sq_end,
    sq_step) -> [n_series x m] f64 grid values`, strip NaN points per series
    (removeNanValues, eval.go:1150), then run the SAME rollup machinery over
    the resident grid — no separate kernel.

    Returns (out, counts, samples_scanned) like rollup_eval."""
    if sq_step == 0:
        sq_step = step
    sq_start = start - (window + sq_step + MAX_SILENCE_INTERVAL_MS)
    sq_end = end + sq_step
    sq_start, sq_end = align_start_end(sq_start, sq_end, sq_step)
    inner_vals = np.asarray(inner_eval(sq_start, sq_end, sq_step),
                            dtype=np.float64)
    sq_ts = np.arange(sq_start, sq_end + 1, sq_step, dtype=np.int64)
    n_series, m = inner_vals.shape
    assert m == len(sq_ts), "inner grid shape mismatch"
    # removeNanValues per series -> CSR
    keep = ~np.isnan(inner_vals)
    counts_per_series = keep.sum(axis=1)
    offsets = np.zeros(n_series + 1, dtype=np.uint64)
    np.cumsum(counts_per_series, out=offsets[1:])
    vals = inner_vals[keep]
    ts = np.broadcast_to(sq_ts, inner_vals.shape)[keep].astype(np.int64)
    plan = RollupPlan(outer_func, start, end, step, window=window,
                      lookback_delta=lookback_delta, arg=arg, arg2=arg2,
                      aggr=aggr,
                      # subquery input is already stale-free grid data
                      keep_stale_nans=True)
    return rollup_eval(plan, ts, vals, offsets, group_ids=group_ids,
                       n_groups=n_groups)


COLAGG_OPS = {
    "median": 0, "quantile": 1, "mad": 2, "stddev": 3, "stdvar": 4,
    "mode": 5, "distinct": 6, "share": 7, "zscore": 8, "iqr_bounds": 9,
    "sum": 10, "min": 11, "max": 12, "avg": 13, "count": 14, "sum2": 15,
    "geomean": 16, "group": 17,
}


def colagg(op, values, group_rows, group_offsets, phi=0.0):
    """Non-incremental cross-series aggregate (aggr.go long tail) over
    member rows per group.  Returns out [n_groups x n_grid] for reducing
    ops, (lower, upper) for iqr_bounds, or a rewritten values matrix for
    share/zscore."""
    init()
    lib = _load_lib()
    opid = COLAGG_OPS[op] if isinstance(op, str) else int(op)
    v = np.ascontiguousarray(values, dtype=np.float64)
    gr = np.ascontiguousarray(group_rows, dtype=np.uint32)
    go = np.ascontiguousarray(group_offsets, dtype=np.uint64)
    n_series, n_grid = v.shape
    n_groups = len(go) - 1
    dp = ctypes.POINTER(ctypes.c_double)
    per_series = opid in (7, 8)
    bounds = opid == 9
    out = None if per_series else np.empty((n_groups, n_grid), np.float64)
    out2 = np.empty((n_groups, n_grid), np.float64) if bounds else None
    vout = np.empty_like(v) if per_series else None
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_colagg(
        ctypes.c_int32(opid), v.ctypes.data_as(dp),
        ctypes.c_uint32(n_series), ctypes.c_uint32(n_grid),
        gr.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        go.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        ctypes.c_uint32(n_groups), ctypes.c_double(phi),
        out.ctypes.data_as(dp) if out is not None else None,
        out2.ctypes.data_as(dp) if out2 is not None else None,
        vout.ctypes.data_as(dp) if vout is not None else None,
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_colagg failed ({rc}): "
                         f"{errbuf.value.decode()}")
    if bounds:
        return out, out2
    if per_series:
        return vout
    return out


def colagg_filter(mode, values, group_of, b1, b2):
    """Outlier series filter (aggrFuncOutliersIQR / OutliersMAD): returns
    uint8 flags per series."""
    init()
    lib = _load_lib()
    v = np.ascontiguousarray(values, dtype=np.float64)
    gof = np.ascontiguousarray(group_of, dtype=np.int32)
    b1 = np.ascontiguousarray(b1, dtype=np.float64)
    b2 = np.ascontiguousarray(b2, dtype=np.float64)
    n_series, n_grid = v.shape
    flags = np.zeros(n_series, dtype=np.uint8)
    dp = ctypes.POINTER(ctypes.c_double)
    errbuf = ctypes.create_string_buffer(256)
    rc = lib.vmgpu_colagg_filter(
        ctypes.c_int32(0 if mode == "iqr" else 1), v.ctypes.data_as(dp),
        gof.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        ctypes.c_uint32(n_series), ctypes.c_uint32(n_grid),
        b1.ctypes.data_as(dp), b2.ctypes.data_as(dp),
        ctypes.c_uint32(b1.shape[0]),
        flags.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        errbuf, ctypes.c_size_t(256))
    if rc != 0:
        raise VmGpuError(f"vmgpu_colagg_filter failed ({rc}): "
                         f"{errbuf.value.decode()}")
    return flags


def quantiles_over_time_plans(phi_label, phis, start, end, step, **kwargs):
    """quantiles_over_time(q, "phi_label", phi1, ...) expansion
    (newRollupQuantiles, rollup.go): one quantile_over_time plan per phi,
    tagged phi_label=%g."""
    return [("%g" % phi,
             RollupPlan("quantile_over_time", start, end, step,
                        arg=float(phi), **kwargs)) for phi in phis]


def topk_merge_shards(shard_ids, shard_summaries, k, reverse=False):
    """Cross-shard topk merge (DESIGN §4 / SURVEY §5): each GPU selects its
    local k candidates (vmgpu_topk_range) with their summary values; the
    global top k is selected from the gathered candidates.  Matches the
    single-batch selection whenever per-shard candidate sets are complete
    (they are: a global top-k member is necessarily in its shard's local
    top k).  Returns [(shard, local_id), ...] of the global selection,
    ordered by summary (lessWithNaNs ordering: NaNs sort last for topk,
    first for bottomk — aggr.go:1262)."""
    cands = []
    for shard, (ids, sums) in enumerate(zip(shard_ids, shard_summaries)):
        for i, sv in zip(ids, sums):
            cands.append((shard, int(i), float(sv)))

    def sort_key(c):
        v = c[2]
        nan = v != v
        # topk: biggest first, NaN last; bottomk: smallest first, NaN last
        if reverse:
            return (nan, v)
        return (nan, -v)

    cands.sort(key=sort_key)
    kk = max(int(k), 0)
    return [(s, i) for s, i, _ in cands[:kk]]
