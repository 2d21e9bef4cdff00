"""CPU-side checks of the product C-ABI library: it builds, loads, exports
every symbol include/vmgpu.h declares, and fails LOUDLY without a GPU
(no silent CPU fallback).  No compute calls here (no GPU in CI)."""
import ctypes
import os
import re
import subprocess

import numpy as np
import pytest

from conftest import REPO_ROOT

LIB = os.path.join(REPO_ROOT, "victoriametrics_amd", "libvmgpu.so")
HEADER = os.path.join(REPO_ROOT, "include", "vmgpu.h")


def _ensure_built():
    if not os.path.exists(LIB):
        subprocess.run(
            ["sh", "build.sh"],
            cwd=os.path.join(REPO_ROOT, "victoriametrics_amd", "csrc"),
            check=True)
    return LIB


def test_lib_exports_all_header_symbols():
    lib = ctypes.CDLL(_ensure_built())
    header = open(HEADER).read()
    declared = re.findall(r"^(?:int|void)\s+(vmgpu_\w+)\(", header, re.M)
    assert len(declared) >= 9, f"header should declare the ABI; found {declared}"
    for sym in declared:
        assert hasattr(lib, sym), f"libvmgpu.so missing exported symbol {sym}"


def test_host_finalize_matches_oracle():
    """vmgpu_aggr_finalize_host is host-side product code (the post-allreduce
    finalize) — check against the oracle finalize on CPU."""
    import oracle
    _ensure_built()
    lib = ctypes.CDLL(LIB)
    rng = np.random.default_rng(7)
    for op_name, op_id in [("sum", 1), ("min", 2), ("max", 3), ("avg", 4),
                           ("count", 5), ("sum2", 6), ("geomean", 7),
                           ("group", 8)]:
        v = rng.random(64) * 10 + 0.1
        c = (rng.random(64) * 3).astype(np.int64).astype(np.float64)
        v_ref = v.copy()
        c_ref = c.copy()
        if op_name in ("count", "group"):
            # count/group track everything in the values row
            v[c == 0] = 0.0
            v_ref = v.copy()
        oracle.lib().vm_aggr_finalize(
            ctypes.c_int(op_id),
            v_ref.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            c_ref.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.c_int64(64))
        lib.vmgpu_aggr_finalize_host(
            ctypes.c_int32(op_id),
            v.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            c.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.c_uint64(64))
        assert np.array_equal(v, v_ref, equal_nan=True), op_name


def test_engine_fails_loudly_without_gpu():
    import torch
    from victoriametrics_amd import engine
    _ensure_built()
    if torch.cuda.is_available():
        pytest.skip("GPU present; loud-failure path not applicable")
    with pytest.raises(engine.VmGpuError):
        engine.init()


def test_plan_validation():
    from victoriametrics_amd import engine
    with pytest.raises(engine.VmGpuError):
        engine.RollupPlan("not_a_func", 0, 100, 10)
    with pytest.raises(engine.VmGpuError):
        engine.RollupPlan("rate", 0, 100, 0)
    with pytest.raises(engine.VmGpuError):
        engine.RollupPlan("rate", 100, 0, 10)
    p = engine.RollupPlan("rate", 0, 90, 10, window=300)
    assert p.n_grid == 10
    assert list(p.timestamps()) == list(range(0, 100, 10))
    # preFunc wiring (getRollupConfigs, rollup.go:374-516)
    assert p._c.remove_counter_resets == 1
    assert p._c.may_adjust_window == 1
    assert p._c.samples_scanned_per_call == 2
    assert p._c.drop_stale_nans == 1
    assert p._c.max_staleness_interval == 0  # lookback_delta == 0
    p2 = engine.RollupPlan("rate", 0, 90, 10, window=300, lookback_delta=50)
    assert p2._c.max_staleness_interval == 350  # lookback + window
    p3 = engine.RollupPlan("default_rollup", 0, 90, 10)
    assert p3._c.drop_stale_nans == 0
    assert p3._c.is_default_rollup == 1
    p4 = engine.RollupPlan("sum_over_time", 0, 90, 10)
    assert p4._c.remove_counter_resets == 0
    assert p4._c.may_adjust_window == 0
    assert p4._c.samples_scanned_per_call == 0


def test_func_id_tables_match_oracle():
    import oracle
    from victoriametrics_amd import engine
    assert engine.FUNC_IDS == oracle.FUNC_IDS
    assert engine.AGGR_IDS == oracle.AGGR_IDS
    assert engine.REMOVE_COUNTER_RESETS_FUNCS == oracle.REMOVE_COUNTER_RESETS_FUNCS
    assert engine.SAMPLES_SCANNED_PER_CALL == oracle.SAMPLES_SCANNED_PER_CALL
    assert engine.CAN_ADJUST_WINDOW_FUNCS == oracle.CAN_ADJUST_WINDOW_FUNCS


def test_tracer_and_limits():
    from victoriametrics_amd.tracer import Tracer, new_child, donef, done
    from victoriametrics_amd import limits
    import json as _json
    import pytest as _pytest
    # nil-safe: disabled tracer is None and helpers swallow everything
    qt = Tracer.new(False, "q %s", "x")
    assert qt is None
    assert new_child(qt, "child") is None
    donef(qt, "done")
    done(qt)
    qt = Tracer.new(True, "promql %s", "sum(rate(m[5m]))")
    c = qt.new_child("rollup over %d series", 4)
    c.donef("kernel %.3f ms", 1.25)
    qt.done()
    tree = _json.loads(qt.to_json())
    assert "promql" in tree["message"]
    assert tree["children"][0]["message"].endswith("kernel 1.250 ms")
    assert "duration_msec" in tree
    # limits
    limits.validate_max_points(100)
    with _pytest.raises(limits.QueryLimitError):
        limits.validate_max_points(limits.max_points_per_timeseries + 1)
    old = limits.max_memory_per_query
    limits.max_memory_per_query = 1000
    try:
        with _pytest.raises(limits.QueryLimitError):
            limits.check_rollup_memory(1000, 1000)
        limits.check_rollup_memory(2, 2)
    finally:
        limits.max_memory_per_query = old
    d = limits.Deadline(1e-9)
    import time as _t
    _t.sleep(0.01)
    with _pytest.raises(limits.QueryLimitError):
        d.check("test stage")


def test_rollup_plan_respects_max_points():
    from victoriametrics_amd import limits
    from victoriametrics_amd.engine import RollupPlan, VmGpuError
    import pytest as _pytest
    with _pytest.raises(limits.QueryLimitError):
        RollupPlan("rate", 0, limits.max_points_per_timeseries * 2_000, 1_000)


def test_plan_flag_wiring():
    """getRollupConfigs wiring pins (rollup.go:374-516 + eval.go:2108):
    per-function preFunc/window/staleness flags."""
    from victoriametrics_amd.engine import RollupPlan
    p = RollupPlan("rate", 0, 10_000, 1_000, window=5_000,
                   lookback_delta=300_000)
    assert p._c.remove_counter_resets == 1
    assert p._c.may_adjust_window == 1
    assert p._c.samples_scanned_per_call == 2
    assert p._c.drop_stale_nans == 1
    # stalenessInterval = lookbackDelta + window when lookback != 0
    assert p._c.max_staleness_interval == 305_000
    p2 = RollupPlan("rate", 0, 10_000, 1_000, window=5_000)
    assert p2._c.max_staleness_interval == 0
    d = RollupPlan("default_rollup", 0, 10_000, 1_000)
    assert d._c.is_default_rollup == 1
    assert d._c.drop_stale_nans == 0       # keeps stale markers
    assert d._c.samples_scanned_per_call == 1
    assert d._c.remove_counter_resets == 0
    a = RollupPlan("avg_over_time", 0, 10_000, 1_000, window=5_000)
    assert a._c.may_adjust_window == 0
    assert a._c.remove_counter_resets == 0
    assert a._c.samples_scanned_per_call == 0  # O(window) func
    s = RollupPlan("stale_samples_over_time", 0, 10_000, 1_000,
                   window=5_000)
    assert s._c.drop_stale_nans == 0
    q = RollupPlan("quantile_over_time", 0, 10_000, 1_000, window=5_000,
                   arg=0.9)
    assert q._c.arg == 0.9


def test_rollup_dispatch_covers_reference_map():
    from victoriametrics_amd import engine
    # the 80 keys of rollupFuncs (rollup.go:31-112): 70 device funcs
    # (FUNC_IDS incl. aliases), 7 rollup_* multi-result expansions
    # (rollup_fake_plans), aggr_over_time/quantiles_over_time plan
    # expansions, and the 2 documented host-side multi-output funcs
    reference_names = {
        "absent_over_time", "aggr_over_time", "ascent_over_time",
        "avg_over_time", "changes", "changes_prometheus",
        "count_eq_over_time", "count_gt_over_time", "count_le_over_time",
        "count_ne_over_time", "count_over_time", "count_values_over_time",
        "decreases_over_time", "default_rollup", "delta", "delta_prometheus",
        "deriv", "deriv_fast", "descent_over_time", "distinct_over_time",
        "duration_over_time", "first_over_time", "geomean_over_time",
        "histogram_over_time", "hoeffding_bound_lower",
        "hoeffding_bound_upper", "holt_winters", "idelta", "ideriv",
        "increase", "increase_prometheus", "increase_pure",
        "increases_over_time", "integrate", "iqr_over_time", "irate", "lag",
        "last_over_time", "lifetime", "mad_over_time", "max_over_time",
        "median_over_time", "min_over_time", "mode_over_time",
        "outlier_iqr_over_time", "predict_linear", "present_over_time",
        "quantile_over_time", "quantiles_over_time", "range_over_time",
        "rate", "rate_over_sum", "resets", "rollup", "rollup_candlestick",
        "rollup_delta", "rollup_deriv", "rollup_increase", "rollup_rate",
        "rollup_scrape_interval", "scrape_interval", "share_eq_over_time",
        "share_gt_over_time", "share_le_over_time", "stale_samples_over_time",
        "stddev_over_time", "stdvar_over_time", "sum_eq_over_time",
        "sum_gt_over_time", "sum_le_over_time", "sum_over_time",
        "sum2_over_time", "tfirst_over_time", "timestamp",
        "timestamp_with_name", "tlast_change_over_time", "tlast_over_time",
        "tmax_over_time", "tmin_over_time", "zscore_over_time"}
    expansions = {"aggr_over_time", "quantiles_over_time"}
    # host multi-series rollups (timeseriesMap side channel, rollup.go:1490)
    from victoriametrics_amd import rollup_multi
    host_multi_output = {"count_values_over_time", "histogram_over_time"}
    assert callable(rollup_multi.count_values_over_time)
    assert callable(rollup_multi.histogram_over_time)
    handled = (set(engine.FUNC_IDS) | set(engine.ROLLUP_FAKE_FUNCS) |
               expansions | host_multi_output)
    missing = reference_names - handled
    assert not missing, f"rollupFuncs without an entry point: {missing}"
    assert callable(engine.aggr_over_time_plans)
    assert callable(engine.quantiles_over_time_plans)
    assert callable(engine.rollup_fake_plans)


def test_finalize_rollup_metric_name():
    # doRollupForTimeseries naming (eval.go:2009-2018) +
    # rollupFuncsKeepMetricName (rollup.go:267-287)
    from victoriametrics_amd import engine
    from victoriametrics_amd.metric_name import MetricName
    src = MetricName(b"http_requests", [(b"job", b"api")])
    # rate resets the metric group
    mn = engine.finalize_rollup_metric_name(src, "rate")
    assert mn.metric_group == b"" and mn.get_tag_value("job") == b"api"
    assert src.metric_group == b"http_requests"  # source untouched
    # keep-list funcs keep it
    for f in ("avg_over_time", "default_rollup", "timestamp_with_name",
              "rollup_candlestick"):
        assert engine.finalize_rollup_metric_name(
            src, f).metric_group == b"http_requests", f
    # timestamp (without _with_name) resets
    assert engine.finalize_rollup_metric_name(
        src, "timestamp").metric_group == b""
    # keep_metric_names modifier overrides
    assert engine.finalize_rollup_metric_name(
        src, "rate", keep_metric_names=True).metric_group == b"http_requests"
    # the rollup tag for multi-result expansions
    mn = engine.finalize_rollup_metric_name(src, "rollup_candlestick",
                                            rollup_tag="high")
    assert mn.get_tag_value("rollup") == b"high"
    # coherence: every keep-list entry is a known rollup name
    known = (set(engine.FUNC_IDS) | set(engine.ROLLUP_FAKE_FUNCS) |
             {"quantiles_over_time"})
    assert engine.ROLLUP_KEEP_METRIC_NAME_FUNCS <= known


def test_plan_with_offset():
    # eval.go:954-1008: grid shifted back by the offset; reported
    # timestamps unshifted; candlestick auto-applies offset -step
    from victoriametrics_amd import engine
    start, end, step = 1_000_000, 1_600_000, 200_000
    plan, ts = engine.plan_with_offset("rate", start, end, step,
                                       offset_ms=3_600_000, window=300_000)
    assert plan._c.start == start - 3_600_000
    assert plan._c.end == end - 3_600_000
    assert list(ts) == [1_000_000, 1_200_000, 1_400_000, 1_600_000]
    # no offset: identity
    plan, ts = engine.plan_with_offset("rate", start, end, step,
                                       window=300_000)
    assert plan._c.start == start and list(ts)[0] == start
    # candlestick sub-plan: evaluation shifted FORWARD one step, reported
    # timestamps still the original grid
    plan, ts = engine.plan_with_offset("max_over_time", start, end, step,
                                       parent_func="rollup_candlestick")
    assert plan._c.start == start + step and plan._c.end == end + step
    assert list(ts) == [1_000_000, 1_200_000, 1_400_000, 1_600_000]


def test_aggregate_absent_over_time():
    from victoriametrics_amd import engine
    import math
    import numpy as np
    nan = math.nan
    # rows are per-series absent_over_time outputs
    rows = [[1.0, nan, 1.0, nan], [1.0, 1.0, nan, nan]]
    out = engine.aggregate_absent_over_time(rows, 4)
    v = out[0].values
    assert v[0] == 1.0
    assert all(math.isnan(x) for x in v[1:])
    # no input series at all -> all 1s
    out = engine.aggregate_absent_over_time([], 3)
    assert list(out[0].values) == [1.0, 1.0, 1.0]


def test_incremental_aggr_callbacks_covered():
    # incrementalAggrFuncCallbacksMap (aggr_incremental.go:18-66): the 9
    # incremental aggregates; 8 run fused in the rollup kernels
    # (AGGR_IDS), `any` via the representative-group-id rewrite
    from victoriametrics_amd import engine
    reference = {"sum", "min", "max", "avg", "count", "sum2", "geomean",
                 "any", "group"}
    fused = set(engine.AGGR_IDS) - {"none"}
    assert reference - fused == {"any"}
    assert callable(engine.any_representative_group_ids)


def test_packed_create_fails_loudly_without_init():
    """vmgpu_batch_create_packed through the C-ABI without vmgpu_init (or
    without a GPU): a loud error, never a silent fallback."""
    import torch
    from victoriametrics_amd import engine
    _ensure_built()
    if torch.cuda.is_available():
        pytest.skip("GPU present; loud-failure path not applicable")
    import oracle
    ts = np.arange(4, dtype=np.int64) * 15_000 + 1_000_000
    vals = np.arange(4, dtype=np.int64)
    packed, nb, sbs = oracle.pack_blocks(ts, vals,
                                         np.asarray([0, 4], np.uint64))
    with pytest.raises(engine.VmGpuError):
        engine.SeriesBatch.from_packed(packed, nb, sbs)


def test_host_alloc_symbols_exported():
    """The pinned-staging entry points exist in the built library (the cgo
    side links them)."""
    import ctypes
    lib = ctypes.CDLL(LIB)
    for sym in ("vmgpu_batch_create_packed", "vmgpu_host_alloc",
                "vmgpu_host_free"):
        assert hasattr(lib, sym), sym
