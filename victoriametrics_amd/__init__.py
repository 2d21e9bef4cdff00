"""victoriametrics_amd — MI355X-native rollup/aggregation engine for the
VictoriaMetrics vmselect query path.

Product surface (mirrors the reference's promql operator seam; see
include/vmgpu.h and DESIGN.md):

    from victoriametrics_amd import engine
    plan = engine.RollupPlan("rate", start, end, step, window=300_000)
    out, counts, scanned = engine.rollup_eval(plan, ts, vals, offsets)

The compute path is hand-written HIP for gfx950 (victoriametrics_amd/csrc/);
there is no CPU fallback — calls raise on machines without a usable GPU or
without the built extension.
"""
from . import engine  # noqa: F401

__all__ = ["engine"]
