"""Query resource limits — mirrors of the knobs that shape the hot path
(SURVEY §5 config):

- max_points_per_timeseries  -search.maxPointsPerTimeseries (30k default;
  validated in getRollupConfigs, rollup.go:462 via
  rollupConfig.getMaxPoints)
- max_memory_per_query       -search.maxMemoryPerQuery (memory_limiter.go:
  the evaluator refuses queries whose intermediate buffers exceed the
  budget; error text mirrors eval.go's rollup memory estimate message)
- max_series_per_aggr_func   -search.maxSeriesPerAggrFunc (count_values)
- deadline                   searchutil.Deadline (netstorage.go:102): a
  wall-clock budget checked before expensive stages.
"""
import time

max_points_per_timeseries = 30_000
max_memory_per_query = 0          # 0 = unlimited, bytes otherwise
max_series_per_aggr_func = 100_000


class QueryLimitError(RuntimeError):
    pass


def validate_max_points(n_grid):
    """rollupConfig grid validation (rollup.go:462)."""
    if n_grid > max_points_per_timeseries:
        raise QueryLimitError(
            f"the number of points per timeseries {n_grid} exceeds "
            f"-search.maxPointsPerTimeseries={max_points_per_timeseries}; "
            "either increase the limit or reduce (end-start)/step")


def check_rollup_memory(n_series, n_grid, grouped_rows=None):
    """evalRollupFuncNoCache's rollup memory estimate (eval.go:1899 area):
    refuse before allocating when the output matrix exceeds the budget."""
    if max_memory_per_query <= 0:
        return
    rows = grouped_rows if grouped_rows is not None else n_series
    need = rows * n_grid * 8
    if need > max_memory_per_query:
        raise QueryLimitError(
            f"not enough memory for processing the query: the query "
            f"needs ~{need} bytes for {rows} series x {n_grid} points; "
            f"-search.maxMemoryPerQuery={max_memory_per_query}; either "
            "increase the limit or reduce the number of series/points")


class Deadline:
    """searchutil.Deadline mirror: wall-clock budget with the reference's
    exceeded-error shape."""

    def __init__(self, timeout_s, flag_hint="-search.maxQueryDuration"):
        self.deadline = time.monotonic() + timeout_s
        self.timeout_s = timeout_s
        self.flag_hint = flag_hint

    def exceeded(self):
        return time.monotonic() > self.deadline

    def check(self, what=""):
        if self.exceeded():
            raise QueryLimitError(
                f"cannot complete {what or 'the query'} in "
                f"{self.timeout_s:.3f} seconds; possible solutions: reduce "
                f"query load; increase {self.flag_hint}")
