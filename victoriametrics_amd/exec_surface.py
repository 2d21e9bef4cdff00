"""The Exec tail (exec.go:37-131): everything applied to evaluated series
before they leave as `netstorage.Result`s — first-point truncation for
instant queries, conditional sort by metric name, duplicate-output
detection, the max-response-series guard and round_digits rounding.
Host-side in the reference, host-side here."""
import math

import numpy as np

from .binary_op import remove_empty_series
from .decimal import round_to_decimal_digits

# maySortResults (exec.go:105-131): these top-level expressions already
# ordered their output; everything else is sorted by metric name
_SORTED_FUNCS = {"sort", "sort_desc", "limit_offset", "sort_by_label",
                 "sort_by_label_desc", "sort_by_label_numeric",
                 "sort_by_label_numeric_desc"}
_SORTED_AGGRS = {"topk", "bottomk", "outliersk",
                 "topk_max", "topk_min", "topk_avg", "topk_median",
                 "topk_last", "bottomk_max", "bottomk_min", "bottomk_avg",
                 "bottomk_median", "bottomk_last"}


def may_sort_results(kind, name=""):
    """kind: "func" | "aggr" | "binop" | anything else."""
    name = name.lower()
    if kind == "func":
        return name not in _SORTED_FUNCS
    if kind == "aggr":
        return name not in _SORTED_AGGRS
    if kind == "binop":
        return name != "or"  # issue 4763: keep `a or b` order
    return True


def _sorted_tags(mn):
    return sorted((bytes(k), bytes(v)) for k, v in mn.tags)


def metric_name_sort_key(mn):
    """metricNameLess (exec.go:170-192): group first, then the sorted tag
    list lexicographically (key, then value), shorter prefix first."""
    return (bytes(mn.metric_group), _sorted_tags(mn))


def sort_series_by_metric_name(series_list):
    series_list.sort(key=lambda s: metric_name_sort_key(s.mn))
    return series_list


class DuplicateOutputSeriesError(ValueError):
    pass


def timeseries_to_result(series_list, may_sort, round_digits=100,
                         first_point_only=False, max_response_series=0):
    """timeseriesToResult + the Exec tail (exec.go:70-103): returns the
    final Series list.  Raises DuplicateOutputSeriesError on duplicate
    naming and ValueError past max_response_series."""
    if first_point_only:
        for s in series_list:
            s.values = np.asarray(s.values, np.float64)[:1]
    series_list = remove_empty_series(series_list)
    if may_sort:
        sort_series_by_metric_name(series_list)
    seen = set()
    for s in series_list:
        k = s.mn.marshal_sorted()
        if k in seen:
            raise DuplicateOutputSeriesError(
                "duplicate output timeseries: %r" % (s.mn,))
        seen.add(k)
    if 0 < max_response_series < len(series_list):
        raise ValueError(
            "the response contains more than -search.maxResponseSeries=%d "
            "time series: %d series" % (max_response_series,
                                        len(series_list)))
    if round_digits < 100:
        for s in series_list:
            s.values = round_to_decimal_digits(s.values, round_digits)
    return series_list
