import math
import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def decode_float(x):
    """Decode golden-vector float encoding ('nan'/'inf'/'-inf'/'stale_nan')."""
    if isinstance(x, str):
        if x == "nan":
            return math.nan
        if x == "inf":
            return math.inf
        if x == "-inf":
            return -math.inf
        if x == "stale_nan":
            import oracle
            return oracle.stale_nan()
        raise ValueError(f"bad float encoding: {x}")
    return float(x)


def decode_floats(xs):
    return [decode_float(x) for x in xs]


def assert_values_equal(got, expected, rel=1e-13, context=""):
    """testRowsEqual semantics (rollup_test.go:1564-1607): NaN matches NaN,
    finite values compared at relative precision."""
    assert len(got) == len(expected), f"{context}: len {len(got)} != {len(expected)}"
    for i, (g, e) in enumerate(zip(got, expected)):
        if math.isnan(e):
            assert math.isnan(g), f"{context}[{i}]: got {g}, want NaN"
            continue
        assert not math.isnan(g), f"{context}[{i}]: got NaN, want {e}"
        if e == 0:
            assert g == 0 or abs(g) < 1e-300, f"{context}[{i}]: got {g}, want 0"
        elif math.isinf(e):
            assert g == e, f"{context}[{i}]: got {g}, want {e}"
        else:
            assert abs(g - e) / abs(e) <= rel, f"{context}[{i}]: got {g}, want {e}"
