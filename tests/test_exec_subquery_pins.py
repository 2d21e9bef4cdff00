"""Subquery pins transcribed from TestExecSuccess (exec_test.go:8279-8323):
rate() over subquery grids, including subquery step defaulting, the
alignStartEnd extension and offset-shifted evaluation — the engine side is
engine.rollup_subquery (evalRollupFuncWithSubquery, eval.go:1033) feeding
the device rollup, so these run on GPU.  Expected arrays are the
reference's own outputs, verbatim."""
import numpy as np
import pytest

from victoriametrics_amd import engine

pytestmark = pytest.mark.gpu

START_MS = 1000_000
END_MS = 2000_000
STEP_MS = 200_000


@pytest.fixture(scope="module", autouse=True)
def _init():
    engine.init()


def inner_2000_minus_time(sq_start, sq_end, sq_step):
    ts = np.arange(sq_start, sq_end + 1, sq_step, dtype=np.int64)
    return (2000.0 - ts / 1e3).reshape(1, -1)


def _run(start, end, window, sq_step):
    out, _, _ = engine.rollup_subquery(
        "rate", start, end, STEP_MS, window, sq_step,
        inner_2000_minus_time)
    return np.asarray(out).ravel()


def test_rate_subquery_default_step():
    # `rate((2000-time())[100s:])` :8279 -> [5, 4, 3, 2, 1, 0]
    # (empty step defaults to the outer step, eval.go:1041-1043)
    got = _run(START_MS, END_MS, 100_000, 0)
    np.testing.assert_allclose(got, [5, 4, 3, 2, 1, 0], rtol=1e-12, atol=0)


def test_rate_subquery_100s_step():
    # `rate((2000-time())[100s:100s])` :8290 -> [0, 0, 6, 4, 2, 0]
    got = _run(START_MS, END_MS, 100_000, 100_000)
    np.testing.assert_allclose(got, [0, 0, 6, 4, 2, 0], rtol=1e-12, atol=0)


def test_rate_subquery_offset():
    # `rate((2000-time())[100s:100s] offset 100s)` :8301 ->
    # [0, 0, 7, 5, 3, 1]: the grid shifts back by the offset and results
    # are reported on the original timestamps (eval.go:954-1008)
    off = 100_000
    got = _run(START_MS - off, END_MS - off, 100_000, 100_000)
    np.testing.assert_allclose(got, [0, 0, 7, 5, 3, 1], rtol=1e-12, atol=0)


def test_rate_subquery_double_offset():
    # `rate((2000-time())[100s:100s] offset 100s)[:] offset 100s` :8312 ->
    # [0, 0, 0, 7, 5, 3]: the outer subquery shifts the whole inner
    # evaluation back another 100s; with default inner step the outer
    # [:] rollup is default_rollup (last value) over the shifted grid.
    off = 100_000

    def inner(sq_start, sq_end, sq_step):
        out, _, _ = engine.rollup_subquery(
            "rate", sq_start - off, sq_end - off, sq_step, 100_000, 100_000,
            inner_2000_minus_time)
        return np.asarray(out)

    out, _, _ = engine.rollup_subquery(
        "default_rollup", START_MS - off, END_MS - off, STEP_MS, 0, 0, inner)
    got = np.asarray(out).ravel()
    np.testing.assert_allclose(got, [0, 0, 0, 7, 5, 3], rtol=1e-12, atol=0)
