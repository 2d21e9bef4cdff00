"""Pins the codec/decimal/merge oracle against the reference's own test
vectors (tests/golden/codec_golden.json ← lib/encoding, lib/decimal,
lib/storage/dedup, netstorage merge tests), plus round-trip properties with
the write path (marshalInt64Array) mirroring lib/storage block marshaling."""
import json
import math
import os

import numpy as np
import pytest

from conftest import decode_float, decode_floats

import oracle

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "codec_golden.json")))


@pytest.mark.parametrize("case", GOLDEN["nearest_delta"],
                         ids=lambda c: f"nd_{c['va'][:3]}_{c['pb']}")
def test_nearest_delta_marshal_pins(case):
    data, first = oracle.marshal_nearest_delta(case["va"], case["pb"])
    assert first == case["first"]
    assert data.hex() == case["hex"]
    if len(case["va"]) >= 1:
        back = oracle.unmarshal_nearest_delta(data, first, len(case["va"]))
        if case["pb"] >= 64:
            assert list(back) == case["va"]


@pytest.mark.parametrize("case", GOLDEN["nearest_delta2"],
                         ids=lambda c: f"nd2_{c['va'][:3]}_{c['pb']}")
def test_nearest_delta2_marshal_pins(case):
    data, first = oracle.marshal_nearest_delta(case["va"], case["pb"], delta2=True)
    assert first == case["first"]
    assert data.hex() == case["hex"]


def test_nearest_delta_roundtrip_lossless():
    rng = np.random.default_rng(42)
    for n in (1, 2, 3, 10, 1000, 8192):
        a = rng.integers(-10**12, 10**12, n)
        data, first = oracle.marshal_nearest_delta(a, 64)
        back = oracle.unmarshal_nearest_delta(data, first, n)
        assert np.array_equal(a, back), f"n={n}"
        if n >= 2:
            data2, first2 = oracle.marshal_nearest_delta(a, 64, delta2=True)
            back2 = oracle.unmarshal_nearest_delta(data2, first2, n, delta2=True)
            assert np.array_equal(a, back2), f"delta2 n={n}"


def test_varint_roundtrip():
    vals = [0, 1, -1, 63, -64, 64, -65, 127, 128, 2**20, -2**20, 2**40,
            -2**40, 2**62, -2**62, 2**63 - 1, -2**63]
    data = oracle.marshal_varint64s(vals)
    back, used = oracle.unmarshal_varint64s(data, len(vals))
    assert used == len(data)
    assert list(back) == vals


@pytest.mark.parametrize("case", GOLDEN["is_const"])
def test_is_const(case):
    a = np.asarray(case["a"], dtype=np.int64)
    got = oracle._codec_lib().vm_is_const(
        a.ctypes.data_as(oracle.ctypes.POINTER(oracle.ctypes.c_int64)) if len(a) else None,
        oracle.ctypes.c_int64(len(a)))
    assert bool(got) == case["ok"], case


@pytest.mark.parametrize("case", GOLDEN["is_delta_const"])
def test_is_delta_const(case):
    a = np.asarray(case["a"], dtype=np.int64)
    got = oracle._codec_lib().vm_is_delta_const(
        a.ctypes.data_as(oracle.ctypes.POINTER(oracle.ctypes.c_int64)) if len(a) else None,
        oracle.ctypes.c_int64(len(a)))
    assert bool(got) == case["ok"], case


@pytest.mark.parametrize("case", GOLDEN["is_gauge"])
def test_is_gauge(case):
    a = np.asarray(case["a"], dtype=np.int64)
    got = oracle._codec_lib().vm_is_gauge(
        a.ctypes.data_as(oracle.ctypes.POINTER(oracle.ctypes.c_int64)) if len(a) else None,
        oracle.ctypes.c_int64(len(a)))
    assert bool(got) == case["ok"], case


@pytest.mark.parametrize("case", GOLDEN["positive_float_to_decimal"],
                         ids=lambda c: f"pftd_{c['f']}")
def test_positive_float_to_decimal_pins(case):
    v, e = oracle.positive_float_to_decimal(float(case["f"]))
    assert (v, e) == (case["v"], case["e"]), case


def test_from_float_specials():
    assert oracle.decimal_from_float(0.0) == (0, 0)
    assert oracle.decimal_from_float(math.inf) == (2**63 - 1, 0)
    assert oracle.decimal_from_float(-math.inf) == (-2**63, 0)
    assert oracle.decimal_from_float(oracle.stale_nan()) == (2**63 - 2, 0)
    v, e = oracle.decimal_from_float(-0.001130435)
    assert (v, e) == (-1130435, -9)


def test_decimal_to_float_roundtrip():
    rng = np.random.default_rng(3)
    for f in list(rng.normal(0, 1e6, 200)) + [1.0, -1.0, 0.1, 123.4567,
                                              1e300, -1e300, 1e-300]:
        v, e = oracle.decimal_from_float(float(f))
        back = oracle.decimal_to_float(v, e)
        if f != 0:
            assert abs(back - f) <= abs(f) * 1e-11, (f, v, e, back)


def test_append_decimal_to_float_specials():
    va = [123, 2**63 - 1, -2**63, 2**63 - 2, -5]
    out = oracle.decimal_append_to_float(va, 1)
    assert out[0] == 1230.0
    assert out[1] == math.inf
    assert out[2] == -math.inf
    assert oracle.lib().vm_is_stale_nan(out[3])
    assert out[4] == -50.0


def test_marshal_int64_array_types():
    # const
    data, mt, first = oracle.marshal_int64_array([7] * 100)
    assert (mt, first, data) == (oracle.MT_CONST, 7, b"")
    assert list(oracle.unmarshal_int64_array(data, 100, mt, first)) == [7] * 100
    # delta const
    a = list(range(0, 500, 5))
    data, mt, first = oracle.marshal_int64_array(a)
    assert mt == oracle.MT_DELTA_CONST and first == 0
    assert list(oracle.unmarshal_int64_array(data, len(a), mt, first)) == a
    # counter -> delta2 (+zstd for big blocks)
    rng = np.random.default_rng(8428)
    c = np.cumsum(rng.integers(100, 200, 4096))
    data, mt, first = oracle.marshal_int64_array(c, 64)
    assert mt in (oracle.MT_ZSTD_NEAREST_DELTA2, oracle.MT_NEAREST_DELTA2)
    assert np.array_equal(oracle.unmarshal_int64_array(data, len(c), mt, first), c)
    # gauge -> delta (+zstd)
    g = (np.cumsum(rng.standard_normal(4096)) * 1000).astype(np.int64)
    data, mt, first = oracle.marshal_int64_array(g, 64)
    assert mt in (oracle.MT_ZSTD_NEAREST_DELTA, oracle.MT_NEAREST_DELTA)
    assert np.array_equal(oracle.unmarshal_int64_array(data, len(g), mt, first), g)
    # small block never zstd (minCompressibleBlockSize=128)
    s = np.cumsum(rng.integers(100, 200, 8))
    data, mt, first = oracle.marshal_int64_array(s, 64)
    assert mt in (oracle.MT_NEAREST_DELTA2, oracle.MT_NEAREST_DELTA)


def test_lossy_precision_roundtrip_bounds():
    # precisionBits=p keeps values within 2^-p relative error (encoding contract)
    rng = np.random.default_rng(5)
    a = np.cumsum(rng.integers(1000, 2000, 1000))
    for pb in (8, 16, 32):
        data, mt, first = oracle.marshal_int64_array(a, pb)
        back = oracle.unmarshal_int64_array(data, len(a), mt, first)
        rel = np.abs(back - a) / np.maximum(np.abs(a), 1)
        assert rel.max() <= 2.0 ** (1 - pb), (pb, rel.max())


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["needs_dedup_via_dedup"]))
def test_needs_dedup(i, case):
    ts = np.asarray(case["ts"], dtype=np.int64)
    vals = np.arange(len(ts), dtype=np.float64)
    t2, v2 = oracle.deduplicate_samples(ts, vals, case["interval"])
    changed = len(t2) != len(ts) or not np.array_equal(t2, ts)
    assert changed == case["changed"], case


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["dedup_identical_ts"]))
def test_dedup_identical_timestamps(i, case):
    vals = decode_floats(case["vals"])
    t2, v2 = oracle.deduplicate_samples(case["ts"], vals, case["interval"])
    assert list(t2) == case["exp_ts"], case
    exp = decode_floats(case["exp_vals"])
    for got, want in zip(v2, exp):
        if math.isnan(want):
            assert oracle.lib().vm_is_stale_nan(got)
        else:
            assert got == want, case


@pytest.mark.parametrize("i,case", enumerate(GOLDEN["merge_sort_blocks"]))
def test_merge_sort_blocks(i, case):
    blocks = [(b["ts"], decode_floats(b["vals"])) for b in case["blocks"]]
    t, v = oracle.merge_sort_blocks(blocks, case["dedup"])
    assert list(t) == case["exp_ts"], f"case {i}"
    assert list(v) == decode_floats(case["exp_vals"]), f"case {i}"


def test_block_marshal_pipeline_roundtrip():
    """End-to-end block write→read: values through FromFloat/decimal scale +
    marshalInt64Array (MarshalValues), timestamps through marshalInt64Array
    (MarshalTimestamps), both at precisionBits=64 — mirrors
    lib/storage/block.go MarshalData/UnmarshalData."""
    rng = np.random.default_rng(11)
    n = 2000
    ts = 1_000_000_000_000 + np.cumsum(rng.integers(10_000, 20_000, n))
    vals_f = np.round(np.cumsum(rng.random(n) * 10), 3)
    va, scale = oracle.float_to_decimal(vals_f)
    tdata, tmt, tfirst = oracle.marshal_int64_array(ts, 64)
    vdata, vmt, vfirst = oracle.marshal_int64_array(va, 64)
    ts_back = oracle.unmarshal_int64_array(tdata, n, tmt, tfirst)
    va_back = oracle.unmarshal_int64_array(vdata, n, vmt, vfirst)
    assert np.array_equal(ts_back, ts)
    assert np.array_equal(va_back, va)
    f_back = oracle.decimal_append_to_float(va_back, scale)
    assert np.allclose(f_back, vals_f, rtol=1e-12, atol=0)


def test_ensure_non_decreasing_reference_vectors():
    # TestEnsureNonDecreasingSequence (lib/encoding/encoding_test.go:71-81)
    # verbatim — decode-side timestamp clamping
    cases = [
        ([], -1234, -34, []),
        ([123], -1234, -1234, [-1234]),
        ([123], -1234, 345, [345]),
        ([-23, -14], -23, -14, [-23, -14]),
        ([-23, -14], -25, 0, [-25, 0]),
        ([0, -1, 10, 5, 6, 7], 2, 8, [2, 2, 8, 8, 8, 8]),
        ([0, -1, 10, 5, 6, 7], -2, 8, [-2, -1, 8, 8, 8, 8]),
        ([0, -1, 10, 5, 6, 7], -2, 12, [-2, -1, 10, 10, 10, 12]),
        ([1, 2, 1, 3, 4, 5], 1, 5, [1, 2, 2, 3, 4, 5]),
    ]
    for a, vmin, vmax, want in cases:
        got = oracle.ensure_non_decreasing(
            np.asarray(a, np.int64), vmin, vmax)
        assert list(got) == want, (a, vmin, vmax, list(got), want)
