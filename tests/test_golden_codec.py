"""Oracle vs the reference's own golden vectors (tests/golden/*.json).

Vector provenance (data transcribed from the reference's tests):
  encoder_vectors.json  <- src/dbnode/encoding/m3tsz/encoder_test.go:54-393,
                           iterator_test.go:44-385
  production_streams.json <- encoder_benchmark_test.go:36-47 (10 streams),
                           iterator_test.go:400 (decode-regression stream)
"""
import base64
import ctypes
import json
import os
import struct

import numpy as np
import pytest

import oracle

GOLDEN = os.path.join(os.path.dirname(__file__), "golden")


@pytest.fixture(scope="module")
def vectors():
    with open(os.path.join(GOLDEN, "encoder_vectors.json")) as f:
        return json.load(f)


@pytest.fixture(scope="module")
def production():
    with open(os.path.join(GOLDEN, "production_streams.json")) as f:
        return json.load(f)


def _run_writer(fn, *args):
    out = np.zeros(64, dtype=np.uint8)
    pos = ctypes.c_int32(0)
    n = fn(*args, oracle._pu8(out), 64, ctypes.byref(pos))
    assert n >= 0, n
    return list(out[:n]), pos.value


def test_xxhash64():
    assert oracle.xxhash64(b"") == 0xEF46DB3751D8E999  # emptyAnnotationChecksum
    assert oracle.xxhash64(b"hello") == 0x26C7827D889F6DA3


def test_dod_unchanged(vectors):
    L = oracle.lib()
    for case in vectors["dod_unchanged"]["cases"]:
        b, p = _run_writer(L.oracle_test_write_dod_unchanged, 0,
                           case["delta_ns"], case["unit"])
        assert b == case["bytes"] and p == case["pos"], case


def test_dod_changed(vectors):
    L = oracle.lib()
    for case in vectors["dod_changed"]["cases"]:
        b, p = _run_writer(L.oracle_test_write_dod_changed, 0, case["delta_ns"])
        assert b == case["bytes"] and p == case["pos"], case


def test_write_xor(vectors):
    L = oracle.lib()
    for case in vectors["write_xor"]["cases"]:
        b, p = _run_writer(L.oracle_test_write_xor, int(case["prev_xor"], 16),
                           int(case["cur_xor"], 16))
        assert b == case["bytes"] and p == case["pos"], case


def test_write_annotation(vectors):
    L = oracle.lib()
    for case in vectors["write_annotation"]["cases"]:
        ann = np.asarray(case["annotation"], dtype=np.uint8)
        out = np.zeros(64, dtype=np.uint8)
        pos = ctypes.c_int32(0)
        n = L.oracle_test_write_annotation(
            oracle._pu8(ann) if len(ann) else None, len(ann),
            oracle._pu8(out), 64, ctypes.byref(pos))
        assert list(out[:n]) == case["bytes"] and pos.value == case["pos"], case


def test_write_timeunit(vectors):
    L = oracle.lib()
    for case in vectors["write_timeunit"]["cases"]:
        out = np.zeros(64, dtype=np.uint8)
        pos = ctypes.c_int32(0)
        ch = ctypes.c_int32(0)
        n = L.oracle_test_write_timeunit(case["unit"], oracle._pu8(out), 64,
                                         ctypes.byref(pos), ctypes.byref(ch))
        assert list(out[:n]) == case["bytes"], case
        assert pos.value == case["pos"] and bool(ch.value) == case["changed"], case


def test_read_next_timestamp(vectors):
    L = oracle.lib()
    for case in vectors["read_next_timestamp"]["cases"]:
        buf = np.asarray(case["bytes"], dtype=np.uint8)
        od = ctypes.c_int64(0)
        r = L.oracle_test_read_next_timestamp(oracle._pu8(buf), len(buf),
                                              case["unit"], case["prev_delta_ns"],
                                              ctypes.byref(od))
        assert r == 0 and od.value == case["expected_delta_ns"], case


def test_read_next_value(vectors):
    L = oracle.lib()
    for case in vectors["read_next_value"]["cases"]:
        buf = np.asarray(case["bytes"], dtype=np.uint8)
        ox = ctypes.c_uint64(0)
        ob = ctypes.c_uint64(0)
        r = L.oracle_test_read_next_value(oracle._pu8(buf), len(buf),
                                          int(case["prev_bits"], 16),
                                          int(case["prev_xor"], 16),
                                          ctypes.byref(ox), ctypes.byref(ob))
        assert r == 0, case
        assert ox.value == int(case["expected_xor"], 16), case
        assert ob.value == int(case["expected_bits"], 16), case


def _case_points(fs, case):
    base = fs["base_ns"]
    ts, vals, units, anns = [], [], [], []
    for ptd in case["points"]:
        dt = ptd.get("dt_s", 0) * 10**9 + ptd.get("dt_ns", 0)
        ts.append(base + dt)
        vals.append(ptd["value"])
        units.append(ptd["unit"])
        anns.append(bytes(ptd["annotation"]) if "annotation" in ptd else b"")
    return ts, vals, units, anns


def test_full_stream_encode(vectors):
    fs = vectors["full_streams"]
    for case in fs["cases"]:
        ts, vals, units, anns = _case_points(fs, case)
        enc = oracle.encode_series(ts, vals, units=units, annotations=anns,
                                   start_ns=fs["start_ns"], int_optimized=False)
        assert list(enc) == case["stream_bytes"], case["name"]
        if "raw_buffer" in case:
            raw, pos = oracle.encode_series(ts, vals, units=units, annotations=anns,
                                            start_ns=fs["start_ns"],
                                            int_optimized=False, raw=True)
            assert list(raw) == case["raw_buffer"] and pos == case["raw_pos"], case["name"]


def test_full_stream_decode(vectors):
    fs = vectors["full_streams"]
    for case in fs["cases"]:
        ts, vals, units, _ = _case_points(fs, case)
        dec = oracle.decode_series(bytes(np.asarray(case["stream_bytes"], dtype=np.uint8)),
                                   int_optimized=False, with_annotations=True)
        assert list(dec["ts"]) == ts, case["name"]
        assert list(dec["vals"]) == vals, case["name"]
        assert list(dec["units"]) == units, case["name"]
        if "decoded_annotations" in case:
            exp = [bytes(a) if a else None for a in case["decoded_annotations"]]
            assert dec["annotations"] == exp, case["name"]


def test_production_streams_decode(production):
    """All 10 embedded production streams decode end-to-end: 719-720 pts,
    strictly monotonic 10s-cadence millisecond-unit timestamps (the
    empirical compression range 0.41-2.97 B/pt pins the value grammar)."""
    for i, b64 in enumerate(production["samples"]):
        data = base64.b64decode(b64)
        dec = oracle.decode_series(data, int_optimized=True)
        n = len(dec["ts"])
        assert n in (719, 720), (i, n)
        assert np.all(np.diff(dec["ts"]) > 0), i
        assert set(dec["units"].tolist()) == {2}, i  # Millisecond
        bpp = len(data) / n
        assert 0.40 < bpp < 3.0, (i, bpp)


def test_production_streams_reencode_byte_exact(production):
    """decode -> encode round trip reproduces every production stream
    byte-for-byte (start time = the stream's own first 64 bits)."""
    for i, b64 in enumerate(production["samples"]):
        data = base64.b64decode(b64)
        start = struct.unpack(">q", data[:8])[0]
        dec = oracle.decode_series(data, int_optimized=True)
        enc = oracle.encode_series(dec["ts"], dec["vals"], units=dec["units"],
                                   start_ns=start, int_optimized=True)
        assert enc == data, f"stream {i} re-encode differs"


def test_regression_stream_decodes(production):
    """iterator_test.go:396-412: the decoding-regression stream must decode
    without error (it was written by an older encoder: decode-only)."""
    data = base64.b64decode(production["regression"])
    dec = oracle.decode_series(data, int_optimized=True)
    assert len(dec["ts"]) == 150
    assert np.all(np.diff(dec["ts"]) > 0)
