"""Unaggregated metric wire parser (m3_amd/csrc/unagg.cpp) vs the oracle
writer restatement — the m3aggregator ingest boundary (§8f row 4 tail).
Framing (zigzag varint sizes), proto3 payloads, metadatas passthrough,
truncation/type failure modes."""
import numpy as np
import pytest

from oracle import unagg_writer as uw
from m3_amd import engine
from m3_amd.engine import parse_unaggregated, M3GpuError

pytestmark = pytest.mark.skipif(not engine.engine_available(),
                                reason="libm3gpu.so not built")


def test_unagg_roundtrip_mixed():
    rng = np.random.default_rng(73)
    msgs = []
    expect = []
    for i in range(200):
        kind = i % 3
        mid = f"svc.m{i:04d}".encode()
        md = b"\x0a" + bytes([i % 100 + 1]) + bytes(i % 100 + 1)
        if kind == 0:
            v = int(rng.integers(-10**12, 10**12))
            msgs.append(uw.with_metadatas(1, uw.counter(mid, v), md))
            expect.append(("counter", mid, v, None, md))
        elif kind == 1:
            vals = np.round(rng.random(int(rng.integers(1, 50))) * 1e3, 3)
            msgs.append(uw.with_metadatas(2, uw.batch_timer(mid, vals), md))
            expect.append(("batch_timer", mid, None, vals, md))
        else:
            v = float(rng.random() * 1e6)
            msgs.append(uw.with_metadatas(3, uw.gauge(mid, v), md))
            expect.append(("gauge", mid, None, np.array([v]), md))
    out = parse_unaggregated(uw.encode_stream(msgs))
    assert len(out) == 200
    for m, (t, mid, cv, vals, md) in zip(out, expect):
        assert m["type"] == t
        assert m["id"] == mid
        assert m["metadatas"] == md
        if cv is not None:
            assert m["counter_value"] == cv
            assert m["values"][0] == float(cv)
        else:
            assert np.array_equal(m["values"].view(np.uint64),
                                  vals.view(np.uint64))


def test_unagg_timed_metrics():
    mid = b"timed.metric"
    tm = uw.timed_metric(3, mid, 1427162462 * 10**9, 42.5, b"note")
    msgs = [uw.with_metadatas(6, tm),
            uw.with_metadatas(7, tm, metadatas=b"\x0a\x02hi")]
    out = parse_unaggregated(uw.encode_stream(msgs))
    for m in out:
        assert m["id"] == mid
        assert m["metric_type"] == 3
        assert m["time_nanos"] == 1427162462 * 10**9
        assert m["values"][0] == 42.5
        assert m["annotation"] == b"note"
    assert out[0]["type"] == "timed_with_metadatas"
    assert out[1]["type"] == "timed_with_storage_policy"


def test_unagg_negative_counter_and_zero_omission():
    # value 0 omitted by proto3 -> parses as 0; negative -> 10-byte varint
    msgs = [uw.with_metadatas(1, uw.counter(b"a", 0)),
            uw.with_metadatas(1, uw.counter(b"b", -7))]
    out = parse_unaggregated(uw.encode_stream(msgs))
    assert out[0]["counter_value"] == 0
    assert out[1]["counter_value"] == -7


def test_unagg_client_time_and_annotation():
    msgs = [uw.with_metadatas(3, uw.gauge(b"g", 1.5, b"ann", 12345))]
    m = parse_unaggregated(uw.encode_stream(msgs))[0]
    assert m["time_nanos"] == 12345
    assert m["annotation"] == b"ann"


def test_unagg_errors():
    good = uw.encode_stream([uw.with_metadatas(1, uw.counter(b"x", 5))])
    with pytest.raises(M3GpuError, match="truncated"):
        parse_unaggregated(good[:-2])
    # forwarded union type rejected
    fw = uw.with_metadatas(4, uw.timed_metric(1, b"f", 1, 2.0))
    with pytest.raises(M3GpuError, match="type"):
        parse_unaggregated(uw.encode_stream([fw]))
    # zero/negative size prefix
    with pytest.raises(M3GpuError, match="size"):
        parse_unaggregated(b"\x00" + good)


def test_unagg_unknown_fields_skipped():
    """Future fields (higher numbers, any wire type) skip cleanly."""
    base = uw.counter(b"c", 9)
    base += uw.pv_uvarint(9 << 3 | 0) + uw.pv_uvarint(777)   # varint
    base += uw.pv_uvarint(10 << 3 | 2) + uw.pv_uvarint(3) + b"xyz"
    base += uw.pv_uvarint(11 << 3 | 1) + b"\x00" * 8          # fixed64
    m = parse_unaggregated(uw.encode_stream([uw.with_metadatas(1, base)]))[0]
    assert m["counter_value"] == 9


def test_unagg_empty_stream():
    assert parse_unaggregated(b"") == []


def test_unagg_large_batch_timer():
    """Packed-doubles payload beyond one length byte (500 values = 4000 B)
    and a large id exercise multi-byte varint lengths."""
    vals = np.arange(500, dtype=np.float64) * 0.5
    mid = b"x" * 300  # bin16 id
    msg = uw.with_metadatas(2, uw.batch_timer(mid, vals))
    m = parse_unaggregated(uw.encode_stream([msg]))[0]
    assert m["id"] == mid
    assert np.array_equal(m["values"], vals)


def test_unagg_maximal_varint_length_rejected():
    """A bytes-field length encoded as a maximal 10-byte varint (~2^64)
    must fail with a clean truncation/proto error, not wrap the bounds
    check and abort the process (unagg.cpp UaRd::bytes overflow fix)."""
    # counter payload: field 1 (id, wt=2) with length 2^64-1
    max_len = uw.pv_uvarint((1 << 64) - 1)
    assert len(max_len) == 10 and max_len[-1] == 1
    bad_counter = uw.pv_uvarint(1 << 3 | 2) + max_len  # no body follows
    msg = uw.with_metadatas(1, bad_counter)
    with pytest.raises(M3GpuError, match="truncated|proto"):
        parse_unaggregated(uw.encode_stream([msg]))
    # 10th byte > 1 exceeds uint64 (Go binary.ReadUvarint overflow rule)
    over = b"\xff" * 9 + b"\x02"
    bad2 = uw.pv_uvarint(1 << 3 | 2) + over
    with pytest.raises(M3GpuError, match="truncated|proto"):
        parse_unaggregated(uw.encode_stream([uw.with_metadatas(1, bad2)]))
    # same class inside skip() of an unknown length-delimited field
    base = uw.counter(b"c", 7) + uw.pv_uvarint(9 << 3 | 2) + max_len
    with pytest.raises(M3GpuError, match="truncated|proto"):
        parse_unaggregated(uw.encode_stream([uw.with_metadatas(1, base)]))


def test_unagg_metadata_fields_separated():
    """StagedMetadatas (field 2) and StoragePolicy (field 3) of a
    TimedMetricWithStoragePolicy come back as separate tagged segments,
    not only as one unsplittable concatenation."""
    sp = b"\x08\x0a\x10\x3c"  # opaque fake StoragePolicy proto bytes
    md = b"\x0a\x05hello"
    payload = (uw.pv_uvarint(1 << 3 | 2) +
               uw.pv_uvarint(len(uw.timed_metric(2, b"tm", 5, 1.5))) +
               uw.timed_metric(2, b"tm", 5, 1.5) +
               uw.pv_uvarint(2 << 3 | 2) + uw.pv_uvarint(len(md)) + md +
               uw.pv_uvarint(3 << 3 | 2) + uw.pv_uvarint(len(sp)) + sp)
    msg = (uw.pv_varint_field(1, 7) + uw.pv_uvarint(8 << 3 | 2) +
           uw.pv_uvarint(len(payload)) + payload)
    m = parse_unaggregated(uw.encode_stream([msg]))[0]
    assert m["metadatas"] == md + sp  # legacy concat view unchanged
    assert m["metadata_fields"] == [(2, md), (3, sp)]
