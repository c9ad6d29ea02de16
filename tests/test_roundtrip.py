"""Encode->decode round trips over the reference's test distributions.

Mirrors src/dbnode/encoding/m3tsz/roundtrip_test.go:38-285 with the
testgen/generate.go:30-44 value distributions (seeded here for determinism):
counter 12-digit ints, timer 7.6, small gauges 0.1, precise 2.16, negative
5.3, mixed-sign 3-digit ints, mixed, int-overflow at +-2^63, and the
precision fixture 187.80131100000006. All run for intOptimized true AND false.
"""
import numpy as np
import pytest

import oracle

START = 1427162462 * 10**9
ENC_START = 1427162400 * 10**9


def gen_float(rng, num_dig, num_dec, n):
    """testgen.GenerateFloatVal: numDig digits . numDec decimals (as parsed
    floats). Behavior-alike generator (values, not the Go RNG stream)."""
    dig = rng.integers(0, 10**num_dig if num_dig else 1, n)
    if num_dec == 0:
        return dig.astype(np.float64)
    dec = rng.integers(0, 10**num_dec, n)
    return np.array([float(f"{d}.{x}") for d, x in zip(dig, dec)])


def _roundtrip(ts, vals, units=None, anns=None):
    for intopt in (True, False):
        enc = oracle.encode_series(ts, vals, units=units, annotations=anns,
                                   start_ns=ENC_START, int_optimized=intopt)
        dec = oracle.decode_series(enc, int_optimized=intopt)
        assert np.array_equal(dec["ts"], ts), f"ts mismatch intopt={intopt}"
        assert np.array_equal(dec["vals"], vals), (
            f"val mismatch intopt={intopt} at "
            f"{np.argwhere(np.asarray(dec['vals']) != np.asarray(vals))[:5]}")
        if units is not None:
            assert np.array_equal(dec["units"], units)


def _ts(rng, n, max_step_s=1200):
    steps = rng.integers(1, max_step_s, n)
    return START + np.cumsum(steps) * 10**9


DISTS = {
    "counter": (12, 0),
    "timer": (7, 6),
    "small_gauge": (0, 1),
    "precise_gauge": (2, 16),
}


@pytest.mark.parametrize("dist", list(DISTS))
def test_distribution_roundtrip(dist):
    num_dig, num_dec = DISTS[dist]
    rng = np.random.default_rng(hash(dist) % 2**32)
    for it in range(8):
        n = 500
        vals = gen_float(rng, num_dig, num_dec, n)
        _roundtrip(_ts(rng, n), vals)


def test_negative_gauge_roundtrip():
    rng = np.random.default_rng(5)
    vals = -gen_float(rng, 5, 3, 500)
    _roundtrip(_ts(rng, 500), vals)


def test_mixed_sign_int_roundtrip():
    rng = np.random.default_rng(6)
    vals = gen_float(rng, 3, 0, 500)
    vals[rng.random(500) < 0.5] *= -1
    _roundtrip(_ts(rng, 500), vals)


def test_mixed_roundtrip():
    rng = np.random.default_rng(7)
    n = 500
    vals = np.empty(n)
    cur = float(gen_float(rng, 3, 16, 1)[0])
    for i in range(n):
        r = rng.random()
        if r < 0.1:
            cur = float(gen_float(rng, 5, 0, 1)[0])
        elif r < 0.2:
            cur = float(gen_float(rng, 3, 16, 1)[0])
        vals[i] = cur
    _roundtrip(_ts(rng, n, 7200), vals)


def test_int_overflow_roundtrip():
    """roundtrip_test.go:270-285 generateOverflowDatapoints."""
    li = float(2**63 - 1 - 1)   # float64(MaxInt64-1) == 2^63
    ln = float(-(2**63) + 1)    # float64(MinInt64+1) == -2^63
    vals = [li, 10, ln, 10, ln, li, -12, li, 14.5, li, ln, 12.34858499392, li]
    ts = [START + i * 10**9 for i in range(len(vals))]
    _roundtrip(ts, vals)


def test_precision_fixture():
    """roundtrip_test.go:95-108: 187.80131100000006 x100 at 1m cadence."""
    vals = [187.80131100000006] * 100
    ts = [START + i * 60 * 10**9 for i in range(100)]
    _roundtrip(ts, vals)


def test_unit_changes_and_annotations():
    """validateRoundTrip :119-158: ms unit at i=0, us at i=10; annotations
    foo (i<5), bar (i<7), 64-byte at i=10; repeats must be deduped."""
    rng = np.random.default_rng(42)
    n = 300
    ts = _ts(rng, n)
    vals = gen_float(rng, 7, 6, n)
    units = np.full(n, 1, np.uint8)
    units[0] = 2
    units[10] = 3
    long_ann = b"long annotation " * 4  # 64 bytes
    anns = [b"foo" if i < 5 else (b"bar" if i < 7 else (long_ann if i == 10 else b""))
            for i in range(n)]
    for intopt in (True, False):
        enc = oracle.encode_series(ts, vals, units=units, annotations=anns,
                                   start_ns=ENC_START, int_optimized=intopt)
        dec = oracle.decode_series(enc, int_optimized=intopt, with_annotations=True)
        assert np.array_equal(dec["ts"], ts)
        assert np.array_equal(dec["vals"], vals)
        assert np.array_equal(dec["units"], units)
        # annotation written only when the checksum changes
        exp = [None] * n
        exp[0] = b"foo"
        exp[5] = b"bar"
        exp[10] = long_ann
        assert dec["annotations"] == exp


def test_nan_inf_roundtrip():
    vals = [1.5, float("inf"), float("-inf"), 3.0, float("nan"), 3.0, 0.0, -0.0]
    ts = [START + i * 10**9 for i in range(len(vals))]
    for intopt in (True, False):
        enc = oracle.encode_series(ts, vals, start_ns=ENC_START, int_optimized=intopt)
        dec = oracle.decode_series(enc, int_optimized=intopt)
        got = np.asarray(dec["vals"])
        exp = np.asarray(vals)
        if intopt:
            # int-optimized mode folds -0.0 into the repeat opcode after 0.0
            # (encoder.go:200-206 valDiff==0): Go == equality, like the
            # reference's own roundtrip assertions.
            eq = (got == exp) | (np.isnan(got) & np.isnan(exp))
            assert eq.all(), (got, exp)
        else:
            assert np.array_equal(got.view(np.uint64), exp.view(np.uint64))


def test_empty_and_single():
    enc = oracle.encode_series([], [], start_ns=ENC_START)
    assert enc == b""
    enc = oracle.encode_series([START], [42.0], start_ns=ENC_START)
    dec = oracle.decode_series(enc)
    assert list(dec["ts"]) == [START] and list(dec["vals"]) == [42.0]


def test_batch_matches_series():
    """oracle_encode_batch/oracle_decode_batch agree with the per-series API."""
    rng = np.random.default_rng(11)
    nseries, npts = 64, 240
    base = START + np.arange(nseries, dtype=np.int64)[:, None] * 10**9 * 3600
    ts = base + np.arange(npts, dtype=np.int64)[None, :] * 10 * 10**9
    vals = np.round(rng.random((nseries, npts)) * 1e6, 3)
    counts = np.full(nseries, npts, np.uint32)
    rows, lens = oracle.encode_batch(ts, vals, counts)
    blob = np.concatenate([rows[i, :lens[i]] for i in range(nseries)])
    offs = np.zeros(nseries + 1, np.uint64)
    offs[1:] = np.cumsum(lens)
    dts, dvals, dcounts = oracle.decode_batch(blob, offs, stride=npts + 8)
    assert np.array_equal(dcounts, counts)
    assert np.array_equal(dts[:, :npts], ts)
    assert np.array_equal(dvals[:, :npts], vals)
    for i in range(0, nseries, 17):
        single = oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
        assert bytes(rows[i, :lens[i]]) == single


def test_pack_streams_64b_alignment():
    """pack_streams emits 64B-aligned offsets (one HBM line per stream
    chunk — m3gpu.h layout note); the ABI minimum is 8B."""
    from m3_amd.engine import pack_streams
    streams = [b"\x01" * n for n in (1, 63, 64, 65, 129, 0, 7)]
    blob, offsets, lens = pack_streams(streams)
    assert np.all(offsets % 64 == 0)
    assert int(offsets[-1]) == len(blob)
    for i, s in enumerate(streams):
        o = int(offsets[i])
        assert bytes(blob[o:o + len(s)]) == s
        # zero pad up to the next stream
        assert not blob[o + len(s):int(offsets[i + 1])].any()


def test_oracle_roundtrip_annotations_and_unit_changes():
    """Oracle self-check: annotations interleaved with time-unit changes
    round-trip exactly (annotation markers chain with timeunit markers —
    tryReadMarker loops, timestamp_iterator.go:174-235)."""
    rng = np.random.default_rng(31337)
    start = 1427162462 * 10**9
    n = 60
    ts = start + np.cumsum(rng.integers(1, 50, n)) * 10**9
    # switch units mid-stream: s -> ms -> s
    units = np.full(n, 1, np.uint8)
    units[20:40] = 2
    ts[20:40] = (ts[20:40] // 10**6) * 10**6
    vals = np.round(rng.random(n) * 100, 2)
    anns = [None] * n
    anns[0], anns[19], anns[20], anns[45] = b"a0", b"pre", b"at-switch", b"zz"
    blob = oracle.encode_series(ts, vals, units=units, annotations=anns,
                                start_ns=start)
    dec = oracle.decode_series(blob, with_annotations=True)
    assert np.array_equal(dec["ts"], ts)
    assert np.array_equal(dec["vals"], vals)
    assert np.array_equal(dec["units"], units)
    got = [(i, a) for i, a in enumerate(dec["annotations"]) if a is not None]
    assert got == [(0, b"a0"), (19, b"pre"), (20, b"at-switch"), (45, b"zz")]
