"""ReaderIterator-shaped wrappers (m3_amd/iterators.py) over decoded
batches: Next/Current/Err semantics per dbnode/encoding/types.go:197-203."""
import numpy as np
import pytest

import oracle
from m3_amd.iterators import BatchIterators, SliceReaderIterator

START = 1427162462 * 10**9


def test_iterator_semantics():
    rng = np.random.default_rng(103)
    nseries, npts = 8, 25
    ts = START + np.cumsum(rng.integers(1, 60, (nseries, npts)), axis=1) * 10**9
    vals = np.round(rng.random((nseries, npts)) * 100, 2)
    counts = rng.integers(1, npts + 1, nseries)
    its = BatchIterators(ts, vals, counts)
    assert len(its) == nseries
    for i, it in enumerate(its):
        got = []
        while it.Next():
            t, v, u = it.Current()
            got.append((t, v))
            assert u == 1
        assert it.Err() is None
        assert got == [(int(ts[i, j]), float(vals[i, j]))
                       for j in range(counts[i])]
        assert not it.Next()  # exhausted stays exhausted
        it.Close()


def test_iterator_current_before_next():
    it = SliceReaderIterator(np.array([1]), np.array([2.0]), 1)
    with pytest.raises(RuntimeError):
        it.Current()
    assert it.Next()
    assert it.Current() == (1, 2.0, 1)


def test_iterator_sticky_error():
    """A per-series error makes the iterator yield nothing and report the
    mapped error string — like a sticky it.Err() on one reference
    iterator."""
    ts = np.zeros((1, 4), np.int64)
    vals = np.zeros((1, 4), np.float64)
    its = BatchIterators(ts, vals, np.array([4]), errs=np.array([1]))
    it = its.iterator(0)
    assert not it.Next()
    assert it.Err() == "eof"


def test_iterator_over_oracle_decode():
    """End-to-end CPU check: encode with the oracle, decode with the
    oracle, iterate with the reference-shaped wrapper."""
    rng = np.random.default_rng(107)
    npts = 40
    ts = START + np.cumsum(rng.integers(1, 30, npts)) * 10**9
    vals = np.round(rng.random(npts) * 10, 1)
    blob = oracle.encode_series(ts, vals, start_ns=int(ts[0]))
    from m3_amd.engine import pack_streams
    b, off, lens = pack_streams([blob])
    o_ts, o_vals, o_counts = oracle.decode_batch(b, off, stride=npts + 4)
    it = BatchIterators(o_ts, o_vals, o_counts).iterator(0)
    out = []
    while it.Next():
        t, v, _ = it.Current()
        out.append((t, v))
    assert out == list(zip(ts.tolist(), vals.tolist()))


def test_iterator_annotations_sticky():
    """CurrentAnnotation() mirrors the reference's sticky PrevAnt
    (iterator.go:226-231) from annotation-set events."""
    from m3_amd.iterators import SliceReaderIterator
    ts = np.arange(6, dtype=np.int64)
    vals = np.arange(6, dtype=np.float64)
    events = [(0, b"a"), (2, b"b"), (3, b"c")]
    it = SliceReaderIterator(ts, vals, 6, ann_events=events)
    seen = []
    while it.Next():
        seen.append(it.CurrentAnnotation())
    assert seen == [b"a", b"a", b"b", b"c", b"c", b"c"]
    # no events: always None
    it = SliceReaderIterator(ts, vals, 3)
    it.Next()
    assert it.CurrentAnnotation() is None


def test_ann_region_helpers_cpu():
    """parse_ann_region / decoded_ann_per_point are pure host logic: build
    a region by hand (layout: m3gpu.h) and check both views."""
    from m3_amd.engine import parse_ann_region, decoded_ann_per_point
    region = np.zeros(128, np.uint8)
    events = [(0, b"aa"), (3, b"bcd")]
    tail = 128
    evb = region[4:].view(np.uint32)
    for j, (pt, data) in enumerate(events):
        tail -= len(data)
        region[tail:tail + len(data)] = np.frombuffer(data, np.uint8)
        evb[j * 3:j * 3 + 3] = [pt, tail, len(data)]
    region[:4].view(np.uint32)[0] = len(events)
    assert parse_ann_region(region) == [(0, b"aa"), (3, b"bcd")]
    assert decoded_ann_per_point(region, 5) == [b"aa", b"aa", b"aa", b"bcd",
                                                b"bcd"]
    empty = np.zeros(16, np.uint8)
    assert parse_ann_region(empty) == []
    assert decoded_ann_per_point(empty, 3) == [None, None, None]
