"""Commit log reader (m3_amd/csrc/commitlog.cpp) vs the oracle writer
restatement (oracle/commitlog_writer.py) — the commitlog half of §8f
row 1. Chunk container, record framing, fast-encoder msgpack formats,
series metadata registration, checksum failure modes."""
import struct
import zlib

import numpy as np
import pytest

from oracle import commitlog_writer as clw
from m3_amd import engine
from m3_amd.engine import CommitLog, M3GpuError

START = 1427162462 * 10**9
pytestmark = pytest.mark.skipif(not engine.engine_available(),
                                reason="libm3gpu.so not built")


def make_entries(rng, nseries=12, npts=40, with_annotations=False):
    """Interleaved per-series entries, like real ingest traffic."""
    entries = []
    per = {}
    for i in range(nseries):
        ts = START + np.cumsum(rng.integers(1, 60, npts)) * 10**9
        vals = np.round(rng.random(npts) * 1000, 3)
        per[i] = (ts, vals)
    for t in range(npts):
        for i in range(nseries):
            ant = b"ann" + bytes([t]) if (with_annotations and i == 0 and
                                          t % 7 == 0) else None
            tags = b"t=" + bytes([i]) if i % 2 == 0 else None
            entries.append((i, f"cl.series.{i:03d}".encode(), i % 4,
                            int(per[i][0][t]), float(per[i][1][t]), 4, ant,
                            tags))
    return entries, per


def test_commitlog_roundtrip(tmp_path):
    rng = np.random.default_rng(43)
    entries, per = make_entries(rng)
    path = tmp_path / "commitlog-0-0.db"
    clw.write_commitlog(str(path), entries, index=7)
    with CommitLog(path) as cl:
        assert cl.index == 7
        assert cl.num_entries == len(entries)
        assert cl.num_series == 12
        series = cl.series()
        for i, m in enumerate(series):  # first-seen order == index order
            assert m["unique_index"] == i
            assert m["id"] == f"cl.series.{i:03d}".encode()
            assert m["namespace"] == b"default"
            assert m["shard"] == i % 4
            assert m["tags"] == (b"t=" + bytes([i]) if i % 2 == 0 else b"")
            ts, vals = per[i]
            assert np.array_equal(m["ts"], ts)
            assert np.array_equal(m["vals"].view(np.uint64),
                                  vals.view(np.uint64))
            assert np.all(m["units"] == 4)


def test_commitlog_many_small_chunks(tmp_path):
    """Tiny flush size forces many chunks; records still parse as one
    continuous stream (chunk_reader.go Read crossing chunks)."""
    rng = np.random.default_rng(47)
    entries, per = make_entries(rng, nseries=5, npts=30)
    path = tmp_path / "commitlog-0-1.db"
    clw.write_commitlog(str(path), entries, flush_size=96)
    raw = open(path, "rb").read()
    # count chunks
    nchunks, pos = 0, 0
    while pos < len(raw):
        size = struct.unpack_from("<I", raw, pos)[0]
        pos += 12 + size
        nchunks += 1
    assert nchunks > 20
    with CommitLog(path) as cl:
        assert cl.num_entries == len(entries)
        for i, m in enumerate(cl.series()):
            assert np.array_equal(m["ts"], per[i][0])


def test_commitlog_record_spanning_chunks(tmp_path):
    """A record split across two chunks must still decode (the reference
    reader supports it even though its writer avoids it)."""
    rec_info = clw.encode_log_info(0)
    md = clw.encode_log_metadata(b"sp", b"ns", 1, None)
    rec_entry = clw.encode_log_entry(9, 0, md, START, 5.5, 4, None)
    framed = clw.uvarint(len(rec_info)) + rec_info + \
        clw.uvarint(len(rec_entry)) + rec_entry
    cut = len(framed) - 7  # split inside the last record
    out = bytearray()
    for payload in (framed[:cut], framed[cut:]):
        size = struct.pack("<I", len(payload))
        out += size + struct.pack("<I", zlib.adler32(size)) + \
            struct.pack("<I", zlib.adler32(payload)) + payload
    path = tmp_path / "commitlog-0-2.db"
    path.write_bytes(bytes(out))
    with CommitLog(path) as cl:
        assert cl.num_entries == 1
        m = cl.series()[0]
        assert m["id"] == b"sp" and m["ts"][0] == START and m["vals"][0] == 5.5


def test_commitlog_checksum_failures(tmp_path):
    rng = np.random.default_rng(53)
    entries, _ = make_entries(rng, nseries=3, npts=10)
    path = tmp_path / "commitlog-0-3.db"
    clw.write_commitlog(str(path), entries)
    raw = bytearray(open(path, "rb").read())
    # corrupt payload byte -> data checksum
    bad = bytearray(raw)
    bad[20] ^= 0xFF
    p = tmp_path / "bad1.db"
    p.write_bytes(bytes(bad))
    with pytest.raises(M3GpuError, match="chunk_checksum"):
        CommitLog(p)
    # corrupt size field -> size checksum
    bad = bytearray(raw)
    bad[0] ^= 0x01
    p = tmp_path / "bad2.db"
    p.write_bytes(bytes(bad))
    with pytest.raises(M3GpuError, match="chunk_checksum"):
        CommitLog(p)
    # truncate mid-chunk
    p = tmp_path / "bad3.db"
    p.write_bytes(bytes(raw[:len(raw) - 5]))
    with pytest.raises(M3GpuError, match="truncated"):
        CommitLog(p)


def test_commitlog_missing_metadata(tmp_path):
    """An entry for a never-registered series index errors like
    errCommitLogReaderMissingMetadata."""
    w = clw.ChunkedWriter()
    w.write_record(clw.encode_log_info(0))
    w.write_record(clw.encode_log_entry(3, 0, None, START, 1.0, 4, None))
    path = tmp_path / "commitlog-0-4.db"
    path.write_bytes(w.bytes())
    with pytest.raises(M3GpuError, match="missing_metadata"):
        CommitLog(path)


def test_commitlog_annotations(tmp_path):
    rng = np.random.default_rng(59)
    entries, per = make_entries(rng, nseries=3, npts=21, with_annotations=True)
    path = tmp_path / "commitlog-0-5.db"
    clw.write_commitlog(str(path), entries)
    with CommitLog(path) as cl:
        m = cl.series()[0]
        assert len(m["annotations"]) == 3  # t = 0, 7, 14
        for (pi, b), t in zip(m["annotations"], (0, 7, 14)):
            assert pi == t
            assert b == b"ann" + bytes([t])


def test_commitlog_empty(tmp_path):
    w = clw.ChunkedWriter()
    w.write_record(clw.encode_log_info(4))
    path = tmp_path / "commitlog-0-6.db"
    path.write_bytes(w.bytes())
    with CommitLog(path) as cl:
        assert cl.index == 4
        assert cl.num_entries == 0
        assert cl.num_series == 0


def test_commitlog_directory_merge(tmp_path):
    """Multi-file bootstrap: series identified by (namespace, id) across
    files (unique_index is per-file), files consumed in log-index order."""
    rng = np.random.default_rng(101)
    t0 = START
    # file index 1: series A(idx 0) + B(idx 1)
    e1 = [(0, b"series.A", 0, t0 + 1 * 10**9, 1.0, 4, None, None),
          (1, b"series.B", 0, t0 + 2 * 10**9, 2.0, 4, None, None),
          (0, b"series.A", 0, t0 + 3 * 10**9, 3.0, 4, None, None)]
    # file index 2: series A has a DIFFERENT unique index; C is new
    e2 = [(7, b"series.A", 0, t0 + 4 * 10**9, 4.0, 4, None, None),
          (9, b"series.C", 0, t0 + 5 * 10**9, 5.0, 4, None, None)]
    clw.write_commitlog(str(tmp_path / "commitlog-0-5.db"), e2, index=2)
    clw.write_commitlog(str(tmp_path / "commitlog-0-4.db"), e1, index=1)
    from m3_amd.engine import commitlog_read_dir
    meta = commitlog_read_dir(tmp_path)
    by_id = {m["id"]: m for m in meta}
    assert set(by_id) == {b"series.A", b"series.B", b"series.C"}
    # A merged across files in index order (1 then 2)
    assert by_id[b"series.A"]["ts"].tolist() == [t0 + 1 * 10**9,
                                                 t0 + 3 * 10**9,
                                                 t0 + 4 * 10**9]
    assert by_id[b"series.A"]["vals"].tolist() == [1.0, 3.0, 4.0]
    assert by_id[b"series.C"]["vals"].tolist() == [5.0]


def test_commitlog_empty_chunk_tolerated(tmp_path):
    """A zero-payload chunk (valid header, size 0) contributes nothing and
    parsing continues with the next chunk."""
    rec = clw.uvarint(len(clw.encode_log_info(5))) + clw.encode_log_info(5)
    size0 = struct.pack("<I", 0)
    empty_chunk = size0 + struct.pack("<I", zlib.adler32(size0)) + \
        struct.pack("<I", zlib.adler32(b""))
    size1 = struct.pack("<I", len(rec))
    chunk1 = size1 + struct.pack("<I", zlib.adler32(size1)) + \
        struct.pack("<I", zlib.adler32(rec)) + rec
    path = tmp_path / "commitlog-0-9.db"
    path.write_bytes(empty_chunk + chunk1 + empty_chunk)
    with CommitLog(path) as cl:
        assert cl.index == 5
        assert cl.num_entries == 0


def test_commitlog_maximal_varint_record_len(tmp_path):
    """A checksum-valid chunk whose record-length uvarint encodes ~2^64
    must fail with a clean truncation error, not wrap the bounds check
    and read past the payload (commitlog.cpp ClRd overflow fix)."""
    payload = clw.uvarint((1 << 64) - 1)  # 10-byte maximal uvarint
    size = struct.pack("<I", len(payload))
    chunk = size + struct.pack("<I", zlib.adler32(size)) + \
        struct.pack("<I", zlib.adler32(payload)) + payload
    path = tmp_path / "commitlog-0-0.db"
    path.write_bytes(chunk)
    with pytest.raises(M3GpuError, match="truncated"):
        CommitLog(path)
    # a uvarint whose 10th byte exceeds 1 overflows uint64 entirely
    payload = b"\xff" * 9 + b"\x02"
    size = struct.pack("<I", len(payload))
    chunk = size + struct.pack("<I", zlib.adler32(size)) + \
        struct.pack("<I", zlib.adler32(payload)) + payload
    path.write_bytes(chunk)
    with pytest.raises(M3GpuError, match="truncated"):
        CommitLog(path)
