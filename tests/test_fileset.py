"""Fileset volume reader (m3_amd/csrc/fileset.cpp, pure host code) vs the
oracle writer restatement (oracle/fileset_writer.py) — the §8f row 1
ingestion boundary. Parity pinning per DESIGN.md: codec blocks are
golden-pinned; the container format is restated from the reference's own
byte-level decoder (msgpack/decoder_fast.go) + digest scheme; no
reference-generated binary fixtures exist (pure-Go writer, no Go
toolchain), so these tests pin writer<->reader agreement plus every
validation path."""
import struct
import zlib

import numpy as np
import pytest

import oracle
from oracle import fileset_writer as fsw
from m3_amd import engine
from m3_amd.engine import FilesetVolume, M3GpuError

BLOCK_START = 1427162400 * 10**9
pytestmark = pytest.mark.skipif(not engine.engine_available(),
                                reason="libm3gpu.so not built")


def make_series(rng, n, npts_max=50):
    series = []
    raw = {}
    for i in range(n):
        npts = int(rng.integers(2, npts_max))
        ts = BLOCK_START + np.cumsum(rng.integers(1, 60, npts)) * 10**9
        vals = np.round(rng.random(npts) * 100, 2)
        blob = oracle.encode_series(ts, vals, start_ns=int(ts[0]))
        sid = f"series.{i:04d}".encode()
        tags = b"tagblob" + bytes([i % 251]) if i % 3 == 0 else None
        series.append((sid, blob, tags))
        raw[sid] = (ts, vals, blob)
    return series, raw


def test_fileset_roundtrip(tmp_path):
    rng = np.random.default_rng(11)
    series, raw = make_series(rng, 37)
    fsw.write_volume(str(tmp_path), BLOCK_START, series)
    with FilesetVolume(str(tmp_path), BLOCK_START) as v:
        assert v.block_start == BLOCK_START
        assert v.num_entries == 37
        assert v.major_version == 1 and v.minor_version == 1
        assert v.block_size == 2 * 3600 * 10**9
        ents = v.entries()
        # sorted by data offset asc == write order here
        offs = [e[2] for e in ents]
        assert offs == sorted(offs)
        for sid, size, off, ck, tags in ents:
            ts, vals, blob = raw[sid]
            assert size == len(blob)
            assert ck == zlib.adler32(blob)
        # tags round-trip
        by_id = {e[0]: e for e in ents}
        for sid, blob, tags in series:
            assert by_id[sid][4] == (tags or b"")
        blob_arr, offsets, lens = v.pack()
        assert np.all(offsets % 64 == 0)
        for i, (sid, size, off, ck, tags) in enumerate(ents):
            got = bytes(blob_arr[int(offsets[i]):int(offsets[i]) + int(lens[i])])
            assert got == raw[sid][2], sid
        # the packed blob decodes bit-exactly (oracle CPU decode; oracle
        # offsets carry an n+1 end sentinel)
        stride = 64
        o_off = np.concatenate([offsets, [np.uint64(len(blob_arr))]])
        o_ts, o_vals, o_counts = oracle.decode_batch(blob_arr, o_off,
                                                     stride=stride)
        for i, (sid, size, off, ck, tags) in enumerate(ents):
            ts, vals, _ = raw[sid]
            assert o_counts[i] == len(ts)
            assert np.array_equal(o_ts[i, :len(ts)], ts)
            assert np.array_equal(o_vals[i, :len(ts)], vals)


def test_fileset_corruption_detected(tmp_path):
    rng = np.random.default_rng(13)
    series, _ = make_series(rng, 8)
    paths = fsw.write_volume(str(tmp_path), BLOCK_START, series)
    by_suffix = {p.split("-")[-1].split(".")[0]: p for p in paths}

    def flip(path, pos=None):
        data = bytearray(open(path, "rb").read())
        pos = len(data) // 2 if pos is None else pos
        data[pos] ^= 0xFF
        open(path, "wb").write(bytes(data))
        return data

    for suffix, expect in [("info", "digest"), ("index", "digest"),
                           ("summaries", "digest"), ("bloomfilter", "digest"),
                           ("data", "digest"), ("digest", "digest"),
                           ("checkpoint", "digest")]:
        orig = open(by_suffix[suffix], "rb").read()
        flip(by_suffix[suffix])
        with pytest.raises(M3GpuError):
            FilesetVolume(str(tmp_path), BLOCK_START)
        open(by_suffix[suffix], "wb").write(orig)
    # sanity: intact volume still opens
    FilesetVolume(str(tmp_path), BLOCK_START).close()


def test_fileset_data_checksum_mismatch(tmp_path):
    """Corrupt a data block AND fix up the file-level digests: the
    per-entry checksum (read.go:393-397) must still catch it."""
    rng = np.random.default_rng(17)
    series, _ = make_series(rng, 4)
    paths = fsw.write_volume(str(tmp_path), BLOCK_START, series)
    by_suffix = {p.split("-")[-1].split(".")[0]: p for p in paths}
    data = bytearray(open(by_suffix["data"], "rb").read())
    data[3] ^= 0x01
    open(by_suffix["data"], "wb").write(bytes(data))
    # recompute digest + checkpoint files
    dig = bytearray(open(by_suffix["digest"], "rb").read())
    dig[16:20] = struct.pack("<I", zlib.adler32(bytes(data)))
    open(by_suffix["digest"], "wb").write(bytes(dig))
    open(by_suffix["checkpoint"], "wb").write(
        struct.pack("<I", zlib.adler32(bytes(dig))))
    with pytest.raises(M3GpuError, match="data_checksum"):
        FilesetVolume(str(tmp_path), BLOCK_START)


def test_fileset_legacy_names(tmp_path):
    rng = np.random.default_rng(19)
    series, raw = make_series(rng, 5)
    fsw.write_volume(str(tmp_path), BLOCK_START, series, legacy_names=True)
    with FilesetVolume(str(tmp_path), BLOCK_START, volume_index=0) as v:
        assert v.num_entries == 5
        assert {e[0] for e in v.entries()} == set(raw)


def test_fileset_empty_volume(tmp_path):
    fsw.write_volume(str(tmp_path), BLOCK_START, [])
    with FilesetVolume(str(tmp_path), BLOCK_START) as v:
        assert v.num_entries == 0
        blob, offsets, lens = v.pack()
        assert len(offsets) == 0


def test_fileset_missing_checkpoint(tmp_path):
    rng = np.random.default_rng(23)
    series, _ = make_series(rng, 3)
    paths = fsw.write_volume(str(tmp_path), BLOCK_START, series)
    import os
    os.remove([p for p in paths if "checkpoint" in p][0])
    with pytest.raises(M3GpuError, match="io"):
        FilesetVolume(str(tmp_path), BLOCK_START)


def test_fileset_v1_entries_accepted(tmp_path):
    """Older V1 index entries (5 fields, no tags/checksum) decode fine
    (decoder.go:405-409 'actual < 6')."""
    rng = np.random.default_rng(29)
    series, raw = make_series(rng, 6)
    paths = fsw.write_volume(str(tmp_path), BLOCK_START, series)
    by_suffix = {p.split("-")[-1].split(".")[0]: p for p in paths}
    # rebuild the index file with V1-style entries
    index = bytearray()
    data = open(by_suffix["data"], "rb").read()
    entries = []
    off = 0
    for idx, (sid, blob, tags) in enumerate(series):
        entries.append((sid, idx, off, len(blob), zlib.adler32(blob)))
        off += len(blob)
    for sid, idx, off_, size, ck in sorted(entries):
        rec = fsw.root_object(fsw.INDEX_ENTRY_VERSION, fsw.INDEX_ENTRY)
        rec += fsw.mp_array_len(5)
        rec += fsw.mp_int(idx) + fsw.mp_bytes(sid) + fsw.mp_int(size)
        rec += fsw.mp_int(off_) + fsw.mp_int(ck)
        index += rec
    open(by_suffix["index"], "wb").write(bytes(index))
    dig = bytearray(open(by_suffix["digest"], "rb").read())
    dig[4:8] = struct.pack("<I", zlib.adler32(bytes(index)))
    open(by_suffix["digest"], "wb").write(bytes(dig))
    open(by_suffix["checkpoint"], "wb").write(
        struct.pack("<I", zlib.adler32(bytes(dig))))
    with FilesetVolume(str(tmp_path), BLOCK_START) as v:
        ents = v.entries()
        assert len(ents) == 6
        for sid, size, off_, ck, tags in ents:
            assert tags == b""
            assert raw[sid][2] == data[off_:off_ + size]


def test_fileset_entry_checksum_mismatch(tmp_path):
    """A V3 entry whose trailing checksum disagrees is rejected even when
    the file-level digest is consistent (decoder.go:443-445)."""
    rng = np.random.default_rng(31)
    series, _ = make_series(rng, 2)
    paths = fsw.write_volume(str(tmp_path), BLOCK_START, series)
    by_suffix = {p.split("-")[-1].split(".")[0]: p for p in paths}
    sid, blob, _ = series[0]
    rec_good = fsw.encode_index_entry(0, sid, len(blob), 0,
                                      zlib.adler32(blob), None)
    rec_bad = bytearray(rec_good)
    rec_bad[-1] ^= 0x01  # perturb the trailing checksum varint
    index = bytes(rec_bad)
    open(by_suffix["index"], "wb").write(index)
    dig = bytearray(open(by_suffix["digest"], "rb").read())
    dig[4:8] = struct.pack("<I", zlib.adler32(index))
    open(by_suffix["digest"], "wb").write(bytes(dig))
    open(by_suffix["checkpoint"], "wb").write(
        struct.pack("<I", zlib.adler32(bytes(dig))))
    # info still says 2 entries; first entry fails its checksum
    with pytest.raises(M3GpuError, match="entry_checksum|msgpack"):
        FilesetVolume(str(tmp_path), BLOCK_START)


def test_fileset_huge_entries_count_rejected(tmp_path):
    """A digest-valid volume whose info file claims entries=2^62 must
    return a schema error instead of letting an unbounded reserve()
    throw past the C ABI (fileset.cpp fs_parse_entries bound)."""
    rng = np.random.default_rng(17)
    series, _ = make_series(rng, 3)
    fsw.write_volume(str(tmp_path), BLOCK_START, series)

    def p(suffix):
        return tmp_path / f"fileset-{BLOCK_START}-0-{suffix}.db"

    info = fsw.encode_index_info(BLOCK_START, 2 * 3600 * 10**9, 1 << 62,
                                 1, 64, 3)
    p("info").write_bytes(info)
    # recompute digests + checkpoint so only the count is wrong
    names = ("info", "index", "summaries", "bloomfilter", "data")
    dig = b"".join(struct.pack("<I", fsw.digest32(p(s).read_bytes()))
                   for s in names)
    p("digest").write_bytes(dig)
    p("checkpoint").write_bytes(struct.pack("<I", fsw.digest32(dig)))
    with pytest.raises(M3GpuError, match="schema"):
        FilesetVolume(str(tmp_path), BLOCK_START)
