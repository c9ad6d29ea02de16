"""Fileset volume writer — TEST INFRASTRUCTURE ONLY.

Restates the write side of the reference's dbnode fileset persistence
(src/dbnode/persist/fs/write.go:251-655, msgpack/encoder.go:139-432,
msgpack/schema.go:39-110, digest/{buffer.go,writer.go}, fs.go:27-51,
files.go:1729-1745) to generate volumes for testing the product reader
(m3_amd/csrc/fileset.cpp). Like everything under oracle/, this module may
only be imported by tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg — never by the product path.

Parity pinning note: the reference writer is pure Go
and no Go toolchain exists in this environment, so these bytes cannot be
pinned against reference-generated files. What IS pinned:
  - the data blocks (the golden-pinned M3TSZ codec),
  - every digest (hash/adler32 == zlib.adler32),
  - the msgpack layer, restated from the reference's own byte-level decoder
    (msgpack/decoder_fast.go:176-410) and the vmihailenco/msgpack v2.8.3
    encoding rules (go.mod:90/208): positive ints use the compact unsigned
    families (fixint/uint8/16/32/64), negatives the signed families, bytes
    use bin8/16/32, nil bytes encode as nil (0xc0), array lengths <=15 as
    fixarray;
  - structure/order, from write.go and encoder.go cited above.
The bloom filter CONTENTS are a restatement of the standard partitioned
bloom construction (github.com/m3db/bloom/v4 v4.0.0-20200901140942 is a
go.mod dependency whose source is not vendored here); ingestion never
reads the bitset (read.go only digest-validates it), so only its m/k info
fields and digest matter for reader parity.
"""
import os
import struct
import zlib
from math import ceil, log

# ------------------------------ msgpack ---------------------------------


def mp_uint(v):
    """vmihailenco v2 EncodeUint64: most compact unsigned family."""
    if v < 0:
        raise ValueError("mp_uint negative")
    if v <= 0x7F:
        return bytes([v])
    if v <= 0xFF:
        return bytes([0xCC, v])
    if v <= 0xFFFF:
        return b"\xcd" + struct.pack(">H", v)
    if v <= 0xFFFFFFFF:
        return b"\xce" + struct.pack(">I", v)
    return b"\xcf" + struct.pack(">Q", v)


def mp_int(v):
    """vmihailenco v2 EncodeInt64: non-negative -> unsigned families,
    negative -> neg fixint / int8 / int16 / int32 / int64."""
    if v >= 0:
        return mp_uint(v)
    if v >= -32:
        return struct.pack("b", v)
    if v >= -(1 << 7):
        return b"\xd0" + struct.pack(">b", v)
    if v >= -(1 << 15):
        return b"\xd1" + struct.pack(">h", v)
    if v >= -(1 << 31):
        return b"\xd2" + struct.pack(">i", v)
    return b"\xd3" + struct.pack(">q", v)


def mp_bytes(v):
    """EncodeBytes: nil -> 0xc0; else bin8/16/32."""
    if v is None:
        return b"\xc0"
    n = len(v)
    if n <= 0xFF:
        return bytes([0xC4, n]) + bytes(v)
    if n <= 0xFFFF:
        return b"\xc5" + struct.pack(">H", n) + bytes(v)
    return b"\xc6" + struct.pack(">I", n) + bytes(v)


def mp_array_len(n):
    if n <= 15:
        return bytes([0x90 | n])
    if n <= 0xFFFF:
        return b"\xdc" + struct.pack(">H", n)
    return b"\xdd" + struct.pack(">I", n)


# object types / schema constants (msgpack/schema.go:39-110)
ROOT_OBJECT, INDEX_INFO, INDEX_SUMMARIES_INFO, INDEX_BLOOM_FILTER_INFO, \
    INDEX_ENTRY, INDEX_SUMMARY = 1, 2, 3, 4, 5, 6
CURR_FIELDS = {ROOT_OBJECT: 2, INDEX_INFO: 11, INDEX_SUMMARIES_INFO: 1,
               INDEX_BLOOM_FILTER_INFO: 2, INDEX_ENTRY: 7, INDEX_SUMMARY: 3}
INDEX_INFO_VERSION = 1
INDEX_ENTRY_VERSION = 1
INDEX_SUMMARY_VERSION = 1
MAJOR_VERSION = 1  # persist/schema/types.go:32
MINOR_VERSION = 1  # persist/schema/types.go:37


def root_object(version, objtype):
    """encodeRootObject (encoder.go:380-384)."""
    return mp_int(version) + mp_array_len(CURR_FIELDS[ROOT_OBJECT]) + mp_int(objtype)


def encode_index_info(block_start, block_size, entries, summaries,
                      bloom_m, bloom_k, snapshot_time=0, file_type=1,
                      snapshot_id=None, volume_index=0):
    """EncodeIndexInfo V5 (encoder.go:139-158, 280-304). file_type 1 =
    FileSetFlushType (persist types)."""
    out = root_object(INDEX_INFO_VERSION, INDEX_INFO)
    out += mp_array_len(CURR_FIELDS[INDEX_INFO])
    out += mp_int(block_start)
    out += mp_int(block_size)
    out += mp_int(entries)
    out += mp_int(MAJOR_VERSION)
    out += mp_array_len(CURR_FIELDS[INDEX_SUMMARIES_INFO]) + mp_int(summaries)
    out += (mp_array_len(CURR_FIELDS[INDEX_BLOOM_FILTER_INFO]) +
            mp_int(bloom_m) + mp_int(bloom_k))
    out += mp_int(snapshot_time)
    out += mp_int(file_type)
    out += mp_bytes(snapshot_id)  # nil for flush filesets (write.go:606-609)
    out += mp_int(volume_index)
    out += mp_int(MINOR_VERSION)
    return out


def encode_index_entry(index, series_id, size, offset, data_checksum,
                       encoded_tags=None):
    """EncodeIndexEntry V3 (encoder.go:160-180, 328-339): trailing adler32
    of the entry's own preceding bytes."""
    out = root_object(INDEX_ENTRY_VERSION, INDEX_ENTRY)
    out += mp_array_len(CURR_FIELDS[INDEX_ENTRY])
    out += mp_int(index)
    out += mp_bytes(series_id)
    out += mp_int(size)
    out += mp_int(offset)
    out += mp_int(data_checksum)
    out += mp_bytes(encoded_tags)
    out += mp_int(zlib.adler32(out) & 0xFFFFFFFF)
    return out


def encode_index_summary(index, series_id, index_entry_offset):
    """EncodeIndexSummary (encoder.go:182-190, 341-346)."""
    out = root_object(INDEX_SUMMARY_VERSION, INDEX_SUMMARY)
    out += mp_array_len(CURR_FIELDS[INDEX_SUMMARY])
    out += mp_int(index)
    out += mp_bytes(series_id)
    out += mp_int(index_entry_offset)
    return out


# ------------------------------ bloom -----------------------------------


def bloom_estimate(n, p):
    """Standard bloom sizing (m3db/bloom v4 EstimateFalsePositiveRate
    restatement — write.go:414-419 passes (numSeries, fpRate))."""
    if n < 1:
        n = 1
    m = max(1, int(ceil(-(n * log(p)) / (log(2) ** 2))))
    k = max(1, int(ceil(log(2) * m / n)))
    return m, k


def bloom_bitset_bytes(m):
    """BitSet().Write serializes ceil(m/64) uint64 words (LE)."""
    words = (m + 63) // 64
    return bytearray(words * 8)


def bloom_add(bitset, m, k, series_id):
    """Best-effort bitset population (double hashing). The reference's
    exact hash family lives in the un-vendored m3db/bloom dependency;
    ingestion never reads these bits (read.go digest-validates only)."""
    import hashlib
    d = hashlib.sha256(bytes(series_id)).digest()
    h1 = int.from_bytes(d[:8], "little")
    h2 = int.from_bytes(d[8:16], "little") | 1
    for i in range(k):
        bit = (h1 + i * h2) % m
        bitset[(bit >> 3)] |= 1 << (bit & 7)


# ------------------------------ writer ----------------------------------


def digest32(b):
    return zlib.adler32(bytes(b)) & 0xFFFFFFFF


def write_volume(shard_dir, block_start_ns, series, block_size_ns=2 * 3600 * 10**9,
                 volume_index=0, summaries_percent=0.03,
                 bloom_fp_percent=0.02, legacy_names=False):
    """Write a complete flush-type fileset volume.

    series: list of (id: bytes, data: bytes, encoded_tags: bytes|None).
    Mirrors writer.close() ordering (write.go:373-443): data written in
    Write() call order, index sorted by ID (write.go:457), summaries every
    summaryEvery-th entry, bloom over all IDs, info last; digests file then
    checkpoint. Returns the paths written.
    """
    os.makedirs(shard_dir, exist_ok=True)

    def path(suffix):
        if legacy_names:
            name = f"fileset-{block_start_ns}-{suffix}.db"
        else:
            name = f"fileset-{block_start_ns}-{volume_index}-{suffix}.db"
        return os.path.join(shard_dir, name)

    # data file: concatenation in Write() order (write.go:251-326)
    data = bytearray()
    entries = []  # (id, tags, index, offset, size, checksum)
    for idx, (sid, blob, tags) in enumerate(series):
        if len(blob) == 0:
            continue  # writeAll skips empty (write.go:300-302)
        checksum = digest32(blob)
        entries.append([bytes(sid), tags, idx, len(data), len(blob), checksum])
        data += blob

    # index sorted by ID (write.go:446-501)
    entries_by_id = sorted(entries, key=lambda e: e[0])
    n = max(1, len(entries))
    summaries_approx = len(entries) * summaries_percent
    summary_every = int(len(entries) // summaries_approx) if summaries_approx > 0 else 0

    bloom_m, bloom_k = bloom_estimate(n, bloom_fp_percent)
    bitset = bloom_bitset_bytes(bloom_m)

    index = bytearray()
    summaries = bytearray()
    nsummaries = 0
    offset = 0
    for i, (sid, tags, idx, doff, size, checksum) in enumerate(entries_by_id):
        bloom_add(bitset, bloom_m, bloom_k, sid)
        rec = encode_index_entry(idx, sid, size, doff, checksum, tags)
        if summary_every == 0 or i % summary_every == 0:
            summaries += encode_index_summary(idx, sid, offset)
            nsummaries += 1
        index += rec
        offset += len(rec)

    info = encode_index_info(block_start_ns, block_size_ns, len(entries),
                             nsummaries, bloom_m, bloom_k,
                             volume_index=volume_index)

    files = {
        "info": bytes(info),
        "index": bytes(index),
        "summaries": bytes(summaries),
        "bloomfilter": bytes(bitset),
        "data": bytes(data),
    }
    for suffix, content in files.items():
        with open(path(suffix), "wb") as f:
            f.write(content)

    # digests file: info, index, summaries, bloom, data (write.go:381-390)
    dig = b"".join(struct.pack("<I", digest32(files[s]))
                   for s in ("info", "index", "summaries", "bloomfilter", "data"))
    with open(path("digest"), "wb") as f:
        f.write(dig)
    # checkpoint: digest of the digest file (write.go:337-347, 638-655)
    with open(path("checkpoint"), "wb") as f:
        f.write(struct.pack("<I", digest32(dig)))
    return [path(s) for s in
            ("info", "index", "summaries", "bloomfilter", "data", "digest",
             "checkpoint")]
