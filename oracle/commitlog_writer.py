"""Commit log writer — TEST INFRASTRUCTURE ONLY.

Restates the reference's commit log write path
(src/dbnode/persist/fs/commitlog/writer.go:143-301 chunked container,
msgpack/encoder_fast.go:50-313 record encoding, msgpack/schema.go:170-183
fixed headers) to generate log files for the product reader
(m3_amd/csrc/commitlog.cpp). oracle/-only import rules apply.

Byte fidelity notes:
  - chunk header = size u32 LE + adler32(size bytes) + adler32(payload)
    (writer.go:43-58, 343-392);
  - records are uvarint-length-prefixed (writer.go:282-301) and a record is
    flushed into a new chunk when it would not fit the current buffer
    (mirroring bufio semantics: records never span chunks when written by
    the reference writer, though readers must support spanning);
  - LogEntry/LogMetadata bytes follow the FAST encoders exactly
    (encoder_fast.go: compact unsigned families for non-negative ints,
    bin8/16/32 bytes, nil for absent bytes, float64 as 0xcb BE), including
    the reference's LogMetadata header declaring 3 fields while writing 4
    (schema.go:109 vs encoder.go:372-378);
  - per-series metadata is attached to the first entry of each unique
    index (writer.go:205-221).
"""
import struct
import zlib

from .fileset_writer import mp_int, mp_uint, mp_bytes, mp_array_len

LOG_INFO, LOG_ENTRY, LOG_METADATA = 7, 8, 9


def _root(objtype, version=1):
    return mp_int(version) + mp_array_len(2) + mp_int(objtype)


# fixed headers precomputed at init in the reference (schema.go:170-183)
LOG_ENTRY_HEADER = _root(LOG_ENTRY) + mp_array_len(7)
LOG_METADATA_HEADER = _root(LOG_METADATA) + mp_array_len(3)


def mp_float64(v):
    return b"\xcb" + struct.pack(">d", v)


def uvarint(v):
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def encode_log_info(index):
    """encodeLogInfo (encoder.go:348-357): two deprecated varints + index."""
    return _root(LOG_INFO) + mp_array_len(3) + mp_int(0) + mp_int(0) + mp_int(index)


def encode_log_metadata(series_id, namespace, shard, encoded_tags=None):
    """EncodeLogMetadataFast (encoder_fast.go:72-87)."""
    return (LOG_METADATA_HEADER + mp_bytes(series_id) + mp_bytes(namespace) +
            mp_uint(shard) + mp_bytes(encoded_tags))


def encode_log_entry(index, create_ns, metadata, timestamp_ns, value, unit,
                     annotation=None):
    """EncodeLogEntryFast (encoder_fast.go:50-68)."""
    return (LOG_ENTRY_HEADER + mp_uint(index) + mp_int(create_ns) +
            mp_bytes(metadata) + mp_int(timestamp_ns) + mp_float64(value) +
            mp_uint(unit) + mp_bytes(annotation))


class ChunkedWriter:
    """fsChunkWriter + bufio flush-on-boundary (writer.go:282-301,343-392)."""

    def __init__(self, flush_size=65536):
        self.flush_size = flush_size
        self.buf = bytearray()
        self.out = bytearray()

    def write_record(self, record):
        framed = uvarint(len(record)) + record
        if self.buf and len(self.buf) + len(framed) > self.flush_size:
            self.flush()
        self.buf += framed

    def flush(self):
        if not self.buf:
            return
        payload = bytes(self.buf)
        size = struct.pack("<I", len(payload))
        self.out += size
        self.out += struct.pack("<I", zlib.adler32(size))
        self.out += struct.pack("<I", zlib.adler32(payload))
        self.out += payload
        self.buf.clear()

    def bytes(self):
        self.flush()
        return bytes(self.out)


def write_commitlog(path, entries, index=0, flush_size=65536,
                    namespace=b"default", create_ns=0):
    """Write a commit log file.

    entries: iterable of (unique_index: int, series_id: bytes,
    shard: int, timestamp_ns: int, value: float, unit: int,
    annotation: bytes|None, encoded_tags: bytes|None). Metadata is emitted
    with the first entry of each unique index, like writer.go:195-243.
    """
    w = ChunkedWriter(flush_size)
    w.write_record(encode_log_info(index))
    seen = set()
    for (uidx, sid, shard, ts, val, unit, annotation, tags) in entries:
        metadata = None
        if uidx not in seen:
            metadata = encode_log_metadata(sid, namespace, shard, tags)
            seen.add(uidx)
        w.write_record(encode_log_entry(uidx, create_ns, metadata, ts, val,
                                        unit, annotation))
    with open(path, "wb") as f:
        f.write(w.bytes())
    return path
