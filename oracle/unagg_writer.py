"""Unaggregated-metric wire writer — TEST INFRASTRUCTURE ONLY.

Restates the m3aggregator ingest encoding
(src/metrics/encoding/protobuf/unaggregated_encoder.go:195-236 framing +
gogo-generated metricpb marshaling, src/metrics/generated/proto/metricpb/
{metric.proto,composite.proto}) to generate wire buffers for the product
parser (m3_amd/csrc/unagg.cpp). oracle/-only import rules apply.

Framing: each message = Go binary.PutVarint (ZIGZAG varint) of the
protobuf size, then the MetricWithMetadatas bytes. Protobuf encoding is
standard proto3 wire format with fields in ascending order (gogo
marshalers emit ascending); repeated doubles are packed.
"""
import struct


def pv_uvarint(v):
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def pv_varint_field(field, v):
    """int64 field: two's complement varint (10 bytes when negative)."""
    return pv_uvarint(field << 3 | 0) + pv_uvarint(v & 0xFFFFFFFFFFFFFFFF)


def pv_bytes_field(field, b):
    if b is None or len(b) == 0:
        return b""  # proto3 omits empty
    return pv_uvarint(field << 3 | 2) + pv_uvarint(len(b)) + bytes(b)


def pv_double_field(field, v):
    return pv_uvarint(field << 3 | 1) + struct.pack("<d", v)


def pv_packed_doubles(field, vals):
    if not len(vals):
        return b""
    payload = b"".join(struct.pack("<d", v) for v in vals)
    return pv_uvarint(field << 3 | 2) + pv_uvarint(len(payload)) + payload


def counter(mid, value, annotation=None, client_time_nanos=0):
    out = pv_bytes_field(1, mid)
    if value:
        out += pv_varint_field(2, value)
    out += pv_bytes_field(3, annotation)
    if client_time_nanos:
        out += pv_varint_field(4, client_time_nanos)
    return out


def batch_timer(mid, values, annotation=None, client_time_nanos=0):
    out = pv_bytes_field(1, mid)
    out += pv_packed_doubles(2, values)
    out += pv_bytes_field(3, annotation)
    if client_time_nanos:
        out += pv_varint_field(4, client_time_nanos)
    return out


def gauge(mid, value, annotation=None, client_time_nanos=0):
    out = pv_bytes_field(1, mid)
    if value != 0.0:
        out += pv_double_field(2, value)
    out += pv_bytes_field(3, annotation)
    if client_time_nanos:
        out += pv_varint_field(4, client_time_nanos)
    return out


def timed_metric(metric_type, mid, time_nanos, value, annotation=None):
    out = b""
    if metric_type:
        out += pv_varint_field(1, metric_type)
    out += pv_bytes_field(2, mid)
    if time_nanos:
        out += pv_varint_field(3, time_nanos)
    out += pv_double_field(4, value)
    out += pv_bytes_field(5, annotation)
    return out


def with_metadatas(union_type, metric_bytes, metadatas=b"\x0a\x00"):
    """MetricWithMetadatas: 1=type, (type+1)=payload {1: metric,
    2: metadatas}. Default metadatas = a minimal non-empty StagedMetadatas
    blob (opaque passthrough)."""
    payload = (pv_uvarint(1 << 3 | 2) + pv_uvarint(len(metric_bytes)) +
               metric_bytes)
    if metadatas:
        payload += (pv_uvarint(2 << 3 | 2) + pv_uvarint(len(metadatas)) +
                    metadatas)
    return (pv_varint_field(1, union_type) +
            pv_uvarint((union_type + 1) << 3 | 2) +
            pv_uvarint(len(payload)) + payload)


def frame(msg):
    """unaggregated_encoder framing: zigzag varint size + message."""
    n = len(msg)
    zz = (n << 1) ^ (n >> 63)
    return pv_uvarint(zz) + msg


def encode_stream(messages):
    """messages: list of MetricWithMetadatas byte blobs -> wire buffer."""
    return b"".join(frame(m) for m in messages)
