"""Oracle bindings — TEST INFRASTRUCTURE ONLY.

CPU restatement of the reference M3TSZ codec + m3aggregator rollup, used as
the parity checker and the `cpu_baseline` leg of bench.py. Only tests/,
__graft_entry__.smoke() and bench.py may import this package; the product
path (m3_amd) must never route through it.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "liboracle.so")


def build():
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            build()
        _lib = ctypes.CDLL(_LIB_PATH)
        _setup(_lib)
    return _lib


c_i64 = ctypes.c_int64
c_u64 = ctypes.c_uint64
c_i32 = ctypes.c_int32
c_u32 = ctypes.c_uint32
c_u8 = ctypes.c_uint8
c_f64 = ctypes.c_double
c_int = ctypes.c_int
P = ctypes.POINTER


def _setup(L):
    L.oracle_encode_series.restype = c_i64
    L.oracle_encode_series.argtypes = [P(c_i64), P(c_f64), P(c_u8), P(c_i32), P(c_u8),
                                       c_i32, c_i64, c_int, P(c_u8), c_i64]
    L.oracle_encode_series_raw.restype = c_i64
    L.oracle_encode_series_raw.argtypes = [P(c_i64), P(c_f64), P(c_u8), P(c_i32), P(c_u8),
                                           c_i32, c_i64, c_int, P(c_u8), c_i64, P(c_i32)]
    L.oracle_decode_series.restype = c_i64
    L.oracle_decode_series.argtypes = [P(c_u8), c_i64, c_int, c_u8,
                                       P(c_i64), P(c_f64), P(c_u8),
                                       P(c_i32), P(c_u8), c_i64, c_i64]
    L.oracle_encode_batch.restype = c_int
    L.oracle_encode_batch.argtypes = [P(c_i64), P(c_f64), P(c_u32), c_i64, c_i64,
                                      c_int, c_u8, P(c_u8), c_i64, P(c_u32), c_int]
    L.oracle_decode_batch.restype = c_int
    L.oracle_decode_batch.argtypes = [P(c_u8), P(c_u64), c_i64, c_int, c_u8,
                                      P(c_i64), P(c_f64), P(c_u32), c_i64, c_int]
    L.oracle_xxhash64.restype = c_u64
    L.oracle_xxhash64.argtypes = [P(c_u8), ctypes.c_size_t]
    L.oracle_ckms_list_len.restype = c_i64
    L.oracle_ckms_list_len.argtypes = [P(c_f64), c_i64, P(c_f64), c_int, c_f64, c_int]
    L.oracle_ckms_quantiles.restype = c_int
    L.oracle_ckms_quantiles.argtypes = [P(c_f64), c_i64, P(c_f64), c_int, c_f64, c_int,
                                        P(c_f64), P(c_f64), P(c_f64)]
    L.oracle_rollup_series.restype = c_int
    L.oracle_rollup_series.argtypes = [P(c_i64), P(c_f64), c_i64, c_int, c_i64, c_i64,
                                       c_i64, P(c_i32), c_int, P(c_f64), P(c_i64)]
    L.oracle_rollup_batch.restype = c_int
    L.oracle_rollup_batch.argtypes = [P(c_i64), P(c_f64), P(c_u32), c_i64, c_i64,
                                      c_int, c_i64, c_i64, P(c_i32), c_int,
                                      P(c_f64), P(c_i64), c_int]
    L.oracle_rollup_batch_opts.restype = c_int
    L.oracle_rollup_batch_opts.argtypes = L.oracle_rollup_batch.argtypes + [c_f64, c_int]
    # test wrappers
    for name in ("oracle_test_write_dod_unchanged", "oracle_test_write_dod_changed",
                 "oracle_test_write_xor", "oracle_test_write_annotation",
                 "oracle_test_write_timeunit"):
        getattr(L, name).restype = c_i64
    L.oracle_test_write_dod_unchanged.argtypes = [c_i64, c_i64, c_u8, P(c_u8), c_i64, P(c_i32)]
    L.oracle_test_write_dod_changed.argtypes = [c_i64, c_i64, P(c_u8), c_i64, P(c_i32)]
    L.oracle_test_write_xor.argtypes = [c_u64, c_u64, P(c_u8), c_i64, P(c_i32)]
    L.oracle_test_write_annotation.argtypes = [P(c_u8), c_i64, P(c_u8), c_i64, P(c_i32)]
    L.oracle_test_write_timeunit.argtypes = [c_u8, P(c_u8), c_i64, P(c_i32), P(c_i32)]
    L.oracle_test_read_next_timestamp.restype = c_int
    L.oracle_test_read_next_timestamp.argtypes = [P(c_u8), c_i64, c_u8, c_i64, P(c_i64)]
    L.oracle_test_read_next_value.restype = c_int
    L.oracle_test_read_next_value.argtypes = [P(c_u8), c_i64, c_u64, c_u64, P(c_u64), P(c_u64)]
    L.oracle_test_convert_to_int_float.restype = c_int
    L.oracle_test_convert_to_int_float.argtypes = [c_f64, c_u8, P(c_f64), P(c_u8), P(c_i32)]


def _pu8(arr):
    return arr.ctypes.data_as(P(c_u8))


def encode_series(ts_ns, vals, units=None, annotations=None, start_ns=None,
                  int_optimized=True, raw=False):
    """Encode one series; returns bytes (finalized stream) or (bytes, pos) if raw."""
    ts = np.ascontiguousarray(ts_ns, dtype=np.int64)
    v = np.ascontiguousarray(vals, dtype=np.float64)
    n = len(ts)
    if start_ns is None:
        start_ns = int(ts[0]) if n else 0
    u = None
    if units is not None:
        u = np.ascontiguousarray(units, dtype=np.uint8)
    ann_off = ann_bytes = None
    if annotations is not None:
        offs = [0]
        blob = bytearray()
        for a in annotations:
            if a:
                blob.extend(a)
            offs.append(len(blob))
        ann_off = np.asarray(offs, dtype=np.int32)
        ann_bytes = np.frombuffer(bytes(blob), dtype=np.uint8) if blob else np.zeros(1, np.uint8)
    cap = 32 + n * 24
    if annotations is not None:
        # annotation markers: 11-bit marker + varint len + bytes each
        cap += sum(16 + len(a) for a in annotations if a)
    out = np.zeros(cap, dtype=np.uint8)
    L = lib()
    args = [ts.ctypes.data_as(P(c_i64)), v.ctypes.data_as(P(c_f64)),
            u.ctypes.data_as(P(c_u8)) if u is not None else None,
            ann_off.ctypes.data_as(P(c_i32)) if ann_off is not None else None,
            _pu8(ann_bytes) if ann_bytes is not None else None,
            n, start_ns, 1 if int_optimized else 0, _pu8(out), cap]
    if raw:
        pos = c_i32(0)
        r = L.oracle_encode_series_raw(*args, ctypes.byref(pos))
        if r < 0:
            raise RuntimeError(f"oracle encode error {r}")
        return bytes(out[:r]), pos.value
    r = L.oracle_encode_series(*args)
    if r < 0:
        raise RuntimeError(f"oracle encode error {r}")
    return bytes(out[:r])


def decode_series(data, int_optimized=True, default_unit=1, cap=None,
                  with_annotations=False):
    """Decode one stream. Returns dict with ts, vals, units (+ annotations)."""
    buf = np.frombuffer(bytes(data), dtype=np.uint8)
    if cap is None:
        cap = max(16, len(buf) * 9)
    ts = np.zeros(cap, dtype=np.int64)
    vals = np.zeros(cap, dtype=np.float64)
    units = np.zeros(cap, dtype=np.uint8)
    L = lib()
    if with_annotations:
        ann_lens = np.zeros(cap, dtype=np.int32)
        ann_cap = len(buf) + 4096
        ann_bytes = np.zeros(ann_cap, dtype=np.uint8)
        n = L.oracle_decode_series(_pu8(buf), len(buf), 1 if int_optimized else 0,
                                   default_unit, ts.ctypes.data_as(P(c_i64)),
                                   vals.ctypes.data_as(P(c_f64)), _pu8(units),
                                   ann_lens.ctypes.data_as(P(c_i32)), _pu8(ann_bytes),
                                   ann_cap, cap)
        if n < 0:
            raise RuntimeError(f"oracle decode error {n}")
        anns = []
        off = 0
        for i in range(n):
            if ann_lens[i] >= 0:
                anns.append(bytes(ann_bytes[off:off + ann_lens[i]]))
                off += ann_lens[i]
            else:
                anns.append(None)
        return dict(ts=ts[:n], vals=vals[:n], units=units[:n], annotations=anns)
    n = L.oracle_decode_series(_pu8(buf), len(buf), 1 if int_optimized else 0,
                               default_unit, ts.ctypes.data_as(P(c_i64)),
                               vals.ctypes.data_as(P(c_f64)), _pu8(units),
                               None, None, 0, cap)
    if n < 0:
        raise RuntimeError(f"oracle decode error {n}")
    return dict(ts=ts[:n], vals=vals[:n], units=units[:n])


def encode_batch(ts_ns, vals, counts, int_optimized=True, unit=1,
                 out_stride=None, nthreads=0):
    """Encode SoA batch (nseries x stride rows). Returns (blob_rows, lens)."""
    ts = np.ascontiguousarray(ts_ns, dtype=np.int64)
    v = np.ascontiguousarray(vals, dtype=np.float64)
    c = np.ascontiguousarray(counts, dtype=np.uint32)
    nseries, stride = ts.shape
    if out_stride is None:
        out_stride = 32 + int(stride) * 24
    out = np.zeros((nseries, out_stride), dtype=np.uint8)
    lens = np.zeros(nseries, dtype=np.uint32)
    if nthreads <= 0:
        nthreads = os.cpu_count()
    r = lib().oracle_encode_batch(
        ts.ctypes.data_as(P(c_i64)), v.ctypes.data_as(P(c_f64)),
        c.ctypes.data_as(P(c_u32)), nseries, stride,
        1 if int_optimized else 0, unit, _pu8(out), out_stride,
        lens.ctypes.data_as(P(c_u32)), nthreads)
    if r != 0:
        raise RuntimeError(f"oracle encode_batch error {r}")
    return out, lens


def decode_batch(blobs, offsets, int_optimized=True, default_unit=1, stride=1500,
                 nthreads=0, out_ts=None, out_vals=None, out_counts=None):
    """Decode packed streams (blobs: uint8 array, offsets: uint64 nseries+1)."""
    b = np.ascontiguousarray(blobs, dtype=np.uint8)
    off = np.ascontiguousarray(offsets, dtype=np.uint64)
    nseries = len(off) - 1
    if out_ts is None:
        out_ts = np.zeros((nseries, stride), dtype=np.int64)
        out_vals = np.zeros((nseries, stride), dtype=np.float64)
        out_counts = np.zeros(nseries, dtype=np.uint32)
    if nthreads <= 0:
        nthreads = os.cpu_count()
    r = lib().oracle_decode_batch(
        _pu8(b), off.ctypes.data_as(P(c_u64)), nseries,
        1 if int_optimized else 0, default_unit,
        out_ts.ctypes.data_as(P(c_i64)), out_vals.ctypes.data_as(P(c_f64)),
        out_counts.ctypes.data_as(P(c_u32)), stride, nthreads)
    if r != 0:
        raise RuntimeError(f"oracle decode_batch error {r}")
    return out_ts, out_vals, out_counts


def ckms_quantiles(values, quantiles, eps=1e-3, every=1024):
    """Full CKMS stream (reference defaults): returns (quantile values, min, max)."""
    v = np.ascontiguousarray(values, dtype=np.float64)
    q = np.ascontiguousarray(quantiles, dtype=np.float64)
    out = np.zeros(len(q), dtype=np.float64)
    mn = c_f64(0.0)
    mx = c_f64(0.0)
    lib().oracle_ckms_quantiles(v.ctypes.data_as(P(c_f64)), len(v),
                                q.ctypes.data_as(P(c_f64)), len(q), eps, every,
                                out.ctypes.data_as(P(c_f64)),
                                ctypes.byref(mn), ctypes.byref(mx))
    return out, mn.value, mx.value


def ckms_list_len(values, quantiles, eps=1e-3, every=1024):
    """Post-flush CKMS sample-list length (sizing aid for engine cap tests)."""
    v = np.ascontiguousarray(values, dtype=np.float64)
    q = np.ascontiguousarray(quantiles, dtype=np.float64)
    return int(lib().oracle_ckms_list_len(
        v.ctypes.data_as(P(c_f64)), len(v),
        q.ctypes.data_as(P(c_f64)), len(q), eps, every))


METRIC_COUNTER, METRIC_GAUGE, METRIC_TIMER = 0, 1, 2

AGG = dict(last=1, min=2, max=3, mean=4, median=5, count=6, sum=7, sumsq=8,
           stdev=9, p10=10, p20=11, p30=12, p40=13, p50=14, p60=15, p70=16,
           p80=17, p90=18, p95=19, p99=20, p999=21, p9999=22, p25=23, p75=24)


def rollup_batch(ts_ns, vals, counts, metric_type, window_ns, nbuckets,
                 agg_types, nthreads=0, want_window_ts=True,
                 eps=1e-3, every=1024):
    ts = np.ascontiguousarray(ts_ns, dtype=np.int64)
    v = np.ascontiguousarray(vals, dtype=np.float64)
    c = np.ascontiguousarray(counts, dtype=np.uint32)
    nseries, stride = ts.shape
    aggs = np.asarray([AGG[a] if isinstance(a, str) else a for a in agg_types],
                      dtype=np.int32)
    out = np.zeros((nseries, nbuckets, len(aggs)), dtype=np.float64)
    wts = np.zeros((nseries, nbuckets), dtype=np.int64) if want_window_ts else None
    if nthreads <= 0:
        nthreads = os.cpu_count()
    r = lib().oracle_rollup_batch_opts(
        ts.ctypes.data_as(P(c_i64)), v.ctypes.data_as(P(c_f64)),
        c.ctypes.data_as(P(c_u32)), nseries, stride,
        metric_type, window_ns, nbuckets,
        aggs.ctypes.data_as(P(c_i32)), len(aggs),
        out.ctypes.data_as(P(c_f64)),
        wts.ctypes.data_as(P(c_i64)) if wts is not None else None, nthreads,
        eps, every)
    if r != 0:
        raise RuntimeError(f"oracle rollup error {r}")
    return (out, wts) if want_window_ts else out


def xxhash64(data):
    buf = np.frombuffer(bytes(data), dtype=np.uint8)
    if len(buf) == 0:
        buf = np.zeros(1, dtype=np.uint8)
        return lib().oracle_xxhash64(_pu8(buf), 0)
    return lib().oracle_xxhash64(_pu8(buf), len(buf))


def merge_batch(ts, vals, counts, out_stride=None, nthreads=0):
    """Replica-deduplicating merge (MultiReaderIterator semantics over one
    slice of R replica iterators). ts/vals: [R, nseries, stride]; counts:
    [R, nseries]. Returns (out_ts, out_vals, out_counts, out_errs)."""
    ts = np.ascontiguousarray(ts, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    counts = np.ascontiguousarray(counts, dtype=np.uint32)
    r, nseries, stride = ts.shape
    if out_stride is None:
        out_stride = stride * r
    out_ts = np.zeros((nseries, out_stride), dtype=np.int64)
    out_vals = np.zeros((nseries, out_stride), dtype=np.float64)
    out_counts = np.zeros(nseries, dtype=np.uint32)
    out_errs = np.zeros(nseries, dtype=np.int32)
    L = lib()
    L.oracle_merge_batch.restype = c_int
    L.oracle_merge_batch.argtypes = [P(c_i64), P(c_f64), P(c_u32), c_int,
                                     c_i64, c_i64, P(c_i64), P(c_f64),
                                     P(c_u32), c_i64, P(c_i32), c_int]
    if nthreads <= 0:
        nthreads = os.cpu_count()
    L.oracle_merge_batch(ts.ctypes.data_as(P(c_i64)),
                         vals.ctypes.data_as(P(c_f64)),
                         counts.ctypes.data_as(P(c_u32)), r, nseries, stride,
                         out_ts.ctypes.data_as(P(c_i64)),
                         out_vals.ctypes.data_as(P(c_f64)),
                         out_counts.ctypes.data_as(P(c_u32)), out_stride,
                         out_errs.ctypes.data_as(P(c_i32)), nthreads)
    return out_ts, out_vals, out_counts, out_errs
