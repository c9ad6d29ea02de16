"""ctypes binding of libm3gpu.so (include/m3gpu.h) + torch-tensor surface.

Device entry points (`*_dev`) take torch CUDA tensors (ROCm) and enqueue on
the current torch stream; host entry points take numpy arrays and block.
"""
import ctypes
import os

import numpy as np

c_i64 = ctypes.c_int64
c_u64 = ctypes.c_uint64
c_i32 = ctypes.c_int32
c_u32 = ctypes.c_uint32
c_u8 = ctypes.c_uint8
c_f64 = ctypes.c_double
c_int = ctypes.c_int
c_vp = ctypes.c_void_p
P = ctypes.POINTER

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "csrc", "libm3gpu.so")

METRIC_COUNTER, METRIC_GAUGE, METRIC_TIMER = 0, 1, 2

M3GPU_AGG = dict(last=1, min=2, max=3, mean=4, median=5, count=6, sum=7,
                 sumsq=8, stdev=9, p10=10, p20=11, p30=12, p40=13, p50=14,
                 p60=15, p70=16, p80=17, p90=18, p95=19, p99=20, p999=21,
                 p9999=22, p25=23, p75=24)

SERIES_ERRORS = {0: "ok", 1: "eof", 2: "dod_overflow", 3: "no_scheme",
                 4: "invalid_mult", 5: "annotation", 6: "capacity",
                 7: "unsorted", 8: "bucket_overflow"}


class M3GpuError(RuntimeError):
    pass


_lib = None


def engine_available():
    return os.path.exists(_LIB_PATH)


def lib():
    """Load libm3gpu.so. Raises loudly when the HIP engine is missing —
    the product path has no CPU fallback."""
    global _lib
    if _lib is None:
        if not engine_available():
            raise M3GpuError(
                f"libm3gpu.so not found at {_LIB_PATH}; build it with "
                f"`make -C m3_amd/csrc` (hipcc --offload-arch=gfx950). "
                f"The m3_amd product path has no CPU fallback.")
        L = ctypes.CDLL(_LIB_PATH)
        L.m3gpu_init.restype = c_int
        L.m3gpu_init.argtypes = [c_int]
        L.m3gpu_last_error.restype = ctypes.c_char_p
        L.m3gpu_decode_batch_dev.restype = c_int
        L.m3gpu_decode_batch_dev.argtypes = [c_vp, c_vp, c_vp, c_u32, c_int, c_u8,
                                             c_vp, c_vp, c_vp, c_vp, c_u32, c_vp]
        L.m3gpu_decode_batch_dev_perm.restype = c_int
        L.m3gpu_decode_batch_dev_perm.argtypes = [c_vp, c_vp, c_vp, c_vp, c_u32,
                                                  c_int, c_u8, c_vp, c_vp, c_vp,
                                                  c_vp, c_u32, c_vp]
        L.m3gpu_decode_batch_dev_ann.restype = c_int
        L.m3gpu_decode_batch_dev_ann.argtypes = [c_vp, c_vp, c_vp, c_u32,
                                                 c_int, c_u8, c_vp, c_vp, c_vp,
                                                 c_vp, c_u32, c_vp, c_u32, c_vp]
        L.m3gpu_decode_batch.restype = c_int
        L.m3gpu_decode_batch.argtypes = [P(c_u8), c_u64, P(c_u64), P(c_u32), c_u32,
                                         c_int, c_u8, P(c_i64), P(c_f64), P(c_u32),
                                         P(c_i32), c_u32]
        L.m3gpu_encode_batch_dev.restype = c_int
        L.m3gpu_encode_batch_dev.argtypes = [c_vp, c_vp, c_vp, c_u32, c_u32, c_int,
                                             c_u8, c_vp, c_u32, c_vp, c_vp, c_vp]
        L.m3gpu_encode_batch.restype = c_int
        L.m3gpu_encode_batch.argtypes = [P(c_i64), P(c_f64), P(c_u32), c_u32, c_u32,
                                         c_int, c_u8, P(c_u8), c_u32, P(c_u32), P(c_i32)]
        L.m3gpu_compact_dev.restype = c_int
        L.m3gpu_compact_dev.argtypes = [c_vp, c_u32, c_vp, c_vp, c_u32, c_vp, c_vp]
        L.m3gpu_rollup_batch_dev.restype = c_int
        L.m3gpu_rollup_batch_dev.argtypes = [c_vp, c_vp, c_vp, c_u32, c_int, c_u8,
                                             c_int, c_i64, c_u32, P(c_i32), c_int,
                                             c_vp, c_vp, c_vp, c_vp]
        L.m3gpu_rollup_batch.restype = c_int
        L.m3gpu_rollup_batch.argtypes = [P(c_u8), c_u64, P(c_u64), P(c_u32), c_u32,
                                         c_int, c_u8, c_int, c_i64, c_u32, P(c_i32),
                                         c_int, P(c_f64), P(c_i64), P(c_i32)]
        _lib = L
    return _lib


def _check(rc, what):
    if rc != 0:
        raise M3GpuError(f"{what} failed ({rc}): {lib().m3gpu_last_error().decode()}")


def _raise_series_errors(errs, what):
    errs = np.asarray(errs)
    bad = np.nonzero(errs)[0]
    if len(bad):
        i = int(bad[0])
        raise M3GpuError(
            f"{what}: {len(bad)} series failed; first: series {i} -> "
            f"{SERIES_ERRORS.get(int(errs[i]), errs[i])}")


def pack_streams(streams):
    """Pack a list of encoded streams into (blob, offsets, lens) with the
    64-byte-aligned zero-padded layout (one stream chunk = one HBM line,
    so the decode ring's chunk refills are line-aligned; the C ABI itself
    requires only 8B)."""
    n = len(streams)
    lens = np.fromiter((len(s) for s in streams), dtype=np.uint32, count=n)
    padded = (lens.astype(np.uint64) + 63) & ~np.uint64(63)
    offsets = np.zeros(n + 1, dtype=np.uint64)
    np.cumsum(padded, out=offsets[1:])
    blob = np.zeros(int(offsets[-1]), dtype=np.uint8)
    for i, s in enumerate(streams):
        o = int(offsets[i])
        blob[o:o + len(s)] = np.frombuffer(bytes(s), dtype=np.uint8)
    return blob, offsets, lens


def _np(a, dt):
    return np.ascontiguousarray(a, dtype=dt)


def _pp(a, ct):
    return a.ctypes.data_as(P(ct))


# ------------------------- host-pointer surface -------------------------

def decode_batch(blob, offsets, lens, stride, int_optimized=True,
                 default_unit=1, check_errors=True):
    blob = _np(blob, np.uint8)
    offsets = _np(offsets, np.uint64)
    lens = _np(lens, np.uint32)
    nseries = len(lens)
    out_ts = np.empty((nseries, stride), dtype=np.int64)
    out_vals = np.empty((nseries, stride), dtype=np.float64)
    out_counts = np.empty(nseries, dtype=np.uint32)
    out_errs = np.empty(nseries, dtype=np.int32)
    rc = lib().m3gpu_decode_batch(
        _pp(blob, c_u8), len(blob), _pp(offsets, c_u64), _pp(lens, c_u32),
        nseries, 1 if int_optimized else 0, default_unit,
        _pp(out_ts, c_i64), _pp(out_vals, c_f64), _pp(out_counts, c_u32),
        _pp(out_errs, c_i32), stride)
    _check(rc, "m3gpu_decode_batch")
    if check_errors:
        _raise_series_errors(out_errs, "decode")
    return out_ts, out_vals, out_counts, out_errs


def encode_batch(ts, vals, counts, int_optimized=True, unit=1,
                 out_stride=None, check_errors=True):
    ts = _np(ts, np.int64)
    vals = _np(vals, np.float64)
    counts = _np(counts, np.uint32)
    nseries, stride = ts.shape
    if out_stride is None:
        out_stride = (24 * stride + 32 + 7) & ~7
    out_bytes = np.zeros((nseries, out_stride), dtype=np.uint8)
    out_lens = np.empty(nseries, dtype=np.uint32)
    out_errs = np.empty(nseries, dtype=np.int32)
    rc = lib().m3gpu_encode_batch(
        _pp(ts, c_i64), _pp(vals, c_f64), _pp(counts, c_u32), nseries, stride,
        1 if int_optimized else 0, unit, _pp(out_bytes, c_u8), out_stride,
        _pp(out_lens, c_u32), _pp(out_errs, c_i32))
    _check(rc, "m3gpu_encode_batch")
    if check_errors:
        _raise_series_errors(out_errs, "encode")
    return out_bytes, out_lens, out_errs


def rollup_batch(blob, offsets, lens, metric_type, window_ns, nbuckets,
                 agg_types, int_optimized=True, default_unit=1,
                 check_errors=True):
    blob = _np(blob, np.uint8)
    offsets = _np(offsets, np.uint64)
    lens = _np(lens, np.uint32)
    nseries = len(lens)
    aggs = np.asarray([M3GPU_AGG[a] if isinstance(a, str) else a
                       for a in agg_types], dtype=np.int32)
    out = np.empty((nseries, nbuckets, len(aggs)), dtype=np.float64)
    wts = np.empty((nseries, nbuckets), dtype=np.int64)
    errs = np.empty(nseries, dtype=np.int32)
    rc = lib().m3gpu_rollup_batch(
        _pp(blob, c_u8), len(blob), _pp(offsets, c_u64), _pp(lens, c_u32),
        nseries, 1 if int_optimized else 0, default_unit, metric_type,
        window_ns, nbuckets, _pp(aggs, c_i32), len(aggs),
        _pp(out, c_f64), _pp(wts, c_i64), _pp(errs, c_i32))
    _check(rc, "m3gpu_rollup_batch")
    if check_errors:
        _raise_series_errors(errs, "rollup")
    return out, wts, errs


# ------------------------- torch device surface -------------------------

def _dev_ptr(t):
    return ctypes.c_void_p(t.data_ptr())


def _torch_stream():
    import torch
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def decode_batch_dev(d_blob, d_offsets, d_lens, out_ts, out_vals, out_counts,
                     out_errs, int_optimized=True, default_unit=1, d_perm=None):
    """All args are torch CUDA tensors; enqueues on the current torch stream.
    out_ts: int64 [nseries, stride]; out_vals: float64 [nseries, stride].
    d_perm (int32, optional): scheduling permutation — e.g.
    torch.argsort(d_lens) so each wavefront decodes similar-cost streams."""
    nseries = d_lens.numel()
    stride = out_ts.shape[1]
    rc = lib().m3gpu_decode_batch_dev_perm(
        _dev_ptr(d_blob), _dev_ptr(d_offsets), _dev_ptr(d_lens),
        _dev_ptr(d_perm) if d_perm is not None else None, nseries,
        1 if int_optimized else 0, default_unit, _dev_ptr(out_ts),
        _dev_ptr(out_vals), _dev_ptr(out_counts), _dev_ptr(out_errs), stride,
        _torch_stream())
    _check(rc, "m3gpu_decode_batch_dev")


def decode_batch_dev_ann(d_blob, d_offsets, d_lens, out_ts, out_vals,
                         out_counts, out_errs, out_ann, int_optimized=True,
                         default_unit=1):
    """Annotation-capturing decode (materializes ReaderIterator.Current()'s
    third return, iterator.go:226-231). out_ann: uint8 [nseries, ann_stride]
    device tensor; each series' region holds
    [u32 n_events][n_events x {u32 point, u32 off, u32 len}][...bytes]
    (bytes grow from the tail). Decode with parse_ann_region / iterate with
    decoded_ann_per_point."""
    nseries = d_lens.numel()
    stride = out_ts.shape[1]
    rc = lib().m3gpu_decode_batch_dev_ann(
        _dev_ptr(d_blob), _dev_ptr(d_offsets), _dev_ptr(d_lens), nseries,
        1 if int_optimized else 0, default_unit, _dev_ptr(out_ts),
        _dev_ptr(out_vals), _dev_ptr(out_counts), _dev_ptr(out_errs), stride,
        _dev_ptr(out_ann), out_ann.shape[1], _torch_stream())
    _check(rc, "m3gpu_decode_batch_dev_ann")


def parse_ann_region(region):
    """Parse one series' annotation region (host uint8 array) into a list of
    (point_idx, bytes) set-events in stream order."""
    region = np.asarray(region, dtype=np.uint8)
    n = int(region[:4].view(np.uint32)[0])
    out = []
    for i in range(n):
        ev = region[4 + i * 12: 4 + i * 12 + 12].view(np.uint32)
        point, off, ln = int(ev[0]), int(ev[1]), int(ev[2])
        out.append((point, bytes(region[off:off + ln])))
    return out


def decoded_ann_per_point(region, npts):
    """Materialize the sticky PrevAnt view (iterator.go:226-231): the
    annotation returned with each of npts datapoints (None until the first
    set-event)."""
    events = parse_ann_region(region)
    out = [None] * npts
    cur = None
    j = 0
    for p in range(npts):
        while j < len(events) and events[j][0] <= p:
            cur = events[j][1]
            j += 1
        out[p] = cur
    return out


def encode_batch_dev(d_ts, d_vals, d_counts, out_bytes, out_lens, out_errs,
                     int_optimized=True, unit=1):
    nseries, stride = d_ts.shape
    out_stride = out_bytes.shape[1]
    rc = lib().m3gpu_encode_batch_dev(
        _dev_ptr(d_ts), _dev_ptr(d_vals), _dev_ptr(d_counts), nseries, stride,
        1 if int_optimized else 0, unit, _dev_ptr(out_bytes), out_stride,
        _dev_ptr(out_lens), _dev_ptr(out_errs), _torch_stream())
    _check(rc, "m3gpu_encode_batch_dev")


def compact_dev(d_src, src_stride, d_lens, d_dst_offsets, d_dst):
    nseries = d_lens.numel()
    rc = lib().m3gpu_compact_dev(
        _dev_ptr(d_src), src_stride, _dev_ptr(d_lens), _dev_ptr(d_dst_offsets),
        nseries, _dev_ptr(d_dst), _torch_stream())
    _check(rc, "m3gpu_compact_dev")


def regather_dev(d_blob, d_offsets, d_lens, d_perm, d_dst_offsets, d_dst):
    """Layout pass: physically reorder packed streams into perm order
    (k_regather) so each decoding wavefront's 64 streams are neighbors in
    HBM. Pair with decode_batch_dev(d_perm=None) on the result."""
    L = lib()
    if not hasattr(L.m3gpu_regather_dev, "_configured"):
        L.m3gpu_regather_dev.restype = c_int
        L.m3gpu_regather_dev.argtypes = [c_vp, c_vp, c_vp, c_vp, c_vp, c_u32,
                                         c_vp, c_vp]
        L.m3gpu_regather_dev._configured = True
    rc = L.m3gpu_regather_dev(
        _dev_ptr(d_blob), _dev_ptr(d_offsets), _dev_ptr(d_lens),
        _dev_ptr(d_perm), _dev_ptr(d_dst_offsets), d_lens.numel(),
        _dev_ptr(d_dst), _torch_stream())
    _check(rc, "m3gpu_regather_dev")


def rollup_batch_dev(d_blob, d_offsets, d_lens, metric_type, window_ns,
                     nbuckets, agg_types, out, out_window_ts, out_errs,
                     int_optimized=True, default_unit=1,
                     eps=1e-3, every=1024):
    """eps/every: CKMS stream options (reference defaults; cm
    options.go:30-32). Non-default values route via the _opts ABI."""
    nseries = d_lens.numel()
    aggs = np.asarray([M3GPU_AGG[a] if isinstance(a, str) else a
                       for a in agg_types], dtype=np.int32)
    L = lib()
    if not hasattr(L.m3gpu_rollup_batch_dev_opts, "_configured"):
        L.m3gpu_rollup_batch_dev_opts.restype = c_int
        L.m3gpu_rollup_batch_dev_opts.argtypes =             list(L.m3gpu_rollup_batch_dev.argtypes) + [c_f64, c_int]
        L.m3gpu_rollup_batch_dev_opts._configured = True
    rc = L.m3gpu_rollup_batch_dev_opts(
        _dev_ptr(d_blob), _dev_ptr(d_offsets), _dev_ptr(d_lens), nseries,
        1 if int_optimized else 0, default_unit, metric_type, window_ns,
        nbuckets, _pp(aggs, c_i32), len(aggs), _dev_ptr(out),
        _dev_ptr(out_window_ts), _dev_ptr(out_errs), _torch_stream(),
        eps, every)
    _check(rc, "m3gpu_rollup_batch_dev")


def aggregate_tiles_dev(torch, d_blob, d_offsets, d_lens, metric_type,
                        window_ns, nbuckets, agg, int_optimized=True,
                        default_unit=1, unit_out=1):
    """AggregateTiles-shaped backend (what dbnode's large-tiles job,
    storage/shard.go:2682, would drive): fused decode -> windowed rollup with
    m3aggregator semantics -> re-encode one downsampled M3TSZ stream per
    series. Each non-empty tile emits (window-END timestamp, agg value)
    (list.go:541-543); empty tiles emit nothing. Returns (d_tile_bytes
    [n, out_stride] uint8, d_tile_lens int32, d_counts int32).
    All work stays on-device (rollup -> gap compaction -> encode)."""
    nseries = d_lens.numel()
    aggs = [agg, "count"]
    out = torch.empty((nseries, nbuckets, 2), dtype=torch.float64, device=d_blob.device)
    wts = torch.empty((nseries, nbuckets), dtype=torch.int64, device=d_blob.device)
    errs = torch.empty(nseries, dtype=torch.int32, device=d_blob.device)
    rollup_batch_dev(d_blob, d_offsets, d_lens, metric_type, window_ns,
                     nbuckets, aggs, out, wts, errs,
                     int_optimized=int_optimized, default_unit=default_unit)
    # compact out the empty tiles (stable: valid buckets keep time order);
    # when no series has gaps (the common dense case) skip the gather
    valid = out[:, :, 1] > 0
    counts = valid.sum(1).to(torch.int32)
    if bool((counts == nbuckets).all().item()):
        ts_c = wts
        vals_c = out[:, :, 0]
    else:
        order = torch.argsort((~valid).to(torch.int8), dim=1, stable=True)
        ts_c = torch.gather(wts, 1, order)
        vals_c = torch.gather(out[:, :, 0].contiguous(), 1, order)
    out_stride = (24 * nbuckets + 32 + 15) & ~15
    d_tile_bytes = torch.zeros((nseries, out_stride), dtype=torch.uint8,
                               device=d_blob.device)
    d_tile_lens = torch.empty(nseries, dtype=torch.int32, device=d_blob.device)
    d_tile_errs = torch.empty(nseries, dtype=torch.int32, device=d_blob.device)
    encode_batch_dev(ts_c.contiguous(), vals_c.contiguous(), counts,
                     d_tile_bytes, d_tile_lens, d_tile_errs,
                     int_optimized=int_optimized, unit=unit_out)
    torch.cuda.synchronize()
    _raise_series_errors(errs.cpu().numpy(), "tiles rollup")
    _raise_series_errors(d_tile_errs.cpu().numpy(), "tiles encode")
    return d_tile_bytes, d_tile_lens, counts


def merge_batch_dev(d_ts, d_vals, d_counts, nreplicas, out_ts, out_vals,
                    out_counts, out_errs):
    """Replica-deduplicating merge (MultiReaderIterator semantics,
    IterateLastPushed). d_ts/d_vals: [nreplicas*nseries, stride] rows
    (replica-major); outputs [nseries, out_stride]."""
    nseries = d_counts.numel() // nreplicas
    stride = d_ts.shape[-1]
    out_stride = out_ts.shape[-1]
    L = lib()
    if not hasattr(L.m3gpu_merge_batch_dev, "_configured"):
        L.m3gpu_merge_batch_dev.restype = c_int
        L.m3gpu_merge_batch_dev.argtypes = [c_vp, c_vp, c_vp, c_u32, c_u32,
                                            c_u32, c_vp, c_vp, c_vp, c_vp,
                                            c_u32, c_vp]
        L.m3gpu_merge_batch_dev._configured = True
    rc = L.m3gpu_merge_batch_dev(
        _dev_ptr(d_ts), _dev_ptr(d_vals), _dev_ptr(d_counts), nreplicas,
        nseries, stride, _dev_ptr(out_ts), _dev_ptr(out_vals),
        _dev_ptr(out_counts), _dev_ptr(out_errs), out_stride, _torch_stream())
    _check(rc, "m3gpu_merge_batch_dev")


# ========================= fileset volume reader =========================

FS_ERRORS = {-101: "io", -102: "checkpoint", -103: "digest", -104: "msgpack",
             -105: "schema", -106: "entry_checksum", -107: "data_checksum",
             -108: "bounds", -109: "badhandle", -110: "capacity"}


def _fs_configure(L):
    if getattr(L, "_fs_configured", False):
        return
    L.m3gpu_fileset_open.restype = c_int
    L.m3gpu_fileset_open.argtypes = [ctypes.c_char_p, c_i64, c_int]
    L.m3gpu_fileset_close.restype = c_int
    L.m3gpu_fileset_close.argtypes = [c_int]
    L.m3gpu_fileset_last_error.restype = ctypes.c_char_p
    L.m3gpu_fileset_info.restype = c_int
    L.m3gpu_fileset_info.argtypes = [c_int] + [P(c_i64)] * 5 + [P(c_int)] + [P(c_i64)] * 3
    L.m3gpu_fileset_entry.restype = c_int
    L.m3gpu_fileset_entry.argtypes = [c_int, c_i64, P(c_i64), P(c_i64),
                                      P(c_i64), P(c_vp), P(c_i64), P(c_vp),
                                      P(c_i64)]
    L.m3gpu_fileset_packed_size.restype = c_i64
    L.m3gpu_fileset_packed_size.argtypes = [c_int]
    L.m3gpu_fileset_pack.restype = c_int
    L.m3gpu_fileset_pack.argtypes = [c_int, P(c_u8), c_u64, P(c_u64), P(c_u32)]
    L._fs_configured = True


class FilesetVolume:
    """A validated, opened fileset volume (native reader fileset.cpp).

    Replaces the reference's DataFileSetReader bulk loop
    (persist/fs/read.go:413-457) for ingestion: entries come back sorted
    by data offset ascending with IDs/tags, and pack() yields the
    decode-batch blob layout for the GPU codec.
    """

    def __init__(self, shard_dir, block_start_ns, volume_index=0):
        L = lib()
        _fs_configure(L)
        h = L.m3gpu_fileset_open(str(shard_dir).encode(), block_start_ns,
                                 volume_index)
        if h < 0:
            detail = L.m3gpu_fileset_last_error().decode()
            raise M3GpuError(
                f"fileset open failed: {FS_ERRORS.get(h, h)} ({detail})")
        self._h = h
        self._lib = L
        info = [c_i64() for _ in range(5)]
        vol = c_int()
        extra = [c_i64() for _ in range(3)]
        L.m3gpu_fileset_info(h, *[ctypes.byref(x) for x in info],
                             ctypes.byref(vol),
                             *[ctypes.byref(x) for x in extra])
        self.block_start = info[0].value
        self.block_size = info[1].value
        self.num_entries = info[2].value
        self.major_version = info[3].value
        self.minor_version = info[4].value
        self.volume_index = vol.value
        self.bloom_m, self.bloom_k, self.summaries = (x.value for x in extra)

    def close(self):
        if self._h >= 0:
            self._lib.m3gpu_fileset_close(self._h)
            self._h = -1

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def entries(self):
        """[(id: bytes, size, data_offset, checksum, tags: bytes)] sorted by
        data offset ascending."""
        out = []
        sz, off, ck, idl, tgl = c_i64(), c_i64(), c_i64(), c_i64(), c_i64()
        idp, tgp = c_vp(), c_vp()
        for i in range(self.num_entries):
            rc = self._lib.m3gpu_fileset_entry(
                self._h, i, ctypes.byref(sz), ctypes.byref(off),
                ctypes.byref(ck), ctypes.byref(idp), ctypes.byref(idl),
                ctypes.byref(tgp), ctypes.byref(tgl))
            _check(rc, "m3gpu_fileset_entry")
            sid = ctypes.string_at(idp, idl.value) if idl.value else b""
            tags = ctypes.string_at(tgp, tgl.value) if tgl.value else b""
            out.append((sid, sz.value, off.value, ck.value, tags))
        return out

    def pack(self):
        """Repack data blocks into the decode-batch layout: returns
        (blob: uint8 ndarray, offsets: uint64 ndarray, lens: uint32)."""
        total = self._lib.m3gpu_fileset_packed_size(self._h)
        if total < 0:
            raise M3GpuError("packed_size failed")
        blob = np.zeros(max(int(total), 16), np.uint8)
        offsets = np.zeros(self.num_entries, np.uint64)
        lens = np.zeros(self.num_entries, np.uint32)
        rc = self._lib.m3gpu_fileset_pack(
            self._h, blob.ctypes.data_as(P(c_u8)), blob.nbytes,
            offsets.ctypes.data_as(P(c_u64)), lens.ctypes.data_as(P(c_u32)))
        _check(rc, "m3gpu_fileset_pack")
        return blob, offsets, lens


def fileset_ingest_dev(torch, shard_dir, block_start_ns, volume_index=0,
                       stride=None, int_optimized=True, default_unit=1,
                       device="cuda:0"):
    """End-to-end volume ingestion on the GPU: open+validate the volume
    (native reader), repack, H2D, decode every block with the HIP codec.
    Returns (ids, d_ts, d_vals, d_counts, d_errs).

    Replaces the reference's bootstrap/fileset read into memory
    (read.go:413-457 feeding m3tsz.NewReaderIterator per block) with one
    batched decode."""
    with FilesetVolume(shard_dir, block_start_ns, volume_index) as v:
        ids = [e[0] for e in v.entries()]
        blob, offsets, lens = v.pack()
        if stride is None:
            # worst case: every remaining byte after the 11B header is a
            # 1-bit zero-DoD point (+ guard)
            stride = int(max(1, (lens.max() - 11)) * 8 + 8) if len(lens) else 8
        n = v.num_entries
    d_blob = torch.from_numpy(blob).to(device)
    d_off = torch.from_numpy(offsets.astype(np.int64)).to(device)
    d_lens = torch.from_numpy(lens.astype(np.int32)).to(device)
    d_ts = torch.zeros((n, stride), dtype=torch.int64, device=device)
    d_vals = torch.zeros((n, stride), dtype=torch.float64, device=device)
    d_counts = torch.empty(n, dtype=torch.int32, device=device)
    d_errs = torch.empty(n, dtype=torch.int32, device=device)
    decode_batch_dev(d_blob, d_off, d_lens, d_ts, d_vals, d_counts, d_errs,
                     int_optimized=int_optimized, default_unit=default_unit)
    return ids, d_ts, d_vals, d_counts, d_errs


# =========================== commit log reader ===========================

CL_ERRORS = dict(FS_ERRORS)
CL_ERRORS.update({-111: "chunk_checksum", -112: "missing_metadata",
                  -113: "truncated"})


def _cl_configure(L):
    if getattr(L, "_cl_configured", False):
        return
    L.m3gpu_commitlog_open.restype = c_int
    L.m3gpu_commitlog_open.argtypes = [ctypes.c_char_p]
    L.m3gpu_commitlog_close.restype = c_int
    L.m3gpu_commitlog_close.argtypes = [c_int]
    L.m3gpu_commitlog_last_error.restype = ctypes.c_char_p
    L.m3gpu_commitlog_info.restype = c_int
    L.m3gpu_commitlog_info.argtypes = [c_int, P(c_i64), P(c_i64), P(c_i64)]
    L.m3gpu_commitlog_series.restype = c_int
    L.m3gpu_commitlog_series.argtypes = [c_int, c_i64, P(c_u64), P(c_vp),
                                         P(c_i64), P(c_vp), P(c_i64),
                                         P(c_u32), P(c_vp), P(c_i64),
                                         P(c_i64), P(c_i64)]
    L.m3gpu_commitlog_series_points.restype = c_int
    L.m3gpu_commitlog_series_points.argtypes = [c_int, c_i64, P(c_i64),
                                                P(c_f64), P(c_u8)]
    L.m3gpu_commitlog_series_annotation.restype = c_int
    L.m3gpu_commitlog_series_annotation.argtypes = [c_int, c_i64, c_i64,
                                                    P(c_i64), P(c_vp),
                                                    P(c_i64)]
    L._cl_configured = True


class CommitLog:
    """A parsed, checksum-validated commit log file (commitlog.cpp).

    Replaces the reference's commitlog Reader loop
    (persist/fs/commitlog/reader.go:161-209) for bootstrap: series in
    first-seen order, each with its datapoints in log order."""

    def __init__(self, path):
        L = lib()
        _cl_configure(L)
        h = L.m3gpu_commitlog_open(str(path).encode())
        if h < 0:
            detail = L.m3gpu_commitlog_last_error().decode()
            raise M3GpuError(
                f"commitlog open failed: {CL_ERRORS.get(h, h)} ({detail})")
        self._h = h
        self._lib = L
        idx, ne, ns = c_i64(), c_i64(), c_i64()
        L.m3gpu_commitlog_info(h, ctypes.byref(idx), ctypes.byref(ne),
                               ctypes.byref(ns))
        self.index = idx.value
        self.num_entries = ne.value
        self.num_series = ns.value

    def close(self):
        if self._h >= 0:
            self._lib.m3gpu_commitlog_close(self._h)
            self._h = -1

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def series(self):
        """[{id, namespace, shard, tags, unique_index, ts, vals, units,
        annotations: [(point_idx, bytes)]}] in first-seen order."""
        out = []
        for i in range(self.num_series):
            ui = c_u64()
            idp, nsp, tgp = c_vp(), c_vp(), c_vp()
            idl, nsl, tgl = c_i64(), c_i64(), c_i64()
            shard = c_u32()
            npts, nant = c_i64(), c_i64()
            rc = self._lib.m3gpu_commitlog_series(
                self._h, i, ctypes.byref(ui), ctypes.byref(idp),
                ctypes.byref(idl), ctypes.byref(nsp), ctypes.byref(nsl),
                ctypes.byref(shard), ctypes.byref(tgp), ctypes.byref(tgl),
                ctypes.byref(npts), ctypes.byref(nant))
            _check(rc, "m3gpu_commitlog_series")
            n = npts.value
            ts = np.empty(n, np.int64)
            vals = np.empty(n, np.float64)
            units = np.empty(n, np.uint8)
            rc = self._lib.m3gpu_commitlog_series_points(
                self._h, i, ts.ctypes.data_as(P(c_i64)),
                vals.ctypes.data_as(P(c_f64)), units.ctypes.data_as(P(c_u8)))
            _check(rc, "m3gpu_commitlog_series_points")
            annotations = []
            for j in range(nant.value):
                pi, ap, al = c_i64(), c_vp(), c_i64()
                self._lib.m3gpu_commitlog_series_annotation(
                    self._h, i, j, ctypes.byref(pi), ctypes.byref(ap),
                    ctypes.byref(al))
                annotations.append((pi.value, ctypes.string_at(ap, al.value)))
            out.append(dict(
                unique_index=ui.value,
                id=ctypes.string_at(idp, idl.value) if idl.value else b"",
                namespace=ctypes.string_at(nsp, nsl.value) if nsl.value else b"",
                shard=shard.value,
                tags=ctypes.string_at(tgp, tgl.value) if tgl.value else b"",
                ts=ts, vals=vals, units=units, annotations=annotations))
        return out


def commitlog_bootstrap_dev(torch, path, int_optimized=True, device="cuda:0"):
    """Bootstrap-from-commitlog on the GPU: parse + validate the log
    (native reader), group per series, batch-encode every series into
    M3TSZ blocks with the HIP encoder. Returns (series_meta, d_bytes,
    d_lens, d_errs) where series_meta is the CommitLog.series() list.

    Replaces the reference's commitlog bootstrapper read-and-re-encode
    (bootstrap/bootstrapper/commitlog + series buffer encoders) with one
    batched GPU encode."""
    with CommitLog(path) as cl:
        meta = cl.series()
    if not meta:
        return [], None, None, None
    n = len(meta)
    width = max(len(m["ts"]) for m in meta)
    ts = np.zeros((n, width), np.int64)
    vals = np.zeros((n, width), np.float64)
    counts = np.zeros(n, np.int32)
    for i, m in enumerate(meta):
        k = len(m["ts"])
        ts[i, :k] = m["ts"]
        vals[i, :k] = m["vals"]
        counts[i] = k
    d_ts = torch.from_numpy(ts).to(device)
    d_vals = torch.from_numpy(vals).to(device)
    d_counts = torch.from_numpy(counts).to(device)
    out_stride = (24 * width + 32 + 15) & ~15
    d_bytes = torch.zeros((n, out_stride), dtype=torch.uint8, device=device)
    d_lens = torch.empty(n, dtype=torch.int32, device=device)
    d_errs = torch.empty(n, dtype=torch.int32, device=device)
    encode_batch_dev(d_ts, d_vals, d_counts, d_bytes, d_lens, d_errs,
                     int_optimized=int_optimized)
    return meta, d_bytes, d_lens, d_errs


# ==================== unaggregated metric wire parser ====================

UA_ERRORS = {-121: "truncated", -122: "proto", -123: "type",
             -124: "badhandle", -125: "size"}
UA_TYPES = {1: "counter", 2: "batch_timer", 3: "gauge",
            5: "timed_with_metadata", 6: "timed_with_metadatas",
            7: "timed_with_storage_policy"}


def parse_unaggregated(buf):
    """Parse an unaggregated-metric wire buffer (m3aggregator ingest,
    metrics/encoding/protobuf framing) with the native parser (unagg.cpp).
    Returns a list of dicts: {type, metric_type, id, values, counter_value,
    time_nanos, annotation, metadatas}."""
    L = lib()
    if not getattr(L, "_ua_configured", False):
        L.m3gpu_unagg_parse.restype = c_int
        L.m3gpu_unagg_parse.argtypes = [P(c_u8), c_u64]
        L.m3gpu_unagg_close.restype = c_int
        L.m3gpu_unagg_close.argtypes = [c_int]
        L.m3gpu_unagg_last_error.restype = ctypes.c_char_p
        L.m3gpu_unagg_count.restype = c_i64
        L.m3gpu_unagg_count.argtypes = [c_int]
        L.m3gpu_unagg_metric.restype = c_int
        L.m3gpu_unagg_metric.argtypes = [c_int, c_i64, P(c_i32), P(c_i32),
                                         P(c_vp), P(c_i64), P(c_i64),
                                         P(c_i64), P(c_i64), P(c_vp),
                                         P(c_i64), P(c_vp), P(c_i64)]
        L.m3gpu_unagg_values.restype = c_int
        L.m3gpu_unagg_values.argtypes = [c_int, c_i64, P(c_f64)]
        L.m3gpu_unagg_metadata_count.restype = c_i64
        L.m3gpu_unagg_metadata_count.argtypes = [c_int, c_i64]
        L.m3gpu_unagg_metadata.restype = c_int
        L.m3gpu_unagg_metadata.argtypes = [c_int, c_i64, c_i64, P(c_i32),
                                           P(c_vp), P(c_i64)]
        L._ua_configured = True
    b = np.frombuffer(bytes(buf), dtype=np.uint8)
    h = L.m3gpu_unagg_parse(b.ctypes.data_as(P(c_u8)), b.nbytes)
    if h < 0:
        detail = L.m3gpu_unagg_last_error().decode()
        raise M3GpuError(
            f"unagg parse failed: {UA_ERRORS.get(h, h)} ({detail})")
    try:
        out = []
        for i in range(L.m3gpu_unagg_count(h)):
            ut, mt = c_i32(), c_i32()
            idp, anp, mdp = c_vp(), c_vp(), c_vp()
            idl, nv, cv, tn, anl, mdl = (c_i64() for _ in range(6))
            rc = L.m3gpu_unagg_metric(
                h, i, ctypes.byref(ut), ctypes.byref(mt), ctypes.byref(idp),
                ctypes.byref(idl), ctypes.byref(nv), ctypes.byref(cv),
                ctypes.byref(tn), ctypes.byref(anp), ctypes.byref(anl),
                ctypes.byref(mdp), ctypes.byref(mdl))
            _check(rc, "m3gpu_unagg_metric")
            vals = np.empty(nv.value, np.float64)
            if nv.value:
                L.m3gpu_unagg_values(h, i, vals.ctypes.data_as(P(c_f64)))
            meta_fields = []
            for j in range(L.m3gpu_unagg_metadata_count(h, i)):
                fnum, fp, fl = c_i32(), c_vp(), c_i64()
                rc = L.m3gpu_unagg_metadata(
                    h, i, j, ctypes.byref(fnum), ctypes.byref(fp),
                    ctypes.byref(fl))
                _check(rc, "m3gpu_unagg_metadata")
                meta_fields.append(
                    (fnum.value,
                     ctypes.string_at(fp, fl.value) if fl.value else b""))
            out.append(dict(
                metadata_fields=meta_fields,
                type=UA_TYPES.get(ut.value, ut.value),
                metric_type=mt.value,
                id=ctypes.string_at(idp, idl.value) if idl.value else b"",
                values=vals,
                counter_value=cv.value,
                time_nanos=tn.value,
                annotation=ctypes.string_at(anp, anl.value) if anl.value else b"",
                metadatas=ctypes.string_at(mdp, mdl.value) if mdl.value else b"",
            ))
        return out
    finally:
        L.m3gpu_unagg_close(h)


def commitlog_read_dir(dirpath):
    """Read every commit log file in a directory in log-index order and
    merge series ACROSS files by (namespace, id) — unique_index is only
    unique within one file (writer.go seen-bitset is per file). Returns
    the same series-dict list as CommitLog.series(), with points in
    (file index, log order)."""
    import glob as _glob
    files = []
    for p in sorted(_glob.glob(os.path.join(str(dirpath), "commitlog-*.db"))):
        cl = CommitLog(p)
        files.append((cl.index, p, cl))
    files.sort(key=lambda t: t[0])
    merged = {}
    order = []
    for _, _, cl in files:
        try:
            for m in cl.series():
                key = (m["namespace"], m["id"])
                if key not in merged:
                    merged[key] = m
                    order.append(key)
                else:
                    g = merged[key]
                    g["ts"] = np.concatenate([g["ts"], m["ts"]])
                    g["vals"] = np.concatenate([g["vals"], m["vals"]])
                    g["units"] = np.concatenate([g["units"], m["units"]])
                    base = len(g["annotations"])
                    g["annotations"].extend(
                        (base + pi, b) for pi, b in m["annotations"])
        finally:
            cl.close()
    return [merged[k] for k in order]


def commitlog_bootstrap_dir_dev(torch, dirpath, int_optimized=True,
                                device="cuda:0"):
    """Bootstrap from a DIRECTORY of commit logs (the reference commitlog
    bootstrapper's whole-volume job): merge per series across files, then
    batch-encode blocks with the HIP encoder."""
    meta = commitlog_read_dir(dirpath)
    if not meta:
        return [], None, None, None
    n = len(meta)
    width = max(len(m["ts"]) for m in meta)
    ts = np.zeros((n, width), np.int64)
    vals = np.zeros((n, width), np.float64)
    counts = np.zeros(n, np.int32)
    for i, m in enumerate(meta):
        k = len(m["ts"])
        ts[i, :k] = m["ts"]
        vals[i, :k] = m["vals"]
        counts[i] = k
    d_ts = torch.from_numpy(ts).to(device)
    d_vals = torch.from_numpy(vals).to(device)
    d_counts = torch.from_numpy(counts).to(device)
    out_stride = (24 * width + 32 + 15) & ~15
    d_bytes = torch.zeros((n, out_stride), dtype=torch.uint8, device=device)
    d_lens = torch.empty(n, dtype=torch.int32, device=device)
    d_errs = torch.empty(n, dtype=torch.int32, device=device)
    encode_batch_dev(d_ts, d_vals, d_counts, d_bytes, d_lens, d_errs,
                     int_optimized=int_optimized)
    return meta, d_bytes, d_lens, d_errs
