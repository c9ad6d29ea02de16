"""m3_amd — MI355X-native M3TSZ block codec + windowed rollup engine.

The product path of this framework: batched M3TSZ encode/decode and the
fused decode->downsample rollup, executed by hand-written HIP/CDNA4 kernels
(libm3gpu.so, one series per wavefront) behind the C ABI in include/m3gpu.h.

This package is the host-side mirror of the reference's pluggable hot-path
interfaces (dbnode encoding.Encoder/ReaderIterator bulk path + m3aggregator
elem rollup semantics) for Python callers; the Go-facing cgo surface binds
the same C ABI directly.

Fails loudly if the HIP engine is missing: there is NO CPU fallback here —
the oracle/ package is test infrastructure only and is never imported by
this product path.
"""
from .iterators import (  # noqa: F401
    BatchIterators,
    SliceReaderIterator,
)
from .engine import (  # noqa: F401
    CL_ERRORS,
    CommitLog,
    commitlog_bootstrap_dev,
    commitlog_bootstrap_dir_dev,
    commitlog_read_dir,
    FS_ERRORS,
    FilesetVolume,
    fileset_ingest_dev,
    M3GPU_AGG,
    METRIC_COUNTER,
    METRIC_GAUGE,
    METRIC_TIMER,
    SERIES_ERRORS,
    M3GpuError,
    compact_dev,
    decode_batch,
    decode_batch_dev,
    encode_batch,
    encode_batch_dev,
    engine_available,
    lib,
    merge_batch_dev,
    pack_streams,
    parse_unaggregated,
    regather_dev,
    aggregate_tiles_dev,
    rollup_batch,
    rollup_batch_dev,
)
