/*
 * m3gpu.h — C ABI of the MI355X-native M3TSZ + rollup engine (libm3gpu.so).
 *
 * This is the drop-in boundary for the reference's pluggable codec/rollup
 * hot path. On the Go side the reference injects codec constructors into
 * pools (src/dbnode/storage/options.go:498-510, dbnode/server/server.go:
 * 1786-1793, dbnode/client/options.go:504, query/pools/query_pools.go:208)
 * and creates per-window aggregations via typeSpecificElemBase.NewAggregation
 * (aggregator/aggregator/generic_elem.go:94-95). A cgo wrapper implementing
 * encoding.Encoder / encoding.ReaderIterator (dbnode/encoding/types.go:39-96,
 * :197-203) and the aggregation batch path binds to the entry points below —
 * see INTEGRATION.md for the cgo stubs a maintainer would add.
 *
 * Conventions:
 *  - All functions return 0 on success, a negative M3GPU_ERR_* on failure;
 *    m3gpu_last_error() gives a thread-local message.
 *  - `_dev` entry points take DEVICE pointers and an optional hipStream_t
 *    (as void*); they enqueue async work (caller synchronizes the stream).
 *    Host-pointer convenience forms (no suffix) do alloc+H2D+kernel+D2H
 *    internally and block — these are what a cgo caller uses directly.
 *  - Encoded streams are the reference's on-disk M3TSZ block format
 *    (src/dbnode/encoding/m3tsz, incl. the EOS-marker tail), bit-exact.
 *  - Packed stream layout: blobs[offsets[i] .. offsets[i]+lens[i]) is series
 *    i's stream. The kernels REQUIRE offsets[i] to be 8-byte aligned and
 *    every stream zero-padded to an 8-byte boundary (refills are aligned
 *    u64 loads; the zero pad reproduces reader64's zero-filled tail word).
 *    pack_streams and the fileset/regather packers emit 64-byte
 *    alignment (one HBM line per stream chunk), which satisfies this.
 *    lens[] are true stream lengths.
 *  - Time units use the reference's xtime.Unit byte values
 *    (src/x/time/unit.go:30-42): 1=s, 2=ms, 3=us, 4=ns.
 */
#ifndef M3GPU_H
#define M3GPU_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum {
    M3GPU_OK = 0,
    M3GPU_ERR_HIP = -1,        /* HIP runtime failure (see m3gpu_last_error) */
    M3GPU_ERR_BADARG = -2,     /* invalid argument */
    /* per-series error codes written to out_errs[i] (positive) */
    M3GPU_SERIES_OK = 0,
    M3GPU_SERIES_EOF = 1,          /* truncated stream */
    M3GPU_SERIES_DOD_OVERFLOW = 2, /* timestamp_encoder.go:216-221 */
    M3GPU_SERIES_NO_SCHEME = 3,    /* errNoTimeSchemaForUnit */
    M3GPU_SERIES_INVALID_MULT = 4, /* errInvalidMultiplier */
    M3GPU_SERIES_ANNOTATION = 5,   /* bad annotation */
    M3GPU_SERIES_CAPACITY = 6,     /* stride/out buffer too small */
    M3GPU_SERIES_UNSORTED = 7,     /* rollup input crosses buckets backwards */
    M3GPU_SERIES_BUCKET_OVERFLOW = 8, /* rollup bucket beyond CKMS capacity
                                         (sample list > 3072; unreachable for
                                         the reference-default eps/cadence) */
};

/* metric types for the rollup entry (aggregator/aggregation) */
enum { M3GPU_METRIC_COUNTER = 0, M3GPU_METRIC_GAUGE = 1, M3GPU_METRIC_TIMER = 2 };

/* aggregation type ids — the reference's metrics/aggregation Type enum
 * (src/metrics/aggregation/type.go:31-56) */
enum {
    M3GPU_AGG_LAST = 1, M3GPU_AGG_MIN = 2, M3GPU_AGG_MAX = 3, M3GPU_AGG_MEAN = 4,
    M3GPU_AGG_MEDIAN = 5, M3GPU_AGG_COUNT = 6, M3GPU_AGG_SUM = 7,
    M3GPU_AGG_SUMSQ = 8, M3GPU_AGG_STDEV = 9,
    M3GPU_AGG_P10 = 10, M3GPU_AGG_P20 = 11, M3GPU_AGG_P30 = 12, M3GPU_AGG_P40 = 13,
    M3GPU_AGG_P50 = 14, M3GPU_AGG_P60 = 15, M3GPU_AGG_P70 = 16, M3GPU_AGG_P80 = 17,
    M3GPU_AGG_P90 = 18, M3GPU_AGG_P95 = 19, M3GPU_AGG_P99 = 20,
    M3GPU_AGG_P999 = 21, M3GPU_AGG_P9999 = 22, M3GPU_AGG_P25 = 23, M3GPU_AGG_P75 = 24,
};

int m3gpu_init(int device);
void m3gpu_shutdown(void);
const char* m3gpu_last_error(void);

/* -------- batched decode (replaces m3tsz ReaderIterator bulk reads:
 * src/dbnode/encoding/m3tsz/iterator.go:81-219 over xio.Reader64) --------
 * One series per wavefront. Outputs SoA rows: series i writes
 * out_ts[i*stride .. i*stride+count) and out_vals likewise;
 * out_counts[i] = decoded points; out_errs[i] = M3GPU_SERIES_*. */
int m3gpu_decode_batch_dev(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t stride, void* hip_stream);

/* Scheduling variant: d_perm (device, nseries int32) permutes which stream
 * each lane decodes (outputs still land at the stream's own row). Passing a
 * length-sorted order gives every wavefront 64 similar-cost streams. */
int m3gpu_decode_batch_dev_perm(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    const int32_t* d_perm,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t stride, void* hip_stream);

/* Annotation-capturing variant (materializes Current()'s third return,
 * iterator.go:226-231 / timestamp_iterator.go:327-356): d_out_ann is a
 * per-series region of ann_stride bytes (4-aligned, >= 16) laid out as
 *   [u32 n_events][n_events x {u32 point, u32 off, u32 len}][...bytes]
 * where `point` is the 0-based index of the first datapoint the
 * annotation applies to (sticky until replaced), and off/len locate the
 * bytes within the region (they grow from the tail). A region too small
 * for a series' annotations flags M3GPU_SERIES_CAPACITY on that series. */
int m3gpu_decode_batch_dev_ann(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t stride,
    uint8_t* d_out_ann, uint32_t ann_stride, void* hip_stream);

/* host-pointer convenience form of the annotation-capturing decode */
int m3gpu_decode_batch_ann(
    const uint8_t* blobs, uint64_t blobs_len,
    const uint64_t* offsets, const uint32_t* lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts,
    int32_t* out_errs, uint32_t stride,
    uint8_t* out_ann, uint32_t ann_stride);

int m3gpu_decode_batch(
    const uint8_t* blobs, uint64_t blobs_len,
    const uint64_t* offsets, const uint32_t* lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts,
    int32_t* out_errs, uint32_t stride);

/* -------- batched encode (replaces m3tsz.Encoder bulk writes:
 * src/dbnode/encoding/m3tsz/encoder.go:89-250) --------
 * Series i encodes counts[i] points from ts/vals rows (stride elements per
 * row); start time = first timestamp (as dbnode series buffers do). Output:
 * stream bytes at d_out_bytes + i*out_stride (finalized, EOS tail included),
 * length in out_lens[i]. out_stride must be a multiple of 8 and hold the
 * worst case (~20 B/pt + 16). Fixed unit, no annotations (bulk path). */
int m3gpu_encode_batch_dev(
    const int64_t* d_ts, const double* d_vals, const uint32_t* d_counts,
    uint32_t nseries, uint32_t stride, int int_optimized, uint8_t unit,
    uint8_t* d_out_bytes, uint32_t out_stride, uint32_t* d_out_lens,
    int32_t* d_out_errs, void* hip_stream);

int m3gpu_encode_batch(
    const int64_t* ts, const double* vals, const uint32_t* counts,
    uint32_t nseries, uint32_t stride, int int_optimized, uint8_t unit,
    uint8_t* out_bytes, uint32_t out_stride, uint32_t* out_lens,
    int32_t* out_errs);

/* Pack strided encoder output into the tight 8B-aligned blob layout:
 * series i's first lens[i] bytes (rounded up to whole 8B words, whose pad
 * bytes the encoder zeroes) move from d_src + i*src_stride to
 * d_dst + d_dst_offsets[i]. d_dst must be pre-zeroed. */
int m3gpu_compact_dev(
    const uint8_t* d_src, uint32_t src_stride, const uint32_t* d_lens,
    const uint64_t* d_dst_offsets, uint32_t nseries, uint8_t* d_dst,
    void* hip_stream);

/* -------- fused decode -> windowed rollup (replaces decode +
 * elem.AddValue/Consume: aggregator/aggregator/generic_elem.go:219-235,
 * 424-485 with aggregation/{counter,gauge,timer}.go semantics) --------
 * Decodes each stream and aggregates into ceil-aligned windows of window_ns:
 * bucket b covers [base + b*window, base + (b+1)*window) where base =
 * truncate(first timestamp). Emits naggs doubles per bucket in agg_types[]
 * order (d_out[i*nbuckets*naggs + b*naggs + k]) and the window-END timestamp
 * (list.go:541-543) in d_out_window_ts[i*nbuckets + b]. Quantiles use the
 * production CKMS semantics (exact for <=64 values per bucket; more sets
 * M3GPU_SERIES_BUCKET_OVERFLOW). agg_types is a HOST pointer (naggs <= 16). */
int m3gpu_rollup_batch_dev(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int metric_type, int64_t window_ns, uint32_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* d_out, int64_t* d_out_window_ts, int32_t* d_out_errs,
    void* hip_stream);

/* As above with explicit CKMS stream options (quantile/cm/options.go:30-32;
 * defaults eps=1e-3, insert_and_compress_every=1024). eps in (0, 0.5),
 * every in [1, 1024] (deep-tier buffers are sized for a 2*1024 peak). */
int m3gpu_rollup_batch_dev_opts(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int metric_type, int64_t window_ns, uint32_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* d_out, int64_t* d_out_window_ts, int32_t* d_out_errs,
    void* hip_stream, double eps, int insert_and_compress_every);

/* -------- replica-deduplicating merge (replaces MultiReaderIterator over
 * one slice of R replica iterators: dbnode/encoding/multi_reader_iterator.go
 * :62-155 + iterators.go:56-237, IterateLastPushed default). Decoded replica
 * rows (replica-major: row r*nseries+i) merge into one row per series with
 * equal timestamps deduped (last in values-order wins) and decreasing
 * timestamps flagged (err 100 = errOutOfOrderIterator). R <= 4. -------- */
int m3gpu_merge_batch_dev(
    const int64_t* d_ts, const double* d_vals, const uint32_t* d_counts,
    uint32_t nreplicas, uint32_t nseries, uint32_t stride,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t out_stride, void* hip_stream);

int m3gpu_rollup_batch(
    const uint8_t* blobs, uint64_t blobs_len,
    const uint64_t* offsets, const uint32_t* lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int metric_type, int64_t window_ns, uint32_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* out, int64_t* out_window_ts, int32_t* out_errs);

/* Layout pass: physically reorder packed streams so dst series i holds
 * src series perm[i] (dst_offsets precomputed from lens[perm], 16B
 * aligned). Lays a batch out in wave-schedule order for L2 locality. */
int m3gpu_regather_dev(
    const uint8_t* d_src, const uint64_t* d_src_offsets,
    const uint32_t* d_lens, const int32_t* d_perm,
    const uint64_t* d_dst_offsets, uint32_t nseries, uint8_t* d_dst,
    void* hip_stream);

/* Largest bucket size for which the CKMS stream with the given eps
 * provably never merges for this agg set (quantiles extracted, sorted
 * unique): up to this depth the wave-tier rollup's quantiles are exact
 * order statistics. For tests and host-side sizing. */
int m3gpu_ckms_exact_cap(const int32_t* agg_types, int naggs, double eps);

/* ======================= fileset volume reader =======================
 * Native reader for the reference's dbnode fileset volumes (persist/fs
 * read.go:145-457 + msgpack/decoder.go + digest): open a volume from a
 * shard directory (reference naming fileset-<blockStartNs>-<volume>-
 * <suffix>.db, legacy no-volume names for volume 0), validate the
 * checkpoint, all five file digests, every index-entry checksum (V3) and
 * every data-block checksum, then expose the entries sorted by data
 * offset ascending and repack the blocks into the decode-batch blob
 * layout above. Pure host code (usable without a GPU). */

/* Returns a handle >= 0, or a negative M3GPU_FS_ERR_* code. */
int m3gpu_fileset_open(const char* shard_dir, int64_t block_start_ns,
                       int volume_index);
int m3gpu_fileset_close(int handle);
const char* m3gpu_fileset_last_error(void);
int m3gpu_fileset_info(int handle, int64_t* block_start, int64_t* block_size,
                       int64_t* entries, int64_t* major_version,
                       int64_t* minor_version, int* volume_index,
                       int64_t* bloom_m, int64_t* bloom_k,
                       int64_t* summaries);
int m3gpu_fileset_entry(int handle, int64_t i, int64_t* size, int64_t* offset,
                        int64_t* data_checksum, const uint8_t** id,
                        int64_t* id_len, const uint8_t** tags,
                        int64_t* tags_len);
int64_t m3gpu_fileset_packed_size(int handle);
int m3gpu_fileset_pack(int handle, uint8_t* blob, uint64_t blob_cap,
                       uint64_t* offsets, uint32_t* lens);

enum {
    M3GPU_FS_ERR_IO = -101,
    M3GPU_FS_ERR_CHECKPOINT = -102,
    M3GPU_FS_ERR_DIGEST = -103,
    M3GPU_FS_ERR_MSGPACK = -104,
    M3GPU_FS_ERR_SCHEMA = -105,
    M3GPU_FS_ERR_ENTRY_CHECKSUM = -106,
    M3GPU_FS_ERR_DATA_CHECKSUM = -107,
    M3GPU_FS_ERR_BOUNDS = -108,
    M3GPU_FS_ERR_BADHANDLE = -109,
    M3GPU_FS_ERR_CAPACITY = -110,
};

/* ======================== commit log reader =========================
 * Native reader for the reference's commit log files (persist/fs/
 * commitlog: 12-byte chunk headers with adler32 size/data checksums,
 * uvarint-framed msgpack LogInfo/LogEntry records, nested LogMetadata on
 * each series' first entry). Series come back in first-seen order with
 * datapoints in log order — the bootstrap-from-commitlog input to the GPU
 * batch encoder. Pure host code. */
int m3gpu_commitlog_open(const char* path); /* handle >= 0 or error */
int m3gpu_commitlog_close(int handle);
const char* m3gpu_commitlog_last_error(void);
int m3gpu_commitlog_info(int handle, int64_t* index, int64_t* num_entries,
                         int64_t* num_series);
int m3gpu_commitlog_series(int handle, int64_t i, uint64_t* unique_index,
                           const uint8_t** id, int64_t* id_len,
                           const uint8_t** ns, int64_t* ns_len,
                           uint32_t* shard, const uint8_t** tags,
                           int64_t* tags_len, int64_t* num_points,
                           int64_t* num_annotations);
int m3gpu_commitlog_series_points(int handle, int64_t i, int64_t* ts,
                                  double* vals, uint8_t* units);
int m3gpu_commitlog_series_annotation(int handle, int64_t i, int64_t j,
                                      int64_t* point_index,
                                      const uint8_t** bytes, int64_t* len);

enum {
    M3GPU_CL_ERR_CHUNK_CHECKSUM = -111,
    M3GPU_CL_ERR_MISSING_METADATA = -112,
    M3GPU_CL_ERR_TRUNCATED = -113,
};

/* ================= unaggregated metric wire parser =================
 * Batch parser for the m3aggregator ingest wire format
 * (metrics/encoding/protobuf): zigzag-varint size-prefixed metricpb
 * MetricWithMetadatas messages. Untimed counter/batch-timer/gauge and
 * timed unions are parsed; metadatas/policies kept as opaque bytes.
 * Pure host code; grouped values feed the GPU encode/rollup path. */
int m3gpu_unagg_parse(const uint8_t* buf, uint64_t len); /* handle or <0 */
int m3gpu_unagg_close(int handle);
const char* m3gpu_unagg_last_error(void);
int64_t m3gpu_unagg_count(int handle);
int m3gpu_unagg_metric(int handle, int64_t i, int32_t* union_type,
                       int32_t* metric_type, const uint8_t** id,
                       int64_t* id_len, int64_t* num_values,
                       int64_t* counter_value, int64_t* time_nanos,
                       const uint8_t** annotation, int64_t* annotation_len,
                       const uint8_t** metadatas, int64_t* metadatas_len);
int m3gpu_unagg_values(int handle, int64_t i, double* out);
/* Field-separated wrapper metadata (StagedMetadatas field 2, StoragePolicy
 * field 3, ...): `metadatas` above is a concatenation and cannot be split
 * back into the individual reference protos. */
int64_t m3gpu_unagg_metadata_count(int handle, int64_t i);
int m3gpu_unagg_metadata(int handle, int64_t i, int64_t j, int32_t* field,
                         const uint8_t** bytes, int64_t* len);

enum {
    M3GPU_UA_ERR_TRUNCATED = -121,
    M3GPU_UA_ERR_PROTO = -122,
    M3GPU_UA_ERR_TYPE = -123,
    M3GPU_UA_ERR_BADHANDLE = -124,
    M3GPU_UA_ERR_SIZE = -125,
};

#ifdef __cplusplus
}
#endif
#endif
