/*
 * m3tsz_oracle.h — CPU oracle for the M3TSZ block codec.
 *
 * TEST INFRASTRUCTURE ONLY. This oracle is a line-faithful C restatement of
 * the reference Go implementation (m3db/m3) of the M3TSZ codec, used solely
 * as the parity checker and CPU baseline for the MI355X HIP implementation.
 * Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * call into this library. The product path (m3_amd + libm3gpu.so) must never
 * route through it.
 *
 * Reference behavior restated from (file:line cited per function in the .c):
 *   src/dbnode/encoding/m3tsz/{m3tsz,encoder,iterator,timestamp_encoder,
 *     timestamp_iterator,float_encoder_iterator,int_sig_bits_tracker}.go
 *   src/dbnode/encoding/{istream,ostream,scheme,encoding}.go
 *   src/dbnode/x/xio/reader64.go
 *   src/x/time/unit.go, src/x/time/time.go
 *
 * Parity pinned by the reference's own golden vectors (tests/golden/):
 *   encoder_test.go:54-393, iterator_test.go:44-412,
 *   encoder_benchmark_test.go:36-47 (10 production streams).
 */
#ifndef M3TSZ_ORACLE_H
#define M3TSZ_ORACLE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* xtime.Unit values (src/x/time/unit.go:30-42) */
enum {
    M3_UNIT_NONE = 0,
    M3_UNIT_SECOND = 1,
    M3_UNIT_MILLISECOND = 2,
    M3_UNIT_MICROSECOND = 3,
    M3_UNIT_NANOSECOND = 4,
    M3_UNIT_MINUTE = 5,
    M3_UNIT_HOUR = 6,
    M3_UNIT_DAY = 7,
    M3_UNIT_YEAR = 8,
};

/* Error codes (negative returns) */
enum {
    M3_ERR_EOF = 1,               /* io.EOF */
    M3_ERR_DOD_OVERFLOW = 2,      /* deltaOfDelta overflows 32 bits (timestamp_encoder.go:216-221) */
    M3_ERR_NO_SCHEME = 3,         /* errNoTimeSchemaForUnit */
    M3_ERR_INVALID_MULT = 4,      /* errInvalidMultiplier */
    M3_ERR_ANNOTATION = 5,        /* bad annotation length / too few bytes */
    M3_ERR_CAPACITY = 6,          /* caller-provided buffer too small */
};

/* -------- single-series full codec (golden-vector surface) -------- */

/* Encode one series. Returns final stream length in bytes (head+EOS tail), or
 * -err. ann_offsets: per-point annotation byte ranges into ann_bytes
 * (ann_offsets[i]..ann_offsets[i+1]); NULL => no annotations. units: per-point
 * time unit; NULL => all `default_unit`. */
int64_t oracle_encode_series(
    const int64_t* ts_ns, const double* vals, const uint8_t* units,
    const int32_t* ann_offsets, const uint8_t* ann_bytes,
    int32_t npts, int64_t start_ns, int int_optimized,
    uint8_t* out, int64_t out_cap);

/* Same but returns the raw (un-finalized) ostream buffer and bit pos,
 * mirroring encoder.os.RawBytes() — used to pin the encoder_test vectors
 * that check the raw buffer. out_pos receives pos (bits used in last byte). */
int64_t oracle_encode_series_raw(
    const int64_t* ts_ns, const double* vals, const uint8_t* units,
    const int32_t* ann_offsets, const uint8_t* ann_bytes,
    int32_t npts, int64_t start_ns, int int_optimized,
    uint8_t* out, int64_t out_cap, int32_t* out_pos);

/* Decode one stream. Returns number of points, or -err.
 * out_units may be NULL. out_ann_lens (per point, 0 = no annotation on that
 * point) and out_ann_bytes (concatenated) may be NULL. */
int64_t oracle_decode_series(
    const uint8_t* data, int64_t len, int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint8_t* out_units,
    int32_t* out_ann_lens, uint8_t* out_ann_bytes, int64_t ann_cap,
    int64_t cap);

/* -------- batch surface (threaded; test-data generation + cpu_baseline) ---- */

/* Encode nseries series of counts[i] points each from SoA rows
 * (ts_ns[i*stride + j], vals[i*stride + j]), all with the same fixed unit and
 * no annotations; start time of series i = ts_ns[i*stride] (start == first
 * datapoint timestamp, as dbnode buffers do). Output: blobs at
 * out_bytes + i*out_stride, lengths in out_lens. Returns 0 or -err. */
int oracle_encode_batch(
    const int64_t* ts_ns, const double* vals, const uint32_t* counts,
    int64_t nseries, int64_t stride, int int_optimized, uint8_t unit,
    uint8_t* out_bytes, int64_t out_stride, uint32_t* out_lens, int nthreads);

/* Decode nseries streams at blobs+offsets[i] (len = offsets[i+1]-offsets[i])
 * into SoA rows of stride `stride`. Returns 0 or -err (first error wins). */
int oracle_decode_batch(
    const uint8_t* blobs, const uint64_t* offsets, int64_t nseries,
    int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts, int64_t stride,
    int nthreads);

/* xxhash64 (annotation dedupe, timestamp_encoder.go:164-170; algorithm is the
 * public XXH64, seed 0 — reference depends on github.com/cespare/xxhash/v2
 * v2.1.2 which is not vendored; reimplemented from the published algorithm). */
uint64_t oracle_xxhash64(const uint8_t* data, size_t len);

#ifdef __cplusplus
}
#endif
#endif
