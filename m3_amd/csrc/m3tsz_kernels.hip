/*
 * m3tsz_kernels.hip — MI355X (gfx950/CDNA4) native M3TSZ block codec and
 * fused decode->rollup engine, plus the C-ABI host layer (include/m3gpu.h).
 *
 * This is a from-scratch GPU design of the reference's hot path (reference
 * cites per function; format: src/dbnode/encoding/m3tsz + aggregator
 * semantics), NOT a port of its Go code:
 *
 *  - One SERIES PER WAVEFRONT. M3TSZ is a sequentially-dependent
 *    variable-length code, so all 64 lanes run the (wave-uniform) bit parser
 *    redundantly — branches cost s_branch, not divergence — and the wave's
 *    parallelism is spent on the memory system:
 *      decode: lane (n & 63) captures point n in registers; every 64 points
 *        the wave stores 64x(8B ts + 8B val) contiguously — fully coalesced
 *        512B transactions. The compressed stream is read as aligned
 *        big-endian u64 words at one wave-uniform address (L1 broadcast).
 *      encode: lane (w & 63) captures output word w; every 64 words the wave
 *        stores 512B coalesced.
 *      rollup: bucket values are staged to LDS, quantiles picked by a
 *        rank-select over lanes (no sort network needed), reproducing the
 *        reference CKMS walk exactly (see ckms_small_n semantics below).
 *  - Streams are decoded from 8-byte-aligned offsets, zero-padded to an
 *    8-byte boundary (layout contract in m3gpu.h; pack_streams emits 16B
 *    alignment, which satisfies it) so every refill is an aligned u64 load.
 *  - Wave64 only; no CUDA shims, no hipified code.
 *
 * Compile: hipcc --offload-arch=gfx950 -O3 -ffp-contract=off (bit-exact f64:
 * convertToIntFloat m3tsz.go:78-119, decode accumulation iterator.go:168-175).
 */
#include <hip/hip_runtime.h>
#include <atomic>
#include <math.h>
#include <string.h>
#include <stdio.h>
#include "../../include/m3gpu.h"

/* ========================= shared constants ========================= */
/* m3tsz.go:28-62, scheme.go:28-52, x/time/unit.go:30-42 */

#define WAVE 64
#define WAVES_PER_BLOCK 4
#define BLOCK_THREADS (WAVE * WAVES_PER_BLOCK)

namespace m3 {

__device__ __constant__ int64_t UNIT_NS_D[9] = {
    0, 1000000000LL, 1000000LL, 1000LL, 1LL,
    60000000000LL, 3600000000000LL, 86400000000000LL, 31536000000000000LL};

#define MARKER_OPCODE 0x100ULL
#define MARKER_BITS 11 /* 9 opcode + 2 value */
#define MARKER_EOS 0
#define MARKER_ANNOTATION 1
#define MARKER_TIMEUNIT 2

#define MAX_MULT 6
#define NUM_SIG_BITS 6
#define NUM_MULT_BITS 3
#define SIG_DIFF_THRESHOLD 3
#define SIG_REPEAT_THRESHOLD 5

__device__ __forceinline__ int unit_valid(uint8_t u) { return u > 0 && u < 9; }

/* scheme.go:42-52: {7,9,12} bucket value bits; default 32 (s/ms) / 64 (us/ns) */
__device__ __forceinline__ int scheme_default_bits(uint8_t unit) {
    if (unit == 1 || unit == 2) return 32;
    if (unit == 3 || unit == 4) return 64;
    return 0;
}

__device__ __forceinline__ uint8_t num_sig(uint64_t v) { /* encoding.go:29-31 */
    return (uint8_t)(64 - (v ? __builtin_clzll(v) : 64));
}
__device__ __forceinline__ int64_t sign_extend(uint64_t v, uint32_t nbits) {
    uint32_t sh = 64 - nbits;
    return ((int64_t)(v << sh)) >> sh;
}
/* 10^m for m in [0,6] as a pure VALU select chain (see exp10_table note:
 * an indexed const array would cost a global load + vmcnt(0) per use) */
__device__ __forceinline__ double exp10_sel(uint8_t m) {
    double r = 1.0;
    r = (m == 1) ? 10.0 : r;
    r = (m == 2) ? 100.0 : r;
    r = (m == 3) ? 1000.0 : r;
    r = (m == 4) ? 10000.0 : r;
    r = (m == 5) ? 100000.0 : r;
    r = (m == 6) ? 1000000.0 : r;
    return r;
}

/* Go float64->int64 (amd64 CVTTSD2SQ): out-of-range/NaN -> INT64_MIN */
__device__ __forceinline__ int64_t go_f2i(double v) {
    if (!(v >= -9223372036854775808.0 && v < 9223372036854775808.0)) return INT64_MIN;
    return (int64_t)v;
}
__device__ __forceinline__ uint64_t f2bits(double v) { return __double_as_longlong(v); }
__device__ __forceinline__ double bits2f(uint64_t b) { return __longlong_as_double((long long)b); }

/* Go math.Modf: Modf(+-Inf) = (+-Inf, NaN). The finite path is exact
 * trunc-toward-zero + subtract (v - trunc(v) is exact in IEEE — Sterbenz
 * for |v| >= 1, and trunc(v) == 0 below), one v_trunc_f64 + v_add_f64
 * instead of the libm modf sequence. */
__device__ __forceinline__ double go_modf(double v, double* ip) {
    if (isinf(v)) { *ip = v; return __longlong_as_double(0x7ff8000000000000LL); }
    double i = trunc(v);
    *ip = i;
    return v - i;
}

/* ===================== device bit reader ===================== */
/* istream.go:73-115 over reader64.go:40-80, with the m3gpu.h layout
 * contract: stream starts 8B-aligned, buffer zero-padded to 8B. */

/* Per-lane LDS input ring geometry (decode kernel): 8 buffered words per
 * lane (one 64B chunk = one HBM line), rows padded to 9 so that both the
 * per-lane ds_read_b64 pulls and the refill ds_write_b64 bursts are
 * bank-conflict-free (b64 reads bank on (a/4)%64 in 2x32 lane groups,
 * writes on (a/4)%32 in 4x16 groups; the 9-word row stride makes lane
 * bases land on distinct bank pairs). */
#ifndef IN_WORDS
#define IN_WORDS 4
#endif
#ifndef IN_STRIDE
#define IN_STRIDE 4
#endif

struct BitReader {
    /* 128-bit register lookahead (`a` = next bits, `b` = following word,
     * both byte-swapped to stream bit order; `p` = bits of `a` already
     * consumed): peek64() is branch-free, consume() crosses at most one
     * word per call (every field is <= 64 bits). EOF semantics match
     * istream.go:73-115 over reader64.go:40-80: `bits_left` counts the
     * un-consumed TRUE stream bits, reads past it fail per series; the
     * blob's zero padding reproduces reader64's zero-filled partial tail
     * word, so the window itself can always hold 128 physical bits.
     *
     * Words arrive one of two ways:
     *  - LDS ring (decode kernel, `lds` set): refill() tops the per-lane
     *    ring up to 16 words once per 8-point tile — each lane issues up
     *    to 8 back-to-back loads of CONSECUTIVE words of its own stream,
     *    so the per-lane 64B line is fetched once and consumed fully
     *    (L1-amortized), and the long-latency global gathers leave the
     *    per-point dependency chain entirely: the parser's word pulls are
     *    cheap ds_read_b64s.
     *  - direct (`lds` null, rollup/merge kernels): a 1-deep register
     *    prefetch (`pfw`) beyond the window, i.e. 3 words of lookahead,
     *    keeps the refill load issued ~128 bits before first use. */
    const uint64_t* words;
    uint64_t a, b, c;   /* c: ring-word prefetch — the ds_read latency sits
                         * between one crossing and the NEXT, not in the
                         * peek chain */
    uint32_t p;
    int64_t bits_left;  /* un-consumed stream bits (from the true byte len) */
    uint32_t wnext;     /* next word index to pull into the window */
    uint32_t wtotal;    /* ceil(len/8) */
    uint64_t* lds;      /* this lane's ring base, or nullptr */
    uint32_t rfill;     /* words fetched into the ring so far */
    uint64_t pend[IN_WORDS]; /* deferred refill: loads issued at tile start */
    uint32_t pend_n;
    uint64_t pfw;       /* direct mode: extra prefetched word */

    __device__ __forceinline__ uint64_t next_word() {
        uint64_t w;
        if (lds) {
            /* ring hit is the steady state (refill runs per tile); a
             * mid-tile underrun (adversarially dense streams consuming
             * >16 words in 8 points) falls back to a direct load —
             * correct, just slower */
            if (wnext < rfill) w = lds[wnext & (IN_WORDS - 1)];
            else w = (wnext < wtotal) ? __builtin_bswap64(words[wnext]) : 0;
        } else {
            w = pfw;
            uint32_t nn = wnext + 1;
            pfw = (nn < wtotal) ? __builtin_bswap64(words[nn]) : 0;
        }
        wnext++;
        return w;
    }

    /* Deferred per-tile ring top-up, CHUNK-granular: one refill = the next
     * whole 64B line of this lane's stream (8 consecutive u64 loads issued
     * back-to-back — the first fetches the line, the rest hit L1 within
     * the burst; with 64B-aligned stream packing every line is fetched
     * exactly once, no inter-tile L1 retention needed). refill_issue()
     * starts the loads at TILE START only when the whole ring is free;
     * refill_commit() writes them to the ring at TILE END, after the
     * tile's output stores have issued — gfx9 vmcnt retires in issue
     * order, so the commit's counted wait isolates these loads and never
     * drains the in-flight output stores. The loads' latency hides under
     * the whole tile's parsing. */
#ifndef RF_LOW
#define RF_LOW 3 /* refill when <= this many ring words remain. Topping up
                  * every tile (word-granular bursts, L1-amortized) measured
                  * faster than whole-chunk-only refills (RF_LOW 0): the ring
                  * then never dries mid-tile, so the parser's underrun
                  * fallback (an in-chain global gather whose wait also
                  * drains in-flight output stores) stays off the hot path */
#endif
    __device__ __forceinline__ void refill_issue(bool active) {
        if (!lds) return;
        if (rfill < wnext) rfill = wnext; /* resync: underrun consumed
                                           * [rfill, wnext) directly */
        uint32_t used = rfill - wnext;
        pend_n = 0;
        if (active && used <= RF_LOW && rfill < wtotal) {
            uint32_t n = wtotal - rfill;
            uint32_t room = IN_WORDS - used;
            if (n > room) n = room;
            pend_n = n;
#pragma unroll
            for (uint32_t i = 0; i < IN_WORDS; i++)
                if (i < pend_n) pend[i] = __builtin_bswap64(words[rfill + i]);
        }
    }
    __device__ __forceinline__ void refill_commit() {
        if (!lds) return;
#pragma unroll
        for (uint32_t i = 0; i < IN_WORDS; i++)
            if (i < pend_n) lds[(rfill + i) & (IN_WORDS - 1)] = pend[i];
        rfill += pend_n;
        pend_n = 0;
    }

    __device__ __forceinline__ void init(const uint8_t* base, uint64_t off,
                                         uint32_t l, uint64_t* ring) {
        words = (const uint64_t*)(base + off);
        wtotal = (l + 7) >> 3;
        bits_left = (int64_t)l * 8;
        lds = ring;
        wnext = 0;
        rfill = 0;
        pend_n = 0;
        if (ring) {
            refill_issue(true);
            refill_commit();
        } else {
            pfw = wtotal ? __builtin_bswap64(words[0]) : 0;
        }
        a = next_word();
        b = next_word();
        c = next_word();
        p = 0;
    }

    /* next 64 physical bits (zero-padded past the stream tail); branch-free */
    __device__ __forceinline__ uint64_t peek64() const {
        return (a << p) | (p ? (b >> (64 - p)) : 0);
    }
    /* advance n <= 64 bits already validated against bits_left */
    __device__ __forceinline__ void consume(uint32_t n) {
        bits_left -= n;
        p += n;
        if (p >= 64) {
            a = b;
            b = c;
            c = next_word();
            p -= 64;
        }
    }
    __device__ __forceinline__ int read_bits(uint32_t n, uint64_t* out) {
        if (bits_left < (int64_t)n) return M3GPU_SERIES_EOF;
        *out = n ? (peek64() >> (64 - n)) : 0;
        consume(n);
        return 0;
    }
    __device__ __forceinline__ int peek_bits(uint32_t n, uint64_t* out) {
        if (bits_left < (int64_t)n) return M3GPU_SERIES_EOF;
        *out = n ? (peek64() >> (64 - n)) : 0;
        return 0;
    }
};

/* ===================== device decoder state ===================== */
/* Port of the reference decode state machine: iterator.go:47-219,
 * timestamp_iterator.go:41-361, float_encoder_iterator.go:105-165. */

struct Decoder {
    BitReader r;
    int64_t prev_time, prev_time_delta;
    int64_t unit_ns; /* cached UNIT_NS_D[time_unit] (0 when unit invalid) */
    double mult_pow;  /* cached 10^mult (mult changes rarely; keeping the
                       * select chain out of the per-point value emit) */
    double int_val;
    uint64_t prev_float_bits, prev_xor;
    uint8_t time_unit, scheme_unit, mult, sig;
    uint8_t default_unit;
    bool have_scheme, tu_changed, done, is_float;
    bool int_optimized;

    __device__ __forceinline__ void set_unit(uint8_t tu) {
        time_unit = tu;
        unit_ns = unit_valid(tu) ? UNIT_NS_D[tu] : 0;
    }

    /* optional annotation capture (m3gpu_decode_batch_dev_ann): events
     * {point, off, len} grow from the front of the per-series region,
     * annotation bytes from the tail; collision flags CAPACITY */
    uint8_t* ann_region;
    uint32_t ann_stride_, ann_events, ann_tail, points;
    int ann_err;

    __device__ __forceinline__ void init(const uint8_t* base, uint64_t off, uint32_t len,
                         bool intopt, uint8_t dunit, uint64_t* ring = nullptr,
                         uint8_t* ann = nullptr, uint32_t ann_stride = 0) {
        r.init(base, off, len, ring);
        prev_time = 0; prev_time_delta = 0;
        int_val = 0; prev_float_bits = 0; prev_xor = 0;
        time_unit = 0; unit_ns = 0; scheme_unit = 0; mult = 0; sig = 0;
        mult_pow = 1.0;
        default_unit = dunit;
        have_scheme = false; tu_changed = false; done = false; is_float = false;
        int_optimized = intopt;
        ann_region = ann;
        ann_stride_ = ann_stride;
        ann_events = 0;
        ann_tail = ann_stride;
        points = 0;
        ann_err = 0;
    }

    /* timestamp_iterator.go:115-135 */
    __device__ __forceinline__ int read_time_unit() {
        uint64_t tu_bits;
        int err = r.read_bits(8, &tu_bits);
        if (err) return err;
        uint8_t tu = (uint8_t)tu_bits;
        if (unit_valid(tu) && tu != time_unit) {
            tu_changed = true;
            if (scheme_default_bits(tu)) { have_scheme = true; scheme_unit = tu; }
        }
        set_unit(tu);
        return 0;
    }

    /* binary.ReadVarint via ReadByte */
    __device__ __forceinline__ int read_varint(int64_t* out) {
        uint64_t ux = 0;
        int shift = 0;
        for (int i = 0; i < 10; i++) {
            uint64_t b;
            int err = r.read_bits(8, &b);
            if (err) return err;
            ux |= (b & 0x7f) << shift;
            if (!(b & 0x80)) {
                int64_t x = (int64_t)(ux >> 1);
                if (ux & 1) x = ~x;
                *out = x;
                return 0;
            }
            shift += 7;
        }
        return M3GPU_SERIES_ANNOTATION;
    }

    /* timestamp_iterator.go:327-356. Without a capture region the
     * annotation bytes are skipped; with one (the _ann decode surface)
     * each annotation-set event is recorded as {point, off, len} so the
     * caller can materialize Current()'s sticky PrevAnt
     * (iterator.go:226-231) by carrying events forward. */
    __device__ __forceinline__ int skip_annotation() {
        int64_t alen;
        int err = read_varint(&alen);
        if (err) return err;
        alen += 1;
        if (alen <= 0) return M3GPU_SERIES_ANNOTATION;
        uint8_t* dst = nullptr;
        if (ann_region && !ann_err) {
            uint32_t ev_end = 4 + (ann_events + 1) * 12;
            if (alen > (int64_t)ann_tail || ev_end > ann_tail - (uint32_t)alen) {
                ann_err = M3GPU_SERIES_CAPACITY; /* region full: flag, keep
                                                  * decoding values */
            } else {
                ann_tail -= (uint32_t)alen;
                dst = ann_region + ann_tail;
                uint32_t* ev = (uint32_t*)(ann_region + 4 + ann_events * 12);
                ev[0] = points; /* index of the point this starts applying to */
                ev[1] = ann_tail;
                ev[2] = (uint32_t)alen;
                ann_events++;
            }
        }
        for (int64_t i = 0; i < alen; i++) {
            uint64_t b;
            err = r.read_bits(8, &b);
            if (err) return err;
            if (dst) dst[i] = (uint8_t)b;
        }
        return 0;
    }
    __device__ __forceinline__ void ann_finish(int* err_io) {
        if (!ann_region) return;
        *(uint32_t*)ann_region = ann_events;
        if (ann_err && *err_io == 0) *err_io = ann_err;
    }

    /* timestamp_iterator.go:307-325 */
    __device__ __forceinline__ int read_full_timestamp(int64_t* dod) {
        if (!scheme_default_bits(time_unit)) return M3GPU_SERIES_NO_SCHEME;
        have_scheme = true;
        scheme_unit = time_unit;
        uint64_t bits;
        int err = r.read_bits(64, &bits);
        if (err) return err;
        *dod = (int64_t)bits;
        return 0;
    }

    /* timestamp_iterator.go:250-305 */
    __device__ __forceinline__ int read_dod(int64_t* out) {
        if (tu_changed) return read_full_timestamp(out);
        if (!have_scheme) return M3GPU_SERIES_NO_SCHEME;
        uint64_t cb;
        int err = r.read_bits(1, &cb);
        if (err) return err;
        if (cb == 0) { *out = 0; return 0; }
        const uint32_t opcodes[3] = {0x2, 0x6, 0xe};
        const uint32_t vbits[3] = {7, 9, 12};
#pragma unroll
        for (int i = 0; i < 3; i++) {
            uint64_t nxt;
            err = r.read_bits(1, &nxt);
            if (err) { *out = 0; return 0; } /* swallowed (:271-274) */
            cb = (cb << 1) | nxt;
            if (cb == opcodes[i]) {
                uint64_t db;
                err = r.read_bits(vbits[i], &db);
                if (err) return err;
                if (!unit_valid(time_unit)) { *out = 0; return 0; }
                *out = sign_extend(db, vbits[i]) * unit_ns;
                return 0;
            }
        }
        uint32_t dbits = (uint32_t)scheme_default_bits(scheme_unit);
        uint64_t db;
        err = r.read_bits(dbits, &db);
        if (err) return err;
        if (!unit_valid(time_unit)) { *out = 0; return 0; }
        *out = sign_extend(db, dbits) * unit_ns;
        return 0;
    }

    /* readMarkerOrDeltaOfDelta (:237-248) with tryReadMarker (:174-235)
     * unrolled into a loop (annotation/timeunit markers chain), and the
     * 11-bit marker peek FUSED with the DoD bucket decode: one peek
     * classifies {dod==0 | marker | bucket} (GPU-relevant consequence (ii)
     * of the grammar). Bit-identical to the sequential reference reads;
     * falls back to the bit-by-bit path near end-of-stream (short peek)
     * and for unit-change/no-scheme states. */
    __device__ __forceinline__ int read_marker_or_dod(int64_t* out) {
        /* dominant path: regular cadence emits dod == 0 = a single 0 bit,
         * which can never be a marker (markers start with 1). */
        if (!tu_changed && have_scheme) {
            uint64_t b1;
            if (r.peek_bits(1, &b1) == 0 && b1 == 0) {
                r.consume(1);
                *out = 0;
                return 0;
            }
        }
        for (;;) {
            uint64_t ov;
            /* markers are checked BEFORE any scheme/unit-change gating
             * (tryReadMarker runs first in the reference) */
            if (r.peek_bits(MARKER_BITS, &ov) != 0) return read_dod(out);
            bool fast = !tu_changed && have_scheme;
            if (fast && (ov >> 10) == 0) { /* zero bucket: 1-cadence fast path */
                r.consume(1);
                *out = 0;
                return 0;
            }
            if ((ov >> 2) == MARKER_OPCODE) {
                uint64_t marker = ov & 0x3;
                int err;
                if (marker == MARKER_EOS) {
                    r.consume(MARKER_BITS);
                    done = true;
                    *out = 0;
                    return 0;
                } else if (marker == MARKER_ANNOTATION) {
                    r.consume(MARKER_BITS);
                    err = skip_annotation();
                    if (err) return err;
                    continue;
                } else if (marker == MARKER_TIMEUNIT) {
                    r.consume(MARKER_BITS);
                    err = read_time_unit();
                    if (err) return err;
                    continue;
                }
                /* unknown marker value: falls through and parses as a
                 * bucket (:232-234) — the 10... prefix selects bucket 0,
                 * exactly as the sequential reads would. */
            }
            if (!fast) return read_dod(out); /* unit change / no scheme */
            /* bucket select: count leading ones of the top 4 bits */
            uint32_t top4 = (uint32_t)(ov >> (MARKER_BITS - 4)) & 0xF;
            uint32_t L = __builtin_clz(~(top4 << 28)); /* in [1,4] */
            uint32_t ob = (L <= 3) ? L + 1 : 4;        /* opcode bits */
            uint32_t vb = (L == 1) ? 7 : (L == 2) ? 9 : (L == 3) ? 12
                          : (uint32_t)scheme_default_bits(scheme_unit);
            r.consume(ob);
            uint64_t db;
            int err = r.read_bits(vb, &db);
            if (err) return err;
            if (!unit_valid(time_unit)) { *out = 0; return 0; } /* swallowed */
            *out = sign_extend(db, vb) * unit_ns;
            return 0;
        }
    }

    /* timestamp_iterator.go:137-161 + initialTimeUnit */
    __device__ __forceinline__ int read_first_timestamp() {
        uint64_t nt_bits;
        int err = r.read_bits(64, &nt_bits);
        if (err) return err;
        int64_t nt = (int64_t)nt_bits;
        if (time_unit == 0 && unit_valid(default_unit) &&
            nt % UNIT_NS_D[default_unit] == 0) {
            set_unit(default_unit);
        }
        if (scheme_default_bits(time_unit)) { have_scheme = true; scheme_unit = time_unit; }
        int64_t dod;
        err = read_marker_or_dod(&dod);
        if (err) return err;
        if (!done) prev_time_delta += dod;
        prev_time = nt + prev_time_delta;
        return 0;
    }

    /* timestamp_iterator.go:80-113 */
    __device__ __forceinline__ int read_timestamp(bool* first) {
        *first = false;
        int err;
        if (prev_time != 0) {
            int64_t dod;
            err = read_marker_or_dod(&dod);
            if (err == 0 && !done) {
                prev_time_delta += dod;
                prev_time += prev_time_delta;
            }
        } else {
            *first = true;
            err = read_first_timestamp();
        }
        if (err) return err;
        if (tu_changed) { prev_time_delta = 0; tu_changed = false; }
        return 0;
    }

    /* float_encoder_iterator.go:105-165 */
    __device__ __forceinline__ int read_full_float() {
        uint64_t vb;
        int err = r.read_bits(64, &vb);
        if (err) return err;
        prev_float_bits = vb;
        prev_xor = vb;
        return 0;
    }
    __device__ __forceinline__ int read_next_float() {
        uint64_t cb;
        int err = r.read_bits(1, &cb);
        if (err) return err;
        if (cb == 0) { prev_xor = 0; return 0; }
        uint64_t nxt;
        err = r.read_bits(1, &nxt);
        if (err) return err;
        cb = (cb << 1) | nxt;
        if (cb == 0x2) { /* contained */
            uint32_t lead = prev_xor ? __builtin_clzll(prev_xor) : 64;
            uint32_t trail = prev_xor ? __builtin_ctzll(prev_xor) : 0;
            uint32_t nmean = 64 - lead - trail;
            uint64_t mb;
            err = r.read_bits(nmean, &mb);
            if (err) return err;
            prev_xor = mb << trail;
            prev_float_bits ^= prev_xor;
            return 0;
        }
        uint64_t lm;
        err = r.read_bits(12, &lm);
        if (err) return err;
        uint64_t lead = (lm & 4032) >> 6;
        uint64_t nmean = (lm & 63) + 1;
        uint64_t mb;
        err = r.read_bits((uint32_t)nmean, &mb);
        if (err) return err;
        uint64_t trail = 64 - lead - nmean;
        prev_xor = mb << trail;
        prev_float_bits ^= prev_xor;
        return 0;
    }

    /* iterator.go:178-219 */
    __device__ __forceinline__ int read_int_sig_mult() {
        uint64_t b;
        int err = r.read_bits(1, &b);
        if (err) return err;
        if (b == 1) { /* opcodeUpdateSig */
            err = r.read_bits(1, &b);
            if (err) return err;
            if (b == 0) sig = 0;
            else {
                uint64_t s;
                err = r.read_bits(NUM_SIG_BITS, &s);
                if (err) return err;
                sig = (uint8_t)s + 1;
            }
        }
        err = r.read_bits(1, &b);
        if (err) return err;
        if (b == 1) { /* opcodeUpdateMult */
            uint64_t m;
            err = r.read_bits(NUM_MULT_BITS, &m);
            if (err) return err;
            mult = (uint8_t)m;
            if (mult > MAX_MULT) return M3GPU_SERIES_INVALID_MULT;
            mult_pow = exp10_sel(mult);
        }
        return 0;
    }
    __device__ __forceinline__ int read_int_val_diff() {
        if (sig == 64) { /* readIntValDiffSlow */
            uint64_t sb;
            int err = r.read_bits(1, &sb);
            if (err) return err;
            double sgn = (sb == 1) ? 1.0 : -1.0;
            uint64_t bits;
            err = r.read_bits(sig, &bits);
            if (err) return err;
            int_val += sgn * (double)bits;
            return 0;
        }
        uint64_t bits;
        int err = r.read_bits((uint32_t)sig + 1, &bits);
        if (err) return err;
        double sgn = -1.0;
        if ((bits >> sig) == 1) { sgn = 1.0; bits ^= (1ULL << sig); }
        int_val += sgn * (double)bits;
        return 0;
    }

    /* iterator.go:108-176 */
    __device__ __forceinline__ int read_first_value() {
        if (!int_optimized) return read_full_float();
        uint64_t b;
        int err = r.read_bits(1, &b);
        if (err) return err;
        if (b == 1) { /* float mode */
            err = read_full_float();
            if (err) return err;
            is_float = true;
            return 0;
        }
        err = read_int_sig_mult();
        if (err) return err;
        return read_int_val_diff();
    }
    __device__ __forceinline__ int read_next_value() {
        if (!int_optimized) return read_next_float();
        uint64_t b;
        int err = r.read_bits(1, &b);
        if (err) return err;
        if (b == 0) { /* opcodeUpdate */
            err = r.read_bits(1, &b);
            if (err) return err;
            if (b == 1) return 0; /* repeat */
            err = r.read_bits(1, &b);
            if (err) return err;
            if (b == 1) { /* -> float mode */
                err = read_full_float();
                if (err) return err;
                is_float = true;
                return 0;
            }
            err = read_int_sig_mult();
            if (err) return err;
            err = read_int_val_diff();
            if (err) return err;
            is_float = false;
            return 0;
        }
        if (is_float) return read_next_float();
        return read_int_val_diff();
    }

    /* Fully fused fast path: ONE 64-bit peek decodes the timestamp field
     * AND the common int-mode value field, then ONE consume advances the
     * window. Grammar consequence (i) from SURVEY.md Appendix A: short ts
     * fields (<=16 bits for buckets 0-2) plus a no-update int diff
     * (2+sig bits) fit one 64-bit window. Falls through to the stepwise
     * path (bit-identical) for markers, default buckets, float mode,
     * wide sig, unit changes and near-EOS. Returns 1/0/-err like next().
     * Returns -1000 to mean "take the general path" (nothing consumed). */
    /* XOR field fused from w2 (value bits left-aligned), pre = bits already
     * classified before the XOR control (ts field + optional mode prefix).
     * Returns 0 (consumed + state updated) or -1000 (take stepwise path). */
    __device__ __forceinline__ int fused_xor(uint64_t w2, uint32_t pre) {
        if (!(w2 >> 63)) { /* '0': same value */
            r.consume(pre + 1);
            prev_xor = 0;
            return 0;
        }
        if ((w2 >> 62) == 0x2) { /* '10' contained */
            uint32_t lead = prev_xor ? __builtin_clzll(prev_xor) : 64;
            uint32_t trail = prev_xor ? __builtin_ctzll(prev_xor) : 0;
            uint32_t nmean = 64 - lead - trail;
            if (pre + 2 + nmean > 64) return -1000;
            uint64_t mb = nmean ? ((w2 << 2) >> (64 - nmean)) : 0;
            r.consume(pre + 2 + nmean);
            prev_xor = mb << trail;
            prev_float_bits ^= prev_xor;
            return 0;
        }
        /* '11' + 6b lead + 6b (nmean-1), then payload (may exceed the peek) */
        uint64_t lead = (w2 >> 56) & 0x3f;
        uint64_t nmean = ((w2 >> 50) & 0x3f) + 1;
        r.consume(pre + 14);
        uint64_t mb;
        int err = r.read_bits((uint32_t)nmean, &mb);
        if (err) return -err; /* same EOF point as the stepwise reads */
        uint64_t trail = 64 - lead - nmean;
        prev_xor = mb << trail;
        prev_float_bits ^= prev_xor;
        return 0;
    }

    /* Branch-minimized fast path: ONE branch-free 64-bit peek classifies
     * the timestamp field (dod==0 | bucket 0-2) arithmetically — no
     * per-bucket branching — and the int-mode value field (no-update diff
     * | repeat) with pure select chains, then a single consume advances
     * the window. Per-lane divergence survives only at: the int/float
     * mode split, the rare escapes (markers, default buckets, sig/mode
     * updates, near-EOS) which return -1000 and fall to the stepwise
     * path (bit-identical semantics), and the window crossing inside
     * consume(). Grammar consequence (i), SURVEY.md Appendix A. */
    __device__ __forceinline__ int next_fused(int64_t* t, double* v) {
        if (tu_changed || !have_scheme || prev_time == 0 || r.bits_left < 64)
            return -1000;
        const uint64_t w = r.peek64();
        /* ---- timestamp: '0' => dod 0; '1'^L 0 + {7,9,12}b => bucket ---- */
        const uint64_t b0 = w >> 63;
        uint32_t tsbits = 1;
        int64_t dod = 0;
        if (b0) { /* branched: regular-cadence waves (dod==0 = a single
                   * 0 bit, the dominant shape) skip the bucket block */
            const uint32_t L = (uint32_t)__builtin_clzll(~w); /* leading 1s */
            if (L >= 4 || (w >> 55) == MARKER_OPCODE)
                return -1000; /* default bucket / EOS|annot|timeunit marker */
            const uint32_t vb = (0xC970u >> (L * 4)) & 0xFu; /* {7,9,12} */
            tsbits = L + 1 + vb;
            /* unit_ns==0 swallows invalid units (:271-274) */
            dod = sign_extend((w << (L + 1)) >> (64 - vb), vb) * unit_ns;
        }
        const uint64_t w2 = w << tsbits; /* tsbits <= 16 here */

        if (int_optimized && !is_float) {
            /* int mode: '1' sign+sig diff | '01' repeat | '000' sig/mult
             * update + diff (encoder.go:200-250) | '001' float switch ->
             * stepwise. Covering the update opcode here matters: adaptive
             * sig tracking (int_sig_bits_tracker.go) re-widths every few
             * points on random-walk data, and each stepwise escape drags
             * the whole wave through the big slow path. */
            const uint64_t v0 = w2 >> 63;
            const uint64_t v1 = (w2 >> 62) & 1;
            uint32_t nb;
            if (v0) { /* opcodeNoUpdate: sign + sig diff */
                if (sig > 45u || tsbits + 2u + sig > 64u)
                    return -1000; /* wide sig: field may exceed the peek */
                nb = 2u + sig;
                const uint64_t field = (w2 << 1) >> (63u - sig);
                const uint64_t sbit = (field >> sig) & 1;
                const uint64_t mag = field ^ (sbit << sig);
                int_val += (sbit ? 1.0 : -1.0) * (double)mag;
            } else if (v1) {
                /* repeat: int_val EXACTLY unchanged (adding 0.0 would flip
                 * -0.0; DESIGN.md §6 repeat quirk) */
                nb = 2;
            } else if ((w2 >> 61) & 1) {
                return -1000; /* '001' -> float-mode switch: stepwise */
            } else {
                /* '000' + readIntSigMult + readIntValDiff
                 * (iterator.go:178-219) from the same peek */
                const uint64_t u = w2 << 3;
                uint8_t nsig = sig;
                uint32_t off;
                if (u >> 63) { /* opcodeUpdateSig */
                    const uint64_t z = (u >> 62) & 1;
                    nsig = z ? (uint8_t)(((u >> 56) & 0x3f) + 1) : 0;
                    off = z ? 8u : 2u;
                } else {
                    off = 1u;
                }
                const uint64_t m = u << off;
                uint8_t nmult = mult;
                uint32_t moff;
                if (m >> 63) { /* opcodeUpdateMult */
                    nmult = (uint8_t)((m >> 60) & 0x7);
                    moff = 4u;
                    if (nmult > MAX_MULT)
                        return -1000; /* stepwise raises invalid_mult */
                } else {
                    moff = 1u;
                }
                if (nsig > 45u || tsbits + 3u + off + moff + 1u + nsig > 64u)
                    return -1000;
                const uint64_t f = m << moff;
                const uint64_t sbit = f >> 63;
                const uint64_t mag =
                    nsig ? ((f << 1) >> ((64u - nsig) & 63u)) : 0;
                int_val += (sbit ? 1.0 : -1.0) * (double)mag;
                sig = nsig;
                if (nmult != mult) { mult = nmult; mult_pow = exp10_sel(nmult); }
                nb = 3u + off + moff + 1u + nsig;
            }
            r.consume(tsbits + nb);
            prev_time_delta += dod;
            prev_time += prev_time_delta;
            *t = prev_time;
            /* the f64 divide is ~25 VALU ops: branch it so mult==0 waves
             * (the common case) never pay for a select-discarded divide */
            double outv = int_val;
            if (mult != 0) outv = int_val / mult_pow;
            *v = outv;
            return 1;
        }
        /* float value (pure float stream, or int-optimized float mode) */
        int rx;
        if (!int_optimized) {
            rx = fused_xor(w2, tsbits);
        } else if (w2 >> 63) { /* '1' + XOR */
            rx = fused_xor(w2 << 1, tsbits + 1);
        } else if ((w2 >> 62) & 1) { /* '01' repeat */
            r.consume(tsbits + 2);
            rx = 0;
        } else {
            return -1000; /* '00...' mode change: stepwise */
        }
        if (rx == -1000) return -1000; /* nothing consumed */
        if (rx) return rx; /* negative error, same EOF point as stepwise */
        prev_time_delta += dod;
        prev_time += prev_time_delta;
        *t = prev_time;
        *v = bits2f(prev_float_bits);
        return 1;
    }

    /* One point. Returns 1 = value in (*t,*v), 0 = done, -err on error. */
    __device__ __forceinline__ int next(int64_t* t, double* v) {
        if (done) return 0;
        int f = next_fused(t, v);
        if (f != -1000) {
            points += (f == 1);
            return f;
        }
        bool first;
        int err = read_timestamp(&first);
        if (err) return -err;
        if (done) return 0;
        err = first ? read_first_value() : read_next_value();
        if (err) return -err;
        *t = prev_time;
        if (!int_optimized || is_float) *v = bits2f(prev_float_bits);
        else if (mult != 0) *v = int_val / mult_pow;
        else *v = int_val;
        points++;
        return 1;
    }
    /* select chain, NOT an indexed array: a dynamically-indexed local
     * const table lowers to a .rodata global_load whose vmcnt(0) wait
     * (gfx9 vmcnt counts stores too) drains every in-flight output store
     * right in the per-point value chain */
    __device__ __forceinline__ double exp10_table(uint8_t m) {
        return exp10_sel(m);
    }
};

/* ========================= decode kernel ========================= */
/* ONE SERIES PER LANE: a wave decodes 64 independent streams in parallel,
 * one point per active lane per iteration. The VLC parser state lives in
 * per-lane VGPRs; data-dependent branches diverge only where lanes disagree
 * (host-side batches group similar series for lane coherence, but any order
 * is correct).
 *
 * Memory structure (v6):
 *  - INPUT through a per-lane LDS ring (BitReader::refill): the global
 *    gathers leave the per-point chain — once per 8-point tile each lane
 *    bursts up to 8 loads of consecutive words of its own stream (per-lane
 *    64B lines fetched once, L1-amortized), and the parser pulls words
 *    with cheap conflict-free ds_read_b64s.
 *  - OUTPUT accumulated in per-lane REGISTERS (8 ts + 8 val slots, indexed
 *    by the unrolled tile counter) and flushed as 4+4 back-to-back 16B
 *    stores into the lane's own rows: consecutive stores complete each
 *    64B line, so write-combining holds WRITE_SIZE at the algorithmic
 *    16 B/pt with no LDS transpose and no cross-lane shuffles. */

#define DEC_TILE 8

__global__ void __launch_bounds__(BLOCK_THREADS, 4) /* hold the 128-VGPR
    granule: ring + tiles = 40 KB/block -> 4 blocks/CU, 4 waves/SIMD */
k_decode_batch(const uint8_t* __restrict__ blobs,
               const uint64_t* __restrict__ offsets,
               const uint32_t* __restrict__ lens,
               const int32_t* __restrict__ perm,
               uint32_t nseries, int int_optimized, uint8_t default_unit,
               int64_t* __restrict__ out_ts, double* __restrict__ out_vals,
               uint32_t* __restrict__ out_counts, int32_t* __restrict__ out_errs,
               uint32_t stride,
               uint8_t* __restrict__ out_ann, uint32_t ann_stride) {
    const uint32_t lane = threadIdx.x & (WAVE - 1);
    const uint32_t wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const uint32_t slot = blockIdx.x * BLOCK_THREADS + wave * WAVE + lane;
    /* optional scheduling permutation (e.g. length-sorted): waves then get
     * 64 similar-cost streams, removing intra-wave and intra-CU skew.
     * Purely a schedule: outputs still land in series order. */
    const uint32_t series = (perm && slot < nseries) ? (uint32_t)perm[slot] : slot;

    /* input ring + output tile. The tile is AoS — (ts, val) PAIRS, one
     * ds_write_b128 per point instead of two b64s — with a pair-granular
     * XOR column swizzle (col ^ (lane & 7)). */
    __shared__ uint64_t ring_all[WAVES_PER_BLOCK][WAVE][IN_STRIDE];
    __shared__ longlong2 tile_all[WAVES_PER_BLOCK][WAVE][DEC_TILE];
    uint64_t* ring = ring_all[wave][lane];
    longlong2 (*tile)[DEC_TILE] = tile_all[wave];

    const bool in_range = slot < nseries;
    Decoder d;
    d.init(blobs, in_range ? offsets[series] : 0, in_range ? lens[series] : 0,
           int_optimized != 0, default_unit, ring,
           (out_ann && in_range) ? out_ann + (uint64_t)series * ann_stride
                                 : nullptr,
           ann_stride);

    bool running = in_range;
    int err = 0;
    uint32_t cnt = 0;
    uint32_t k = 0;

    const bool discard = out_ts == nullptr; /* parse-only diagnostic */

    /* transposed flush: lane l covers row (l>>3)+8j, point (l&7) — one
     * ds_read_b128 pulls the (ts, val) pair, then two 8B stores whose
     * addresses are consecutive across the 8 lanes of a row = full
     * 64B-line utilization (WRITE_SIZE stays at the algorithmic 16 B/pt) */
    auto flush = [&](uint32_t base_pt) {
        if (discard) return;
        __builtin_amdgcn_wave_barrier();
        const uint32_t p = lane & (DEC_TILE - 1);
        const uint32_t r0 = lane / DEC_TILE;
        const uint32_t pt = base_pt + p;
        if (__all(cnt >= base_pt + DEC_TILE)) {
            /* all 64 rows full: unconditional stores, no per-row counts */
            for (uint32_t j = 0; j < DEC_TILE; j++) {
                uint32_t rr = r0 + j * (WAVE / DEC_TILE);
                uint32_t pc = p ^ (rr & 7);
                uint64_t row = (uint64_t)__shfl((int)series, (int)rr);
                longlong2 pr = tile[rr][pc];
                out_ts[row * stride + pt] = pr.x;
                out_vals[row * stride + pt] = __longlong_as_double(pr.y);
            }
        } else {
            for (uint32_t j = 0; j < DEC_TILE; j++) {
                uint32_t rr = r0 + j * (WAVE / DEC_TILE);
                uint32_t pc = p ^ (rr & 7);
                uint32_t cc = (uint32_t)__shfl((int)cnt, (int)rr);
                uint64_t row = (uint64_t)__shfl((int)series, (int)rr);
                if (pt < cc) {
                    longlong2 pr = tile[rr][pc];
                    out_ts[row * stride + pt] = pr.x;
                    out_vals[row * stride + pt] = __longlong_as_double(pr.y);
                }
            }
        }
        __builtin_amdgcn_wave_barrier();
    };

    /* capacity is tile-hoisted: the per-point check only runs on the rare
     * tile that could cross `stride` */
    auto step = [&](uint32_t j, bool check_cap) {
        if (running) {
            int64_t t;
            double v;
            int rstat = d.next(&t, &v);
            if (rstat <= 0) {
                err = -rstat;
                running = false;
            } else if (check_cap && cnt >= stride) {
                err = M3GPU_SERIES_CAPACITY;
                running = false;
            } else {
                if (!discard) {
                    uint32_t col = j ^ (lane & 7);
                    longlong2 pr;
                    pr.x = t;
                    pr.y = __double_as_longlong(v);
                    tile[lane][col] = pr;
                }
                cnt++;
            }
        }
    };

    while (__any(running)) {
        d.r.refill_issue(running);
        /* NOT unrolled: the fully-inlined parser is ~10k instructions —
         * eight copies would blow the instruction cache */
        /* wave-uniform capacity flag: cnt <= k + j, so the per-point
         * check matters only on a tile that could cross `stride` */
        const bool check_cap = k + DEC_TILE > stride;
#pragma unroll 1
        for (uint32_t j = 0; j < DEC_TILE; j++) step(j, check_cap);
        /* commit BEFORE the flush: the commit's counted vmcnt wait then
         * covers only this tile's loads plus the PREVIOUS tile's stores
         * (issued a whole tile ago, long retired) — never the 16 stores
         * the flush is about to issue */
        d.r.refill_commit();
        flush(k);
        k += DEC_TILE;
    }

    if (in_range) {
        d.ann_finish(&err);
        out_counts[series] = cnt;
        out_errs[series] = err;
    }
}

/* ===================== device encoder ===================== */
/* Word-based big-endian bit emitter producing the byte-identical stream of
 * ostream.go:133-221; lane (w & 63) stages word w, coalesced 512B flushes. */

struct BitWriter {
    /* Per-lane big-endian bit emitter producing the byte-identical stream
     * of ostream.go:133-221. Each lane owns one output row; completed 64-bit
     * words store directly (8B per store; rows padded to 8B). */
    uint64_t acc;       /* left-aligned bit accumulator */
    uint32_t used;      /* bits used in acc */
    uint32_t nwords;    /* full words emitted */
    uint64_t pend;      /* buffered even word: pairs store as aligned 16B */
    uint64_t* out;      /* 16B-aligned output row */
    uint32_t cap_words;
    int err;

    __device__ __forceinline__ void init(uint8_t* row, uint32_t cap_bytes) {
        acc = 0; used = 0; nwords = 0; pend = 0;
        out = (uint64_t*)row;
        cap_words = cap_bytes / 8;
        err = 0;
    }
    __device__ __forceinline__ void emit_word(uint64_t w) {
        if (nwords >= cap_words) { err = M3GPU_SERIES_CAPACITY; return; }
        if (nwords & 1) {
            ulonglong2 pr;
            pr.x = pend;
            pr.y = __builtin_bswap64(w);
            *(ulonglong2*)(out + nwords - 1) = pr;
        } else {
            pend = __builtin_bswap64(w);
        }
        nwords++;
    }
    __device__ __forceinline__ void write_bits(uint64_t v, uint32_t n) {
        if (n == 0 || err) return;
        v = (n >= 64) ? v : (v & ((1ULL << n) - 1)); /* low n bits */
        uint32_t space = 64 - used;
        if (n <= space) {
            acc |= v << (space - n);
            used += n;
            if (used == 64) { emit_word(acc); acc = 0; used = 0; }
        } else {
            uint32_t lo = n - space;      /* remainder */
            acc |= v >> lo;
            emit_word(acc);
            acc = (lo >= 64) ? 0 : (v << (64 - lo));
            used = lo;
        }
    }
    __device__ __forceinline__ void write_bit(uint32_t b) { write_bits(b, 1); }
    /* Flush the buffered word + partial tail word. Returns byte length. */
    __device__ __forceinline__ uint32_t finish() {
        if (nwords & 1) out[nwords - 1] = pend; /* odd word count: flush pend */
        uint32_t nbytes = nwords * 8;
        if (used > 0) {
            if (nwords >= cap_words) { err = M3GPU_SERIES_CAPACITY; return 0; }
            out[nwords] = __builtin_bswap64(acc); /* zero-padded */
            nbytes += (used + 7) / 8;
        }
        return nbytes;
    }
};

/* m3tsz.go:78-119 convertToIntFloat — bit-sensitive: -ffp-contract=off.
 * math.Nextafter on a strictly-positive finite double is a 1-ulp step,
 * which for IEEE positives is an integer inc/dec of the bit pattern —
 * exact, and much cheaper than the libm nextafter sequence inside this
 * up-to-7-iteration loop (val > 0 is guaranteed here: val == 0 has
 * r == 0 and returns first; toward-0 steps down, toward i+1 > val steps
 * up). */
__device__ __forceinline__ double go_next_down(double x) { /* x > 0 finite */
    return __longlong_as_double(__double_as_longlong(x) - 1);
}
__device__ __forceinline__ double go_next_up(double x) { /* x > 0 finite */
    return __longlong_as_double(__double_as_longlong(x) + 1);
}

__device__ __forceinline__ int convert_to_int_float(double v, uint8_t cur_max_mult,
                                    double* out_val, uint8_t* out_mult, bool* out_is_float) {
    const double MAXINT = 9223372036854775808.0;
    if (cur_max_mult == 0 && v < MAXINT) {
        double i, r;
        r = go_modf(v, &i);
        if (r == 0) { *out_val = i; *out_mult = 0; *out_is_float = false; return 0; }
    }
    if (cur_max_mult > MAX_MULT) return M3GPU_SERIES_INVALID_MULT;
    double sign = 1.0;
    if (v < 0) sign = -1.0;
    for (uint8_t m = cur_max_mult; m <= MAX_MULT; m++) {
        double val = v * exp10_sel(m) * sign;
        if (val >= 1e13) break;
        double i, r;
        r = go_modf(val, &i);
        if (r == 0) { *out_val = sign * i; *out_mult = m; *out_is_float = false; return 0; }
        else if (r < 0.1) {
            if (go_next_down(val) <= i) { *out_val = sign * i; *out_mult = m; *out_is_float = false; return 0; }
        } else if (r > 0.9) {
            double nxt = i + 1;
            if (go_next_up(val) >= nxt) { *out_val = sign * nxt; *out_mult = m; *out_is_float = false; return 0; }
        }
    }
    *out_val = v; *out_mult = 0; *out_is_float = true;
    return 0;
}

/* Encoder state: encoder.go:42-61 + TimestampEncoder + sig tracker. The bulk
 * path has a fixed unit and no annotations (m3gpu.h contract). */
struct Encoder {
    BitWriter w;
    int64_t unit_ns; /* cached UNIT_NS_D[unit] — the bulk path has one
                      * fixed unit; an indexed read per point would be a
                      * .rodata global load + vmcnt(0) in the hot chain */
    int64_t prev_time, prev_time_delta;
    uint64_t prev_xor, prev_float_bits;
    double int_val;
    uint8_t time_unit, max_mult;
    uint8_t num_sig_state, cur_highest_lower_sig, num_lower_sig;
    bool has_written_first, is_float, int_optimized;
    uint32_t num_encoded;

    __device__ __forceinline__ void init(uint8_t* row, uint32_t cap_bytes,
                         int64_t start_ns, bool intopt, uint8_t default_unit) {
        w.init(row, cap_bytes);
        prev_time = start_ns;
        prev_time_delta = 0;
        prev_xor = 0; prev_float_bits = 0;
        int_val = 0;
        /* initialTimeUnit (timestamp_encoder.go:248-259) */
        unit_ns = unit_valid(default_unit) ? UNIT_NS_D[default_unit] : 0;
        time_unit = (unit_valid(default_unit) && unit_ns != 0 &&
                     start_ns % unit_ns == 0) ? default_unit : 0;
        max_mult = 0;
        num_sig_state = 0; cur_highest_lower_sig = 0; num_lower_sig = 0;
        has_written_first = false; is_float = false;
        int_optimized = intopt;
        num_encoded = 0;
    }

    __device__ __forceinline__ void write_marker(uint32_t marker) {
        w.write_bits(MARKER_OPCODE, 9);
        w.write_bits(marker, 2);
    }

    /* unit_ns takes one of a few fixed values (unit.go:30-42); constant-
     * divisor arms (the branch is wave-uniform) let the compiler emit
     * magic-multiply sequences instead of a ~40-op runtime 64-bit
     * division per encoded point */
    __device__ __forceinline__ int64_t div_unit(int64_t x) const {
        switch (unit_ns) {
        case 1LL: return x;
        case 1000LL: return x / 1000LL;
        case 1000000LL: return x / 1000000LL;
        case 1000000000LL: return x / 1000000000LL;
        default: return x / unit_ns;
        }
    }

    /* timestamp_encoder.go:205-246 */
    __device__ __forceinline__ int write_dod_unchanged(int64_t prev_delta, int64_t cur_delta, uint8_t unit) {
        if (!unit_valid(unit)) return M3GPU_SERIES_NO_SCHEME;
        int64_t dod = div_unit(cur_delta - prev_delta);
        if (unit == 1 || unit == 2) {
            if ((int64_t)(int32_t)dod != dod) return M3GPU_SERIES_DOD_OVERFLOW;
        }
        int dbits = scheme_default_bits(unit);
        if (!dbits) return M3GPU_SERIES_NO_SCHEME;
        if (dod == 0) { w.write_bits(0, 1); return 0; }
        /* bucket select arithmetically (the table-lookup form vectorizes
         * into per-lane .rodata global loads): dod fits b two's-complement
         * bits iff bits(|folded|)+1 <= b */
        const uint64_t m = (uint64_t)(dod < 0 ? ~dod : dod);
        const uint32_t nb = (m ? 64u - (uint32_t)__builtin_clzll(m) : 0u) + 1u;
        if (nb <= 12u) {
            const uint32_t sel = (nb <= 7u) ? 0u : (nb <= 9u) ? 1u : 2u;
            const uint32_t opc = (sel == 0) ? 0x2u : (sel == 1) ? 0x6u : 0xeu;
            const uint32_t vb = (sel == 0) ? 7u : (sel == 1) ? 9u : 12u;
            w.write_bits(opc, sel + 2u);
            w.write_bits((uint64_t)dod, vb);
            return 0;
        }
        w.write_bits(0xf, 4);
        w.write_bits((uint64_t)dod, dbits);
        return 0;
    }

    /* timestamp_encoder.go:72-129 (fixed unit, no annotations) */
    __device__ __forceinline__ int write_time(int64_t cur_time, uint8_t unit) {
        if (!has_written_first) {
            w.write_bits((uint64_t)prev_time, 64);
            has_written_first = true;
        }
        bool tu_changed = false;
        if (unit_valid(unit) && unit != time_unit) { /* maybeWriteTimeUnitChange */
            write_marker(MARKER_TIMEUNIT);
            w.write_bits(unit, 8);
            time_unit = unit;
            unit_ns = UNIT_NS_D[unit];
            tu_changed = true;
        }
        int64_t time_delta = cur_time - prev_time;
        prev_time = cur_time;
        if (tu_changed) {
            w.write_bits((uint64_t)(time_delta - prev_time_delta), 64);
            prev_time_delta = 0;
            return 0;
        }
        int err = write_dod_unchanged(prev_time_delta, time_delta, unit);
        prev_time_delta = time_delta;
        return err;
    }

    /* float_encoder_iterator.go:69-103 */
    __device__ __forceinline__ void write_full_float(uint64_t val) {
        prev_float_bits = val;
        prev_xor = val;
        w.write_bits(val, 64);
    }
    __device__ __forceinline__ void write_xor(uint64_t cur_xor) {
        if (cur_xor == 0) { w.write_bits(0, 1); return; }
        uint32_t pl = prev_xor ? __builtin_clzll(prev_xor) : 64;
        uint32_t pt = prev_xor ? __builtin_ctzll(prev_xor) : 0;
        uint32_t cl = __builtin_clzll(cur_xor);
        uint32_t ct = __builtin_ctzll(cur_xor);
        if (cl >= pl && ct >= pt) {
            w.write_bits(0x2, 2);
            w.write_bits(cur_xor >> pt, 64 - pl - pt);
            return;
        }
        w.write_bits(0x3, 2);
        w.write_bits(cl, 6);
        uint32_t nmean = 64 - cl - ct;
        w.write_bits(nmean - 1, 6);
        w.write_bits(cur_xor >> ct, nmean);
    }
    __device__ __forceinline__ void write_next_float(uint64_t val) {
        uint64_t x = prev_float_bits ^ val;
        write_xor(x);
        prev_xor = x;
        prev_float_bits = val;
    }

    /* int_sig_bits_tracker.go:35-91 */
    __device__ __forceinline__ void tracker_write_int_val_diff(uint64_t val_bits, bool neg) {
        w.write_bit(neg ? 1 : 0);
        w.write_bits(val_bits, num_sig_state);
    }
    __device__ __forceinline__ void tracker_write_int_sig(uint8_t s) {
        if (num_sig_state != s) {
            w.write_bit(1);
            if (s == 0) w.write_bit(0);
            else { w.write_bit(1); w.write_bits((uint64_t)(s - 1), NUM_SIG_BITS); }
        } else {
            w.write_bit(0);
        }
        num_sig_state = s;
    }
    __device__ __forceinline__ uint8_t tracker_track_new_sig(uint8_t nsig) {
        uint8_t new_sig = num_sig_state;
        if (nsig > num_sig_state) {
            new_sig = nsig;
        } else if (num_sig_state - nsig >= SIG_DIFF_THRESHOLD) {
            if (num_lower_sig == 0) cur_highest_lower_sig = nsig;
            else if (nsig > cur_highest_lower_sig) cur_highest_lower_sig = nsig;
            num_lower_sig++;
            if (num_lower_sig >= SIG_REPEAT_THRESHOLD) {
                new_sig = cur_highest_lower_sig;
                num_lower_sig = 0;
            }
        } else {
            num_lower_sig = 0;
        }
        return new_sig;
    }

    /* encoder.go:233-250 */
    __device__ __forceinline__ void write_int_sig_mult(uint8_t s, uint8_t m, bool float_changed) {
        tracker_write_int_sig(s);
        if (m > max_mult) {
            w.write_bit(1);
            w.write_bits(m, NUM_MULT_BITS);
            max_mult = m;
        } else if (num_sig_state == s && max_mult == m && float_changed) {
            w.write_bit(1);
            w.write_bits(max_mult, NUM_MULT_BITS);
        } else {
            w.write_bit(0);
        }
    }

    /* encoder.go:112-146 */
    __device__ __forceinline__ int write_first_value(double v) {
        if (!int_optimized) { write_full_float(f2bits(v)); return 0; }
        double val; uint8_t m; bool isf;
        int err = convert_to_int_float(v, 0, &val, &m, &isf);
        if (err) return err;
        if (isf) {
            w.write_bit(1);
            write_full_float(f2bits(v));
            is_float = true;
            max_mult = m;
            return 0;
        }
        w.write_bit(0);
        int_val = val;
        bool neg_diff = true;
        if (val < 0) { neg_diff = false; val = -1 * val; }
        uint64_t val_bits = (uint64_t)go_f2i(val);
        uint8_t nsig = num_sig(val_bits);
        write_int_sig_mult(nsig, m, false);
        tracker_write_int_val_diff(val_bits, neg_diff);
        return 0;
    }

    /* encoder.go:174-231 */
    __device__ __forceinline__ void write_float_val(uint64_t val, uint8_t m) {
        if (!is_float) {
            w.write_bit(0); w.write_bit(0); w.write_bit(1);
            write_full_float(val);
            is_float = true;
            max_mult = m;
            return;
        }
        if (val == prev_float_bits) { w.write_bit(0); w.write_bit(1); return; }
        w.write_bit(1);
        write_next_float(val);
    }
    __device__ __forceinline__ void write_int_val(double val, uint8_t m, bool isf, double val_diff) {
        if (val_diff == 0 && isf == is_float && m == max_mult) {
            w.write_bit(0); w.write_bit(1);
            return;
        }
        bool neg = false;
        if (val_diff < 0) { neg = true; val_diff = -1 * val_diff; }
        uint64_t diff_bits = (uint64_t)go_f2i(val_diff);
        uint8_t nsig = num_sig(diff_bits);
        uint8_t new_sig = tracker_track_new_sig(nsig);
        bool float_changed = (isf != is_float);
        if (m > max_mult || num_sig_state != new_sig || float_changed) {
            w.write_bit(0); w.write_bit(0); w.write_bit(0);
            write_int_sig_mult(new_sig, m, float_changed);
            tracker_write_int_val_diff(diff_bits, neg);
            is_float = false;
        } else {
            w.write_bit(1);
            tracker_write_int_val_diff(diff_bits, neg);
        }
        int_val = val;
    }

    /* encoder.go:148-172 */
    __device__ __forceinline__ int write_next_value(double v) {
        if (!int_optimized) { write_next_float(f2bits(v)); return 0; }
        double val; uint8_t m; bool isf;
        int err = convert_to_int_float(v, max_mult, &val, &m, &isf);
        if (err) return err;
        double val_diff = 0;
        if (!isf) val_diff = int_val - val;
        if (isf || val_diff >= 9223372036854775808.0 || val_diff <= -9223372036854775808.0) {
            write_float_val(f2bits(val), m);
            return 0;
        }
        write_int_val(val, m, isf, val_diff);
        return 0;
    }

    __device__ __forceinline__ int encode(int64_t t, double v, uint8_t unit) {
        int err = write_time(t, unit);
        if (err) return err;
        err = num_encoded == 0 ? write_first_value(v) : write_next_value(v);
        if (err == 0) num_encoded++;
        return err ? err : w.err;
    }

    /* Finalize: appending the EOS marker to the live bitstream produces
     * exactly head[:len-1] + Tail(lastByte, pos) (scheme.go:198-212). */
    __device__ __forceinline__ uint32_t finalize() {
        if (num_encoded == 0 && w.nwords == 0 && w.used == 0) return 0;
        write_marker(MARKER_EOS);
        return w.finish();
    }
};

__global__ void __launch_bounds__(BLOCK_THREADS)
k_encode_batch(const int64_t* __restrict__ ts, const double* __restrict__ vals,
               const uint32_t* __restrict__ counts, uint32_t nseries,
               uint32_t stride, int int_optimized, uint8_t unit,
               uint8_t* __restrict__ out_bytes, uint32_t out_stride,
               uint32_t* __restrict__ out_lens, int32_t* __restrict__ out_errs) {
    /* ONE SERIES PER LANE (like decode). Input points stage through an LDS
     * tile filled cooperatively — fill step j: lane l loads row (l>>3)+8j,
     * point (l&7): 8 consecutive 8B addresses per row = full 64B-line
     * utilization — instead of 64 independent 8B gathers per point. */
    const uint32_t lane = threadIdx.x & (WAVE - 1);
    const uint32_t wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const uint32_t s_base = blockIdx.x * BLOCK_THREADS + wave * WAVE;
    const uint32_t series = s_base + lane;

    /* unpadded tiles with an XOR column swizzle (col ^ (row & 7)): bank
     * conflict-free for both per-lane access and the coalesced tile pass,
     * and the exact 32 KB/block admits 5 blocks/CU (5 waves/SIMD). */
    __shared__ int64_t ts_tile_all[WAVES_PER_BLOCK][WAVE][DEC_TILE];
    __shared__ double val_tile_all[WAVES_PER_BLOCK][WAVE][DEC_TILE];
    int64_t (*ts_tile)[DEC_TILE] = ts_tile_all[wave];
    double (*val_tile)[DEC_TILE] = val_tile_all[wave];

    const bool in_range = series < nseries;
    uint32_t n = in_range ? counts[series] : 0;

    Encoder e;
    if (in_range)
        e.init(out_bytes + (uint64_t)series * out_stride, out_stride,
               n ? ts[(uint64_t)series * stride] : 0, int_optimized != 0, unit);

    auto fill_tile = [&](uint32_t base_pt) {
        __builtin_amdgcn_wave_barrier();
        const uint32_t p = lane & (DEC_TILE - 1);
        const uint32_t r0 = lane / DEC_TILE;
        for (uint32_t j = 0; j < DEC_TILE; j++) {
            uint32_t r = r0 + j * (WAVE / DEC_TILE);
            uint32_t c = (uint32_t)__shfl((int)n, (int)r);
            uint32_t pt = base_pt + p;
            if (pt < c) {
                uint64_t row = (uint64_t)(s_base + r) * stride + pt;
                uint32_t pc = p ^ (r & 7);
                ts_tile[r][pc] = ts[row];
                val_tile[r][pc] = vals[row];
            }
        }
        __builtin_amdgcn_wave_barrier();
    };

    bool running = in_range && n > 0;
    int err = 0;
    uint32_t len = 0;
    uint32_t j = 0;
    while (__any(running)) {
        if ((j & (DEC_TILE - 1)) == 0) fill_tile(j);
        if (running) {
            uint32_t col = (j & (DEC_TILE - 1)) ^ (lane & 7);
            int64_t t = ts_tile[lane][col];
            double v = val_tile[lane][col];
            err = e.encode(t, v, unit);
            if (err) {
                running = false;
            } else if (j + 1 == n) {
                len = e.finalize();
                err = e.w.err;
                running = false;
            }
        }
        j++;
    }
    if (in_range) {
        out_lens[series] = err ? 0 : len;
        out_errs[series] = err;
    }
}

/* ===================== fused decode -> rollup kernel ===================== */
/* Reference semantics:
 *  bucket assignment  generic_elem.go:219-221 (truncate to window)
 *  output timestamp   list.go:541-543 (window END)
 *  Counter            counter.go:52-131  (int64 sum/min/max/count/sumSq)
 *  Gauge              gauge.go:73-165    (NaN rules :87-98; last by ts order)
 *  Timer              timer.go:56-153 over the CKMS stream; for <= 64 values
 *                     per bucket the production-default CKMS (eps=1e-3,
 *                     insertAndCompressEvery=1024) equals the no-compression
 *                     calcQuantiles walk (stream.go:231-277), implemented
 *                     here exactly (incl. the one-emission-per-sample shift
 *                     for colliding ranks); >64 values flags
 *                     M3GPU_SERIES_BUCKET_OVERFLOW.
 */

#define QCAP 512 /* wave-kernel LDS staging capacity per series */
#define MAX_AGGS 16

struct RollupPlan {
    int32_t agg_types[MAX_AGGS];
    int8_t qidx[MAX_AGGS];   /* index into qs, or -1 */
    double qs[MAX_AGGS];     /* sorted unique quantiles */
    int32_t nq;
    int32_t naggs;
    /* Largest bucket size for which the production-default CKMS
     * (eps=1e-3, insertAndCompressEvery=1024) provably never compresses
     * for THIS quantile set, so its quantiles are exact order statistics
     * (the calcQuantiles walk). Computed on the host; buckets beyond it
     * flag M3GPU_SERIES_BUCKET_OVERFLOW rather than approximate. */
    int32_t exact_cap;
    /* Largest bucket size for which EVERY requested quantile resolves to
     * the bucket MAX under the reference's walk (all-high quantile sets,
     * e.g. p90+): the lane kernel then needs no value staging at all —
     * running min/max suffice. 0 = extreme mode not applicable. */
    int32_t extreme_cap;
    /* CKMS stream options (quantile/cm/options.go:30-32 defaults:
     * eps=1e-3, insertAndCompressEvery=1024); configurable via the
     * *_opts ABI entries. every is capped at 1024 by validation (the
     * deep-tier LDS buffers are sized for a 2*1024 peak). */
    double eps;
    int32_t every;
};

struct BucketState {
    int64_t isum, imin, imax, isumsq;
    double fsum, fsumsq, fmin, fmax, last;
    int64_t last_at;
    int64_t count;

    __device__ __forceinline__ void reset() {
        isum = 0; isumsq = 0;
        imin = INT64_MAX; imax = INT64_MIN;   /* counter.go:44-47 */
        fsum = 0; fsumsq = 0;
        fmin = __longlong_as_double(0x7ff8000000000000LL); /* NaN, gauge.go:56-59 */
        fmax = __longlong_as_double(0x7ff8000000000000LL);
        last = 0; last_at = 0;
        count = 0;
    }
};

__device__ __forceinline__ double m3_stdev(int64_t count, double sum_sq, double sum) { /* common.go:29-36 */
    int64_t div = count * (count - 1);
    if (div == 0) return 0.0;
    return sqrt(((double)count * sum_sq - sum * sum) / (double)div);
}

__global__ void __launch_bounds__(BLOCK_THREADS)
k_rollup_batch(const uint8_t* __restrict__ blobs,
               const uint64_t* __restrict__ offsets,
               const uint32_t* __restrict__ lens,
               const int32_t* __restrict__ select, /* optional series subset */
               uint32_t nseries, int int_optimized, uint8_t default_unit,
               int metric_type, int64_t window_ns, uint32_t nbuckets,
               RollupPlan plan,
               double* __restrict__ out, int64_t* __restrict__ out_window_ts,
               int32_t* __restrict__ out_errs) {
    /* readfirstlane makes the series index provably wave-uniform: the
     * whole parser then compiles to scalar (SGPR) code with scalar
     * branches instead of exec-mask divergence sequences. */
    const uint32_t wave = __builtin_amdgcn_readfirstlane(threadIdx.x / WAVE);
    const uint32_t lane = threadIdx.x % WAVE;
    const uint32_t slot = blockIdx.x * WAVES_PER_BLOCK + wave;
    const uint32_t series = __builtin_amdgcn_readfirstlane(
        (select && slot < nseries) ? (uint32_t)select[slot] : slot);
    /* per-wave bucket value staging for quantiles (timer) */
    __shared__ double qvals_all[WAVES_PER_BLOCK][QCAP];
    __shared__ double sorted_all[WAVES_PER_BLOCK][QCAP];
    double* qvals = qvals_all[wave];
    double* sorted = sorted_all[wave];
    if (slot >= nseries) return;

    Decoder d;
    d.init(blobs, offsets[series], lens[series], int_optimized != 0, default_unit);

    double* out_row = out + (uint64_t)series * nbuckets * plan.naggs;
    int64_t* wts_row = out_window_ts ? out_window_ts + (uint64_t)series * nbuckets : nullptr;

    BucketState bs;
    bs.reset();
    int64_t base = 0;
    int64_t cur_bucket = -1;
    uint32_t nq_in_bucket = 0;
    int err = 0;

    auto emit_bucket = [&](int64_t b) {
        if (b < 0 || b >= (int64_t)nbuckets) return;
        if (wts_row && lane == 0) wts_row[b] = base + (b + 1) * window_ns;
        uint32_t n = nq_in_bucket;
        /* quantile targets: the no-compression calcQuantiles walk closed
         * form: k_i = max(rank_i, k_{i-1}+1), value = sorted[min(k_i,n)-1];
         * n<=3: sorted[min(int(q*n), n-1)] (quantilesFromBuf) */
        if (metric_type == M3GPU_METRIC_TIMER && plan.nq > 0 && n > 0) {
            /* rank-select: each lane ranks elements lane, lane+64, ...
             * LDS ops from one wave retire in order; the fences only stop
             * compiler reordering around the cross-lane LDS use. */
            __builtin_amdgcn_wave_barrier();
            __threadfence_block();
            for (uint32_t e = lane; e < n; e += WAVE) {
                double v = qvals[e];
                uint32_t rank = 0;
                for (uint32_t j = 0; j < n; j++) {
                    double o = qvals[j];
                    rank += (o < v) || (o == v && j < e);
                }
                sorted[rank] = v;
            }
            __builtin_amdgcn_wave_barrier();
            __threadfence_block();
        }
        if (lane < (uint32_t)plan.naggs) {
            int32_t t = plan.agg_types[lane];
            int8_t qi = plan.qidx[lane];
            double r = 0;
            if (metric_type == M3GPU_METRIC_COUNTER) { /* counter.go:112-131 */
                switch (t) {
                case M3GPU_AGG_MIN: r = (double)bs.imin; break;
                case M3GPU_AGG_MAX: r = (double)bs.imax; break;
                case M3GPU_AGG_MEAN: r = bs.count ? (double)bs.isum / (double)bs.count : 0; break;
                case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                case M3GPU_AGG_SUM: r = (double)bs.isum; break;
                case M3GPU_AGG_SUMSQ: r = (double)bs.isumsq; break;
                case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, (double)bs.isumsq, (double)bs.isum); break;
                default: r = 0; break;
                }
            } else if (metric_type == M3GPU_METRIC_GAUGE) { /* gauge.go:144-165 */
                switch (t) {
                case M3GPU_AGG_LAST: r = bs.last; break;
                case M3GPU_AGG_MIN: r = bs.fmin; break;
                case M3GPU_AGG_MAX: r = bs.fmax; break;
                case M3GPU_AGG_MEAN: r = bs.count ? bs.fsum / (double)bs.count : 0.0; break;
                case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                case M3GPU_AGG_SUM: r = bs.fsum; break;
                case M3GPU_AGG_SUMSQ: r = bs.fsumsq; break;
                case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, bs.fsumsq, bs.fsum); break;
                default: r = 0; break;
                }
            } else { /* timer.go:131-153 */
                if (qi >= 0) {
                    if (n == 0) r = 0.0; /* empty stream Quantile -> 0 */
                    else if (n <= 3) { /* quantilesFromBuf :210-229 */
                        int idx = (int)(plan.qs[qi] * (double)n);
                        if (idx >= (int)n) idx = n - 1;
                        r = sorted[idx];
                    } else {
                        /* calcQuantiles walk closed form over the sorted
                         * unique quantile list (one emission per sample) */
                        int k = 0;
                        for (int i = 0; i <= qi; i++) {
                            int rank = (int)ceil(plan.qs[i] * (double)n);
                            k = (i == 0) ? rank : ((rank > k + 1) ? rank : k + 1);
                        }
                        if (k > (int)n) k = n;
                        r = sorted[k - 1];
                    }
                } else {
                    switch (t) {
                    case M3GPU_AGG_MIN: r = n ? sorted[0] : 0.0; break;   /* Quantile(0) */
                    case M3GPU_AGG_MAX: r = n ? sorted[n - 1] : 0.0; break;
                    case M3GPU_AGG_MEAN: r = bs.count ? bs.fsum / (double)bs.count : 0.0; break;
                    case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                    case M3GPU_AGG_SUM: r = bs.fsum; break;
                    case M3GPU_AGG_SUMSQ: r = bs.fsumsq; break;
                    case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, bs.fsumsq, bs.fsum); break;
                    default: r = 0; break;
                    }
                }
            }
            out_row[(uint64_t)b * plan.naggs + lane] = r;
        }
    };

    for (;;) {
        int64_t t;
        double v;
        int rstat = d.next(&t, &v);
        if (rstat < 0) { err = -rstat; break; }
        bool have = rstat == 1;
        int64_t b = -1;
        if (have) {
            if (cur_bucket < 0) base = (t / window_ns) * window_ns; /* Truncate */
            if (t < base) { err = M3GPU_SERIES_UNSORTED; break; }
            b = (t - base) / window_ns;
            if (b >= (int64_t)nbuckets) { err = M3GPU_SERIES_CAPACITY; break; }
            if (b < cur_bucket) { err = M3GPU_SERIES_UNSORTED; break; }
        }
        if (!have || b != cur_bucket) {
            if (cur_bucket >= 0) emit_bucket(cur_bucket);
            /* emit empty buckets in any gap (and the tail after the stream) */
            int64_t stop = have ? b : (int64_t)nbuckets;
            for (int64_t eb = (cur_bucket < 0 ? 0 : cur_bucket + 1); eb < stop; eb++) {
                bs.reset();
                nq_in_bucket = 0;
                emit_bucket(eb);
            }
            if (!have) break;
            bs.reset();
            nq_in_bucket = 0;
            cur_bucket = b;
        }
        /* accumulate */
        if (metric_type == M3GPU_METRIC_COUNTER) { /* counter.go:52-78 */
            int64_t iv = go_f2i(v);
            bs.isum += iv;
            bs.count++;
            if (bs.imax < iv) bs.imax = iv;
            if (bs.imin > iv) bs.imin = iv;
            bs.isumsq += iv * iv;
        } else if (metric_type == M3GPU_METRIC_GAUGE) { /* gauge.go:73-103 */
            if (bs.last_at == 0 || t > bs.last_at) { bs.last_at = t; bs.last = v; }
            bs.count++;
            if (!isnan(v)) {
                bs.fsum += v;
                if (isnan(bs.fmax) || bs.fmax < v) bs.fmax = v;
                if (isnan(bs.fmin) || bs.fmin > v) bs.fmin = v;
                bs.fsumsq += v * v;
            }
        } else { /* timer.go:56-75 */
            bs.count++;
            bs.fsum += v;
            bs.fsumsq += v * v;
            if (plan.nq > 0) {
                if (nq_in_bucket >= (uint32_t)plan.exact_cap ||
                    nq_in_bucket >= QCAP) {
                    err = M3GPU_SERIES_BUCKET_OVERFLOW;
                    break;
                }
                if (lane == 0) qvals[nq_in_bucket] = v;
                nq_in_bucket++;
            }
        }
    }
    if (lane == 0) out_errs[series] = err;
}

/* ===================== blob regather (layout pass) ===================== */
/* Physically reorders packed streams: dst series i <- src series perm[i].
 * Used to lay the blob out in wave-schedule order (length-sorted), so the
 * 64 lanes of each decoding wavefront read NEIGHBORING memory instead of
 * scattered streams — the L2-locality companion to the perm scheduling.
 * One series per wavefront, u64 copies (offsets are 16B aligned). */
__global__ void __launch_bounds__(BLOCK_THREADS)
k_regather(const uint8_t* __restrict__ src, const uint64_t* __restrict__ src_offsets,
           const uint32_t* __restrict__ lens, const int32_t* __restrict__ perm,
           const uint64_t* __restrict__ dst_offsets, uint32_t nseries,
           uint8_t* __restrict__ dst) {
    const uint32_t wave = __builtin_amdgcn_readfirstlane(threadIdx.x / WAVE);
    const uint32_t lane = threadIdx.x % WAVE;
    const uint32_t i = blockIdx.x * WAVES_PER_BLOCK + wave;
    if (i >= nseries) return;
    const uint32_t s = (uint32_t)perm[i];
    const uint64_t* sp = (const uint64_t*)(src + src_offsets[s]);
    uint64_t* dp = (uint64_t*)(dst + dst_offsets[i]);
    uint32_t nwords = (lens[s] + 7) / 8;
    for (uint32_t w = lane; w < nwords; w += WAVE) dp[w] = sp[w];
}

/* ===================== stream compaction kernel ===================== */
/* Packs strided encoder output rows into the tight 8B-aligned blob layout
 * (m3gpu.h): one series per wavefront, u64 copies, coalesced within a row. */
__global__ void __launch_bounds__(BLOCK_THREADS)
k_compact(const uint8_t* __restrict__ src, uint32_t src_stride,
          const uint32_t* __restrict__ lens,
          const uint64_t* __restrict__ dst_offsets, uint32_t nseries,
          uint8_t* __restrict__ dst) {
    /* readfirstlane makes the series index provably wave-uniform: the
     * whole parser then compiles to scalar (SGPR) code with scalar
     * branches instead of exec-mask divergence sequences. */
    const uint32_t wave = __builtin_amdgcn_readfirstlane(threadIdx.x / WAVE);
    const uint32_t lane = threadIdx.x % WAVE;
    const uint32_t series = blockIdx.x * WAVES_PER_BLOCK + wave;
    if (series >= nseries) return;
    const uint64_t* s = (const uint64_t*)(src + (uint64_t)series * src_stride);
    uint64_t* d = (uint64_t*)(dst + dst_offsets[series]);
    uint32_t nwords = (lens[series] + 7) / 8;
    for (uint32_t w = lane; w < nwords; w += WAVE) d[w] = s[w];
}


/* -------- series-per-lane fused rollup --------
 * Same semantics as k_rollup_batch but one series per LANE (64 parsers per
 * wave, like k_decode_batch). Timer quantile values stage in a per-lane LDS
 * row kept sorted by insertion (bucket sizes <= RQCAP_LANE); larger buckets
 * flag M3GPU_SERIES_BUCKET_OVERFLOW and the host retries those series on the
 * wave-per-series kernel (64-deep staging). */
#define RQCAP_LANE 16

#define RQ_NONE 0
#define RQ_STAGED 1
#define RQ_EXTREME 2

template <int QMODE, int METRIC>
__global__ void __launch_bounds__(BLOCK_THREADS, QMODE == RQ_STAGED ? 3 : 4)
/* staged: 40KB LDS (qbuf + ring) pins 3 blocks/CU, so 3 waves/SIMD with no
   spills; other modes target the 128-VGPR granule (4 waves/SIMD) */
k_rollup_lane(const uint8_t* __restrict__ blobs,
              const uint64_t* __restrict__ offsets,
              const uint32_t* __restrict__ lens,
              uint32_t nseries, int int_optimized, uint8_t default_unit,
              int64_t window_ns, uint32_t nbuckets,
              RollupPlan plan,
              double* __restrict__ out, int64_t* __restrict__ out_window_ts,
              int32_t* __restrict__ out_errs) {
    const uint32_t lane = threadIdx.x & (WAVE - 1);
    const uint32_t wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const uint32_t series = blockIdx.x * BLOCK_THREADS + wave * WAVE + lane;

    constexpr bool WITH_Q = QMODE == RQ_STAGED;
    __shared__ double qbuf_all[WITH_Q ? WAVES_PER_BLOCK : 1]
                              [WITH_Q ? WAVE : 1][WITH_Q ? RQCAP_LANE : 1];
    double* qrow = WITH_Q ? qbuf_all[wave][lane] : nullptr;
    /* per-lane input ring, as in k_decode_batch */
    __shared__ uint64_t rring_all[WAVES_PER_BLOCK][WAVE][IN_STRIDE];
    uint64_t* rring = rring_all[wave][lane];

    const bool in_range = series < nseries;
    Decoder d;
    d.init(blobs, in_range ? offsets[series] : 0, in_range ? lens[series] : 0,
           int_optimized != 0, default_unit, rring);

    double* out_row = out + (uint64_t)series * nbuckets * plan.naggs;
    int64_t* wts_row = out_window_ts ? out_window_ts + (uint64_t)series * nbuckets : nullptr;

    BucketState bs;
    bs.reset();
    int64_t base = 0;
    int64_t cur_bucket = -1;
    int64_t cur_lo = 0, cur_hi = 0; /* current bucket ts bounds */
    uint32_t nq = 0;
    int err = 0;
    bool running = in_range;

    auto emit_bucket = [&](int64_t b) {
        if (b < 0 || b >= (int64_t)nbuckets) return;
        if (wts_row) wts_row[b] = base + (b + 1) * window_ns;
        /* even naggs: emit value pairs as aligned 16B stores (the rows are
         * naggs*8B apart, so paired stores halve the line-granular write
         * amplification of the scattered per-lane emission) */
        const bool paired = (plan.naggs & 1) == 0;
        double pend = 0;
        for (int k = 0; k < plan.naggs; k++) {
            int32_t t = plan.agg_types[k];
            int8_t qi = plan.qidx[k];
            double r = 0;
            if (METRIC == M3GPU_METRIC_COUNTER) { /* counter.go:112-131 */
                switch (t) {
                case M3GPU_AGG_MIN: r = (double)bs.imin; break;
                case M3GPU_AGG_MAX: r = (double)bs.imax; break;
                case M3GPU_AGG_MEAN: r = bs.count ? (double)bs.isum / (double)bs.count : 0; break;
                case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                case M3GPU_AGG_SUM: r = (double)bs.isum; break;
                case M3GPU_AGG_SUMSQ: r = (double)bs.isumsq; break;
                case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, (double)bs.isumsq, (double)bs.isum); break;
                default: break;
                }
            } else if (METRIC == M3GPU_METRIC_GAUGE) { /* gauge.go:144-165 */
                switch (t) {
                case M3GPU_AGG_LAST: r = bs.last; break;
                case M3GPU_AGG_MIN: r = bs.fmin; break;
                case M3GPU_AGG_MAX: r = bs.fmax; break;
                case M3GPU_AGG_MEAN: r = bs.count ? bs.fsum / (double)bs.count : 0.0; break;
                case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                case M3GPU_AGG_SUM: r = bs.fsum; break;
                case M3GPU_AGG_SUMSQ: r = bs.fsumsq; break;
                case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, bs.fsumsq, bs.fsum); break;
                default: break;
                }
            } else { /* timer.go:131-153; qrow holds the sorted bucket values */
                if (QMODE == RQ_EXTREME && qi >= 0) {
                    /* all-high quantile set: every requested quantile
                     * resolves to the bucket max for n <= plan.extreme_cap
                     * (host-verified against the walk AND quantilesFromBuf;
                     * deeper buckets took the overflow path) */
                    r = bs.count ? bs.fmax : 0.0;
                } else if (WITH_Q && qi >= 0) {
                    if (nq == 0) r = 0.0; /* empty stream Quantile -> 0 */
                    else if (nq <= 3) { /* quantilesFromBuf :210-229 */
                        uint32_t idx = (uint32_t)(plan.qs[qi] * (double)nq);
                        if (idx >= nq) idx = nq - 1;
                        r = qrow[idx];
                    } else { /* calcQuantiles walk closed form */
                        int kk = 0;
                        for (int i = 0; i <= qi; i++) {
                            int rank = (int)ceil(plan.qs[i] * (double)nq);
                            kk = (i == 0) ? rank : ((rank > kk + 1) ? rank : kk + 1);
                        }
                        if (kk > (int)nq) kk = nq;
                        r = qrow[kk - 1];
                    }
                } else {
                    switch (t) {
                    case M3GPU_AGG_MIN:
                        r = (QMODE == RQ_EXTREME)
                                ? (bs.count ? bs.fmin : 0.0)
                                : ((WITH_Q && nq) ? qrow[0] : 0.0);
                        break;
                    case M3GPU_AGG_MAX:
                        r = (QMODE == RQ_EXTREME)
                                ? (bs.count ? bs.fmax : 0.0)
                                : ((WITH_Q && nq) ? qrow[nq - 1] : 0.0);
                        break;
                    case M3GPU_AGG_MEAN: r = bs.count ? bs.fsum / (double)bs.count : 0.0; break;
                    case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                    case M3GPU_AGG_SUM: r = bs.fsum; break;
                    case M3GPU_AGG_SUMSQ: r = bs.fsumsq; break;
                    case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, bs.fsumsq, bs.fsum); break;
                    default: break;
                    }
                }
            }
            if (paired) {
                if (k & 1) {
                    double2 pr;
                    pr.x = pend;
                    pr.y = r;
                    *(double2*)(out_row + (uint64_t)b * plan.naggs + k - 1) = pr;
                } else {
                    pend = r;
                }
            } else {
                out_row[(uint64_t)b * plan.naggs + k] = r;
            }
        }
    };

    while (__any(running)) {
        d.r.refill_issue(running);
#pragma unroll 1
        for (uint32_t jt = 0; jt < DEC_TILE; jt++) {
        if (running) {
            int64_t t;
            double v;
            int rstat = d.next(&t, &v);
            if (rstat < 0) {
                err = -rstat;
                running = false;
            } else {
                bool have = rstat == 1;
                int64_t b = -1;
                bool ok = true;
                if (have) {
                    if (cur_bucket >= 0 && t >= cur_lo && t < cur_hi) {
                        /* same bucket: two compares instead of a 64-bit
                         * division per point */
                        b = cur_bucket;
                    } else {
                        if (cur_bucket < 0) base = (t / window_ns) * window_ns; /* Truncate */
                        if (t < base) { err = M3GPU_SERIES_UNSORTED; running = false; ok = false; }
                        else {
                            b = (t - base) / window_ns;
                            if (b >= (int64_t)nbuckets) { err = M3GPU_SERIES_CAPACITY; running = false; ok = false; }
                            else if (b < cur_bucket) { err = M3GPU_SERIES_UNSORTED; running = false; ok = false; }
                        }
                    }
                }
                if (ok && (!have || b != cur_bucket)) {
                    if (cur_bucket >= 0) emit_bucket(cur_bucket);
                    int64_t stop = have ? b : (int64_t)nbuckets;
                    for (int64_t eb = (cur_bucket < 0 ? 0 : cur_bucket + 1); eb < stop; eb++) {
                        bs.reset();
                        nq = 0;
                        emit_bucket(eb);
                    }
                    if (!have) { running = false; ok = false; }
                    else {
                        bs.reset();
                        nq = 0;
                        cur_bucket = b;
                        cur_lo = base + b * window_ns;
                        cur_hi = cur_lo + window_ns;
                    }
                }
                if (ok) {
                    if (METRIC == M3GPU_METRIC_COUNTER) { /* counter.go:52-78 */
                        int64_t iv = go_f2i(v);
                        bs.isum += iv;
                        bs.count++;
                        if (bs.imax < iv) bs.imax = iv;
                        if (bs.imin > iv) bs.imin = iv;
                        bs.isumsq += iv * iv;
                    } else if (METRIC == M3GPU_METRIC_GAUGE) { /* gauge.go:73-103 */
                        if (bs.last_at == 0 || t > bs.last_at) { bs.last_at = t; bs.last = v; }
                        bs.count++;
                        if (!isnan(v)) {
                            bs.fsum += v;
                            if (isnan(bs.fmax) || bs.fmax < v) bs.fmax = v;
                            if (isnan(bs.fmin) || bs.fmin > v) bs.fmin = v;
                            bs.fsumsq += v * v;
                        }
                    } else { /* timer.go:56-75 */
                        if (QMODE == RQ_EXTREME &&
                            (isnan(v) ||
                             bs.count >= (int64_t)plan.extreme_cap)) {
                            /* NaN ordering and deep buckets keep the exact
                             * staged semantics: retry on the wave tier */
                            err = M3GPU_SERIES_BUCKET_OVERFLOW;
                            running = false;
                        } else {
                            if (QMODE == RQ_EXTREME) {
                                if (bs.count == 0) {
                                    bs.fmin = v;
                                    bs.fmax = v;
                                } else {
                                    if (v < bs.fmin) bs.fmin = v;
                                    if (v > bs.fmax) bs.fmax = v;
                                }
                            }
                            bs.count++;
                            bs.fsum += v;
                            bs.fsumsq += v * v;
                            if (WITH_Q && plan.nq > 0) {
                                if (nq >= RQCAP_LANE ||
                                    nq >= (uint32_t)plan.exact_cap) {
                                    err = M3GPU_SERIES_BUCKET_OVERFLOW;
                                    running = false;
                                    bs.count--; /* not accumulated */
                                } else {
                                    /* insertion into this lane's sorted row */
                                    uint32_t j = nq;
                                    while (j > 0 && qrow[j - 1] > v) { qrow[j] = qrow[j - 1]; j--; }
                                    qrow[j] = v;
                                    nq++;
                                }
                            }
                        }
                    }
                }
            }
        }
        }
        d.r.refill_commit();
    }
    if (in_range) out_errs[series] = err;
}


__global__ void __launch_bounds__(BLOCK_THREADS)
k_count_errcode(const int32_t* __restrict__ errs, uint32_t n, int32_t code,
                uint32_t* __restrict__ out) {
    /* grid-stride count of one per-series error code: lets the rollup
     * dispatch test for tier retries with a 4-byte D2H instead of pulling
     * and scanning the whole error array every call */
    uint32_t cnt = 0;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x)
        if (errs[i] == code) cnt++;
    if (__any(cnt != 0)) {
        cnt += __shfl_down(cnt, 32); cnt += __shfl_down(cnt, 16);
        cnt += __shfl_down(cnt, 8); cnt += __shfl_down(cnt, 4);
        cnt += __shfl_down(cnt, 2); cnt += __shfl_down(cnt, 1);
        if ((threadIdx.x & (WAVE - 1)) == 0 && cnt) atomicAdd(out, cnt);
    }
}

/* ===================== replica-deduplicating merge =====================
 * MultiReaderIterator semantics over one slice of R (<=4) replica rows of
 * already-decoded points (multi_reader_iterator.go:62-155 +
 * iterators.go:56-237, IterateLastPushed default): the `values` order starts
 * as replica order and mutates only by swap-with-tail on exhaustion; among
 * equal earliest timestamps the LAST in `values` order wins; equal-to-prev
 * timestamps are deduped; a decreasing one is errOutOfOrderIterator.
 * One series per lane; order bookkeeping packed in a u32 (a byte per slot)
 * so everything stays in registers. */
#define MERGE_MAX_R 4
#define MERGE_ERR_OUT_OF_ORDER 100

__global__ void __launch_bounds__(BLOCK_THREADS)
k_merge(const int64_t* __restrict__ ts, const double* __restrict__ vals,
        const uint32_t* __restrict__ counts, uint32_t nreplicas,
        uint32_t nseries, uint32_t stride,
        int64_t* __restrict__ out_ts, double* __restrict__ out_vals,
        uint32_t* __restrict__ out_counts, int32_t* __restrict__ out_errs,
        uint32_t out_stride) {
    const uint32_t series = blockIdx.x * BLOCK_THREADS + threadIdx.x;
    if (series >= nseries) return;

    const int64_t* tr[MERGE_MAX_R];
    const double* vr[MERGE_MAX_R];
    uint32_t cnt[MERGE_MAX_R];
    uint32_t cur[MERGE_MAX_R];
    int64_t cur_ts[MERGE_MAX_R];
    double cur_val[MERGE_MAX_R];

    uint32_t ord = 0; /* packed replica ids, byte per slot, slot 0 = lowest */
    uint32_t nvals = 0;
#pragma unroll
    for (uint32_t r = 0; r < MERGE_MAX_R; r++) {
        if (r < nreplicas) {
            tr[r] = ts + ((uint64_t)r * nseries + series) * stride;
            vr[r] = vals + ((uint64_t)r * nseries + series) * stride;
            cnt[r] = counts[(uint64_t)r * nseries + series];
            cur[r] = 0;
            if (cnt[r] > 0) {
                cur_ts[r] = tr[r][0];
                cur_val[r] = vr[r][0];
                ord |= r << (8 * nvals);
                nvals++;
            }
        }
    }

    int64_t* orow_ts = out_ts + (uint64_t)series * out_stride;
    double* orow_val = out_vals + (uint64_t)series * out_stride;
    uint32_t n = 0;
    int64_t prev_at = 0;
    bool first = true;
    int err = 0;

    while (nvals > 0) {
        /* earliest scan in `ord` order; last min wins (IterateLastPushed) */
        int64_t earliest = INT64_MAX;
        uint32_t winner = 0;
#pragma unroll
        for (uint32_t i = 0; i < MERGE_MAX_R; i++) {
            if (i < nvals) {
                uint32_t r = (ord >> (8 * i)) & 0xff;
                int64_t t = cur_ts[r];
                if (t <= earliest) {
                    earliest = t;
                    winner = r;
                }
            }
        }
        if (!first && earliest < prev_at) { err = MERGE_ERR_OUT_OF_ORDER; break; }
        if (first || earliest != prev_at) {
            if (n >= out_stride) { err = M3GPU_SERIES_CAPACITY; break; }
            orow_ts[n] = earliest;
            orow_val[n] = cur_val[winner];
            n++;
            prev_at = earliest;
            first = false;
        }
        /* advance every replica at `earliest`; swap-with-tail on exhaustion
         * (ascending slot order with re-examination == the reference's
         * removal order, see DESIGN.md merge note) */
        for (uint32_t i = 0; i < nvals;) {
            uint32_t r = (ord >> (8 * i)) & 0xff;
            if (cur_ts[r] == earliest) {
                cur[r]++;
                if (cur[r] >= cnt[r]) {
                    uint32_t tail = (ord >> (8 * (nvals - 1))) & 0xff;
                    ord = (ord & ~(0xffu << (8 * i))) | (tail << (8 * i));
                    nvals--;
                    continue; /* re-examine the swapped-in slot */
                }
                cur_ts[r] = tr[r][cur[r]];
                cur_val[r] = vr[r][cur[r]];
            }
            i++;
        }
    }
    out_counts[series] = n;
    out_errs[series] = err;
}


/* ===================== full-CKMS deep-bucket rollup =====================
 * Third tier of the timer-quantile chain (after the per-lane <=16 staging
 * and the wave-kernel exact-order-statistics cap): the reference CKMS
 * stream WITH compression (quantile/cm/stream.go:77-429), ported from the
 * oracle's restatement onto LDS arrays. One WAVE per workgroup, one series
 * per workgroup (rare retry path; correctness, not throughput):
 *  - samples double-buffer (value,numRanks,delta) replaces the linked list;
 *    the insert cursor walk becomes a two-pointer merge (equivalent:
 *    incoming v inserted before the first list element with value >= v,
 *    delta = that element's numRanks+delta-1; the tail appends delta 0);
 *  - compress walks backward tracking the surviving next element; removed
 *    samples are compacted after the walk (same removal order);
 *  - in the AddBatch(1)/Flush usage the compress cursor is always nil on
 *    insert entry, so the compValue/compressMinRank-in-insert branch of the
 *    reference is dead (stream.go:284-287,303) and omitted;
 *  - buffers bufLess/bufMore are value MULTISETS here (the reference's heap
 *    order only permutes ties, which insert() cannot distinguish);
 *  - capacity overflow (sample list or buffer > CKMS_CAP) flags
 *    M3GPU_SERIES_BUCKET_OVERFLOW — never an approximation. */
#define CKMS_CAP 3072
#define CKMS_BUF 2080 /* bufMore peak = carried bufLess (<=1024) + 1024 new */

struct CkmsLds {
    double* val[2];
    int32_t* nr[2];
    int32_t* dl[2];
    double* less;
    double* more;
};

struct CkmsState {
    int cur;           /* active buffer index */
    int32_t len;       /* samples */
    int32_t nless, nmore;
    int64_t num_values;
    int32_t counter;   /* insertAndCompressCounter */
    int err;
};

__device__ __forceinline__ void ckms_reset(CkmsState& st) {
    st.cur = 0;
    st.len = 0;
    st.nless = 0;
    st.nmore = 0;
    st.num_values = 0;
    st.counter = 0;
}

/* wave-parallel ascending sort of buf[0..n) into tmp[0..n) (rank-select) */
__device__ __forceinline__ void ckms_sort(const double* buf, double* tmp,
                                          int32_t n, uint32_t lane) {
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    for (int32_t e = lane; e < n; e += WAVE) {
        double v = buf[e];
        int32_t rank = 0;
        for (int32_t j = 0; j < n; j++) {
            double o = buf[j];
            rank += (o < v) || (o == v && j < e);
        }
        tmp[rank] = v;
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
}

/* stream.go:362-377 threshold(rank), exact int64 truncation */
__device__ __forceinline__ int64_t ckms_threshold(const RollupPlan& plan,
                                                  int64_t num_vals, int64_t rank) {
    int64_t min_val = INT64_MAX;
    double eps = 2.0 * plan.eps;
    for (int i = 0; i < plan.nq; i++) {
        int64_t qmin;
        if (rank >= (int64_t)(plan.qs[i] * (double)num_vals))
            qmin = (int64_t)(eps * (double)rank / plan.qs[i]);
        else
            qmin = (int64_t)(eps * (double)(num_vals - rank) / (1.0 - plan.qs[i]));
        if (qmin < min_val) min_val = qmin;
    }
    return min_val;
}

/* insert sorted batch (stream.go:280-331) + compress (:333-401); lane 0
 * does the sequential merge/walk, the wave sorts. `sorted` is dedicated LDS scratch. */
__device__ void ckms_insert_compress(CkmsLds& L, CkmsState& st,
                                     const RollupPlan& plan, double* sorted,
                                     uint32_t lane) {
    int32_t m = st.nmore;
    ckms_sort(L.more, sorted, m, lane);
    if (st.len + m > CKMS_CAP) { /* uniform: all lanes see the same state */
        st.err = M3GPU_SERIES_BUCKET_OVERFLOW;
        return;
    }
    if (lane == 0) {
        int src = st.cur, dst = st.cur ^ 1;
        int32_t n = st.len;
        {
            /* two-pointer merge == the reference's cursor walk */
            int32_t i = 0, j = 0, o = 0;
            while (i < n) {
                if (j < m && sorted[j] <= L.val[src][i]) {
                    L.val[dst][o] = sorted[j];
                    L.nr[dst][o] = 1;
                    L.dl[dst][o] = L.nr[src][i] + L.dl[src][i] - 1;
                    o++; j++;
                } else {
                    L.val[dst][o] = L.val[src][i];
                    L.nr[dst][o] = L.nr[src][i];
                    L.dl[dst][o] = L.dl[src][i];
                    o++; i++;
                }
            }
            while (j < m) { /* PushBack tail, delta 0 (:316-328) */
                L.val[dst][o] = sorted[j];
                L.nr[dst][o] = 1;
                L.dl[dst][o] = 0;
                o++; j++;
            }
            st.cur = dst;
            st.len = o;
            st.num_values += m;
            /* compress (:333-401); entry cursor nil in this usage */
            int32_t len = st.len;
            if (len >= 3) {
                int b = st.cur;
                int64_t cmr = st.num_values - 1 - L.nr[b][len - 2];
                int32_t nxt = len - 2; /* surviving element after curr */
                for (int32_t c = len - 3; c >= 1; c--) {
                    int64_t max_rank = cmr + L.nr[b][c] + L.dl[b][c];
                    int64_t thr = ckms_threshold(plan, st.num_values, max_rank);
                    cmr -= L.nr[b][c];
                    int64_t test_val = L.nr[b][c] + L.nr[b][nxt] + L.dl[b][nxt];
                    if (test_val <= thr) {
                        L.nr[b][nxt] += L.nr[b][c];
                        L.nr[b][c] = -1; /* removed mark */
                    } else {
                        nxt = c;
                    }
                }
                /* compact, preserving order */
                int32_t o2 = 0;
                for (int32_t c = 0; c < len; c++) {
                    if (L.nr[b][c] >= 0) {
                        if (o2 != c) {
                            L.val[b][o2] = L.val[b][c];
                            L.nr[b][o2] = L.nr[b][c];
                            L.dl[b][o2] = L.dl[b][c];
                        }
                        o2++;
                    }
                }
                st.len = o2;
            }
        }
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    /* uniform parts every lane must perform itself: the buffer swap
     * (resetInsertCursor, stream.go:425-429) swaps POINTERS held per lane */
    double* t = L.less; L.less = L.more; L.more = t;
    st.nmore = st.nless;
    st.nless = 0;
    /* lane-0-computed scalars: broadcast via readfirstlane */
    st.cur = __builtin_amdgcn_readfirstlane(st.cur);
    st.len = __builtin_amdgcn_readfirstlane(st.len);
    st.num_values = ((int64_t)(uint32_t)__builtin_amdgcn_readfirstlane(
                         (uint32_t)(st.num_values >> 32)) << 32) |
                    (uint32_t)__builtin_amdgcn_readfirstlane((uint32_t)st.num_values);
}

/* stream.go:77-116 AddBatch(values[0..1)) — one value at a time, exactly
 * like Timer.AddBatch -> stream per-value adds in the rollup loop. */
__device__ __forceinline__ void ckms_add(CkmsLds& L, CkmsState& st,
                                         const RollupPlan& plan, double v,
                                         double* sorted, uint32_t lane) {
    if (st.err) return;
    if (st.len == 0 && st.nmore == 0 && st.nless == 0 && st.num_values == 0) {
        /* first value becomes the first sample (:88-97) */
        if (lane == 0) {
            L.val[st.cur][0] = v;
            L.nr[st.cur][0] = 1;
            L.dl[st.cur][0] = 0;
        }
        st.len = 1;
        st.num_values = 1;
        __builtin_amdgcn_wave_barrier();
        __threadfence_block();
        return;
    }
    double insert_point = L.val[st.cur][0]; /* insertCursor == head here */
    if (st.nless >= CKMS_BUF || st.nmore >= CKMS_BUF) {
        st.err = M3GPU_SERIES_BUCKET_OVERFLOW;
        return;
    }
    if (v < insert_point) {
        if (lane == 0) L.less[st.nless] = v;
        st.nless++;
    } else {
        if (lane == 0) L.more[st.nmore] = v;
        st.nmore++;
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    if (st.counter == plan.every) { /* insertAndCompressEvery, checked BEFORE ++ */
        ckms_insert_compress(L, st, plan, sorted, lane);
        st.counter = 0;
    }
    st.counter++;
}

/* stream.go:123-137 Flush */
__device__ __forceinline__ void ckms_flush(CkmsLds& L, CkmsState& st,
                                           const RollupPlan& plan,
                                           double* sorted, uint32_t lane) {
    while (!st.err && (st.nless > 0 || st.nmore > 0)) {
        if (st.nmore == 0) { /* resetInsertCursor: swap */
            double* t = L.less; L.less = L.more; L.more = t;
            st.nmore = st.nless;
            st.nless = 0;
        }
        ckms_insert_compress(L, st, plan, sorted, lane);
    }
}

/* stream.go:231-277 calcQuantiles (+ :210-229 quantilesFromBuf), filling
 * computed[0..nq) for the plan's sorted unique quantile list. computed[]
 * must be zeroed per bucket first: the reference leaves un-emitted
 * quantiles at the zero value (the post-loop condition can fail for the
 * top quantiles at certain numValues — real behavior, preserved). Lane 0
 * only; thr_rank/thr_thresh are LDS scratch (>= plan.nq entries). */
__device__ void ckms_calc_quantiles(const CkmsLds& L, const CkmsState& st,
                                    const RollupPlan& plan, double* computed,
                                    int64_t* thr_rank, int64_t* thr_thresh) {
    if (plan.nq == 0 || st.num_values == 0) return;
    const int b = st.cur;
    if (st.num_values <= 3) { /* quantilesFromBuf over the sample list */
        for (int i = 0; i < plan.nq; i++) {
            int32_t idx = (int32_t)(plan.qs[i] * (double)st.len);
            if (idx >= st.len) idx = st.len - 1;
            computed[i] = L.val[b][idx];
        }
        return;
    }
    for (int i = 0; i < plan.nq; i++) {
        int64_t rank = (int64_t)ceil(plan.qs[i] * (double)st.num_values);
        thr_rank[i] = rank;
        thr_thresh[i] =
            (int64_t)ceil((double)ckms_threshold(plan, st.num_values, rank) / 2.0);
    }
    int64_t min_rank = 0, max_rank = 0;
    int idx = 0;
    int32_t prev = 0;
    for (int32_t c = 0; c < st.len && idx < plan.nq; c++) {
        max_rank = min_rank + L.nr[b][c] + L.dl[b][c];
        if (max_rank > thr_rank[idx] + thr_thresh[idx] || min_rank > thr_rank[idx]) {
            computed[idx] = L.val[b][prev];
            idx++;
        }
        min_rank += L.nr[b][c];
        prev = c;
    }
    for (int i = idx; i < plan.nq; i++) { /* post-loop (:268-276), note >= */
        if (max_rank >= thr_rank[i] + thr_thresh[i] || min_rank > thr_rank[i])
            computed[i] = L.val[b][prev];
    }
}

/* ------------- the deep-bucket rollup kernel (third tier) -------------
 * One series per 64-thread workgroup (one wave), full 150 KB dynamic LDS
 * for the CKMS state. Only TIMER series with quantile aggs can reach this
 * tier (nothing else sets BUCKET_OVERFLOW). Decode/bucket walk identical
 * to k_rollup_batch. */
#define CKMS_BLOCK 64

__global__ void __launch_bounds__(CKMS_BLOCK)
k_rollup_ckms(const uint8_t* __restrict__ blobs,
              const uint64_t* __restrict__ offsets,
              const uint32_t* __restrict__ lens,
              const int32_t* __restrict__ select, uint32_t nseries,
              int int_optimized, uint8_t default_unit,
              int64_t window_ns, uint32_t nbuckets, RollupPlan plan,
              double* __restrict__ out, int64_t* __restrict__ out_window_ts,
              int32_t* __restrict__ out_errs) {
    extern __shared__ uint8_t lds_raw[];
    const uint32_t lane = threadIdx.x % WAVE;
    const uint32_t slot = blockIdx.x;
    if (slot >= nseries) return;
    const uint32_t series = __builtin_amdgcn_readfirstlane(
        select ? (uint32_t)select[slot] : slot);

    /* carve the dynamic LDS (layout mirrored by rollup_ckms_lds_bytes) */
    uint8_t* p = lds_raw;
    CkmsLds L;
    L.val[0] = (double*)p; p += CKMS_CAP * 8;
    L.val[1] = (double*)p; p += CKMS_CAP * 8;
    L.nr[0] = (int32_t*)p; p += CKMS_CAP * 4;
    L.nr[1] = (int32_t*)p; p += CKMS_CAP * 4;
    L.dl[0] = (int32_t*)p; p += CKMS_CAP * 4;
    L.dl[1] = (int32_t*)p; p += CKMS_CAP * 4;
    L.less = (double*)p; p += CKMS_BUF * 8;
    L.more = (double*)p; p += CKMS_BUF * 8;
    double* sorted = (double*)p; p += CKMS_BUF * 8;
    double* computed = (double*)p; p += MAX_AGGS * 8;
    int64_t* thr_rank = (int64_t*)p; p += MAX_AGGS * 8;
    int64_t* thr_thresh = (int64_t*)p; p += MAX_AGGS * 8;

    Decoder d;
    d.init(blobs, offsets[series], lens[series], int_optimized != 0, default_unit);

    double* out_row = out + (uint64_t)series * nbuckets * plan.naggs;
    int64_t* wts_row = out_window_ts ? out_window_ts + (uint64_t)series * nbuckets : nullptr;

    BucketState bs;
    bs.reset();
    CkmsState st;
    ckms_reset(st);
    st.err = 0;
    int64_t base = 0;
    int64_t cur_bucket = -1;
    int err = 0;

    auto emit_bucket = [&](int64_t bk) {
        if (bk < 0 || bk >= (int64_t)nbuckets) return;
        if (wts_row && lane == 0) wts_row[bk] = base + (bk + 1) * window_ns;
        ckms_flush(L, st, plan, sorted, lane);
        if (st.err) return;
        if (lane == 0) {
            for (int i = 0; i < plan.nq; i++) computed[i] = 0.0;
            ckms_calc_quantiles(L, st, plan, computed, thr_rank, thr_thresh);
        }
        __builtin_amdgcn_wave_barrier();
        __threadfence_block();
        if (lane < (uint32_t)plan.naggs) { /* timer.go:131-153 ValueOf */
            int32_t t = plan.agg_types[lane];
            int8_t qi = plan.qidx[lane];
            double r = 0;
            int cb = st.cur;
            if (qi >= 0) {
                r = st.len ? computed[qi] : 0.0; /* Quantile(q): computed */
            } else {
                switch (t) {
                case M3GPU_AGG_MIN: r = st.len ? L.val[cb][0] : 0.0; break;
                case M3GPU_AGG_MAX: r = st.len ? L.val[cb][st.len - 1] : 0.0; break;
                case M3GPU_AGG_MEAN: r = bs.count ? bs.fsum / (double)bs.count : 0.0; break;
                case M3GPU_AGG_COUNT: r = (double)bs.count; break;
                case M3GPU_AGG_SUM: r = bs.fsum; break;
                case M3GPU_AGG_SUMSQ: r = bs.fsumsq; break;
                case M3GPU_AGG_STDEV: r = m3_stdev(bs.count, bs.fsumsq, bs.fsum); break;
                default: r = 0; break;
                }
            }
            out_row[(uint64_t)bk * plan.naggs + lane] = r;
        }
    };

    for (;;) {
        int64_t t;
        double v;
        int rstat = d.next(&t, &v);
        if (rstat < 0) { err = -rstat; break; }
        bool have = rstat == 1;
        int64_t b = -1;
        if (have) {
            if (cur_bucket < 0) base = (t / window_ns) * window_ns;
            if (t < base) { err = M3GPU_SERIES_UNSORTED; break; }
            b = (t - base) / window_ns;
            if (b >= (int64_t)nbuckets) { err = M3GPU_SERIES_CAPACITY; break; }
            if (b < cur_bucket) { err = M3GPU_SERIES_UNSORTED; break; }
        }
        if (!have || b != cur_bucket) {
            if (cur_bucket >= 0) {
                emit_bucket(cur_bucket);
                if (st.err) { err = st.err; break; }
            }
            int64_t stop = have ? b : (int64_t)nbuckets;
            for (int64_t eb = (cur_bucket < 0 ? 0 : cur_bucket + 1); eb < stop; eb++) {
                bs.reset();
                ckms_reset(st);
                emit_bucket(eb);
            }
            if (!have) break;
            bs.reset();
            ckms_reset(st);
            cur_bucket = b;
        }
        /* timer.go:56-75 AddBatch, one value at a time */
        bs.count++;
        bs.fsum += v;
        bs.fsumsq += v * v;
        ckms_add(L, st, plan, v, sorted, lane);
        if (st.err) { err = st.err; break; }
    }
    if (lane == 0) out_errs[series] = err;
}

} // namespace m3

/* ============================ C-ABI host layer ============================ */

#include <mutex>
#include <climits>

static __thread char g_err[512];
static int g_device = 0;
static bool g_inited = false;

static int set_hip_err(const char* what, hipError_t e) {
    snprintf(g_err, sizeof(g_err), "%s: %s", what, hipGetErrorString(e));
    return M3GPU_ERR_HIP;
}
#define HIP_TRY(expr) do { hipError_t _e = (expr); if (_e != hipSuccess) return set_hip_err(#expr, _e); } while (0)

extern "C" {

const char* m3gpu_last_error(void) { return g_err; }

int m3gpu_init(int device) {
    HIP_TRY(hipSetDevice(device));
    g_device = device;
    g_inited = true;
    return M3GPU_OK;
}

void m3gpu_shutdown(void) { g_inited = false; }

/* wave-per-series kernels (rollup, compact): 4 series per block */
static inline uint32_t grid_for(uint32_t nseries) {
    return (nseries + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
}
/* lane-per-series kernels (decode, encode): 256 series per block */
static inline uint32_t grid_lane(uint32_t nseries) {
    return (nseries + BLOCK_THREADS - 1) / BLOCK_THREADS;
}

int m3gpu_decode_batch_dev(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t stride, void* hip_stream) {
    return m3gpu_decode_batch_dev_perm(d_blobs, d_offsets, d_lens, NULL,
                                       nseries, int_optimized, default_unit,
                                       d_out_ts, d_out_vals, d_out_counts,
                                       d_out_errs, stride, hip_stream);
}

int m3gpu_decode_batch_dev_perm(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    const int32_t* d_perm,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t stride, void* hip_stream) {
    if (!nseries) return M3GPU_OK;
    hipStream_t s = (hipStream_t)hip_stream;
    hipLaunchKernelGGL(m3::k_decode_batch, dim3(grid_lane(nseries)),
                       dim3(BLOCK_THREADS), 0, s,
                       d_blobs, d_offsets, d_lens, d_perm, nseries,
                       int_optimized, default_unit, d_out_ts, d_out_vals,
                       d_out_counts, d_out_errs, stride, (uint8_t*)NULL, 0);
    HIP_TRY(hipGetLastError());
    return M3GPU_OK;
}

int m3gpu_decode_batch_dev_ann(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t stride,
    uint8_t* d_out_ann, uint32_t ann_stride, void* hip_stream) {
    if (!nseries) return M3GPU_OK;
    if (ann_stride % 4 || ann_stride < 16) {
        snprintf(g_err, sizeof(g_err),
                 "ann_stride must be 4-byte aligned and >= 16");
        return M3GPU_ERR_BADARG;
    }
    hipStream_t s = (hipStream_t)hip_stream;
    hipLaunchKernelGGL(m3::k_decode_batch, dim3(grid_lane(nseries)),
                       dim3(BLOCK_THREADS), 0, s,
                       d_blobs, d_offsets, d_lens, (const int32_t*)NULL,
                       nseries, int_optimized, default_unit, d_out_ts,
                       d_out_vals, d_out_counts, d_out_errs, stride,
                       d_out_ann, ann_stride);
    HIP_TRY(hipGetLastError());
    return M3GPU_OK;
}

int m3gpu_encode_batch_dev(
    const int64_t* d_ts, const double* d_vals, const uint32_t* d_counts,
    uint32_t nseries, uint32_t stride, int int_optimized, uint8_t unit,
    uint8_t* d_out_bytes, uint32_t out_stride, uint32_t* d_out_lens,
    int32_t* d_out_errs, void* hip_stream) {
    if (!nseries) return M3GPU_OK;
    if (out_stride % 8) {
        snprintf(g_err, sizeof(g_err), "out_stride must be a multiple of 8");
        return M3GPU_ERR_BADARG;
    }
    hipStream_t s = (hipStream_t)hip_stream;
    hipLaunchKernelGGL(m3::k_encode_batch, dim3(grid_lane(nseries)),
                       dim3(BLOCK_THREADS), 0, s,
                       d_ts, d_vals, d_counts, nseries, stride, int_optimized,
                       unit, d_out_bytes, out_stride, d_out_lens, d_out_errs);
    HIP_TRY(hipGetLastError());
    return M3GPU_OK;
}

int m3gpu_compact_dev(
    const uint8_t* d_src, uint32_t src_stride, const uint32_t* d_lens,
    const uint64_t* d_dst_offsets, uint32_t nseries, uint8_t* d_dst,
    void* hip_stream) {
    if (!nseries) return M3GPU_OK;
    hipStream_t s = (hipStream_t)hip_stream;
    hipLaunchKernelGGL(m3::k_compact, dim3(grid_for(nseries)),
                       dim3(BLOCK_THREADS), 0, s,
                       d_src, src_stride, d_lens, d_dst_offsets, nseries, d_dst);
    HIP_TRY(hipGetLastError());
    return M3GPU_OK;
}

int m3gpu_regather_dev(
    const uint8_t* d_src, const uint64_t* d_src_offsets,
    const uint32_t* d_lens, const int32_t* d_perm,
    const uint64_t* d_dst_offsets, uint32_t nseries, uint8_t* d_dst,
    void* hip_stream) {
    if (!nseries) return M3GPU_OK;
    hipStream_t s = (hipStream_t)hip_stream;
    hipLaunchKernelGGL(m3::k_regather, dim3(grid_for(nseries)),
                       dim3(BLOCK_THREADS), 0, s,
                       d_src, d_src_offsets, d_lens, d_perm, d_dst_offsets,
                       nseries, d_dst);
    HIP_TRY(hipGetLastError());
    return M3GPU_OK;
}

static double agg_quantile_of_host(int32_t t) {
    switch (t) {
    case M3GPU_AGG_MEDIAN: case M3GPU_AGG_P50: return 0.5;
    case M3GPU_AGG_P10: return 0.1;
    case M3GPU_AGG_P20: return 0.2;
    case M3GPU_AGG_P25: return 0.25;
    case M3GPU_AGG_P30: return 0.3;
    case M3GPU_AGG_P40: return 0.4;
    case M3GPU_AGG_P60: return 0.6;
    case M3GPU_AGG_P70: return 0.7;
    case M3GPU_AGG_P75: return 0.75;
    case M3GPU_AGG_P80: return 0.8;
    case M3GPU_AGG_P90: return 0.9;
    case M3GPU_AGG_P95: return 0.95;
    case M3GPU_AGG_P99: return 0.99;
    case M3GPU_AGG_P999: return 0.999;
    case M3GPU_AGG_P9999: return 0.9999;
    default: return -1.0;
    }
}

/* Exported standalone form of the plan builder's no-compression cap (the
 * exact simulation of stream.go:362-385 thresholds; see the inline copy
 * in m3gpu_rollup_batch_dev_opts). For tests and host-side sizing. */
extern "C" int m3gpu_ckms_exact_cap(const int32_t* agg_types, int naggs,
                                    double eps) {
    double qs[MAX_AGGS];
    int nq = 0;
    for (int i = 0; i < naggs && i < MAX_AGGS; i++) {
        double q = agg_quantile_of_host(agg_types[i]);
        if (q >= 0) qs[nq++] = q;
    }
    for (int i = 1; i < nq; i++) {
        double k = qs[i];
        int j = i - 1;
        while (j >= 0 && qs[j] > k) { qs[j + 1] = qs[j]; j--; }
        qs[j + 1] = k;
    }
    int m = 0;
    for (int i = 0; i < nq; i++)
        if (m == 0 || qs[m - 1] != qs[i]) qs[m++] = qs[i];
    nq = m;
    int cap = QCAP;
    double eps2 = 2.0 * eps;
    for (int v = 4; v <= QCAP + 1; v++) {
        bool merges = false;
        for (int mr = 0; mr <= v && !merges; mr++) {
            long long thr = LLONG_MAX;
            for (int i = 0; i < nq; i++) {
                long long qmin;
                if (mr >= (long long)(qs[i] * (double)v))
                    qmin = (long long)(eps2 * (double)mr / qs[i]);
                else
                    qmin = (long long)(eps2 * (double)(v - mr) / (1.0 - qs[i]));
                if (qmin < thr) thr = qmin;
            }
            if (nq > 0 && thr >= 2) merges = true;
        }
        if (merges) { cap = v - 1; break; }
    }
    return cap;
}

int m3gpu_rollup_batch_dev_opts(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int metric_type, int64_t window_ns, uint32_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* d_out, int64_t* d_out_window_ts, int32_t* d_out_errs,
    void* hip_stream, double eps, int every) {
    if (!nseries) return M3GPU_OK;
    if (naggs < 1 || naggs > MAX_AGGS || window_ns <= 0) {
        snprintf(g_err, sizeof(g_err), "bad naggs/window");
        return M3GPU_ERR_BADARG;
    }
    if (!(eps > 0.0 && eps < 0.5) || every < 1 || every > 1024) {
        snprintf(g_err, sizeof(g_err),
                 "eps must be in (0,0.5), every in [1,1024]");
        return M3GPU_ERR_BADARG;
    }
    m3::RollupPlan plan;
    memset(&plan, 0, sizeof(plan));
    plan.naggs = naggs;
    plan.eps = eps;
    plan.every = every;
    /* sorted unique quantile list + per-agg mapping (timer stream is
     * registered with the sorted unique quantile set) */
    double qs[MAX_AGGS];
    int nq = 0;
    for (int i = 0; i < naggs; i++) {
        plan.agg_types[i] = agg_types[i];
        plan.qidx[i] = -1;
        double q = -1;
        switch (agg_types[i]) {
        case M3GPU_AGG_MEDIAN: q = 0.5; break;
        case M3GPU_AGG_P10: q = 0.1; break;
        case M3GPU_AGG_P20: q = 0.2; break;
        case M3GPU_AGG_P25: q = 0.25; break;
        case M3GPU_AGG_P30: q = 0.3; break;
        case M3GPU_AGG_P40: q = 0.4; break;
        case M3GPU_AGG_P50: q = 0.5; break;
        case M3GPU_AGG_P60: q = 0.6; break;
        case M3GPU_AGG_P70: q = 0.7; break;
        case M3GPU_AGG_P75: q = 0.75; break;
        case M3GPU_AGG_P80: q = 0.8; break;
        case M3GPU_AGG_P90: q = 0.9; break;
        case M3GPU_AGG_P95: q = 0.95; break;
        case M3GPU_AGG_P99: q = 0.99; break;
        case M3GPU_AGG_P999: q = 0.999; break;
        case M3GPU_AGG_P9999: q = 0.9999; break;
        default: break;
        }
        if (q >= 0) { qs[nq++] = q; }
    }
    /* sort + dedupe */
    for (int i = 1; i < nq; i++) {
        double k = qs[i];
        int j = i - 1;
        while (j >= 0 && qs[j] > k) { qs[j + 1] = qs[j]; j--; }
        qs[j + 1] = k;
    }
    int m = 0;
    for (int i = 0; i < nq; i++)
        if (m == 0 || qs[m - 1] != qs[i]) qs[m++] = qs[i];
    nq = m;
    plan.nq = nq;
    for (int i = 0; i < nq; i++) plan.qs[i] = qs[i];
    /* exact_cap: largest n such that for every numVals v <= n and every
     * maxRank mr <= v, the compress merge threshold (stream.go:362-385,
     * int64 truncation exactly as the reference computes it) stays < 2 =
     * the minimum testVal — i.e. compression NEVER merges and quantiles
     * are exact order statistics. */
    plan.exact_cap = QCAP;
    {
        double eps2 = 2.0 * plan.eps;
        for (int v = 4; v <= QCAP + 1; v++) /* minSamplesToCompress=3 */ {
            bool merges = false;
            for (int mr = 0; mr <= v && !merges; mr++) {
                long long thr = LLONG_MAX;
                for (int i = 0; i < nq; i++) {
                    long long qmin;
                    if (mr >= (long long)(qs[i] * (double)v))
                        qmin = (long long)(eps2 * (double)mr / qs[i]);
                    else
                        qmin = (long long)(eps2 * (double)(v - mr) / (1.0 - qs[i]));
                    if (qmin < thr) thr = qmin;
                }
                if (nq > 0 && thr >= 2) merges = true;
            }
            if (merges) { plan.exact_cap = v - 1; break; }
        }
    }
    for (int i = 0; i < naggs; i++) {
        double q = -1;
        switch (agg_types[i]) {
        case M3GPU_AGG_MEDIAN: case M3GPU_AGG_P50: q = 0.5; break;
        case M3GPU_AGG_P10: q = 0.1; break;
        case M3GPU_AGG_P20: q = 0.2; break;
        case M3GPU_AGG_P25: q = 0.25; break;
        case M3GPU_AGG_P30: q = 0.3; break;
        case M3GPU_AGG_P40: q = 0.4; break;
        case M3GPU_AGG_P60: q = 0.6; break;
        case M3GPU_AGG_P70: q = 0.7; break;
        case M3GPU_AGG_P75: q = 0.75; break;
        case M3GPU_AGG_P80: q = 0.8; break;
        case M3GPU_AGG_P90: q = 0.9; break;
        case M3GPU_AGG_P95: q = 0.95; break;
        case M3GPU_AGG_P99: q = 0.99; break;
        case M3GPU_AGG_P999: q = 0.999; break;
        case M3GPU_AGG_P9999: q = 0.9999; break;
        default: break;
        }
        if (q >= 0)
            for (int k = 0; k < nq; k++)
                if (plan.qs[k] == q) { plan.qidx[i] = (int8_t)k; break; }
    }

    /* extreme_cap: largest n (bounded by exact_cap) for which EVERY
     * requested quantile resolves to sorted[n-1] — the bucket max — in
     * BOTH regimes the lane kernel reproduces (quantilesFromBuf for n<=3,
     * the calcQuantiles walk with the k_{i-1}+1 shift above). All-high
     * quantile sets (p90+, e.g. the sum/min/max/p99 rollup) then need no
     * per-point value staging at all: running min/max suffice. */
    plan.extreme_cap = 0;
    if (nq > 0) {
        int capn = plan.exact_cap < QCAP ? plan.exact_cap : QCAP;
        for (int n = 1; n <= capn; n++) {
            bool allmax = true;
            if (n <= 3) {
                for (int i = 0; i < nq && allmax; i++) {
                    int idx = (int)(qs[i] * (double)n);
                    if (idx >= n) idx = n - 1;
                    if (idx != n - 1) allmax = false;
                }
            } else {
                int kk = 0;
                for (int i = 0; i < nq; i++) {
                    int rank = (int)ceil(qs[i] * (double)n);
                    kk = (i == 0) ? rank : ((rank > kk + 1) ? rank : kk + 1);
                    int kc = kk > n ? n : kk;
                    if (kc != n) { allmax = false; break; }
                }
            }
            if (!allmax) break;
            plan.extreme_cap = n;
        }
    }

    hipStream_t s = (hipStream_t)hip_stream;
    bool with_q = plan.nq > 0 && metric_type == M3GPU_METRIC_TIMER;
    /* deeper-than-cap buckets overflow to the wave tier; only use extreme
     * mode when the cap covers normal windows (a lone p99 with the
     * production-default eps has exact_cap 13, which bounds it) */
    bool extreme = with_q && plan.extreme_cap >= 8;
    /* metric type is a template parameter: dead per-metric BucketState
     * fields drop out of the register file */
    if (extreme)
        hipLaunchKernelGGL((m3::k_rollup_lane<RQ_EXTREME, M3GPU_METRIC_TIMER>),
                           dim3(grid_lane(nseries)),
                           dim3(BLOCK_THREADS), 0, s,
                           d_blobs, d_offsets, d_lens, nseries, int_optimized,
                           default_unit, window_ns, nbuckets, plan,
                           d_out, d_out_window_ts, d_out_errs);
    else if (with_q)
        hipLaunchKernelGGL((m3::k_rollup_lane<RQ_STAGED, M3GPU_METRIC_TIMER>),
                           dim3(grid_lane(nseries)),
                           dim3(BLOCK_THREADS), 0, s,
                           d_blobs, d_offsets, d_lens, nseries, int_optimized,
                           default_unit, window_ns, nbuckets, plan,
                           d_out, d_out_window_ts, d_out_errs);
    else if (metric_type == M3GPU_METRIC_COUNTER)
        hipLaunchKernelGGL((m3::k_rollup_lane<RQ_NONE, M3GPU_METRIC_COUNTER>),
                           dim3(grid_lane(nseries)),
                           dim3(BLOCK_THREADS), 0, s,
                           d_blobs, d_offsets, d_lens, nseries, int_optimized,
                           default_unit, window_ns, nbuckets, plan,
                           d_out, d_out_window_ts, d_out_errs);
    else if (metric_type == M3GPU_METRIC_GAUGE)
        hipLaunchKernelGGL((m3::k_rollup_lane<RQ_NONE, M3GPU_METRIC_GAUGE>),
                           dim3(grid_lane(nseries)),
                           dim3(BLOCK_THREADS), 0, s,
                           d_blobs, d_offsets, d_lens, nseries, int_optimized,
                           default_unit, window_ns, nbuckets, plan,
                           d_out, d_out_window_ts, d_out_errs);
    else
        hipLaunchKernelGGL((m3::k_rollup_lane<RQ_NONE, M3GPU_METRIC_TIMER>),
                           dim3(grid_lane(nseries)),
                           dim3(BLOCK_THREADS), 0, s,
                           d_blobs, d_offsets, d_lens, nseries, int_optimized,
                           default_unit, window_ns, nbuckets, plan,
                           d_out, d_out_window_ts, d_out_errs);
    HIP_TRY(hipGetLastError());
    if (with_q) {
        /* buckets deeper than the lane tier's capacity overflow to the
         * wave-per-series kernel. The common case is zero overflows: test
         * with a device-side count + 4-byte D2H instead of pulling and
         * scanning the whole error array every call. */
        /* process-lifetime scratch; atomics guard the one-time alloc
         * (host entry points may be called from multiple threads) */
        static std::atomic<uint32_t*> g_ovf{nullptr};
        uint32_t* d_ovf = g_ovf.load(std::memory_order_acquire);
        if (!d_ovf) {
            uint32_t* fresh = nullptr;
            HIP_TRY(hipMalloc(&fresh, sizeof(uint32_t)));
            uint32_t* expected = nullptr;
            if (!g_ovf.compare_exchange_strong(expected, fresh,
                                               std::memory_order_acq_rel)) {
                (void)hipFree(fresh); /* another thread won the race */
            }
            d_ovf = g_ovf.load(std::memory_order_acquire);
        }
        HIP_TRY(hipMemsetAsync(d_ovf, 0, sizeof(uint32_t), s));
        hipLaunchKernelGGL(m3::k_count_errcode, dim3(256), dim3(BLOCK_THREADS),
                           0, s, d_out_errs, nseries,
                           M3GPU_SERIES_BUCKET_OVERFLOW, d_ovf);
        HIP_TRY(hipGetLastError());
        uint32_t h_ovf = 0;
        HIP_TRY(hipMemcpyAsync(&h_ovf, d_ovf, sizeof(uint32_t),
                               hipMemcpyDeviceToHost, s));
        HIP_TRY(hipStreamSynchronize(s));
        if (h_ovf == 0) return M3GPU_OK;
        int32_t* h_errs = (int32_t*)malloc(nseries * sizeof(int32_t));
        if (!h_errs) { snprintf(g_err, sizeof(g_err), "oom"); return M3GPU_ERR_HIP; }
        hipError_t ce = hipMemcpy(h_errs, d_out_errs, nseries * sizeof(int32_t),
                                  hipMemcpyDeviceToHost);
        if (ce != hipSuccess) { free(h_errs); return set_hip_err("errs copy", ce); }
        uint32_t nretry = 0;
        for (uint32_t i = 0; i < nseries; i++)
            if (h_errs[i] == M3GPU_SERIES_BUCKET_OVERFLOW) nretry++;
        if (nretry) {
            int32_t* h_sel = (int32_t*)malloc(nretry * sizeof(int32_t));
            uint32_t k = 0;
            for (uint32_t i = 0; i < nseries; i++)
                if (h_errs[i] == M3GPU_SERIES_BUCKET_OVERFLOW) h_sel[k++] = (int32_t)i;
            int32_t* d_sel = nullptr;
            hipError_t e2 = hipMalloc(&d_sel, nretry * sizeof(int32_t));
            if (e2 == hipSuccess)
                e2 = hipMemcpy(d_sel, h_sel, nretry * sizeof(int32_t), hipMemcpyHostToDevice);
            if (e2 == hipSuccess) {
                hipLaunchKernelGGL(m3::k_rollup_batch, dim3(grid_for(nretry)),
                                   dim3(BLOCK_THREADS), 0, s,
                                   d_blobs, d_offsets, d_lens, d_sel, nretry,
                                   int_optimized, default_unit, metric_type,
                                   window_ns, nbuckets, plan,
                                   d_out, d_out_window_ts, d_out_errs);
                e2 = hipGetLastError();
                if (e2 == hipSuccess) e2 = hipStreamSynchronize(s);
            }
            (void)hipFree(d_sel);
            free(h_sel);
            if (e2 != hipSuccess) { free(h_errs); return set_hip_err("rollup retry", e2); }

            /* Third tier: buckets deeper than the exact-order-statistics cap
             * run the real compressed CKMS (k_rollup_ckms, one series per
             * workgroup, ~122 KB dynamic LDS). Rescan the error flags the
             * wave kernel just rewrote. */
            ce = hipMemcpy(h_errs, d_out_errs, nseries * sizeof(int32_t),
                           hipMemcpyDeviceToHost);
            if (ce != hipSuccess) { free(h_errs); return set_hip_err("errs copy2", ce); }
            uint32_t ndeep = 0;
            for (uint32_t i = 0; i < nseries; i++)
                if (h_errs[i] == M3GPU_SERIES_BUCKET_OVERFLOW) ndeep++;
            if (ndeep) {
                size_t lds_bytes = (size_t)CKMS_CAP * 32 /* val+nr+dl x2 */
                                 + (size_t)CKMS_BUF * 24 /* less+more+sorted */
                                 + (size_t)MAX_AGGS * 24; /* computed+thr */
                hipError_t e3 = hipFuncSetAttribute(
                    reinterpret_cast<const void*>(m3::k_rollup_ckms),
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);
                int32_t* h_sel2 = (int32_t*)malloc(ndeep * sizeof(int32_t));
                uint32_t k2 = 0;
                for (uint32_t i = 0; i < nseries; i++)
                    if (h_errs[i] == M3GPU_SERIES_BUCKET_OVERFLOW)
                        h_sel2[k2++] = (int32_t)i;
                int32_t* d_sel2 = nullptr;
                if (e3 == hipSuccess) e3 = hipMalloc(&d_sel2, ndeep * sizeof(int32_t));
                if (e3 == hipSuccess)
                    e3 = hipMemcpy(d_sel2, h_sel2, ndeep * sizeof(int32_t),
                                   hipMemcpyHostToDevice);
                if (e3 == hipSuccess) {
                    hipLaunchKernelGGL(m3::k_rollup_ckms, dim3(ndeep),
                                       dim3(CKMS_BLOCK), lds_bytes, s,
                                       d_blobs, d_offsets, d_lens, d_sel2, ndeep,
                                       int_optimized, default_unit,
                                       window_ns, nbuckets, plan,
                                       d_out, d_out_window_ts, d_out_errs);
                    e3 = hipGetLastError();
                    if (e3 == hipSuccess) e3 = hipStreamSynchronize(s);
                }
                (void)hipFree(d_sel2);
                free(h_sel2);
                if (e3 != hipSuccess) { free(h_errs); return set_hip_err("rollup ckms", e3); }
            }
        }
        free(h_errs);
    }
    return M3GPU_OK;
}

int m3gpu_rollup_batch_dev(
    const uint8_t* d_blobs, const uint64_t* d_offsets, const uint32_t* d_lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int metric_type, int64_t window_ns, uint32_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* d_out, int64_t* d_out_window_ts, int32_t* d_out_errs,
    void* hip_stream) {
    return m3gpu_rollup_batch_dev_opts(
        d_blobs, d_offsets, d_lens, nseries, int_optimized, default_unit,
        metric_type, window_ns, nbuckets, agg_types, naggs, d_out,
        d_out_window_ts, d_out_errs, hip_stream, 1e-3, 1024);
}

int m3gpu_merge_batch_dev(
    const int64_t* d_ts, const double* d_vals, const uint32_t* d_counts,
    uint32_t nreplicas, uint32_t nseries, uint32_t stride,
    int64_t* d_out_ts, double* d_out_vals, uint32_t* d_out_counts,
    int32_t* d_out_errs, uint32_t out_stride, void* hip_stream) {
    if (!nseries) return M3GPU_OK;
    if (nreplicas < 1 || nreplicas > MERGE_MAX_R) {
        snprintf(g_err, sizeof(g_err), "nreplicas must be 1..%d", MERGE_MAX_R);
        return M3GPU_ERR_BADARG;
    }
    hipStream_t s = (hipStream_t)hip_stream;
    hipLaunchKernelGGL(m3::k_merge, dim3(grid_lane(nseries)),
                       dim3(BLOCK_THREADS), 0, s,
                       d_ts, d_vals, d_counts, nreplicas, nseries, stride,
                       d_out_ts, d_out_vals, d_out_counts, d_out_errs,
                       out_stride);
    HIP_TRY(hipGetLastError());
    return M3GPU_OK;
}

/* ---------- host-pointer convenience forms (the cgo surface) ---------- */

int m3gpu_decode_batch(
    const uint8_t* blobs, uint64_t blobs_len,
    const uint64_t* offsets, const uint32_t* lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts,
    int32_t* out_errs, uint32_t stride) {
    uint8_t* d_blobs = nullptr;
    uint64_t* d_offsets = nullptr;
    uint32_t* d_lens = nullptr;
    int64_t* d_ts = nullptr;
    double* d_vals = nullptr;
    uint32_t* d_counts = nullptr;
    int32_t* d_errs = nullptr;
    uint64_t npts = (uint64_t)nseries * stride;
    int rc = M3GPU_OK;
    HIP_TRY(hipMalloc(&d_blobs, blobs_len));
    HIP_TRY(hipMalloc(&d_offsets, (nseries + 1) * sizeof(uint64_t)));
    HIP_TRY(hipMalloc(&d_lens, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_ts, npts * sizeof(int64_t)));
    HIP_TRY(hipMalloc(&d_vals, npts * sizeof(double)));
    HIP_TRY(hipMalloc(&d_counts, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_errs, nseries * sizeof(int32_t)));
    HIP_TRY(hipMemcpy(d_blobs, blobs, blobs_len, hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_offsets, offsets, (nseries + 1) * sizeof(uint64_t), hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_lens, lens, nseries * sizeof(uint32_t), hipMemcpyHostToDevice));
    rc = m3gpu_decode_batch_dev(d_blobs, d_offsets, d_lens, nseries,
                                int_optimized, default_unit, d_ts, d_vals,
                                d_counts, d_errs, stride, nullptr);
    if (rc == M3GPU_OK) {
        HIP_TRY(hipMemcpy(out_ts, d_ts, npts * sizeof(int64_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_vals, d_vals, npts * sizeof(double), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_counts, d_counts, nseries * sizeof(uint32_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_errs, d_errs, nseries * sizeof(int32_t), hipMemcpyDeviceToHost));
    }
    (void)hipFree(d_blobs); (void)hipFree(d_offsets); (void)hipFree(d_lens);
    (void)hipFree(d_ts); (void)hipFree(d_vals); (void)hipFree(d_counts); (void)hipFree(d_errs);
    return rc;
}

int m3gpu_decode_batch_ann(
    const uint8_t* blobs, uint64_t blobs_len,
    const uint64_t* offsets, const uint32_t* lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts,
    int32_t* out_errs, uint32_t stride,
    uint8_t* out_ann, uint32_t ann_stride) {
    /* host-pointer convenience form of m3gpu_decode_batch_dev_ann (what a
     * cgo ReaderIterator shim calls for annotation-bearing blocks) */
    uint8_t* d_blobs = nullptr;
    uint64_t* d_offsets = nullptr;
    uint32_t* d_lens = nullptr;
    int64_t* d_ts = nullptr;
    double* d_vals = nullptr;
    uint32_t* d_counts = nullptr;
    int32_t* d_errs = nullptr;
    uint8_t* d_ann = nullptr;
    uint64_t npts = (uint64_t)nseries * stride;
    uint64_t ann_total = (uint64_t)nseries * ann_stride;
    int rc = M3GPU_OK;
    HIP_TRY(hipMalloc(&d_blobs, blobs_len));
    HIP_TRY(hipMalloc(&d_offsets, (nseries + 1) * sizeof(uint64_t)));
    HIP_TRY(hipMalloc(&d_lens, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_ts, npts * sizeof(int64_t)));
    HIP_TRY(hipMalloc(&d_vals, npts * sizeof(double)));
    HIP_TRY(hipMalloc(&d_counts, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_errs, nseries * sizeof(int32_t)));
    HIP_TRY(hipMalloc(&d_ann, ann_total));
    HIP_TRY(hipMemset(d_ann, 0, ann_total));
    HIP_TRY(hipMemcpy(d_blobs, blobs, blobs_len, hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_offsets, offsets, (nseries + 1) * sizeof(uint64_t), hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_lens, lens, nseries * sizeof(uint32_t), hipMemcpyHostToDevice));
    rc = m3gpu_decode_batch_dev_ann(d_blobs, d_offsets, d_lens, nseries,
                                    int_optimized, default_unit, d_ts, d_vals,
                                    d_counts, d_errs, stride, d_ann,
                                    ann_stride, nullptr);
    if (rc == M3GPU_OK) {
        HIP_TRY(hipMemcpy(out_ts, d_ts, npts * sizeof(int64_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_vals, d_vals, npts * sizeof(double), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_counts, d_counts, nseries * sizeof(uint32_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_errs, d_errs, nseries * sizeof(int32_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_ann, d_ann, ann_total, hipMemcpyDeviceToHost));
    }
    (void)hipFree(d_blobs); (void)hipFree(d_offsets); (void)hipFree(d_lens);
    (void)hipFree(d_ts); (void)hipFree(d_vals); (void)hipFree(d_counts);
    (void)hipFree(d_errs); (void)hipFree(d_ann);
    return rc;
}

int m3gpu_encode_batch(
    const int64_t* ts, const double* vals, const uint32_t* counts,
    uint32_t nseries, uint32_t stride, int int_optimized, uint8_t unit,
    uint8_t* out_bytes, uint32_t out_stride, uint32_t* out_lens,
    int32_t* out_errs) {
    int64_t* d_ts = nullptr;
    double* d_vals = nullptr;
    uint32_t* d_counts = nullptr;
    uint8_t* d_out = nullptr;
    uint32_t* d_lens = nullptr;
    int32_t* d_errs = nullptr;
    uint64_t npts = (uint64_t)nseries * stride;
    uint64_t outb = (uint64_t)nseries * out_stride;
    int rc;
    HIP_TRY(hipMalloc(&d_ts, npts * sizeof(int64_t)));
    HIP_TRY(hipMalloc(&d_vals, npts * sizeof(double)));
    HIP_TRY(hipMalloc(&d_counts, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_out, outb));
    HIP_TRY(hipMalloc(&d_lens, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_errs, nseries * sizeof(int32_t)));
    HIP_TRY(hipMemcpy(d_ts, ts, npts * sizeof(int64_t), hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_vals, vals, npts * sizeof(double), hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_counts, counts, nseries * sizeof(uint32_t), hipMemcpyHostToDevice));
    HIP_TRY(hipMemset(d_out, 0, outb));
    rc = m3gpu_encode_batch_dev(d_ts, d_vals, d_counts, nseries, stride,
                                int_optimized, unit, d_out, out_stride,
                                d_lens, d_errs, nullptr);
    if (rc == M3GPU_OK) {
        HIP_TRY(hipMemcpy(out_bytes, d_out, outb, hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_lens, d_lens, nseries * sizeof(uint32_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_errs, d_errs, nseries * sizeof(int32_t), hipMemcpyDeviceToHost));
    }
    (void)hipFree(d_ts); (void)hipFree(d_vals); (void)hipFree(d_counts);
    (void)hipFree(d_out); (void)hipFree(d_lens); (void)hipFree(d_errs);
    return rc;
}

int m3gpu_rollup_batch(
    const uint8_t* blobs, uint64_t blobs_len,
    const uint64_t* offsets, const uint32_t* lens,
    uint32_t nseries, int int_optimized, uint8_t default_unit,
    int metric_type, int64_t window_ns, uint32_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* out, int64_t* out_window_ts, int32_t* out_errs) {
    uint8_t* d_blobs = nullptr;
    uint64_t* d_offsets = nullptr;
    uint32_t* d_lens = nullptr;
    double* d_out = nullptr;
    int64_t* d_wts = nullptr;
    int32_t* d_errs = nullptr;
    uint64_t nout = (uint64_t)nseries * nbuckets * naggs;
    int rc;
    HIP_TRY(hipMalloc(&d_blobs, blobs_len));
    HIP_TRY(hipMalloc(&d_offsets, (nseries + 1) * sizeof(uint64_t)));
    HIP_TRY(hipMalloc(&d_lens, nseries * sizeof(uint32_t)));
    HIP_TRY(hipMalloc(&d_out, nout * sizeof(double)));
    HIP_TRY(hipMalloc(&d_wts, (uint64_t)nseries * nbuckets * sizeof(int64_t)));
    HIP_TRY(hipMalloc(&d_errs, nseries * sizeof(int32_t)));
    HIP_TRY(hipMemcpy(d_blobs, blobs, blobs_len, hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_offsets, offsets, (nseries + 1) * sizeof(uint64_t), hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_lens, lens, nseries * sizeof(uint32_t), hipMemcpyHostToDevice));
    rc = m3gpu_rollup_batch_dev(d_blobs, d_offsets, d_lens, nseries,
                                int_optimized, default_unit, metric_type,
                                window_ns, nbuckets, agg_types, naggs,
                                d_out, d_wts, d_errs, nullptr);
    if (rc == M3GPU_OK) {
        HIP_TRY(hipMemcpy(out, d_out, nout * sizeof(double), hipMemcpyDeviceToHost));
        if (out_window_ts)
            HIP_TRY(hipMemcpy(out_window_ts, d_wts, (uint64_t)nseries * nbuckets * sizeof(int64_t), hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(out_errs, d_errs, nseries * sizeof(int32_t), hipMemcpyDeviceToHost));
    }
    (void)hipFree(d_blobs); (void)hipFree(d_offsets); (void)hipFree(d_lens);
    (void)hipFree(d_out); (void)hipFree(d_wts); (void)hipFree(d_errs);
    return rc;
}

} /* extern "C" */
