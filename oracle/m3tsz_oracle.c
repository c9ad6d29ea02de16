/*
 * m3tsz_oracle.c — CPU oracle (C restatement) of the M3TSZ block codec.
 * TEST INFRASTRUCTURE ONLY — see m3tsz_oracle.h header comment.
 *
 * Every function cites the reference file:line whose behavior it restates.
 * Compile with -ffp-contract=off: encode output bytes depend on exact f64
 * arithmetic in convertToIntFloat (m3tsz.go:78-119) and decode int
 * accumulation (iterator.go:168-175).
 */
#include "m3tsz_oracle.h"
#include <string.h>
#include <stdlib.h>
#include <math.h>

/* ============================ constants ============================ */

/* m3tsz.go:28-62 */
#define OPCODE_ZERO_SIG 0x0
#define OPCODE_NONZERO_SIG 0x1
#define NUM_SIG_BITS 6
#define OPCODE_ZERO_VALUE_XOR 0x0
#define OPCODE_CONTAINED_VALUE_XOR 0x2
#define OPCODE_UNCONTAINED_VALUE_XOR 0x3
#define OPCODE_NO_UPDATE_SIG 0x0
#define OPCODE_UPDATE_SIG 0x1
#define OPCODE_UPDATE 0x0
#define OPCODE_NO_UPDATE 0x1
#define OPCODE_UPDATE_MULT 0x1
#define OPCODE_NO_UPDATE_MULT 0x0
#define OPCODE_POSITIVE 0x0
#define OPCODE_NEGATIVE 0x1
#define OPCODE_REPEAT 0x1
#define OPCODE_NO_REPEAT 0x0
#define OPCODE_FLOAT_MODE 0x1
#define OPCODE_INT_MODE 0x0
#define SIG_DIFF_THRESHOLD 3
#define SIG_REPEAT_THRESHOLD 5
#define MAX_MULT 6
#define NUM_MULT_BITS 3

/* scheme.go:28-38 */
#define MARKER_OPCODE 0x100ULL
#define MARKER_OPCODE_BITS 9
#define MARKER_VALUE_BITS 2
#define MARKER_EOS 0
#define MARKER_ANNOTATION 1
#define MARKER_TIMEUNIT 2

static const double MAX_INT = 9223372036854775808.0;  /* float64(math.MaxInt64), m3tsz.go:65 */
static const double MIN_INT = -9223372036854775808.0; /* float64(math.MinInt64), m3tsz.go:66 */
static const double MAX_OPT_INT = 1e13;               /* m3tsz.go:67 */
static const double MULTIPLIERS[7] = {1.0, 10.0, 100.0, 1000.0, 10000.0, 100000.0, 1000000.0}; /* m3tsz.go:131-140 */

/* src/x/time/unit.go unitsToDuration (ns per unit) */
static const int64_t UNIT_NS[9] = {
    0,                       /* None */
    1000000000LL,            /* Second */
    1000000LL,               /* Millisecond */
    1000LL,                  /* Microsecond */
    1LL,                     /* Nanosecond */
    60000000000LL,           /* Minute */
    3600000000000LL,         /* Hour */
    86400000000000LL,        /* Day */
    31536000000000000LL,     /* Year = 365 days */
};
#define UNIT_COUNT 9
static int unit_is_valid(uint8_t u) { return u > 0 && u < UNIT_COUNT; } /* unit.go:88-90 */

/* default time encoding schemes, scheme.go:42-52:
 * buckets {7,9,12} value bits; default 32 (s/ms) or 64 (us/ns). */
static int scheme_default_bits(uint8_t unit) {
    switch (unit) {
    case M3_UNIT_SECOND:
    case M3_UNIT_MILLISECOND: return 32;
    case M3_UNIT_MICROSECOND:
    case M3_UNIT_NANOSECOND: return 64;
    default: return 0; /* no scheme */
    }
}
static const int BUCKET_VALUE_BITS[3] = {7, 9, 12};
/* NewTimeEncodingScheme (scheme.go:125-144): bucket i opcode/bits */
static const uint64_t BUCKET_OPCODE[3] = {0x2, 0x6, 0xe}; /* 10, 110, 1110 */
static const int BUCKET_OPCODE_BITS[3] = {2, 3, 4};
#define DEFAULT_BUCKET_OPCODE 0xfULL /* 1111 */
#define DEFAULT_BUCKET_OPCODE_BITS 4

/* ========================= helpers (encoding.go) ========================= */

static inline uint8_t num_sig(uint64_t v) { /* encoding.go:29-31 */
    return (uint8_t)(64 - (v ? __builtin_clzll(v) : 64));
}
static inline void leading_trailing_zeros(uint64_t v, int* lead, int* trail) { /* encoding.go:35-43 */
    if (v == 0) { *lead = 64; *trail = 0; return; }
    *lead = __builtin_clzll(v);
    *trail = __builtin_ctzll(v);
}
static inline int64_t sign_extend(uint64_t v, uint8_t num_bits) { /* encoding.go:46-49 */
    int shift = 64 - num_bits;
    return ((int64_t)(v << shift)) >> shift;
}
/* Go float64->int64 conversion as produced by gc on amd64 (CVTTSD2SQ):
 * out-of-range and NaN give 0x8000000000000000. Used for
 * uint64(int64(val)) in encoder.go:141,214. */
static inline int64_t go_f2i(double v) {
    if (v >= MAX_INT || v < MIN_INT || v != v) return INT64_MIN;
    return (int64_t)v;
}
static inline uint64_t f2bits(double v) { uint64_t b; memcpy(&b, &v, 8); return b; }
static inline double bits2f(uint64_t b) { double v; memcpy(&v, &b, 8); return v; }

/* ============================== ostream ============================== */
/* ostream.go:33-221. pos = bits used in last byte (0..8). */

typedef struct {
    uint8_t* buf;
    int64_t len, cap;
    int pos;
} m3_ostream;

static void os_init(m3_ostream* os, int64_t cap) {
    os->buf = (uint8_t*)malloc(cap > 16 ? (size_t)cap : 16);
    os->len = 0; os->cap = cap > 16 ? cap : 16; os->pos = 0;
}
static void os_free(m3_ostream* os) { free(os->buf); os->buf = NULL; }
static inline void os_grow_byte(m3_ostream* os, uint8_t v, int np) { /* ostream.go:86-91 */
    if (os->len == os->cap) {
        os->cap *= 2;
        os->buf = (uint8_t*)realloc(os->buf, (size_t)os->cap);
    }
    os->buf[os->len++] = v;
    os->pos = np;
}
static inline int os_has_unused(const m3_ostream* os) { return os->pos > 0 && os->pos < 8; }
static inline void os_fill_unused(m3_ostream* os, uint8_t v) { /* ostream.go:129-131 */
    os->buf[os->len - 1] |= (uint8_t)(v >> os->pos);
}
static inline void os_write_bit(m3_ostream* os, int v) { /* ostream.go:133-141 */
    uint8_t b = (uint8_t)(v << 7);
    if (!os_has_unused(os)) { os_grow_byte(os, b, 1); return; }
    os_fill_unused(os, b);
    os->pos++;
}
static inline void os_write_byte(m3_ostream* os, uint8_t v) { /* ostream.go:143-150 */
    if (!os_has_unused(os)) { os_grow_byte(os, v, 8); return; }
    os_fill_unused(os, v);
    os_grow_byte(os, (uint8_t)(v << (8 - os->pos)), os->pos);
}
static void os_write_bytes(m3_ostream* os, const uint8_t* bytes, int64_t n) { /* ostream.go:152-173 */
    if (!os_has_unused(os)) {
        for (int64_t i = 0; i < n; i++) os_grow_byte(os, bytes[i], 8);
        os->pos = 8;
        return;
    }
    for (int64_t i = 0; i < n; i++) os_write_byte(os, bytes[i]);
}
static void os_write_bits(m3_ostream* os, uint64_t v, int num_bits) { /* ostream.go:180-221 */
    if (num_bits == 0) return;
    if (num_bits > 64) num_bits = 64;
    v <<= (64 - num_bits);
    while (num_bits >= 8) {
        os_write_byte(os, (uint8_t)(v >> 56));
        v <<= 8;
        num_bits -= 8;
    }
    uint8_t remainder = (uint8_t)(v >> 56);
    while (num_bits > 0) {
        uint8_t val = remainder & 0x80;
        if (os_has_unused(os)) { os_fill_unused(os, val); os->pos++; }
        else os_grow_byte(os, val, 1);
        remainder <<= 1;
        num_bits--;
    }
}

/* ============================== istream ============================== */
/* istream.go:30-133 over xio.BytesReader64 (reader64.go:29-86). */

typedef struct {
    const uint8_t* data;
    int64_t len;
    int64_t index;      /* reader64 byte index */
    uint64_t current;   /* istream buffered word (left-aligned) */
    uint8_t remaining;  /* valid bits in current */
} m3_istream;

static void is_init(m3_istream* is, const uint8_t* data, int64_t len) {
    is->data = data; is->len = len; is->index = 0; is->current = 0; is->remaining = 0;
}
/* reader64.go:40-58 */
static int rd64_read(m3_istream* is, uint64_t* word, uint8_t* n) {
    if (is->index + 8 <= is->len) {
        uint64_t res;
        memcpy(&res, is->data + is->index, 8);
        res = __builtin_bswap64(res);
        is->index += 8;
        *word = res; *n = 8;
        return 0;
    }
    if (is->index >= is->len) return -M3_ERR_EOF;
    uint64_t res = 0; uint8_t bytes = 0;
    for (; is->index < is->len; is->index++) { res = (res << 8) | is->data[is->index]; bytes++; }
    *word = res << (64 - 8 * bytes); *n = bytes;
    return 0;
}
/* reader64.go:61-80 */
static int rd64_peek(const m3_istream* is, uint64_t* word, uint8_t* n) {
    if (is->index + 8 <= is->len) {
        uint64_t res;
        memcpy(&res, is->data + is->index, 8);
        *word = __builtin_bswap64(res); *n = 8;
        return 0;
    }
    if (is->index >= is->len) return -M3_ERR_EOF;
    uint64_t res = 0; uint8_t bytes = 0;
    for (int64_t i = is->index; i < is->len; i++) { res = (res << 8) | is->data[i]; bytes++; }
    *word = res << (64 - 8 * bytes); *n = bytes;
    return 0;
}
static inline uint64_t read_bits_in_word(uint64_t w, uint8_t num_bits) { /* istream.go:123-125 */
    return num_bits ? (w >> (64 - num_bits)) : 0;
}
/* istream.go:73-98 */
static int is_read_bits(m3_istream* is, uint8_t num_bits, uint64_t* out) {
    uint64_t res = read_bits_in_word(is->current, num_bits);
    uint8_t remaining = is->remaining;
    if (num_bits <= remaining) {
        is->current = (num_bits >= 64) ? 0 : (is->current << num_bits);
        is->remaining -= num_bits;
        *out = res;
        return 0;
    }
    uint8_t bits_needed = num_bits - remaining;
    uint64_t current; uint8_t nb;
    int err = rd64_read(is, &current, &nb);
    if (err) return err;
    uint8_t n = nb * 8;
    if (n < bits_needed) return -M3_ERR_EOF;
    is->current = (bits_needed >= 64) ? 0 : (current << bits_needed); /* Go: <<64 == 0 */
    is->remaining = n - bits_needed;
    *out = res | (current >> (64 - bits_needed));
    return 0;
}
/* istream.go:101-115 */
static int is_peek_bits(const m3_istream* is, uint8_t num_bits, uint64_t* out) {
    if (num_bits <= is->remaining) { *out = read_bits_in_word(is->current, num_bits); return 0; }
    uint64_t res = read_bits_in_word(is->current, num_bits);
    uint8_t bits_needed = num_bits - is->remaining;
    uint64_t next; uint8_t nb;
    int err = rd64_peek(is, &next, &nb);
    if (err) return err;
    if (8 * nb < bits_needed) return -M3_ERR_EOF;
    *out = res | read_bits_in_word(next, bits_needed);
    return 0;
}

/* ============================ xxhash64 ============================ */
/* Public XXH64, seed 0 (reference uses github.com/cespare/xxhash/v2 v2.1.2,
 * not vendored; annotation dedupe only, timestamp_encoder.go:56,164-170). */

#define P64_1 11400714785074694791ULL
#define P64_2 14029467366897019727ULL
#define P64_3 1609587929392839161ULL
#define P64_4 9650029242287828579ULL
#define P64_5 2870177450012600261ULL

static inline uint64_t rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
static inline uint64_t xx_round(uint64_t acc, uint64_t input) {
    acc += input * P64_2;
    acc = rotl64(acc, 31);
    acc *= P64_1;
    return acc;
}
static inline uint64_t xx_merge(uint64_t acc, uint64_t val) {
    acc ^= xx_round(0, val);
    acc = acc * P64_1 + P64_4;
    return acc;
}
static inline uint64_t le64(const uint8_t* p) { uint64_t v; memcpy(&v, p, 8); return v; }
static inline uint32_t le32(const uint8_t* p) { uint32_t v; memcpy(&v, p, 4); return v; }

uint64_t oracle_xxhash64(const uint8_t* data, size_t len) {
    const uint8_t* p = data;
    const uint8_t* end = data + len;
    uint64_t h;
    if (len >= 32) {
        uint64_t v1 = P64_1 + P64_2, v2 = P64_2, v3 = 0, v4 = (uint64_t)0 - P64_1;
        do {
            v1 = xx_round(v1, le64(p)); p += 8;
            v2 = xx_round(v2, le64(p)); p += 8;
            v3 = xx_round(v3, le64(p)); p += 8;
            v4 = xx_round(v4, le64(p)); p += 8;
        } while (p <= end - 32);
        h = rotl64(v1, 1) + rotl64(v2, 7) + rotl64(v3, 12) + rotl64(v4, 18);
        h = xx_merge(h, v1); h = xx_merge(h, v2); h = xx_merge(h, v3); h = xx_merge(h, v4);
    } else {
        h = P64_5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) { h ^= xx_round(0, le64(p)); h = rotl64(h, 27) * P64_1 + P64_4; p += 8; }
    if (p + 4 <= end) { h ^= (uint64_t)le32(p) * P64_1; h = rotl64(h, 23) * P64_2 + P64_3; p += 4; }
    while (p < end) { h ^= (uint64_t)(*p) * P64_5; h = rotl64(h, 11) * P64_1; p++; }
    h ^= h >> 33; h *= P64_2; h ^= h >> 29; h *= P64_3; h ^= h >> 32;
    return h;
}
#define EMPTY_ANN_CHECKSUM 0xEF46DB3751D8E999ULL /* xxhash.Sum64(nil) */

/* ======================= convertToIntFloat ======================= */
/* m3tsz.go:78-127 — bit-sensitive float math: no fma, exact order. */


/* Go math.Modf: Modf(±Inf) = (±Inf, NaN) — C99 modf returns (±Inf, ±0). */
static inline double go_modf(double v, double* ipart) {
    if (v == (1.0/0.0) || v == (-1.0/0.0)) { *ipart = v; return nan(""); }
    return modf(v, ipart);
}
static int convert_to_int_float(double v, uint8_t cur_max_mult,
                                double* out_val, uint8_t* out_mult, int* out_is_float) {
    if (cur_max_mult == 0 && v < MAX_INT) {
        double i, r;
        r = go_modf(v, &i);
        if (r == 0) { *out_val = i; *out_mult = 0; *out_is_float = 0; return 0; }
    }
    if (cur_max_mult > MAX_MULT) return -M3_ERR_INVALID_MULT;

    double sign = 1.0;
    if (v < 0) sign = -1.0;

    for (uint8_t mult = cur_max_mult; mult <= MAX_MULT; mult++) {
        double val = v * MULTIPLIERS[mult] * sign;
        if (val >= MAX_OPT_INT) break;
        double i, r;
        r = go_modf(val, &i);
        if (r == 0) { *out_val = sign * i; *out_mult = mult; *out_is_float = 0; return 0; }
        else if (r < 0.1) {
            if (nextafter(val, 0) <= i) { *out_val = sign * i; *out_mult = mult; *out_is_float = 0; return 0; }
        } else if (r > 0.9) {
            double next = i + 1;
            if (nextafter(val, next) >= next) { *out_val = sign * next; *out_mult = mult; *out_is_float = 0; return 0; }
        }
    }
    *out_val = v; *out_mult = 0; *out_is_float = 1;
    return 0;
}
static inline double convert_from_int_float(double val, uint8_t mult) { /* m3tsz.go:121-127 */
    if (mult == 0) return val;
    return val / MULTIPLIERS[mult];
}

/* =========================== encoder state =========================== */

typedef struct {
    /* TimestampEncoder (timestamp_encoder.go:37-54) */
    int64_t prev_time;
    int64_t prev_time_delta;
    uint64_t prev_ann_checksum;
    uint8_t time_unit;
    int tu_encoded_manually;
    int has_written_first;
    /* FloatEncoderAndIterator (float_encoder_iterator.go:36-43) */
    uint64_t prev_xor, prev_float_bits;
    /* IntSigBitsTracker (int_sig_bits_tracker.go:27-31) */
    uint8_t num_sig, cur_highest_lower_sig, num_lower_sig;
    /* encoder (encoder.go:42-61) */
    double int_val;
    uint32_t num_encoded;
    uint8_t max_mult;
    int int_optimized;
    int is_float;
} m3tsz_enc;

/* encoder.go:248-258 initialTimeUnit (timestamp_encoder.go:248-259) */
static uint8_t initial_time_unit(int64_t start, uint8_t tu) {
    if (!unit_is_valid(tu)) return M3_UNIT_NONE;
    int64_t tv = UNIT_NS[tu];
    if (start % tv == 0) return tu;
    return M3_UNIT_NONE;
}

static void enc_init(m3tsz_enc* e, int64_t start_ns, int int_optimized, uint8_t default_unit) {
    memset(e, 0, sizeof(*e));
    e->prev_time = start_ns;
    e->time_unit = initial_time_unit(start_ns, default_unit);
    e->prev_ann_checksum = EMPTY_ANN_CHECKSUM;
    e->int_optimized = int_optimized;
}

/* WriteSpecialMarker, scheme.go:217-220 */
static void write_special_marker(m3_ostream* os, int marker) {
    os_write_bits(os, MARKER_OPCODE, MARKER_OPCODE_BITS);
    os_write_bits(os, (uint64_t)marker, MARKER_VALUE_BITS);
}

/* binary.PutVarint (zigzag varint, Go encoding/binary) */
static int put_varint(uint8_t* buf, int64_t x) {
    uint64_t ux = ((uint64_t)x) << 1;
    if (x < 0) ux = ~ux;
    int i = 0;
    while (ux >= 0x80) { buf[i++] = (uint8_t)(ux) | 0x80; ux >>= 7; }
    buf[i++] = (uint8_t)ux;
    return i;
}

/* timestamp_encoder.go:172-195 writeAnnotation */
static void enc_write_annotation(m3tsz_enc* e, m3_ostream* os, const uint8_t* ant, int64_t ant_len) {
    if (ant_len == 0) return; /* shouldWriteAnnotation :164-166 */
    uint64_t checksum = oracle_xxhash64(ant, (size_t)ant_len);
    if (checksum == e->prev_ann_checksum) return;
    write_special_marker(os, MARKER_ANNOTATION);
    uint8_t buf[10];
    int n = put_varint(buf, ant_len - 1); /* :181-183 */
    os_write_bytes(os, buf, n);
    os_write_bytes(os, ant, ant_len);
    e->prev_ann_checksum = checksum;
}

/* timestamp_encoder.go:139-159 */
static int enc_maybe_write_time_unit_change(m3tsz_enc* e, m3_ostream* os, uint8_t time_unit) {
    if (!unit_is_valid(time_unit) || time_unit == e->time_unit) return 0;
    write_special_marker(os, MARKER_TIMEUNIT);
    os_write_byte(os, time_unit); /* WriteTimeUnit :133-137 */
    e->time_unit = time_unit;
    e->tu_encoded_manually = 1;
    return 1;
}

/* timestamp_encoder.go:205-246 */
static int enc_write_dod_unchanged(m3tsz_enc* e, m3_ostream* os,
                                   int64_t prev_delta, int64_t cur_delta, uint8_t time_unit) {
    (void)e;
    if (!unit_is_valid(time_unit)) return -M3_ERR_NO_SCHEME; /* timeUnit.Value() error */
    int64_t u = UNIT_NS[time_unit];
    int64_t dod = (cur_delta - prev_delta) / u; /* ToNormalizedDuration, time.go:55-57 */
    if (time_unit == M3_UNIT_MILLISECOND || time_unit == M3_UNIT_SECOND) {
        int32_t dod32 = (int32_t)dod;
        if ((int64_t)dod32 != dod) return -M3_ERR_DOD_OVERFLOW;
    }
    int default_bits = scheme_default_bits(time_unit);
    if (default_bits == 0) return -M3_ERR_NO_SCHEME;
    if (dod == 0) {
        os_write_bits(os, 0x0, 1); /* zero bucket */
        return 0;
    }
    for (int i = 0; i < 3; i++) {
        int nb = BUCKET_VALUE_BITS[i];
        int64_t bmin = -((int64_t)1 << (nb - 1));
        int64_t bmax = ((int64_t)1 << (nb - 1)) - 1;
        if (dod >= bmin && dod <= bmax) {
            os_write_bits(os, BUCKET_OPCODE[i], BUCKET_OPCODE_BITS[i]);
            os_write_bits(os, (uint64_t)dod, nb);
            return 0;
        }
    }
    os_write_bits(os, DEFAULT_BUCKET_OPCODE, DEFAULT_BUCKET_OPCODE_BITS);
    os_write_bits(os, (uint64_t)dod, default_bits);
    return 0;
}

/* timestamp_encoder.go:103-129 WriteNextTime (+ :197-203 unit-changed dod) */
static int enc_write_next_time(m3tsz_enc* e, m3_ostream* os, int64_t cur_time,
                               const uint8_t* ant, int64_t ant_len, uint8_t time_unit) {
    enc_write_annotation(e, os, ant, ant_len);
    int tu_changed = enc_maybe_write_time_unit_change(e, os, time_unit);

    int64_t time_delta = cur_time - e->prev_time;
    e->prev_time = cur_time;
    if (tu_changed || e->tu_encoded_manually) {
        int64_t dod_ns = time_delta - e->prev_time_delta;
        os_write_bits(os, (uint64_t)dod_ns, 64);
        e->prev_time_delta = 0;
        e->tu_encoded_manually = 0;
        return 0;
    }
    int err = enc_write_dod_unchanged(e, os, e->prev_time_delta, time_delta, time_unit);
    e->prev_time_delta = time_delta;
    return err;
}

/* timestamp_encoder.go:72-101 WriteTime / WriteFirstTime */
static int enc_write_time(m3tsz_enc* e, m3_ostream* os, int64_t cur_time,
                          const uint8_t* ant, int64_t ant_len, uint8_t time_unit) {
    if (!e->has_written_first) {
        os_write_bits(os, (uint64_t)e->prev_time, 64); /* start time, ns (:96-99) */
        int err = enc_write_next_time(e, os, cur_time, ant, ant_len, time_unit);
        if (err) return err;
        e->has_written_first = 1;
        return 0;
    }
    return enc_write_next_time(e, os, cur_time, ant, ant_len, time_unit);
}

/* float_encoder_iterator.go:69-103 */
static void write_full_float(m3tsz_enc* e, m3_ostream* os, uint64_t val) {
    e->prev_float_bits = val;
    e->prev_xor = val;
    os_write_bits(os, val, 64);
}
static void write_xor(m3tsz_enc* e, m3_ostream* os, uint64_t cur_xor) {
    if (cur_xor == 0) { os_write_bits(os, OPCODE_ZERO_VALUE_XOR, 1); return; }
    int prev_lead, prev_trail, cur_lead, cur_trail;
    leading_trailing_zeros(e->prev_xor, &prev_lead, &prev_trail);
    leading_trailing_zeros(cur_xor, &cur_lead, &cur_trail);
    if (cur_lead >= prev_lead && cur_trail >= prev_trail) {
        os_write_bits(os, OPCODE_CONTAINED_VALUE_XOR, 2);
        os_write_bits(os, cur_xor >> prev_trail, 64 - prev_lead - prev_trail);
        return;
    }
    os_write_bits(os, OPCODE_UNCONTAINED_VALUE_XOR, 2);
    os_write_bits(os, (uint64_t)cur_lead, 6);
    int num_meaningful = 64 - cur_lead - cur_trail;
    os_write_bits(os, (uint64_t)(num_meaningful - 1), 6);
    os_write_bits(os, cur_xor >> cur_trail, num_meaningful);
}
static void write_next_float(m3tsz_enc* e, m3_ostream* os, uint64_t val) {
    uint64_t xor = e->prev_float_bits ^ val;
    write_xor(e, os, xor);
    e->prev_xor = xor;
    e->prev_float_bits = val;
}

/* int_sig_bits_tracker.go:35-62 */
static void tracker_write_int_val_diff(m3tsz_enc* e, m3_ostream* os, uint64_t val_bits, int neg) {
    os_write_bit(os, neg ? OPCODE_NEGATIVE : OPCODE_POSITIVE);
    os_write_bits(os, val_bits, e->num_sig);
}
static void tracker_write_int_sig(m3tsz_enc* e, m3_ostream* os, uint8_t sig) {
    if (e->num_sig != sig) {
        os_write_bit(os, OPCODE_UPDATE_SIG);
        if (sig == 0) os_write_bit(os, OPCODE_ZERO_SIG);
        else {
            os_write_bit(os, OPCODE_NONZERO_SIG);
            os_write_bits(os, (uint64_t)(sig - 1), NUM_SIG_BITS);
        }
    } else {
        os_write_bit(os, OPCODE_NO_UPDATE_SIG);
    }
    e->num_sig = sig;
}
/* int_sig_bits_tracker.go:68-91 */
static uint8_t tracker_track_new_sig(m3tsz_enc* e, uint8_t nsig) {
    uint8_t new_sig = e->num_sig;
    if (nsig > e->num_sig) {
        new_sig = nsig;
    } else if (e->num_sig - nsig >= SIG_DIFF_THRESHOLD) {
        if (e->num_lower_sig == 0) e->cur_highest_lower_sig = nsig;
        else if (nsig > e->cur_highest_lower_sig) e->cur_highest_lower_sig = nsig;
        e->num_lower_sig++;
        if (e->num_lower_sig >= SIG_REPEAT_THRESHOLD) {
            new_sig = e->cur_highest_lower_sig;
            e->num_lower_sig = 0;
        }
    } else {
        e->num_lower_sig = 0;
    }
    return new_sig;
}

/* encoder.go:233-250 writeIntSigMult */
static void enc_write_int_sig_mult(m3tsz_enc* e, m3_ostream* os, uint8_t sig, uint8_t mult, int float_changed) {
    tracker_write_int_sig(e, os, sig);
    if (mult > e->max_mult) {
        os_write_bit(os, OPCODE_UPDATE_MULT);
        os_write_bits(os, (uint64_t)mult, NUM_MULT_BITS);
        e->max_mult = mult;
    } else if (e->num_sig == sig && e->max_mult == mult && float_changed) {
        os_write_bit(os, OPCODE_UPDATE_MULT);
        os_write_bits(os, (uint64_t)e->max_mult, NUM_MULT_BITS);
    } else {
        os_write_bit(os, OPCODE_NO_UPDATE_MULT);
    }
}

/* encoder.go:112-146 writeFirstValue */
static int enc_write_first_value(m3tsz_enc* e, m3_ostream* os, double v) {
    if (!e->int_optimized) { write_full_float(e, os, f2bits(v)); return 0; }
    double val; uint8_t mult; int is_float;
    int err = convert_to_int_float(v, 0, &val, &mult, &is_float);
    if (err) return err;
    if (is_float) {
        os_write_bit(os, OPCODE_FLOAT_MODE);
        write_full_float(e, os, f2bits(v));
        e->is_float = 1;
        e->max_mult = mult;
        return 0;
    }
    os_write_bit(os, OPCODE_INT_MODE);
    e->int_val = val;
    int neg_diff = 1;
    if (val < 0) { neg_diff = 0; val = -1 * val; }
    uint64_t val_bits = (uint64_t)go_f2i(val);
    uint8_t nsig = num_sig(val_bits);
    enc_write_int_sig_mult(e, os, nsig, mult, 0);
    tracker_write_int_val_diff(e, os, val_bits, neg_diff);
    return 0;
}

/* encoder.go:174-197 writeFloatVal */
static void enc_write_float_val(m3tsz_enc* e, m3_ostream* os, uint64_t val, uint8_t mult) {
    if (!e->is_float) {
        os_write_bit(os, OPCODE_UPDATE);
        os_write_bit(os, OPCODE_NO_REPEAT);
        os_write_bit(os, OPCODE_FLOAT_MODE);
        write_full_float(e, os, val);
        e->is_float = 1;
        e->max_mult = mult;
        return;
    }
    if (val == e->prev_float_bits) {
        os_write_bit(os, OPCODE_UPDATE);
        os_write_bit(os, OPCODE_REPEAT);
        return;
    }
    os_write_bit(os, OPCODE_NO_UPDATE);
    write_next_float(e, os, val);
}

/* encoder.go:199-231 writeIntVal */
static void enc_write_int_val(m3tsz_enc* e, m3_ostream* os, double val, uint8_t mult, int is_float, double val_diff) {
    if (val_diff == 0 && is_float == e->is_float && mult == e->max_mult) {
        os_write_bit(os, OPCODE_UPDATE);
        os_write_bit(os, OPCODE_REPEAT);
        return;
    }
    int neg = 0;
    if (val_diff < 0) { neg = 1; val_diff = -1 * val_diff; }
    uint64_t val_diff_bits = (uint64_t)go_f2i(val_diff);
    uint8_t nsig = num_sig(val_diff_bits);
    uint8_t new_sig = tracker_track_new_sig(e, nsig);
    int is_float_changed = (is_float != e->is_float);
    if (mult > e->max_mult || e->num_sig != new_sig || is_float_changed) {
        os_write_bit(os, OPCODE_UPDATE);
        os_write_bit(os, OPCODE_NO_REPEAT);
        os_write_bit(os, OPCODE_INT_MODE);
        enc_write_int_sig_mult(e, os, new_sig, mult, is_float_changed);
        tracker_write_int_val_diff(e, os, val_diff_bits, neg);
        e->is_float = 0;
    } else {
        os_write_bit(os, OPCODE_NO_UPDATE);
        tracker_write_int_val_diff(e, os, val_diff_bits, neg);
    }
    e->int_val = val;
}

/* encoder.go:148-172 writeNextValue */
static int enc_write_next_value(m3tsz_enc* e, m3_ostream* os, double v) {
    if (!e->int_optimized) { write_next_float(e, os, f2bits(v)); return 0; }
    double val; uint8_t mult; int is_float;
    int err = convert_to_int_float(v, e->max_mult, &val, &mult, &is_float);
    if (err) return err;
    double val_diff = 0;
    if (!is_float) val_diff = e->int_val - val;
    if (is_float || val_diff >= MAX_INT || val_diff <= MIN_INT) {
        enc_write_float_val(e, os, f2bits(val), mult);
        return 0;
    }
    enc_write_int_val(e, os, val, mult, is_float, val_diff);
    return 0;
}

/* encoder.go:89-110 Encode */
static int enc_encode(m3tsz_enc* e, m3_ostream* os, int64_t t_ns, double v,
                      uint8_t unit, const uint8_t* ant, int64_t ant_len) {
    int err = enc_write_time(e, os, t_ns, ant, ant_len, unit);
    if (err) return err;
    if (e->num_encoded == 0) err = enc_write_first_value(e, os, v);
    else err = enc_write_next_value(e, os, v);
    if (err == 0) e->num_encoded++;
    return err;
}

/* scheme.go:198-212 Tail + encoder.go:394-429 segment construction:
 * final stream = raw[:len-1] + (top `pos` bits of last byte + EOS marker). */
static int64_t enc_finalize(const m3_ostream* os, uint8_t* out, int64_t out_cap) {
    if (os->len == 0) return 0;
    uint8_t last = os->buf[os->len - 1];
    int pos = os->pos;
    m3_ostream tail;
    os_init(&tail, 8);
    os_write_bits(&tail, ((uint64_t)last) >> (8 - pos), pos);
    write_special_marker(&tail, MARKER_EOS);
    int64_t total = (os->len - 1) + tail.len;
    if (total > out_cap) { os_free(&tail); return -M3_ERR_CAPACITY; }
    memcpy(out, os->buf, (size_t)(os->len - 1));
    memcpy(out + os->len - 1, tail.buf, (size_t)tail.len);
    os_free(&tail);
    return total;
}

/* =========================== decoder state =========================== */

typedef struct {
    m3_istream is;
    /* TimestampIterator (timestamp_iterator.go:41-64) */
    int64_t prev_time;
    int64_t prev_time_delta;
    uint8_t time_unit;
    uint8_t default_unit;
    int have_scheme;       /* timeEncodingScheme != nil */
    uint8_t scheme_unit;   /* unit of current scheme */
    int tu_changed;
    int done;
    /* readerIterator (iterator.go:47-64) */
    double int_val;
    uint64_t prev_float_bits, prev_xor;
    uint8_t mult, sig;
    int int_optimized, is_float;
    int err;
    /* annotation scratch */
    uint8_t ann[4096];
    int64_t ann_len; /* -1 = none this point */
} m3tsz_dec;

static void dec_init(m3tsz_dec* d, const uint8_t* data, int64_t len, int int_optimized, uint8_t default_unit) {
    memset(d, 0, sizeof(*d));
    is_init(&d->is, data, len);
    d->default_unit = default_unit;
    d->int_optimized = int_optimized;
    d->ann_len = -1;
}

static int dec_read_dod(m3tsz_dec* d, int64_t* out_dod);

/* timestamp_iterator.go:115-135 ReadTimeUnit */
static int dec_read_time_unit(m3tsz_dec* d) {
    uint64_t tu_bits;
    int err = is_read_bits(&d->is, 8, &tu_bits);
    if (err) return err;
    uint8_t tu = (uint8_t)tu_bits;
    if (unit_is_valid(tu) && tu != d->time_unit) {
        d->tu_changed = 1;
        if (scheme_default_bits(tu) != 0) { d->have_scheme = 1; d->scheme_unit = tu; }
    }
    d->time_unit = tu;
    return 0;
}

/* binary.ReadVarint over istream (ReadByte) */
static int dec_read_varint(m3tsz_dec* d, int64_t* out) {
    uint64_t ux = 0;
    int shift = 0;
    for (int i = 0; i < 10; i++) {
        uint64_t b;
        int err = is_read_bits(&d->is, 8, &b);
        if (err) return err;
        ux |= (b & 0x7f) << shift;
        if ((b & 0x80) == 0) {
            int64_t x = (int64_t)(ux >> 1);
            if (ux & 1) x = ~x;
            *out = x;
            return 0;
        }
        shift += 7;
    }
    return -M3_ERR_ANNOTATION;
}

/* timestamp_iterator.go:327-356 readAnnotation */
static int dec_read_annotation(m3tsz_dec* d) {
    int64_t ant_len;
    int err = dec_read_varint(d, &ant_len);
    if (err) return err;
    ant_len += 1;
    if (ant_len <= 0) return -M3_ERR_ANNOTATION;
    if (ant_len > (int64_t)sizeof(d->ann)) return -M3_ERR_ANNOTATION;
    for (int64_t i = 0; i < ant_len; i++) {
        uint64_t b;
        err = is_read_bits(&d->is, 8, &b);
        if (err) return err;
        d->ann[i] = (uint8_t)b;
    }
    d->ann_len = ant_len;
    return 0;
}

static int dec_read_marker_or_dod(m3tsz_dec* d, int64_t* out_dod);

/* timestamp_iterator.go:174-235 tryReadMarker.
 * Returns: 1 = marker consumed (dod in *out_dod, or Done), 0 = not a marker,
 * negative = error. */
static int dec_try_read_marker(m3tsz_dec* d, int64_t* out_dod) {
    uint64_t opcode_and_value;
    int err = is_peek_bits(&d->is, MARKER_OPCODE_BITS + MARKER_VALUE_BITS, &opcode_and_value);
    if (err) return 0; /* :182-184: peek error => not a marker */
    uint64_t opcode = opcode_and_value >> MARKER_VALUE_BITS;
    if (opcode != MARKER_OPCODE) return 0;
    uint64_t marker = opcode_and_value & ((1 << MARKER_VALUE_BITS) - 1);
    uint64_t discard;
    switch (marker) {
    case MARKER_EOS:
        err = is_read_bits(&d->is, MARKER_OPCODE_BITS + MARKER_VALUE_BITS, &discard);
        if (err) return err;
        d->done = 1;
        *out_dod = 0;
        return 1;
    case MARKER_ANNOTATION:
        err = is_read_bits(&d->is, MARKER_OPCODE_BITS + MARKER_VALUE_BITS, &discard);
        if (err) return err;
        err = dec_read_annotation(d);
        if (err) return err;
        err = dec_read_marker_or_dod(d, out_dod);
        if (err) return err;
        return 1;
    case MARKER_TIMEUNIT:
        err = is_read_bits(&d->is, MARKER_OPCODE_BITS + MARKER_VALUE_BITS, &discard);
        if (err) return err;
        err = dec_read_time_unit(d);
        if (err) return err;
        err = dec_read_marker_or_dod(d, out_dod);
        if (err) return err;
        return 1;
    default:
        return 0;
    }
}

/* timestamp_iterator.go:307-325 readFullTimestamp */
static int dec_read_full_timestamp(m3tsz_dec* d, int64_t* out_dod) {
    if (scheme_default_bits(d->time_unit) == 0) return -M3_ERR_NO_SCHEME;
    d->have_scheme = 1;
    d->scheme_unit = d->time_unit;
    uint64_t dod_bits;
    int err = is_read_bits(&d->is, 64, &dod_bits);
    if (err) return err;
    *out_dod = sign_extend(dod_bits, 64);
    return 0;
}

/* timestamp_iterator.go:250-305 readDeltaOfDelta */
static int dec_read_dod(m3tsz_dec* d, int64_t* out_dod) {
    if (d->tu_changed) return dec_read_full_timestamp(d, out_dod);
    if (!d->have_scheme) return -M3_ERR_NO_SCHEME;
    uint64_t cb;
    int err = is_read_bits(&d->is, 1, &cb);
    if (err) return err;
    if (cb == 0x0) { *out_dod = 0; return 0; } /* zero bucket opcode */
    for (int i = 0; i < 3; i++) {
        uint64_t next_cb;
        err = is_read_bits(&d->is, 1, &next_cb);
        if (err) { *out_dod = 0; return 0; } /* reference swallows this error (:271-274) */
        cb = (cb << 1) | next_cb;
        if (cb == BUCKET_OPCODE[i]) {
            uint64_t dod_bits;
            err = is_read_bits(&d->is, (uint8_t)BUCKET_VALUE_BITS[i], &dod_bits);
            if (err) return err;
            int64_t dod = sign_extend(dod_bits, (uint8_t)BUCKET_VALUE_BITS[i]);
            if (!unit_is_valid(d->time_unit)) { *out_dod = 0; return 0; } /* :284-287 swallowed */
            *out_dod = dod * UNIT_NS[d->time_unit];
            return 0;
        }
    }
    int default_bits = scheme_default_bits(d->scheme_unit);
    uint64_t dod_bits;
    err = is_read_bits(&d->is, (uint8_t)default_bits, &dod_bits);
    if (err) return err;
    int64_t dod = sign_extend(dod_bits, (uint8_t)default_bits);
    if (!unit_is_valid(d->time_unit)) { *out_dod = 0; return 0; }
    *out_dod = dod * UNIT_NS[d->time_unit];
    return 0;
}

/* timestamp_iterator.go:237-248 */
static int dec_read_marker_or_dod(m3tsz_dec* d, int64_t* out_dod) {
    int r = dec_try_read_marker(d, out_dod);
    if (r < 0) return r;
    if (r == 1 || d->done) return 0;
    return dec_read_dod(d, out_dod);
}

/* timestamp_iterator.go:137-161 readFirstTimestamp */
static int dec_read_first_timestamp(m3tsz_dec* d) {
    uint64_t nt_bits;
    int err = is_read_bits(&d->is, 64, &nt_bits);
    if (err) return err;
    int64_t nt = (int64_t)nt_bits;
    if (d->time_unit == M3_UNIT_NONE) d->time_unit = initial_time_unit(nt, d->default_unit);
    if (scheme_default_bits(d->time_unit) != 0) { d->have_scheme = 1; d->scheme_unit = d->time_unit; }
    int64_t dod;
    err = dec_read_marker_or_dod(d, &dod);
    if (err) return err;
    if (!d->done) {
        d->prev_time_delta += dod;
    }
    d->prev_time = nt + d->prev_time_delta;
    return 0;
}

/* timestamp_iterator.go:80-113 ReadTimestamp. Returns 0 ok / -err.
 * Sets *first. Done flag is d->done. */
static int dec_read_timestamp(m3tsz_dec* d, int* first) {
    d->ann_len = -1;
    *first = 0;
    int err;
    if (d->prev_time != 0) {
        int64_t dod;
        err = dec_read_marker_or_dod(d, &dod);
        if (err == 0 && !d->done) {
            d->prev_time_delta += dod;
            d->prev_time += d->prev_time_delta;
        }
    } else {
        *first = 1;
        err = dec_read_first_timestamp(d);
    }
    if (err) return err;
    if (d->tu_changed) {
        d->prev_time_delta = 0;
        d->tu_changed = 0;
    }
    return 0;
}

/* float_encoder_iterator.go:105-165 */
static int dec_read_full_float(m3tsz_dec* d) {
    uint64_t vb;
    int err = is_read_bits(&d->is, 64, &vb);
    if (err) return err;
    d->prev_float_bits = vb;
    d->prev_xor = vb;
    return 0;
}
static int dec_read_next_float(m3tsz_dec* d) {
    uint64_t cb;
    int err = is_read_bits(&d->is, 1, &cb);
    if (err) return err;
    if (cb == OPCODE_ZERO_VALUE_XOR) { d->prev_xor = 0; return 0; }
    uint64_t next_cb;
    err = is_read_bits(&d->is, 1, &next_cb);
    if (err) return err;
    cb = (cb << 1) | next_cb;
    if (cb == OPCODE_CONTAINED_VALUE_XOR) {
        int prev_lead, prev_trail;
        leading_trailing_zeros(d->prev_xor, &prev_lead, &prev_trail);
        uint8_t num_meaningful = (uint8_t)(64 - prev_lead - prev_trail);
        uint64_t meaningful;
        err = is_read_bits(&d->is, num_meaningful, &meaningful);
        if (err) return err;
        d->prev_xor = meaningful << prev_trail;
        d->prev_float_bits ^= d->prev_xor;
        return 0;
    }
    uint64_t lead_and_meaningful;
    err = is_read_bits(&d->is, 12, &lead_and_meaningful);
    if (err) return err;
    uint64_t num_leading = (lead_and_meaningful & 4032) >> 6;
    uint64_t num_meaningful = (lead_and_meaningful & 63) + 1;
    uint64_t meaningful;
    err = is_read_bits(&d->is, (uint8_t)num_meaningful, &meaningful);
    if (err) return err;
    uint64_t num_trailing = 64 - num_leading - num_meaningful;
    d->prev_xor = meaningful << num_trailing;
    d->prev_float_bits ^= d->prev_xor;
    return 0;
}

/* iterator.go:178-219 */
static int dec_read_int_sig_mult(m3tsz_dec* d) {
    uint64_t b;
    int err = is_read_bits(&d->is, 1, &b);
    if (err) return err;
    if (b == OPCODE_UPDATE_SIG) {
        err = is_read_bits(&d->is, 1, &b);
        if (err) return err;
        if (b == OPCODE_ZERO_SIG) d->sig = 0;
        else {
            uint64_t s;
            err = is_read_bits(&d->is, NUM_SIG_BITS, &s);
            if (err) return err;
            d->sig = (uint8_t)s + 1;
        }
    }
    err = is_read_bits(&d->is, 1, &b);
    if (err) return err;
    if (b == OPCODE_UPDATE_MULT) {
        uint64_t m;
        err = is_read_bits(&d->is, NUM_MULT_BITS, &m);
        if (err) return err;
        d->mult = (uint8_t)m;
        if (d->mult > MAX_MULT) return -M3_ERR_INVALID_MULT;
    }
    return 0;
}
static int dec_read_int_val_diff(m3tsz_dec* d) {
    if (d->sig == 64) {
        /* readIntValDiffSlow, iterator.go:212-219 */
        uint64_t sb;
        int err = is_read_bits(&d->is, 1, &sb);
        if (err) return err;
        double sign = (sb == OPCODE_NEGATIVE) ? 1.0 : -1.0;
        uint64_t bits;
        err = is_read_bits(&d->is, d->sig, &bits);
        if (err) return err;
        d->int_val += sign * (double)bits;
        return 0;
    }
    uint64_t bits;
    int err = is_read_bits(&d->is, d->sig + 1, &bits);
    if (err) return err;
    double sign = -1.0;
    if ((bits >> d->sig) == OPCODE_NEGATIVE) {
        sign = 1.0;
        bits ^= (1ULL << d->sig);
    }
    d->int_val += sign * (double)bits;
    return 0;
}

/* iterator.go:108-126 readFirstValue */
static int dec_read_first_value(m3tsz_dec* d) {
    if (!d->int_optimized) return dec_read_full_float(d);
    uint64_t b;
    int err = is_read_bits(&d->is, 1, &b);
    if (err) return err;
    if (b == OPCODE_FLOAT_MODE) {
        err = dec_read_full_float(d);
        if (err) return err;
        d->is_float = 1;
        return 0;
    }
    err = dec_read_int_sig_mult(d);
    if (err) return err;
    return dec_read_int_val_diff(d);
}

/* iterator.go:128-176 readNextValue */
static int dec_read_next_value(m3tsz_dec* d) {
    if (!d->int_optimized) return dec_read_next_float(d);
    uint64_t b;
    int err = is_read_bits(&d->is, 1, &b);
    if (err) return err;
    if (b == OPCODE_UPDATE) {
        err = is_read_bits(&d->is, 1, &b);
        if (err) return err;
        if (b == OPCODE_REPEAT) return 0;
        err = is_read_bits(&d->is, 1, &b);
        if (err) return err;
        if (b == OPCODE_FLOAT_MODE) {
            err = dec_read_full_float(d);
            if (err) return err;
            d->is_float = 1;
            return 0;
        }
        err = dec_read_int_sig_mult(d);
        if (err) return err;
        err = dec_read_int_val_diff(d);
        if (err) return err;
        d->is_float = 0;
        return 0;
    }
    if (d->is_float) return dec_read_next_float(d);
    return dec_read_int_val_diff(d);
}

/* iterator.go:81-106 Next. Returns 1 = value, 0 = done, -err. */
static int dec_next(m3tsz_dec* d, int64_t* t_ns, double* val, uint8_t* unit) {
    if (d->err || d->done) return d->err ? d->err : 0;
    int first;
    int err = dec_read_timestamp(d, &first);
    if (err) { d->err = err; return err; }
    if (d->done) return 0;
    if (!first) err = dec_read_next_value(d);
    else err = dec_read_first_value(d);
    if (err) { d->err = err; return err; }
    *t_ns = d->prev_time;
    if (!d->int_optimized || d->is_float) *val = bits2f(d->prev_float_bits);
    else *val = convert_from_int_float(d->int_val, d->mult);
    *unit = d->time_unit;
    return 1;
}

/* ========================= exported surface ========================= */

static int64_t encode_series_impl(
    const int64_t* ts_ns, const double* vals, const uint8_t* units,
    const int32_t* ann_offsets, const uint8_t* ann_bytes,
    int32_t npts, int64_t start_ns, int int_optimized,
    m3_ostream* os) {
    m3tsz_enc e;
    enc_init(&e, start_ns, int_optimized, M3_UNIT_SECOND); /* opts default, options.go:32 */
    for (int32_t i = 0; i < npts; i++) {
        uint8_t unit = units ? units[i] : M3_UNIT_SECOND;
        const uint8_t* ant = NULL;
        int64_t ant_len = 0;
        if (ann_offsets) {
            ant = ann_bytes + ann_offsets[i];
            ant_len = ann_offsets[i + 1] - ann_offsets[i];
        }
        int err = enc_encode(&e, os, ts_ns[i], vals[i], unit, ant, ant_len);
        if (err) return err;
    }
    return 0;
}

int64_t oracle_encode_series(
    const int64_t* ts_ns, const double* vals, const uint8_t* units,
    const int32_t* ann_offsets, const uint8_t* ann_bytes,
    int32_t npts, int64_t start_ns, int int_optimized,
    uint8_t* out, int64_t out_cap) {
    m3_ostream os;
    os_init(&os, 1024);
    int64_t r = encode_series_impl(ts_ns, vals, units, ann_offsets, ann_bytes,
                                   npts, start_ns, int_optimized, &os);
    if (r < 0) { os_free(&os); return r; }
    int64_t n = enc_finalize(&os, out, out_cap);
    os_free(&os);
    return n;
}

int64_t oracle_encode_series_raw(
    const int64_t* ts_ns, const double* vals, const uint8_t* units,
    const int32_t* ann_offsets, const uint8_t* ann_bytes,
    int32_t npts, int64_t start_ns, int int_optimized,
    uint8_t* out, int64_t out_cap, int32_t* out_pos) {
    m3_ostream os;
    os_init(&os, 1024);
    int64_t r = encode_series_impl(ts_ns, vals, units, ann_offsets, ann_bytes,
                                   npts, start_ns, int_optimized, &os);
    if (r < 0) { os_free(&os); return r; }
    if (os.len > out_cap) { os_free(&os); return -M3_ERR_CAPACITY; }
    memcpy(out, os.buf, (size_t)os.len);
    *out_pos = os.pos;
    int64_t n = os.len;
    os_free(&os);
    return n;
}

int64_t oracle_decode_series(
    const uint8_t* data, int64_t len, int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint8_t* out_units,
    int32_t* out_ann_lens, uint8_t* out_ann_bytes, int64_t ann_cap,
    int64_t cap) {
    m3tsz_dec d;
    dec_init(&d, data, len, int_optimized, default_unit);
    int64_t n = 0;
    int64_t ann_used = 0;
    for (;;) {
        int64_t t = 0; double v = 0; uint8_t u = 0;
        int r = dec_next(&d, &t, &v, &u);
        if (r == 0) break;
        if (r < 0) return r;
        if (n >= cap) return -M3_ERR_CAPACITY;
        out_ts[n] = t;
        out_vals[n] = v;
        if (out_units) out_units[n] = u;
        if (out_ann_lens) {
            if (d.ann_len >= 0) {
                if (ann_used + d.ann_len > ann_cap) return -M3_ERR_CAPACITY;
                memcpy(out_ann_bytes + ann_used, d.ann, (size_t)d.ann_len);
                ann_used += d.ann_len;
                out_ann_lens[n] = (int32_t)d.ann_len;
            } else {
                out_ann_lens[n] = -1;
            }
        }
        n++;
    }
    return n;
}

int oracle_encode_batch(
    const int64_t* ts_ns, const double* vals, const uint32_t* counts,
    int64_t nseries, int64_t stride, int int_optimized, uint8_t unit,
    uint8_t* out_bytes, int64_t out_stride, uint32_t* out_lens, int nthreads) {
    int err = 0;
    (void)nthreads;
#pragma omp parallel for schedule(dynamic, 64) num_threads(nthreads)
    for (int64_t i = 0; i < nseries; i++) {
        if (err) continue;
        m3_ostream os;
        os_init(&os, 1024);
        m3tsz_enc e;
        enc_init(&e, ts_ns[i * stride], int_optimized, M3_UNIT_SECOND);
        int le = 0;
        for (uint32_t j = 0; j < counts[i]; j++) {
            le = enc_encode(&e, &os, ts_ns[i * stride + j], vals[i * stride + j], unit, NULL, 0);
            if (le) break;
        }
        if (!le) {
            int64_t n = enc_finalize(&os, out_bytes + i * out_stride, out_stride);
            if (n < 0) le = (int)-n;
            else out_lens[i] = (uint32_t)n;
        }
        os_free(&os);
        if (le) {
#pragma omp critical
            err = le > 0 ? -le : le;
        }
    }
    return err;
}

int oracle_decode_batch(
    const uint8_t* blobs, const uint64_t* offsets, int64_t nseries,
    int int_optimized, uint8_t default_unit,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts, int64_t stride,
    int nthreads) {
    int err = 0;
    (void)nthreads;
#pragma omp parallel for schedule(dynamic, 64) num_threads(nthreads)
    for (int64_t i = 0; i < nseries; i++) {
        if (err) continue;
        int64_t n = oracle_decode_series(
            blobs + offsets[i], (int64_t)(offsets[i + 1] - offsets[i]),
            int_optimized, default_unit,
            out_ts + i * stride, out_vals + i * stride, NULL,
            NULL, NULL, 0, stride);
        if (n < 0) {
#pragma omp critical
            err = (int)n;
        } else {
            out_counts[i] = (uint32_t)n;
        }
    }
    return err;
}

/* ================== test-only wrappers (golden vectors) ==================
 * These mirror how the reference's unit tests call internals directly:
 * encoder_test.go:54-123,125-155,172-205; iterator_test.go:44-115. */

/* encoder_test.go:54-81 TestWriteDeltaOfDeltaTimeUnitUnchanged */
int64_t oracle_test_write_dod_unchanged(int64_t prev_delta_ns, int64_t cur_delta_ns,
                                        uint8_t unit, uint8_t* out, int64_t cap,
                                        int32_t* out_pos) {
    m3_ostream os; os_init(&os, 64);
    m3tsz_enc e; memset(&e, 0, sizeof(e));
    int err = enc_write_dod_unchanged(&e, &os, prev_delta_ns, cur_delta_ns, unit);
    if (err) { os_free(&os); return err; }
    if (os.len > cap) { os_free(&os); return -M3_ERR_CAPACITY; }
    memcpy(out, os.buf, (size_t)os.len);
    *out_pos = os.pos;
    int64_t n = os.len; os_free(&os);
    return n;
}

/* encoder_test.go:83-101 TestWriteDeltaOfDeltaTimeUnitChanged */
int64_t oracle_test_write_dod_changed(int64_t prev_delta_ns, int64_t cur_delta_ns,
                                      uint8_t* out, int64_t cap, int32_t* out_pos) {
    m3_ostream os; os_init(&os, 64);
    int64_t dod = cur_delta_ns - prev_delta_ns; /* timestamp_encoder.go:197-203 */
    os_write_bits(&os, (uint64_t)dod, 64);
    if (os.len > cap) { os_free(&os); return -M3_ERR_CAPACITY; }
    memcpy(out, os.buf, (size_t)os.len);
    *out_pos = os.pos;
    int64_t n = os.len; os_free(&os);
    return n;
}

/* encoder_test.go:103-123 TestWriteValue (writeXOR with seeded PrevXOR) */
int64_t oracle_test_write_xor(uint64_t prev_xor, uint64_t cur_xor,
                              uint8_t* out, int64_t cap, int32_t* out_pos) {
    m3_ostream os; os_init(&os, 64);
    m3tsz_enc e; memset(&e, 0, sizeof(e));
    e.prev_xor = prev_xor;
    write_xor(&e, &os, cur_xor);
    if (os.len > cap) { os_free(&os); return -M3_ERR_CAPACITY; }
    memcpy(out, os.buf, (size_t)os.len);
    *out_pos = os.pos;
    int64_t n = os.len; os_free(&os);
    return n;
}

/* encoder_test.go:125-155 TestWriteAnnotation (fresh TimestampEncoder) */
int64_t oracle_test_write_annotation(const uint8_t* ant, int64_t ant_len,
                                     uint8_t* out, int64_t cap, int32_t* out_pos) {
    m3_ostream os; os_init(&os, 64);
    m3tsz_enc e; memset(&e, 0, sizeof(e));
    e.prev_ann_checksum = EMPTY_ANN_CHECKSUM;
    enc_write_annotation(&e, &os, ant, ant_len);
    if (os.len > cap) { os_free(&os); return -M3_ERR_CAPACITY; }
    memcpy(out, os.buf, (size_t)os.len);
    *out_pos = os.pos;
    int64_t n = os.len; os_free(&os);
    return n;
}

/* encoder_test.go:172-205 TestWriteTimeUnit (TimeUnit starts at None) */
int64_t oracle_test_write_timeunit(uint8_t unit, uint8_t* out, int64_t cap,
                                   int32_t* out_pos, int32_t* out_changed) {
    m3_ostream os; os_init(&os, 64);
    m3tsz_enc e; memset(&e, 0, sizeof(e));
    e.time_unit = M3_UNIT_NONE;
    *out_changed = enc_maybe_write_time_unit_change(&e, &os, unit);
    if (os.len > cap) { os_free(&os); return -M3_ERR_CAPACITY; }
    memcpy(out, os.buf, (size_t)os.len);
    *out_pos = os.pos;
    int64_t n = os.len; os_free(&os);
    return n;
}

/* iterator_test.go:44-88 TestReaderIteratorReadNextTimestamp */
int oracle_test_read_next_timestamp(const uint8_t* data, int64_t len,
                                    uint8_t unit, int64_t prev_delta_ns,
                                    int64_t* out_delta_ns) {
    m3tsz_dec d;
    dec_init(&d, data, len, 0, M3_UNIT_SECOND);
    d.time_unit = unit;
    d.prev_time_delta = prev_delta_ns;
    if (scheme_default_bits(unit) != 0) { d.have_scheme = 1; d.scheme_unit = unit; }
    int64_t dod;
    int err = dec_read_marker_or_dod(&d, &dod); /* readNextTimestamp :163-172 */
    if (err) return err;
    d.prev_time_delta += dod;
    d.prev_time += d.prev_time_delta;
    *out_delta_ns = d.prev_time_delta;
    return 0;
}

/* iterator_test.go:90-115 TestReaderIteratorReadNextValue */
int oracle_test_read_next_value(const uint8_t* data, int64_t len,
                                uint64_t prev_float_bits, uint64_t prev_xor,
                                uint64_t* out_xor, uint64_t* out_bits) {
    m3tsz_dec d;
    dec_init(&d, data, len, 0, M3_UNIT_SECOND);
    d.prev_float_bits = prev_float_bits;
    d.prev_xor = prev_xor;
    int err = dec_read_next_value(&d);
    if (err) return err;
    *out_xor = d.prev_xor;
    *out_bits = d.prev_float_bits;
    return 0;
}

/* convertToIntFloat direct access (m3tsz_test.go) */
int oracle_test_convert_to_int_float(double v, uint8_t cur_max_mult,
                                     double* out_val, uint8_t* out_mult, int32_t* out_is_float) {
    int isf = 0;
    int err = convert_to_int_float(v, cur_max_mult, out_val, out_mult, &isf);
    *out_is_float = isf;
    return err;
}
