/* unagg.cpp — native batch parser for the reference's unaggregated metric
 * wire format (m3aggregator ingest).
 *
 * Restates src/metrics/encoding/protobuf/{unaggregated_encoder.go:218-236,
 * unaggregated_iterator.go:87-131} framing — each message is a Go
 * binary.PutVarint (ZIGZAG varint) size prefix + a metricpb
 * MetricWithMetadatas protobuf (src/metrics/generated/proto/metricpb/
 * {metric.proto,composite.proto}) — and the protobuf wire layer for the
 * metric payloads:
 *   Counter    {1:id 2:int64 value 3:annotation 4:client_time}
 *   BatchTimer {1:id 2:repeated double values 3:annotation 4:client_time}
 *   Gauge      {1:id 2:double value 3:annotation 4:client_time}
 *   TimedMetric{1:type 2:id 3:time_nanos 4:double value 5:annotation}
 * Union types 1-3 (untimed counter/batch-timer/gauge), 5/6 (timed) and
 * 7 (timed with storage policy) are parsed; metadatas/policies are
 * retained as OPAQUE bytes (control-plane payload, passed through like
 * the reference's aggregator data path hands them to rule matching).
 * Unknown fields are skipped per proto wire rules; unknown union types
 * are an error (unaggregated_iterator.go checks the type enum).
 *
 * Output feeds the GPU path exactly like commitlog bootstrap: group the
 * per-metric values host-side, then batch-encode / rollup on device. */
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>
#include <mutex>

enum {
    M3GPU_UA_ERR_TRUNCATED = -121,
    M3GPU_UA_ERR_PROTO = -122,
    M3GPU_UA_ERR_TYPE = -123,
    M3GPU_UA_ERR_BADHANDLE = -124,
    M3GPU_UA_ERR_SIZE = -125,
};

/* MetricWithMetadatas.Type (composite.proto) */
enum {
    UA_COUNTER_WITH_METADATAS = 1,
    UA_BATCH_TIMER_WITH_METADATAS = 2,
    UA_GAUGE_WITH_METADATAS = 3,
    UA_FORWARDED = 4,
    UA_TIMED_WITH_METADATA = 5,
    UA_TIMED_WITH_METADATAS = 6,
    UA_TIMED_WITH_STORAGE_POLICY = 7,
};

struct UaRd {
    const uint8_t* p;
    size_t n, pos = 0;
    int err = 0;
    bool u64(uint64_t* out) { /* proto varint */
        uint64_t v = 0;
        int shift = 0;
        for (int i = 0; i < 10; i++) {
            if (pos >= n) { err = M3GPU_UA_ERR_TRUNCATED; return false; }
            uint8_t b = p[pos++];
            /* 10th byte may only contribute bit 63 (Go binary.ReadUvarint
             * overflow rule): anything above 1 overflows uint64 */
            if (i == 9 && b > 1) { err = M3GPU_UA_ERR_PROTO; return false; }
            v |= (uint64_t)(b & 0x7f) << shift;
            if (!(b & 0x80)) { *out = v; return true; }
            shift += 7;
        }
        err = M3GPU_UA_ERR_PROTO;
        return false;
    }
    bool fixed64(uint64_t* out) {
        if (pos + 8 > n) { err = M3GPU_UA_ERR_TRUNCATED; return false; }
        uint64_t v = 0;
        for (int i = 7; i >= 0; i--) v = (v << 8) | p[pos + i];
        pos += 8;
        *out = v;
        return true;
    }
    bool bytes(const uint8_t** out, uint64_t* len) {
        uint64_t l;
        if (!u64(&l)) return false;
        /* subtraction form: pos + l wraps for l near 2^64 (pos <= n always) */
        if (l > n - pos) { err = M3GPU_UA_ERR_TRUNCATED; return false; }
        *out = p + pos;
        *len = l;
        pos += l;
        return true;
    }
    bool skip(uint32_t wt) {
        uint64_t d;
        const uint8_t* b;
        switch (wt) {
        case 0: return u64(&d);
        case 1: return fixed64(&d);
        case 2: return bytes(&b, &d);
        case 5:
            if (pos + 4 > n) { err = M3GPU_UA_ERR_TRUNCATED; return false; }
            pos += 4;
            return true;
        default: err = M3GPU_UA_ERR_PROTO; return false;
        }
    }
};

/* one retained wrapper-level length-delimited field (StagedMetadatas,
 * StoragePolicy, ...), kept with its field number so downstream rule
 * matching can recover the individual reference protos */
struct UaMetaField {
    uint32_t field;
    std::vector<uint8_t> bytes;
};

struct UaMetric {
    int32_t union_type;
    int32_t metric_type;    /* metricpb.MetricType for timed metrics */
    std::vector<uint8_t> id;
    std::vector<double> values;   /* counter value widened; gauge; timer batch */
    int64_t counter_value = 0;    /* exact int64 for counters */
    int64_t time_nanos = 0;       /* client_time (untimed) or time (timed) */
    std::vector<uint8_t> annotation;
    std::vector<uint8_t> metadatas;    /* concatenation (back-compat view) */
    std::vector<UaMetaField> meta_fields; /* field-separated segments */
};

struct UaBatch {
    std::vector<UaMetric> metrics;
};

static std::mutex g_ua_mu;
static std::vector<UaBatch*> g_ua_batches;
static char g_ua_err[256];

static int ua_fail(int code, const char* what) {
    snprintf(g_ua_err, sizeof(g_ua_err), "unagg: %s", what);
    return code;
}

/* Counter/Gauge/BatchTimer/TimedMetric payload (metric.proto) */
static bool ua_parse_metric(UaRd& rd, UaMetric& m, bool timed) {
    while (rd.pos < rd.n) {
        uint64_t tag;
        if (!rd.u64(&tag)) return false;
        uint32_t f = (uint32_t)(tag >> 3), wt = (uint32_t)(tag & 7);
        uint64_t v;
        const uint8_t* b;
        uint64_t bl;
        if (timed) {
            switch (f) {
            case 1: if (!rd.u64(&v)) return false; m.metric_type = (int32_t)v; continue;
            case 2: if (!rd.bytes(&b, &bl)) return false; m.id.assign(b, b + bl); continue;
            case 3: if (!rd.u64(&v)) return false; m.time_nanos = (int64_t)v; continue;
            case 4: if (!rd.fixed64(&v)) return false; {
                double d; memcpy(&d, &v, 8); m.values.push_back(d); } continue;
            case 5: if (!rd.bytes(&b, &bl)) return false; m.annotation.assign(b, b + bl); continue;
            default: if (!rd.skip(wt)) return false; continue;
            }
        }
        switch (f) {
        case 1: if (!rd.bytes(&b, &bl)) return false; m.id.assign(b, b + bl); continue;
        case 2:
            if (wt == 0) { /* counter int64 */
                if (!rd.u64(&v)) return false;
                m.counter_value = (int64_t)v;
                m.values.push_back((double)(int64_t)v);
            } else if (wt == 1) { /* gauge double */
                if (!rd.fixed64(&v)) return false;
                double d; memcpy(&d, &v, 8);
                m.values.push_back(d);
            } else if (wt == 2) { /* batch timer packed doubles */
                if (!rd.bytes(&b, &bl)) return false;
                if (bl % 8) { rd.err = M3GPU_UA_ERR_PROTO; return false; }
                for (uint64_t i = 0; i < bl; i += 8) {
                    uint64_t u = 0;
                    for (int k = 7; k >= 0; k--) u = (u << 8) | b[i + k];
                    double d; memcpy(&d, &u, 8);
                    m.values.push_back(d);
                }
            } else { rd.err = M3GPU_UA_ERR_PROTO; return false; }
            continue;
        case 3: if (!rd.bytes(&b, &bl)) return false; m.annotation.assign(b, b + bl); continue;
        case 4: if (!rd.u64(&v)) return false; m.time_nanos = (int64_t)v; continue;
        default: if (!rd.skip(wt)) return false; continue;
        }
    }
    return true;
}

/* XxxWithMetadatas wrapper: field 1 = metric, field 2(+3) = opaque */
static bool ua_parse_union(UaRd& rd, UaMetric& m, bool timed) {
    while (rd.pos < rd.n) {
        uint64_t tag;
        if (!rd.u64(&tag)) return false;
        uint32_t f = (uint32_t)(tag >> 3), wt = (uint32_t)(tag & 7);
        const uint8_t* b;
        uint64_t bl;
        if (f == 1 && wt == 2) {
            if (!rd.bytes(&b, &bl)) return false;
            UaRd sub{b, (size_t)bl};
            if (!ua_parse_metric(sub, m, timed)) { rd.err = sub.err; return false; }
        } else if (wt == 2) { /* metadatas / storage policy: keep opaque,
                               * field-separated (plus the legacy concat) */
            if (!rd.bytes(&b, &bl)) return false;
            m.metadatas.insert(m.metadatas.end(), b, b + bl);
            UaMetaField mf;
            mf.field = f;
            mf.bytes.assign(b, b + bl);
            m.meta_fields.push_back(std::move(mf));
        } else {
            if (!rd.skip(wt)) return false;
        }
    }
    return true;
}

extern "C" {

const char* m3gpu_unagg_last_error(void) { return g_ua_err; }

/* Parse a buffer of consecutive size-prefixed MetricWithMetadatas messages
 * (the unaggregated wire stream). Returns a handle >= 0 or an error. */
int m3gpu_unagg_parse(const uint8_t* buf, uint64_t len) {
    UaBatch* batch = new UaBatch();
    size_t pos = 0;
    while (pos < len) {
        /* Go binary.ReadVarint: zigzag varint size (iterator.go:106-118) */
        uint64_t uv = 0;
        int shift = 0;
        bool ok = false;
        for (int i = 0; i < 10 && pos < len; i++) {
            uint8_t b = buf[pos++];
            uv |= (uint64_t)(b & 0x7f) << shift;
            if (!(b & 0x80)) { ok = true; break; }
            shift += 7;
        }
        if (!ok) { delete batch; return ua_fail(M3GPU_UA_ERR_TRUNCATED, "size varint"); }
        int64_t size = (int64_t)(uv >> 1);
        if (uv & 1) size = ~size;
        if (size <= 0) { delete batch; return ua_fail(M3GPU_UA_ERR_SIZE, "non-positive size"); }
        if ((uint64_t)size > len - pos) {
            delete batch;
            return ua_fail(M3GPU_UA_ERR_TRUNCATED, "message body");
        }
        UaRd rd{buf + pos, (size_t)size};
        pos += (size_t)size;
        /* MetricWithMetadatas: 1=type enum, 2..8=payload by type */
        int32_t utype = 0;
        UaMetric m;
        while (rd.pos < rd.n && !rd.err) {
            uint64_t tag;
            if (!rd.u64(&tag)) break;
            uint32_t f = (uint32_t)(tag >> 3), wt = (uint32_t)(tag & 7);
            if (f == 1 && wt == 0) {
                uint64_t v;
                if (!rd.u64(&v)) break;
                utype = (int32_t)v;
                continue;
            }
            if (wt == 2 && f >= 2 && f <= 8 && f != 5) {
                /* payload field number = union type + 1; field 5
                 * (forwarded) is skipped here - rejected by the type check */
                const uint8_t* b;
                uint64_t bl;
                if (!rd.bytes(&b, &bl)) break;
                UaRd sub{b, (size_t)bl};
                bool timed = f >= 6; /* 6,7,8 carry TimedMetric */
                if (!ua_parse_union(sub, m, timed)) { rd.err = sub.err; break; }
                continue;
            }
            if (!rd.skip(wt)) break;
        }
        if (rd.err) { delete batch; return ua_fail(rd.err, "message parse"); }
        if (utype < UA_COUNTER_WITH_METADATAS ||
            utype > UA_TIMED_WITH_STORAGE_POLICY || utype == UA_FORWARDED) {
            /* forwarded metrics are aggregator-internal pipeline traffic,
             * not ingest; reject like the iterator's type check */
            delete batch;
            return ua_fail(M3GPU_UA_ERR_TYPE, "unsupported union type");
        }
        m.union_type = utype;
        batch->metrics.push_back(std::move(m));
    }
    std::lock_guard<std::mutex> lk(g_ua_mu);
    for (size_t i = 0; i < g_ua_batches.size(); i++) {
        if (!g_ua_batches[i]) { g_ua_batches[i] = batch; return (int)i; }
    }
    g_ua_batches.push_back(batch);
    return (int)g_ua_batches.size() - 1;
}

static UaBatch* ua_get(int h) {
    std::lock_guard<std::mutex> lk(g_ua_mu);
    if (h < 0 || (size_t)h >= g_ua_batches.size()) return nullptr;
    return g_ua_batches[h];
}

int m3gpu_unagg_close(int h) {
    std::lock_guard<std::mutex> lk(g_ua_mu);
    if (h < 0 || (size_t)h >= g_ua_batches.size() || !g_ua_batches[h])
        return M3GPU_UA_ERR_BADHANDLE;
    delete g_ua_batches[h];
    g_ua_batches[h] = nullptr;
    return 0;
}

int64_t m3gpu_unagg_count(int h) {
    UaBatch* b = ua_get(h);
    if (!b) return M3GPU_UA_ERR_BADHANDLE;
    return (int64_t)b->metrics.size();
}

int m3gpu_unagg_metric(int h, int64_t i, int32_t* union_type,
                       int32_t* metric_type, const uint8_t** id,
                       int64_t* id_len, int64_t* num_values,
                       int64_t* counter_value, int64_t* time_nanos,
                       const uint8_t** annotation, int64_t* annotation_len,
                       const uint8_t** metadatas, int64_t* metadatas_len) {
    UaBatch* b = ua_get(h);
    if (!b) return M3GPU_UA_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= b->metrics.size()) return M3GPU_UA_ERR_BADHANDLE;
    const UaMetric& m = b->metrics[(size_t)i];
    if (union_type) *union_type = m.union_type;
    if (metric_type) *metric_type = m.metric_type;
    if (id) *id = m.id.data();
    if (id_len) *id_len = (int64_t)m.id.size();
    if (num_values) *num_values = (int64_t)m.values.size();
    if (counter_value) *counter_value = m.counter_value;
    if (time_nanos) *time_nanos = m.time_nanos;
    if (annotation) *annotation = m.annotation.data();
    if (annotation_len) *annotation_len = (int64_t)m.annotation.size();
    if (metadatas) *metadatas = m.metadatas.data();
    if (metadatas_len) *metadatas_len = (int64_t)m.metadatas.size();
    return 0;
}

/* Field-separated wrapper metadata access (StagedMetadatas field 2,
 * StoragePolicy field 3, RoutingPolicy, unknown retained fields): the
 * concatenated `metadatas` view cannot be split back into protos. */
int64_t m3gpu_unagg_metadata_count(int h, int64_t i) {
    UaBatch* b = ua_get(h);
    if (!b) return M3GPU_UA_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= b->metrics.size()) return M3GPU_UA_ERR_BADHANDLE;
    return (int64_t)b->metrics[(size_t)i].meta_fields.size();
}

int m3gpu_unagg_metadata(int h, int64_t i, int64_t j, int32_t* field,
                         const uint8_t** bytes, int64_t* len) {
    UaBatch* b = ua_get(h);
    if (!b) return M3GPU_UA_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= b->metrics.size()) return M3GPU_UA_ERR_BADHANDLE;
    const UaMetric& m = b->metrics[(size_t)i];
    if (j < 0 || (size_t)j >= m.meta_fields.size()) return M3GPU_UA_ERR_BADHANDLE;
    const UaMetaField& mf = m.meta_fields[(size_t)j];
    if (field) *field = (int32_t)mf.field;
    if (bytes) *bytes = mf.bytes.data();
    if (len) *len = (int64_t)mf.bytes.size();
    return 0;
}

int m3gpu_unagg_values(int h, int64_t i, double* out) {
    UaBatch* b = ua_get(h);
    if (!b) return M3GPU_UA_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= b->metrics.size()) return M3GPU_UA_ERR_BADHANDLE;
    const UaMetric& m = b->metrics[(size_t)i];
    memcpy(out, m.values.data(), m.values.size() * sizeof(double));
    return 0;
}

} /* extern "C" */
