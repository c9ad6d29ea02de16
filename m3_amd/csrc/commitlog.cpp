/* commitlog.cpp — native reader for the reference's commit log files.
 *
 * Restates src/dbnode/persist/fs/commitlog read path for bulk bootstrap:
 *  - chunked container (writer.go:43-58,340-392 / chunk_reader.go:62-120):
 *    each chunk = 12-byte header {size u32 LE, adler32(size bytes) u32 LE,
 *    adler32(payload) u32 LE} + payload; the payload byte stream is
 *    CONTINUOUS across chunks (records may span chunk boundaries,
 *    chunk_reader.go Read);
 *  - records = Go binary.Uvarint length + msgpack bytes (writer.go:282-301,
 *    reader.go:195-209);
 *  - first record is LogInfo (decoder.go:476-493: two deprecated varints +
 *    file index); every other record is a LogEntry
 *    (decoder_fast.go:64-113: index uint, create, metadata bytes,
 *    timestamp, value f64, unit uint, annotation bytes);
 *  - a LogEntry's metadata field carries a nested LogMetadata msgpack blob
 *    (id, namespace, shard, encodedTags — note the reference's header
 *    declares 3 fields but writes/reads 4, schema.go:109 vs
 *    encoder.go:372-378; replicated) the first time a series unique index
 *    appears (writer.go:205-221); an entry whose index was never
 *    registered is an error (reader.go errCommitLogReaderMissingMetadata).
 *
 * Output: series in first-seen order with their datapoints in log order,
 * ready for the GPU batch encoder (the commitlog bootstrapper's
 * read-and-re-encode job).
 */
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>
#include <unordered_map>
#include <mutex>

enum {
    M3GPU_CL_ERR_IO = -101,          /* shares fileset codes where same */
    M3GPU_CL_ERR_MSGPACK = -104,
    M3GPU_CL_ERR_SCHEMA = -105,
    M3GPU_CL_ERR_BADHANDLE = -109,
    M3GPU_CL_ERR_CHUNK_CHECKSUM = -111,
    M3GPU_CL_ERR_MISSING_METADATA = -112,
    M3GPU_CL_ERR_TRUNCATED = -113,
};

/* shared with fileset.cpp (same TU-local copies; kept static) */
static uint32_t cl_adler32(const uint8_t* p, size_t n) {
    const uint32_t MOD = 65521;
    uint32_t a = 1, b = 0;
    while (n > 0) {
        size_t chunk = n > 5552 ? 5552 : n;
        n -= chunk;
        for (size_t i = 0; i < chunk; i++) {
            a += p[i];
            b += a;
        }
        p += chunk;
        a %= MOD;
        b %= MOD;
    }
    return (b << 16) | a;
}

static uint32_t cl_le32(const uint8_t* p) {
    return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
           ((uint32_t)p[3] << 24);
}

struct ClRd { /* permissive msgpack, mirroring decoder_fast.go */
    const uint8_t* p;
    size_t n, pos = 0;
    int err = 0;
    bool need(size_t k) {
        if (pos + k > n) { err = M3GPU_CL_ERR_MSGPACK; return false; }
        return true;
    }
    uint8_t byte() { return need(1) ? p[pos++] : 0; }
    uint64_t be(int k) {
        if (!need((size_t)k)) return 0;
        uint64_t v = 0;
        for (int i = 0; i < k; i++) v = (v << 8) | p[pos + i];
        pos += (size_t)k;
        return v;
    }
    int64_t read_int() {
        uint8_t c = byte();
        if (err) return 0;
        if (c == 0xc0) return 0;
        if (c <= 0x7f) return (int64_t)c;
        if (c >= 0xe0) return (int64_t)(int8_t)c;
        switch (c) {
        case 0xcc: return (int64_t)be(1);
        case 0xd0: return (int64_t)(int8_t)be(1);
        case 0xcd: return (int64_t)be(2);
        case 0xd1: return (int64_t)(int16_t)be(2);
        case 0xce: return (int64_t)be(4);
        case 0xd2: return (int64_t)(int32_t)be(4);
        case 0xcf: case 0xd3: return (int64_t)be(8);
        default: err = M3GPU_CL_ERR_MSGPACK; return 0;
        }
    }
    double read_f64() { /* decodeFloat64 (decoder_fast.go:330-363) */
        uint8_t c = byte();
        if (err) return 0;
        if (c == 0xca) { /* float32 */
            uint32_t i = (uint32_t)be(4);
            float f;
            memcpy(&f, &i, 4);
            return (double)f;
        }
        if (c == 0xcb) {
            uint64_t i = be(8);
            double d;
            memcpy(&d, &i, 8);
            return d;
        }
        err = M3GPU_CL_ERR_MSGPACK;
        return 0;
    }
    int read_array_len() {
        uint8_t c = byte();
        if (err) return 0;
        if (c >= 0x90 && c <= 0x9f) return (int)(c & 0x0f);
        if (c == 0xdc) return (int)be(2);
        if (c == 0xdd) return (int)be(4);
        err = M3GPU_CL_ERR_MSGPACK;
        return 0;
    }
    int64_t read_bytes(const uint8_t** out) {
        uint8_t c = byte();
        if (err) return 0;
        int64_t len;
        if (c == 0xc0) { *out = nullptr; return -1; }
        else if (c >= 0xa0 && c <= 0xbf) len = (int64_t)(c & 0x1f);
        else if (c == 0xd9 || c == 0xc4) len = (int64_t)be(1);
        else if (c == 0xda || c == 0xc5) len = (int64_t)be(2);
        else if (c == 0xdb || c == 0xc6) len = (int64_t)be(4);
        else { err = M3GPU_CL_ERR_MSGPACK; return 0; }
        if (!need((size_t)len)) return 0;
        *out = p + pos;
        pos += (size_t)len;
        return len;
    }
    void skip_value() {
        if (err || !need(1)) return;
        uint8_t c = p[pos];
        if (c <= 0x7f || c >= 0xe0 || c == 0xc0) { pos++; return; }
        const uint8_t* dummy;
        switch (c) {
        case 0xcc: case 0xd0: byte(); be(1); return;
        case 0xcd: case 0xd1: byte(); be(2); return;
        case 0xce: case 0xd2: byte(); be(4); return;
        case 0xcf: case 0xd3: case 0xcb: byte(); be(8); return;
        case 0xca: byte(); be(4); return;
        default: read_bytes(&dummy); return;
        }
    }
    /* version int, root arraylen>=2, objtype int -> extra root fields */
    bool read_root(int expect_type, int* extra) {
        int64_t version = read_int();
        if (err) return false;
        if (version < 1) { err = M3GPU_CL_ERR_SCHEMA; return false; }
        int nf = read_array_len();
        if (err) return false;
        if (nf < 2) { err = M3GPU_CL_ERR_SCHEMA; return false; }
        int64_t t = read_int();
        if (err) return false;
        if (t != expect_type) { err = M3GPU_CL_ERR_SCHEMA; return false; }
        *extra = nf - 2;
        return true;
    }
};

enum { CL_LOG_INFO = 7, CL_LOG_ENTRY = 8, CL_LOG_METADATA = 9 };

struct ClPoint {
    int64_t ts;
    double val;
    uint8_t unit;
};

struct ClSeries {
    uint64_t unique_index;
    std::vector<uint8_t> id, ns, tags;
    uint32_t shard = 0;
    std::vector<ClPoint> points;
    std::vector<std::pair<int64_t, std::vector<uint8_t>>> annotations;
};

struct ClFile {
    int64_t index = 0;
    int64_t num_entries = 0;
    std::vector<ClSeries> series;        /* first-seen order */
    std::unordered_map<uint64_t, size_t> by_index;
};

static std::mutex g_cl_mu;
static std::vector<ClFile*> g_cl_files;
static char g_cl_err[512];

static int cl_fail(int code, const char* what) {
    snprintf(g_cl_err, sizeof(g_cl_err), "commitlog: %s", what);
    return code;
}

/* Go binary.ReadUvarint over the continuous stream. */
static bool cl_uvarint(const std::vector<uint8_t>& s, size_t* pos, uint64_t* out) {
    uint64_t v = 0;
    int shift = 0;
    for (int i = 0; i < 10; i++) {
        if (*pos >= s.size()) return false;
        uint8_t b = s[(*pos)++];
        /* Go binary.ReadUvarint overflow rule: the 10th byte may only
         * contribute bit 63, i.e. must be <= 1 */
        if (i == 9 && b > 1) return false;
        if (b < 0x80) {
            *out = v | ((uint64_t)b << shift);
            return true;
        }
        v |= ((uint64_t)(b & 0x7f)) << shift;
        shift += 7;
    }
    return false;
}

extern "C" {

const char* m3gpu_commitlog_last_error(void) { return g_cl_err; }

int m3gpu_commitlog_open(const char* path) {
    FILE* f = fopen(path, "rb");
    if (!f) return cl_fail(M3GPU_CL_ERR_IO, "open failed");
    std::vector<uint8_t> raw;
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    raw.resize(sz > 0 ? (size_t)sz : 0);
    if (sz > 0 && fread(raw.data(), 1, (size_t)sz, f) != (size_t)sz) {
        fclose(f);
        return cl_fail(M3GPU_CL_ERR_IO, "read failed");
    }
    fclose(f);

    /* 1. validate chunks, build the continuous payload stream */
    std::vector<uint8_t> stream;
    stream.reserve(raw.size());
    size_t pos = 0;
    while (pos < raw.size()) {
        if (pos + 12 > raw.size())
            return cl_fail(M3GPU_CL_ERR_TRUNCATED, "truncated chunk header");
        uint32_t size = cl_le32(&raw[pos]);
        uint32_t ck_size = cl_le32(&raw[pos + 4]);
        uint32_t ck_data = cl_le32(&raw[pos + 8]);
        if (cl_adler32(&raw[pos], 4) != ck_size)
            return cl_fail(M3GPU_CL_ERR_CHUNK_CHECKSUM, "chunk size checksum");
        if (pos + 12 + size > raw.size())
            return cl_fail(M3GPU_CL_ERR_TRUNCATED, "truncated chunk payload");
        if (cl_adler32(&raw[pos + 12], size) != ck_data)
            return cl_fail(M3GPU_CL_ERR_CHUNK_CHECKSUM, "chunk data checksum");
        stream.insert(stream.end(), raw.begin() + pos + 12,
                      raw.begin() + pos + 12 + size);
        pos += 12 + (size_t)size;
    }

    /* 2. parse records */
    ClFile* file = new ClFile();
    size_t sp = 0;
    bool first = true;
    while (sp < stream.size()) {
        uint64_t rec_len = 0;
        size_t mark = sp;
        if (!cl_uvarint(stream, &sp, &rec_len)) {
            if (mark == stream.size()) break; /* clean EOF */
            delete file;
            return cl_fail(M3GPU_CL_ERR_TRUNCATED, "truncated record size");
        }
        /* subtraction form: sp + rec_len wraps for rec_len near 2^64 */
        if (rec_len > stream.size() - sp) {
            delete file;
            return cl_fail(M3GPU_CL_ERR_TRUNCATED, "truncated record");
        }
        ClRd rd{stream.data() + sp, (size_t)rec_len};
        sp += rec_len;
        int extra = 0;
        if (first) { /* LogInfo (decoder.go:476-493) */
            first = false;
            if (!rd.read_root(CL_LOG_INFO, &extra)) {
                delete file;
                return cl_fail(rd.err, "log info root");
            }
            int nf = rd.read_array_len();
            if (rd.err || nf < 3) {
                delete file;
                return cl_fail(M3GPU_CL_ERR_SCHEMA, "log info fields");
            }
            rd.read_int(); /* deprecated start */
            rd.read_int(); /* deprecated duration */
            file->index = rd.read_int();
            if (rd.err) { delete file; return cl_fail(rd.err, "log info"); }
            continue;
        }
        /* LogEntry (decoder_fast.go:64-113) */
        if (!rd.read_root(CL_LOG_ENTRY, &extra)) {
            delete file;
            return cl_fail(rd.err, "log entry root");
        }
        int nf = rd.read_array_len();
        if (rd.err || nf < 7) {
            delete file;
            return cl_fail(M3GPU_CL_ERR_SCHEMA, "log entry fields");
        }
        uint64_t uidx = (uint64_t)rd.read_int();
        rd.read_int(); /* create time (ignored by bootstrap) */
        const uint8_t* meta = nullptr;
        int64_t meta_len = rd.read_bytes(&meta);
        int64_t ts = rd.read_int();
        double val = rd.read_f64();
        uint64_t unit = (uint64_t)rd.read_int();
        const uint8_t* ant = nullptr;
        int64_t ant_len = rd.read_bytes(&ant);
        if (rd.err) { delete file; return cl_fail(rd.err, "log entry"); }

        auto it = file->by_index.find(uidx);
        if (it == file->by_index.end()) {
            if (meta_len <= 0) { /* reader.go errCommitLogReaderMissingMetadata */
                delete file;
                return cl_fail(M3GPU_CL_ERR_MISSING_METADATA, "entry without metadata");
            }
            /* nested LogMetadata (decoder_fast.go:115-151; the declared
             * field count is 3 but 4 fields follow — replicated) */
            ClRd md{meta, (size_t)meta_len};
            int mextra = 0;
            if (!md.read_root(CL_LOG_METADATA, &mextra)) {
                delete file;
                return cl_fail(md.err, "log metadata root");
            }
            int mf = md.read_array_len();
            if (md.err || mf < 3) {
                delete file;
                return cl_fail(M3GPU_CL_ERR_SCHEMA, "log metadata fields");
            }
            ClSeries s;
            s.unique_index = uidx;
            const uint8_t* b = nullptr;
            int64_t bl = md.read_bytes(&b);
            if (bl > 0) s.id.assign(b, b + bl);
            bl = md.read_bytes(&b);
            if (bl > 0) s.ns.assign(b, b + bl);
            s.shard = (uint32_t)md.read_int();
            bl = md.read_bytes(&b);
            if (bl > 0) s.tags.assign(b, b + bl);
            if (md.err) { delete file; return cl_fail(md.err, "log metadata"); }
            file->by_index.emplace(uidx, file->series.size());
            it = file->by_index.find(uidx);
            file->series.push_back(std::move(s));
        }
        ClSeries& s = file->series[it->second];
        if (ant_len > 0)
            s.annotations.emplace_back((int64_t)s.points.size(),
                                       std::vector<uint8_t>(ant, ant + ant_len));
        s.points.push_back(ClPoint{ts, val, (uint8_t)unit});
        file->num_entries++;
    }

    std::lock_guard<std::mutex> lk(g_cl_mu);
    for (size_t i = 0; i < g_cl_files.size(); i++) {
        if (!g_cl_files[i]) { g_cl_files[i] = file; return (int)i; }
    }
    g_cl_files.push_back(file);
    return (int)g_cl_files.size() - 1;
}

static ClFile* cl_get(int h) {
    std::lock_guard<std::mutex> lk(g_cl_mu);
    if (h < 0 || (size_t)h >= g_cl_files.size()) return nullptr;
    return g_cl_files[h];
}

int m3gpu_commitlog_close(int h) {
    std::lock_guard<std::mutex> lk(g_cl_mu);
    if (h < 0 || (size_t)h >= g_cl_files.size() || !g_cl_files[h])
        return M3GPU_CL_ERR_BADHANDLE;
    delete g_cl_files[h];
    g_cl_files[h] = nullptr;
    return 0;
}

int m3gpu_commitlog_info(int h, int64_t* index, int64_t* num_entries,
                         int64_t* num_series) {
    ClFile* f = cl_get(h);
    if (!f) return M3GPU_CL_ERR_BADHANDLE;
    if (index) *index = f->index;
    if (num_entries) *num_entries = f->num_entries;
    if (num_series) *num_series = (int64_t)f->series.size();
    return 0;
}

int m3gpu_commitlog_series(int h, int64_t i, uint64_t* unique_index,
                           const uint8_t** id, int64_t* id_len,
                           const uint8_t** ns, int64_t* ns_len,
                           uint32_t* shard, const uint8_t** tags,
                           int64_t* tags_len, int64_t* num_points,
                           int64_t* num_annotations) {
    ClFile* f = cl_get(h);
    if (!f) return M3GPU_CL_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= f->series.size()) return M3GPU_CL_ERR_BADHANDLE;
    const ClSeries& s = f->series[(size_t)i];
    if (unique_index) *unique_index = s.unique_index;
    if (id) *id = s.id.data();
    if (id_len) *id_len = (int64_t)s.id.size();
    if (ns) *ns = s.ns.data();
    if (ns_len) *ns_len = (int64_t)s.ns.size();
    if (shard) *shard = s.shard;
    if (tags) *tags = s.tags.data();
    if (tags_len) *tags_len = (int64_t)s.tags.size();
    if (num_points) *num_points = (int64_t)s.points.size();
    if (num_annotations) *num_annotations = (int64_t)s.annotations.size();
    return 0;
}

/* Copy series i's datapoints (log order) into caller arrays. */
int m3gpu_commitlog_series_points(int h, int64_t i, int64_t* ts, double* vals,
                                  uint8_t* units) {
    ClFile* f = cl_get(h);
    if (!f) return M3GPU_CL_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= f->series.size()) return M3GPU_CL_ERR_BADHANDLE;
    const ClSeries& s = f->series[(size_t)i];
    for (size_t k = 0; k < s.points.size(); k++) {
        if (ts) ts[k] = s.points[k].ts;
        if (vals) vals[k] = s.points[k].val;
        if (units) units[k] = s.points[k].unit;
    }
    return 0;
}

/* Annotation j of series i: point index it attaches to + bytes. */
int m3gpu_commitlog_series_annotation(int h, int64_t i, int64_t j,
                                      int64_t* point_index,
                                      const uint8_t** bytes, int64_t* len) {
    ClFile* f = cl_get(h);
    if (!f) return M3GPU_CL_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= f->series.size()) return M3GPU_CL_ERR_BADHANDLE;
    const ClSeries& s = f->series[(size_t)i];
    if (j < 0 || (size_t)j >= s.annotations.size())
        return M3GPU_CL_ERR_BADHANDLE;
    if (point_index) *point_index = s.annotations[(size_t)j].first;
    if (bytes) *bytes = s.annotations[(size_t)j].second.data();
    if (len) *len = (int64_t)s.annotations[(size_t)j].second.size();
    return 0;
}

} /* extern "C" */
