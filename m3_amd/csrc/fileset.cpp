/* fileset.cpp — native reader for the reference's dbnode fileset volumes.
 *
 * Restates the read side of src/dbnode/persist/fs (read.go:145-457,
 * msgpack/decoder.go:251-446, msgpack/decoder_fast.go:176-410,
 * digest/{digest.go,buffer.go,writer.go}, fs.go:27-51 naming,
 * files.go:1729-1745 path construction) for bulk ingestion on the MI355X
 * host: open a volume, validate every digest, parse the info and index
 * files, and repack the data blocks into the 64-byte-aligned batch layout
 * the decode kernels consume (include/m3gpu.h; 8B required, 64B = one
 * HBM line per stream chunk for the decode ring's aligned refills).
 *
 * Scope notes:
 *  - flush-type volumes with both current (fileset-<t>-<v>-<suffix>.db) and
 *    legacy (fileset-<t>-<suffix>.db) naming;
 *  - msgpack primitives accept every int/uint family exactly like the
 *    reference's permissive decoder (decoder_fast.go:176-345);
 *  - IndexInfo V5 with graceful decode of older field counts
 *    (decoder.go:251-330); IndexEntry V1/V2/V3 with the V3 trailing
 *    entry-checksum (adler32 of the entry's preceding bytes,
 *    decoder.go:430-446);
 *  - summaries and bloom filter files are digest-validated but not parsed
 *    (the reference reader skips summaries too, read.go:321-323; bloom is a
 *    seek-path accelerator, unused by bulk ingestion);
 *  - entries are sorted by data offset ascending (read.go:352-370) and
 *    every entry's data checksum is verified (read.go:393-397).
 */
#include <cerrno>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>
#include <algorithm>
#include <mutex>

/* ---------------------------------------------------------------- errors */
enum {
    M3GPU_FS_OK = 0,
    M3GPU_FS_ERR_IO = -101,            /* open/read failure */
    M3GPU_FS_ERR_CHECKPOINT = -102,    /* bad checkpoint file */
    M3GPU_FS_ERR_DIGEST = -103,        /* a file digest mismatch */
    M3GPU_FS_ERR_MSGPACK = -104,       /* malformed msgpack */
    M3GPU_FS_ERR_SCHEMA = -105,        /* wrong object type/version/fields */
    M3GPU_FS_ERR_ENTRY_CHECKSUM = -106,/* index entry checksum mismatch */
    M3GPU_FS_ERR_DATA_CHECKSUM = -107, /* data block checksum mismatch */
    M3GPU_FS_ERR_BOUNDS = -108,        /* entry exceeds data file */
    M3GPU_FS_ERR_BADHANDLE = -109,
    M3GPU_FS_ERR_CAPACITY = -110,      /* caller buffer too small */
};

/* -------------------------------------------------------------- adler32 */
/* hash/adler32 (digest.go:36-38): MOD 65521, standard zlib definition. */
static uint32_t fs_adler32(const uint8_t* p, size_t n) {
    const uint32_t MOD = 65521;
    uint32_t a = 1, b = 0;
    while (n > 0) {
        size_t chunk = n > 5552 ? 5552 : n; /* max before 32-bit overflow */
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

/* ------------------------------------------------------ msgpack reading */
/* Permissive primitive decoders mirroring decoder_fast.go:176-410: any
 * int/uint family is accepted for an integer field; bytes accept
 * fixstr/str8-32/bin8-32 and nil (-1 length). */
struct MsgRd {
    const uint8_t* p;
    size_t n;
    size_t pos = 0;
    int err = 0;

    bool need(size_t k) {
        if (pos + k > n) { err = M3GPU_FS_ERR_MSGPACK; return false; }
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
    int64_t read_int() { /* decodeInt: all families */
        uint8_t c = byte();
        if (err) return 0;
        if (c == 0xc0) return 0;                       /* nil -> 0 */
        if (c <= 0x7f) return (int64_t)c;              /* pos fixint */
        if (c >= 0xe0) return (int64_t)(int8_t)c;      /* neg fixint */
        switch (c) {
        case 0xcc: return (int64_t)be(1);              /* uint8 */
        case 0xd0: return (int64_t)(int8_t)be(1);      /* int8 */
        case 0xcd: return (int64_t)be(2);              /* uint16 */
        case 0xd1: return (int64_t)(int16_t)be(2);     /* int16 */
        case 0xce: return (int64_t)be(4);              /* uint32 */
        case 0xd2: return (int64_t)(int32_t)be(4);     /* int32 */
        case 0xcf: case 0xd3: return (int64_t)be(8);   /* uint64/int64 */
        default: err = M3GPU_FS_ERR_MSGPACK; return 0;
        }
    }
    int read_array_len() {
        uint8_t c = byte();
        if (err) return 0;
        if (c >= 0x90 && c <= 0x9f) return (int)(c & 0x0f);
        if (c == 0xdc) return (int)be(2);
        if (c == 0xdd) return (int)be(4);
        err = M3GPU_FS_ERR_MSGPACK;
        return 0;
    }
    /* returns length, -1 for nil; *out points into the buffer */
    int64_t read_bytes(const uint8_t** out) {
        uint8_t c = byte();
        if (err) return 0;
        int64_t len;
        if (c == 0xc0) { *out = nullptr; return -1; }
        else if (c >= 0xa0 && c <= 0xbf) len = (int64_t)(c & 0x1f);
        else if (c == 0xd9 || c == 0xc4) len = (int64_t)be(1);
        else if (c == 0xda || c == 0xc5) len = (int64_t)be(2);
        else if (c == 0xdb || c == 0xc6) len = (int64_t)be(4);
        else { err = M3GPU_FS_ERR_MSGPACK; return 0; }
        if (!need((size_t)len)) return 0;
        *out = p + pos;
        pos += (size_t)len;
        return len;
    }
    void skip_value() { /* skip one object (ints/bytes only occur here) */
        if (err) return;
        uint8_t c = p[pos];
        if (c <= 0x7f || c >= 0xe0 || c == 0xc0) { pos++; return; }
        const uint8_t* dummy;
        switch (c) {
        case 0xcc: case 0xd0: byte(); be(1); return;
        case 0xcd: case 0xd1: byte(); be(2); return;
        case 0xce: case 0xd2: byte(); be(4); return;
        case 0xcf: case 0xd3: byte(); be(8); return;
        case 0xcb: byte(); be(8); return;
        default: read_bytes(&dummy); return;
        }
    }
};

/* object types (msgpack/schema.go:63-76) */
enum {
    FS_ROOT_OBJECT = 1,
    FS_INDEX_INFO = 2,
    FS_INDEX_SUMMARIES_INFO = 3,
    FS_INDEX_BLOOM_FILTER_INFO = 4,
    FS_INDEX_ENTRY = 5,
};

/* decodeRootObject (decoder.go): version int, arraylen(>=2), objtype int.
 * Returns fields-to-skip count from the root array (curr fields - 2 read
 * here is not how it works: root has 2 fields [objectType, object]; any
 * extra root fields are skipped). */
static bool fs_read_root(MsgRd& rd, int expect_type, int* extra_root_fields) {
    int64_t version = rd.read_int();
    if (rd.err) return false;
    if (version < 1) { rd.err = M3GPU_FS_ERR_SCHEMA; return false; }
    int nfields = rd.read_array_len();
    if (rd.err) return false;
    if (nfields < 2) { rd.err = M3GPU_FS_ERR_SCHEMA; return false; }
    int64_t objtype = rd.read_int();
    if (rd.err) return false;
    if (objtype != expect_type) { rd.err = M3GPU_FS_ERR_SCHEMA; return false; }
    *extra_root_fields = nfields - 2;
    return true;
}

struct FsInfo {
    int64_t block_start = 0;
    int64_t block_size = 0;
    int64_t entries = 0;
    int64_t major_version = 0;
    int64_t minor_version = 0;
    int64_t summaries = 0;
    int64_t bloom_m = 0, bloom_k = 0;
    int64_t snapshot_time = 0;
    int64_t file_type = 0;
    int volume_index = 0;
};

/* decodeIndexInfo (decoder.go:251-330), V5 = 11 fields, min 6. */
static int fs_parse_info(const uint8_t* buf, size_t len, FsInfo* out) {
    MsgRd rd{buf, len};
    int extra = 0;
    if (!fs_read_root(rd, FS_INDEX_INFO, &extra)) return rd.err;
    int actual = rd.read_array_len();
    if (rd.err) return rd.err;
    if (actual < 6) return M3GPU_FS_ERR_SCHEMA;
    out->block_start = rd.read_int();
    out->block_size = rd.read_int();
    out->entries = rd.read_int();
    out->major_version = rd.read_int();
    {   /* summaries info: nested object, 1+ fields */
        int n = rd.read_array_len();
        if (rd.err) return rd.err;
        if (n < 1) return M3GPU_FS_ERR_SCHEMA;
        out->summaries = rd.read_int();
        for (int i = 1; i < n; i++) rd.skip_value();
    }
    {   /* bloom filter info: 2+ fields */
        int n = rd.read_array_len();
        if (rd.err) return rd.err;
        if (n < 2) return M3GPU_FS_ERR_SCHEMA;
        out->bloom_m = rd.read_int();
        out->bloom_k = rd.read_int();
        for (int i = 2; i < n; i++) rd.skip_value();
    }
    int consumed = 6;
    if (actual >= 8) { /* V2 fields */
        out->snapshot_time = rd.read_int();
        out->file_type = rd.read_int();
        consumed = 8;
    }
    if (actual >= 9) { /* V3: snapshot id */
        const uint8_t* dummy;
        rd.read_bytes(&dummy);
        consumed = 9;
    }
    if (actual >= 10) { out->volume_index = (int)rd.read_int(); consumed = 10; }
    if (actual >= 11) { out->minor_version = rd.read_int(); consumed = 11; }
    for (int i = consumed; i < actual; i++) rd.skip_value();
    for (int i = 0; i < extra; i++) rd.skip_value();
    return rd.err;
}

struct FsEntry {
    int64_t index;
    int64_t size;
    int64_t offset;
    int64_t data_checksum;
    std::vector<uint8_t> id;
    std::vector<uint8_t> tags;
};

/* decodeIndexEntry (decoder.go:386-446): V3 appends an adler32 of the
 * entry's own bytes (from entry start through the skip of extra fields,
 * excluding the checksum field itself) as a final varint. */
static int fs_parse_entries(const uint8_t* buf, size_t len, int64_t count,
                            std::vector<FsEntry>* out) {
    MsgRd rd{buf, len};
    /* `count` comes from the (digest-valid but possibly corrupt) info file:
     * bound it against the index file size before reserving — each entry
     * occupies at least ~8 bytes, so count > len is always a schema error
     * and an unbounded reserve() would throw past the C ABI. */
    if (count < 0 || (uint64_t)count > (uint64_t)len) return M3GPU_FS_ERR_SCHEMA;
    out->reserve((size_t)count);
    for (int64_t e = 0; e < count; e++) {
        size_t entry_start = rd.pos;
        int extra = 0;
        if (!fs_read_root(rd, FS_INDEX_ENTRY, &extra)) return rd.err;
        int actual = rd.read_array_len();
        if (rd.err) return rd.err;
        if (actual < 5) return M3GPU_FS_ERR_SCHEMA;
        FsEntry ent;
        ent.index = rd.read_int();
        const uint8_t* idp = nullptr;
        int64_t idlen = rd.read_bytes(&idp);
        if (rd.err) return rd.err;
        if (idlen > 0) ent.id.assign(idp, idp + idlen);
        ent.size = rd.read_int();
        ent.offset = rd.read_int();
        ent.data_checksum = rd.read_int();
        int consumed = 5;
        if (actual >= 6) {
            const uint8_t* tp = nullptr;
            int64_t tlen = rd.read_bytes(&tp);
            if (rd.err) return rd.err;
            if (tlen > 0) ent.tags.assign(tp, tp + tlen);
            consumed = 6;
        }
        if (actual >= 7) {
            /* skip any fields beyond tags except the trailing checksum
             * (schema stipulates the checksum is FINAL, decoder.go:427-429) */
            for (int i = consumed; i < actual - 1; i++) rd.skip_value();
            uint32_t got = fs_adler32(buf + entry_start, rd.pos - entry_start);
            int64_t want = rd.read_int();
            if (rd.err) return rd.err;
            if (want != (int64_t)got) return M3GPU_FS_ERR_ENTRY_CHECKSUM;
        } else {
            for (int i = consumed; i < actual; i++) rd.skip_value();
        }
        for (int i = 0; i < extra; i++) rd.skip_value();
        if (rd.err) return rd.err;
        out->push_back(std::move(ent));
    }
    return 0;
}

/* ----------------------------------------------------------- file utils */
static int fs_read_file(const std::string& path, std::vector<uint8_t>* out) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) return M3GPU_FS_ERR_IO;
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    if (sz < 0) { fclose(f); return M3GPU_FS_ERR_IO; }
    out->resize((size_t)sz);
    if (sz > 0 && fread(out->data(), 1, (size_t)sz, f) != (size_t)sz) {
        fclose(f);
        return M3GPU_FS_ERR_IO;
    }
    fclose(f);
    return 0;
}

static uint32_t fs_le32(const uint8_t* p) {
    return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
           ((uint32_t)p[3] << 24);
}

/* fileset-<t>-<v>-<suffix>.db, or legacy fileset-<t>-<suffix>.db when the
 * volume-0 checkpoint only exists under the legacy name
 * (files.go:1729-1736, read.go:178-197). */
static std::string fs_path(const std::string& dir, int64_t t, int vol,
                           const char* suffix, bool legacy) {
    char buf[512];
    if (legacy)
        snprintf(buf, sizeof(buf), "%s/fileset-%lld-%s.db", dir.c_str(),
                 (long long)t, suffix);
    else
        snprintf(buf, sizeof(buf), "%s/fileset-%lld-%d-%s.db", dir.c_str(),
                 (long long)t, vol, suffix);
    return std::string(buf);
}

static bool fs_exists(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (f) fclose(f);
    return f != nullptr;
}

/* ------------------------------------------------------------- volumes */
struct FsVolume {
    FsInfo info;
    std::vector<FsEntry> entries; /* sorted by offset asc */
    std::vector<uint8_t> data;
};

static std::mutex g_fs_mu;
static std::vector<FsVolume*> g_fs_volumes;

static char g_fs_err[512];
static int fs_fail(int code, const char* what, const std::string& path) {
    snprintf(g_fs_err, sizeof(g_fs_err), "%s: %s", what, path.c_str());
    return code;
}

extern "C" {

const char* m3gpu_fileset_last_error(void) { return g_fs_err; }

/* Open + fully validate a fileset volume. Returns handle >= 0 or error. */
int m3gpu_fileset_open(const char* shard_dir, int64_t block_start_ns,
                       int volume_index) {
    std::string dir(shard_dir);
    bool legacy = false;
    if (volume_index == 0 &&
        !fs_exists(fs_path(dir, block_start_ns, 0, "checkpoint", false)) &&
        fs_exists(fs_path(dir, block_start_ns, 0, "checkpoint", true)))
        legacy = true; /* isFirstVolumeLegacy (read.go:178-184) */

    auto path = [&](const char* sfx) {
        return fs_path(dir, block_start_ns, volume_index, sfx, legacy);
    };

    /* 1. checkpoint -> expected digest of the digest file (read.go:203-208) */
    std::vector<uint8_t> buf;
    int rc = fs_read_file(path("checkpoint"), &buf);
    if (rc) return fs_fail(rc, "open checkpoint", path("checkpoint"));
    if (buf.size() != 4)
        return fs_fail(M3GPU_FS_ERR_CHECKPOINT, "checkpoint size", path("checkpoint"));
    uint32_t expect_digest_of_digest = fs_le32(buf.data());

    /* 2. digest file: five LE u32 digests (write.go:381-390), validated
     * against the checkpoint (read.go:310-329) */
    std::vector<uint8_t> dig;
    rc = fs_read_file(path("digest"), &dig);
    if (rc) return fs_fail(rc, "open digest", path("digest"));
    if (dig.size() != 20)
        return fs_fail(M3GPU_FS_ERR_DIGEST, "digest size", path("digest"));
    if (fs_adler32(dig.data(), dig.size()) != expect_digest_of_digest)
        return fs_fail(M3GPU_FS_ERR_DIGEST, "digest-of-digest", path("digest"));
    uint32_t d_info = fs_le32(dig.data() + 0);
    uint32_t d_index = fs_le32(dig.data() + 4);
    uint32_t d_summ = fs_le32(dig.data() + 8);
    uint32_t d_bloom = fs_le32(dig.data() + 12);
    uint32_t d_data = fs_le32(dig.data() + 16);

    /* 3. info file (read.go:331-350) */
    std::vector<uint8_t> info_buf;
    rc = fs_read_file(path("info"), &info_buf);
    if (rc) return fs_fail(rc, "open info", path("info"));
    if (fs_adler32(info_buf.data(), info_buf.size()) != d_info)
        return fs_fail(M3GPU_FS_ERR_DIGEST, "info digest", path("info"));
    FsVolume* vol = new FsVolume();
    rc = fs_parse_info(info_buf.data(), info_buf.size(), &vol->info);
    if (rc) { delete vol; return fs_fail(rc, "parse info", path("info")); }

    /* 4. summaries + bloom: digest-validate only (read.go:321-323) */
    rc = fs_read_file(path("summaries"), &buf);
    if (rc) { delete vol; return fs_fail(rc, "open summaries", path("summaries")); }
    if (fs_adler32(buf.data(), buf.size()) != d_summ) {
        delete vol;
        return fs_fail(M3GPU_FS_ERR_DIGEST, "summaries digest", path("summaries"));
    }
    rc = fs_read_file(path("bloomfilter"), &buf);
    if (rc) { delete vol; return fs_fail(rc, "open bloomfilter", path("bloomfilter")); }
    if (fs_adler32(buf.data(), buf.size()) != d_bloom) {
        delete vol;
        return fs_fail(M3GPU_FS_ERR_DIGEST, "bloom digest", path("bloomfilter"));
    }

    /* 5. index file (read.go:352-370) */
    std::vector<uint8_t> idx_buf;
    rc = fs_read_file(path("index"), &idx_buf);
    if (rc) { delete vol; return fs_fail(rc, "open index", path("index")); }
    if (fs_adler32(idx_buf.data(), idx_buf.size()) != d_index) {
        delete vol;
        return fs_fail(M3GPU_FS_ERR_DIGEST, "index digest", path("index"));
    }
    rc = fs_parse_entries(idx_buf.data(), idx_buf.size(), vol->info.entries,
                          &vol->entries);
    if (rc) { delete vol; return fs_fail(rc, "parse index", path("index")); }
    std::sort(vol->entries.begin(), vol->entries.end(),
              [](const FsEntry& a, const FsEntry& b) { return a.offset < b.offset; });

    /* 6. data file + per-entry checksums (read.go:386-397) */
    rc = fs_read_file(path("data"), &vol->data);
    if (rc) { delete vol; return fs_fail(rc, "open data", path("data")); }
    if (fs_adler32(vol->data.data(), vol->data.size()) != d_data) {
        delete vol;
        return fs_fail(M3GPU_FS_ERR_DIGEST, "data digest", path("data"));
    }
    for (const FsEntry& e : vol->entries) {
        if (e.offset < 0 || e.size < 0 ||
            (uint64_t)(e.offset + e.size) > vol->data.size()) {
            delete vol;
            return fs_fail(M3GPU_FS_ERR_BOUNDS, "entry bounds", path("data"));
        }
        if ((int64_t)fs_adler32(vol->data.data() + e.offset, (size_t)e.size) !=
            e.data_checksum) {
            delete vol;
            return fs_fail(M3GPU_FS_ERR_DATA_CHECKSUM, "data checksum", path("data"));
        }
    }

    std::lock_guard<std::mutex> lk(g_fs_mu);
    for (size_t i = 0; i < g_fs_volumes.size(); i++) {
        if (!g_fs_volumes[i]) { g_fs_volumes[i] = vol; return (int)i; }
    }
    g_fs_volumes.push_back(vol);
    return (int)g_fs_volumes.size() - 1;
}

static FsVolume* fs_get(int handle) {
    std::lock_guard<std::mutex> lk(g_fs_mu);
    if (handle < 0 || (size_t)handle >= g_fs_volumes.size()) return nullptr;
    return g_fs_volumes[handle];
}

int m3gpu_fileset_close(int handle) {
    std::lock_guard<std::mutex> lk(g_fs_mu);
    if (handle < 0 || (size_t)handle >= g_fs_volumes.size() ||
        !g_fs_volumes[handle])
        return M3GPU_FS_ERR_BADHANDLE;
    delete g_fs_volumes[handle];
    g_fs_volumes[handle] = nullptr;
    return 0;
}

int m3gpu_fileset_info(int handle, int64_t* block_start, int64_t* block_size,
                       int64_t* entries, int64_t* major_version,
                       int64_t* minor_version, int* volume_index,
                       int64_t* bloom_m, int64_t* bloom_k,
                       int64_t* summaries) {
    FsVolume* v = fs_get(handle);
    if (!v) return M3GPU_FS_ERR_BADHANDLE;
    if (block_start) *block_start = v->info.block_start;
    if (block_size) *block_size = v->info.block_size;
    if (entries) *entries = v->info.entries;
    if (major_version) *major_version = v->info.major_version;
    if (minor_version) *minor_version = v->info.minor_version;
    if (volume_index) *volume_index = v->info.volume_index;
    if (bloom_m) *bloom_m = v->info.bloom_m;
    if (bloom_k) *bloom_k = v->info.bloom_k;
    if (summaries) *summaries = v->info.summaries;
    return 0;
}

/* Entry metadata (entries ordered by data offset asc). id/tags pointers
 * are into reader-owned memory, valid until close. */
int m3gpu_fileset_entry(int handle, int64_t i, int64_t* size, int64_t* offset,
                        int64_t* data_checksum, const uint8_t** id,
                        int64_t* id_len, const uint8_t** tags,
                        int64_t* tags_len) {
    FsVolume* v = fs_get(handle);
    if (!v) return M3GPU_FS_ERR_BADHANDLE;
    if (i < 0 || (size_t)i >= v->entries.size()) return M3GPU_FS_ERR_BADHANDLE;
    const FsEntry& e = v->entries[(size_t)i];
    if (size) *size = e.size;
    if (offset) *offset = e.offset;
    if (data_checksum) *data_checksum = e.data_checksum;
    if (id) *id = e.id.data();
    if (id_len) *id_len = (int64_t)e.id.size();
    if (tags) *tags = e.tags.data();
    if (tags_len) *tags_len = (int64_t)e.tags.size();
    return 0;
}

/* Size of the packed blob m3gpu_fileset_pack would produce. */
int64_t m3gpu_fileset_packed_size(int handle) {
    FsVolume* v = fs_get(handle);
    if (!v) return M3GPU_FS_ERR_BADHANDLE;
    int64_t total = 0;
    for (const FsEntry& e : v->entries) total += (e.size + 63) & ~63ll;
    return total;
}

/* Repack the volume's data blocks into the decode-batch layout
 * (include/m3gpu.h): 16-byte-aligned offsets, zero padding between
 * blocks. offsets/lens arrays must hold entries elements. */
int m3gpu_fileset_pack(int handle, uint8_t* blob, uint64_t blob_cap,
                       uint64_t* offsets, uint32_t* lens) {
    FsVolume* v = fs_get(handle);
    if (!v) return M3GPU_FS_ERR_BADHANDLE;
    uint64_t off = 0;
    for (size_t i = 0; i < v->entries.size(); i++) {
        const FsEntry& e = v->entries[i];
        uint64_t aligned = ((uint64_t)e.size + 63) & ~63ull;
        if (off + aligned > blob_cap) return M3GPU_FS_ERR_CAPACITY;
        memcpy(blob + off, v->data.data() + e.offset, (size_t)e.size);
        if (aligned > (uint64_t)e.size)
            memset(blob + off + e.size, 0, (size_t)(aligned - e.size));
        offsets[i] = off;
        lens[i] = (uint32_t)e.size;
        off += aligned;
    }
    return 0;
}

} /* extern "C" */
