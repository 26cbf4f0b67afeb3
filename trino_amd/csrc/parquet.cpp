/* parquet.cpp — minimal native Parquet reader (§8f row 1).
 *
 * Re-designed MI355X-first replacement for the reference's Java Parquet
 * column reader on this path (lib/trino-parquet/src/main/java/io/trino/
 * parquet/reader/{ParquetReader,PageReader}.java and reader/flat/<leaf> leaf
 * decoders): columnar decode straight into flat column buffers that upload
 * to HBM unchanged (no row pivot), serving the "SF100 Parquet" scan config.
 *
 * Scope (matches what Trino-written/pyarrow-written TPC-H files use):
 *  - footer FileMetaData via Thrift compact protocol (hand parser, unknown
 *    fields skipped);
 *  - DataPage v1 + v2, required/optional flat columns (max rep level 0);
 *  - encodings PLAIN, RLE/bit-packed hybrid levels, RLE_DICTIONARY /
 *    PLAIN_DICTIONARY indices;
 *  - physical types INT32, INT64, DOUBLE, BYTE_ARRAY (returned as
 *    dictionary ids + dictionary when dict-encoded, else offsets+bytes);
 *  - codecs UNCOMPRESSED, SNAPPY (own decoder — the format is public),
 *    ZSTD (system libzstd.so.1 via dlopen).
 * Anything else returns TG_ERR_UNSUPPORTED with a precise message.
 */
#include "common.h"
#include <dlfcn.h>
#include <cstdio>
#include <string>
#include <vector>

/* ---------- thrift compact protocol ---------- */
struct TIn {
    const uint8_t* p;
    const uint8_t* end;
    bool ok = true;

    uint8_t u8() { if (p >= end) { ok = false; return 0; } return *p++; }
    uint64_t varint()
    {
        uint64_t v = 0;
        int sh = 0;
        while (p < end) {
            uint8_t b = *p++;
            v |= (uint64_t)(b & 0x7F) << sh;
            if (!(b & 0x80)) return v;
            sh += 7;
            if (sh > 63) break;
        }
        ok = false;
        return 0;
    }
    int64_t zigzag() { uint64_t v = varint(); return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }
    void skip_bytes(int64_t n) { if (end - p < n) { ok = false; p = end; } else p += n; }
};

enum TType { T_STOP = 0, T_TRUE = 1, T_FALSE = 2, T_BYTE = 3, T_I16 = 4, T_I32 = 5,
             T_I64 = 6, T_DOUBLE = 7, T_BINARY = 8, T_LIST = 9, T_SET = 10,
             T_MAP = 11, T_STRUCT = 12 };

static void t_skip(TIn& in, int type);

static void t_skip_struct(TIn& in)
{
    while (in.ok) {
        uint8_t b = in.u8();
        if (b == 0) return;
        int type = b & 0x0F;
        if ((b >> 4) == 0) in.zigzag();   /* long-form field id */
        t_skip(in, type);
    }
}

static void t_skip(TIn& in, int type)
{
    switch (type) {
        case T_TRUE: case T_FALSE: break;
        case T_BYTE: in.u8(); break;
        case T_I16: case T_I32: case T_I64: in.zigzag(); break;
        case T_DOUBLE: in.skip_bytes(8); break;
        case T_BINARY: { uint64_t n = in.varint(); in.skip_bytes((int64_t)n); break; }
        case T_LIST: case T_SET: {
            uint8_t h = in.u8();
            uint64_t n = h >> 4;
            int et = h & 0x0F;
            if (n == 15) n = in.varint();
            for (uint64_t i = 0; i < n && in.ok; i++) t_skip(in, et);
            break;
        }
        case T_MAP: {
            uint64_t n = in.varint();
            if (n == 0) break;
            uint8_t kv = in.u8();
            for (uint64_t i = 0; i < n && in.ok; i++) { t_skip(in, kv >> 4); t_skip(in, kv & 0x0F); }
            break;
        }
        case T_STRUCT: t_skip_struct(in); break;
        default: in.ok = false;
    }
}

/* iterate struct fields: calls f(field_id, type, in&) which MUST consume the
 * field value; return false from f to have it skipped here */
template <typename F>
static void t_struct(TIn& in, F&& f)
{
    int16_t fid = 0;
    while (in.ok) {
        uint8_t b = in.u8();
        if (b == 0) return;
        int type = b & 0x0F;
        int delta = b >> 4;
        if (delta == 0) fid = (int16_t)in.zigzag();
        else fid = (int16_t)(fid + delta);
        if (!f(fid, type, in)) t_skip(in, type);
    }
}

template <typename F>
static void t_list(TIn& in, F&& f)
{
    uint8_t h = in.u8();
    uint64_t n = h >> 4;
    int et = h & 0x0F;
    if (n == 15) n = in.varint();
    for (uint64_t i = 0; i < n && in.ok; i++) f(et, in);
}

/* ---------- parquet metadata (subset) ---------- */
struct PqColumnMeta {
    int32_t ptype = -1;            /* physical type */
    int32_t codec = 0;
    int64_t num_values = 0;
    int64_t data_page_offset = -1;
    int64_t dict_page_offset = -1;
    int64_t total_compressed = 0;
    std::string name;              /* last path element */
};
struct PqRowGroup {
    int64_t num_rows = 0;
    std::vector<PqColumnMeta> cols;
};
struct PqSchemaCol { std::string name; int32_t ptype = -1; bool optional = false; };

struct tg_parquet_file {
    std::vector<uint8_t> bytes;    /* whole file (round 1: files are read whole) */
    int64_t num_rows = 0;
    std::vector<PqSchemaCol> schema;     /* leaf columns, schema order */
    std::vector<PqRowGroup> row_groups;
};

static void parse_column_meta(TIn& in, PqColumnMeta* cm)
{
    t_struct(in, [&](int fid, int type, TIn& i2) {
        switch (fid) {
            case 1: cm->ptype = (int32_t)i2.zigzag(); return true;           /* type */
            case 3: { /* path_in_schema: list<string> */
                t_list(i2, [&](int, TIn& i3) {
                    uint64_t n = i3.varint();
                    cm->name.assign((const char*)i3.p, (size_t)n);
                    i3.skip_bytes((int64_t)n);
                });
                return true;
            }
            case 4: cm->codec = (int32_t)i2.zigzag(); return true;
            case 5: cm->num_values = i2.zigzag(); return true;
            case 7: cm->total_compressed = i2.zigzag(); return true;
            case 9: cm->data_page_offset = i2.zigzag(); return true;
            case 11: cm->dict_page_offset = i2.zigzag(); return true;
            default: return false;
        }
    });
}

static bool parse_footer(tg_parquet_file* f)
{
    size_t sz = f->bytes.size();
    if (sz < 12 || memcmp(f->bytes.data() + sz - 4, "PAR1", 4) != 0) return false;
    uint32_t mlen;
    memcpy(&mlen, f->bytes.data() + sz - 8, 4);
    if (mlen + 12 > sz) return false;
    TIn in{f->bytes.data() + sz - 8 - mlen, f->bytes.data() + sz - 8};

    t_struct(in, [&](int fid, int type, TIn& i2) {
        switch (fid) {
            case 2: { /* schema: list<SchemaElement> */
                bool root = true;
                t_list(i2, [&](int, TIn& i3) {
                    PqSchemaCol sc;
                    int32_t num_children = 0;
                    t_struct(i3, [&](int f2, int t2, TIn& i4) {
                        switch (f2) {
                            case 1: sc.ptype = (int32_t)i4.zigzag(); return true;
                            case 3: sc.optional = (i4.zigzag() == 1); return true; /* 0=REQUIRED 1=OPTIONAL 2=REPEATED */
                            case 4: { uint64_t n = i4.varint();
                                      sc.name.assign((const char*)i4.p, (size_t)n);
                                      i4.skip_bytes((int64_t)n); return true; }
                            case 5: num_children = (int32_t)i4.zigzag(); return true;
                            default: return false;
                        }
                    });
                    if (root) { root = false; return; }   /* message root */
                    if (num_children == 0) f->schema.push_back(sc);
                });
                return true;
            }
            case 3: f->num_rows = i2.zigzag(); return true;
            case 4: { /* row_groups */
                t_list(i2, [&](int, TIn& i3) {
                    PqRowGroup rg;
                    t_struct(i3, [&](int f2, int t2, TIn& i4) {
                        switch (f2) {
                            case 1: { /* columns: list<ColumnChunk> */
                                t_list(i4, [&](int, TIn& i5) {
                                    PqColumnMeta cm;
                                    t_struct(i5, [&](int f3, int t3, TIn& i6) {
                                        if (f3 == 3) { parse_column_meta(i6, &cm); return true; }
                                        return false;
                                    });
                                    rg.cols.push_back(cm);
                                });
                                return true;
                            }
                            case 3: rg.num_rows = i4.zigzag(); return true;
                            default: return false;
                        }
                    });
                    f->row_groups.push_back(rg);
                });
                return true;
            }
            default: return false;
        }
    });
    return in.ok && !f->schema.empty();
}

/* ---------- decompression ---------- */
static bool snappy_uncompress(const uint8_t* in, size_t in_len, uint8_t* out, size_t out_cap,
                              size_t* out_len)
{
    /* snappy raw format: varint uncompressed length, then tagged elements */
    size_t ip = 0, op = 0;
    uint64_t ulen = 0;
    int sh = 0;
    while (ip < in_len) {
        uint8_t b = in[ip++];
        ulen |= (uint64_t)(b & 0x7F) << sh;
        sh += 7;
        if (!(b & 0x80)) break;
    }
    if (ulen > out_cap) return false;
    while (ip < in_len) {
        uint8_t tag = in[ip++];
        int t = tag & 3;
        if (t == 0) {                      /* literal */
            size_t len = (tag >> 2) + 1;
            if (len > 60) {
                int nb = (int)len - 60;
                len = 0;
                for (int k = 0; k < nb; k++) len |= (size_t)in[ip++] << (8 * k);
                len += 1;
            }
            if (ip + len > in_len || op + len > out_cap) return false;
            memcpy(out + op, in + ip, len);
            ip += len;
            op += len;
        }
        else {                             /* copy */
            size_t len, off;
            if (t == 1) {
                len = ((tag >> 2) & 7) + 4;
                off = ((size_t)(tag >> 5) << 8) | in[ip++];
            }
            else if (t == 2) {
                len = (tag >> 2) + 1;
                off = (size_t)in[ip] | ((size_t)in[ip + 1] << 8);
                ip += 2;
            }
            else {
                len = (tag >> 2) + 1;
                off = (size_t)in[ip] | ((size_t)in[ip + 1] << 8) |
                      ((size_t)in[ip + 2] << 16) | ((size_t)in[ip + 3] << 24);
                ip += 4;
            }
            if (off == 0 || off > op || op + len > out_cap) return false;
            for (size_t k = 0; k < len; k++) { out[op] = out[op - off]; op++; }
        }
    }
    *out_len = op;
    return op == ulen;
}

typedef size_t (*zstd_decompress_fn)(void*, size_t, const void*, size_t);
typedef unsigned (*zstd_iserr_fn)(size_t);
static zstd_decompress_fn zstd_decompress_p = nullptr;
static zstd_iserr_fn zstd_iserr_p = nullptr;

static bool load_zstd()
{
    if (zstd_decompress_p) return true;
    void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) return false;
    zstd_decompress_p = (zstd_decompress_fn)dlsym(h, "ZSTD_decompress");
    zstd_iserr_p = (zstd_iserr_fn)dlsym(h, "ZSTD_isError");
    return zstd_decompress_p && zstd_iserr_p;
}

/* ---------- RLE / bit-packed hybrid ---------- */
static bool rle_decode(const uint8_t* in, size_t len, int bit_width, int64_t count,
                       int32_t* out)
{
    size_t ip = 0;
    int64_t o = 0;
    int byte_w = (bit_width + 7) / 8;
    while (o < count && ip < len) {
        uint64_t h = 0;
        int sh = 0;
        while (ip < len) {
            uint8_t b = in[ip++];
            h |= (uint64_t)(b & 0x7F) << sh;
            sh += 7;
            if (!(b & 0x80)) break;
        }
        if (h & 1) {                       /* bit-packed: (h>>1) groups of 8 */
            int64_t groups = (int64_t)(h >> 1);
            if (bit_width <= 16) {
                /* one group = 8 values = bit_width BYTES: unpack via a
                 * 128-bit register (scalar byte loop ran ~10x slower) */
                uint64_t mask = (1ull << bit_width) - 1;
                for (int64_t g = 0; g < groups; g++) {
                    unsigned __int128 bits = 0;
                    size_t take = (size_t)bit_width;
                    if (ip + take > len) take = len - ip;
                    memcpy(&bits, in + ip, take);
                    ip += take;
                    if (o + 8 <= count) {
                        #pragma unroll
                        for (int k = 0; k < 8; k++) {
                            out[o + k] = (int32_t)((uint64_t)bits & mask);
                            bits >>= bit_width;
                        }
                        o += 8;
                    }
                    else {
                        for (int k = 0; k < 8; k++) {
                            if (o < count) out[o++] = (int32_t)((uint64_t)bits & mask);
                            bits >>= bit_width;
                        }
                    }
                }
            }
            else {
                int64_t n = groups * 8;
                uint64_t buf = 0;
                int bits = 0;
                for (int64_t k = 0; k < n && o < count + 7; k++) {
                    while (bits < bit_width && ip < len) { buf |= (uint64_t)in[ip++] << bits; bits += 8; }
                    int32_t v = (int32_t)(buf & ((1ull << bit_width) - 1));
                    buf >>= bit_width;
                    bits -= bit_width;
                    if (o < count) out[o++] = v;
                }
            }
        }
        else {                             /* run: (h>>1) copies */
            int64_t n = (int64_t)(h >> 1);
            uint32_t v = 0;
            for (int k = 0; k < byte_w && ip < len; k++) v |= (uint32_t)in[ip++] << (8 * k);
            for (int64_t k = 0; k < n && o < count; k++) out[o++] = (int32_t)v;
        }
    }
    return o == count;
}

/* def-level stream == one RLE run of 1s covering nv values (the dominant
 * null-free OPTIONAL-column case): skip materializing levels entirely */
static bool defs_all_ones(const uint8_t* in, size_t len, int64_t nv)
{
    size_t ip = 0;
    uint64_t h = 0;
    int sh = 0;
    while (ip < len) {
        uint8_t b = in[ip++];
        h |= (uint64_t)(b & 0x7F) << sh;
        sh += 7;
        if (!(b & 0x80)) break;
    }
    if (h & 1) return false;               /* bit-packed: inspect normally */
    if ((int64_t)(h >> 1) < nv) return false;
    return ip < len && in[ip] == 1;        /* run value (bit width 1 => 1 byte) */
}

/* ---------- column reading ---------- */
struct PageHdr {
    int32_t type = -1;             /* 0 data, 2 dict, 3 data_v2 */
    int32_t uncompressed = 0, compressed = 0;
    int32_t num_values = 0;
    int32_t encoding = 0;
    int32_t def_encoding = 3;
    /* v2 */
    int32_t num_nulls = 0, num_rows = 0;
    int32_t def_len = 0, rep_len = 0;
    int32_t v2_compressed = 1;
};

static bool parse_page_header(TIn& in, PageHdr* h)
{
    t_struct(in, [&](int fid, int type, TIn& i2) {
        switch (fid) {
            case 1: h->type = (int32_t)i2.zigzag(); return true;
            case 2: h->uncompressed = (int32_t)i2.zigzag(); return true;
            case 3: h->compressed = (int32_t)i2.zigzag(); return true;
            case 5: /* data_page_header */
                t_struct(i2, [&](int f2, int t2, TIn& i3) {
                    switch (f2) {
                        case 1: h->num_values = (int32_t)i3.zigzag(); return true;
                        case 2: h->encoding = (int32_t)i3.zigzag(); return true;
                        case 3: h->def_encoding = (int32_t)i3.zigzag(); return true;
                        default: return false;
                    }
                });
                return true;
            case 7: /* dictionary_page_header */
                t_struct(i2, [&](int f2, int t2, TIn& i3) {
                    switch (f2) {
                        case 1: h->num_values = (int32_t)i3.zigzag(); return true;
                        case 2: h->encoding = (int32_t)i3.zigzag(); return true;
                        default: return false;
                    }
                });
                return true;
            case 8: /* data_page_header_v2 */
                t_struct(i2, [&](int f2, int t2, TIn& i3) {
                    switch (f2) {
                        case 1: h->num_values = (int32_t)i3.zigzag(); return true;
                        case 2: h->num_nulls = (int32_t)i3.zigzag(); return true;
                        case 3: h->num_rows = (int32_t)i3.zigzag(); return true;
                        case 4: h->encoding = (int32_t)i3.zigzag(); return true;
                        case 5: h->def_len = (int32_t)i3.zigzag(); return true;
                        case 6: h->rep_len = (int32_t)i3.zigzag(); return true;
                        case 7: h->v2_compressed = (t2 == T_TRUE) ? 1 : 0; return true;
                        default: return false;
                    }
                });
                return true;
            default: return false;
        }
    });
    return in.ok;
}

static tg_status decompress(int codec, const uint8_t* in, size_t clen, size_t ulen,
                            std::vector<uint8_t>* out)
{
    out->resize(ulen);
    if (codec == 0) {                          /* UNCOMPRESSED */
        if (clen != ulen) { TG_SET_ERR("parquet: size mismatch on uncompressed page"); return TG_ERR_INVALID_ARG; }
        memcpy(out->data(), in, ulen);
        return TG_OK;
    }
    /* NOTE: v1 pages are always codec-compressed; compressed size equal to
     * uncompressed size is possible and is NOT a stored page (a real snappy
     * page with clen==ulen==3127 broke the former shortcut). */
    if (codec == 1) {                           /* SNAPPY */
        size_t got = 0;
        if (!snappy_uncompress(in, clen, out->data(), ulen, &got) || got != ulen) {
            TG_SET_ERR("parquet: snappy decode failed");
            return TG_ERR_INVALID_ARG;
        }
        return TG_OK;
    }
    if (codec == 6) {                           /* ZSTD */
        if (!load_zstd()) { TG_SET_ERR("parquet: libzstd unavailable"); return TG_ERR_UNSUPPORTED; }
        size_t r = zstd_decompress_p(out->data(), ulen, in, clen);
        if (zstd_iserr_p(r) || r != ulen) { TG_SET_ERR("parquet: zstd decode failed"); return TG_ERR_INVALID_ARG; }
        return TG_OK;
    }
    TG_SET_ERR("parquet: unsupported codec %d (supported: uncompressed/snappy/zstd)", codec);
    return TG_ERR_UNSUPPORTED;
}

extern "C" tg_status tg_parquet_open(tg_session* s, const char* path, tg_parquet_file** out)
{
    (void)s;
    FILE* fp = fopen(path, "rb");
    if (!fp) { TG_SET_ERR("parquet: cannot open %s", path); return TG_ERR_INVALID_ARG; }
    fseek(fp, 0, SEEK_END);
    long sz = ftell(fp);
    fseek(fp, 0, SEEK_SET);
    auto* f = new tg_parquet_file();
    f->bytes.resize((size_t)sz);
    if (fread(f->bytes.data(), 1, (size_t)sz, fp) != (size_t)sz) {
        fclose(fp); delete f;
        TG_SET_ERR("parquet: short read");
        return TG_ERR_INVALID_ARG;
    }
    fclose(fp);
    if (!parse_footer(f)) {
        delete f;
        TG_SET_ERR("parquet: bad footer/metadata");
        return TG_ERR_INVALID_ARG;
    }
    *out = f;
    return TG_OK;
}

extern "C" void tg_parquet_close(tg_parquet_file* f) { delete f; }
extern "C" int64_t tg_parquet_num_rows(tg_parquet_file* f) { return f->num_rows; }
extern "C" int32_t tg_parquet_num_columns(tg_parquet_file* f) { return (int32_t)f->schema.size(); }
extern "C" const char* tg_parquet_column_name(tg_parquet_file* f, int32_t i)
{
    return (i >= 0 && i < (int32_t)f->schema.size()) ? f->schema[i].name.c_str() : "";
}
extern "C" int32_t tg_parquet_physical_type(tg_parquet_file* f, int32_t i)
{
    return (i >= 0 && i < (int32_t)f->schema.size()) ? f->schema[i].ptype : -1;
}

/* Decode one column across all row groups into caller host buffers.
 * Numerics (INT32/INT64/DOUBLE): values -> out_values (elem-sized), nulls ->
 * out_valid bitmap (may be NULL if column required / caller uninterested;
 * null slots are zero-filled). BYTE_ARRAY: pass out_ids (int32 per row) +
 * out_dict_bytes/out_dict_offsets sized via tg_parquet_dict_sizes. */
extern "C" tg_status tg_parquet_read_column(tg_session* s, tg_parquet_file* f, int32_t col,
    void* out_values, uint64_t* out_valid,
    int32_t* out_ids, uint8_t* out_dict_bytes, int64_t dict_bytes_cap,
    int32_t* out_dict_offsets, int32_t* out_dict_count)
{
    (void)s;
    if (col < 0 || col >= (int32_t)f->schema.size()) { TG_SET_ERR("bad column"); return TG_ERR_INVALID_ARG; }
    int ptype = f->schema[col].ptype;
    int esz = ptype == 1 ? 4 : 8;
    bool is_ba = ptype == 6;
    if (ptype != 1 && ptype != 2 && ptype != 5 && !is_ba) {
        TG_SET_ERR("parquet: unsupported physical type %d", ptype);
        return TG_ERR_UNSUPPORTED;
    }
    if (out_valid) {
        int64_t words = (f->num_rows + 63) / 64;
        memset(out_valid, 0xFF, (size_t)words * 8);
    }
    /* BYTE_ARRAY dictionary accumulated across row groups: per-row-group
     * dictionaries are remapped into one global dictionary by value */
    std::vector<uint8_t> gdict_bytes;
    std::vector<int32_t> gdict_offsets{0};

    /* per-row-group start rows: row groups are independent for numeric
     * columns, so they decode in parallel (OpenMP); BYTE_ARRAY columns stay
     * sequential (the global dictionary remap is order-dependent) */
    std::vector<int64_t> rg_base(f->row_groups.size() + 1, 0);
    for (size_t g = 0; g < f->row_groups.size(); g++)
        rg_base[g + 1] = rg_base[g] + f->row_groups[g].num_rows;

    auto read_group = [&](size_t gi) -> tg_status {
        auto& rg = f->row_groups[gi];
        int64_t row = rg_base[gi];
        const PqColumnMeta& cm = rg.cols[col];
        int64_t off = cm.dict_page_offset >= 0 &&
                      (cm.data_page_offset < 0 || cm.dict_page_offset < cm.data_page_offset)
                      ? cm.dict_page_offset : cm.data_page_offset;
        int64_t remaining = cm.num_values;
        /* row-group-local dictionary */
        std::vector<uint8_t> dict_raw;          /* PLAIN dictionary page payload */
        std::vector<int64_t> dict_num(1, 0);
        std::vector<int32_t> ba_dict_remap;     /* local id -> global id */
        int64_t dict_count = 0;

        while (remaining > 0) {
            if (off < 0 || off >= (int64_t)f->bytes.size()) { TG_SET_ERR("parquet: bad page offset"); return TG_ERR_INVALID_ARG; }
            TIn in{f->bytes.data() + off, f->bytes.data() + f->bytes.size()};
            PageHdr h;
            if (!parse_page_header(in, &h)) { TG_SET_ERR("parquet: bad page header"); return TG_ERR_INVALID_ARG; }
            const uint8_t* body = in.p;
            if (getenv("TG_PQ_DEBUG"))
                fprintf(stderr, "[pq] off=%lld type=%d nv=%d enc=%d comp=%d uncomp=%d remaining=%lld row=%lld\n",
                        (long long)off, h.type, h.num_values, h.encoding,
                        h.compressed, h.uncompressed, (long long)remaining, (long long)row);
            off = (body - f->bytes.data()) + h.compressed;

            if (h.type == 2) {                  /* dictionary page */
                tg_status st = decompress(cm.codec, body, h.compressed, h.uncompressed, &dict_raw);
                if (st != TG_OK) return st;
                dict_count = h.num_values;
                if (is_ba) {
                    /* split into values and remap into the global dictionary */
                    ba_dict_remap.resize(dict_count);
                    size_t ip = 0;
                    for (int64_t d = 0; d < dict_count; d++) {
                        uint32_t len;
                        memcpy(&len, dict_raw.data() + ip, 4);
                        ip += 4;
                        /* linear search global dict (dicts are tiny on this path) */
                        int32_t gid = -1;
                        for (size_t g = 0; g + 1 < gdict_offsets.size(); g++) {
                            int32_t glen = gdict_offsets[g + 1] - gdict_offsets[g];
                            if ((uint32_t)glen == len &&
                                memcmp(gdict_bytes.data() + gdict_offsets[g], dict_raw.data() + ip, len) == 0) {
                                gid = (int32_t)g;
                                break;
                            }
                        }
                        if (gid < 0) {
                            gid = (int32_t)gdict_offsets.size() - 1;
                            gdict_bytes.insert(gdict_bytes.end(), dict_raw.data() + ip,
                                               dict_raw.data() + ip + len);
                            gdict_offsets.push_back((int32_t)gdict_bytes.size());
                        }
                        ba_dict_remap[d] = gid;
                        ip += len;
                    }
                }
                continue;
            }
            if (h.type != 0 && h.type != 3) { TG_SET_ERR("parquet: unexpected page type %d", h.type); return TG_ERR_UNSUPPORTED; }

            std::vector<uint8_t> page;
            const uint8_t* levels_src = nullptr;
            const uint8_t* vals = nullptr;
            size_t vals_len = 0;
            int32_t def_len = 0;
            if (h.type == 3) {                  /* v2: levels stored uncompressed before body */
                def_len = h.def_len;
                levels_src = body + h.rep_len;
                const uint8_t* enc_vals = body + h.rep_len + h.def_len;
                size_t enc_len = h.compressed - h.rep_len - h.def_len;
                size_t un_len = h.uncompressed - h.rep_len - h.def_len;
                if (h.v2_compressed) {
                    tg_status st = decompress(cm.codec, enc_vals, enc_len, un_len, &page);
                    if (st != TG_OK) return st;
                    vals = page.data(); vals_len = page.size();
                }
                else { vals = enc_vals; vals_len = un_len; }
            }
            else {
                if (cm.codec == 0) {        /* uncompressed: zero-copy */
                    vals = body;
                    vals_len = (size_t)h.uncompressed;
                }
                else {
                    tg_status st = decompress(cm.codec, body, h.compressed, h.uncompressed, &page);
                    if (st != TG_OK) return st;
                    vals = page.data();
                    vals_len = page.size();
                }
                /* v1: [def level length i32 + RLE] before values (flat schema:
                 * no rep levels; REQUIRED columns have no level section).
                 * Heuristic per spec: optional column => levels present. */
                if (f->schema[col].optional) {
                    memcpy(&def_len, vals, 4);
                    levels_src = vals + 4;
                    vals = vals + 4 + def_len;
                    vals_len -= 4 + def_len;
                }
            }

            int64_t nv = h.num_values;
            std::vector<int32_t> defs;
            int64_t n_nonnull = nv;
            if (levels_src && def_len > 0 &&
                defs_all_ones(levels_src, (size_t)def_len, nv)) {
                /* null-free page: fall through with empty defs */
            }
            else if (levels_src && def_len > 0) {
                defs.resize(nv);
                if (!rle_decode(levels_src, (size_t)def_len, 1, nv, defs.data())) {
                    TG_SET_ERR("parquet: def level decode failed");
                    return TG_ERR_INVALID_ARG;
                }
                n_nonnull = 0;
                for (int64_t i = 0; i < nv; i++) n_nonnull += defs[i];
            }

            /* decode values of this page */
            std::vector<int32_t> idx;
            const uint8_t* plain = vals;
            bool dict_encoded = (h.encoding == 8 || h.encoding == 2);
            if (dict_encoded) {
                int bw = vals[0];
                idx.resize(n_nonnull);
                if (!rle_decode(vals + 1, vals_len - 1, bw, n_nonnull, idx.data())) {
                    TG_SET_ERR("parquet: dict index decode failed");
                    return TG_ERR_INVALID_ARG;
                }
            }
            else if (h.encoding != 0) {
                TG_SET_ERR("parquet: unsupported data encoding %d", h.encoding);
                return TG_ERR_UNSUPPORTED;
            }

            size_t pp = 0;            /* PLAIN cursor */
            int64_t k = 0;            /* non-null cursor */
            if (defs.empty() && !is_ba) {
                /* null-free fast paths (TPC-H columns are REQUIRED) */
                if (!dict_encoded) {
                    memcpy((uint8_t*)out_values + row * esz, vals, (size_t)nv * esz);
                }
                else if (esz == 8) {
                    const uint64_t* dv = (const uint64_t*)dict_raw.data();
                    uint64_t* ov = (uint64_t*)out_values + row;
                    for (int64_t i = 0; i < nv; i++) ov[i] = dv[idx[i]];
                }
                else {
                    const uint32_t* dv = (const uint32_t*)dict_raw.data();
                    uint32_t* ov = (uint32_t*)out_values + row;
                    for (int64_t i = 0; i < nv; i++) ov[i] = dv[idx[i]];
                }
                row += nv;
                remaining -= nv;
                continue;
            }
            if (defs.empty() && is_ba && dict_encoded && out_ids) {
                int32_t* ov = out_ids + row;
                for (int64_t i = 0; i < nv; i++) ov[i] = ba_dict_remap[idx[i]];
                row += nv;
                remaining -= nv;
                continue;
            }
            for (int64_t i = 0; i < nv; i++) {
                int64_t r = row + i;
                bool isnull = !defs.empty() && defs[i] == 0;
                if (isnull) {
                    if (out_valid) out_valid[r >> 6] &= ~(1ull << (r & 63));
                    if (is_ba) { if (out_ids) out_ids[r] = 0; }
                    else memset((uint8_t*)out_values + r * esz, 0, esz);
                    continue;
                }
                if (is_ba) {
                    int32_t gid;
                    if (dict_encoded) gid = ba_dict_remap[idx[k]];
                    else {            /* PLAIN byte array: append to global dict */
                        uint32_t len;
                        memcpy(&len, plain + pp, 4);
                        pp += 4;
                        gid = (int32_t)gdict_offsets.size() - 1;
                        gdict_bytes.insert(gdict_bytes.end(), plain + pp, plain + pp + len);
                        gdict_offsets.push_back((int32_t)gdict_bytes.size());
                        pp += len;
                    }
                    out_ids[r] = gid;
                }
                else if (dict_encoded) {
                    memcpy((uint8_t*)out_values + r * esz, dict_raw.data() + (int64_t)idx[k] * esz, esz);
                }
                else {
                    memcpy((uint8_t*)out_values + r * esz, plain + pp, esz);
                    pp += esz;
                }
                k++;
            }
            row += nv;
            remaining -= nv;
        }
        return TG_OK;
    };

    tg_status gst = TG_OK;
    if (is_ba) {
        for (size_t gi = 0; gi < f->row_groups.size() && gst == TG_OK; gi++)
            gst = read_group(gi);
    }
    else {
        #pragma omp parallel for schedule(dynamic)
        for (int64_t gi = 0; gi < (int64_t)f->row_groups.size(); gi++) {
            if (gst != TG_OK) continue;
            tg_status st = read_group((size_t)gi);
            if (st != TG_OK) gst = st;   /* benign race: any error wins */
        }
    }
    if (gst != TG_OK) return gst;
    if (is_ba) {
        int32_t dc = (int32_t)gdict_offsets.size() - 1;
        if (out_dict_count) *out_dict_count = dc;
        if (out_dict_offsets) memcpy(out_dict_offsets, gdict_offsets.data(), (dc + 1) * 4);
        if (out_dict_bytes) {
            if ((int64_t)gdict_bytes.size() > dict_bytes_cap) { TG_SET_ERR("parquet: dict bytes cap"); return TG_ERR_INVALID_ARG; }
            memcpy(out_dict_bytes, gdict_bytes.data(), gdict_bytes.size());
        }
    }
    return TG_OK;
}

#include <omp.h>

/* Decode several columns concurrently: columns in an outer OMP team, row
 * groups in each column's inner team (nested; ~cores total). The
 * reference's ParquetReader prefetches/decodes column chunks in parallel
 * the same way (reader/ParquetReader.java row-group column chunks). */
extern "C" tg_status tg_parquet_read_columns(tg_session* s, tg_parquet_file* f,
    const int32_t* cols, int32_t n_cols, void** out_values,
    uint64_t** out_valid, int32_t** out_ids, uint8_t** out_dict_bytes,
    const int64_t* dict_caps, int32_t** out_dict_offsets,
    int32_t** out_dict_counts)
{
    /* columns SEQUENTIAL, row groups parallel (each read_column's inner
     * `parallel for` uses the full team). The previous nested-team design
     * (outer team of n_cols, inner teams of threads/n_cols) measured
     * 0.5-3.7 s for 6 columns x 24M rows where the sequential-columns
     * version takes ~0.1 s: with threads/n_cols == 1 every column decoded
     * its row groups serially, and the per-call nested team setup
     * thrashed. */
    tg_status st = TG_OK;
    for (int32_t c = 0; c < n_cols && st == TG_OK; c++) {
        st = tg_parquet_read_column(
            s, f, cols[c], out_values ? out_values[c] : nullptr,
            out_valid ? out_valid[c] : nullptr,
            out_ids ? out_ids[c] : nullptr,
            out_dict_bytes ? out_dict_bytes[c] : nullptr,
            dict_caps ? dict_caps[c] : 0,
            out_dict_offsets ? out_dict_offsets[c] : nullptr,
            out_dict_counts ? out_dict_counts[c] : nullptr);
    }
    return st;
}
