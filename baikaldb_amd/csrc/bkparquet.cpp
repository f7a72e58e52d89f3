// bkparquet.cpp — minimal from-scratch Parquet reader for the cold columnar
// ingestion path. The reference stages OLAP cold data as
// parquet files (src/column/file_manager.h:252-334, parquet_writer.h:119,
// read via Arrow). This replaces that staging for the hot path WITHOUT
// linking Arrow: it parses the thrift-compact footer and PLAIN data pages
// directly and hands host column buffers to the engine
// (bkgpu_table_create/upload).
//
// Supported envelope (v1, stated in DESIGN.md):
//   - format v1 files, "PAR1" magic, thrift-compact FileMetaData
//   - physical types INT64 and DOUBLE (flat schema, no nesting)
//   - uncompressed PLAIN-encoded DataPage v1; multiple pages / row groups
//   - OPTIONAL columns via definition levels (RLE/bit-packed hybrid,
//     max_def_level == 1)
// Everything else (codecs, dictionary pages, v2 pages, nesting) is rejected
// with a clear error, never silently misread.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <algorithm>
#include <string>
#include <vector>

namespace bkparquet {

static thread_local std::string g_err;
static void seterr(const std::string& e) { g_err = e; }

struct Cursor {
    const uint8_t* p;
    const uint8_t* end;
    bool ok = true;
    uint8_t u8() {
        if (p >= end) { ok = false; return 0; }
        return *p++;
    }
    uint64_t varint() {
        uint64_t v = 0;
        int sh = 0;
        while (ok) {
            uint8_t b = u8();
            v |= (uint64_t)(b & 0x7F) << sh;
            if (!(b & 0x80)) break;
            sh += 7;
            if (sh > 63) { ok = false; break; }
        }
        return v;
    }
    int64_t zigzag() {
        uint64_t v = varint();
        return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
    }
    void bytes(size_t n) {
        if ((size_t)(end - p) < n) { ok = false; return; }
        p += n;
    }
    std::string str() {
        uint64_t n = varint();
        if ((uint64_t)(end - p) < n) { ok = false; return ""; }
        std::string s((const char*)p, (size_t)n);
        p += n;
        return s;
    }
};

/* thrift compact element types */
enum { T_STOP = 0, T_TRUE = 1, T_FALSE = 2, T_BYTE = 3, T_I16 = 4, T_I32 = 5,
       T_I64 = 6, T_DOUBLE = 7, T_BINARY = 8, T_LIST = 9, T_SET = 10,
       T_MAP = 11, T_STRUCT = 12 };

static void skip_value(Cursor& c, int type);

static void skip_struct(Cursor& c) {
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) return;
        int type = b & 0xF;
        if ((b >> 4) == 0) (void)c.zigzag();  /* long-form field id */
        skip_value(c, type);
    }
}

static void skip_value(Cursor& c, int type) {
    switch (type) {
        case T_TRUE: case T_FALSE: break;   /* value in type */
        case T_BYTE: case T_I16: case T_I32: case T_I64: (void)c.zigzag(); break;
        case T_DOUBLE: c.bytes(8); break;
        case T_BINARY: { uint64_t n = c.varint(); c.bytes((size_t)n); break; }
        case T_LIST: case T_SET: {
            uint8_t h = c.u8();
            uint64_t n = h >> 4;
            int et = h & 0xF;
            if (n == 0xF) n = c.varint();
            for (uint64_t i = 0; i < n && c.ok; i++) {
                if (et == T_TRUE || et == T_FALSE) c.bytes(1);
                else skip_value(c, et);
            }
            break;
        }
        case T_MAP: {
            uint64_t n = c.varint();
            if (n) {
                uint8_t kv = c.u8();
                for (uint64_t i = 0; i < n && c.ok; i++) {
                    skip_value(c, kv >> 4);
                    skip_value(c, kv & 0xF);
                }
            }
            break;
        }
        case T_STRUCT: skip_struct(c); break;
        default: c.ok = false;
    }
}

/* list header helper: returns count, sets elem type */
static uint64_t list_head(Cursor& c, int* et) {
    uint8_t h = c.u8();
    uint64_t n = h >> 4;
    *et = h & 0xF;
    if (n == 0xF) n = c.varint();
    return n;
}

struct SchemaCol {
    std::string name;
    int physical_type = -1;   /* parquet: 1=INT32? no: 0=BOOL,1=INT32,2=INT64,
                                 4=FLOAT,5=DOUBLE,6=BYTE_ARRAY */
    bool optional = false;
};

struct ColChunk {
    int physical_type = -1;
    int codec = -1;           /* 0 = UNCOMPRESSED */
    int64_t num_values = 0;
    int64_t data_page_offset = -1;
    int64_t dict_page_offset = -1;
    std::vector<std::string> path;
};

struct RowGroup {
    std::vector<ColChunk> cols;
    int64_t num_rows = 0;
};

struct FileMeta {
    std::vector<SchemaCol> schema;   /* leaf columns, root excluded */
    std::vector<RowGroup> groups;
    int64_t num_rows = 0;
};

static void parse_schema_element(Cursor& c, std::vector<SchemaCol>& out,
                                 bool* is_root) {
    SchemaCol sc;
    int num_children = 0;
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        switch (fid) {
            case 1: sc.physical_type = (int)c.zigzag(); break;   /* type */
            case 3: sc.optional = (c.zigzag() == 1); break;      /* repetition */
            case 4: sc.name = c.str(); break;
            case 5: num_children = (int)c.zigzag(); break;
            default: skip_value(c, type);
        }
    }
    if (num_children > 0) *is_root = true;   /* group node (root) */
    else { *is_root = false; out.push_back(sc); }
}

static void parse_column_meta(Cursor& c, ColChunk& cc) {
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        switch (fid) {
            case 1: cc.physical_type = (int)c.zigzag(); break;
            case 3: {   /* path_in_schema: list<string> */
                int et;
                uint64_t n = list_head(c, &et);
                for (uint64_t i = 0; i < n && c.ok; i++)
                    cc.path.push_back(c.str());
                break;
            }
            case 4: cc.codec = (int)c.zigzag(); break;
            case 5: cc.num_values = c.zigzag(); break;
            case 9: cc.data_page_offset = c.zigzag(); break;
            case 11: cc.dict_page_offset = c.zigzag(); break;
            default: skip_value(c, type);
        }
    }
}

static void parse_column_chunk(Cursor& c, ColChunk& cc) {
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        if (fid == 3 && type == T_STRUCT) parse_column_meta(c, cc);
        else skip_value(c, type);
    }
}

static void parse_row_group(Cursor& c, RowGroup& rg) {
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        if (fid == 1 && type == T_LIST) {
            int et;
            uint64_t n = list_head(c, &et);
            for (uint64_t i = 0; i < n && c.ok; i++) {
                ColChunk cc;
                parse_column_chunk(c, cc);
                rg.cols.push_back(cc);
            }
        } else if (fid == 3) {
            rg.num_rows = c.zigzag();
        } else {
            skip_value(c, type);
        }
    }
}

static bool parse_file_meta(Cursor& c, FileMeta& fm) {
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        if (fid == 2 && type == T_LIST) {          /* schema */
            int et;
            uint64_t n = list_head(c, &et);
            for (uint64_t i = 0; i < n && c.ok; i++) {
                bool root;
                parse_schema_element(c, fm.schema, &root);
            }
        } else if (fid == 3) {
            fm.num_rows = c.zigzag();
        } else if (fid == 4 && type == T_LIST) {   /* row_groups */
            int et;
            uint64_t n = list_head(c, &et);
            for (uint64_t i = 0; i < n && c.ok; i++) {
                RowGroup rg;
                parse_row_group(c, rg);
                fm.groups.push_back(rg);
            }
        } else {
            skip_value(c, type);
        }
    }
    return c.ok;
}

/* ---- page header (thrift) ---- */
struct PageHeader {
    int type = -1;             /* 0 = DATA_PAGE, 2 = DICTIONARY_PAGE */
    int32_t uncompressed_size = 0;
    int32_t compressed_size = 0;
    int32_t num_values = 0;
    int encoding = -1;         /* 0 = PLAIN */
    int def_encoding = -1;     /* 3 = RLE */
};

static void parse_data_page_header(Cursor& c, PageHeader& ph) {
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        switch (fid) {
            case 1: ph.num_values = (int32_t)c.zigzag(); break;
            case 2: ph.encoding = (int)c.zigzag(); break;
            case 3: ph.def_encoding = (int)c.zigzag(); break;
            default: skip_value(c, type);
        }
    }
}

static bool parse_page_header(Cursor& c, PageHeader& ph) {
    int fid = 0;
    for (;;) {
        uint8_t b = c.u8();
        if (!c.ok || b == 0) break;
        int type = b & 0xF;
        int delta = b >> 4;
        if (delta == 0) fid = (int)c.zigzag(); else fid += delta;
        switch (fid) {
            case 1: ph.type = (int)c.zigzag(); break;
            case 2: ph.uncompressed_size = (int32_t)c.zigzag(); break;
            case 3: ph.compressed_size = (int32_t)c.zigzag(); break;
            case 5: if (type == T_STRUCT) { parse_data_page_header(c, ph); break; }
                    /* fallthrough to skip for non-struct */
                    skip_value(c, type); break;
            default: skip_value(c, type);
        }
    }
    return c.ok;
}

/* RLE/bit-packed hybrid decoder (parquet Encodings.md):
 *   header = varint; header & 1 ? bit-packed group of (header>>1)*8 values
 *                                : RLE run of (header>>1) copies of 1 value
 * (RLE value stored in ceil(width/8) bytes LE). Decodes into u32. */
static bool rle_hybrid(const uint8_t*& p, const uint8_t* rend, int width,
                       int64_t nvals, uint32_t* out) {
    int vbytes = (width + 7) / 8;
    int64_t i = 0;
    Cursor c{p, rend};
    while (i < nvals && c.p < rend && c.ok) {
        uint64_t h = c.varint();
        if (!c.ok) return false;
        if (h & 1) {              /* bit-packed: (h>>1) groups of 8 values */
            uint64_t groups = h >> 1;
            uint64_t acc = 0;
            int nbits = 0;
            for (uint64_t g = 0; g < groups && i < nvals; g++) {
                for (int k = 0; k < 8 && i < nvals; k++) {
                    while (nbits < width) {
                        if (c.p >= rend) return false;
                        acc |= (uint64_t)(*c.p++) << nbits;
                        nbits += 8;
                    }
                    out[i++] = (uint32_t)(acc & ((width == 32)
                                   ? 0xFFFFFFFFull
                                   : ((1ull << width) - 1)));
                    acc >>= width;
                    nbits -= width;
                }
                /* a full group consumes exactly width bytes; partial final
                 * group already consumed what it needed via the loop */
            }
        } else {                  /* RLE run */
            uint64_t run = h >> 1;
            uint32_t v = 0;
            for (int b = 0; b < vbytes; b++) {
                if (c.p >= rend) return false;
                v |= (uint32_t)(*c.p++) << (8 * b);
            }
            for (uint64_t r = 0; r < run && i < nvals; r++) out[i++] = v;
        }
    }
    p = c.p;
    return i == nvals;
}

/* definition levels, width 1 (flat optional column), v1 page layout:
 * i32 LE byte length then RLE/bit-packed hybrid */
static bool read_def_levels(Cursor& c, int32_t nvals, uint8_t* def) {
    if ((size_t)(c.end - c.p) < 4) return false;
    uint32_t len;
    memcpy(&len, c.p, 4);
    c.p += 4;
    const uint8_t* rend = c.p + len;
    if (rend > c.end) return false;
    std::vector<uint32_t> tmp((size_t)nvals);
    const uint8_t* p = c.p;
    if (!rle_hybrid(p, rend, 1, nvals, tmp.data())) return false;
    for (int32_t i = 0; i < nvals; i++) def[i] = (uint8_t)(tmp[i] & 1);
    c.p = rend;
    return true;
}

struct Reader {
    std::vector<uint8_t> buf;
    FileMeta meta;
};

static bool load_file(const char* path, std::vector<uint8_t>& buf) {
    FILE* f = fopen(path, "rb");
    if (!f) { seterr(std::string("open failed: ") + path); return false; }
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    buf.resize((size_t)n);
    size_t got = fread(buf.data(), 1, (size_t)n, f);
    fclose(f);
    if (got != (size_t)n) { seterr("short read"); return false; }
    return true;
}

static Reader* open_reader(const char* path) {
    Reader* r = new Reader();
    if (!load_file(path, r->buf)) { delete r; return nullptr; }
    const std::vector<uint8_t>& b = r->buf;
    if (b.size() < 12 || memcmp(b.data(), "PAR1", 4) != 0 ||
        memcmp(b.data() + b.size() - 4, "PAR1", 4) != 0) {
        seterr("not a parquet file (PAR1 magic)");
        delete r;
        return nullptr;
    }
    uint32_t mlen;
    memcpy(&mlen, b.data() + b.size() - 8, 4);
    if (mlen + 12ull > b.size()) { seterr("bad footer length"); delete r; return nullptr; }
    Cursor c{b.data() + b.size() - 8 - mlen, b.data() + b.size() - 8};
    if (!parse_file_meta(c, r->meta)) {
        seterr("footer thrift parse failed");
        delete r;
        return nullptr;
    }
    return r;
}

/* decode the dictionary page of a chunk (PLAIN-encoded entries).
 * For INT64/DOUBLE: 8-byte values. For BYTE_ARRAY: [u32 len][bytes]. */
static bool read_dict_page(Reader* r, const ColChunk& cc, int ptype,
                           std::vector<uint64_t>* vals,
                           std::vector<std::string>* strs) {
    Cursor c{r->buf.data() + cc.dict_page_offset,
             r->buf.data() + r->buf.size()};
    PageHeader ph;
    if (!parse_page_header(c, ph)) { seterr("dict page header parse failed"); return false; }
    if (ph.type != 2) { seterr("expected dictionary page"); return false; }
    Cursor pc{c.p, c.p + ph.uncompressed_size};
    /* DictionaryPageHeader num_values lives in field 7; our PageHeader parser
     * keeps only data_page_header, so recover the count from entries. */
    if (ptype == 6) {
        while (pc.p + 4 <= pc.end) {
            uint32_t len;
            memcpy(&len, pc.p, 4);
            pc.p += 4;
            if ((size_t)(pc.end - pc.p) < len) { seterr("dict entry underrun"); return false; }
            strs->emplace_back((const char*)pc.p, (size_t)len);
            pc.p += len;
        }
    } else {
        while (pc.p + 8 <= pc.end) {
            uint64_t v;
            memcpy(&v, pc.p, 8);
            pc.p += 8;
            vals->push_back(v);
        }
    }
    return true;
}

/* read one leaf column across all row groups into host buffers.
 * INT64/DOUBLE -> out is int64/double[num_rows];
 * BYTE_ARRAY   -> out is int32[num_rows] ORDER-PRESERVING dict codes and
 *                 *dict_out receives the sorted word list (code c = c-th
 *                 smallest string, so code order == byte order — the
 *                 invariant the engine's string MIN/MAX/ORDER BY relies on).
 * valid: per-row 1/0 (may be null for required columns).
 * Returns rows read, < 0 on error. */
static int64_t read_column(Reader* r, int col, void* out, uint8_t* valid,
                           std::vector<std::string>* dict_out) {
    const FileMeta& fm = r->meta;
    if (col < 0 || (size_t)col >= fm.schema.size()) { seterr("bad column"); return -1; }
    const SchemaCol& sc = fm.schema[col];
    int ptype = sc.physical_type;
    if (ptype != 2 && ptype != 5 && ptype != 6) {
        seterr("unsupported physical type (need INT64/DOUBLE/BYTE_ARRAY): col " +
               sc.name);
        return -1;
    }
    bool is_str = ptype == 6;
    if (is_str && !dict_out) { seterr("string column needs dict_out"); return -1; }

    /* pass 1 for strings: collect the global word set to build the
     * order-preserving code assignment across all row groups */
    std::vector<std::string> words;   /* sorted unique words */
    if (is_str) {
        std::vector<std::string> all;
        for (const RowGroup& rg : fm.groups) {
            const ColChunk& cc = rg.cols[col];
            if (cc.dict_page_offset >= 0) {
                std::vector<uint64_t> dv;
                std::vector<std::string> ds;
                if (!read_dict_page(r, cc, ptype, &dv, &ds)) return -1;
                for (auto& w : ds) all.push_back(std::move(w));
            }
        }
        /* PLAIN (non-dict) string pages contribute words during pass 2; to
         * keep one code space we scan them here too */
        for (const RowGroup& rg : fm.groups) {
            const ColChunk& cc = rg.cols[col];
            if (cc.dict_page_offset >= 0) continue;
            int64_t remaining = cc.num_values;
            int64_t off = cc.data_page_offset;
            while (remaining > 0) {
                Cursor c{r->buf.data() + off, r->buf.data() + r->buf.size()};
                PageHeader ph;
                if (!parse_page_header(c, ph)) { seterr("page header parse failed"); return -1; }
                const uint8_t* data = c.p;
                off = (int64_t)(data - r->buf.data()) + ph.compressed_size;
                if (ph.type != 0 || ph.encoding != 0) { seterr("unsupported string page"); return -1; }
                Cursor pc{data, data + ph.uncompressed_size};
                std::vector<uint8_t> def((size_t)ph.num_values, 1);
                if (sc.optional && !read_def_levels(pc, ph.num_values, def.data()))
                    { seterr("def level decode failed"); return -1; }
                for (int32_t i = 0; i < ph.num_values; i++) {
                    if (!def[(size_t)i]) continue;
                    if ((size_t)(pc.end - pc.p) < 4) { seterr("string underrun"); return -1; }
                    uint32_t len;
                    memcpy(&len, pc.p, 4);
                    pc.p += 4;
                    if ((size_t)(pc.end - pc.p) < len) { seterr("string underrun"); return -1; }
                    all.emplace_back((const char*)pc.p, (size_t)len);
                    pc.p += len;
                }
                remaining -= ph.num_values;
            }
        }
        std::sort(all.begin(), all.end());
        all.erase(std::unique(all.begin(), all.end()), all.end());
        words = std::move(all);
        if (words.size() > (1u << 30)) { seterr("dictionary too large"); return -1; }
    }
    auto code_of = [&](const std::string& w) -> int32_t {
        return (int32_t)(std::lower_bound(words.begin(), words.end(), w) -
                         words.begin());
    };

    int64_t row = 0;
    std::vector<uint8_t> def;
    std::vector<uint32_t> idx;
    for (const RowGroup& rg : fm.groups) {
        if ((size_t)col >= rg.cols.size()) { seterr("row group missing column"); return -1; }
        const ColChunk& cc = rg.cols[col];
        if (cc.codec != 0) { seterr("compressed parquet unsupported (codec != UNCOMPRESSED)"); return -1; }
        /* chunk dictionary (original parquet code order, NOT our code space) */
        std::vector<uint64_t> dictv;
        std::vector<std::string> dicts;
        std::vector<int32_t> remap;   /* parquet dict idx -> engine code */
        if (cc.dict_page_offset >= 0) {
            if (!read_dict_page(r, cc, ptype, &dictv, &dicts)) return -1;
            if (is_str) {
                remap.resize(dicts.size());
                for (size_t i = 0; i < dicts.size(); i++)
                    remap[i] = code_of(dicts[i]);
            }
        }
        int64_t remaining = cc.num_values;
        int64_t off = cc.data_page_offset;
        while (remaining > 0) {
            if (off < 0 || (size_t)off >= r->buf.size()) { seterr("bad page offset"); return -1; }
            Cursor c{r->buf.data() + off, r->buf.data() + r->buf.size()};
            PageHeader ph;
            if (!parse_page_header(c, ph)) { seterr("page header parse failed"); return -1; }
            const uint8_t* data = c.p;
            off = (int64_t)(data - r->buf.data()) + ph.compressed_size;
            if (ph.type != 0) { seterr("unsupported page type (need v1 data pages)"); return -1; }
            Cursor pc{data, data + ph.uncompressed_size};
            int32_t nv = ph.num_values;
            def.assign((size_t)nv, 1);
            if (sc.optional) {
                if (ph.def_encoding != 3) { seterr("def levels must be RLE"); return -1; }
                if (!read_def_levels(pc, nv, def.data())) { seterr("def level decode failed"); return -1; }
            }
            int64_t npresent = 0;
            for (int32_t i = 0; i < nv; i++) npresent += def[(size_t)i];
            if (ph.encoding == 2 || ph.encoding == 8) {
                /* PLAIN_DICTIONARY / RLE_DICTIONARY indices */
                if (cc.dict_page_offset < 0) { seterr("dict-encoded page without dict"); return -1; }
                if (pc.p >= pc.end) { seterr("missing index bit width"); return -1; }
                int width = *pc.p++;
                idx.assign((size_t)npresent, 0);
                const uint8_t* p = pc.p;
                if (width > 0) {
                    if (!rle_hybrid(p, pc.end, width, npresent, idx.data()))
                        { seterr("index decode failed"); return -1; }
                }
                int64_t k = 0;
                for (int32_t i = 0; i < nv; i++) {
                    if (row >= fm.num_rows) { seterr("row overflow"); return -1; }
                    if (def[(size_t)i]) {
                        uint32_t ix = idx[(size_t)k++];
                        if (is_str) {
                            if (ix >= remap.size()) { seterr("dict index out of range"); return -1; }
                            ((int32_t*)out)[row] = remap[ix];
                        } else {
                            if (ix >= dictv.size()) { seterr("dict index out of range"); return -1; }
                            memcpy((uint8_t*)out + (size_t)row * 8, &dictv[ix], 8);
                        }
                        if (valid) valid[row] = 1;
                    } else {
                        if (is_str) ((int32_t*)out)[row] = 0;
                        else memset((uint8_t*)out + (size_t)row * 8, 0, 8);
                        if (valid) valid[row] = 0;
                    }
                    row++;
                }
            } else if (ph.encoding == 0) {
                for (int32_t i = 0; i < nv; i++) {
                    if (row >= fm.num_rows) { seterr("row overflow"); return -1; }
                    if (def[(size_t)i]) {
                        if (is_str) {
                            uint32_t len;
                            if ((size_t)(pc.end - pc.p) < 4) { seterr("string underrun"); return -1; }
                            memcpy(&len, pc.p, 4);
                            pc.p += 4;
                            if ((size_t)(pc.end - pc.p) < len) { seterr("string underrun"); return -1; }
                            ((int32_t*)out)[row] =
                                code_of(std::string((const char*)pc.p, (size_t)len));
                            pc.p += len;
                        } else {
                            if ((size_t)(pc.end - pc.p) < 8) { seterr("page data underrun"); return -1; }
                            memcpy((uint8_t*)out + (size_t)row * 8, pc.p, 8);
                            pc.p += 8;
                        }
                        if (valid) valid[row] = 1;
                    } else {
                        if (is_str) ((int32_t*)out)[row] = 0;
                        else memset((uint8_t*)out + (size_t)row * 8, 0, 8);
                        if (valid) valid[row] = 0;
                    }
                    row++;
                }
            } else {
                seterr("unsupported page encoding");
                return -1;
            }
            remaining -= nv;
        }
    }
    if (is_str) *dict_out = std::move(words);
    return row;
}

}  // namespace bkparquet

/* ---- C ABI (declared in include/bkgpu.h) ---- */
extern "C" {

typedef struct BkParquet BkParquet;

BkParquet* bkparquet_open(const char* path) {
    return (BkParquet*)bkparquet::open_reader(path);
}

int64_t bkparquet_num_rows(const BkParquet* r) {
    return ((const bkparquet::Reader*)r)->meta.num_rows;
}

int bkparquet_num_cols(const BkParquet* r) {
    return (int)((const bkparquet::Reader*)r)->meta.schema.size();
}

/* BkType of a column: 6 INT64, 12 DOUBLE, 13 STRING(dict); <0 unsupported */
int bkparquet_col_type(const BkParquet* r, int col) {
    const auto& s = ((const bkparquet::Reader*)r)->meta.schema;
    if (col < 0 || (size_t)col >= s.size()) return -1;
    if (s[col].physical_type == 2) return 6;
    if (s[col].physical_type == 5) return 12;
    if (s[col].physical_type == 6) return 13;
    return -2;
}

int bkparquet_col_nullable(const BkParquet* r, int col) {
    const auto& s = ((const bkparquet::Reader*)r)->meta.schema;
    if (col < 0 || (size_t)col >= s.size()) return -1;
    return s[col].optional ? 1 : 0;
}

int bkparquet_col_name(const BkParquet* r, int col, char* out, int cap) {
    const auto& s = ((const bkparquet::Reader*)r)->meta.schema;
    if (col < 0 || (size_t)col >= s.size()) return -1;
    return snprintf(out, (size_t)cap, "%s", s[col].name.c_str());
}

/* read a whole column to host buffers (out sized num_rows * 8 bytes; valid
 * sized num_rows or NULL for required columns). Returns rows read, <0 err. */
int64_t bkparquet_read_column(BkParquet* r, int col, void* out, uint8_t* valid) {
    return bkparquet::read_column((bkparquet::Reader*)r, col, out, valid,
                                  nullptr);
}

/* BYTE_ARRAY column -> int32 ORDER-PRESERVING dict codes + the dictionary.
 * *dict_handle receives an opaque handle (free with bkparquet_dict_free);
 * *dict_n its entry count. */
int64_t bkparquet_read_string_column(BkParquet* r, int col, int32_t* codes,
                                     uint8_t* valid, void** dict_handle,
                                     int64_t* dict_n) {
    auto* d = new std::vector<std::string>();
    int64_t got = bkparquet::read_column((bkparquet::Reader*)r, col, codes,
                                         valid, d);
    if (got < 0) { delete d; return got; }
    if (dict_n) *dict_n = (int64_t)d->size();
    if (dict_handle) *dict_handle = d; else delete d;
    return got;
}

int bkparquet_dict_word(void* dict_handle, int64_t code, char* out, int cap) {
    auto* d = (std::vector<std::string>*)dict_handle;
    if (code < 0 || (size_t)code >= d->size()) return -1;
    return snprintf(out, (size_t)cap, "%s", (*d)[(size_t)code].c_str());
}

void bkparquet_dict_free(void* dict_handle) {
    delete (std::vector<std::string>*)dict_handle;
}

const char* bkparquet_last_error(void) { return bkparquet::g_err.c_str(); }

void bkparquet_close(BkParquet* r) { delete (bkparquet::Reader*)r; }

}  /* extern "C" */

/* ---- glue: stage a parquet file straight into an HBM-resident table ----
 * (the cold-data ingestion slot: region picks up a parquet cold file and
 * hands it to the engine, replacing the synthetic generator staging). */
struct BkColSpec2 { int32_t col_type, dist; int64_t p0, p1; int32_t nf, pad; };
extern "C" void* bkgpu_table_create(int ncols, const void* specs, int64_t nrows);
extern "C" int bkgpu_table_upload(void* t, int col, const void* data,
                                  const uint8_t* valid);
extern "C" void bkgpu_table_free(void* t);

extern "C" int bkgpu_table_set_dict(void* t, int col, const char* concat,
                                    const int64_t* offs, int64_t n);

extern "C" void* bkgpu_table_from_parquet(const char* path) {
    BkParquet* r = bkparquet_open(path);
    if (!r) return nullptr;
    int nc = bkparquet_num_cols(r);
    int64_t nrows = bkparquet_num_rows(r);
    std::vector<BkColSpec2> specs((size_t)nc);
    for (int c = 0; c < nc; c++) {
        int t = bkparquet_col_type(r, c);
        if (t < 0) {
            bkparquet::seterr("unsupported column type in parquet schema");
            bkparquet_close(r);
            return nullptr;
        }
        specs[(size_t)c] = {t, 0, 0, 0,
                            bkparquet_col_nullable(r, c) ? 1 : 0, 0};
    }
    void* tab = bkgpu_table_create(nc, specs.data(), nrows);
    if (!tab) { bkparquet_close(r); return nullptr; }
    std::vector<uint8_t> data((size_t)nrows * 8);
    std::vector<uint8_t> valid;
    for (int c = 0; c < nc; c++) {
        uint8_t* vp = nullptr;
        if (bkparquet_col_nullable(r, c)) {
            valid.assign((size_t)nrows, 1);
            vp = valid.data();
        }
        bool ok;
        if (bkparquet_col_type(r, c) == 13) {
            void* dh = nullptr;
            int64_t dn = 0;
            ok = bkparquet_read_string_column(r, c, (int32_t*)data.data(), vp,
                                              &dh, &dn) == nrows;
            if (ok) ok = bkgpu_table_upload(tab, c, data.data(), vp) == 0;
            if (ok && dh) {
                auto* d = (std::vector<std::string>*)dh;
                std::string concat;
                std::vector<int64_t> offs(d->size() + 1, 0);
                for (size_t i = 0; i < d->size(); i++) {
                    concat += (*d)[i];
                    offs[i + 1] = (int64_t)concat.size();
                }
                ok = bkgpu_table_set_dict(tab, c, concat.data(), offs.data(),
                                          (int64_t)d->size()) == 0;
            }
            if (dh) bkparquet_dict_free(dh);
        } else {
            ok = bkparquet_read_column(r, c, data.data(), vp) == nrows &&
                 bkgpu_table_upload(tab, c, data.data(), vp) == 0;
        }
        if (!ok) {
            bkgpu_table_free(tab);
            bkparquet_close(r);
            return nullptr;
        }
    }
    bkparquet_close(r);
    return tab;
}
