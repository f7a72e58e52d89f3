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

/* RLE/bit-packed hybrid definition levels, bit width 1 (max_def_level 1).
 * Layout (v1 page): i32 LE byte length, then runs:
 *   header = varint; header & 1 ? bit-packed group of (header>>1)*8 values
 *                                : RLE run of (header>>1) copies of 1 value. */
static bool read_def_levels(Cursor& c, int32_t nvals, uint8_t* def) {
    if ((size_t)(c.end - c.p) < 4) return false;
    uint32_t len;
    memcpy(&len, c.p, 4);
    c.p += 4;
    const uint8_t* rend = c.p + len;
    if (rend > c.end) return false;
    int32_t i = 0;
    while (i < nvals && c.p < rend) {
        uint64_t h = c.varint();
        if (!c.ok) return false;
        if (h & 1) {              /* bit-packed: (h>>1) groups of 8, width 1 */
            uint64_t groups = h >> 1;
            for (uint64_t g = 0; g < groups && i < nvals; g++) {
                if (c.p >= rend) return false;
                uint8_t byte = *c.p++;
                for (int b = 0; b < 8 && i < nvals; b++)
                    def[i++] = (byte >> b) & 1;
            }
        } else {                  /* RLE: h>>1 copies of one width-1 value */
            uint64_t run = h >> 1;
            if (c.p >= rend) return false;
            uint8_t v = *c.p++ & 1;
            for (uint64_t r = 0; r < run && i < nvals; r++) def[i++] = v;
        }
    }
    c.p = rend;
    return i == nvals;
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

/* read one leaf column across all row groups into host buffers.
 * out: int64/double array of meta.num_rows; valid: per-row 1/0 (may be null
 * for required columns). Returns rows read, < 0 on error. */
static int64_t read_column(Reader* r, int col, void* out, uint8_t* valid) {
    const FileMeta& fm = r->meta;
    if (col < 0 || (size_t)col >= fm.schema.size()) { seterr("bad column"); return -1; }
    const SchemaCol& sc = fm.schema[col];
    if (sc.physical_type != 2 && sc.physical_type != 5) {
        seterr("unsupported physical type (need INT64 or DOUBLE): col " +
               sc.name);
        return -1;
    }
    int64_t row = 0;
    std::vector<uint8_t> def;
    for (const RowGroup& rg : fm.groups) {
        if ((size_t)col >= rg.cols.size()) { seterr("row group missing column"); return -1; }
        const ColChunk& cc = rg.cols[col];
        if (cc.codec != 0) { seterr("compressed parquet unsupported (codec != UNCOMPRESSED)"); return -1; }
        if (cc.dict_page_offset >= 0) { seterr("dictionary-encoded parquet unsupported (write with use_dictionary=False)"); return -1; }
        int64_t remaining = cc.num_values;
        int64_t off = cc.data_page_offset;
        while (remaining > 0) {
            if (off < 0 || (size_t)off >= r->buf.size()) { seterr("bad page offset"); return -1; }
            Cursor c{r->buf.data() + off, r->buf.data() + r->buf.size()};
            PageHeader ph;
            if (!parse_page_header(c, ph)) { seterr("page header parse failed"); return -1; }
            const uint8_t* data = c.p;
            off = (int64_t)(data - r->buf.data()) + ph.compressed_size;
            if (ph.type != 0) { seterr("unsupported page type (v2 pages / dict page)"); return -1; }
            if (ph.encoding != 0) { seterr("unsupported page encoding (need PLAIN)"); return -1; }
            Cursor pc{data, data + ph.uncompressed_size};
            int32_t nv = ph.num_values;
            def.assign((size_t)nv, 1);
            if (sc.optional) {
                if (ph.def_encoding != 3) { seterr("def levels must be RLE"); return -1; }
                if (!read_def_levels(pc, nv, def.data())) { seterr("def level decode failed"); return -1; }
            }
            size_t esz = 8;
            for (int32_t i = 0; i < nv; i++) {
                if (row >= fm.num_rows) { seterr("row overflow"); return -1; }
                if (def[i]) {
                    if ((size_t)(pc.end - pc.p) < esz) { seterr("page data underrun"); return -1; }
                    memcpy((uint8_t*)out + (size_t)row * esz, pc.p, esz);
                    pc.p += esz;
                    if (valid) valid[row] = 1;
                } else {
                    memset((uint8_t*)out + (size_t)row * esz, 0, esz);
                    if (valid) valid[row] = 0;
                }
                row++;
            }
            remaining -= nv;
        }
    }
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

/* BkType of a column: 6 (BK_INT64) or 12 (BK_DOUBLE); <0 unsupported */
int bkparquet_col_type(const BkParquet* r, int col) {
    const auto& s = ((const bkparquet::Reader*)r)->meta.schema;
    if (col < 0 || (size_t)col >= s.size()) return -1;
    if (s[col].physical_type == 2) return 6;
    if (s[col].physical_type == 5) return 12;
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
    return bkparquet::read_column((bkparquet::Reader*)r, col, out, valid);
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
        if (bkparquet_read_column(r, c, data.data(), vp) != nrows ||
            bkgpu_table_upload(tab, c, data.data(), vp) != 0) {
            bkgpu_table_free(tab);
            bkparquet_close(r);
            return nullptr;
        }
    }
    bkparquet_close(r);
    return tab;
}
