// bkarrow.cpp — from-scratch Arrow IPC (stream-format) serialization of a
// columnar result batch: the byte format the reference store emits for
// vectorized results (/root/reference/src/store/region.cpp:2905-2918 —
// arrow::ipc::SerializeSchema / SerializeRecordBatch into
// response.extra_res) and the MPP exchange ships
// (exchange_sender_node.h:100-132). No Arrow library is linked (none
// exists in this image): the Message/Schema/RecordBatch flatbuffers and
// the IPC framing are written directly, the same way bkparquet.cpp reads
// parquet without Arrow. Validated byte-for-byte by pyarrow round-trip in
// tests/test_arrow_ipc.py.
//
// Format references (public specs, restated):
//  - flatbuffers wire format: root u32 offset; tables = i32 vtable soffset
//    + fields; vtable = [u16 vt_bytes][u16 table_bytes][u16 field_offs...];
//    vectors = [u32 len][elems]; strings = [u32 len][bytes][NUL]; all
//    offsets point to higher addresses (the builder writes back-to-front).
//  - Arrow Message.fbs/Schema.fbs ids: Message{version:0, header_type:1,
//    header:2, bodyLength:3}; header union {Schema=1, RecordBatch=3};
//    Schema{endianness:0, fields:1}; Field{name:0, nullable:1, type_type:2,
//    type:3, dictionary:4, children:5}; Type union {Int=2, FloatingPoint=3,
//    Utf8=5}; Int{bitWidth:0, is_signed:1}; FloatingPoint{precision:0,
//    DOUBLE=2}; RecordBatch{length:0, nodes:1, buffers:2}; structs
//    FieldNode{length:i64, null_count:i64}, Buffer{offset:i64, length:i64}.
//  - IPC stream framing: [0xFFFFFFFF][i32 meta_len][flatbuffer pad8][body];
//    end-of-stream = [0xFFFFFFFF][0x00000000]. MetadataVersion V5 = 4.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <string>
#include <vector>

#include "../../include/bk_common.h"
#include "../../include/bk_arrow.h"

namespace {

/* ---- minimal back-to-front flatbuffer builder ----
 * Positions ("pos") are byte distances from the END of the final buffer;
 * the final size is padded to 8, so pos % 8 == 0 implies the final address
 * is 8-aligned. A reference stored at the 4 bytes ending at pos P holds
 * P - target_pos (forward offset in address space). */
struct FB {
    std::vector<uint8_t> buf;
    size_t head;
    explicit FB(size_t cap = 1 << 16) : buf(cap), head(cap) {}
    size_t pos() const { return buf.size() - head; }
    void grow(size_t need) {
        if (head >= need) return;
        size_t old = buf.size(), add = old + need;
        std::vector<uint8_t> nb(old + add);
        memcpy(&nb[add + head - 0], &buf[head], old - head);
        /* shift: data occupied [head, old) -> now [add+head, add+old) */
        buf.swap(nb);
        head += add;
    }
    void push(const void* p, size_t n) {
        grow(n);
        head -= n;
        memcpy(&buf[head], p, n);
    }
    void pad_to(size_t align) {
        while (pos() % align) {
            grow(1);
            buf[--head] = 0;
        }
    }
    template <typename T>
    void scalar(T v) { push(&v, sizeof v); }
    /* u32 forward reference to an object at `target` pos */
    void ref(size_t target) {
        uint32_t off = (uint32_t)(pos() + 4 - target);
        scalar(off);
    }
    /* padding in a back-to-front builder must be pushed BEFORE an object
     * (it then sits AFTER it in address space, between objects): pre-pad
     * until pos + upcoming bytes hits the target residue */
    void prepad(size_t upcoming, size_t align, size_t residue = 0) {
        while ((pos() + upcoming) % align != residue) {
            uint8_t z = 0;
            push(&z, 1);
        }
    }
    size_t string(const char* s, size_t n) {
        prepad(n + 1 + 4, 4);                        /* len lands 4-aligned */
        uint8_t z = 0;
        push(&z, 1);                                 /* NUL */
        push(s, n);
        scalar((uint32_t)n);
        return pos();
    }
    /* vector of u32 refs to tables */
    size_t ref_vector(const std::vector<size_t>& targets) {
        prepad(4 * targets.size() + 4, 4);
        for (size_t i = targets.size(); i-- > 0;) ref(targets[i]);
        scalar((uint32_t)targets.size());
        return pos();
    }
    /* vector of 16-byte structs (two i64s each): the u32 count 4-aligned
     * with the first element (count addr + 4) 8-aligned -> vector pos
     * ≡ 4 (mod 8) */
    size_t struct16_vector(const std::vector<int64_t>& vals) {
        size_t n = vals.size() / 2;
        prepad(vals.size() * 8 + 4, 8, 4);
        push(vals.data(), vals.size() * 8);
        scalar((uint32_t)n);
        return pos();
    }
};

/* one table field: either an inline scalar or a forward reference */
struct TField {
    int id;
    int size;          /* 1, 2, 4, 8 */
    int align;         /* == size for scalars, 4 for refs */
    bool is_ref;
    uint64_t val;      /* scalar bits or target pos */
};

/* build a table from present fields (any order); returns table pos */
size_t table(FB& fb, const std::vector<TField>& fields) {
    int maxid = -1;
    for (auto& f : fields) maxid = f.id > maxid ? f.id : maxid;
    /* layout: [soffset i32][fields by descending size for natural align] */
    std::vector<TField> lay = fields;
    for (size_t i = 0; i < lay.size(); i++)          /* stable by size desc */
        for (size_t j = i + 1; j < lay.size(); j++)
            if (lay[j].size > lay[i].size) std::swap(lay[i], lay[j]);
    std::vector<uint16_t> voff(maxid + 1, 0);
    int off = 4, max_align = 4;
    for (auto& f : lay) {
        off = (off + f.size - 1) & ~(f.size - 1);
        voff[f.id] = (uint16_t)off;
        off += f.size;
        if (!f.is_ref && f.size > max_align) max_align = f.size;
    }
    int tbytes = off;
    std::vector<uint8_t> blob(tbytes, 0);
    std::vector<std::pair<int, size_t>> refs;        /* local off -> target */
    for (auto& f : lay) {
        if (f.is_ref) refs.push_back({voff[f.id], f.val});
        else memcpy(&blob[voff[f.id]], &f.val, f.size);
    }
    /* alignment: pos of table start must be ≡ 0 mod 4, and for any 8-byte
     * scalar at local off o: (table_pos - o) % 8 == 0. All 8-byte scalars
     * share o = 8-aligned slots... layout puts them first at off 8? With
     * descending-size layout the first 8-byte field lands at off 8 (after
     * soffset+pad), subsequent at 16, ... all ≡ 0 mod 8, so require
     * table_pos % 8 == 0 when max_align == 8. */
    size_t pre = fb.pos() + tbytes;
    size_t need = max_align;
    while ((fb.pos() + tbytes) % need) {
        uint8_t z = 0;
        fb.push(&z, 1);
    }
    (void)pre;
    fb.push(blob.data(), tbytes);
    size_t tpos = fb.pos();
    /* patch refs: field at local off o sits at pos tpos - o; stored u32 =
     * (tpos - o) - target */
    for (auto& r : refs) {
        uint32_t v = (uint32_t)((tpos - r.first) - r.second);
        memcpy(&fb.buf[fb.head + r.first], &v, 4);
    }
    /* vtable: [u16 vt_bytes][u16 table_bytes][u16 offs...] */
    std::vector<uint16_t> vt(2 + maxid + 1);
    vt[0] = (uint16_t)(vt.size() * 2);
    vt[1] = (uint16_t)tbytes;
    for (int i = 0; i <= maxid; i++) vt[2 + i] = voff[i];
    fb.prepad(vt.size() * 2, 2);
    fb.push(vt.data(), vt.size() * 2);
    size_t vpos = fb.pos();
    /* soffset at table start: table_addr - vtable_addr = vpos - tpos */
    int32_t so = (int32_t)(vpos - tpos);
    memcpy(&fb.buf[fb.head + (vpos - tpos)], &so, 4);
    return tpos;
}

TField scal16(int id, uint16_t v) { return {id, 2, 2, false, v}; }
TField scal8b(int id, uint8_t v) { return {id, 1, 1, false, v}; }
TField scal32(int id, uint32_t v) { return {id, 4, 4, false, v}; }
TField scal64(int id, uint64_t v) { return {id, 8, 8, false, v}; }
TField fref(int id, size_t target) { return {id, 4, 4, true, target}; }

/* finish: root ref + pad front so total is 8-aligned (pos-space alignment
 * then holds in address space) */
std::vector<uint8_t> finish(FB& fb, size_t root) {
    fb.prepad(4, 8);            /* root u32 at buffer START, total 8-aligned */
    fb.ref(root);
    return std::vector<uint8_t>(fb.buf.begin() + fb.head, fb.buf.end());
}

/* ---- IPC framing ---- */
void frame(std::string& out, const std::vector<uint8_t>& meta,
           const std::string& body) {
    uint32_t cont = 0xFFFFFFFFu;
    size_t mlen = (meta.size() + 7) & ~7ull;
    /* metadata length includes its own padding; the 8-byte prefix
     * (continuation + length) keeps the flatbuffer 8-aligned */
    int32_t len = (int32_t)mlen;
    out.append((const char*)&cont, 4);
    out.append((const char*)&len, 4);
    out.append((const char*)meta.data(), meta.size());
    out.append(mlen - meta.size(), '\0');
    out.append(body);
}

}  // namespace

namespace {

size_t build_field(FB& fb, const char* name, const BkArrowCol& c) {
    size_t tname = fb.string(name, strlen(name));
    size_t ttype;
    uint8_t ttag;
    if (c.col_type == BK_DOUBLE) {
        ttype = table(fb, {scal16(0, 2)});                /* DOUBLE */
        ttag = 3;                                         /* FloatingPoint */
    } else if (c.col_type == BK_STRING) {
        ttype = table(fb, {});                            /* Utf8 {} */
        ttag = 5;
    } else {
        ttype = table(fb, {scal32(0, 64), scal8b(1, 1)}); /* Int64 signed */
        ttag = 2;
    }
    size_t children = fb.ref_vector({});
    return table(fb, {fref(0, tname), scal8b(1, c.valid != nullptr ? 1 : 0),
                      scal8b(2, ttag), fref(3, ttype), fref(5, children)});
}

void append_body(std::string& body, std::vector<int64_t>& bufs,
                 const void* p, size_t n) {
    bufs.push_back((int64_t)body.size());
    bufs.push_back((int64_t)n);
    body.append((const char*)p, n);
    body.append((8 - body.size() % 8) % 8, '\0');
}

}  // namespace

extern "C" {

/* Serialize the schema message alone (the reference's
 * arrow::ipc::SerializeSchema half, region.cpp:2905: the store returns
 * schema and rows in separate response fields). Caller frees *out with
 * free(). */
int bk_arrow_schema(int ncols, const BkArrowCol* cols,
                    const char* const* names, void** out, int64_t* out_len) {
    FB fb;
    std::vector<size_t> fields(ncols);
    for (int i = ncols - 1; i >= 0; i--)
        fields[i] = build_field(fb, names[i], cols[i]);
    size_t fvec = fb.ref_vector(fields);
    size_t schema = table(fb, {scal16(0, 0), fref(1, fvec)});
    size_t msg = table(fb, {scal16(0, 4),            /* V5 */
                            scal8b(1, 1),            /* header: Schema */
                            fref(2, schema), scal64(3, 0)});
    std::vector<uint8_t> meta = finish(fb, msg);
    std::string s;
    frame(s, meta, std::string());
    *out = malloc(s.size());
    memcpy(*out, s.data(), s.size());
    *out_len = (int64_t)s.size();
    return 0;
}

/* Serialize one record batch message (SerializeRecordBatch half).
 * BK_STRING columns emit utf8 offsets+data from the dict words. */
int bk_arrow_batch(int ncols, const BkArrowCol* cols, int64_t nrows,
                   void** out, int64_t* out_len) {
    std::string body;
    std::vector<int64_t> bufs;        /* pairs (offset, length) */
    std::vector<int64_t> nodes;       /* pairs (length, null_count) */
    for (int i = 0; i < ncols; i++) {
        const BkArrowCol& c = cols[i];
        int64_t nulls = 0;
        if (c.valid) {
            std::vector<uint8_t> bits((nrows + 7) / 8, 0);
            for (int64_t r = 0; r < nrows; r++) {
                if (c.valid[r]) bits[r >> 3] |= 1u << (r & 7);
                else nulls++;
            }
            append_body(body, bufs, bits.data(), bits.size());
        } else {
            bufs.push_back((int64_t)body.size());
            bufs.push_back(0);
        }
        nodes.push_back(nrows);
        nodes.push_back(nulls);
        if (c.col_type == BK_STRING) {
            const int32_t* codes = (const int32_t*)c.data;
            std::vector<int32_t> offs(nrows + 1, 0);
            std::string chars;
            for (int64_t r = 0; r < nrows; r++) {
                if (!c.valid || c.valid[r]) {
                    int32_t code = codes[r];
                    if (code >= 0 && code < c.nwords)
                        chars.append(c.words[code]);
                }
                offs[r + 1] = (int32_t)chars.size();
            }
            append_body(body, bufs, offs.data(), offs.size() * 4);
            append_body(body, bufs, chars.data(), chars.size());
        } else {
            append_body(body, bufs, c.data, (size_t)nrows * 8);
        }
    }
    FB fb;
    size_t nvec = fb.struct16_vector(nodes);
    size_t bvec = fb.struct16_vector(bufs);
    size_t rb = table(fb, {scal64(0, (uint64_t)nrows), fref(1, nvec),
                           fref(2, bvec)});
    size_t msg = table(fb, {scal16(0, 4), scal8b(1, 3),   /* RecordBatch */
                            fref(2, rb),
                            scal64(3, (uint64_t)body.size())});
    std::vector<uint8_t> meta = finish(fb, msg);
    std::string s;
    frame(s, meta, body);
    *out = malloc(s.size());
    memcpy(*out, s.data(), s.size());
    *out_len = (int64_t)s.size();
    return 0;
}

/* Full IPC stream = schema message + batch message + end-of-stream marker
 * — exactly the concatenation a consumer of region.cpp:2905-2918's two
 * response fields reconstructs; pyarrow.ipc.open_stream reads it. */
int bk_arrow_ipc_stream(int ncols, const BkArrowCol* cols, int64_t nrows,
                        const char* const* names, void** out,
                        int64_t* out_len) {
    void *s = nullptr, *b = nullptr;
    int64_t sn = 0, bn = 0;
    if (bk_arrow_schema(ncols, cols, names, &s, &sn) != 0) return -1;
    if (bk_arrow_batch(ncols, cols, nrows, &b, &bn) != 0) {
        free(s);
        return -1;
    }
    uint32_t eos[2] = {0xFFFFFFFFu, 0};
    int64_t total = sn + bn + 8;
    char* o = (char*)malloc(total);
    memcpy(o, s, sn);
    memcpy(o + sn, b, bn);
    memcpy(o + sn + bn, eos, 8);
    free(s);
    free(b);
    *out = o;
    *out_len = total;
    return 0;
}

void bk_arrow_free(void* p) { free(p); }

}  /* extern "C" */
