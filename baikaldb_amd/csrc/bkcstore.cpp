// bkcstore.cpp — CSTORE per-column KV decode (include/bk_cstore.h).
// Restates the reference's column-store read path without RocksDB: the
// caller supplies each column family's KV pairs as a flat ascending
// stream (the iterator's view), and decode performs exactly the
// merge-join TableIterator::get_column runs per field
// (/root/reference/src/engine/table_iterator.cpp:525-597): walk the
// primary-key stream, advance the field's column stream; key match ->
// decode the little-endian value (message_helper.h:291-420), primary key
// ahead -> the row takes field_info.default_expr_value (NULL when the
// default is null — the write side never stores NULL/default fields,
// table_record.cpp:362-470).
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <map>
#include <string>
#include <vector>

#include "../../include/bk_common.h"
#include "../../include/bk_keyenc.h"
#include "../../include/bk_cstore.h"
#include "../../include/bkgpu.h"

namespace {

thread_local char g_cerr[256] = "";
void cerr_set(const char* m) { snprintf(g_cerr, sizeof g_cerr, "%s", m); }

struct Kv {
    std::string key, val;
};

/* parse a [u32 klen][key][u32 vlen][val] stream */
bool parse_kv(const uint8_t* p, int64_t n, std::vector<Kv>& out) {
    int64_t i = 0;
    while (i < n) {
        if (i + 4 > n) return false;
        uint32_t kl;
        memcpy(&kl, p + i, 4);
        i += 4;
        if (i + kl + 4 > n) return false;
        Kv kv;
        kv.key.assign((const char*)p + i, kl);
        i += kl;
        uint32_t vl;
        memcpy(&vl, p + i, 4);
        i += 4;
        if (i + vl > n) return false;
        kv.val.assign((const char*)p + i, vl);
        i += vl;
        out.push_back(std::move(kv));
    }
    return true;
}

void put_be64(std::string& s, uint64_t v) {
    uint64_t b = bk_bswap64(v);
    s.append((const char*)&b, 8);
}
void put_be32(std::string& s, uint32_t v) {
    uint32_t b = bk_bswap32(v);
    s.append((const char*)&b, 4);
}

}  // namespace

extern "C" {

struct BkCstore {
    int64_t nrows = 0;
    int ncols = 0;                       /* pk + nfields */
    std::vector<int32_t> types;
    std::vector<std::vector<int64_t>> ci;
    std::vector<std::vector<double>> cd;
    std::vector<std::vector<int32_t>> codes;
    std::vector<std::vector<uint8_t>> valid;   /* empty = all valid */
    std::vector<std::vector<std::string>> words;
};

const char* bk_cstore_last_error(void) { return g_cerr; }

BkCstore* bk_cstore_decode(int64_t region_id, int64_t table_id,
                           const uint8_t* primary_kv, int64_t primary_len,
                           int nfields, const BkCstoreField* fields) {
    /* prefixes: row keys [enc64(region)][enc64(index_id == table_id)];
     * column keys [enc64(region)][enc32(table_id)][enc32(field_id)]
     * (MutTableKey::replace_i32 at offsets 8 and 12,
     * table_iterator.cpp:355-357) */
    std::string row_prefix;
    put_be64(row_prefix, bk_enc_i64(region_id));
    put_be64(row_prefix, bk_enc_i64(table_id));
    std::vector<Kv> prim;
    if (!parse_kv(primary_kv, primary_len, prim)) {
        cerr_set("malformed primary stream");
        return nullptr;
    }
    BkCstore* c = new BkCstore();
    c->ncols = nfields + 1;
    c->types.assign(c->ncols, BK_INT64);
    c->ci.resize(c->ncols);
    c->cd.resize(c->ncols);
    c->codes.resize(c->ncols);
    c->valid.resize(c->ncols);
    c->words.resize(c->ncols);
    /* pass 1: primary stream -> pk column + pure_pk list */
    std::vector<std::string> pure_pk;
    for (auto& kv : prim) {
        if (kv.key.size() < row_prefix.size() + 8 ||
            memcmp(kv.key.data(), row_prefix.data(), row_prefix.size())) {
            cerr_set("row key prefix mismatch");
            delete c;
            return nullptr;
        }
        const char* pk = kv.key.data() + row_prefix.size();
        uint64_t be;
        memcpy(&be, pk, 8);
        c->ci[0].push_back(bk_dec_i64(bk_bswap64(be)));
        pure_pk.push_back(kv.key.substr(row_prefix.size()));
    }
    c->nrows = (int64_t)pure_pk.size();
    /* pass 2: per field, merge-join its column stream with pure_pk
     * (both ascending, get_column's loop) */
    for (int f = 0; f < nfields; f++) {
        const BkCstoreField& fi = fields[f];
        int col = f + 1;
        c->types[col] = fi.col_type;
        std::string prefix;
        put_be64(prefix, bk_enc_i64(region_id));
        put_be32(prefix, bk_enc_i32((int32_t)table_id));
        put_be32(prefix, bk_enc_i32(fi.field_id));
        std::vector<Kv> ckv;
        if (!parse_kv(fi.kv, fi.kv_len, ckv)) {
            cerr_set("malformed column stream");
            delete c;
            return nullptr;
        }
        std::vector<uint8_t> va(c->nrows, 1);
        bool any_null = false;
        std::map<std::string, int32_t> dict;
        std::vector<std::string> raw_words(c->nrows);
        size_t it = 0;
        for (int64_t r = 0; r < c->nrows; r++) {
            /* advance past keys < this row's pure_pk (the reference's
             * while loop: cmp > 0 -> Next) */
            const std::string want = prefix + pure_pk[r];
            while (it < ckv.size() && ckv[it].key < want) it++;
            bool hit = it < ckv.size() && ckv[it].key == want;
            const std::string* v = hit ? &ckv[it].val : nullptr;
            if (hit) it++;
            if (!v) {   /* default_expr_value */
                if (!fi.has_default) {
                    va[r] = 0;
                    any_null = true;
                    c->ci[col].push_back(0);
                    c->cd[col].push_back(0);
                    raw_words[r].clear();
                    continue;
                }
                if (fi.col_type == BK_DOUBLE) {
                    c->cd[col].push_back(fi.def_d);
                    c->ci[col].push_back(0);
                } else if (fi.col_type == BK_STRING) {
                    raw_words[r] = fi.def_s ? fi.def_s : "";
                    c->ci[col].push_back(0);
                    c->cd[col].push_back(0);
                } else {
                    c->ci[col].push_back(fi.def_i);
                    c->cd[col].push_back(0);
                }
                continue;
            }
            /* decode_field (message_helper.h): little-endian fixed width;
             * strings raw */
            if (fi.col_type == BK_DOUBLE) {
                if (v->size() < 8) {
                    cerr_set("double value underrun");
                    delete c;
                    return nullptr;
                }
                double d;
                memcpy(&d, v->data(), 8);
                c->cd[col].push_back(d);
                c->ci[col].push_back(0);
            } else if (fi.col_type == BK_STRING) {
                raw_words[r] = *v;
                c->ci[col].push_back(0);
                c->cd[col].push_back(0);
            } else {
                if (v->size() < 8) {
                    cerr_set("int64 value underrun");
                    delete c;
                    return nullptr;
                }
                int64_t x;
                memcpy(&x, v->data(), 8);   /* to_little_endian == identity */
                c->ci[col].push_back(x);
                c->cd[col].push_back(0);
            }
        }
        if (fi.col_type == BK_STRING) {
            /* order-preserving dict codes (same policy as the parquet
             * ingest: code order == byte order of the words) */
            for (int64_t r = 0; r < c->nrows; r++)
                if (va[r]) dict.emplace(raw_words[r], 0);
            int32_t next = 0;
            for (auto& kvp : dict) kvp.second = next++;
            c->codes[col].resize(c->nrows, 0);
            for (int64_t r = 0; r < c->nrows; r++)
                c->codes[col][r] = va[r] ? dict[raw_words[r]] : 0;
            c->words[col].reserve(dict.size());
            for (auto& kvp : dict) c->words[col].push_back(kvp.first);
        }
        if (any_null) c->valid[col] = std::move(va);
    }
    return c;
}

int64_t bk_cstore_nrows(const BkCstore* c) { return c->nrows; }
const void* bk_cstore_col(const BkCstore* c, int col) {
    if (col < 0 || col >= c->ncols) return nullptr;
    if (c->types[col] == BK_DOUBLE) return c->cd[col].data();
    if (c->types[col] == BK_STRING) return c->codes[col].data();
    return c->ci[col].data();
}
const uint8_t* bk_cstore_valid(const BkCstore* c, int col) {
    if (col < 0 || col >= c->ncols || c->valid[col].empty()) return nullptr;
    return c->valid[col].data();
}
int64_t bk_cstore_nwords(const BkCstore* c, int col) {
    return (col < 0 || col >= c->ncols) ? 0 : (int64_t)c->words[col].size();
}
const char* bk_cstore_word(const BkCstore* c, int col, int64_t code) {
    if (col < 0 || col >= c->ncols || code < 0 ||
        code >= (int64_t)c->words[col].size())
        return nullptr;
    return c->words[col][code].c_str();
}
void bk_cstore_free(BkCstore* c) { delete c; }

struct BkgTable* bkgpu_table_from_cstore(const BkCstore* c) {
    std::vector<BkColSpec> specs(c->ncols);
    for (int i = 0; i < c->ncols; i++) {
        memset(&specs[i], 0, sizeof(BkColSpec));
        specs[i].col_type = c->types[i];
        specs[i].null_frac_x1e6 = c->valid[i].empty() ? 0 : 1;
    }
    BkgTable* t = bkgpu_table_create(c->ncols, specs.data(), c->nrows);
    if (!t) return nullptr;
    for (int i = 0; i < c->ncols; i++) {
        const uint8_t* va = c->valid[i].empty() ? nullptr
                                                : c->valid[i].data();
        const void* data = bk_cstore_col(c, i);
        if (bkgpu_table_upload(t, i, data, va) != 0) {
            bkgpu_table_free(t);
            return nullptr;
        }
        if (c->types[i] == BK_STRING) {
            std::string concat;
            std::vector<int64_t> offs;
            offs.push_back(0);
            for (auto& w : c->words[i]) {
                concat += w;
                offs.push_back((int64_t)concat.size());
            }
            if (bkgpu_table_set_dict(t, i, concat.c_str(), offs.data(),
                                     (int64_t)c->words[i].size()) != 0) {
                bkgpu_table_free(t);
                return nullptr;
            }
        }
    }
    return t;
}

}  /* extern "C" */
