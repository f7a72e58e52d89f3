/* bk_cstore.h — decode of the reference's CSTORE on-disk KV format (the
 * per-column RocksDB column-family layout) into columnar batches: the
 * second half of SURVEY §8f.1 (cold-data ingestion), restating
 *   /root/reference/src/engine/table_iterator.cpp:322-600 (open_columns +
 *     get_column merge-join of the primary-key stream with each field's
 *     column stream; a missing column key yields the field default),
 *   /root/reference/include/common/mut_table_key.h:113-208 (key byte
 *     encodings: big-endian sign-flipped ints — the bk_keyenc.h encodings
 *     already pinned against the reference's compiled key_encoder.h),
 *   /root/reference/src/common/table_record.cpp:362-470
 *     encode_field_for_cstore (values: little-endian fixed width; strings
 *     raw bytes; NULL/default fields are NOT written), and
 *   /root/reference/include/common/message_helper.h:291-420 decode_field.
 *
 * Key layouts (RocksDB the library is absent here; the BYTES are what
 * matter — streams arrive as flat [u32 klen][key][u32 vlen][value]
 * sequences in ascending key order, exactly the iterator's view):
 *   row key    = [enc_i64(region)][enc_i64(index_id)][pure_pk]
 *   column key = [enc_i64(region)][enc_i32(table_id)][enc_i32(field_id)]
 *                [pure_pk]            (table_iterator.cpp:355-357 replace)
 *   pure_pk    = MutTableKey-encoded primary key fields (v1: one INT64)
 */
#ifndef BK_CSTORE_H
#define BK_CSTORE_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BkCstoreField {
    int32_t field_id;
    int32_t col_type;       /* BK_INT64 / BK_DOUBLE / BK_STRING */
    const uint8_t* kv;      /* this field's column-CF stream */
    int64_t kv_len;
    /* default_expr_value (field_info.default_expr_value): rows with no
     * column key take it; has_default == 0 means the default is NULL */
    int32_t has_default;
    int64_t def_i;
    double  def_d;
    const char* def_s;
} BkCstoreField;

typedef struct BkCstore BkCstore;  /* decoded columnar batch (host) */

/* Decode the primary stream + nfields column streams. Returns NULL on
 * malformed input (bk_cstore_last_error describes it). */
BkCstore* bk_cstore_decode(int64_t region_id, int64_t table_id,
                           const uint8_t* primary_kv, int64_t primary_len,
                           int nfields, const BkCstoreField* fields);
const char* bk_cstore_last_error(void);
int64_t bk_cstore_nrows(const BkCstore* c);
/* column 0 = the primary key (INT64 decoded from the row keys); columns
 * 1..nfields = the declared fields in order */
const void* bk_cstore_col(const BkCstore* c, int col);   /* i64/f64/i32 */
const uint8_t* bk_cstore_valid(const BkCstore* c, int col); /* NULL: none */
int64_t bk_cstore_nwords(const BkCstore* c, int col);
const char* bk_cstore_word(const BkCstore* c, int col, int64_t code);
void bk_cstore_free(BkCstore* c);

/* decode + upload into an HBM-resident engine table (column order: pk,
 * then fields; BK_STRING fields land as order-preserving dict codes with
 * the word list attached) — the end-to-end KV -> HBM -> GROUP BY path. */
struct BkgTable;
struct BkgTable* bkgpu_table_from_cstore(const BkCstore* c);

#ifdef __cplusplus
}
#endif
#endif /* BK_CSTORE_H */
