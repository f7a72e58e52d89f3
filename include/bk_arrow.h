/* bk_arrow.h — from-scratch Arrow IPC (stream format) serialization of a
 * columnar result batch: the vectorized-result bytes the reference store
 * returns to the frontend (/root/reference/src/store/region.cpp:2905-2918,
 * arrow::ipc::SerializeSchema / SerializeRecordBatch into
 * response.extra_res) and the MPP exchange ships over brpc
 * (exchange_sender_node.h:100-132). Implemented without an Arrow library
 * (baikaldb_amd/csrc/bkarrow.cpp writes the Message/Schema/RecordBatch
 * flatbuffers and the framing directly); validated by pyarrow round-trip
 * in tests/test_arrow_ipc.py. */
#ifndef BK_ARROW_H
#define BK_ARROW_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BkArrowCol {
    int32_t col_type;            /* BK_INT64 / BK_DOUBLE / BK_STRING */
    const void* data;            /* i64 / f64 / i32 dict codes (host) */
    const uint8_t* valid;        /* byte-per-row validity or NULL */
    const char* const* words;    /* BK_STRING: words[code], NUL-terminated */
    int64_t nwords;
} BkArrowCol;

/* Schema message alone (the SerializeSchema half — the store returns
 * schema and rows in separate response fields). *out is malloc'd; free
 * with bk_arrow_free. */
int bk_arrow_schema(int ncols, const BkArrowCol* cols,
                    const char* const* names, void** out, int64_t* out_len);
/* One record batch message (the SerializeRecordBatch half). BK_STRING
 * columns emit utf8 offsets+data resolved through the dictionary. */
int bk_arrow_batch(int ncols, const BkArrowCol* cols, int64_t nrows,
                   void** out, int64_t* out_len);
/* Full IPC stream: schema + batch + end-of-stream marker — what a consumer
 * of the two response fields reconstructs; pyarrow.ipc.open_stream reads
 * it directly. */
int bk_arrow_ipc_stream(int ncols, const BkArrowCol* cols, int64_t nrows,
                        const char* const* names, void** out,
                        int64_t* out_len);
void bk_arrow_free(void* p);

#ifdef __cplusplus
}
#endif
#endif /* BK_ARROW_H */
