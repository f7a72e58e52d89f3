/* bkgpu.h — C-ABI of the MI355X-native OLAP execution engine ("bkgpu").
 *
 * This is the drop-in boundary for baikalStore: the host-side
 * exec nodes (C++ mirror of ExecNode, see include/bk_exec.h) call ONLY this
 * flat C API, exactly as the reference's row nodes hand off to Arrow Acero
 * today (src/runtime/arrow_io_excutor.cpp:265, src/store/region.cpp:2793-2923).
 * No C++ or torch types cross this boundary: plain pointers and sizes.
 *
 * Entry points and the reference interface each replaces:
 *  - bkgpu_table_*        : the columnar batch source standing in for
 *                           RocksdbScanNode's decoded output
 *                           (src/exec/rocksdb_scan_node.cpp:748-800; the
 *                           KV-decode itself is below the boundary)
 *  - bkgpu_filter_agg     : FilterNode::get_next + AggNode::open/process
 *                           (src/exec/filter_node.cpp:736-795,
 *                            src/exec/agg_node.cpp:405-545) fused into one
 *                           GPU pipeline pass
 *  - bkgpu_agg_merge      : the db-side MERGE_AGG combine
 *                           (src/exec/agg_node.cpp:29,539-543) — used by the
 *                           multi-GPU region-set merge (RCCL over xGMI)
 *  - bkgpu_agg_fetch      : AggNode::get_next finalize+emit
 *                           (src/exec/agg_node.cpp:548-573,
 *                            src/expr/agg_fn_call.cpp:927-975)
 *  - bkgpu_sort_topk      : SortNode + TopNSorter (src/exec/sort_node.cpp:278-440,
 *                           include/runtime/topn_sorter.h:32-63)
 *
 * All functions return 0 on success, negative on error (mirrors ExecNode's
 * int-return error convention, include/exec/exec_node.h:140-153);
 * bkgpu_last_error() returns a description.
 */
#ifndef BKGPU_H
#define BKGPU_H

#include <stdint.h>
#include "bk_common.h"

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BkgTable BkgTable;     /* columnar table resident in HBM */
typedef struct BkgAggOut BkgAggOut;   /* aggregation result resident in HBM */

/* ---- device / error ---- */
int         bkgpu_device_count(void);
int         bkgpu_set_device(int dev);
const char* bkgpu_last_error(void);
int         bkgpu_sync(void);

/* ---- table lifecycle ---- */
/* Create a table of nrows rows whose columns follow specs[0..ncols).
 * Storage: BK_INT64 -> int64, BK_DOUBLE -> double, BK_STRING -> int32 dict
 * codes. Columns with null_frac_x1e6 > 0 also get a validity byte array. */
BkgTable* bkgpu_table_create(int ncols, const BkColSpec* specs, int64_t nrows);
/* Fill all columns on device with the deterministic shared generator
 * (bk_datagen.h) for global row ids [row_begin, row_begin+nrows). */
int  bkgpu_table_generate(BkgTable* t, uint64_t seed, int64_t row_begin);
/* Upload host column data (and optional validity bytes) instead. */
int  bkgpu_table_upload(BkgTable* t, int col, const void* data, const uint8_t* valid);
/* Upload ARBITRARY strings into a BK_STRING column: builds the
 * order-preserving dictionary host-side (code order == byte order, the
 * parquet/cstore ingest policy) and uploads int32 codes — the drop-in
 * path for non-dictionary VARCHAR (ExprValue STRING compares /
 * mut_table_key.h:196-208 string keys map onto integer code compares).
 * offs[nrows+1] delimits each row's bytes; NULL rows need valid[r]==0. */
int  bkgpu_table_upload_strings(BkgTable* t, int col, const char* bytes,
                                const int64_t* offs, const uint8_t* valid);
/* literal -> code for string predicates: mode 0 exact (-1 absent),
 * mode 1 lower_bound (s < L <=> code < lower_bound(L), etc.) */
int64_t bkgpu_table_dict_code(const BkgTable* t, int col, const char* word,
                              int64_t wlen, int mode);
int64_t bkgpu_table_nrows(const BkgTable* t);
/* Narrow the physical storage of integer-typed columns (col = -1: all)
 * whose all-rows value range fits 1/2/4 bytes to frame-of-reference
 * deltas — value-preserving (kernels reconstruct exact int64s), cuts the
 * column's HBM traffic 2-8x. Re-generate/upload transparently re-widens. */
int  bkgpu_table_compact(BkgTable* t, int col);
/* Physical bytes/elem of a column after compaction (introspection). */
int  bkgpu_table_col_width(const BkgTable* t, int col);
/* Append a derived BK_STRING column: newcode[r] = remap[oldcode[r]] — the
 * engine-side compilation of a unary string scalar fn (upper/lower/substr,
 * src/expr/internal_functions.cpp via fn_manager.cpp:97-137) on a dict
 * column: the caller transforms the word list, dedups/sorts it (codes stay
 * order-preserving) and passes old->new code map (len ncodes). Returns the
 * new column index, usable in GROUP BY / ORDER BY / MIN/MAX. */
int bkgpu_table_derive_remap(BkgTable* t, int src_col, const int32_t* remap,
                             int64_t ncodes, int64_t new_ncodes);
/* Append a derived EXPRESSION column: dst[r] = eval(prog, row r) — the
 * projection of an arbitrary-depth expression tree (ScalarFnCall::get_value,
 * src/expr/scalar_fn_call.cpp:194-225) compiled to a postfix program (same
 * BkExprOp encoding as BkQuerySpec.prog). out_type is the program's result
 * domain (BK_INT64/BK_DOUBLE, reference arg-cast rule). The new column is a
 * first-class WINDOW fn input / ORDER BY key / out_col. Returns the new
 * column index. */
int bkgpu_table_derive_prog(BkgTable* t, const BkExprOp* prog, int32_t len,
                            int32_t out_type);
int32_t bkgpu_table_col_type(const BkgTable* t, int col);
int32_t bkgpu_table_ncols(const BkgTable* t);
void bkgpu_table_free(BkgTable* t);

/* ---- fused scan+filter+aggregate ---- */
/* Run the SELECT pipeline `WHERE conjuncts GROUP BY group aggs` over rows
 * [row_begin, row_end) of t. expected_groups sizes the hash table (engine
 * retries with a larger table on overflow). Returns NULL on error. */
BkgAggOut* bkgpu_filter_agg(BkgTable* t, const BkQuerySpec* q,
                            int64_t row_begin, int64_t row_end,
                            int64_t expected_groups);

int64_t bkgpu_agg_ngroups(const BkgAggOut* o);
int64_t bkgpu_agg_rows_passed(const BkgAggOut* o);
/* total device time of the aggregate pipeline (HIP events on the launch
 * stream), for bench roofline only */
double  bkgpu_agg_kernel_ms(const BkgAggOut* o);
/* per-kernel breakdown: fills ms[0..n) and 16-byte names; returns n.
 * Fused path: {fused_agg}; partitioned path: {histo, totals, scan, offsets,
 * scatter, part_agg} (pipelined large ranges report one {pipeline} entry);
 * sort-dedup path: {dedup_mat, sort, scan, emit}. */
int     bkgpu_agg_breakdown(const BkgAggOut* o, char* names, double* ms, int cap);

/* ---- partial-aggregate exchange (multi-GPU merge over RCCL) ----
 * Compact wire format of one partial result, an SoA byte blob:
 *   [ flags: u32 * n ][ k0: u64 * n ][ k1: u64 * n ][ states: u64 * n * 2*naggs ]
 * (AVG state = {sum double bits, count}; SUM = {val bits, nonnull count};
 *  MIN/MAX = {order-encoded u64, nonnull count}; COUNT = {count, _}.) */
int64_t bkgpu_agg_export_bytes(const BkgAggOut* o);
/* Write the blob to dst (DEVICE pointer, capacity cap bytes). */
int  bkgpu_agg_export(const BkgAggOut* o, void* dst, int64_t cap);
/* Merge a peer blob (DEVICE pointer, n groups) into o with the reference's
 * AggFnCall::merge semantics (src/expr/agg_fn_call.cpp:781-830). */
int  bkgpu_agg_merge(BkgAggOut* o, const void* blob, int64_t n_groups);

/* ---- hash-partitioned exchange (the MPP repartition the reference's
 * ExchangeSenderNode does over brpc, exchange_sender_node.h:228-235;
 * here it feeds an RCCL all-to-all over xGMI): the partial result's
 * groups split by key hash into `nparts` disjoint wire blobs, every rank
 * merges the blobs of ITS part in parallel — replacing the serialized
 * gather-to-rank-0 merge. The hash is the engine's key_hash, identical
 * on every rank by construction. ---- */
/* per-part group counts (runs compact if needed) */
int  bkgpu_agg_part_counts(const BkgAggOut* o, int nparts, int64_t* counts);
/* write part `part`'s blob (DEVICE dst; layout = the standard wire blob
 * with part_groups groups — pass the count bkgpu_agg_part_counts gave) */
int  bkgpu_agg_export_part(const BkgAggOut* o, int nparts, int part,
                           void* dst, int64_t part_groups);
/* fresh empty result for `q` (merge target for received part blobs) */
BkgAggOut* bkgpu_agg_empty(const BkQuerySpec* q, int64_t expected_groups);

/* SQL LIKE (include/bk_like.h: LikePredicate::like + like_one restated,
 * predicate.h:502-573 / predicate.cpp:509-530) — the host compiles LIKE
 * patterns against dictionary words into BK_OP_IN_BITMAP pushdowns with
 * these. charset: 0 Binary, 1 UTF8, 2 GBK. bkgpu_like_match returns
 * 1/0/-1 (-1 = invalid sequence); bkgpu_like_one applies the reference's
 * GBK->Binary retry and returns 1/0. */
int bkgpu_like_match(const char* target, int64_t tlen, const char* pattern,
                     int64_t plen, int charset, char escape_char);
int bkgpu_like_one(const char* target, int64_t tlen, const char* pattern,
                   int64_t plen, int charset, char escape_char);

/* Level-1 aggregate via SORT-based dedup: radix-sorts the spec-packed group
 * key per passing row and emits one DENSE table slot per unique key — the
 * path for high-cardinality DISTINCT (dedup cardinality ~ rows), where the
 * hash table degrades. Results identical to bkgpu_filter_agg.
 * Requires every group key packed into <= 56 declared group_bits and a row
 * range < 2^32; returns NULL (with bkgpu_last_error) otherwise — callers
 * fall back to bkgpu_filter_agg. */
BkgAggOut* bkgpu_filter_agg_sorted(BkgTable* t, const BkQuerySpec* q,
                                   int64_t row_begin, int64_t row_end);

/* ---- DISTINCT rollup (reference planner rewrite, agg_node.cpp:247-258) ----
 * Fold a level-1 aggregate grouped by (user group keys + distinct col) into
 * the level-2 result grouped by the user keys alone. q2: n_group = level-1
 * n_group - 1 (0 or 1), aggs may be BK_AGG_COUNT_DISTINCT/BK_AGG_SUM_DISTINCT
 * (synthesized from the dedup key; src_idx[a] = -1) or plain aggs merged
 * additively from the level-1 state at src_idx[a]. For multi-GPU, exchange
 * LEVEL-1 blobs (bkgpu_agg_merge dedups (g,d) pairs) before rolling up. */
BkgAggOut* bkgpu_agg_rollup(const BkgAggOut* in, const BkQuerySpec* q2,
                            const int32_t* src_idx, int64_t expected_groups);

/* ---- result fetch (finalize + emit) ----
 * Downloads up to max_groups finalized groups to host arrays (each sized by
 * caller: flags[n], enc[n*BK_MAX_GROUP], out_i/out_d/out_has[naggs*n]).
 * Groups arrive in canonical order (sorted by the reference MutTableKey byte
 * order) iff sorted != 0. Finalization follows agg_fn_call.cpp:927-975. */
int64_t bkgpu_agg_fetch(BkgAggOut* o, int sorted, int64_t max_groups,
                        uint8_t* flags, uint64_t* enc,
                        int64_t* out_i, double* out_d, uint8_t* out_has);
void bkgpu_agg_free(BkgAggOut* o);

/* ---- ORDER BY ... LIMIT top-N ----
 * Select the `limit` smallest rows of [row_begin,row_end) passing q's filter
 * under `order` (INT64/DOUBLE/dict keys, nullable columns ordered per
 * is_null_first; ties broken by arrival index, topn_sorter.h:46-54), write
 * their global row ids in final order to out_rows (host array). Returns
 * count written, negative on error. */
int64_t bkgpu_sort_topk(BkgTable* t, const BkQuerySpec* q,
                        const BkOrderSpec* order, int norder,
                        int64_t row_begin, int64_t row_end,
                        int64_t limit, int64_t* out_rows);
double bkgpu_topk_kernel_ms(void);

/* ---- window functions (WindowNode, window_node.cpp; non-frame mode and
 * ROWS frames — frame_rows=1, bounds f_pre/f_fol rows, negative =
 * UNBOUNDED; RowFrameWindowProcessor semantics for SUM/COUNT/AVG and
 * FIRST/LAST/NTH_VALUE): rows sorted by (partition, order, arrival);
 * outputs fn-major host arrays out_i/out_d/out_null[f*n + i]; out_rowids
 * receives the sorted global row ids. Returns rows produced, <0 error. ---- */
int64_t bkgpu_window(BkgTable* t, const BkQuerySpec* q, int32_t part_col,
                     const BkOrderSpec* order, int norder,
                     const BkWindowFn* fns, int nfns,
                     int32_t frame_rows, int64_t f_pre, int64_t f_fol,
                     int64_t row_begin, int64_t row_end,
                     int64_t* out_rowids, int64_t* out_i, double* out_d,
                     uint8_t* out_null);
/* PARTITION BY multiple columns (window_node.cpp evaluates every
 * partition expr; a new partition starts when any changes). n_part <= 3,
 * n_part + norder <= 4 sort keys. bkgpu_window == n_part <= 1 case. */
int64_t bkgpu_window_multi(BkgTable* t, const BkQuerySpec* q,
                     const int32_t* part_cols, int32_t n_part,
                     const BkOrderSpec* order, int norder,
                     const BkWindowFn* fns, int nfns,
                     int32_t frame_rows, int64_t frame_pre, int64_t frame_fol,
                     int64_t row_begin, int64_t row_end,
                     int64_t* out_rowids, int64_t* out_i, double* out_d,
                     uint8_t* out_null);

/* ---- SELECT without GROUP BY (FilterNode row emission, filter_node.cpp:
 * 736-795): collect up to limit passing global row ids (order unspecified);
 * materialize columns for row ids. Host arrays. ---- */
int64_t bkgpu_filter_collect(BkgTable* t, const BkQuerySpec* q,
                             int64_t row_begin, int64_t row_end,
                             int64_t limit, int64_t* out_rowids_host);
int bkgpu_gather(BkgTable* t, int col, const int64_t* rowids_host, int64_t n,
                 int64_t* out_i, double* out_d, uint8_t* out_null);
/* upload an opaque byte buffer (e.g. a BK_OP_IN_BITMAP dict-code accept
 * bitmap) to device; free with bkgpu_free_ptr */
void* bkgpu_upload_bytes(const void* data, int64_t n);
void  bkgpu_free_ptr(void* p);
/* release pooled device buffers */
void bkgpu_pool_trim(void);

/* ---- cold columnar ingestion: minimal parquet reader ----
 * The reference stages OLAP cold data as parquet (src/column/file_manager.h:
 * 252-334, parquet_writer.h:119). This reads v1 uncompressed PLAIN files
 * (INT64/DOUBLE, optional columns) without Arrow; unsupported features are
 * rejected with bkparquet_last_error(), never misread. */
typedef struct BkParquet BkParquet;
BkParquet* bkparquet_open(const char* path);
int64_t    bkparquet_num_rows(const BkParquet* r);
int        bkparquet_num_cols(const BkParquet* r);
int        bkparquet_col_type(const BkParquet* r, int col);  /* BkType or <0 */
int        bkparquet_col_nullable(const BkParquet* r, int col);
int        bkparquet_col_name(const BkParquet* r, int col, char* out, int cap);
int64_t    bkparquet_read_column(BkParquet* r, int col, void* out, uint8_t* valid);
/* BYTE_ARRAY column -> int32 ORDER-PRESERVING dict codes (code order ==
 * byte order of the words: the invariant string MIN/MAX/ORDER BY rely on) */
int64_t    bkparquet_read_string_column(BkParquet* r, int col, int32_t* codes,
                                        uint8_t* valid, void** dict_handle,
                                        int64_t* dict_n);
int        bkparquet_dict_word(void* dict_handle, int64_t code, char* out, int cap);
void       bkparquet_dict_free(void* dict_handle);
const char* bkparquet_last_error(void);
void       bkparquet_close(BkParquet* r);
/* parquet file -> HBM-resident table (create + upload all columns; string
 * columns keep their dictionary on the table, bkgpu_table_dict_word) */
BkgTable*  bkgpu_table_from_parquet(const char* path);
int        bkgpu_table_set_dict(BkgTable* t, int col, const char* concat,
                                const int64_t* offs, int64_t n);
int        bkgpu_table_dict_word(const BkgTable* t, int col, int64_t code,
                                 char* out, int cap);

#ifdef __cplusplus
}
#endif
#endif /* BKGPU_H */
