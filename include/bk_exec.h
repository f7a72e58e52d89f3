/* bk_exec.h — C-ABI of the C++ host execution layer ("bkexec"): the
 * ExecNode-surface mirror that makes the GPU engine a drop-in for
 * baikalStore's SELECT pipeline.
 *
 * The C++ classes inside libbkgpu.so mirror, name for name and signature for
 * signature, the reference's plugin surface:
 *   ExecNode::init(const pb::PlanNode&) / open(RuntimeState*) /
 *   get_next(RuntimeState*, RowBatch*, bool* eos) / close(RuntimeState*)
 *     -> include/exec/exec_node.h:88,140-153
 *   node construction from a flattened pre-order plan
 *     -> ExecNode::create_tree / create_exec_node,
 *        src/exec/exec_node.cpp:396-414, proto/plan.proto:495-510
 *   driver loop: Region::select_normal's  while(!eos){root->get_next(batch)}
 *     -> src/store/region.cpp:3166-3216 (re-created by bkexec_run /
 *        bkexec_get_next below)
 *
 * This header is the flat C view of that surface for embedders and tests
 * (a cgo/JNI/ctypes binding would wrap exactly these entry points; the C++
 * embedding in a baikalStore build uses the classes directly, see
 * INTEGRATION.md).
 */
#ifndef BK_EXEC_H
#define BK_EXEC_H

#include <stdint.h>
#include "bk_common.h"
#include "bkgpu.h"

#ifdef __cplusplus
extern "C" {
#endif

/* One node of the flattened pre-order plan (pb::PlanNode subset;
 * proto/plan.proto:10-23 node types, plan.proto:495-510 flattening). */
typedef struct BkPlanNodeDesc {
    int32_t node_type;     /* BkNodeType */
    int32_t num_children;  /* pre-order flattening, like pb::Plan */
    int64_t limit;         /* LIMIT_NODE / SORT_NODE limit, -1 none */
    int64_t offset;        /* LIMIT_NODE rows skipped before emitting
                              (limit_node.h:21-41 _offset semantics) */
    /* SCAN_NODE payload */
    BkgTable* table;       /* the region's columnar source */
    /* WHERE/TABLE_FILTER_NODE payload */
    int32_t    n_conjuncts;
    BkConjunct conjuncts[BK_MAX_CONJUNCTS];
    /* AGG/MERGE_AGG payload */
    int32_t    n_group;
    int32_t    group_cols[BK_MAX_GROUP];
    int32_t    group_bits[BK_MAX_GROUP];   /* key packing, bk_common.h */
    int64_t    group_base[BK_MAX_GROUP];
    int32_t    n_aggs;
    BkAggSpec  aggs[BK_MAX_AGGS];
    int64_t    expected_groups;
    int32_t    distinct_bits;      /* declared width/base of DISTINCT column
                                      encodings: lets the level-1 (keys + d)
                                      pass pack into one word so the sort-
                                      dedup level 1 qualifies (bkdedup.inc);
                                      0 = undeclared (hash path only) */
    int32_t    _pad_d;
    int64_t    distinct_base;
    /* SORT_NODE payload */
    int32_t     n_order;
    BkOrderSpec order[4];
    int32_t     n_out_cols;          /* columns materialized per output row */
    int32_t     out_cols[BK_MAX_COLS];
    /* WINDOW_NODE payload (window_node.cpp); reuses n_order/order for the
     * in-partition sort and n_out_cols/out_cols for the materialized input
     * columns. frame_mode: 0 non-frame, 1 ROWS [f_pre PRECEDING, f_fol
     * FOLLOWING] (negative = UNBOUNDED), 2 RANGE UNBOUNDED..CURRENT,
     * 3 RANGE CURRENT..UNBOUNDED (see bkgpu_window). */
    int32_t     part_col;            /* -1 = single whole-set partition */
    int32_t     n_winfns;
    BkWindowFn  winfns[BK_MAX_WINFNS];
    int32_t     frame_mode;
    int32_t     _pad_w;
    int64_t     frame_pre, frame_fol;
} BkPlanNodeDesc;

typedef struct BkExecTree BkExecTree;      /* root ExecNode + RuntimeState */

/* ExecNode::create_tree equivalent: builds the node tree from the pre-order
 * array. Returns NULL on error (bkgpu_last_error()). */
BkExecTree* bkexec_create_tree(const BkPlanNodeDesc* nodes, int n_nodes);

/* root->open(&state) — may drain children (AggNode/SortNode semantics,
 * agg_node.cpp:405-505 / sort_node.cpp:278-347; on this engine the drain IS
 * the fused GPU pipeline). 0 ok, <0 error. */
int bkexec_open(BkExecTree* t);

/* root->get_next(&state, batch, &eos): fills up to `capacity` rows.
 * A row is returned as tagged slot values (BkType tag + i64/f64 payload +
 * null flag). Slot layout per root node type:
 *   AGG root:  [group cols...][agg outputs...]
 *   SORT root: [out_cols...]
 * Writes row-major into out_tag/out_i/out_d/out_null (capacity*n_slots).
 * Returns rows produced (>=0) and sets *eos, or <0 on error. */
int64_t bkexec_get_next(BkExecTree* t, int64_t capacity, int32_t* out_tag,
                        int64_t* out_i, double* out_d, uint8_t* out_null,
                        int* eos);

int bkexec_n_slots(const BkExecTree* t);

/* RuntimeState counters the store reports back (runtime_state.h:237-270) */
int64_t bkexec_num_scan_rows(const BkExecTree* t);
int64_t bkexec_num_filter_rows(const BkExecTree* t);
int64_t bkexec_num_rows_returned(const BkExecTree* t);

/* root->close(&state) + destroy tree (idempotent reset semantics) */
void bkexec_close(BkExecTree* t);

/* dict word materialization for BK_STRING group outputs (the host dict is
 * deterministic from the generation seed; a real embedding would plug its
 * own dictionary here) */
int bkexec_dict_word(uint64_t dict_seed, int64_t code, char* out, int cap);

#ifdef __cplusplus
}
#endif
#endif /* BK_EXEC_H */
