// bkexec.cpp — C++ host execution layer: the ExecNode plugin-surface mirror
// that makes the MI355X engine a drop-in for baikalStore's SELECT pipeline.
//
// The classes here keep the reference's names, signatures, argument meaning
// and error behaviour:
//   ExecNode::init/open/get_next/close      include/exec/exec_node.h:88,140-153
//   ExecNode::create_tree/create_exec_node  src/exec/exec_node.cpp:396-414
//   RuntimeState counters                   include/runtime/runtime_state.h:237-270
//   RowBatch / MemRow containers            include/runtime/row_batch.h:24-231,
//                                           include/mem_row/mem_row.h:28-215
//   FilterNode::get_next                    src/exec/filter_node.cpp:736-795
//   AggNode::open/get_next                  src/exec/agg_node.cpp:405-587
//   SortNode::open/get_next (top-N)         src/exec/sort_node.cpp:278-440
//   LimitNode                               (reached_limit, exec_node.h:186)
//
// Like the reference's Acero alternative path (region.cpp:2793-2923), the
// blocking nodes compile their subtree into ONE engine pipeline at open()
// and stream materialized rows from get_next() — but the engine here is the
// gfx950 kernel pipeline behind include/bkgpu.h, not a CPU library. There is
// NO row-at-a-time CPU fallback: a plan that reaches ScanNode::get_next in
// row mode returns an error instead of silently computing on the host.

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <memory>
#include <string>
#include <vector>

#include "../../include/bk_common.h"
#include "../../include/bk_like.h"
#include "../../include/bk_keyenc.h"
#include "../../include/bk_datagen.h"
#include "../../include/bkgpu.h"
#include "../../include/bk_exec.h"

namespace bkexec {

/* ---- ExprValue-lite (include/common/expr_value.h:35-52 tagged union) ---- */
struct ExprValue {
    int32_t type = BK_NULL_TYPE;   /* BkType */
    bool is_null_ = true;
    int64_t i = 0;
    double d = 0.0;
    bool is_null() const { return is_null_; }
};

/* ---- MemRow (mem_row.h:28-215; one tuple, slot-addressed) ---- */
struct MemRow {
    std::vector<ExprValue> slots;
    explicit MemRow(int n) : slots(n) {}
    const ExprValue& get_value(int slot) const { return slots[slot]; }
    void set_value(int slot, const ExprValue& v) { slots[slot] = v; }
};

/* ---- RowBatch (row_batch.h:24-231) ---- */
class RowBatch {
public:
    explicit RowBatch(size_t capacity = 1024) : _capacity(capacity) {}
    bool is_full() const { return _rows.size() >= _capacity; }
    size_t size() const { return _rows.size(); }
    size_t capacity() const { return _capacity; }
    void set_capacity(size_t c) { _capacity = c; }
    void move_row(std::unique_ptr<MemRow> row) { _rows.emplace_back(std::move(row)); }
    std::unique_ptr<MemRow>& get_row() { return _rows[_idx]; }
    void next() { _idx++; }
    void reset() { _idx = 0; }
    bool is_traverse_over() const { return _idx >= _rows.size(); }
    void clear() { _rows.clear(); _idx = 0; }
    void truncate(size_t n) { if (n < _rows.size()) _rows.resize(n); }
    void drop_range(size_t pos, size_t k) {
        if (k == 0 || pos >= _rows.size()) return;
        size_t end = pos + k < _rows.size() ? pos + k : _rows.size();
        _rows.erase(_rows.begin() + (ptrdiff_t)pos,
                    _rows.begin() + (ptrdiff_t)end);
    }
private:
    std::vector<std::unique_ptr<MemRow>> _rows;
    size_t _idx = 0;
    size_t _capacity;
};

/* ---- RuntimeState (runtime_state.h:83-660 subset: the counters and error
 * surface the store reports back) ---- */
class RuntimeState {
public:
    int64_t num_scan_rows() const { return _num_scan_rows; }
    int64_t num_filter_rows() const { return _num_filter_rows; }
    void inc_num_scan_rows(int64_t n) { _num_scan_rows += n; }
    void inc_num_filter_rows(int64_t n) { _num_filter_rows += n; }
    bool is_cancelled() const { return _cancelled; }
    void cancel() { _cancelled = true; }
    int error_code = 0;
    std::string error_msg;
    size_t row_batch_capacity = 1024;
    int64_t _num_scan_rows = 0;
    int64_t _num_filter_rows = 0;
    bool _cancelled = false;
};

/* ---- ExecNode base (exec_node.h:79-531 subset) ---- */
class ExecNode {
public:
    virtual ~ExecNode() {
        for (auto c : _children) delete c;   /* nodes own children, exec_node.h:83-87 */
    }
    virtual int init(const BkPlanNodeDesc& node) {
        _node_type = node.node_type;
        _limit = node.limit;
        return 0;
    }
    virtual int open(RuntimeState* state) {
        for (auto c : _children) {
            int ret = c->open(state);
            if (ret < 0) return ret;
        }
        return 0;
    }
    virtual int get_next(RuntimeState* state, RowBatch* batch, bool* eos) {
        if (_children.empty()) { *eos = true; return 0; }
        return _children[0]->get_next(state, batch, eos);
    }
    virtual void close(RuntimeState* state) {
        for (auto c : _children) c->close(state);
        _num_rows_returned = 0;
    }
    void add_child(ExecNode* c) { _children.push_back(c); }
    std::vector<ExecNode*>& children() { return _children; }
    bool reached_limit() const {
        return _limit > 0 && _num_rows_returned >= _limit;  /* exec_node.h:186 */
    }
    int32_t node_type() const { return _node_type; }

    static ExecNode* create_exec_node(const BkPlanNodeDesc& node);
    /* pre-order flattened plan -> tree (exec_node.cpp create_tree) */
    static int create_tree(const BkPlanNodeDesc* nodes, int n, ExecNode** root);

    int32_t _node_type = 0;
    int64_t _limit = -1;
    int64_t _num_rows_returned = 0;
protected:
    std::vector<ExecNode*> _children;
private:
    static int build(const BkPlanNodeDesc* nodes, int n, int* pos, ExecNode** out);
};

/* ---- ScanNode: the region's columnar source. Row-mode get_next is an
 * explicit error: the compute path is the GPU pipeline, never host rows. */
class ScanNode : public ExecNode {
public:
    int init(const BkPlanNodeDesc& node) override {
        ExecNode::init(node);
        _table = node.table;
        if (!_table) return -1;
        return 0;
    }
    int get_next(RuntimeState* state, RowBatch*, bool*) override {
        state->error_msg = "ScanNode row-mode get_next: no CPU fallback; "
                           "plan must be rooted in a GPU-compiled node";
        return -1;
    }
    BkgTable* table() const { return _table; }
private:
    BkgTable* _table = nullptr;
};

/* ---- FilterNode (filter_node.cpp:605-795): holds the conjuncts. When it is
 * an interior node, the blocking parent compiles them into the pipeline;
 * when it is the (effective) root, open() runs the GPU filter and get_next()
 * streams materialized rows. ---- */
class FilterNode : public ExecNode {
public:
    int init(const BkPlanNodeDesc& node) override {
        ExecNode::init(node);
        _n_conjuncts = node.n_conjuncts;
        memcpy(_conjuncts, node.conjuncts, sizeof(_conjuncts));
        return 0;
    }
    int n_conjuncts() const { return _n_conjuncts; }
    const BkConjunct* conjuncts() const { return _conjuncts; }
    /* As the EFFECTIVE ROOT (SELECT without GROUP BY/ORDER BY) FilterNode
     * emits the passing rows itself (filter_node.cpp:736-795): the GPU
     * filter runs over BOUNDED row-range chunks (BK_FETCH_CHUNK rows,
     * default 4M) and get_next() streams each chunk's survivors before
     * scanning the next range, so host memory stays O(chunk) for a
     * filter-only SELECT over any table size — the streamed analogue of
     * the reference scan's batch iterator (rocksdb_scan_node.cpp
     * get_next loop), not a whole-result materialization. */
    int get_next(RuntimeState* state, RowBatch* batch, bool* eos) override;
    int n_slots();  /* lazily resolved from the scan's table */
    void close(RuntimeState* state) override {
        ExecNode::close(state);
        _opened = false;
        _eos_source = false;
        _scan_pos = 0;
        _rowids.clear();
        _cols_i.clear(); _cols_d.clear(); _cols_n.clear();
        _iter = 0;
    }
private:
    int fetch_chunk(RuntimeState* state);  /* 1 = chunk ready, 0 = drained */
    int32_t _n_conjuncts = 0;
    BkConjunct _conjuncts[BK_MAX_CONJUNCTS] = {};
    bool _opened = false;
    bool _eos_source = false;
    int _ncols = 0;
    int64_t _nrows = 0;
    int64_t _scan_pos = 0;
    std::vector<int64_t> _rowids;          /* current chunk's survivors */
    std::vector<std::vector<int64_t>> _cols_i;
    std::vector<std::vector<double>> _cols_d;
    std::vector<std::vector<uint8_t>> _cols_n;
    std::vector<int32_t> _col_types;
    int64_t _iter = 0;                     /* cursor within the chunk */
};

/* helpers to locate the pipeline pieces below a blocking node */
static ScanNode* find_scan(ExecNode* n);

int FilterNode::n_slots() {
    if (_ncols == 0) {
        ScanNode* scan = find_scan(this);
        if (scan) _ncols = bkgpu_table_ncols(scan->table());
    }
    return _ncols;
}

static int64_t fetch_chunk_rows() {
    /* read per chunk (once every few million rows), not cached, so a
     * long-lived process can be re-tuned between queries */
    const char* e = getenv("BK_FETCH_CHUNK");
    int64_t n = e ? atoll(e) : 0;
    return n > 0 ? n : (int64_t)(4 << 20);  /* 4M rows per chunk */
}

int FilterNode::fetch_chunk(RuntimeState* state) {
    ScanNode* scan = find_scan(this);
    if (!scan) { state->error_msg = "FilterNode: no scan below"; return -1; }
    BkgTable* t = scan->table();
    if (!_opened) {
        _nrows = bkgpu_table_nrows(t);
        _ncols = bkgpu_table_ncols(t);
        _col_types.resize(_ncols);
        _cols_i.assign(_ncols, {});
        _cols_d.assign(_ncols, {});
        _cols_n.assign(_ncols, {});
        for (int c = 0; c < _ncols; c++)
            _col_types[c] = bkgpu_table_col_type(t, c);
        _scan_pos = 0;
        _opened = true;
    }
    BkQuerySpec q{};
    q.n_conjuncts = _n_conjuncts;
    memcpy(q.conjuncts, _conjuncts, sizeof(q.conjuncts));
    const int64_t chunk = fetch_chunk_rows();
    while (_scan_pos < _nrows) {
        int64_t end = std::min(_nrows, _scan_pos + chunk);
        int64_t span = end - _scan_pos;
        _rowids.resize(span);
        int64_t got = bkgpu_filter_collect(t, &q, _scan_pos, end, span,
                                           _rowids.data());
        if (got < 0) { state->error_msg = bkgpu_last_error(); return -1; }
        _rowids.resize(got);
        /* arrival order: the reference scan emits rows in iterator order;
         * chunks advance in row order, so sorting within the chunk keeps
         * the global stream ordered. */
        std::sort(_rowids.begin(), _rowids.end());
        state->inc_num_scan_rows(span);
        state->inc_num_filter_rows(span - got);
        _scan_pos = end;
        if (got == 0) continue;
        for (int c = 0; c < _ncols; c++) {
            _cols_i[c].resize(got);
            _cols_d[c].resize(got);
            _cols_n[c].resize(got);
            if (bkgpu_gather(t, c, _rowids.data(), got,
                             _cols_i[c].data(), _cols_d[c].data(),
                             _cols_n[c].data()) != 0) {
                state->error_msg = bkgpu_last_error();
                return -1;
            }
        }
        _iter = 0;
        return 1;
    }
    return 0;
}

int FilterNode::get_next(RuntimeState* state, RowBatch* batch, bool* eos) {
    while (true) {
        if (state->is_cancelled()) { *eos = true; return 0; }
        if (reached_limit()) { *eos = true; return 0; }
        if (_iter >= (int64_t)_rowids.size()) {
            if (_eos_source) { *eos = true; return 0; }
            int r = fetch_chunk(state);
            if (r < 0) return r;
            if (r == 0) { _eos_source = true; *eos = true; return 0; }
        }
        if (batch->is_full()) return 0;
        auto row = std::make_unique<MemRow>(_ncols);
        for (int c = 0; c < _ncols; c++) {
            ExprValue v;
            v.type = _col_types[c];
            v.is_null_ = _cols_n[c][_iter] != 0;
            if (!v.is_null_) {
                if (v.type == BK_DOUBLE) v.d = _cols_d[c][_iter];
                else v.i = _cols_i[c][_iter];
            }
            row->set_value(c, v);
        }
        batch->move_row(std::move(row));
        _num_rows_returned++;
        _iter++;
    }
}

static ScanNode* find_scan(ExecNode* n) {
    if (!n) return nullptr;
    if (n->node_type() == BK_SCAN_NODE) return static_cast<ScanNode*>(n);
    for (auto c : n->children()) {
        ScanNode* s = find_scan(c);
        if (s) return s;
    }
    return nullptr;
}
static FilterNode* find_filter(ExecNode* n) {
    if (!n) return nullptr;
    if (n->node_type() == BK_TABLE_FILTER_NODE ||
        n->node_type() == BK_WHERE_FILTER_NODE)
        return static_cast<FilterNode*>(n);
    for (auto c : n->children()) {
        FilterNode* f = find_filter(c);
        if (f) return f;
    }
    return nullptr;
}

struct FetchedGroups {
    int64_t n = 0;
    std::vector<uint8_t> flags;
    std::vector<uint64_t> enc;
    std::vector<int64_t> out_i;
    std::vector<double> out_d;
    std::vector<uint8_t> out_has;
};

/* output tag of an aggregate (finalize types, agg_fn_call.cpp:927-975) */
static int32_t agg_out_type(int agg_type, int32_t in_type) {
    switch (agg_type) {
        case BK_AGG_COUNT_STAR:
        case BK_AGG_COUNT:
        case BK_AGG_COUNT_DISTINCT: return BK_INT64;  /* agg_fn_call.cpp:96-99 */
        case BK_AGG_AVG:
        case BK_AGG_AVG_DISTINCT: return BK_DOUBLE;
        default:           return in_type;  /* SUM/MIN/MAX keep input type */
    }
}

/* ---- AggNode (agg_node.cpp:405-587). open() drains the child pipeline —
 * here: compiles {scan, filter, group, aggs} into one bkgpu_filter_agg call
 * (the Acero-slot pattern, region.cpp:2793) — and get_next() emits finalized
 * group rows. MERGE_AGG shares the implementation (the engine's aggregate
 * states already merge, agg_node.cpp:29,539-543). ---- */
class AggNode : public ExecNode {
public:
    int init(const BkPlanNodeDesc& node) override {
        ExecNode::init(node);
        _desc = node;
        return 0;
    }
    int open(RuntimeState* state) override {
        int ret = ExecNode::open(state);
        if (ret < 0) return ret;
        ScanNode* scan = find_scan(this);
        if (!scan) { state->error_msg = "AggNode: no scan below"; return -1; }
        FilterNode* filter = find_filter(this);
        BkQuerySpec q{};
        if (filter) {
            q.n_conjuncts = filter->n_conjuncts();
            memcpy(q.conjuncts, filter->conjuncts(), sizeof(q.conjuncts));
        }
        q.n_group = _desc.n_group;
        BkgTable* t = scan->table();
        for (int k = 0; k < q.n_group; k++) {
            q.group_cols[k] = _desc.group_cols[k];
            q.group_types[k] = bkgpu_table_col_type(t, _desc.group_cols[k]);
            q.group_bits[k] = _desc.group_bits[k];
            q.group_base[k] = _desc.group_base[k];
        }
        q.n_aggs = _desc.n_aggs;
        for (int a = 0; a < q.n_aggs; a++) {
            q.aggs[a] = _desc.aggs[a];
            const BkAggSpec& as = _desc.aggs[a];
            if (as.arith && as.col2 >= 0) {
                /* expression input: DOUBLE domain iff either operand is
                 * DOUBLE (AggFnCall input cast, agg_fn_call.cpp:496-555) */
                int t1 = bkgpu_table_col_type(t, as.col);
                int t2 = bkgpu_table_col_type(t, as.col2);
                q.agg_in_types[a] = (t1 == BK_DOUBLE || t2 == BK_DOUBLE)
                                        ? BK_DOUBLE : BK_INT64;
            } else {
                q.agg_in_types[a] = as.col >= 0
                    ? bkgpu_table_col_type(t, as.col) : BK_INT64;
            }
        }
        _q = q;
        int64_t expected = _desc.expected_groups > 0 ? _desc.expected_groups : 65536;
        /* DISTINCT aggs: the reference's planner rewrite (agg_node.cpp:
         * 247-258) — per distinct column, level 1 groups by (user keys +
         * that column) and bkgpu_agg_rollup folds the dedup key out. The
         * reference's MULTI_COUNT_DISTINCT (several distinct columns) runs
         * one such pass per column; every pass shares the same filter and
         * group key so the canonical-sorted group sets align row for row
         * and the output columns stitch together. */
        std::vector<int> dcols;            /* distinct columns, deduped */
        std::vector<int> agg_dcol(q.n_aggs, -1);
        for (int a = 0; a < q.n_aggs; a++) {
            int at = q.aggs[a].agg_type;
            if (at < BK_AGG_COUNT_DISTINCT) continue;
            int c = q.aggs[a].col;
            size_t j = 0;
            while (j < dcols.size() && dcols[j] != c) j++;
            if (j == dcols.size()) dcols.push_back(c);
            agg_dcol[a] = (int)j;
        }
        bool has_distinct = !dcols.empty();
        /* passes: pass 0 = plain aggs (or everything when no distinct);
         * pass 1+j = distinct col j. src[a] = (pass, idx-in-pass). */
        struct Src { int pass; int idx; };
        std::vector<Src> src(q.n_aggs);
        std::vector<BkgAggOut*> outs;
        int64_t nrows_t = bkgpu_table_nrows(t);
        if (!has_distinct) {
            BkgAggOut* out = bkgpu_filter_agg(t, &q, 0, nrows_t, expected);
            if (!out) { state->error_msg = bkgpu_last_error(); return -1; }
            outs.push_back(out);
            for (int a = 0; a < q.n_aggs; a++) src[a] = {0, a};
        } else {
            if (q.n_group > 2) {
                state->error_msg = "DISTINCT aggs support <= 2 group keys";
                return -1;
            }
            if (q.n_group == 2 && (!q.group_bits[0] || !q.group_bits[1])) {
                /* 2 user keys + d = 3 level-1 keys: they must pack into
                 * the two 64-bit key words via declared widths */
                state->error_msg = "DISTINCT with 2 group keys needs "
                                   "group_bits declared for both";
                return -1;
            }
            /* pass 0: the plain aggs (always run — it also anchors the
             * group set when a group has rows but only NULL d values) */
            BkQuerySpec q0 = q;
            q0.n_aggs = 0;
            for (int a = 0; a < q.n_aggs; a++) {
                if (agg_dcol[a] >= 0) continue;
                src[a] = {0, q0.n_aggs};
                q0.aggs[q0.n_aggs] = q.aggs[a];
                q0.agg_in_types[q0.n_aggs] = q.agg_in_types[a];
                q0.n_aggs++;
            }
            if (q0.n_aggs == 0) {
                q0.aggs[0].agg_type = BK_AGG_COUNT_STAR;
                q0.aggs[0].col = -1;
                q0.agg_in_types[0] = BK_INT64;
                q0.n_aggs = 1;
            }
            BkgAggOut* p0 = bkgpu_filter_agg(t, &q0, 0, nrows_t, expected);
            if (!p0) { state->error_msg = bkgpu_last_error(); return -1; }
            outs.push_back(p0);
            for (size_t j = 0; j < dcols.size(); j++) {
                BkQuerySpec q1 = q;
                q1.n_group = q.n_group + 1;
                q1.group_cols[q.n_group] = dcols[j];
                q1.group_types[q.n_group] = bkgpu_table_col_type(t, dcols[j]);
                q1.aggs[0].agg_type = BK_AGG_COUNT_STAR;
                q1.aggs[0].col = -1;
                q1.agg_in_types[0] = BK_INT64;
                q1.n_aggs = 1;
                /* level-2 spec: only this column's distinct aggs */
                BkQuerySpec q2 = q;
                q2.n_aggs = 0;
                int32_t src_idx[BK_MAX_AGGS];
                for (int a = 0; a < q.n_aggs; a++) {
                    if (agg_dcol[a] != (int)j) continue;
                    src[a] = {(int)j + 1, q2.n_aggs};
                    src_idx[q2.n_aggs] = -1;
                    q2.aggs[q2.n_aggs] = q.aggs[a];
                    q2.agg_in_types[q2.n_aggs] = q.agg_in_types[a];
                    q2.n_aggs++;
                }
                /* high-cardinality dedup: try the sort-based level 1 when
                 * the (keys + d) widths are declared and pack (bkdedup.inc);
                 * fall back to the hash path otherwise — same results */
                BkgAggOut* l1 = nullptr;
                if (_desc.distinct_bits > 0) {
                    q1.group_bits[q.n_group] = _desc.distinct_bits;
                    q1.group_base[q.n_group] = _desc.distinct_base;
                }
                /* high-cardinality dedup: try the sort-based level 1 (packs
                 * via declared widths or column stats, bkdedup.inc); falls
                 * back to the hash path when the shape does not qualify */
                if (expected * 16 >= (1ll << 22))
                    l1 = bkgpu_filter_agg_sorted(t, &q1, 0, nrows_t);
                if (!l1)
                    l1 = bkgpu_filter_agg(t, &q1, 0, nrows_t, expected * 16);
                BkgAggOut* r = l1 ? bkgpu_agg_rollup(l1, &q2, src_idx,
                                                     expected) : nullptr;
                if (l1) bkgpu_agg_free(l1);
                if (!r) {
                    state->error_msg = bkgpu_last_error();
                    for (auto* o : outs) bkgpu_agg_free(o);
                    return -1;
                }
                outs.push_back(r);
            }
        }
        state->inc_num_scan_rows(nrows_t);
        state->inc_num_filter_rows(nrows_t - bkgpu_agg_rows_passed(outs[0]));
        /* fetch every pass in canonical key order; group sets align (same
         * filter + same user keys), assemble the output columns */
        int64_t n = bkgpu_agg_ngroups(outs[0]);
        _g.n = n;
        _g.flags.resize(n ? n : 1);
        _g.enc.resize((n ? n : 1) * BK_MAX_GROUP);
        _g.out_i.resize((size_t)(n ? n : 1) * q.n_aggs);
        _g.out_d.resize((size_t)(n ? n : 1) * q.n_aggs);
        _g.out_has.resize((size_t)(n ? n : 1) * q.n_aggs);
        int64_t got = -1;
        for (size_t p = 0; p < outs.size(); p++) {
            int pn_aggs = 0;
            for (int a = 0; a < q.n_aggs; a++)
                if (src[a].pass == (int)p && src[a].idx + 1 > pn_aggs)
                    pn_aggs = src[a].idx + 1;
            if (p == 0 && pn_aggs == 0) pn_aggs = 1;  /* synthetic count(*) */
            std::vector<uint8_t> flags(n ? n : 1);
            std::vector<uint64_t> enc((n ? n : 1) * BK_MAX_GROUP);
            std::vector<int64_t> oi((size_t)(n ? n : 1) * pn_aggs);
            std::vector<double> od((size_t)(n ? n : 1) * pn_aggs);
            std::vector<uint8_t> oh((size_t)(n ? n : 1) * pn_aggs);
            int64_t gp = bkgpu_agg_fetch(outs[p], /*sorted=*/1, n,
                                         flags.data(), enc.data(), oi.data(),
                                         od.data(), oh.data());
            if (gp < 0 || (p > 0 && gp != got)) {
                state->error_msg = gp < 0 ? bkgpu_last_error()
                                          : "distinct pass group mismatch";
                for (auto* o : outs) bkgpu_agg_free(o);
                return -1;
            }
            if (p == 0) {
                got = gp;
                memcpy(_g.flags.data(), flags.data(), (size_t)(gp ? gp : 1));
                memcpy(_g.enc.data(), enc.data(),
                       (size_t)(gp ? gp : 1) * BK_MAX_GROUP * 8);
            }
            for (int a = 0; a < q.n_aggs; a++) {
                if (src[a].pass != (int)p) continue;
                for (int64_t g = 0; g < gp; g++) {
                    size_t di = (size_t)a * gp + g;
                    size_t si = (size_t)src[a].idx * gp + g;
                    _g.out_i[di] = oi[si];
                    _g.out_d[di] = od[si];
                    _g.out_has[di] = oh[si];
                }
            }
        }
        for (auto* o : outs) bkgpu_agg_free(o);
        _g.n = got;
        _iter = 0;
        return 0;
    }
    int get_next(RuntimeState* state, RowBatch* batch, bool* eos) override {
        while (true) {
            if (state->is_cancelled()) { *eos = true; return 0; } /* agg_node.cpp:450 */
            if (reached_limit() || _iter >= _g.n) { *eos = true; return 0; }
            if (batch->is_full()) return 0;
            auto row = std::make_unique<MemRow>(n_slots());
            int s = 0;
            for (int k = 0; k < _q.n_group; k++, s++) {
                ExprValue v;
                v.type = _q.group_types[k];
                if ((_g.flags[_iter] >> (7 - k)) & 1) {
                    v.is_null_ = true;
                } else {
                    v.is_null_ = false;
                    uint64_t e = _g.enc[_iter * BK_MAX_GROUP + k];
                    if (v.type == BK_DOUBLE) v.d = bk_dec_f64(e);
                    else if (v.type == BK_STRING) v.i = (int64_t)(uint32_t)e;
                    else v.i = bk_dec_i64(e);
                }
                row->set_value(s, v);
            }
            for (int a = 0; a < _q.n_aggs; a++, s++) {
                ExprValue v;
                v.type = agg_out_type(_q.aggs[a].agg_type, _q.agg_in_types[a]);
                size_t idx = (size_t)a * _g.n + _iter;
                if (!_g.out_has[idx]) {
                    v.is_null_ = true;
                } else {
                    v.is_null_ = false;
                    if (v.type == BK_DOUBLE) v.d = _g.out_d[idx];
                    else v.i = _g.out_i[idx];
                }
                row->set_value(s, v);
            }
            batch->move_row(std::move(row));
            _num_rows_returned++;
            _iter++;
        }
    }
    void close(RuntimeState* state) override {
        ExecNode::close(state);
        _g = FetchedGroups{};
        _iter = 0;
    }
    int n_slots() const { return _q.n_group + _q.n_aggs; }
    const BkQuerySpec& spec() const { return _q; }
private:
    BkPlanNodeDesc _desc{};
    BkQuerySpec _q{};
    FetchedGroups _g;
    int64_t _iter = 0;
};

/* ---- SortNode + TopN (sort_node.cpp:278-440, topn_sorter.h:32-63): open()
 * drains the child via the GPU top-N selection, get_next() emits the
 * materialized out_cols of the selected rows in final order. ---- */
class SortNode : public ExecNode {
public:
    int init(const BkPlanNodeDesc& node) override {
        ExecNode::init(node);
        _desc = node;
        if (_desc.n_out_cols <= 0) return -1;
        return 0;
    }
    int open(RuntimeState* state) override {
        int ret = ExecNode::open(state);
        if (ret < 0) return ret;
        ScanNode* scan = find_scan(this);
        if (!scan) { state->error_msg = "SortNode: no scan below"; return -1; }
        FilterNode* filter = find_filter(this);
        BkQuerySpec q{};
        if (filter) {
            q.n_conjuncts = filter->n_conjuncts();
            memcpy(q.conjuncts, filter->conjuncts(), sizeof(q.conjuncts));
        }
        BkgTable* t = scan->table();
        int64_t limit = _limit > 0 ? _limit : bkgpu_table_nrows(t);
        _rowids.resize(limit > 0 ? limit : 1);
        int64_t got = bkgpu_sort_topk(t, &q, _desc.order, _desc.n_order, 0,
                                      bkgpu_table_nrows(t), limit, _rowids.data());
        if (got < 0) { state->error_msg = bkgpu_last_error(); return -1; }
        _rowids.resize(got);
        state->inc_num_scan_rows(bkgpu_table_nrows(t));
        /* materialize out_cols */
        _cols_i.assign(_desc.n_out_cols, {});
        _cols_d.assign(_desc.n_out_cols, {});
        _cols_n.assign(_desc.n_out_cols, {});
        _col_types.resize(_desc.n_out_cols);
        for (int c = 0; c < _desc.n_out_cols; c++) {
            int col = _desc.out_cols[c];
            _col_types[c] = bkgpu_table_col_type(t, col);
            _cols_i[c].resize(got ? got : 1);
            _cols_d[c].resize(got ? got : 1);
            _cols_n[c].resize(got ? got : 1);
            if (got > 0 &&
                bkgpu_gather(t, col, _rowids.data(), got, _cols_i[c].data(),
                             _cols_d[c].data(), _cols_n[c].data()) != 0) {
                state->error_msg = bkgpu_last_error();
                return -1;
            }
        }
        _iter = 0;
        return 0;
    }
    int get_next(RuntimeState* state, RowBatch* batch, bool* eos) override {
        while (true) {
            if (state->is_cancelled()) { *eos = true; return 0; }
            if (reached_limit() || _iter >= (int64_t)_rowids.size()) {
                *eos = true;
                return 0;
            }
            if (batch->is_full()) return 0;
            auto row = std::make_unique<MemRow>(n_slots());
            for (int c = 0; c < _desc.n_out_cols; c++) {
                ExprValue v;
                v.type = _col_types[c];
                v.is_null_ = _cols_n[c][_iter] != 0;
                if (!v.is_null_) {
                    if (v.type == BK_DOUBLE) v.d = _cols_d[c][_iter];
                    else v.i = _cols_i[c][_iter];
                }
                row->set_value(c, v);
            }
            batch->move_row(std::move(row));
            _num_rows_returned++;
            _iter++;
        }
    }
    void close(RuntimeState* state) override {
        ExecNode::close(state);
        _rowids.clear();
        _iter = 0;
    }
    int n_slots() const { return _desc.n_out_cols; }
    const std::vector<int32_t>& col_types() const { return _col_types; }
private:
    BkPlanNodeDesc _desc{};
    std::vector<int64_t> _rowids;
    std::vector<std::vector<int64_t>> _cols_i;
    std::vector<std::vector<double>> _cols_d;
    std::vector<std::vector<uint8_t>> _cols_n;
    std::vector<int32_t> _col_types;
    int64_t _iter = 0;
};

/* ---- LimitNode ---- */
/* ---- WindowNode (window_node.cpp, NON-FRAME mode; fns evaluated by
 * bkgpu_window over rows sorted by (partition, order, arrival)). Slots:
 * [out_cols...][window fn outputs...]. ---- */
class WindowNode : public ExecNode {
public:
    int init(const BkPlanNodeDesc& node) override {
        ExecNode::init(node);
        _desc = node;
        if (_desc.n_winfns <= 0) return -1;
        return 0;
    }
    static int32_t fn_out_type(const BkWindowFn& f, BkgTable* t) {
        switch (f.fn_type) {
            case BK_WIN_COUNT_STAR: case BK_WIN_COUNT: case BK_WIN_ROW_NUMBER:
            case BK_WIN_RANK: case BK_WIN_DENSE_RANK: case BK_WIN_NTILE:
                return BK_INT64;                 /* window_fn_call.cpp:213-258 */
            case BK_WIN_AVG: case BK_WIN_PERCENT_RANK: case BK_WIN_CUME_DIST:
                return BK_DOUBLE;
            default:
                return bkgpu_table_col_type(t, f.col);  /* input type */
        }
    }
    int open(RuntimeState* state) override {
        int ret = ExecNode::open(state);
        if (ret < 0) return ret;
        ScanNode* scan = find_scan(this);
        if (!scan) { state->error_msg = "WindowNode: no scan below"; return -1; }
        FilterNode* filter = find_filter(this);
        BkQuerySpec q{};
        if (filter) {
            q.n_conjuncts = filter->n_conjuncts();
            memcpy(q.conjuncts, filter->conjuncts(), sizeof(q.conjuncts));
        }
        BkgTable* t = scan->table();
        int64_t nrows = bkgpu_table_nrows(t);
        _rowids.resize(nrows ? nrows : 1);
        _win_i.resize((size_t)_desc.n_winfns * (nrows ? nrows : 1));
        _win_d.resize((size_t)_desc.n_winfns * (nrows ? nrows : 1));
        _win_n.resize((size_t)_desc.n_winfns * (nrows ? nrows : 1));
        int64_t got = bkgpu_window(t, &q, _desc.part_col, _desc.order,
                                   _desc.n_order, _desc.winfns, _desc.n_winfns,
                                   _desc.frame_mode,
                                   _desc.frame_mode ? _desc.frame_pre : -1,
                                   _desc.frame_mode ? _desc.frame_fol : -1,
                                   0, nrows, _rowids.data(), _win_i.data(),
                                   _win_d.data(), _win_n.data());
        if (got < 0) { state->error_msg = bkgpu_last_error(); return -1; }
        _n = got;
        state->inc_num_scan_rows(nrows);
        state->inc_num_filter_rows(nrows - got);
        _fn_types.resize(_desc.n_winfns);
        for (int f = 0; f < _desc.n_winfns; f++)
            _fn_types[f] = fn_out_type(_desc.winfns[f], t);
        /* materialize out_cols of the sorted rows */
        _cols_i.assign(_desc.n_out_cols, {});
        _cols_d.assign(_desc.n_out_cols, {});
        _cols_n.assign(_desc.n_out_cols, {});
        _col_types.resize(_desc.n_out_cols);
        for (int c = 0; c < _desc.n_out_cols; c++) {
            int col = _desc.out_cols[c];
            _col_types[c] = bkgpu_table_col_type(t, col);
            _cols_i[c].resize(_n ? _n : 1);
            _cols_d[c].resize(_n ? _n : 1);
            _cols_n[c].resize(_n ? _n : 1);
            if (_n > 0 &&
                bkgpu_gather(t, col, _rowids.data(), _n, _cols_i[c].data(),
                             _cols_d[c].data(), _cols_n[c].data()) != 0) {
                state->error_msg = bkgpu_last_error();
                return -1;
            }
        }
        _iter = 0;
        return 0;
    }
    int get_next(RuntimeState* state, RowBatch* batch, bool* eos) override {
        while (true) {
            if (state->is_cancelled()) { *eos = true; return 0; }
            if (reached_limit() || _iter >= _n) { *eos = true; return 0; }
            if (batch->is_full()) return 0;
            auto row = std::make_unique<MemRow>(n_slots());
            int s = 0;
            for (int c = 0; c < _desc.n_out_cols; c++, s++) {
                ExprValue v;
                v.type = _col_types[c];
                v.is_null_ = _cols_n[c][_iter] != 0;
                if (!v.is_null_) {
                    if (v.type == BK_DOUBLE) v.d = _cols_d[c][_iter];
                    else v.i = _cols_i[c][_iter];
                }
                row->set_value(s, v);
            }
            for (int f = 0; f < _desc.n_winfns; f++, s++) {
                ExprValue v;
                v.type = _fn_types[f];
                size_t idx = (size_t)f * _n + _iter;
                v.is_null_ = _win_n[idx] != 0;
                if (!v.is_null_) {
                    if (v.type == BK_DOUBLE) v.d = _win_d[idx];
                    else v.i = _win_i[idx];
                }
                row->set_value(s, v);
            }
            batch->move_row(std::move(row));
            _num_rows_returned++;
            _iter++;
        }
    }
    void close(RuntimeState* state) override {
        ExecNode::close(state);
        _rowids.clear(); _win_i.clear(); _win_d.clear(); _win_n.clear();
        _iter = 0; _n = 0;
    }
    int n_slots() const { return _desc.n_out_cols + _desc.n_winfns; }

private:
    BkPlanNodeDesc _desc{};
    std::vector<int64_t> _rowids;
    std::vector<int64_t> _win_i;
    std::vector<double> _win_d;
    std::vector<uint8_t> _win_n;
    std::vector<int32_t> _fn_types;
    std::vector<std::vector<int64_t>> _cols_i;
    std::vector<std::vector<double>> _cols_d;
    std::vector<std::vector<uint8_t>> _cols_n;
    std::vector<int32_t> _col_types;
    int64_t _iter = 0, _n = 0;
};

class LimitNode : public ExecNode {
    int64_t _offset = 0;
    int64_t _num_rows_skipped = 0;
public:
    int init(const BkPlanNodeDesc& node) override {
        ExecNode::init(node);
        _offset = node.offset;
        return 0;
    }
    void close(RuntimeState* state) override {
        ExecNode::close(state);
        _num_rows_skipped = 0;   /* idempotent reset, limit_node.h:16-18 */
    }
    int get_next(RuntimeState* state, RowBatch* batch, bool* eos) override {
        if (_children.empty()) { *eos = true; return 0; }
        if (reached_limit()) { *eos = true; return 0; }
        size_t before = batch->size();
        int ret = _children[0]->get_next(state, batch, eos);
        if (ret < 0) return ret;
        /* OFFSET: drop leading rows until _offset are skipped
         * (limit_node.cpp _num_rows_skipped loop) */
        while (_num_rows_skipped < _offset && batch->size() > before) {
            size_t have = batch->size() - before;
            size_t need = (size_t)(_offset - _num_rows_skipped);
            size_t drop = have < need ? have : need;
            batch->drop_range(before, drop);   /* leading rows of THIS
                                                  child batch only */
            _num_rows_skipped += (int64_t)drop;
            if (batch->size() == before && !*eos) {
                ret = _children[0]->get_next(state, batch, eos);
                if (ret < 0) return ret;
            }
        }
        _num_rows_returned += (int64_t)(batch->size() - before);
        if (_limit > 0 && _num_rows_returned > _limit) {
            /* truncate the overshoot so exactly _limit rows are emitted */
            batch->truncate(batch->size() - (size_t)(_num_rows_returned - _limit));
            _num_rows_returned = _limit;
        }
        if (reached_limit()) *eos = true;
        return 0;
    }
};

/* ---- factory (exec_node.cpp:396-414 switch) ---- */
ExecNode* ExecNode::create_exec_node(const BkPlanNodeDesc& node) {
    switch (node.node_type) {
        case BK_SCAN_NODE:         return new ScanNode();
        case BK_SORT_NODE:         return new SortNode();
        case BK_WINDOW_NODE:       return new WindowNode();
        case BK_AGG_NODE:
        case BK_MERGE_AGG_NODE:    return new AggNode();
        case BK_TABLE_FILTER_NODE:
        case BK_WHERE_FILTER_NODE: return new FilterNode();
        case BK_LIMIT_NODE:        return new LimitNode();
        default:                   return nullptr;
    }
}

int ExecNode::build(const BkPlanNodeDesc* nodes, int n, int* pos, ExecNode** out) {
    if (*pos >= n) return -1;
    const BkPlanNodeDesc& d = nodes[*pos];
    (*pos)++;
    ExecNode* node = create_exec_node(d);
    if (!node) return -1;
    if (node->init(d) != 0) { delete node; return -1; }
    for (int c = 0; c < d.num_children; c++) {
        ExecNode* child = nullptr;
        if (build(nodes, n, pos, &child) != 0) { delete node; return -1; }
        node->add_child(child);
    }
    *out = node;
    return 0;
}

int ExecNode::create_tree(const BkPlanNodeDesc* nodes, int n, ExecNode** root) {
    int pos = 0;
    if (build(nodes, n, &pos, root) != 0) return -1;
    if (pos != n) { delete *root; *root = nullptr; return -1; }
    return 0;
}

}  // namespace bkexec

/* ================= C ABI (include/bk_exec.h) ================= */

struct BkExecTree {
    bkexec::ExecNode* root = nullptr;
    bkexec::RuntimeState state;
    bkexec::RowBatch batch;
    bool opened = false;
    bool batch_eos = false;
    int64_t rows_returned = 0;
    ~BkExecTree() { delete root; }
};

static int tree_n_slots(const BkExecTree* t) {
    using namespace bkexec;
    ExecNode* n = t->root;
    while (n && n->node_type() == BK_LIMIT_NODE && !n->children().empty())
        n = n->children()[0];
    if (!n) return 0;
    if (n->node_type() == BK_AGG_NODE || n->node_type() == BK_MERGE_AGG_NODE)
        return static_cast<AggNode*>(n)->n_slots();
    if (n->node_type() == BK_SORT_NODE)
        return static_cast<SortNode*>(n)->n_slots();
    if (n->node_type() == BK_WINDOW_NODE)
        return static_cast<WindowNode*>(n)->n_slots();
    if (n->node_type() == BK_TABLE_FILTER_NODE ||
        n->node_type() == BK_WHERE_FILTER_NODE)
        return static_cast<FilterNode*>(n)->n_slots();
    return 0;
}

extern "C" BkExecTree* bkexec_create_tree(const BkPlanNodeDesc* nodes, int n_nodes) {
    auto* t = new BkExecTree();
    if (bkexec::ExecNode::create_tree(nodes, n_nodes, &t->root) != 0) {
        delete t;
        return nullptr;
    }
    return t;
}

extern "C" int bkexec_open(BkExecTree* t) {
    int ret = t->root->open(&t->state);
    if (ret == 0) t->opened = true;
    return ret;
}

extern "C" int bkexec_n_slots(const BkExecTree* t) { return tree_n_slots(t); }

extern "C" int64_t bkexec_get_next(BkExecTree* t, int64_t capacity,
                                   int32_t* out_tag, int64_t* out_i,
                                   double* out_d, uint8_t* out_null, int* eos) {
    using namespace bkexec;
    if (!t->opened) return -1;
    int n_slots = tree_n_slots(t);
    /* Region::select_normal's driver loop (region.cpp:3166-3216) */
    RowBatch batch((size_t)capacity);
    bool e = false;
    int ret = t->root->get_next(&t->state, &batch, &e);
    if (ret < 0) return ret;
    int64_t out = 0;
    for (batch.reset(); !batch.is_traverse_over(); batch.next(), out++) {
        MemRow* row = batch.get_row().get();
        for (int s = 0; s < n_slots; s++) {
            const ExprValue& v = row->get_value(s);
            int64_t idx = out * n_slots + s;
            out_tag[idx] = v.type;
            out_null[idx] = v.is_null() ? 1 : 0;
            out_i[idx] = v.i;
            out_d[idx] = v.d;
        }
    }
    t->rows_returned += out;
    *eos = e ? 1 : 0;
    return out;
}

extern "C" int64_t bkexec_num_scan_rows(const BkExecTree* t) {
    return t->state.num_scan_rows();
}
extern "C" int64_t bkexec_num_filter_rows(const BkExecTree* t) {
    return t->state.num_filter_rows();
}
extern "C" int64_t bkexec_num_rows_returned(const BkExecTree* t) {
    return t->rows_returned;
}

extern "C" void bkexec_close(BkExecTree* t) {
    if (t->root) t->root->close(&t->state);  /* idempotent reset */
    delete t;
}

extern "C" int bkexec_dict_word(uint64_t dict_seed, int64_t code, char* out,
                                int cap) {
    return bk_dict_word(dict_seed, code, out, cap);
}

/* SQL LIKE export (include/bk_like.h; LikePredicate::like restated) for
 * binding-level pattern->dict-bitmap compilation and the golden tests.
 * Returns 1/0/-1 (-1 = invalid sequence, the reference's boost::none). */
extern "C" int bkgpu_like_match(const char* target, int64_t tlen,
                                const char* pattern, int64_t plen,
                                int charset, char escape_char) {
    return bk_like_match(target, (size_t)tlen, pattern, (size_t)plen,
                         charset, escape_char);
}
extern "C" int bkgpu_like_one(const char* target, int64_t tlen,
                              const char* pattern, int64_t plen,
                              int charset, char escape_char) {
    return bk_like_one(target, (size_t)tlen, pattern, (size_t)plen,
                       charset, escape_char);
}
