/* oracle.c — CPU restatement of baidu/BaikalDB's OLAP row-engine hot path:
 * the ScanNode -> FilterNode -> AggNode / SortNode pipeline plus the
 * ExprValue/ScalarFnCall/AggFnCall evaluator semantics.
 *
 * *** TEST INFRASTRUCTURE ONLY ***
 * This file is the parity ORACLE and the reported CPU baseline
 * (bench.py cpu_baseline leg). It must never be linked into, imported by, or
 * fallen back to from the product GPU path. Only tests/, __graft_entry__.smoke()
 * and bench.py's cpu_baseline leg may call it.
 *
 * Reference sites restated (file:line against /root/reference):
 *  - filter:      FilterNode::need_copy         src/exec/filter_node.cpp:726-734
 *                 (row kept iff every conjunct is non-NULL and truthy)
 *  - comparisons: operators.cpp:79-105 (typed eq/ne/gt/ge/lt/le, NULL=>NULL)
 *                 with args cast to fn arg types, scalar_fn_call.cpp:219-225
 *  - group key:   ExecNode::encode_exprs_key    src/exec/exec_node.cpp:555-571
 *                 (null-flag byte + MutTableKey::append_value per value,
 *                  mut_table_key.h:113-208, big-endian sign-flipped,
 *                  key_encoder.h:104-173; strings: bytes + '\0')
 *  - hash agg:    AggNode::process_row_batch    src/exec/agg_node.cpp:507-545
 *  - agg fns:     AggFnCall::initialize/update/merge/finalize
 *                 src/expr/agg_fn_call.cpp:370-455, 496-555, 719-830, 927-975
 *                 (COUNT: int64 ++; SUM: starts NULL, first add adopts the
 *                  value's own type then adds in that type (int64 wraps);
 *                  AVG: {double sum, int64 count} pair, agg_fn_call.h:41-47;
 *                  MIN/MAX: ExprValue::compare, expr_value.h:895-945)
 *  - zero-row     AggNode::open                 src/exec/agg_node.cpp:490-505
 *    aggregate:   (no GROUP BY + no rows => single all-initialized row, as the
 *                  db-side merger produces for end-to-end SQL semantics)
 *  - sort/top-N:  SortNode + Sorter/TopNSorter  src/exec/sort_node.cpp:278-440,
 *                 src/runtime/sorter.cpp:18-110, include/runtime/topn_sorter.h:32-63
 *                 (comparator: mem_row_compare.cpp:18-40 — per key: NULLs equal,
 *                  is_null_first decides NULL order regardless of asc/desc;
 *                  ties broken by arrival index, topn_sorter.h:46-54)
 *
 * Parity pinning: the JSON fixtures under tests/golden hold vectors produced
 * by the reference's OWN code compiled in place (oracle/_ref, Makefile target
 * 'ref'): key_encoder.h encode/decode (keyenc_golden.json, covering the
 * expectation classes of test/test_key_encoder.cpp) and datetime.h's packed
 * bit-layout extractions (datetime_golden.json). The reference's
 * test_expr_value.cpp cases are proto-round-trip self-compares and add no
 * cross-implementation signal; ExprValue compare/arith semantics are instead
 * pinned by restatement citations + the numpy cross-checks in tests/.
 *
 * Internally the oracle groups by a fixed-width encoded key (null-flag byte +
 * order-preserving u64 per group column — the same internal key the GPU path
 * uses) and materializes the reference's variable-length MutTableKey byte
 * string only when results are fetched. The mapping fixed-width -> byte key
 * is injective for a fixed group-column type list, so group identity is
 * exactly the reference's.
 */

#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <pthread.h>
#include "../include/bk_common.h"
#include "../include/bk_datagen.h"
#include "../include/bk_keyenc.h"

#define ORC_EXPORT __attribute__((visibility("default")))

/* ------------------------------------------------------------------ */
/* Column container                                                    */
/* ------------------------------------------------------------------ */

typedef struct OrcCol {
    int32_t  type;      /* BkType: BK_INT64, BK_DOUBLE, BK_STRING(dict i32) */
    void*    data;      /* int64_t* / double* / int32_t* */
    uint8_t* valid;     /* NULL => all valid */
} OrcCol;

/* ------------------------------------------------------------------ */
/* Synthetic generation (shared bit-exact generator, bk_datagen.h)     */
/* ------------------------------------------------------------------ */

ORC_EXPORT int orc_generate_column(const BkColSpec* cs, uint64_t seed, uint32_t col,
                                   int64_t row_begin, int64_t row_end,
                                   void* out_data, uint8_t* out_valid) {
    if (cs->col_type == BK_INT64 || cs->col_type == BK_STRING ||
        cs->col_type == BK_DATETIME) {
        for (int64_t r = row_begin; r < row_end; r++) {
            int64_t v = bk_gen_i64(cs, seed, (uint64_t)r, col);
            if (cs->col_type == BK_STRING) ((int32_t*)out_data)[r - row_begin] = (int32_t)v;
            else                           ((int64_t*)out_data)[r - row_begin] = v;
        }
    } else if (cs->col_type == BK_DOUBLE) {
        for (int64_t r = row_begin; r < row_end; r++) {
            ((double*)out_data)[r - row_begin] = bk_gen_f64(cs, seed, (uint64_t)r, col);
        }
    } else {
        return -1;
    }
    if (out_valid) {
        for (int64_t r = row_begin; r < row_end; r++) {
            out_valid[r - row_begin] =
                (uint8_t)bk_cell_valid(seed, (uint64_t)r, col, cs->null_frac_x1e6);
        }
    }
    return 0;
}

/* Deterministic dict word for a code (shared generator, bk_datagen.h). */
ORC_EXPORT int orc_dict_word(uint64_t seed, int64_t code, char* out, int cap) {
    return bk_dict_word(seed, code, out, cap);
}

/* ------------------------------------------------------------------ */
/* Filter (need_copy, filter_node.cpp:726-734)                          */
/* ------------------------------------------------------------------ */

static inline int cell_is_valid(const OrcCol* c, int64_t r) {
    return c->valid == NULL || c->valid[r];
}

static inline int64_t cell_i64(const OrcCol* c, int64_t r) {
    if (c->type == BK_STRING) return (int64_t)((int32_t*)c->data)[r];
    return ((int64_t*)c->data)[r];
}

static inline double cell_f64_cast(const OrcCol* c, int64_t r) {
    /* ExprValue::get_numberic<double> (expr_value.h:341-409): numeric types
     * convert by value. */
    if (c->type == BK_DOUBLE) return ((double*)c->data)[r];
    return (double)cell_i64(c, r);
}

/* postfix expression programs (bk_common.h BkExprOp): restates the same
 * flattening of ScalarFnCall::get_value trees the engine evaluates
 * (scalar_fn_call.cpp:194-225; arg-cast rule 219-225). */
typedef struct { int valid; int64_t i; double d; } OrcPVal;
static OrcPVal orc_eval_prog(const OrcCol* cols, const BkQuerySpec* q,
                             int32_t begin, int32_t len, int64_t r) {
    int64_t si[BK_MAX_PROG_DEPTH];
    double sd[BK_MAX_PROG_DEPTH];
    int sv[BK_MAX_PROG_DEPTH];
    int sp = 0;
    for (int32_t k = 0; k < len; k++) {
        const BkExprOp* e = &q->prog[begin + k];
        switch ((BkProgOp)e->op) {
            case BK_PROG_COL: {
                const OrcCol* c = &cols[e->arg];
                sv[sp] = cell_is_valid(c, r);
                if (c->type == BK_DOUBLE) {
                    sd[sp] = ((double*)c->data)[r];
                    si[sp] = (int64_t)sd[sp];
                } else {
                    si[sp] = cell_i64(c, r);
                    sd[sp] = (double)si[sp];
                }
                sp++;
            } break;
            case BK_PROG_LIT_I:
                si[sp] = e->lit_i;
                sd[sp] = (double)e->lit_i;
                sv[sp] = 1;
                sp++;
                break;
            case BK_PROG_LIT_D:
                sd[sp] = e->lit_d;
                si[sp] = (int64_t)e->lit_d;
                sv[sp] = 1;
                sp++;
                break;
            case BK_PROG_ARITH: {
                sp--;
                int v = sv[sp - 1] && sv[sp];
                if (e->domain == BK_DOUBLE) {
                    double a = sd[sp - 1], b = sd[sp];
                    double o = e->arg == BK_ARITH_ADD   ? a + b
                               : e->arg == BK_ARITH_SUB ? a - b
                                                        : a * b;
                    sd[sp - 1] = o;
                    si[sp - 1] = (int64_t)o;
                } else {
                    uint64_t a = (uint64_t)si[sp - 1], b = (uint64_t)si[sp];
                    int64_t o = (int64_t)(e->arg == BK_ARITH_ADD   ? a + b
                                          : e->arg == BK_ARITH_SUB ? a - b
                                                                   : a * b);
                    si[sp - 1] = o;
                    sd[sp - 1] = (double)o;
                }
                sv[sp - 1] = v;
            } break;
            default:                           /* BK_PROG_FN */
                si[sp - 1] = bk_scalar_fn(e->arg, si[sp - 1]);
                sd[sp - 1] = (double)si[sp - 1];
                break;
        }
    }
    OrcPVal out;
    out.valid = sv[0];
    out.i = si[0];
    out.d = sd[0];
    return out;
}

/* returns 1 iff row passes every clause: standalone conjuncts AND
 * together; or_group members OR within the clause (CNF; see BkConjunct).
 * NULL or false rejects a standalone term; a NULL OR-member is just not
 * true. */
static int orc_row_passes(const OrcCol* cols, const BkQuerySpec* q, int64_t r) {
    uint32_t or_seen = 0, or_sat = 0;
    for (int32_t j = 0; j < q->n_conjuncts; j++) {
        const BkConjunct* cj = &q->conjuncts[j];
        if (cj->prog_len > 0) {
            OrcPVal p = orc_eval_prog(cols, q, cj->prog_begin, cj->prog_len,
                                      r);
            int pass = 0;
            if (p.valid) {
                int cmp = cj->cmp_type == BK_DOUBLE
                              ? ((p.d > cj->lit_d) - (p.d < cj->lit_d))
                              : ((p.i > cj->lit_i) - (p.i < cj->lit_i));
                switch ((BkCmpOp)cj->op) {
                    case BK_OP_EQ: pass = (cmp == 0); break;
                    case BK_OP_NE: pass = (cmp != 0); break;
                    case BK_OP_GT: pass = (cmp > 0);  break;
                    case BK_OP_GE: pass = (cmp >= 0); break;
                    case BK_OP_LT: pass = (cmp < 0);  break;
                    default:       pass = (cmp <= 0); break;
                }
            }
            if (cj->or_group == 0) {
                if (!pass) return 0;
            } else {
                or_seen |= 1u << (cj->or_group & 31);
                if (pass) or_sat |= 1u << (cj->or_group & 31);
            }
            continue;
        }
        const OrcCol* c = &cols[cj->col];
        if (!cell_is_valid(c, r)) {
            if (cj->or_group == 0) return 0;  /* NULL operand => reject */
            or_seen |= 1u << (cj->or_group & 31);
            continue;
        }
        if (cj->arith) {
            /* binary-arith predicate: either operand NULL => NULL */
            const OrcCol* c2 = &cols[cj->col2];
            if (!cell_is_valid(c2, r)) {
                if (cj->or_group == 0) return 0;
                or_seen |= 1u << (cj->or_group & 31);
                continue;
            }
        }
        int cmp; /* sign of (expr - lit) */
        if (cj->cmp_type == BK_DOUBLE) {
            double v = cell_f64_cast(c, r);
            if (cj->arith) {
                double b = cell_f64_cast(&cols[cj->col2], r);
                v = cj->arith == BK_ARITH_ADD ? v + b
                    : cj->arith == BK_ARITH_SUB ? v - b : v * b;
            }
            cmp = (v > cj->lit_d) - (v < cj->lit_d);
        } else { /* BK_INT64 or BK_STRING dict-code compare */
            int64_t v = cell_i64(c, r);
            if (cj->fn) v = bk_scalar_fn(cj->fn, v);
            if (cj->arith) {
                uint64_t ua = (uint64_t)v;
                uint64_t ub = (uint64_t)cell_i64(&cols[cj->col2], r);
                v = (int64_t)(cj->arith == BK_ARITH_ADD ? ua + ub
                    : cj->arith == BK_ARITH_SUB ? ua - ub : ua * ub);
            }
            cmp = (v > cj->lit_i) - (v < cj->lit_i);
        }
        int pass;
        if (cj->op == BK_OP_IN_BITMAP || cj->op == BK_OP_NOT_IN_BITMAP) {
            int64_t v = cell_i64(c, r);
            if (cj->fn) v = bk_scalar_fn(cj->fn, v);
            const uint8_t* bm = (const uint8_t*)(uintptr_t)cj->lit_i;
            int hit = v >= 0 && v < cj->n_in && ((bm[v >> 3] >> (v & 7)) & 1);
            pass = (cj->op == BK_OP_IN_BITMAP) ? hit : !hit;
        } else if (cj->op == BK_OP_IN || cj->op == BK_OP_NOT_IN) {
            /* predicate.h InPredicate semantics over literal lists; big
             * lists (> BK_MAX_INLIST) are a sorted host array in lit_i */
            int64_t v = cell_i64(c, r);
            if (cj->fn) v = bk_scalar_fn(cj->fn, v);
            int found = 0;
            if (cj->n_in <= BK_MAX_INLIST) {
                for (int32_t m = 0; m < cj->n_in; m++)
                    if (cj->in_list[m] == v) { found = 1; break; }
            } else {
                const int64_t* a = (const int64_t*)(uintptr_t)cj->lit_i;
                int32_t lo = 0, hi = cj->n_in - 1;
                while (lo <= hi) {
                    int32_t mid = (lo + hi) >> 1;
                    if (a[mid] == v) { found = 1; break; }
                    if (a[mid] < v) lo = mid + 1; else hi = mid - 1;
                }
            }
            pass = (cj->op == BK_OP_IN) ? found : !found;
        } else {
            switch ((BkCmpOp)cj->op) {
                case BK_OP_EQ: pass = (cmp == 0); break;
                case BK_OP_NE: pass = (cmp != 0); break;
                case BK_OP_GT: pass = (cmp > 0);  break;
                case BK_OP_GE: pass = (cmp >= 0); break;
                case BK_OP_LT: pass = (cmp < 0);  break;
                case BK_OP_LE: pass = (cmp <= 0); break;
                default: pass = 0;
            }
        }
        if (cj->or_group == 0) {
            if (!pass) return 0;
        } else {
            or_seen |= 1u << (cj->or_group & 31);
            if (pass) or_sat |= 1u << (cj->or_group & 31);
        }
    }
    return (or_sat & or_seen) == or_seen;
}

/* ------------------------------------------------------------------ */
/* Hash aggregate                                                      */
/* ------------------------------------------------------------------ */

/* Per-group aggregate state — one entry per agg call.
 * ExprValue-typed per the reference: has==0 <=> intermediate is NULL. */
typedef struct OrcAggState {
    int64_t i;    /* COUNT / SUM(int64) / MIN/MAX(int64|dict) */
    double  d;    /* SUM(double) / AVG sum / MIN/MAX(double) */
    int64_t cnt;  /* AVG count */
    uint8_t has;  /* 0 => NULL intermediate */
} OrcAggState;

typedef struct OrcGroup {
    uint8_t  flag;      /* null-flag byte, exec_node.cpp:556-567 */
    uint64_t e[BK_MAX_GROUP]; /* order-preserving encodes of group values */
    OrcAggState st[BK_MAX_AGGS];
} OrcGroup;

typedef struct OrcMap {
    uint64_t  cap;      /* power of 2 */
    uint64_t  n;
    uint64_t* hashes;   /* 0 = empty (hashes stored |1) */
    OrcGroup* groups;
} OrcMap;

static uint64_t orc_key_hash(uint8_t flag, const uint64_t* e, int ng) {
    uint64_t h = bk_mix64(0x9E1Eull ^ flag);
    for (int i = 0; i < ng; i++) h = bk_mix64(h ^ e[i]);
    return h | 1;
}

static void orc_map_init(OrcMap* m, uint64_t cap) {
    uint64_t c = 64;
    while (c < cap) c <<= 1;
    m->cap = c; m->n = 0;
    m->hashes = (uint64_t*)calloc(c, sizeof(uint64_t));
    m->groups = (OrcGroup*)calloc(c, sizeof(OrcGroup));
}

static void orc_map_free(OrcMap* m) { free(m->hashes); free(m->groups); }

static OrcGroup* orc_map_find_or_insert(OrcMap* m, uint8_t flag,
                                        const uint64_t* e, int ng, int* created);

static void orc_map_grow(OrcMap* m, int ng) {
    OrcMap bigger;
    orc_map_init(&bigger, m->cap * 2);
    for (uint64_t i = 0; i < m->cap; i++) {
        if (!m->hashes[i]) continue;
        int created;
        OrcGroup* g = orc_map_find_or_insert(&bigger, m->groups[i].flag,
                                             m->groups[i].e, ng, &created);
        memcpy(g->st, m->groups[i].st, sizeof(g->st));
    }
    orc_map_free(m);
    *m = bigger;
}

static OrcGroup* orc_map_find_or_insert(OrcMap* m, uint8_t flag,
                                        const uint64_t* e, int ng, int* created) {
    uint64_t h = orc_key_hash(flag, e, ng);
    uint64_t mask = m->cap - 1;
    uint64_t i = h & mask;
    for (;;) {
        if (!m->hashes[i]) {
            m->hashes[i] = h;
            OrcGroup* g = &m->groups[i];
            g->flag = flag;
            for (int k = 0; k < ng; k++) g->e[k] = e[k];
            m->n++;
            *created = 1;
            return g;
        }
        if (m->hashes[i] == h) {
            OrcGroup* g = &m->groups[i];
            int same = (g->flag == flag);
            for (int k = 0; same && k < ng; k++) same = (g->e[k] == e[k]);
            if (same) { *created = 0; return g; }
        }
        i = (i + 1) & mask;
    }
}

/* AggFnCall::update (agg_fn_call.cpp:496-555). in_valid==0 means the input
 * cell is SQL NULL. vtype is the input column's BkType. */
static inline void orc_agg_update(OrcAggState* s, int agg_type, int vtype,
                                  int in_valid, int64_t vi, double vd) {
    switch (agg_type) {
        case BK_AGG_COUNT_STAR:
            s->i++; s->has = 1; return;
        case BK_AGG_COUNT:
            if (in_valid) { s->i++; } s->has = 1; return;
        case BK_AGG_SUM:
            if (!in_valid) return;
            if (vtype == BK_DOUBLE) {
                if (!s->has) { s->d = vd; s->has = 1; }
                else s->d += vd;                      /* expr_value.h:869-871 */
            } else {
                if (!s->has) { s->i = vi; s->has = 1; }
                else s->i = (int64_t)((uint64_t)s->i + (uint64_t)vi); /* int64 wrap */
            }
            return;
        case BK_AGG_AVG:
            if (!in_valid) return;
            s->d += (vtype == BK_DOUBLE) ? vd : (double)vi; /* get_numberic<double> */
            s->cnt++; s->has = 1;
            return;
        case BK_AGG_MIN:
            if (!in_valid) return;
            if (vtype == BK_DOUBLE) {
                if (!s->has || s->d > vd) s->d = vd;
            } else {
                if (!s->has || s->i > vi) s->i = vi;
            }
            s->has = 1; return;
        case BK_AGG_MAX:
            if (!in_valid) return;
            if (vtype == BK_DOUBLE) {
                if (!s->has || s->d < vd) s->d = vd;
            } else {
                if (!s->has || s->i < vi) s->i = vi;
            }
            s->has = 1; return;
        default: return;
    }
}

/* AggFnCall::merge (agg_fn_call.cpp:781-830): partials combine. */
static inline void orc_agg_merge(OrcAggState* dst, const OrcAggState* src,
                                 int agg_type, int vtype) {
    switch (agg_type) {
        case BK_AGG_COUNT_STAR:
        case BK_AGG_COUNT:
        case BK_AGG_COUNT_DISTINCT:
            if (src->has) { dst->i += src->i; dst->has = 1; } return;
        case BK_AGG_SUM_DISTINCT:
        case BK_AGG_SUM:
            if (!src->has) return;
            if (vtype == BK_DOUBLE) {
                if (!dst->has) { dst->d = src->d; dst->has = 1; }
                else dst->d += src->d;
            } else {
                if (!dst->has) { dst->i = src->i; dst->has = 1; }
                else dst->i = (int64_t)((uint64_t)dst->i + (uint64_t)src->i);
            }
            return;
        case BK_AGG_AVG:
        case BK_AGG_AVG_DISTINCT:
            if (!src->has) return;
            dst->d += src->d; dst->cnt += src->cnt; dst->has = 1; return;
        case BK_AGG_MIN:
            if (!src->has) return;
            if (vtype == BK_DOUBLE) { if (!dst->has || dst->d > src->d) dst->d = src->d; }
            else                    { if (!dst->has || dst->i > src->i) dst->i = src->i; }
            dst->has = 1; return;
        case BK_AGG_MAX:
            if (!src->has) return;
            if (vtype == BK_DOUBLE) { if (!dst->has || dst->d < src->d) dst->d = src->d; }
            else                    { if (!dst->has || dst->i < src->i) dst->i = src->i; }
            dst->has = 1; return;
        default: return;
    }
}

/* encode one group-column value to its order-preserving u64 (same internal
 * key the GPU uses). */
static inline uint64_t orc_enc_group(const OrcCol* c, int64_t r, int32_t fn) {
    if (fn) return bk_enc_i64(bk_scalar_fn(fn, cell_i64(c, r)));
    switch (c->type) {
        case BK_INT64:
        case BK_DATETIME: return bk_enc_i64(((int64_t*)c->data)[r]);
        case BK_DOUBLE: return bk_enc_f64(((double*)c->data)[r]);
        case BK_STRING: return (uint64_t)(uint32_t)((int32_t*)c->data)[r];
        default:        return 0;
    }
}

/* ------------------------------------------------------------------ */
/* filter+aggregate over a row range (one "region"'s work)             */
/* ------------------------------------------------------------------ */

typedef struct OrcAggTask {
    const OrcCol*      cols;
    const BkQuerySpec* q;
    int64_t            row_begin, row_end;
    OrcMap             map;
    int64_t            rows_passed;
} OrcAggTask;

static void* orc_agg_worker(void* arg) {
    OrcAggTask* t = (OrcAggTask*)arg;
    const BkQuerySpec* q = t->q;
    const OrcCol* cols = t->cols;
    orc_map_init(&t->map, 1024);
    int ng = q->n_group;
    for (int64_t r = t->row_begin; r < t->row_end; r++) {
        if (!orc_row_passes(cols, q, r)) continue;
        t->rows_passed++;
        uint8_t flag = 0;
        uint64_t e[BK_MAX_GROUP] = {0, 0};
        for (int k = 0; k < ng; k++) {
            const OrcCol* c = &cols[q->group_cols[k]];
            if (!cell_is_valid(c, r)) flag |= (uint8_t)(0x01u << (7 - k)); /* exec_node.cpp:561 */
            else e[k] = orc_enc_group(c, r, q->group_fns[k]);
        }
        int created;
        if (t->map.n * 10 >= t->map.cap * 6) orc_map_grow(&t->map, ng);
        OrcGroup* g = orc_map_find_or_insert(&t->map, flag, e, ng, &created);
        for (int32_t a = 0; a < q->n_aggs; a++) {
            const BkAggSpec* as = &q->aggs[a];
            int vtype = q->agg_in_types[a];
            int in_valid = 1;
            int64_t vi = 0; double vd = 0.0;
            if (as->prog_len > 0) {
                /* postfix expression input */
                OrcPVal p = orc_eval_prog(cols, q, as->prog_begin,
                                          as->prog_len, r);
                in_valid = p.valid;
                vi = p.i;
                vd = p.d;
            } else if (as->col >= 0) {
                const OrcCol* c = &cols[as->col];
                in_valid = cell_is_valid(c, r);
                if (as->arith) {
                    /* expression input (col ARITH col2): NULL if either
                     * operand NULL; computed in the agg_in_type domain */
                    const OrcCol* c2 = &cols[as->col2];
                    in_valid = in_valid && cell_is_valid(c2, r);
                    if (in_valid) {
                        if (vtype == BK_DOUBLE) {
                            double av = cell_f64_cast(c, r);
                            double bv = cell_f64_cast(c2, r);
                            vd = as->arith == BK_ARITH_ADD ? av + bv
                                 : as->arith == BK_ARITH_SUB ? av - bv
                                 : av * bv;
                        } else {
                            uint64_t av = (uint64_t)cell_i64(c, r);
                            uint64_t bv = (uint64_t)cell_i64(c2, r);
                            vi = (int64_t)(as->arith == BK_ARITH_ADD ? av + bv
                                 : as->arith == BK_ARITH_SUB ? av - bv
                                 : av * bv);
                        }
                    }
                } else if (in_valid) {
                    if (c->type == BK_DOUBLE) vd = ((double*)c->data)[r];
                    else vi = cell_i64(c, r);
                }
            }
            orc_agg_update(&g->st[a], as->agg_type, vtype, in_valid, vi, vd);
        }
    }
    return NULL;
}

/* ------------------------------------------------------------------ */
/* Result materialization                                              */
/* ------------------------------------------------------------------ */

typedef struct OrcAggResult {
    int64_t  ngroups;
    int64_t  rows_passed;  /* rows surviving the filter (RuntimeState counters) */
    /* reference MutTableKey byte strings, concatenated */
    uint8_t* key_bytes;
    int64_t* key_off;      /* ngroups+1 offsets */
    /* per agg a, per group g: value at [a*ngroups + g] */
    int64_t* out_i;        /* int64-typed outputs (COUNT, SUM int64, MIN/MAX int64) */
    double*  out_d;        /* double-typed outputs (SUM double, AVG, MIN/MAX double) */
    uint8_t* out_has;      /* 0 => SQL NULL output */
    /* raw group-key components for programmatic checks */
    uint8_t*  g_flag;
    uint64_t* g_enc;       /* [g*BK_MAX_GROUP + k] */
} OrcAggResult;

typedef struct KeySortRef { const OrcGroup* g; } KeySortRef;

static int s_sort_ng;
static int orc_group_cmp(const void* a, const void* b) {
    const OrcGroup* ga = ((const KeySortRef*)a)->g;
    const OrcGroup* gb = ((const KeySortRef*)b)->g;
    /* order by the reference byte-key memcmp order == (flag, then per present
     * column big-endian bytes). Comparing (flag, e[k] with nulls as omitted)
     * — since both keys share the type list, compare flag first then encoded
     * values; a null column (bit set in flag) contributes nothing. */
    if (ga->flag != gb->flag) return ga->flag < gb->flag ? -1 : 1;
    for (int k = 0; k < s_sort_ng; k++) {
        int a_null = (ga->flag >> (7 - k)) & 1;
        if (a_null) continue;
        if (ga->e[k] != gb->e[k]) return ga->e[k] < gb->e[k] ? -1 : 1;
    }
    return 0;
}

/* Build the reference MutTableKey byte string for one group
 * (exec_node.cpp:555-571 + mut_table_key.h append_value). */
static int64_t orc_build_key_bytes(const OrcGroup* g, const BkQuerySpec* q,
                                   uint64_t dict_seed, uint8_t* out /* may be NULL */) {
    int64_t len = 0;
    if (out) out[0] = g->flag;
    len = 1;
    for (int k = 0; k < q->n_group; k++) {
        if ((g->flag >> (7 - k)) & 1) continue;  /* null values omitted */
        int t = q->group_types[k];
        if (t == BK_INT64 || t == BK_DOUBLE || t == BK_DATETIME) {
            if (out) {
                uint64_t be = bk_bswap64(g->e[k]);  /* already sign-flip encoded */
                memcpy(out + len, &be, 8);
            }
            len += 8;
        } else if (t == BK_STRING) {
            char word[64];
            int wl = orc_dict_word(dict_seed, (int64_t)g->e[k], word, sizeof word);
            if (out) { memcpy(out + len, word, (size_t)wl); out[len + wl] = 0; }
            len += wl + 1;  /* append_string: bytes + '\0', mut_table_key.h:166-169 */
        }
    }
    return len;
}

static OrcAggResult* orc_materialize_parts(OrcMap* parts, int nparts,
                                           const BkQuerySpec* q,
                                           int64_t rows_passed,
                                           uint64_t dict_seed, int sort_keys);

/* COUNT/SUM(DISTINCT d) via the reference's planner rewrite
 * (agg_node.cpp:247-258): level 1 = filter+aggregate grouped by (user group
 * keys + d) — each surviving group is one deduped (g, d) pair — then fold
 * into the level-2 result keyed by the user keys alone. Mirrors
 * bkgpu_agg_rollup: src_idx[a] >= 0 merges level-1 agg state a additively,
 * src_idx[a] < 0 synthesizes COUNT_DISTINCT/SUM_DISTINCT from the dedup key. */
ORC_EXPORT OrcAggResult* orc_filter_agg_distinct(
        const OrcCol* cols, int ncols, const BkQuerySpec* q1,
        const BkQuerySpec* q2, const int32_t* src_idx,
        int64_t row_begin, int64_t row_end, int nthreads,
        uint64_t dict_seed, int sort_keys) {
    (void)ncols;
    if (q1->n_group != q2->n_group + 1 || q2->n_group > 2) return NULL;
    int ng2 = q2->n_group;
    if (nthreads < 1) nthreads = 1;
    if (nthreads > 128) nthreads = 128;
    int64_t n = row_end - row_begin;
    if ((int64_t)nthreads > n && n > 0) nthreads = (int)n;
    if (n <= 0) nthreads = 1;
    OrcAggTask* tasks = (OrcAggTask*)calloc((size_t)nthreads, sizeof(OrcAggTask));
    pthread_t* tids = (pthread_t*)calloc((size_t)nthreads, sizeof(pthread_t));
    int64_t chunk = (n + nthreads - 1) / nthreads;
    for (int t = 0; t < nthreads; t++) {
        tasks[t].cols = cols; tasks[t].q = q1;
        tasks[t].row_begin = row_begin + (int64_t)t * chunk;
        tasks[t].row_end = tasks[t].row_begin + chunk;
        if (tasks[t].row_end > row_end) tasks[t].row_end = row_end;
        if (tasks[t].row_begin > row_end) tasks[t].row_begin = row_end;
        if (nthreads == 1) orc_agg_worker(&tasks[t]);
        else pthread_create(&tids[t], NULL, orc_agg_worker, &tasks[t]);
    }
    if (nthreads > 1)
        for (int t = 0; t < nthreads; t++) pthread_join(tids[t], NULL);
    int64_t rows_passed = 0;
    for (int t = 0; t < nthreads; t++) rows_passed += tasks[t].rows_passed;

    /* serial level-1 merge (MERGE_AGG over (g,d)) */
    OrcMap l1;
    orc_map_init(&l1, 4096);
    for (int t = 0; t < nthreads; t++) {
        OrcMap* mt = &tasks[t].map;
        for (uint64_t i = 0; i < mt->cap; i++) {
            if (!mt->hashes[i]) continue;
            int created;
            if (l1.n * 10 >= l1.cap * 6) orc_map_grow(&l1, q1->n_group);
            OrcGroup* g = orc_map_find_or_insert(&l1, mt->groups[i].flag,
                                                 mt->groups[i].e, q1->n_group,
                                                 &created);
            for (int32_t a = 0; a < q1->n_aggs; a++)
                orc_agg_merge(&g->st[a], &mt->groups[i].st[a],
                              q1->aggs[a].agg_type, q1->agg_in_types[a]);
        }
        orc_map_free(mt);
    }
    free(tasks); free(tids);

    /* rollup */
    OrcMap out;
    orc_map_init(&out, 1024);
    for (uint64_t i = 0; i < l1.cap; i++) {
        if (!l1.hashes[i]) continue;
        OrcGroup* gi = &l1.groups[i];
        uint8_t f2 = (uint8_t)(gi->flag & ((0xFFu << (8 - ng2)) & 0xFFu));
        int d_null = (gi->flag >> (7 - ng2)) & 1;
        uint64_t e_d = gi->e[ng2];
        uint64_t e2[BK_MAX_GROUP] = {0, 0};
        for (int k = 0; k < ng2; k++) e2[k] = gi->e[k];
        int created;
        if (out.n * 10 >= out.cap * 6) orc_map_grow(&out, ng2);
        OrcGroup* g2 = orc_map_find_or_insert(&out, f2, e2, ng2, &created);
        for (int32_t a = 0; a < q2->n_aggs; a++) {
            int at = q2->aggs[a].agg_type;
            int vt = q2->agg_in_types[a];
            if (at == BK_AGG_COUNT_DISTINCT) {
                orc_agg_update(&g2->st[a], BK_AGG_COUNT, vt, !d_null, 0, 0);
            } else if (at == BK_AGG_SUM_DISTINCT || at == BK_AGG_AVG_DISTINCT) {
                if (!d_null) {
                    int64_t vi = 0; double vd = 0.0;
                    if (vt == BK_DOUBLE) vd = bk_dec_f64(e_d);
                    else vi = bk_dec_i64(e_d);
                    orc_agg_update(&g2->st[a],
                                   at == BK_AGG_AVG_DISTINCT ? BK_AGG_AVG
                                                             : BK_AGG_SUM,
                                   vt, 1, vi, vd);
                }
            } else {
                orc_agg_merge(&g2->st[a], &gi->st[src_idx[a]], at, vt);
            }
        }
    }
    orc_map_free(&l1);

    /* zero-row, no-GROUP-BY => one all-initialized row (agg_node.cpp:490-505) */
    if (out.n == 0 && ng2 == 0) {
        int created;
        uint64_t e[BK_MAX_GROUP] = {0, 0};
        OrcGroup* g = orc_map_find_or_insert(&out, 0, e, 0, &created);
        for (int32_t a = 0; a < q2->n_aggs; a++) {
            int at = q2->aggs[a].agg_type;
            if (at == BK_AGG_COUNT_STAR || at == BK_AGG_COUNT ||
                at == BK_AGG_COUNT_DISTINCT)
                g->st[a].has = 1;
        }
    }

    OrcAggResult* res = orc_materialize_parts(&out, 1, q2, rows_passed,
                                              dict_seed, sort_keys);
    orc_map_free(&out);
    return res;
}

ORC_EXPORT void orc_agg_result_free(OrcAggResult* res) {
    if (!res) return;
    free(res->key_bytes); free(res->key_off);
    free(res->out_i); free(res->out_d); free(res->out_has);
    free(res->g_flag); free(res->g_enc);
    free(res);
}

/* Finalize (agg_fn_call.cpp:927-975) one group's agg into output arrays. */
static void orc_finalize_group(const OrcGroup* g, const BkQuerySpec* q,
                               int64_t gi, int64_t ngroups, OrcAggResult* res) {
    for (int32_t a = 0; a < q->n_aggs; a++) {
        const OrcAggState* s = &g->st[a];
        int64_t idx = (int64_t)a * ngroups + gi;
        int at = q->aggs[a].agg_type;
        int vtype = q->agg_in_types[a];
        switch (at) {
            case BK_AGG_COUNT_STAR:
            case BK_AGG_COUNT:
            case BK_AGG_COUNT_DISTINCT:  /* COUNT-shaped state */
                res->out_i[idx] = s->i; res->out_has[idx] = 1; break;
            case BK_AGG_SUM_DISTINCT:    /* SUM-shaped state */
            case BK_AGG_SUM:
                if (!s->has) { res->out_has[idx] = 0; break; }
                if (vtype == BK_DOUBLE) res->out_d[idx] = s->d;
                else res->out_i[idx] = s->i;
                res->out_has[idx] = 1; break;
            case BK_AGG_AVG:
            case BK_AGG_AVG_DISTINCT:
                if (!s->has || s->cnt == 0) { res->out_has[idx] = 0; break; }
                res->out_d[idx] = s->d / (double)s->cnt;   /* agg_fn_call.cpp:958 */
                res->out_has[idx] = 1; break;
            case BK_AGG_MIN:
            case BK_AGG_MAX:
                if (!s->has) { res->out_has[idx] = 0; break; }
                if (vtype == BK_DOUBLE) res->out_d[idx] = s->d;
                else res->out_i[idx] = s->i;
                res->out_has[idx] = 1; break;
            default: res->out_has[idx] = 0;
        }
    }
}

typedef struct OrcMergeTask {
    OrcAggTask* tasks;
    int ntasks;
    const BkQuerySpec* q;
    int part, nparts;
    OrcMap out;
} OrcMergeTask;

static void* orc_merge_worker(void* arg) {
    OrcMergeTask* m = (OrcMergeTask*)arg;
    const BkQuerySpec* q = m->q;
    orc_map_init(&m->out, 1024);
    for (int t = 0; t < m->ntasks; t++) {
        OrcMap* mt = &m->tasks[t].map;
        for (uint64_t i = 0; i < mt->cap; i++) {
            if (!mt->hashes[i]) continue;
            if ((int)(mt->hashes[i] % (uint64_t)m->nparts) != m->part) continue;
            int created;
            if (m->out.n * 10 >= m->out.cap * 6) orc_map_grow(&m->out, q->n_group);
            OrcGroup* g = orc_map_find_or_insert(&m->out, mt->groups[i].flag,
                                                 mt->groups[i].e, q->n_group,
                                                 &created);
            for (int32_t a = 0; a < q->n_aggs; a++)
                orc_agg_merge(&g->st[a], &mt->groups[i].st[a],
                              q->aggs[a].agg_type, q->agg_in_types[a]);
        }
    }
    return NULL;
}

/* collect groups from `parts`, canonical-sort, build the result arrays.
 * Does NOT free parts. */
static OrcAggResult* orc_materialize_parts(OrcMap* parts, int nparts,
                                           const BkQuerySpec* q,
                                           int64_t rows_passed,
                                           uint64_t dict_seed, int sort_keys) {
    uint64_t total_groups = 0;
    for (int t = 0; t < nparts; t++) total_groups += parts[t].n;
    int64_t ngroups = (int64_t)total_groups;
    KeySortRef* refs = (KeySortRef*)malloc((size_t)(ngroups > 0 ? ngroups : 1)
                                           * sizeof(KeySortRef));
    int64_t gi = 0;
    for (int t = 0; t < nparts; t++)
        for (uint64_t i = 0; i < parts[t].cap; i++)
            if (parts[t].hashes[i]) refs[gi++].g = &parts[t].groups[i];
    if (sort_keys) {
        s_sort_ng = q->n_group;
        qsort(refs, (size_t)ngroups, sizeof(KeySortRef), orc_group_cmp);
    }

    OrcAggResult* res = (OrcAggResult*)calloc(1, sizeof(OrcAggResult));
    res->ngroups = ngroups;
    res->rows_passed = rows_passed;
    res->key_off = (int64_t*)malloc((size_t)(ngroups + 1) * sizeof(int64_t));
    int64_t total = 0;
    for (int64_t g2 = 0; g2 < ngroups; g2++) {
        res->key_off[g2] = total;
        total += orc_build_key_bytes(refs[g2].g, q, dict_seed, NULL);
    }
    res->key_off[ngroups] = total;
    res->key_bytes = (uint8_t*)malloc((size_t)(total > 0 ? total : 1));
    int64_t na = (int64_t)q->n_aggs * (ngroups > 0 ? ngroups : 1);
    res->out_i = (int64_t*)calloc((size_t)na, sizeof(int64_t));
    res->out_d = (double*)calloc((size_t)na, sizeof(double));
    res->out_has = (uint8_t*)calloc((size_t)na, 1);
    res->g_flag = (uint8_t*)calloc((size_t)(ngroups > 0 ? ngroups : 1), 1);
    res->g_enc = (uint64_t*)calloc((size_t)(ngroups > 0 ? ngroups : 1) * BK_MAX_GROUP,
                                   sizeof(uint64_t));
    for (int64_t g2 = 0; g2 < ngroups; g2++) {
        orc_build_key_bytes(refs[g2].g, q, dict_seed, res->key_bytes + res->key_off[g2]);
        orc_finalize_group(refs[g2].g, q, g2, ngroups, res);
        res->g_flag[g2] = refs[g2].g->flag;
        for (int k = 0; k < BK_MAX_GROUP; k++)
            res->g_enc[g2 * BK_MAX_GROUP + k] = refs[g2].g->e[k];
    }
    free(refs);
    return res;
}

/* Run filter+aggregate over [row_begin,row_end) with nthreads shards
 * (each shard mirrors one region's AggNode, merged like MERGE_AGG —
 * agg_node.cpp:29,539-543). Results sorted by reference key bytes iff
 * sort_keys != 0. */
ORC_EXPORT OrcAggResult* orc_filter_agg(const OrcCol* cols, int ncols,
                                        const BkQuerySpec* q,
                                        int64_t row_begin, int64_t row_end,
                                        int nthreads, uint64_t dict_seed,
                                        int sort_keys) {
    (void)ncols;
    if (nthreads < 1) nthreads = 1;
    int64_t n = row_end - row_begin;
    if (nthreads > 128) nthreads = 128;
    if ((int64_t)nthreads > n && n > 0) nthreads = (int)n;
    if (n <= 0) nthreads = 1;

    OrcAggTask* tasks = (OrcAggTask*)calloc((size_t)nthreads, sizeof(OrcAggTask));
    pthread_t* tids = (pthread_t*)calloc((size_t)nthreads, sizeof(pthread_t));
    int64_t chunk = nthreads > 0 ? (n + nthreads - 1) / nthreads : 0;
    for (int t = 0; t < nthreads; t++) {
        tasks[t].cols = cols; tasks[t].q = q;
        tasks[t].row_begin = row_begin + (int64_t)t * chunk;
        tasks[t].row_end = tasks[t].row_begin + chunk;
        if (tasks[t].row_end > row_end) tasks[t].row_end = row_end;
        if (tasks[t].row_begin > row_end) tasks[t].row_begin = row_end;
        if (nthreads == 1) orc_agg_worker(&tasks[t]);
        else pthread_create(&tids[t], NULL, orc_agg_worker, &tasks[t]);
    }
    if (nthreads > 1)
        for (int t = 0; t < nthreads; t++) pthread_join(tids[t], NULL);

    /* merge partials (MERGE_AGG path, agg_node.cpp:539-543) — parallel by
     * key-hash partition: merge thread t owns groups with hash%T == t, so
     * the combine scales with cores instead of serializing. */
    int64_t rows_passed = 0;
    for (int t = 0; t < nthreads; t++) rows_passed += tasks[t].rows_passed;
    int nmerge = nthreads;
    OrcMergeTask* mtasks = (OrcMergeTask*)calloc((size_t)nmerge, sizeof(OrcMergeTask));
    pthread_t* mtids = (pthread_t*)calloc((size_t)nmerge, sizeof(pthread_t));
    for (int t = 0; t < nmerge; t++) {
        mtasks[t].tasks = tasks;
        mtasks[t].ntasks = nthreads;
        mtasks[t].q = q;
        mtasks[t].part = t;
        mtasks[t].nparts = nmerge;
        if (nmerge == 1) orc_merge_worker(&mtasks[t]);
        else pthread_create(&mtids[t], NULL, orc_merge_worker, &mtasks[t]);
    }
    if (nmerge > 1)
        for (int t = 0; t < nmerge; t++) pthread_join(mtids[t], NULL);
    for (int t = 0; t < nthreads; t++) orc_map_free(&tasks[t].map);
    /* steal the merged maps into one logical view */
    OrcMap* parts = (OrcMap*)calloc((size_t)nmerge, sizeof(OrcMap));
    for (int t = 0; t < nmerge; t++) parts[t] = mtasks[t].out;
    free(mtasks); free(mtids);
    OrcMap* m0 = &parts[0];  /* used only for the zero-row special case */

    /* zero-row, no-GROUP-BY query => one all-initialized row
     * (agg_node.cpp:490-505; COUNT 0, SUM/AVG/MIN/MAX NULL) */
    uint64_t total_groups = 0;
    for (int t = 0; t < nmerge; t++) total_groups += parts[t].n;
    if (total_groups == 0 && q->n_group == 0) {
        int created;
        uint64_t e[BK_MAX_GROUP] = {0, 0};
        OrcGroup* g = orc_map_find_or_insert(m0, 0, e, 0, &created);
        for (int32_t a = 0; a < q->n_aggs; a++) {
            int at = q->aggs[a].agg_type;
            if (at == BK_AGG_COUNT_STAR || at == BK_AGG_COUNT) g->st[a].has = 1;
        }
    }

    OrcAggResult* res = orc_materialize_parts(parts, nmerge, q, rows_passed,
                                              dict_seed, sort_keys);
    for (int t = 0; t < nmerge; t++) orc_map_free(&parts[t]);
    free(parts);
    free(tasks); free(tids);
    return res;
}

/* ------------------------------------------------------------------ */
/* ORDER BY ... LIMIT top-N (TopNSorter semantics)                     */
/* ------------------------------------------------------------------ */

/* comparator per mem_row_compare.cpp:18-40 + arrival-index tie-break
 * (topn_sorter.h:46-54). Returns <0 if a orders before b. */
static int orc_row_order_cmp(const OrcCol* cols, const BkOrderSpec* order, int norder,
                             int64_t ra, int64_t rb) {
    for (int i = 0; i < norder; i++) {
        const OrcCol* c = &cols[order[i].col];
        int va = cell_is_valid(c, ra), vb = cell_is_valid(c, rb);
        if (!va && !vb) continue;
        if (!va) return order[i].is_null_first ? -1 : 1;
        if (!vb) return order[i].is_null_first ? 1 : -1;
        int cmp;
        if (c->type == BK_DOUBLE) {
            double x = ((double*)c->data)[ra], y = ((double*)c->data)[rb];
            cmp = (x > y) - (x < y);
        } else {
            int64_t x = cell_i64(c, ra), y = cell_i64(c, rb);
            cmp = (x > y) - (x < y);
        }
        if (cmp != 0) return order[i].is_asc ? cmp : -cmp;
    }
    return (ra > rb) - (ra < rb);  /* stable by arrival order */
}

typedef struct OrcTopCtx {
    const OrcCol* cols; const BkOrderSpec* order; int norder;
} OrcTopCtx;
static OrcTopCtx s_top_ctx;
static int orc_top_qsort_cmp(const void* a, const void* b) {
    return orc_row_order_cmp(s_top_ctx.cols, s_top_ctx.order, s_top_ctx.norder,
                             *(const int64_t*)a, *(const int64_t*)b);
}

/* max-heap of the current top-N (root = worst kept row) */
static void orc_heap_siftdown(int64_t* h, int64_t n, int64_t i, const OrcTopCtx* c) {
    for (;;) {
        int64_t l = 2 * i + 1, r = l + 1, m = i;
        if (l < n && orc_row_order_cmp(c->cols, c->order, c->norder, h[l], h[m]) > 0) m = l;
        if (r < n && orc_row_order_cmp(c->cols, c->order, c->norder, h[r], h[m]) > 0) m = r;
        if (m == i) return;
        int64_t t = h[i]; h[i] = h[m]; h[m] = t;
        i = m;
    }
}

/* Select + order the top `limit` rows of [row_begin,row_end) that pass the
 * filter (SortNode drains FilterNode, sort_node.cpp:278-347). Returns number
 * written to out_rows (global row indices, in final output order). */
ORC_EXPORT int64_t orc_sort_topk(const OrcCol* cols, int ncols, const BkQuerySpec* q,
                                 const BkOrderSpec* order, int norder,
                                 int64_t row_begin, int64_t row_end,
                                 int64_t limit, int64_t* out_rows) {
    (void)ncols;
    OrcTopCtx ctx = { cols, order, norder };
    int64_t* heap = (int64_t*)malloc((size_t)limit * sizeof(int64_t));
    int64_t hn = 0;
    for (int64_t r = row_begin; r < row_end; r++) {
        if (q && q->n_conjuncts > 0 && !orc_row_passes(cols, q, r)) continue;
        if (hn < limit) {
            heap[hn++] = r;
            if (hn == limit)  /* heapify */
                for (int64_t i = hn / 2 - 1; i >= 0; i--) orc_heap_siftdown(heap, hn, i, &ctx);
        } else if (limit > 0 &&
                   orc_row_order_cmp(cols, order, norder, r, heap[0]) < 0) {
            heap[0] = r;
            orc_heap_siftdown(heap, hn, 0, &ctx);
        }
    }
    /* hn < limit => never heapified: contents are in arrival order, which
     * qsort below handles the same way. */
    s_top_ctx = ctx;
    qsort(heap, (size_t)hn, sizeof(int64_t), orc_top_qsort_cmp);
    memcpy(out_rows, heap, (size_t)hn * sizeof(int64_t));
    free(heap);
    return hn;
}

/* ------------------------------------------------------------------ */
/* WindowNode, NON-FRAME mode (window_node.cpp:39-41: every fn sees its
 * whole partition; fn semantics window_fn_call.cpp:364-700).
 * Rows sorted by (partition asc nulls-first, order keys, arrival), then
 * partitions walked start-to-end. Mirrors bkgpu_window.                */
/* ------------------------------------------------------------------ */

static int orc_cell_eq(const OrcCol* c, int64_t ra, int64_t rb) {
    int va = cell_is_valid(c, ra), vb = cell_is_valid(c, rb);
    if (va != vb) return 0;
    if (!va) return 1;   /* NULLs equal (mem_row_compare.cpp:18-40) */
    if (c->type == BK_DOUBLE)
        return ((double*)c->data)[ra] == ((double*)c->data)[rb];
    return cell_i64(c, ra) == cell_i64(c, rb);
}

static void orc_win_value(const OrcCol* c, int64_t r, int64_t idx,
                          int64_t* out_i, double* out_d, uint8_t* out_null) {
    if (!cell_is_valid(c, r)) { out_null[idx] = 1; return; }
    out_null[idx] = 0;
    if (c->type == BK_DOUBLE) out_d[idx] = ((double*)c->data)[r];
    else out_i[idx] = cell_i64(c, r);
}

ORC_EXPORT int64_t orc_window_multi(const OrcCol* cols, int ncols,
                              const BkQuerySpec* q,
                              const int32_t* part_cols, int32_t n_part,
                              const BkOrderSpec* order, int norder,
                              const BkWindowFn* fns, int nfns,
                              int32_t frame_rows, int64_t f_pre, int64_t f_fol,
                              int64_t row_begin, int64_t row_end,
                              int64_t* out_rows, int64_t* out_i,
                              double* out_d, uint8_t* out_null) {
    BkOrderSpec full[4];
    int nf = 0;
    for (int k = 0; k < n_part && nf < 4; k++) {
        full[nf].col = part_cols[k]; full[nf].is_asc = 1;
        full[nf].is_null_first = 1; nf++;
    }
    for (int k = 0; k < norder && nf < 4; k++) full[nf++] = order[k];
    int64_t n = orc_sort_topk(cols, ncols, q, full, nf, row_begin, row_end,
                              row_end - row_begin, out_rows);
    if (n <= 0) return n;

    int64_t ps = 0;
    while (ps < n) {
        int64_t pe = ps + 1;
        for (; pe < n; pe++) {
            int eq = 1;
            for (int k = 0; k < n_part && eq; k++)
                eq = orc_cell_eq(&cols[part_cols[k]], out_rows[pe],
                                 out_rows[pe - 1]);
            if (!eq) break;
        }
        int64_t pn = pe - ps;
        /* per-partition aggregate states */
        for (int f = 0; f < nfns; f++) {
            int ft = fns[f].fn_type;
            if (ft > BK_WIN_MAX) continue;
            if (frame_rows && ft <= BK_WIN_MAX) {
                /* ROWS frame (RowFrameWindowProcessor): per-row recompute
                 * over [fl, fr] clamped inside the partition */
                const OrcCol* fc = fns[f].col >= 0 ? &cols[fns[f].col] : NULL;
                int64_t rgl = ps, rgr = ps - 1;  /* current peer range */
                for (int64_t j = ps; j < pe; j++) {
                    if (j > rgr) {   /* advance the peer group for RANGE */
                        rgl = j; rgr = j;
                        while (rgr + 1 < pe) {
                            int same = 1;
                            for (int k = 0; k < norder; k++)
                                if (!orc_cell_eq(&cols[order[k].col],
                                                 out_rows[rgr + 1],
                                                 out_rows[j]))
                                    { same = 0; break; }
                            if (!same) break;
                            rgr++;
                        }
                    }
                    int64_t fl, fr;
                    if (frame_rows == 2) { fl = ps; fr = rgr; }
                    else if (frame_rows == 3) { fl = rgl; fr = pe - 1; }
                    else if (frame_rows == 4) {
                        /* RANGE by VALUE over the single ASC int order key */
                        const OrcCol* okc = &cols[order[0].col];
                        if (!cell_is_valid(okc, out_rows[j])) {
                            fl = rgl; fr = rgr;   /* null peers */
                        } else {
                            int64_t v = cell_i64(okc, out_rows[j]);
                            fl = j; fr = j;
                            while (fl > ps && cell_is_valid(okc, out_rows[fl - 1]) &&
                                   (f_pre < 0 ||
                                    cell_i64(okc, out_rows[fl - 1]) >= v - f_pre))
                                fl--;
                            while (fr + 1 < pe &&
                                   cell_is_valid(okc, out_rows[fr + 1]) &&
                                   (f_fol < 0 ||
                                    cell_i64(okc, out_rows[fr + 1]) <= v + f_fol))
                                fr++;
                        }
                    }
                    else {
                        fl = f_pre >= 0 && j - f_pre > ps ? j - f_pre : ps;
                        fr = f_fol >= 0 && j + f_fol < pe - 1 ? j + f_fol
                                                              : pe - 1;
                    }
                    int64_t idx = (int64_t)f * n + j;
                    out_i[idx] = 0; out_d[idx] = 0.0; out_null[idx] = 0;
                    int64_t fcnt = 0, fvi = 0;
                    double fvd = 0.0;
                    for (int64_t jj = fl; jj <= fr; jj++) {
                        int64_t r = out_rows[jj];
                        if (ft == BK_WIN_COUNT_STAR) { fcnt++; continue; }
                        if (!cell_is_valid(fc, r)) continue;
                        if (ft == BK_WIN_SUM) {
                            if (fc->type == BK_DOUBLE)
                                fvd += ((double*)fc->data)[r];
                            else fvi = (int64_t)((uint64_t)fvi +
                                                 (uint64_t)cell_i64(fc, r));
                        } else if (ft == BK_WIN_AVG) {
                            fvd += cell_f64_cast(fc, r);
                        } else if (ft == BK_WIN_MIN || ft == BK_WIN_MAX) {
                            if (fc->type == BK_DOUBLE) {
                                double v = ((double*)fc->data)[r];
                                if (!fcnt || (ft == BK_WIN_MIN ? v < fvd
                                                               : v > fvd))
                                    fvd = v;
                            } else {
                                int64_t v = cell_i64(fc, r);
                                if (!fcnt || (ft == BK_WIN_MIN ? v < fvi
                                                               : v > fvi))
                                    fvi = v;
                            }
                        }
                        fcnt++;
                    }
                    if (ft == BK_WIN_COUNT_STAR || ft == BK_WIN_COUNT) {
                        out_i[idx] = fcnt;
                    } else if (!fcnt) {
                        out_null[idx] = 1;
                    } else if (ft == BK_WIN_AVG) {
                        out_d[idx] = fvd / (double)fcnt;
                    } else if (fc->type == BK_DOUBLE) {
                        out_d[idx] = fvd;
                    } else {
                        out_i[idx] = fvi;
                    }
                }
                continue;
            }
            const OrcCol* c = fns[f].col >= 0 ? &cols[fns[f].col] : NULL;
            int64_t cnt = 0, vi = 0;
            double vd = 0.0;
            int has = 0;
            for (int64_t j = ps; j < pe; j++) {
                int64_t r = out_rows[j];
                if (ft == BK_WIN_COUNT_STAR) { cnt++; continue; }
                if (!cell_is_valid(c, r)) continue;
                switch (ft) {
                    case BK_WIN_COUNT: cnt++; break;
                    case BK_WIN_SUM:
                        if (c->type == BK_DOUBLE) vd += ((double*)c->data)[r];
                        else vi = (int64_t)((uint64_t)vi +
                                            (uint64_t)cell_i64(c, r));
                        cnt++;
                        break;
                    case BK_WIN_AVG: vd += cell_f64_cast(c, r); cnt++; break;
                    case BK_WIN_MIN:
                        if (c->type == BK_DOUBLE) {
                            double v = ((double*)c->data)[r];
                            if (!has || v < vd) vd = v;
                        } else {
                            int64_t v = cell_i64(c, r);
                            if (!has || v < vi) vi = v;
                        }
                        has = 1; cnt++;
                        break;
                    case BK_WIN_MAX:
                        if (c->type == BK_DOUBLE) {
                            double v = ((double*)c->data)[r];
                            if (!has || v > vd) vd = v;
                        } else {
                            int64_t v = cell_i64(c, r);
                            if (!has || v > vi) vi = v;
                        }
                        has = 1; cnt++;
                        break;
                    default: break;
                }
            }
            for (int64_t j = ps; j < pe; j++) {
                int64_t idx = (int64_t)f * n + j;
                out_i[idx] = 0; out_d[idx] = 0.0; out_null[idx] = 0;
                switch (ft) {
                    case BK_WIN_COUNT_STAR:
                    case BK_WIN_COUNT: out_i[idx] = cnt; break;
                    case BK_WIN_SUM:
                        if (!cnt) { out_null[idx] = 1; break; }
                        if (c->type == BK_DOUBLE) out_d[idx] = vd;
                        else out_i[idx] = vi;
                        break;
                    case BK_WIN_AVG:
                        if (!cnt) { out_null[idx] = 1; break; }
                        out_d[idx] = vd / (double)cnt;
                        break;
                    case BK_WIN_MIN:
                    case BK_WIN_MAX:
                        if (!cnt) { out_null[idx] = 1; break; }
                        if (c->type == BK_DOUBLE) out_d[idx] = vd;
                        else out_i[idx] = vi;
                        break;
                    default: break;
                }
            }
        }
        /* rank / positional fns */
        int64_t peer_head = ps, peers_seen = 1, peer_end = ps;
        for (int64_t j = ps; j < pe; j++) {
            if (j > ps) {
                int changed = 0;
                for (int k = 0; k < norder; k++)
                    if (!orc_cell_eq(&cols[order[k].col], out_rows[j],
                                     out_rows[j - 1])) { changed = 1; break; }
                if (changed) { peer_head = j; peers_seen++; }
            }
            if (j >= peer_end) {   /* advance to current peer group's end */
                peer_end = j + 1;
                while (peer_end < pe) {
                    int same = 1;
                    for (int k = 0; k < norder; k++)
                        if (!orc_cell_eq(&cols[order[k].col],
                                         out_rows[peer_end], out_rows[j]))
                            { same = 0; break; }
                    if (!same) break;
                    peer_end++;
                }
            }
            for (int f = 0; f < nfns; f++) {
                int ft = fns[f].fn_type;
                if (ft <= BK_WIN_MAX) continue;
                int64_t idx = (int64_t)f * n + j;
                out_i[idx] = 0; out_d[idx] = 0.0; out_null[idx] = 0;
                int64_t fl = ps, fr = pe - 1;
                if (frame_rows == 1) {
                    if (f_pre >= 0 && j - f_pre > ps) fl = j - f_pre;
                    if (f_fol >= 0 && j + f_fol < pe - 1) fr = j + f_fol;
                } else if (frame_rows == 2) {
                    fr = peer_end - 1;
                } else if (frame_rows == 3) {
                    fl = peer_head;
                } else if (frame_rows == 4) {
                    const OrcCol* okc = &cols[order[0].col];
                    if (!cell_is_valid(okc, out_rows[j])) {
                        fl = peer_head;
                        fr = peer_end - 1;
                    } else {
                        int64_t v = cell_i64(okc, out_rows[j]);
                        fl = j; fr = j;
                        while (fl > ps && cell_is_valid(okc, out_rows[fl - 1]) &&
                               (f_pre < 0 ||
                                cell_i64(okc, out_rows[fl - 1]) >= v - f_pre))
                            fl--;
                        while (fr + 1 < pe &&
                               cell_is_valid(okc, out_rows[fr + 1]) &&
                               (f_fol < 0 ||
                                cell_i64(okc, out_rows[fr + 1]) <= v + f_fol))
                            fr++;
                    }
                }
                switch (ft) {
                    case BK_WIN_ROW_NUMBER: out_i[idx] = j - ps + 1; break;
                    case BK_WIN_RANK: out_i[idx] = peer_head - ps + 1; break;
                    case BK_WIN_DENSE_RANK: out_i[idx] = peers_seen; break;
                    case BK_WIN_PERCENT_RANK:
                        out_d[idx] = pn > 1
                            ? (double)(peer_head - ps) / (double)(pn - 1)
                            : 0.0;
                        break;
                    case BK_WIN_FIRST_VALUE:
                        if (fr < fl) { out_null[idx] = 1; break; }
                        orc_win_value(&cols[fns[f].col], out_rows[fl], idx,
                                      out_i, out_d, out_null);
                        break;
                    case BK_WIN_LAST_VALUE:
                        if (fr < fl) { out_null[idx] = 1; break; }
                        orc_win_value(&cols[fns[f].col], out_rows[fr], idx,
                                      out_i, out_d, out_null);
                        break;
                    case BK_WIN_NTH_VALUE: {
                        int64_t jj = fl + fns[f].param - 1;
                        if (jj >= fl && jj <= fr)
                            orc_win_value(&cols[fns[f].col], out_rows[jj], idx,
                                          out_i, out_d, out_null);
                        else out_null[idx] = 1;
                        break;
                    }
                    case BK_WIN_LEAD:
                    case BK_WIN_LAG: {
                        int64_t off = fns[f].param > 0 ? fns[f].param : 1;
                        int64_t jj = ft == BK_WIN_LEAD ? j + off : j - off;
                        if (jj >= ps && jj < pe) {
                            orc_win_value(&cols[fns[f].col], out_rows[jj], idx,
                                          out_i, out_d, out_null);
                        } else if (fns[f].has_def) {
                            if (cols[fns[f].col].type == BK_DOUBLE)
                                out_d[idx] = fns[f].def_d;
                            else out_i[idx] = fns[f].def_i;
                        } else {
                            out_null[idx] = 1;
                        }
                        break;
                    }
                    case BK_WIN_CUME_DIST:
                        out_d[idx] = (double)(peer_end - ps) / (double)pn;
                        break;
                    case BK_WIN_NTILE: {
                        int64_t k2 = fns[f].param > 0 ? fns[f].param : 1;
                        int64_t quot = pn / k2, rem = pn % k2;
                        int64_t jl = j - ps, fat = rem * (quot + 1);
                        out_i[idx] = jl < fat
                            ? jl / (quot + 1) + 1
                            : rem + (quot > 0 ? (jl - fat) / quot : 0) + 1;
                        break;
                    }
                    default: out_null[idx] = 1; break;
                }
            }
        }
        ps = pe;
    }
    return n;
}

/* ------------------------------------------------------------------ */
/* small exported helpers for tests                                    */
/* ------------------------------------------------------------------ */

ORC_EXPORT int64_t orc_scalar_fn(int32_t fn, int64_t v) { return bk_scalar_fn(fn, v); }
ORC_EXPORT uint64_t orc_encode_i64(int64_t v)  { return bk_enc_i64(v); }
ORC_EXPORT int64_t  orc_decode_i64(uint64_t u) { return bk_dec_i64(u); }
ORC_EXPORT uint64_t orc_encode_f64(double v)   { return bk_enc_f64(v); }
ORC_EXPORT double   orc_decode_f64(uint64_t u) { return bk_dec_f64(u); }
ORC_EXPORT uint64_t orc_mix64(uint64_t x)      { return bk_mix64(x); }
ORC_EXPORT uint64_t orc_cell_bits(uint64_t seed, uint64_t row, uint32_t col) {
    return bk_cell_bits(seed, row, col);
}

/* single-partition-column compatibility entry */
ORC_EXPORT int64_t orc_window(const OrcCol* cols, int ncols,
                              const BkQuerySpec* q, int32_t part_col,
                              const BkOrderSpec* order, int norder,
                              const BkWindowFn* fns, int nfns,
                              int32_t frame_rows, int64_t f_pre, int64_t f_fol,
                              int64_t row_begin, int64_t row_end,
                              int64_t* out_rows, int64_t* out_i,
                              double* out_d, uint8_t* out_null) {
    int32_t pc[1] = {part_col};
    return orc_window_multi(cols, ncols, q, pc, part_col >= 0 ? 1 : 0,
                            order, norder, fns, nfns, frame_rows, f_pre,
                            f_fol, row_begin, row_end, out_rows, out_i,
                            out_d, out_null);
}
