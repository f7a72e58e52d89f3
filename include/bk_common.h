// bk_common.h — shared plain-C descriptors for the MI355X-native BaikalDB OLAP
// hot-path engine ("bkgpu") and its CPU oracle.
//
// These structs cross the C-ABI boundary (include/bkgpu.h) and are also the
// vocabulary of the CPU oracle (oracle/oracle.c). No C++/torch types here.
//
// Type tags mirror baidu/BaikalDB proto/common.proto:46-72 (pb::PrimitiveType)
// so that a baikalStore host embedding this engine can pass its own enum
// values through unchanged.
#ifndef BK_COMMON_H
#define BK_COMMON_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- pb::PrimitiveType (proto/common.proto:46) ---- */
typedef enum BkType {
    BK_INVALID_TYPE = 0,
    BK_NULL_TYPE    = 1,
    BK_BOOL         = 2,
    BK_INT8         = 3,
    BK_INT16        = 4,
    BK_INT32        = 5,
    BK_INT64        = 6,
    BK_UINT8        = 7,
    BK_UINT16       = 8,
    BK_UINT32       = 9,
    BK_UINT64       = 10,
    BK_FLOAT        = 11,
    BK_DOUBLE       = 12,
    BK_STRING       = 13,   /* stored dict-encoded: int32 codes + host dict */
    BK_DATETIME     = 14,   /* MySQL-packed u64 in an int64 column
                               (yearmonth<<46 | day<<41 | hour<<36 |
                                minute<<30 | second<<24; reference
                                include/common/datetime.h:35-68) */
} BkType;

/* ---- pb::PlanNodeType subset (proto/plan.proto:10-23) ---- */
typedef enum BkNodeType {
    BK_SCAN_NODE         = 1,
    BK_SORT_NODE         = 2,
    BK_AGG_NODE          = 4,
    BK_MERGE_AGG_NODE    = 5,
    BK_TABLE_FILTER_NODE = 6,
    BK_LIMIT_NODE        = 11,
    BK_WHERE_FILTER_NODE = 12,
    BK_WINDOW_NODE       = 42,
} BkNodeType;

/* ---- comparison ops of src/expr/operators.cpp:79-105 (eq/ne/gt/ge/lt/le) ---- */
typedef enum BkCmpOp {
    BK_OP_EQ = 0,
    BK_OP_NE = 1,
    BK_OP_GT = 2,
    BK_OP_GE = 3,
    BK_OP_LT = 4,
    BK_OP_LE = 5,
    /* IN-list predicates (src/expr/predicate.h InPredicate; NULL operand =>
     * NULL => row rejected; literal lists carry no NULLs — the planner only
     * pushes literal IN lists). n_in <= BK_MAX_INLIST: literals inline in
     * in_list. n_in > BK_MAX_INLIST: lit_i carries a pointer (device pointer
     * for the GPU engine, host pointer for the oracle) to a SORTED int64
     * array of n_in literals, probed by binary search. */
    BK_OP_IN     = 6,
    BK_OP_NOT_IN = 7,
    /* dict-code bitmap membership: lit_i carries a pointer (device pointer
     * for the GPU engine, host pointer for the oracle) to a bitmap of
     * n_in bits, bit c set <=> dict code c accepted. This is how a LIKE (or
     * any dict-valued predicate) pushes down onto a dictionary column: the
     * embedder matches the pattern against its dictionary once and ships
     * the accept set (the cstore-dict pushdown pattern). */
    BK_OP_IN_BITMAP     = 8,
    BK_OP_NOT_IN_BITMAP = 9,
} BkCmpOp;

#define BK_MAX_INLIST 16

/* ---- AggFnCall::AggType subset (include/expr/agg_fn_call.h:52-58) ---- */
typedef enum BkAggType {
    BK_AGG_COUNT_STAR = 0,
    BK_AGG_COUNT      = 1,
    BK_AGG_SUM        = 2,
    BK_AGG_AVG        = 3,
    BK_AGG_MIN        = 4,
    BK_AGG_MAX        = 5,
    /* DISTINCT aggregates (reference "count_distinct"/"sum_distinct",
     * agg_fn_call.cpp:35-78). Executed as the reference's planner rewrite
     * (agg_node.cpp:247-258): level 1 groups by (group_keys + distinct col),
     * level 2 rolls the dedup key up via bkgpu_agg_rollup. These types only
     * appear in the LEVEL-2 (rollup) spec; their states are COUNT-/SUM-
     * shaped, so export/merge/fetch treat them like COUNT/SUM. */
    BK_AGG_COUNT_DISTINCT = 6,
    BK_AGG_SUM_DISTINCT   = 7,
    BK_AGG_AVG_DISTINCT   = 8,   /* {sum, count} over the dedup keys;
                                    finalizes as sum/count (reference
                                    "avg_distinct", agg_fn_call.cpp:39) */
} BkAggType;

/* ---- synthetic column distributions (bench configs) ---- */
typedef enum BkDist {
    BK_DIST_UNIFORM_I64 = 0,  /* uniform integer in [p0, p1) */
    BK_DIST_CUBESKEW    = 1,  /* integer-only skewed in [0, p0) (density ~ k^-2/3) */
    BK_DIST_DICT        = 2,  /* dict code uniform in [0, p0) (VARCHAR via dict) */
    BK_DIST_SUMU16      = 3,  /* approx N(0,1) double: sum of 4 u16 minus mean, scaled */
    BK_DIST_ZIPFOCT     = 4,  /* log-uniform ("Zipf-1-like") integer in [0, p0):
                                 octave picked uniformly, value uniform in octave
                                 => density ~ 1/(k+1); integer-only, CPU==GPU */
    BK_DIST_DATETIME    = 5,  /* valid packed DATETIME in years [2019, 2026) */
} BkDist;

/* One generated column. Physical storage by type:
 *   BK_INT64  -> int64_t[nrows]
 *   BK_DOUBLE -> double[nrows]
 *   BK_STRING -> int32_t[nrows] dict codes (dict strings generated on host)
 * Optional validity: uint8_t[nrows], 1 = present, 0 = SQL NULL. A column with
 * null_frac_x1e6 == 0 has no validity array (all rows valid). */
typedef struct BkColSpec {
    int32_t col_type;        /* BkType */
    int32_t dist;            /* BkDist */
    int64_t p0, p1;          /* distribution params */
    int32_t null_frac_x1e6;  /* NULL fraction in parts-per-million */
    int32_t _pad;
} BkColSpec;

/* One WHERE conjunct: <col> <op> <literal>, evaluated with the reference's
 * SQL ternary NULL logic (NULL operand => conjunct NULL => row rejected,
 * src/exec/filter_node.cpp:726-734) and the reference's arg-cast rule
 * (src/expr/scalar_fn_call.cpp:219-225: both args cast to the fn arg type).
 * cmp_type selects the typed comparator of src/expr/operators.cpp:79-105:
 *   BK_INT64  -> int64 compare, lit_i
 *   BK_DOUBLE -> double compare (int64 col cast to double), lit_d
 *   BK_STRING -> dict-code equality (EQ/NE only; dict codes are unique per
 *                string so code equality == string equality)
 */
/* unary scalar fn applied to the column value before the compare — the
 * pushed-down datetime extraction calls (reference internal_functions.cpp
 * hour/minute/second/month/year/dayofmonth, fn_manager.cpp:219-230;
 * bit layout datetime.h:35-45 + datetime.cpp:410-419). */
typedef enum BkScalarFn {
    BK_FN_NONE   = 0,
    BK_FN_YEAR   = 1,
    BK_FN_MONTH  = 2,
    BK_FN_DAY    = 3,   /* dayofmonth */
    BK_FN_HOUR   = 4,
    BK_FN_MINUTE = 5,
    BK_FN_SECOND = 6,
} BkScalarFn;

typedef struct BkConjunct {
    int32_t col;
    int32_t op;        /* BkCmpOp */
    int32_t cmp_type;  /* BkType */
    int32_t n_in;      /* IN-list length (BK_OP_IN / BK_OP_NOT_IN) */
    int64_t lit_i;
    double  lit_d;
    int64_t in_list[BK_MAX_INLIST];  /* int64 / dict-code IN literals */
    int32_t fn;        /* BkScalarFn on the column value (int64 paths only) */
    int32_t or_group;  /* 0 = standalone AND term; >0 = OR-clause id: members
                          sharing an id OR together, clauses AND together —
                          how an OR expr tree pushed into FilterNode::
                          need_copy (filter_node.cpp:726-734) evaluates in
                          CNF. SQL ternary: a NULL member is simply not
                          true. */
    int32_t col2;      /* binary-arith predicate (a OP b <cmp> lit): second
                          column; -1 = none. operators.cpp add/minus/
                          multiply semantics: int64 wraps, mixed/double in
                          IEEE f64 (cmp_type decides the domain). Either
                          operand NULL => NULL. */
    int32_t arith;     /* BkArith */
    /* deeper expression LHS: a postfix program in BkQuerySpec.prog
     * (prog_len > 0 replaces col/col2/fn; the compare stays cmp_type) */
    int32_t prog_begin;
    int32_t prog_len;
} BkConjunct;

typedef enum BkArith {
    BK_ARITH_NONE = 0,
    BK_ARITH_ADD = 1,
    BK_ARITH_SUB = 2,
    BK_ARITH_MUL = 3,
} BkArith;

/* One aggregate call (reference: src/expr/agg_fn_call.cpp:496-555 update,
 * 719-830 merge, 927-975 finalize). col == -1 for COUNT(*). */
typedef struct BkAggSpec {
    int32_t agg_type;  /* BkAggType */
    int32_t col;       /* input column, -1 for COUNT_STAR */
    int32_t col2;      /* expression input (col ARITH col2): -1 = plain
                          column. agg_in_types[] carries the result domain
                          (DOUBLE iff either operand is DOUBLE — the
                          reference casts AggFnCall inputs the same way,
                          agg_fn_call.cpp:496-555). NULL if either operand
                          NULL. */
    int32_t arith;     /* BkArith */
    /* deeper expression input: postfix program (replaces col/col2 when
     * prog_len > 0; agg_in_types[] still carries the result domain) */
    int32_t prog_begin;
    int32_t prog_len;
} BkAggSpec;

/* ---- postfix (RPN) expression programs ----
 * ScalarFnCall::get_value walks arbitrary expression trees per row
 * (src/expr/scalar_fn_call.cpp:194-225); the planner flattens such trees
 * into these programs. Each ARITH op declares its COMPUTE DOMAIN,
 * mirroring the reference's arg-cast rule (children cast to the fn's
 * declared arg types, scalar_fn_call.cpp:219-225): INT64 ops wrap like
 * operators.cpp, DOUBLE ops compute IEEE f64 with int operands cast.
 * Any NULL operand makes the result NULL. */
typedef enum BkProgOp {
    BK_PROG_COL   = 0,   /* push column value (arg = column index) */
    BK_PROG_LIT_I = 1,   /* push int64 literal */
    BK_PROG_LIT_D = 2,   /* push double literal */
    BK_PROG_ARITH = 3,   /* pop b, a -> push a OP b (arg = BkArith) */
    BK_PROG_FN    = 4,   /* pop a -> push fn(a) (arg = BkScalarFn, int64) */
} BkProgOp;

typedef struct BkExprOp {
    int32_t op;        /* BkProgOp */
    int32_t arg;
    int32_t domain;    /* BK_INT64 or BK_DOUBLE (ARITH compute domain) */
    int32_t _pad;
    int64_t lit_i;
    double  lit_d;
} BkExprOp;

#define BK_MAX_PROG_POOL 32   /* ops shared by all programs of one query */
#define BK_MAX_PROG_DEPTH 6   /* max operand-stack depth */

/* ---- window functions (reference src/expr/window_fn_call.cpp:20-38 name
 * map; executed in the reference's NON-FRAME mode, window_node.cpp:39-41 —
 * every fn sees the whole partition). ---- */
typedef enum BkWinType {
    BK_WIN_COUNT_STAR   = 0,
    BK_WIN_COUNT        = 1,
    BK_WIN_SUM          = 2,
    BK_WIN_AVG          = 3,
    BK_WIN_MIN          = 4,
    BK_WIN_MAX          = 5,
    BK_WIN_ROW_NUMBER   = 10,
    BK_WIN_RANK         = 11,
    BK_WIN_DENSE_RANK   = 12,
    BK_WIN_PERCENT_RANK = 13,
    BK_WIN_FIRST_VALUE  = 14,
    BK_WIN_LAST_VALUE   = 15,
    BK_WIN_NTH_VALUE    = 16,   /* param = n (1-based) */
    BK_WIN_LEAD         = 17,   /* param = offset; out-of-partition => NULL */
    BK_WIN_LAG          = 18,
    BK_WIN_CUME_DIST    = 19,
    BK_WIN_NTILE        = 20,   /* param = n buckets */
} BkWinType;

typedef struct BkWindowFn {
    int32_t fn_type;   /* BkWinType */
    int32_t col;       /* input column, -1 for COUNT_STAR / pure rank fns */
    int64_t param;     /* NTH_VALUE n / LEAD/LAG offset */
    /* LEAD/LAG optional literal default (window_fn_call.cpp:144-150):
     * returned when the offset row leaves the partition; NULL otherwise */
    int32_t has_def;
    int32_t _pad;
    int64_t def_i;
    double  def_d;
} BkWindowFn;

#define BK_MAX_WINFNS 8

/* ORDER BY key (reference: include/mem_row/mem_row_compare.h:23-45). */
typedef struct BkOrderSpec {
    int32_t col;
    int32_t is_asc;        /* 1 asc, 0 desc */
    int32_t is_null_first; /* NULLs first? */
    int32_t _pad;
} BkOrderSpec;

/* Limits chosen for the hot path (north_star queries use <=2 group cols,
 * <=3 conjuncts, <=4 aggregates; we allow a little headroom). */
#define BK_MAX_COLS      16
#define BK_MAX_CONJUNCTS 8
#define BK_MAX_GROUP     4
#define BK_MAX_AGGS      8

/* A full query descriptor over one columnar table: the pb::Plan subset the
 * store receives for the SELECT pipeline (SCAN -> FILTER -> AGG), flattened. */
typedef struct BkQuerySpec {
    int32_t    n_conjuncts;
    int32_t    n_group;
    int32_t    n_aggs;
    int32_t    _pad;
    BkConjunct conjuncts[BK_MAX_CONJUNCTS];
    int32_t    group_cols[BK_MAX_GROUP];
    int32_t    group_types[BK_MAX_GROUP];  /* BkType of each group col */
    /* >2 group keys pack into the engine's two 64-bit key words. The CALLER
     * declares, per key, a bit width and a base VALUE (inclusive minimum);
     * the engine packs (enc(v) - enc(base)) into `bits` bits. bits == 0
     * (the default) means a full 64-bit word — with <= 2 keys and all-zero
     * bits this reproduces the plain two-word layout. Declaring the widths
     * in the QUERY (not from data statistics) keeps the packed keys
     * identical on every rank, so partial-aggregate blobs merge across
     * GPUs. Values outside [base, base + 2^bits) are caller error.
     * Widths must sum to <= 64 bits per word (keys never straddle words);
     * DOUBLE keys require bits == 0. */
    int32_t    group_bits[BK_MAX_GROUP];
    int64_t    group_base[BK_MAX_GROUP];
    /* optional BkScalarFn per group key (GROUP BY year(c) etc. — the
     * reference evaluates fn exprs inside encode_exprs_key,
     * exec_node.cpp:555-571). Only int64/DATETIME keys; the key's output
     * type is INT64. */
    int32_t    group_fns[BK_MAX_GROUP];
    BkAggSpec  aggs[BK_MAX_AGGS];
    int32_t    agg_in_types[BK_MAX_AGGS];  /* BkType of each agg input col */
    /* shared postfix-program pool (BkConjunct/BkAggSpec prog_begin/len) */
    int32_t    n_prog;
    int32_t    _pad2;
    BkExprOp   prog[BK_MAX_PROG_POOL];
} BkQuerySpec;

#ifdef __cplusplus
}
#endif
#endif /* BK_COMMON_H */
