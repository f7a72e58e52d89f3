# plan.py — host-side mirror of the pb::Plan subset the store receives for
# the SELECT pipeline (reference: proto/plan.proto SCAN/FILTER/AGG/SORT/LIMIT
# nodes; built by ExecNode::create_tree, src/exec/exec_node.cpp:396-414).
#
# A QueryPlan flattens the node tree the way Region::select's exec tree would
# (AggNode -> FilterNode -> ScanNode) into the BkQuerySpec descriptor that
# crosses the C-ABI (include/bk_common.h).
import ctypes as C

# mirror include/bk_common.h (shared with oracle/bindings.py)
TYPE_INT64, TYPE_DOUBLE, TYPE_STRING, TYPE_DATETIME = 6, 12, 13, 14
OP_EQ, OP_NE, OP_GT, OP_GE, OP_LT, OP_LE = 0, 1, 2, 3, 4, 5
AGG_COUNT_STAR, AGG_COUNT, AGG_SUM, AGG_AVG, AGG_MIN, AGG_MAX = 0, 1, 2, 3, 4, 5
AGG_COUNT_DISTINCT, AGG_SUM_DISTINCT = 6, 7
BK_MAX_GROUP, BK_MAX_CONJ, BK_MAX_AGGS = 4, 8, 8

_OPS = {"=": OP_EQ, "!=": OP_NE, ">": OP_GT, ">=": OP_GE, "<": OP_LT,
        "<=": OP_LE, "in": 6, "not_in": 7,
        "in_bitmap": 8, "not_in_bitmap": 9}
_ARITH = {"add": 1, "sub": 2, "mul": 3}
_FNS = {"year": 1, "month": 2, "day": 3, "dayofmonth": 3, "hour": 4,
        "minute": 5, "second": 6}
_AGGS = {"count_star": AGG_COUNT_STAR, "count": AGG_COUNT, "sum": AGG_SUM,
         "avg": AGG_AVG, "min": AGG_MIN, "max": AGG_MAX,
         "count_distinct": AGG_COUNT_DISTINCT,
         "sum_distinct": AGG_SUM_DISTINCT, "avg_distinct": 8}


class BkConjunct(C.Structure):
    _fields_ = [("col", C.c_int32), ("op", C.c_int32),
                ("cmp_type", C.c_int32), ("n_in", C.c_int32),
                ("lit_i", C.c_int64), ("lit_d", C.c_double),
                ("in_list", C.c_int64 * 16),
                ("fn", C.c_int32), ("or_group", C.c_int32),
                ("col2", C.c_int32), ("arith", C.c_int32),
                ("prog_begin", C.c_int32), ("prog_len", C.c_int32)]


class BkAggSpec(C.Structure):
    _fields_ = [("agg_type", C.c_int32), ("col", C.c_int32),
                ("col2", C.c_int32), ("arith", C.c_int32),
                ("prog_begin", C.c_int32), ("prog_len", C.c_int32)]


class BkExprOp(C.Structure):
    _fields_ = [("op", C.c_int32), ("arg", C.c_int32),
                ("domain", C.c_int32), ("_pad", C.c_int32),
                ("lit_i", C.c_int64), ("lit_d", C.c_double)]


BK_MAX_PROG_POOL = 32
PROG_COL, PROG_LIT_I, PROG_LIT_D, PROG_ARITH, PROG_FN = 0, 1, 2, 3, 4


def compile_expr(e, col_types, pool):
    """RPN-compile a nested expression tree into `pool` (list of op dicts);
    returns the expression's domain type. This is the planner half of the
    engine's postfix programs: ScalarFnCall::get_value walks such trees per
    row (scalar_fn_call.cpp:194-225); the compute domain of every ARITH
    node follows the reference's arg-cast rule (children cast to the fn's
    arg types, scalar_fn_call.cpp:219-225 — DOUBLE iff either child is).

    Operands: int = column index; float = double literal;
    ("liti", v) / ("litf", v) explicit literals;
    ("add"|"sub"|"mul", a, b); scalar fns ("year"|..., a)."""
    if isinstance(e, bool):
        raise ValueError("bool is not an expression operand")
    if isinstance(e, int):
        pool.append(dict(op=PROG_COL, arg=e))
        return TYPE_DOUBLE if col_types[e] == TYPE_DOUBLE else TYPE_INT64
    if isinstance(e, float):
        pool.append(dict(op=PROG_LIT_D, lit_d=e))
        return TYPE_DOUBLE
    tag = e[0]
    if tag == "liti":
        pool.append(dict(op=PROG_LIT_I, lit_i=int(e[1])))
        return TYPE_INT64
    if tag == "litf":
        pool.append(dict(op=PROG_LIT_D, lit_d=float(e[1])))
        return TYPE_DOUBLE
    if tag in _ARITH:
        d1 = compile_expr(e[1], col_types, pool)
        d2 = compile_expr(e[2], col_types, pool)
        dom = TYPE_DOUBLE if TYPE_DOUBLE in (d1, d2) else TYPE_INT64
        pool.append(dict(op=PROG_ARITH, arg=_ARITH[tag], domain=dom))
        return dom
    if tag in _FNS:
        compile_expr(e[1], col_types, pool)
        pool.append(dict(op=PROG_FN, arg=_FNS[tag]))
        return TYPE_INT64
    raise ValueError(f"bad expression node: {e!r}")


def expr_is_deep(e):
    """True when the tuple expression needs a postfix program (anything
    beyond the legacy one-arith / one-fn shapes with plain column refs)."""
    if not isinstance(e, tuple):
        return False
    if e[0] in ("liti", "litf"):
        return True
    return any(isinstance(x, (tuple, float)) for x in e[1:])


def spec_add_prog(q, ops):
    """Write compiled ops into q.prog; returns (begin, len)."""
    begin = q.n_prog
    if begin + len(ops) > BK_MAX_PROG_POOL:
        raise ValueError("expression program pool overflow")
    for i, o in enumerate(ops):
        dst = q.prog[begin + i]
        dst.op = o.get("op", 0)
        dst.arg = o.get("arg", 0)
        dst.domain = o.get("domain", TYPE_INT64)
        dst.lit_i = o.get("lit_i", 0)
        dst.lit_d = o.get("lit_d", 0.0)
    q.n_prog = begin + len(ops)
    return begin, len(ops)


_WINFNS = {"count_star": 0, "count": 1, "sum": 2, "avg": 3, "min": 4,
           "max": 5, "row_number": 10, "rank": 11, "dense_rank": 12,
           "percent_rank": 13, "first_value": 14, "last_value": 15,
           "nth_value": 16, "lead": 17, "lag": 18, "cume_dist": 19,
           "ntile": 20}


class BkWindowFn(C.Structure):
    _fields_ = [("fn_type", C.c_int32), ("col", C.c_int32),
                ("param", C.c_int64),
                ("has_def", C.c_int32), ("_pad", C.c_int32),
                ("def_i", C.c_int64), ("def_d", C.c_double)]


class BkOrderSpec(C.Structure):
    _fields_ = [("col", C.c_int32), ("is_asc", C.c_int32),
                ("is_null_first", C.c_int32), ("_pad", C.c_int32)]


class BkQuerySpec(C.Structure):
    _fields_ = [("n_conjuncts", C.c_int32), ("n_group", C.c_int32),
                ("n_aggs", C.c_int32), ("_pad", C.c_int32),
                ("conjuncts", BkConjunct * BK_MAX_CONJ),
                ("group_cols", C.c_int32 * BK_MAX_GROUP),
                ("group_types", C.c_int32 * BK_MAX_GROUP),
                ("group_bits", C.c_int32 * BK_MAX_GROUP),
                ("group_base", C.c_int64 * BK_MAX_GROUP),
                ("group_fns", C.c_int32 * BK_MAX_GROUP),
                ("aggs", BkAggSpec * BK_MAX_AGGS),
                ("agg_in_types", C.c_int32 * BK_MAX_AGGS),
                ("n_prog", C.c_int32), ("_pad2", C.c_int32),
                ("prog", BkExprOp * BK_MAX_PROG_POOL)]


class QueryPlan:
    """SELECT <aggs> FROM t WHERE <conjuncts> GROUP BY <group>.

    conjuncts: (col, op_str, literal) — cmp type inferred like the planner's
               arg-typing (scalar_fn_call.cpp:219-225): double if either side
               double, else int64 (dict codes compare as ints for =/!=).
    aggs:      (name, col) with name in count_star/count/sum/avg/min/max.
    """

    def __init__(self, col_types, conjuncts=(), group=(), aggs=(),
                 group_bits=(), group_base=(), distinct_bits=0,
                 distinct_base=0):
        self.col_types = list(col_types)
        self.conjuncts = list(conjuncts)
        self.group = list(group)
        self.aggs = list(aggs)
        # >2 group keys pack into two 64-bit words: per-key bit width and
        # base value, declared by the caller (bk_common.h BkQuerySpec)
        self.group_bits = list(group_bits)
        self.group_base = list(group_base)
        # declared width/base of the DISTINCT column's encoding: lets the
        # level-1 (keys + d) plan pack into one word so the sort-dedup
        # level 1 qualifies (engine.filter_agg_sorted)
        self.distinct_bits = distinct_bits
        self.distinct_base = distinct_base

    def to_spec(self):
        q = BkQuerySpec()
        q.n_conjuncts = len(self.conjuncts)
        for i, cjt in enumerate(self.conjuncts):
            # (col, op, lit) or (col, op, lit, or_group): members sharing an
            # or_group > 0 OR together, clauses AND together (CNF pushdown of
            # an OR expr tree, filter_node.cpp:726-734)
            col, op, lit = cjt[0], cjt[1], cjt[2]
            cj = q.conjuncts[i]
            cj.or_group = cjt[3] if len(cjt) > 3 else 0
            if not (0 <= cj.or_group <= 31):
                # engine + oracle fold clause ids with & 31: out-of-range
                # ids would silently merge clauses congruent mod 32
                raise ValueError(f"or_group {cj.or_group} out of range 0..31")
            cj.col2 = -1
            if expr_is_deep(col):
                ops = []
                dom = compile_expr(col, self.col_types, ops)
                cj.prog_begin, cj.prog_len = spec_add_prog(q, ops)
                cj.col = 0
                cj.op = _OPS[op] if isinstance(op, str) else op
                if dom == TYPE_DOUBLE or isinstance(lit, float):
                    cj.cmp_type = TYPE_DOUBLE
                    cj.lit_d = float(lit)
                else:
                    cj.cmp_type = TYPE_INT64
                    cj.lit_i = int(lit)
                continue
            if isinstance(col, tuple) and col[0] in _ARITH:
                # ("add"|"sub"|"mul", c1, c2): binary-arith predicate; the
                # compare domain is DOUBLE iff either column is DOUBLE
                cj.arith = _ARITH[col[0]]
                cj.col2 = col[2]
                if (self.col_types[col[1]] == TYPE_DOUBLE or
                        self.col_types[col[2]] == TYPE_DOUBLE):
                    lit = float(lit)
                col = col[1]
            elif isinstance(col, tuple):  # ("hour", col): pushed-down scalar fn
                cj.fn = _FNS[col[0]]
                col = col[1]
            cj.col = col
            cj.op = _OPS[op] if isinstance(op, str) else op
            ct = self.col_types[col]
            if cj.op >= 8:  # bitmap membership: lit = (device_ptr, n_bits)
                cj.cmp_type = TYPE_INT64
                cj.lit_i, cj.n_in = int(lit[0]), int(lit[1])
            elif cj.op >= 6:  # IN / NOT IN: small lists inline; big lists
                cj.cmp_type = TYPE_INT64  # as (device_ptr, n) sorted arrays
                if isinstance(lit, tuple) and len(lit) == 2 and \
                        isinstance(lit[0], int) and lit[1] > len(q.conjuncts[i].in_list):
                    cj.lit_i, cj.n_in = int(lit[0]), int(lit[1])
                else:
                    cj.n_in = len(lit)
                    for m, v in enumerate(lit):
                        cj.in_list[m] = int(v)
            elif ct == TYPE_DOUBLE or isinstance(lit, float):
                cj.cmp_type = TYPE_DOUBLE
                cj.lit_d = float(lit)
            else:
                cj.cmp_type = TYPE_INT64
                cj.lit_i = int(lit)
        q.n_group = len(self.group)
        for i, col in enumerate(self.group):
            if isinstance(col, tuple):   # ("year", col): GROUP BY fn(col)
                q.group_fns[i] = _FNS[col[0]]
                col = col[1]
            q.group_cols[i] = col
            q.group_types[i] = self.col_types[col]
            if i < len(self.group_bits):
                q.group_bits[i] = self.group_bits[i]
            if i < len(self.group_base):
                q.group_base[i] = self.group_base[i]
        q.n_aggs = len(self.aggs)
        for i, (name, col) in enumerate(self.aggs):
            q.aggs[i].agg_type = _AGGS[name] if isinstance(name, str) else name
            q.aggs[i].col2 = -1
            if expr_is_deep(col):
                ops = []
                dom = compile_expr(col, self.col_types, ops)
                (q.aggs[i].prog_begin,
                 q.aggs[i].prog_len) = spec_add_prog(q, ops)
                q.aggs[i].col = 0
                q.agg_in_types[i] = dom
                continue
            if isinstance(col, tuple):
                # expression input: ("add"|"sub"|"mul", a, b) — the domain is
                # DOUBLE iff either operand is DOUBLE (AggFnCall input cast,
                # agg_fn_call.cpp:496-555)
                arith, a_c, b_c = _ARITH[col[0]], col[1], col[2]
                if (self.col_types[a_c] == TYPE_STRING or
                        self.col_types[b_c] == TYPE_STRING):
                    raise ValueError("arith agg input needs numeric columns")
                q.aggs[i].arith = arith
                q.aggs[i].col = a_c
                q.aggs[i].col2 = b_c
                q.agg_in_types[i] = (TYPE_DOUBLE
                                     if TYPE_DOUBLE in (self.col_types[a_c],
                                                        self.col_types[b_c])
                                     else TYPE_INT64)
                continue
            q.aggs[i].col = col
            q.agg_in_types[i] = self.col_types[col] if col >= 0 else TYPE_INT64
        return q

    def has_distinct(self):
        return any((_AGGS[n] if isinstance(n, str) else n) >= AGG_COUNT_DISTINCT
                   for n, _ in self.aggs)

    def split_distinct(self):
        """The reference planner's multi-distinct rewrite (agg_node.cpp:
        247-258): return (level1_plan, level2_spec, src_idx) where level 1
        groups by (user keys + distinct col) and carries the plain aggs, and
        level 2 (bkgpu_agg_rollup / orc_filter_agg_distinct) folds the dedup
        key back out. Envelope: one distinct column, <= 2 user group keys
        (2 keys need group_bits declared so (keys + d) packs into the two
        64-bit key words).
        """
        dist = [(i, n, c) for i, (n, c) in enumerate(self.aggs)
                if (_AGGS[n] if isinstance(n, str) else n) >= AGG_COUNT_DISTINCT]
        dcols = {c for _, _, c in dist}
        if len(dcols) != 1:
            raise ValueError("exactly one DISTINCT column supported")
        if len(self.group) > 2:
            raise ValueError("DISTINCT aggs support <= 2 group keys")
        if len(self.group) == 2 and (len(self.group_bits) < 2 or
                                     not all(self.group_bits[:2])):
            raise ValueError("DISTINCT with 2 group keys needs group_bits "
                             "declared for both")
        dcol = dcols.pop()
        plain = [(n, c) for n, c in self.aggs
                 if (_AGGS[n] if isinstance(n, str) else n) < AGG_COUNT_DISTINCT]
        nk = len(self.group)
        gb = [self.group_bits[i] if i < len(self.group_bits) else 0
              for i in range(nk)]
        gv = [self.group_base[i] if i < len(self.group_base) else 0
              for i in range(nk)]
        l1 = QueryPlan(self.col_types, conjuncts=self.conjuncts,
                       group=self.group + [dcol],
                       aggs=plain or [("count_star", -1)],
                       group_bits=gb + [self.distinct_bits],
                       group_base=gv + [self.distinct_base])
        q2 = BkQuerySpec()
        q2.n_conjuncts = 0
        q2.n_group = len(self.group)
        for i, col in enumerate(self.group):
            q2.group_cols[i] = col
            q2.group_types[i] = self.col_types[col]
            if i < len(self.group_bits):
                q2.group_bits[i] = self.group_bits[i]
            if i < len(self.group_base):
                q2.group_base[i] = self.group_base[i]
        q2.n_aggs = len(self.aggs)
        src_idx = (C.c_int32 * len(self.aggs))()
        next_plain = 0
        for i, (name, col) in enumerate(self.aggs):
            t = _AGGS[name] if isinstance(name, str) else name
            q2.aggs[i].agg_type = t
            q2.aggs[i].col = col
            q2.agg_in_types[i] = self.col_types[col] if col >= 0 else TYPE_INT64
            if t >= AGG_COUNT_DISTINCT:
                src_idx[i] = -1
            else:
                src_idx[i] = next_plain
                next_plain += 1
        return l1, q2, src_idx
