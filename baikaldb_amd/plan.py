# plan.py — host-side mirror of the pb::Plan subset the store receives for
# the SELECT pipeline (reference: proto/plan.proto SCAN/FILTER/AGG/SORT/LIMIT
# nodes; built by ExecNode::create_tree, src/exec/exec_node.cpp:396-414).
#
# A QueryPlan flattens the node tree the way Region::select's exec tree would
# (AggNode -> FilterNode -> ScanNode) into the BkQuerySpec descriptor that
# crosses the C-ABI (include/bk_common.h).
import ctypes as C

# mirror include/bk_common.h (shared with oracle/bindings.py)
TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13
OP_EQ, OP_NE, OP_GT, OP_GE, OP_LT, OP_LE = 0, 1, 2, 3, 4, 5
AGG_COUNT_STAR, AGG_COUNT, AGG_SUM, AGG_AVG, AGG_MIN, AGG_MAX = 0, 1, 2, 3, 4, 5
BK_MAX_GROUP, BK_MAX_CONJ, BK_MAX_AGGS = 2, 8, 8

_OPS = {"=": OP_EQ, "!=": OP_NE, ">": OP_GT, ">=": OP_GE, "<": OP_LT,
        "<=": OP_LE, "in": 6, "not_in": 7,
        "in_bitmap": 8, "not_in_bitmap": 9}
_AGGS = {"count_star": AGG_COUNT_STAR, "count": AGG_COUNT, "sum": AGG_SUM,
         "avg": AGG_AVG, "min": AGG_MIN, "max": AGG_MAX}


class BkConjunct(C.Structure):
    _fields_ = [("col", C.c_int32), ("op", C.c_int32),
                ("cmp_type", C.c_int32), ("n_in", C.c_int32),
                ("lit_i", C.c_int64), ("lit_d", C.c_double),
                ("in_list", C.c_int64 * 16)]


class BkAggSpec(C.Structure):
    _fields_ = [("agg_type", C.c_int32), ("col", C.c_int32)]


class BkOrderSpec(C.Structure):
    _fields_ = [("col", C.c_int32), ("is_asc", C.c_int32),
                ("is_null_first", C.c_int32), ("_pad", C.c_int32)]


class BkQuerySpec(C.Structure):
    _fields_ = [("n_conjuncts", C.c_int32), ("n_group", C.c_int32),
                ("n_aggs", C.c_int32), ("_pad", C.c_int32),
                ("conjuncts", BkConjunct * BK_MAX_CONJ),
                ("group_cols", C.c_int32 * BK_MAX_GROUP),
                ("group_types", C.c_int32 * BK_MAX_GROUP),
                ("aggs", BkAggSpec * BK_MAX_AGGS),
                ("agg_in_types", C.c_int32 * BK_MAX_AGGS)]


class QueryPlan:
    """SELECT <aggs> FROM t WHERE <conjuncts> GROUP BY <group>.

    conjuncts: (col, op_str, literal) — cmp type inferred like the planner's
               arg-typing (scalar_fn_call.cpp:219-225): double if either side
               double, else int64 (dict codes compare as ints for =/!=).
    aggs:      (name, col) with name in count_star/count/sum/avg/min/max.
    """

    def __init__(self, col_types, conjuncts=(), group=(), aggs=()):
        self.col_types = list(col_types)
        self.conjuncts = list(conjuncts)
        self.group = list(group)
        self.aggs = list(aggs)

    def to_spec(self):
        q = BkQuerySpec()
        q.n_conjuncts = len(self.conjuncts)
        for i, (col, op, lit) in enumerate(self.conjuncts):
            cj = q.conjuncts[i]
            cj.col = col
            cj.op = _OPS[op] if isinstance(op, str) else op
            ct = self.col_types[col]
            if cj.op >= 8:  # bitmap membership: lit = (device_ptr, n_bits)
                cj.cmp_type = TYPE_INT64
                cj.lit_i, cj.n_in = int(lit[0]), int(lit[1])
            elif cj.op >= 6:  # IN / NOT IN: lit is a list of int/dict literals
                cj.cmp_type = TYPE_INT64
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
            q.group_cols[i] = col
            q.group_types[i] = self.col_types[col]
        q.n_aggs = len(self.aggs)
        for i, (name, col) in enumerate(self.aggs):
            q.aggs[i].agg_type = _AGGS[name] if isinstance(name, str) else name
            q.aggs[i].col = col
            q.agg_in_types[i] = self.col_types[col] if col >= 0 else TYPE_INT64
        return q
