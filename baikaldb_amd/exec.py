# exec.py — ctypes view of the C++ ExecNode mirror (include/bk_exec.h):
# builds the flattened pre-order plan the way pb::Plan carries it
# (proto/plan.proto:495-510) and drives the open/get_next/close loop the way
# Region::select_normal does (region.cpp:3166-3216). Used by tests and
# INTEGRATION examples; a store embedding uses the C++ classes directly.
import ctypes as C

import numpy as np

from .engine import _load
from .plan import (BkQuerySpec, BkConjunct, BkAggSpec, BkOrderSpec,  # noqa
                   BkWindowFn, _WINFNS,
                   BK_MAX_GROUP, BK_MAX_CONJ, BK_MAX_AGGS, _OPS, _AGGS,
                   TYPE_INT64, TYPE_DOUBLE, TYPE_STRING)

BK_MAX_COLS = 16
SCAN, SORT, AGG, MERGE_AGG, TABLE_FILTER, LIMIT, WHERE_FILTER = 1, 2, 4, 5, 6, 11, 12
WINDOW = 42


class BkPlanNodeDesc(C.Structure):
    _fields_ = [("node_type", C.c_int32), ("num_children", C.c_int32),
                ("limit", C.c_int64), ("offset", C.c_int64),
                ("table", C.c_void_p),
                ("n_conjuncts", C.c_int32),
                ("conjuncts", BkConjunct * BK_MAX_CONJ),
                ("n_group", C.c_int32),
                ("group_cols", C.c_int32 * BK_MAX_GROUP),
                ("group_bits", C.c_int32 * BK_MAX_GROUP),
                ("group_base", C.c_int64 * BK_MAX_GROUP),
                ("n_aggs", C.c_int32),
                ("aggs", BkAggSpec * BK_MAX_AGGS),
                ("expected_groups", C.c_int64),
                ("distinct_bits", C.c_int32), ("_pad_d", C.c_int32),
                ("distinct_base", C.c_int64),
                ("n_order", C.c_int32),
                ("order", BkOrderSpec * 4),
                ("n_out_cols", C.c_int32),
                ("out_cols", C.c_int32 * BK_MAX_COLS),
                ("part_col", C.c_int32),
                ("n_winfns", C.c_int32),
                ("winfns", BkWindowFn * 8),
                ("frame_mode", C.c_int32),
                ("_pad_w", C.c_int32),
                ("frame_pre", C.c_int64), ("frame_fol", C.c_int64)]


def _bind(lib):
    lib.bkexec_create_tree.restype = C.c_void_p
    lib.bkexec_create_tree.argtypes = [C.POINTER(BkPlanNodeDesc), C.c_int]
    lib.bkexec_open.restype = C.c_int
    lib.bkexec_open.argtypes = [C.c_void_p]
    lib.bkexec_n_slots.restype = C.c_int
    lib.bkexec_n_slots.argtypes = [C.c_void_p]
    lib.bkexec_get_next.restype = C.c_int64
    lib.bkexec_get_next.argtypes = [C.c_void_p, C.c_int64,
                                    C.POINTER(C.c_int32), C.POINTER(C.c_int64),
                                    C.POINTER(C.c_double), C.POINTER(C.c_uint8),
                                    C.POINTER(C.c_int)]
    for f in ("bkexec_num_scan_rows", "bkexec_num_filter_rows",
              "bkexec_num_rows_returned"):
        getattr(lib, f).restype = C.c_int64
        getattr(lib, f).argtypes = [C.c_void_p]
    lib.bkexec_close.argtypes = [C.c_void_p]
    lib.bkexec_dict_word.restype = C.c_int
    lib.bkexec_dict_word.argtypes = [C.c_uint64, C.c_int64, C.c_char_p, C.c_int]
    return lib


def scan_node(table):
    d = BkPlanNodeDesc()
    d.node_type, d.num_children = SCAN, 0
    d.limit = -1
    d.table = table.handle
    return d


def filter_node(col_types, conjuncts, num_children=1):
    d = BkPlanNodeDesc()
    d.node_type, d.num_children = WHERE_FILTER, num_children
    d.limit = -1
    d.n_conjuncts = len(conjuncts)
    for i, (col, op, lit) in enumerate(conjuncts):
        cj = d.conjuncts[i]
        if isinstance(col, tuple):   # ("hour", col): pushed-down scalar fn
            from .plan import _FNS
            cj.fn = _FNS[col[0]]
            col = col[1]
        cj.col = col
        cj.op = _OPS[op] if isinstance(op, str) else op
        if col_types[col] == TYPE_DOUBLE or isinstance(lit, float):
            cj.cmp_type, cj.lit_d = TYPE_DOUBLE, float(lit)
        else:
            cj.cmp_type, cj.lit_i = TYPE_INT64, int(lit)
    return d


def window_node(part_col, order, fns, out_cols, num_children=1, limit=-1,
                frame=None):
    """WINDOW_NODE (window_node.cpp, non-frame): fns = (name, col[, param]);
    slots = [out_cols...][fn outputs...]."""
    d = BkPlanNodeDesc()
    d.node_type, d.num_children = WINDOW, num_children
    d.limit = limit
    d.part_col = part_col
    d.n_order = len(order)
    for i, (col, is_asc, null_first) in enumerate(order):
        d.order[i].col, d.order[i].is_asc, d.order[i].is_null_first = \
            col, is_asc, null_first
    d.n_winfns = len(fns)
    for i, f in enumerate(fns):
        name = f[0]
        d.winfns[i].fn_type = _WINFNS[name] if isinstance(name, str) else name
        d.winfns[i].col = f[1]
        d.winfns[i].param = f[2] if len(f) > 2 else 0
    d.n_out_cols = len(out_cols)
    for i, c in enumerate(out_cols):
        d.out_cols[i] = c
    if frame == "range_upc":
        d.frame_mode = 2
    elif frame == "range_crf":
        d.frame_mode = 3
    elif isinstance(frame, tuple) and frame and frame[0] == "range_val":
        d.frame_mode, d.frame_pre, d.frame_fol = 4, frame[1], frame[2]
    elif frame is not None:
        d.frame_mode, d.frame_pre, d.frame_fol = 1, frame[0], frame[1]
    return d


def agg_node(group, aggs, expected_groups=1 << 16, merge=False, num_children=1,
             limit=-1, group_bits=(), group_base=(), distinct_bits=0,
             distinct_base=0):
    d = BkPlanNodeDesc()
    d.node_type, d.num_children = (MERGE_AGG if merge else AGG), num_children
    d.limit = limit
    d.n_group = len(group)
    for i, c in enumerate(group):
        d.group_cols[i] = c
        if i < len(group_bits):
            d.group_bits[i] = group_bits[i]
        if i < len(group_base):
            d.group_base[i] = group_base[i]
    d.n_aggs = len(aggs)
    for i, (name, col) in enumerate(aggs):
        d.aggs[i].agg_type = _AGGS[name] if isinstance(name, str) else name
        d.aggs[i].col = col
    d.expected_groups = expected_groups
    d.distinct_bits = distinct_bits
    d.distinct_base = distinct_base
    return d


def sort_node(order, out_cols, limit, num_children=1):
    d = BkPlanNodeDesc()
    d.node_type, d.num_children = SORT, num_children
    d.limit = limit
    d.n_order = len(order)
    for i, (col, asc, nf) in enumerate(order):
        d.order[i].col, d.order[i].is_asc, d.order[i].is_null_first = col, asc, nf
    d.n_out_cols = len(out_cols)
    for i, c in enumerate(out_cols):
        d.out_cols[i] = c
    return d


def limit_node(limit, num_children=1, offset=0):
    d = BkPlanNodeDesc()
    d.node_type, d.num_children = LIMIT, num_children
    d.limit = limit
    d.offset = offset
    return d


class ExecTree:
    """create_tree + Region::select_normal driver."""

    def __init__(self, nodes):
        self.lib = _bind(_load())
        arr = (BkPlanNodeDesc * len(nodes))(*nodes)
        self.handle = self.lib.bkexec_create_tree(arr, len(nodes))
        if not self.handle:
            raise RuntimeError("bkexec_create_tree failed")

    def open(self):
        rc = self.lib.bkexec_open(self.handle)
        if rc < 0:
            raise RuntimeError("bkexec_open failed")

    def fetch_all(self, batch=1024):
        """Drive get_next to eos; returns (tags, vals_i, vals_d, nulls) each
        shaped (rows, n_slots)."""
        ns = self.lib.bkexec_n_slots(self.handle)
        tags, vi, vd, nn = [], [], [], []
        eos = C.c_int(0)
        while not eos.value:
            t = np.zeros(batch * ns, dtype=np.int32)
            i = np.zeros(batch * ns, dtype=np.int64)
            d = np.zeros(batch * ns, dtype=np.float64)
            u = np.zeros(batch * ns, dtype=np.uint8)
            got = self.lib.bkexec_get_next(
                self.handle, batch,
                t.ctypes.data_as(C.POINTER(C.c_int32)),
                i.ctypes.data_as(C.POINTER(C.c_int64)),
                d.ctypes.data_as(C.POINTER(C.c_double)),
                u.ctypes.data_as(C.POINTER(C.c_uint8)), C.byref(eos))
            if got < 0:
                raise RuntimeError("bkexec_get_next failed")
            if got:
                tags.append(t[:got * ns].reshape(got, ns))
                vi.append(i[:got * ns].reshape(got, ns))
                vd.append(d[:got * ns].reshape(got, ns))
                nn.append(u[:got * ns].reshape(got, ns))
        z = lambda lst, dt: (np.concatenate(lst) if lst else
                             np.zeros((0, ns), dtype=dt))
        return (z(tags, np.int32), z(vi, np.int64), z(vd, np.float64),
                z(nn, np.uint8))

    @property
    def num_scan_rows(self):
        return self.lib.bkexec_num_scan_rows(self.handle)

    @property
    def num_filter_rows(self):
        return self.lib.bkexec_num_filter_rows(self.handle)

    def close(self):
        if self.handle:
            self.lib.bkexec_close(self.handle)
            self.handle = None
