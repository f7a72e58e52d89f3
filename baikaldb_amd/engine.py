# engine.py — ctypes host driver over the C-ABI of include/bkgpu.h.
# This is plumbing for tests/bench; the production embedding is the C++ host
# layer mirroring ExecNode. No CPU fallback exists: if
# the HIP extension is missing on a GPU machine, constructing GpuEngine fails.
import ctypes as C
import os

import numpy as np

from .plan import (BkQuerySpec, BkOrderSpec, BkWindowFn, QueryPlan,
                   BK_MAX_GROUP, _WINFNS)
from .plan import TYPE_INT64, TYPE_DOUBLE, TYPE_STRING

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_HERE, "libbkgpu.so")


def substr_ref(w: str, start: int, ln=None) -> str:
    """SQL SUBSTR exactly as /root/reference/src/expr/internal_functions.cpp
    substr(): 1-based BYTE positions (std::string indexing); start<0 counts
    from the string end (pos = size+start, no extra -1); start 0 / past-end
    -> empty; len<=0 -> empty; the 2-arg form (ln=None) runs to the end.
    Byte slices that split a multibyte char survive via surrogateescape
    (the reference's raw bytes do the same)."""
    b = w.encode("utf-8", "surrogateescape")
    pos = len(b) + start if start < 0 else start - 1
    if pos < 0 or pos >= len(b):
        return ""
    if ln is None:
        out = b[pos:]
    elif ln <= 0:
        return ""
    else:
        out = b[pos:pos + ln]
    return out.decode("utf-8", "surrogateescape")


def resolve_string_fn(fn, named=None):
    """Resolve a derive_string_fn spec to a word->word callable, restating
    /root/reference/src/expr/internal_functions.cpp semantics (all positions
    are BYTE positions, as std::string indexes):
    - ("substr", start[, len]): substr_ref above
    - ("left", len): first len bytes; len<=0 -> ""
    - ("right", len): last len bytes (whole string when len > size);
      len<=0 -> ""
    - a callable passes through; a str looks up the named map."""
    if isinstance(fn, tuple) and fn[0] == "substr":
        start = fn[1]
        ln = fn[2] if len(fn) > 2 else None
        return lambda w: substr_ref(w, start, ln)
    if isinstance(fn, tuple) and fn[0] == "left":
        ln = fn[1]
        return (lambda w: "") if ln <= 0 else (
            lambda w: w.encode("utf-8", "surrogateescape")[:ln]
                       .decode("utf-8", "surrogateescape"))
    if isinstance(fn, tuple) and fn[0] == "right":
        ln = fn[1]
        return (lambda w: "") if ln <= 0 else (
            lambda w: w.encode("utf-8", "surrogateescape")[-ln:]
                       .decode("utf-8", "surrogateescape"))
    if callable(fn):
        return fn
    return named[fn]


class NativeEngineMissing(RuntimeError):
    pass


class _BkColSpec(C.Structure):
    _fields_ = [("col_type", C.c_int32), ("dist", C.c_int32),
                ("p0", C.c_int64), ("p1", C.c_int64),
                ("null_frac_x1e6", C.c_int32), ("_pad", C.c_int32)]


def _load():
    if not os.path.exists(_LIB):
        raise NativeEngineMissing(
            f"native HIP engine not built: {_LIB} missing — run "
            f"__graft_entry__.build() (there is no CPU fallback)")
    lib = C.CDLL(_LIB)
    lib.bkgpu_last_error.restype = C.c_char_p
    lib.bkgpu_device_count.restype = C.c_int
    lib.bkgpu_set_device.argtypes = [C.c_int]
    lib.bkgpu_table_create.restype = C.c_void_p
    lib.bkgpu_table_create.argtypes = [C.c_int, C.POINTER(_BkColSpec), C.c_int64]
    lib.bkgpu_table_generate.argtypes = [C.c_void_p, C.c_uint64, C.c_int64]
    lib.bkgpu_table_upload.argtypes = [C.c_void_p, C.c_int, C.c_void_p, C.c_void_p]
    lib.bkgpu_table_upload_strings.restype = C.c_int
    lib.bkgpu_table_upload_strings.argtypes = [C.c_void_p, C.c_int,
                                               C.c_char_p,
                                               C.POINTER(C.c_int64),
                                               C.c_void_p]
    lib.bkgpu_table_dict_code.restype = C.c_int64
    lib.bkgpu_table_dict_code.argtypes = [C.c_void_p, C.c_int, C.c_char_p,
                                          C.c_int64, C.c_int]
    lib.bkgpu_table_dict_word.restype = C.c_int
    lib.bkgpu_table_dict_word.argtypes = [C.c_void_p, C.c_int, C.c_int64,
                                          C.c_char_p, C.c_int]
    lib.bkgpu_table_nrows.restype = C.c_int64
    lib.bkgpu_table_nrows.argtypes = [C.c_void_p]
    lib.bkgpu_table_free.argtypes = [C.c_void_p]
    lib.bkgpu_filter_agg.restype = C.c_void_p
    lib.bkgpu_filter_agg.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec),
                                     C.c_int64, C.c_int64, C.c_int64]
    lib.bkgpu_agg_ngroups.restype = C.c_int64
    lib.bkgpu_agg_ngroups.argtypes = [C.c_void_p]
    lib.bkgpu_agg_rows_passed.restype = C.c_int64
    lib.bkgpu_agg_rows_passed.argtypes = [C.c_void_p]
    lib.bkgpu_agg_kernel_ms.restype = C.c_double
    lib.bkgpu_agg_kernel_ms.argtypes = [C.c_void_p]
    lib.bkgpu_agg_breakdown.restype = C.c_int
    lib.bkgpu_agg_breakdown.argtypes = [C.c_void_p, C.c_char_p,
                                        C.POINTER(C.c_double), C.c_int]
    lib.bkgpu_agg_export_bytes.restype = C.c_int64
    lib.bkgpu_agg_export_bytes.argtypes = [C.c_void_p]
    lib.bkgpu_agg_export.argtypes = [C.c_void_p, C.c_void_p, C.c_int64]
    lib.bkgpu_agg_merge.argtypes = [C.c_void_p, C.c_void_p, C.c_int64]
    lib.bkgpu_agg_part_counts.restype = C.c_int
    lib.bkgpu_agg_part_counts.argtypes = [C.c_void_p, C.c_int,
                                          C.POINTER(C.c_int64)]
    lib.bkgpu_agg_export_part.restype = C.c_int
    lib.bkgpu_agg_export_part.argtypes = [C.c_void_p, C.c_int, C.c_int,
                                          C.c_void_p, C.c_int64]
    lib.bkgpu_agg_empty.restype = C.c_void_p
    lib.bkgpu_agg_empty.argtypes = [C.POINTER(BkQuerySpec), C.c_int64]
    lib.bkgpu_agg_nfilled.restype = C.c_int64
    lib.bkgpu_agg_nfilled.argtypes = [C.c_void_p]
    lib.bkgpu_window.restype = C.c_int64
    lib.bkgpu_window.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec), C.c_int32,
                                 C.POINTER(BkOrderSpec), C.c_int,
                                 C.POINTER(BkWindowFn), C.c_int,
                                 C.c_int32, C.c_int64, C.c_int64,
                                 C.c_int64, C.c_int64,
                                 C.POINTER(C.c_int64), C.POINTER(C.c_int64),
                                 C.POINTER(C.c_double), C.POINTER(C.c_uint8)]
    lib.bkgpu_window_multi.restype = C.c_int64
    lib.bkgpu_window_multi.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec),
                                 C.POINTER(C.c_int32), C.c_int32,
                                 C.POINTER(BkOrderSpec), C.c_int,
                                 C.POINTER(BkWindowFn), C.c_int,
                                 C.c_int32, C.c_int64, C.c_int64,
                                 C.c_int64, C.c_int64,
                                 C.POINTER(C.c_int64), C.POINTER(C.c_int64),
                                 C.POINTER(C.c_double), C.POINTER(C.c_uint8)]
    lib.bkgpu_agg_rollup.restype = C.c_void_p
    lib.bkgpu_agg_rollup.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec),
                                     C.POINTER(C.c_int32), C.c_int64]
    lib.bkgpu_table_derive_remap.restype = C.c_int
    lib.bkgpu_table_derive_remap.argtypes = [C.c_void_p, C.c_int,
                                             C.POINTER(C.c_int32), C.c_int64,
                                             C.c_int64]
    lib.bkgpu_table_derive_prog.restype = C.c_int
    lib.bkgpu_table_derive_prog.argtypes = [C.c_void_p, C.c_void_p,
                                            C.c_int32, C.c_int32]
    lib.bkgpu_filter_agg_sorted.restype = C.c_void_p
    lib.bkgpu_filter_agg_sorted.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec),
                                            C.c_int64, C.c_int64]
    lib.bkgpu_agg_fetch.restype = C.c_int64
    lib.bkgpu_agg_fetch.argtypes = [C.c_void_p, C.c_int, C.c_int64,
                                    C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                    C.POINTER(C.c_int64), C.POINTER(C.c_double),
                                    C.POINTER(C.c_uint8)]
    lib.bkgpu_agg_free.argtypes = [C.c_void_p]
    lib.bkgpu_sort_topk.restype = C.c_int64
    lib.bkgpu_sort_topk.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec),
                                    C.POINTER(BkOrderSpec), C.c_int,
                                    C.c_int64, C.c_int64, C.c_int64,
                                    C.POINTER(C.c_int64)]
    lib.bkgpu_topk_kernel_ms.restype = C.c_double
    lib.bkgpu_sync.restype = C.c_int
    lib.bkgpu_filter_collect.restype = C.c_int64
    lib.bkgpu_filter_collect.argtypes = [C.c_void_p, C.POINTER(BkQuerySpec),
                                         C.c_int64, C.c_int64, C.c_int64,
                                         C.POINTER(C.c_int64)]
    lib.bkgpu_gather.restype = C.c_int
    lib.bkgpu_gather.argtypes = [C.c_void_p, C.c_int, C.POINTER(C.c_int64),
                                 C.c_int64, C.POINTER(C.c_int64),
                                 C.POINTER(C.c_double), C.POINTER(C.c_uint8)]
    lib.bkgpu_table_col_type.restype = C.c_int32
    lib.bkgpu_table_col_type.argtypes = [C.c_void_p, C.c_int]
    lib.bkgpu_table_compact.restype = C.c_int
    lib.bkgpu_table_compact.argtypes = [C.c_void_p, C.c_int]
    lib.bkgpu_table_col_width.restype = C.c_int
    lib.bkgpu_table_col_width.argtypes = [C.c_void_p, C.c_int]
    lib.bkgpu_upload_bytes.restype = C.c_void_p
    lib.bkgpu_upload_bytes.argtypes = [C.c_void_p, C.c_int64]
    lib.bkgpu_free_ptr.argtypes = [C.c_void_p]
    return lib


class GpuTable:
    def __init__(self, engine, handle, col_types, nrows):
        self.engine = engine
        self.handle = handle
        self.col_types = col_types
        self.nrows = nrows

    def free(self):
        if self.handle:
            self.engine.lib.bkgpu_table_free(self.handle)
            self.handle = None


class AggResult:
    def __init__(self, engine, handle, plan):
        self.engine = engine
        self.handle = handle
        self.plan = plan

    @property
    def ngroups(self):
        return self.engine.lib.bkgpu_agg_ngroups(self.handle)

    @property
    def rows_passed(self):
        return self.engine.lib.bkgpu_agg_rows_passed(self.handle)

    @property
    def nfilled(self):
        """Group count via the table fill counter (no compact) — valid
        for insert-only hash results (the exchange merge target)."""
        return self.engine.lib.bkgpu_agg_nfilled(self.handle)

    @property
    def kernel_ms(self):
        return self.engine.lib.bkgpu_agg_kernel_ms(self.handle)

    def breakdown(self):
        names = C.create_string_buffer(16 * 8)
        ms = (C.c_double * 8)()
        n = self.engine.lib.bkgpu_agg_breakdown(self.handle, names, ms, 8)
        return {names.raw[16 * i:16 * (i + 1)].split(b"\0")[0].decode(): ms[i]
                for i in range(n)}

    def export_bytes(self):
        return self.engine.lib.bkgpu_agg_export_bytes(self.handle)

    def export_to(self, dev_ptr, cap):
        rc = self.engine.lib.bkgpu_agg_export(self.handle, C.c_void_p(dev_ptr), cap)
        self.engine._check(rc, "agg_export")

    def merge_blob(self, dev_ptr, n_groups):
        rc = self.engine.lib.bkgpu_agg_merge(self.handle, C.c_void_p(dev_ptr), n_groups)
        self.engine._check(rc, "agg_merge")

    def part_counts(self, nparts):
        """Per-part group counts for the hash-partitioned exchange (the
        repartition ExchangeSenderNode does, exchange_sender_node.h:228)."""
        out = (C.c_int64 * nparts)()
        rc = self.engine.lib.bkgpu_agg_part_counts(self.handle, nparts, out)
        self.engine._check(rc, "agg_part_counts")
        return list(out)

    def export_part(self, nparts, part, dev_ptr, part_groups):
        rc = self.engine.lib.bkgpu_agg_export_part(
            self.handle, nparts, part, C.c_void_p(dev_ptr), part_groups)
        self.engine._check(rc, "agg_export_part")

    def fetch(self, sorted=True, max_groups=None):
        n = self.ngroups if max_groups is None else min(max_groups, self.ngroups)
        n = max(n, 0)
        na = self.plan and len(self.plan.aggs) or 0
        flags = np.zeros(max(n, 1), dtype=np.uint8)
        enc = np.zeros(max(n, 1) * BK_MAX_GROUP, dtype=np.uint64)
        out_i = np.zeros(max(na * n, 1), dtype=np.int64)
        out_d = np.zeros(max(na * n, 1), dtype=np.float64)
        out_has = np.zeros(max(na * n, 1), dtype=np.uint8)
        got = self.engine.lib.bkgpu_agg_fetch(
            self.handle, 1 if sorted else 0, n,
            flags.ctypes.data_as(C.POINTER(C.c_uint8)),
            enc.ctypes.data_as(C.POINTER(C.c_uint64)),
            out_i.ctypes.data_as(C.POINTER(C.c_int64)),
            out_d.ctypes.data_as(C.POINTER(C.c_double)),
            out_has.ctypes.data_as(C.POINTER(C.c_uint8)))
        self.engine._check(int(got), "agg_fetch")
        return {
            "ngroups": int(got),
            "rows_passed": self.rows_passed,
            "flags": flags[:got],
            "enc": enc[:got * BK_MAX_GROUP].reshape(got, BK_MAX_GROUP),
            "agg_i": out_i[:na * got].reshape(na, got) if got else
                     np.zeros((na, 0), np.int64),
            "agg_d": out_d[:na * got].reshape(na, got) if got else
                     np.zeros((na, 0), np.float64),
            "agg_has": out_has[:na * got].reshape(na, got) if got else
                       np.zeros((na, 0), np.uint8),
        }

    def free(self):
        if self.handle:
            self.engine.lib.bkgpu_agg_free(self.handle)
            self.handle = None


class GpuEngine:
    def __init__(self, device=0):
        self.lib = _load()
        if self.lib.bkgpu_device_count() < 1:
            raise NativeEngineMissing("no HIP device visible")
        self._check(self.lib.bkgpu_set_device(device), "set_device")

    def _check(self, rc, what):
        if rc < 0:
            raise RuntimeError(f"bkgpu {what}: {self.lib.bkgpu_last_error().decode()}")
        return rc

    def create_table(self, specs, nrows):
        """specs: list of (col_type, dist, p0, p1, null_frac_x1e6)."""
        arr = (_BkColSpec * len(specs))()
        types = []
        for i, s in enumerate(specs):
            arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1, arr[i].null_frac_x1e6 = s
            types.append(s[0])
        h = self.lib.bkgpu_table_create(len(specs), arr, nrows)
        if not h:
            raise RuntimeError(f"table_create: {self.lib.bkgpu_last_error().decode()}")
        return GpuTable(self, h, types, nrows)

    def generate(self, table, seed, row_begin=0, compact=None):
        self._check(self.lib.bkgpu_table_generate(table.handle, seed, row_begin),
                    "table_generate")
        # narrow physical column encoding (bkgpu_table_compact): on by
        # default, BK_NARROW=0 disables, compact=False per call
        if compact is None:
            compact = os.environ.get("BK_NARROW", "1") != "0"
        if compact:
            self.compact(table)

    def compact(self, table, col=-1):
        """Narrow integer columns to frame-of-reference u8/u16/u32 storage
        (value-preserving; see bkgpu_table_compact in include/bkgpu.h)."""
        self._check(self.lib.bkgpu_table_compact(table.handle, col),
                    "table_compact")

    def col_width(self, table, col):
        return self.lib.bkgpu_table_col_width(table.handle, col)

    def upload(self, table, col, data, valid=None):
        vptr = valid.ctypes.data_as(C.c_void_p) if valid is not None else None
        self._check(self.lib.bkgpu_table_upload(
            table.handle, col, data.ctypes.data_as(C.c_void_p), vptr), "table_upload")

    def upload_strings(self, table, col, strings, valid=None):
        """Arbitrary (non-dictionary) VARCHAR: the engine builds the
        order-preserving dictionary and stores int32 codes — string
        GROUP BY / MIN / MAX / ORDER BY and range predicates then run as
        integer code ops with identical semantics."""
        byts = bytearray()
        offs = np.zeros(len(strings) + 1, dtype=np.int64)
        for i, w in enumerate(strings):
            if valid is None or valid[i]:
                byts += w.encode() if isinstance(w, str) else bytes(w)
            offs[i + 1] = len(byts)
        vptr = valid.ctypes.data_as(C.c_void_p) if valid is not None else None
        self._check(self.lib.bkgpu_table_upload_strings(
            table.handle, col, bytes(byts),
            offs.ctypes.data_as(C.POINTER(C.c_int64)), vptr),
            "upload_strings")

    def dict_word(self, table, col, code, cap=256):
        buf = C.create_string_buffer(cap)
        if self.lib.bkgpu_table_dict_word(table.handle, col, code, buf,
                                          cap) < 0:
            return None
        return buf.value.decode()

    def dict_code(self, table, col, word, mode=0):
        """mode 0: exact code (-1 absent); mode 1: lower_bound — turns a
        string range literal into a code literal."""
        w = word.encode() if isinstance(word, str) else bytes(word)
        r = self.lib.bkgpu_table_dict_code(table.handle, col, w, len(w),
                                           mode)
        if r == -2:
            raise RuntimeError(self.lib.bkgpu_last_error().decode())
        return int(r)

    def filter_agg(self, table, plan: QueryPlan, row_begin=0, row_end=None,
                   expected_groups=1 << 16):
        if row_end is None:
            row_end = table.nrows
        q = plan.to_spec()
        h = self.lib.bkgpu_filter_agg(table.handle, C.byref(q), row_begin, row_end,
                                      expected_groups)
        if not h:
            raise RuntimeError(f"filter_agg: {self.lib.bkgpu_last_error().decode()}")
        return AggResult(self, h, plan)

    def agg_empty(self, plan: QueryPlan, expected_groups=1 << 16):
        """Fresh empty aggregate result (merge target for the partitioned
        exchange's received blobs)."""
        q = plan.to_spec()
        h = self.lib.bkgpu_agg_empty(C.byref(q), expected_groups)
        if not h:
            raise RuntimeError(f"agg_empty: {self.lib.bkgpu_last_error().decode()}")
        return AggResult(self, h, plan)

    # unary string scalar fns the engine compiles to dict remaps
    # (internal_functions.cpp upper/lower/reverse/substr via
    # fn_manager.cpp:97-137). The reference transforms BYTES (::toupper /
    # ::tolower per byte, std::reverse on bytes) — so upper/lower touch only
    # ASCII letters (UTF-8 continuation bytes pass through, matching C
    # tolower), and reverse is a byte reversal (surrogateescape keeps
    # non-UTF-8 results round-trippable, as the reference's raw bytes are).
    _UP = str.maketrans("abcdefghijklmnopqrstuvwxyz",
                        "ABCDEFGHIJKLMNOPQRSTUVWXYZ")
    _LO = str.maketrans("ABCDEFGHIJKLMNOPQRSTUVWXYZ",
                        "abcdefghijklmnopqrstuvwxyz")
    STRING_FNS = {
        "upper": lambda w: w.translate(GpuEngine._UP),
        "lower": lambda w: w.translate(GpuEngine._LO),
        "reverse": lambda w: w.encode("utf-8", "surrogateescape")[::-1]
                              .decode("utf-8", "surrogateescape"),
    }

    def derive_string_fn(self, table, col, fn, words):
        """GROUP BY fn(varchar_col): transform the column's word list
        (words[code] for code 0..ncodes), dedup + sort the transformed words
        (order-preserving new codes), append a derived dict column with
        newcode = remap[oldcode] (bkgpu_table_derive_remap). Returns
        (new_col_index, new_words). fn: a STRING_FNS name, a callable, or
        ("substr", start[, len]) / ("left", len) / ("right", len) mirroring
        internal_functions.cpp (byte positions, like std::string)."""
        f = resolve_string_fn(fn, self.STRING_FNS)
        tw = [f(w) for w in words]
        new_words = sorted(set(tw))
        idx = {w: i for i, w in enumerate(new_words)}
        remap = (C.c_int32 * len(words))(*[idx[w] for w in tw])
        nc = self.lib.bkgpu_table_derive_remap(table.handle, col, remap,
                                               len(words), len(new_words))
        if nc < 0:
            raise RuntimeError(
                f"derive_remap: {self.lib.bkgpu_last_error().decode()}")
        table.col_types.append(table.col_types[col])
        return nc, new_words

    def derive_expr(self, table, expr):
        """Append a derived expression column = eval(expr) per row (the
        ScalarFnCall projection, scalar_fn_call.cpp:194-225 compiled to a
        postfix program): makes WINDOW fn inputs / ORDER BY keys / out_cols
        expression-valued with no kernel changes. expr uses the
        plan.compile_expr operand grammar. Returns the new column index."""
        from .plan import BkExprOp, compile_expr, TYPE_INT64, TYPE_DOUBLE
        pool = []
        dom = compile_expr(expr, table.col_types, pool)
        arr = (BkExprOp * len(pool))()
        for i, o in enumerate(pool):
            arr[i].op = o.get("op", 0)
            arr[i].arg = o.get("arg", 0)
            arr[i].domain = o.get("domain", TYPE_INT64)
            arr[i].lit_i = o.get("lit_i", 0)
            arr[i].lit_d = o.get("lit_d", 0.0)
        nc = self.lib.bkgpu_table_derive_prog(
            table.handle, C.cast(arr, C.c_void_p), len(pool), dom)
        if nc < 0:
            raise RuntimeError(
                f"derive_prog: {self.lib.bkgpu_last_error().decode()}")
        table.col_types.append(dom)
        return nc

    def filter_agg_sorted(self, table, plan: QueryPlan, row_begin=0,
                          row_end=None):
        """GROUP BY via sort-based dedup (bkgpu_filter_agg_sorted): for
        group counts approaching the row count, where the hash table
        degrades. Raises if the plan's keys don't pack into <=56 declared
        group_bits."""
        if row_end is None:
            row_end = table.nrows
        q = plan.to_spec()
        h = self.lib.bkgpu_filter_agg_sorted(table.handle, C.byref(q),
                                             row_begin, row_end)
        if not h:
            raise RuntimeError(
                f"filter_agg_sorted: {self.lib.bkgpu_last_error().decode()}")
        return AggResult(self, h, plan)

    # above this expected level-1 cardinality, COUNT/SUM(DISTINCT) uses the
    # sort-dedup level 1 when the keys qualify (BK_DEDUP_SORT=0/1 overrides)
    DEDUP_SORT_MIN_L1 = 1 << 22

    def _l1_sorted_eligible(self, l1_plan, expected_l1_groups):
        import os
        env = os.environ.get("BK_DEDUP_SORT")
        if env is not None:
            return env != "0"
        # no declared-width requirement: bkgpu_filter_agg_sorted auto-packs
        # from column stats and errors out (-> hash fallback) otherwise
        return expected_l1_groups >= self.DEDUP_SORT_MIN_L1

    def filter_agg_distinct(self, table, plan: QueryPlan, row_begin=0,
                            row_end=None, expected_l1_groups=1 << 18,
                            expected_groups=1 << 14):
        """COUNT/SUM(DISTINCT d): level-1 filter_agg grouped by
        (user keys + d), then bkgpu_agg_rollup (the reference's multi-
        distinct planner rewrite, agg_node.cpp:247-258). High-cardinality
        level 1 switches to the sort-dedup path (same results)."""
        l1_plan, q2, src_idx = plan.split_distinct()
        if self._l1_sorted_eligible(l1_plan, expected_l1_groups):
            try:
                l1 = self.filter_agg_sorted(table, l1_plan, row_begin, row_end)
            except RuntimeError:
                l1 = self.filter_agg(table, l1_plan, row_begin, row_end,
                                     expected_groups=expected_l1_groups)
        else:
            l1 = self.filter_agg(table, l1_plan, row_begin, row_end,
                                 expected_groups=expected_l1_groups)
        try:
            h = self.lib.bkgpu_agg_rollup(l1.handle, C.byref(q2), src_idx,
                                          expected_groups)
            if not h:
                raise RuntimeError(
                    f"agg_rollup: {self.lib.bkgpu_last_error().decode()}")
        finally:
            l1.free()
        return AggResult(self, h, plan)

    def sort_topk(self, table, order, limit, plan: QueryPlan = None,
                  row_begin=0, row_end=None):
        if row_end is None:
            row_end = table.nrows
        q = (plan or QueryPlan(table.col_types)).to_spec()
        oarr = (BkOrderSpec * len(order))()
        for i, (col, is_asc, null_first) in enumerate(order):
            oarr[i].col, oarr[i].is_asc, oarr[i].is_null_first = col, is_asc, null_first
        out = np.empty(max(limit, 1), dtype=np.int64)
        n = self.lib.bkgpu_sort_topk(table.handle, C.byref(q), oarr, len(order),
                                     row_begin, row_end, limit,
                                     out.ctypes.data_as(C.POINTER(C.c_int64)))
        self._check(int(n), "sort_topk")
        return out[:n].copy()

    def topk_kernel_ms(self):
        return self.lib.bkgpu_topk_kernel_ms()

    def window(self, table, fns, part_col=-1, order=(), plan: QueryPlan = None,
               frame=None, row_begin=0, row_end=None):
        """WindowNode (non-frame): fns = (name, col[, param]). Returns dict
        with sorted rowids and fn-major out_i/out_d/out_null."""
        if row_end is None:
            row_end = table.nrows
        q = (plan or QueryPlan(table.col_types)).to_spec()
        oarr = (BkOrderSpec * max(len(order), 1))()
        for i, (col, is_asc, null_first) in enumerate(order):
            oarr[i].col, oarr[i].is_asc, oarr[i].is_null_first = \
                col, is_asc, null_first
        farr = (BkWindowFn * len(fns))()
        for i, f in enumerate(fns):
            name, col = f[0], f[1]
            farr[i].fn_type = _WINFNS[name] if isinstance(name, str) else name
            farr[i].col = col
            farr[i].param = f[2] if len(f) > 2 else 0
            if len(f) > 3 and f[3] is not None:
                farr[i].has_def = 1
                if isinstance(f[3], float):
                    farr[i].def_d = f[3]
                else:
                    farr[i].def_i = int(f[3])
        cap = row_end - row_begin
        rowids = np.empty(cap, dtype=np.int64)
        out_i = np.zeros(len(fns) * cap, dtype=np.int64)
        out_d = np.zeros(len(fns) * cap, dtype=np.float64)
        out_null = np.zeros(len(fns) * cap, dtype=np.uint8)
        if frame is None:
            fr, fpre, ffol = 0, -1, -1
        elif frame == "range_upc":    # RANGE UNBOUNDED PRECEDING..CURRENT ROW
            fr, fpre, ffol = 2, -1, -1
        elif frame == "range_crf":    # RANGE CURRENT ROW..UNBOUNDED FOLLOWING
            fr, fpre, ffol = 3, -1, -1
        elif isinstance(frame, tuple) and frame[0] == "range_val":
            fr, fpre, ffol = 4, frame[1], frame[2]
        else:
            fr, fpre, ffol = 1, frame[0], frame[1]
        if isinstance(part_col, (list, tuple)):
            parr = (C.c_int32 * max(len(part_col), 1))(*part_col)
            n = self.lib.bkgpu_window_multi(
                table.handle, C.byref(q), parr, len(part_col), oarr,
                len(order), farr, len(fns), fr, fpre, ffol, row_begin,
                row_end,
                rowids.ctypes.data_as(C.POINTER(C.c_int64)),
                out_i.ctypes.data_as(C.POINTER(C.c_int64)),
                out_d.ctypes.data_as(C.POINTER(C.c_double)),
                out_null.ctypes.data_as(C.POINTER(C.c_uint8)))
        else:
            n = self.lib.bkgpu_window(
                table.handle, C.byref(q), part_col, oarr, len(order),
                farr, len(fns), fr, fpre, ffol, row_begin, row_end,
            rowids.ctypes.data_as(C.POINTER(C.c_int64)),
            out_i.ctypes.data_as(C.POINTER(C.c_int64)),
            out_d.ctypes.data_as(C.POINTER(C.c_double)),
            out_null.ctypes.data_as(C.POINTER(C.c_uint8)))
        self._check(int(n), "window")
        n = int(n)
        nf = len(fns)
        return {
            "n": n,
            "rowids": rowids[:n].copy(),
            "out_i": out_i[:nf * n].reshape(nf, n).copy(),
            "out_d": out_d[:nf * n].reshape(nf, n).copy(),
            "out_null": out_null[:nf * n].reshape(nf, n).copy(),
        }

    def upload_bytes(self, data: bytes):
        p = self.lib.bkgpu_upload_bytes(data, len(data))
        if not p:
            raise RuntimeError("upload_bytes failed")
        return p

    def free_ptr(self, p):
        self.lib.bkgpu_free_ptr(C.c_void_p(p))

    def sync(self):
        self._check(self.lib.bkgpu_sync(), "sync")
