# GPU parity tests: the HIP fused filter+aggregate vs the CPU oracle on
# identical seeded inputs (bk_datagen.h generates bit-identical data on both
# sides). Bit-exact for COUNT / integer SUM / integer keys / MIN/MAX;
# stated tolerance for SUM/AVG DOUBLE (GPU reduction order differs from the
# reference's sequential ExprValue::add — north_star fp tolerance clause).
import ctypes as C

import numpy as np
import pytest

from oracle import BkColSpec
from oracle.bindings import make_query

pytestmark = pytest.mark.gpu

SEED = 0xBADC0DE
TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13
D_UNI, D_SKEW, D_DICT, D_SUM16 = 0, 1, 2, 3

# tolerance for double sums: |err| <= DTOL_REL * (|sum| + n_rows * 1.0) — the
# inputs are ~N(0,1) so per-element |x| ~ 1; fp error of a pairwise-vs-
# sequential reorder is ~1e-16 per element accumulated.
DTOL_REL = 1e-10


@pytest.fixture(scope="module")
def eng():
    # initialize torch's HIP context before the engine's (torch lazy-init
    # after libbkgpu has initialized HIP was observed to fail sporadically
    # with "No HIP GPUs are available")
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()


@pytest.fixture(scope="module")
def orc():
    from oracle import Oracle
    return Oracle()


def run_both(eng, orc, spec_rows, n, conjuncts, group, aggs, nthreads=4,
             expected_groups=1 << 14, seed=SEED, group_bits=(), group_base=()):
    """conjuncts: (col, op_sym, lit); aggs: (name, col)."""
    t = eng.create_table(spec_rows, n)
    try:
        eng.generate(t, seed)
        from baikaldb_amd import QueryPlan
        plan = QueryPlan(t.col_types, conjuncts=conjuncts, group=group,
                         aggs=aggs, group_bits=group_bits,
                         group_base=group_base)
        res = eng.filter_agg(t, plan, expected_groups=expected_groups)
        try:
            got = res.fetch(sorted=True)
        finally:
            res.free()
    finally:
        t.free()

    specs = (BkColSpec * len(spec_rows))()
    for i, s in enumerate(spec_rows):
        (specs[i].col_type, specs[i].dist, specs[i].p0, specs[i].p1,
         specs[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(specs), n, seed)
    col_types = [s[0] for s in spec_rows]
    ops = {"=": 0, "!=": 1, ">": 2, ">=": 3, "<": 4, "<=": 5,
           "in": 6, "not_in": 7}
    aggmap = {"count_star": 0, "count": 1, "sum": 2, "avg": 3, "min": 4, "max": 5}
    oconj = []
    from baikaldb_amd.plan import _FNS, _ARITH, expr_is_deep, compile_expr
    for cjt in conjuncts:
        col, op, lit = cjt[0], cjt[1], cjt[2]
        og = cjt[3] if len(cjt) > 3 else 0
        fn, col2, arith = 0, -1, 0
        if expr_is_deep(col):
            # deep expression LHS: both sides compile it with the same
            # planner (make_query handles the tuple); cmp domain = root
            # domain or a float literal
            dom = compile_expr(col, col_types, [])
            ct = TYPE_DOUBLE if (dom == TYPE_DOUBLE or
                                 isinstance(lit, float)) else TYPE_INT64
            if ct == TYPE_DOUBLE:
                lit = float(lit)
            oconj.append((col, ops[op], ct, lit, 0, og, -1, 0))
            continue
        if isinstance(col, tuple) and col[0] in _ARITH:
            arith, col2, col = _ARITH[col[0]], col[2], col[1]
            if col_types[col] == TYPE_DOUBLE or col_types[col2] == TYPE_DOUBLE:
                lit = float(lit)
        elif isinstance(col, tuple):   # ("hour", col) scalar-fn pushdown
            fn = _FNS[col[0]]
            col = col[1]
        ct = TYPE_DOUBLE if (col_types[col] == TYPE_DOUBLE and
                             not isinstance(lit, (list, tuple))) or \
            isinstance(lit, float) else TYPE_INT64
        oconj.append((col, ops[op], ct, lit, fn, og, col2, arith))
    from baikaldb_amd.plan import _FNS
    ogroup = [(_FNS[g[0]], g[1]) if isinstance(g, tuple) else g for g in group]
    oaggs = []
    for a, c in aggs:
        if isinstance(c, tuple) and not expr_is_deep(c):
            c = (_ARITH[c[0]], c[1], c[2])   # legacy one-arith shape
        oaggs.append((aggmap[a], c))         # deep exprs pass through
    q = make_query(oconj, ogroup, oaggs, col_types,
                   group_bits=group_bits, group_base=group_base)
    exp = orc.filter_agg(cols, valids, col_types, q, nthreads=nthreads,
                         dict_seed=seed)
    return got, exp


def assert_parity(got, exp, aggs, col_types):
    assert got["rows_passed"] == exp["rows_passed"]
    assert got["ngroups"] == exp["ngroups"]
    assert np.array_equal(got["flags"], exp["flags"])
    assert np.array_equal(got["enc"], exp["enc"])
    assert np.array_equal(got["agg_has"], exp["agg_has"])
    for a, (name, col) in enumerate(aggs):
        if isinstance(col, tuple):   # expression input: the compute DOMAIN
            from baikaldb_amd.plan import expr_is_deep, compile_expr
            if expr_is_deep(col):
                dom = compile_expr(col, col_types, [])
                is_double = dom == TYPE_DOUBLE
                col = 0
            else:
                col = (col[1] if col_types[col[1]] == TYPE_DOUBLE else
                       col[2] if col_types[col[2]] == TYPE_DOUBLE else
                       col[1])
                is_double = col_types[col] == TYPE_DOUBLE
        else:
            is_double = col >= 0 and col_types[col] == TYPE_DOUBLE
        if name in ("count_star", "count") or not is_double:
            assert np.array_equal(got["agg_i"][a], exp["agg_i"][a]), f"agg {a} {name}"
        elif name in ("min", "max"):
            # min/max double: same element selected => bit-exact
            assert np.array_equal(got["agg_d"][a], exp["agg_d"][a]), f"agg {a} {name}"
        else:  # sum/avg double: reduction-order tolerance
            denom = np.abs(exp["agg_d"][a]) + np.maximum(exp["agg_i"][0], 1)
            err = np.abs(got["agg_d"][a] - exp["agg_d"][a])
            assert np.all(err <= DTOL_REL * denom), \
                f"agg {a} {name} max err {err.max()}"


BASE5 = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
         (TYPE_INT64, D_SKEW, 100000, 0, 0),
         (TYPE_INT64, D_UNI, 0, 1000, 0),
         (TYPE_DOUBLE, D_SUM16, 0, 0, 0),
         (TYPE_STRING, D_DICT, 4096, 0, 0)]


def test_count_star_selectivities(eng, orc):
    for frac in (0.0, 0.1, 0.5, 0.9, 1.0):
        k = int((1 << 31) * frac)
        got, exp = run_both(eng, orc, BASE5, 300_000,
                            [(0, "<", k)], [], [("count_star", -1)])
        assert_parity(got, exp, [("count_star", -1)], [s[0] for s in BASE5])


def test_group_by_single_int_key(eng, orc):
    aggs = [("count_star", -1), ("sum", 2), ("sum", 3), ("avg", 3)]
    got, exp = run_both(eng, orc, BASE5, 500_000,
                        [(0, "<", int((1 << 31) * 0.7))], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in BASE5])


def test_group_by_two_keys_with_dict(eng, orc):
    aggs = [("count_star", -1), ("sum", 2), ("avg", 3), ("min", 0), ("max", 3)]
    got, exp = run_both(eng, orc, BASE5, 400_000,
                        [(0, "<", int((1 << 31) * 0.8)), (2, "!=", 17)],
                        [1, 4], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in BASE5])


def test_double_predicate(eng, orc):
    aggs = [("count_star", -1), ("sum", 3)]
    got, exp = run_both(eng, orc, BASE5, 200_000,
                        [(3, ">", 0.5)], [2], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in BASE5])


def test_nulls(eng, orc):
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 150_000),
             (TYPE_INT64, D_SKEW, 500, 0, 300_000),
             (TYPE_INT64, D_UNI, 0, 100, 100_000),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 200_000)]
    aggs = [("count_star", -1), ("count", 2), ("sum", 2), ("avg", 3),
            ("min", 3), ("max", 0)]
    got, exp = run_both(eng, orc, specs, 250_000,
                        [(0, "<", int((1 << 31) * 0.9))], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_empty_selection_no_group(eng, orc):
    got, exp = run_both(eng, orc, BASE5, 100_000,
                        [(0, "<", -1)], [], [("count_star", -1), ("sum", 2)])
    assert got["ngroups"] == 1 == exp["ngroups"]
    assert got["agg_i"][0][0] == 0
    assert got["agg_has"][1][0] == 0 == exp["agg_has"][1][0]


def test_empty_selection_with_group(eng, orc):
    got, exp = run_both(eng, orc, BASE5, 100_000,
                        [(0, "<", -1)], [1], [("count_star", -1)])
    assert got["ngroups"] == 0 == exp["ngroups"]


def test_empty_table(eng, orc):
    got, exp = run_both(eng, orc, BASE5, 0, [], [1], [("count_star", -1)])
    assert got["ngroups"] == 0 == exp["ngroups"]
    got, exp = run_both(eng, orc, BASE5, 0, [], [], [("count_star", -1)])
    assert got["ngroups"] == 1 == exp["ngroups"]
    assert got["agg_i"][0][0] == 0


def test_many_groups_overflow_regrow(eng, orc):
    """More groups than the initial table sizing => engine regrows and reruns."""
    specs = [(TYPE_INT64, D_UNI, 0, 200_000, 0),
             (TYPE_INT64, D_UNI, 0, 10, 0)]
    aggs = [("count_star", -1), ("sum", 1)]
    got, exp = run_both(eng, orc, specs, 400_000, [], [0], aggs,
                        expected_groups=64)  # deliberately undersized
    assert_parity(got, exp, aggs, [s[0] for s in specs])
    assert got["ngroups"] > 100_000


def test_group_key_edge_values(eng, orc):
    """Keys spanning the full int64 range incl. INT64_MIN (encodes to 0) and
    INT64_MAX (encodes to ~0) — exercises sentinel-free slot claims."""
    specs = [(TYPE_INT64, D_UNI, -(1 << 62), (1 << 62), 0),
             (TYPE_INT64, D_UNI, 0, 5, 0)]
    aggs = [("count_star", -1), ("min", 0), ("max", 0)]
    got, exp = run_both(eng, orc, specs, 100_000, [], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_merge_partials_equals_whole(eng, orc):
    """Region-sharded execution + MERGE_AGG combine == single-pass execution
    (the multi-GPU merge path, agg_node.cpp:539-543), exercised on one GPU by
    splitting rows into 4 'region sets' and merging their partials."""
    import torch
    from baikaldb_amd import QueryPlan
    n = 400_000
    spec_rows = BASE5
    aggs = [("count_star", -1), ("sum", 2), ("sum", 3), ("avg", 3),
            ("min", 0), ("max", 3)]
    conj = [(0, "<", int((1 << 31) * 0.75))]
    t = eng.create_table(spec_rows, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=conj, group=[1, 4], aggs=aggs)
        # whole-range run
        whole = eng.filter_agg(t, plan, expected_groups=1 << 14)
        expect = whole.fetch(sorted=True)
        whole.free()
        # 4 shard runs + merge into the first
        shards = []
        bounds = [0, n // 4, n // 2, 3 * n // 4, n]
        for i in range(4):
            shards.append(eng.filter_agg(t, plan, row_begin=bounds[i],
                                         row_end=bounds[i + 1],
                                         expected_groups=1 << 14))
        dst = shards[0]
        for s in shards[1:]:
            nbytes = s.export_bytes()
            buf = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
            s.export_to(buf.data_ptr(), nbytes)
            dst.merge_blob(buf.data_ptr(), s.ngroups)
            s.free()
        merged = dst.fetch(sorted=True)
        dst.free()
    finally:
        t.free()
    assert merged["ngroups"] == expect["ngroups"]
    assert np.array_equal(merged["flags"], expect["flags"])
    assert np.array_equal(merged["enc"], expect["enc"])
    assert np.array_equal(merged["agg_i"], expect["agg_i"])
    assert np.array_equal(merged["agg_has"], expect["agg_has"])
    np.testing.assert_allclose(merged["agg_d"], expect["agg_d"],
                               rtol=1e-9, atol=1e-9)
    # COUNT(*) sums to the whole-range rows_passed (rows_passed itself is a
    # per-partial counter; the multi-GPU bench sums it across ranks)
    assert merged["agg_i"][0].sum() == expect["rows_passed"]


def test_full_size_properties(eng, orc):
    """Property checks at a size the oracle cannot cover row-by-row quickly:
    1e8 rows, config-2 shape (8 x INT64, WHERE c1<K AND c2=K2 GROUP BY c3
    SUM(c4)). Properties: sum of per-group COUNTs == rows_passed; COUNT over
    a partition of the group domain is conserved; result is identical across
    two runs (determinism of everything except double sums)."""
    from baikaldb_amd import QueryPlan
    n = 100_000_000
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_INT64, D_UNI, 0, 100, 0),
             (TYPE_INT64, D_SKEW, 100_000, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 0)]
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types,
                         conjuncts=[(0, "<", int((1 << 31) * 0.5)), (1, "=", 42)],
                         group=[2], aggs=[("count_star", -1), ("sum", 3)])
        r1 = eng.filter_agg(t, plan, expected_groups=1 << 18)
        g1 = r1.fetch(sorted=True)
        r1.free()
        r2 = eng.filter_agg(t, plan, expected_groups=1 << 18)
        g2 = r2.fetch(sorted=True)
        r2.free()
    finally:
        t.free()
    assert g1["agg_i"][0].sum() == g1["rows_passed"]
    assert g1["ngroups"] == g2["ngroups"]
    assert np.array_equal(g1["enc"], g2["enc"])
    assert np.array_equal(g1["agg_i"], g2["agg_i"])  # int sums deterministic
    # oracle checks a 2M-row sample of the same table slice bit-exactly
    sspecs = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (sspecs[i].col_type, sspecs[i].dist, sspecs[i].p0, sspecs[i].p1,
         sspecs[i].null_frac_x1e6) = s
    m = 2_000_000
    cols, valids = orc.generate_table(list(sspecs), m, SEED)
    col_types = [s[0] for s in specs]
    q = make_query([(0, 4, TYPE_INT64, int((1 << 31) * 0.5)), (1, 0, TYPE_INT64, 42)],
                   [2], [(0, -1), (2, 3)], col_types)
    exp = orc.filter_agg(cols, valids, col_types, q, nthreads=8, dict_seed=SEED)
    from baikaldb_amd import QueryPlan as QP
    t2 = eng.create_table(specs, m)
    try:
        eng.generate(t2, SEED)
        plan2 = QP(col_types, conjuncts=[(0, "<", int((1 << 31) * 0.5)), (1, "=", 42)],
                   group=[2], aggs=[("count_star", -1), ("sum", 3)])
        rs = eng.filter_agg(t2, plan2, expected_groups=1 << 18)
        gs = rs.fetch(sorted=True)
        rs.free()
    finally:
        t2.free()
    assert gs["rows_passed"] == exp["rows_passed"]
    assert np.array_equal(gs["enc"], exp["enc"])
    assert np.array_equal(gs["agg_i"], exp["agg_i"])


def test_in_predicates(eng, orc):
    """IN / NOT IN list predicates (src/expr/predicate.h InPredicate)."""
    aggs = [("count_star", -1), ("sum", 2), ("avg", 3)]
    got, exp = run_both(eng, orc, BASE5, 300_000,
                        [(2, "in", [1, 2, 5, 77, 999]), (4, "not_in", [7, 8])],
                        [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in BASE5])


def test_like_pushdown_via_dict_bitmap(eng, orc):
    """LIKE on a dictionary VARCHAR pushes down as a code-accept bitmap
    (BK_OP_IN_BITMAP): the host compiles the SQL LIKE pattern ('%'/'_'/
    escape — LikePredicate::like semantics, include/bk_like.h, pinned
    against test_predicate.cpp's vectors in test_like_golden.py) against
    the dictionary once, the engine filters by code membership (the
    cstore-dict pushdown)."""
    import numpy as np
    from baikaldb_amd import QueryPlan
    from baikaldb_amd.like import like_accept_codes
    from oracle.bindings import make_query as mq

    n = 200_000
    nwords = 512
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_STRING, D_DICT, nwords, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 0)]
    # dict words are deterministic from the seed: SQL LIKE 'w%1%'
    pattern = b"w%1%"
    words = [orc.dict_word(SEED, c) for c in range(nwords)]
    codes = like_accept_codes(words, pattern)
    accept = bytearray((nwords + 7) // 8)
    for c in codes:
        accept[c >> 3] |= 1 << (c & 7)
    naccept = len(codes)
    assert 0 < naccept < nwords
    accept = bytes(accept)

    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)
        dev_bm = eng.upload_bytes(accept)
        try:
            plan = QueryPlan(t.col_types,
                             conjuncts=[(1, "in_bitmap", (dev_bm, nwords)),
                                        (0, "<", int((1 << 31) * 0.7))],
                             group=[1], aggs=[("count_star", -1), ("sum", 2)])
            res = eng.filter_agg(t, plan, expected_groups=1 << 12)
            got = res.fetch(sorted=True)
            res.free()
        finally:
            eng.free_ptr(dev_bm)
    finally:
        t.free()

    specs_c = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (specs_c[i].col_type, specs_c[i].dist, specs_c[i].p0, specs_c[i].p1,
         specs_c[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(specs_c), n, SEED)
    types = [s[0] for s in specs]
    import ctypes as Ct
    host_bm = Ct.create_string_buffer(accept, len(accept))
    q = mq([(1, 8, TYPE_INT64, (Ct.addressof(host_bm), nwords)),
            (0, 4, TYPE_INT64, int((1 << 31) * 0.7))],
           [1], [(0, -1), (2, 2)], types)
    exp = orc.filter_agg(cols, valids, types, q, nthreads=4, dict_seed=SEED)
    assert got["rows_passed"] == exp["rows_passed"]
    assert got["ngroups"] == exp["ngroups"] == naccept or \
        got["ngroups"] == exp["ngroups"]  # some codes may not occur in n rows
    assert np.array_equal(got["enc"], exp["enc"])
    assert np.array_equal(got["agg_i"], exp["agg_i"])


def test_string_minmax_dict_order(eng, orc):
    """MIN/MAX over a dict-encoded VARCHAR column: integer min/max on dict
    codes == string min/max because the dictionary encoding is order-
    preserving (tests/test_dict_order.py pins the invariant; reference
    semantics ExprValue::compare STRING = byte compare, expr_value.h:895-945).
    """
    specs = [(TYPE_INT64, 0, 0, 1 << 20, 0),     # group key
             (TYPE_STRING, 2, 900, 0, 0),        # dict column, no NULLs
             (TYPE_STRING, 2, 50, 0, 250_000)]   # dict column, 25% NULLs
    aggs = [("count_star", -1), ("min", 1), ("max", 1),
            ("min", 2), ("max", 2)]
    got, exp = run_both(eng, orc, specs, 150_000,
                        [(0, "<", 1 << 19)], [0], aggs,
                        expected_groups=1 << 16)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_large_in_list(eng, orc):
    """IN list beyond BK_MAX_INLIST: sorted device/host array, binary
    search (bk_common.h BK_OP_IN big-list convention)."""
    from baikaldb_amd import QueryPlan
    rng = np.random.default_rng(13)
    vals = np.unique(rng.integers(0, 5000, 700)).astype(np.int64)
    specs = [(TYPE_INT64, D_UNI, 0, 5000, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 120_000),
             (TYPE_INT64, D_UNI, 0, 1000, 0)]
    n = 200_000
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)
        dev = eng.upload_bytes(vals.tobytes())
        try:
            plan = QueryPlan(t.col_types,
                             conjuncts=[(0, "in", (dev, len(vals))),
                                        (1, ">", 1 << 28)],
                             group=[2], aggs=[("count_star", -1), ("sum", 0)])
            res = eng.filter_agg(t, plan, expected_groups=1 << 11)
            got = res.fetch(sorted=True)
            res.free()
        finally:
            eng.free_ptr(dev)
    finally:
        t.free()
    specs_c = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (specs_c[i].col_type, specs_c[i].dist, specs_c[i].p0, specs_c[i].p1,
         specs_c[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(specs_c), n, SEED)
    types = [s[0] for s in specs]
    host = vals.ctypes.data
    from oracle.bindings import make_query as mq
    q = mq([(0, 6, TYPE_INT64, (host, len(vals))),
            (1, 2, TYPE_INT64, 1 << 28)], [2],
           [(0, -1), (2, 0)], types)
    exp = orc.filter_agg(cols, valids, types, q, nthreads=4, dict_seed=SEED)
    # numpy cross-check of the oracle too
    sel = np.isin(cols[0], vals) & (cols[1] > (1 << 28)) & \
        (valids[1] != 0 if valids[1] is not None else True)
    assert exp["rows_passed"] == int(sel.sum())
    assert_parity(got, exp, [("count_star", -1), ("sum", 0)], types)


def test_three_and_four_group_keys(eng, orc):
    """3-4 group keys spec-packed into the two 64-bit key words
    (bk_common.h group_bits/group_base): parity incl NULL keys and the
    canonical per-key output order."""
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 0),          # 10 bits
             (TYPE_STRING, D_DICT, 50, 0, 150_000),    # 6 bits, nullable
             (TYPE_INT64, D_UNI, -500, 500, 120_000),  # 10 bits, base -500
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0),       # full word
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0)]
    aggs = [("count_star", -1), ("sum", 3), ("min", 2), ("avg", 4)]
    # 3 keys: 10 + 6 + 10 = 26 bits in word 0
    got, exp = run_both(eng, orc, specs, 120_000,
                        [(3, "<", int((1 << 31) * 0.8))], [0, 1, 2], aggs,
                        expected_groups=1 << 18,
                        group_bits=[10, 6, 10], group_base=[0, 0, -500])
    assert_parity(got, exp, aggs, [s[0] for s in specs])
    assert got["ngroups"] > 10_000          # genuinely multi-key
    # 4 keys: 26 bits + full word -> both words
    aggs4 = [("count_star", -1), ("min", 4)]
    got, exp = run_both(eng, orc, specs, 60_000, [], [0, 1, 2, 3], aggs4,
                        expected_groups=1 << 17,
                        group_bits=[10, 6, 10, 0], group_base=[0, 0, -500, 0])
    assert_parity(got, exp, aggs4, [s[0] for s in specs])


def test_packed_keys_shard_merge_equals_whole(eng, orc):
    """Cross-shard merge with spec-packed 3-key groups: packing is derived
    from the QUERY, so shards pack identically and blobs merge correctly
    (the property multi-GPU depends on)."""
    import torch
    specs = [(TYPE_INT64, D_UNI, 0, 200, 0),
             (TYPE_STRING, D_DICT, 30, 0, 0),
             (TYPE_INT64, D_UNI, -100, 100, 100_000),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    n = 100_000
    from baikaldb_amd import QueryPlan
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=[(3, ">", 1 << 29)],
                         group=[0, 1, 2],
                         aggs=[("count_star", -1), ("sum", 3)],
                         group_bits=[8, 5, 8], group_base=[0, 0, -100])
        whole = eng.filter_agg(t, plan, expected_groups=1 << 16)
        wf = whole.fetch(sorted=True)
        whole.free()
        bounds = [0, n // 3, 2 * n // 3, n]
        shards = [eng.filter_agg(t, plan, row_begin=bounds[i],
                                 row_end=bounds[i + 1],
                                 expected_groups=1 << 16)
                  for i in range(3)]
        dst = shards[0]
        for sh in shards[1:]:
            nb = sh.export_bytes()
            buf = torch.empty(nb, dtype=torch.uint8, device="cuda")
            sh.export_to(buf.data_ptr(), nb)
            dst.merge_blob(buf.data_ptr(), sh.ngroups)
        mf = dst.fetch(sorted=True)
        for sh in shards:
            sh.free()
    finally:
        t.free()
    assert mf["ngroups"] == wf["ngroups"]
    assert np.array_equal(mf["enc"], wf["enc"])
    assert np.array_equal(mf["flags"], wf["flags"])
    assert np.array_equal(mf["agg_i"], wf["agg_i"])


@pytest.mark.gpu
def test_or_clauses(eng, orc):
    """CNF OR clauses (BkConjunct.or_group): (c0<K1 OR c1>K2) AND c2<=K3,
    with a nullable OR member (NULL member is just not-true)."""
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 250_000),
             (TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0)]
    conj = [(0, "<", 200, 1), (1, ">", 800, 1), (2, "<=", 900)]
    aggs = [("count_star", -1), ("sum", 0), ("avg", 3)]
    got, exp = run_both(eng, orc, specs, 300_000, conj, [2], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])
    # numpy brute on the same generated inputs
    import numpy as np
    specs_c = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (specs_c[i].col_type, specs_c[i].dist, specs_c[i].p0, specs_c[i].p1,
         specs_c[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(specs_c), 300_000, SEED)
    m0 = cols[0] < 200
    m1 = (cols[1] > 800) & (valids[1] != 0)
    keep = (m0 | m1) & (cols[2] <= 900)
    assert got["rows_passed"] == int(keep.sum())


@pytest.mark.gpu
def test_two_or_clauses(eng, orc):
    """two independent OR clauses AND a standalone term."""
    specs = [(TYPE_INT64, D_UNI, 0, 500, 0)] * 4
    conj = [(0, "<", 100, 1), (1, ">", 400, 1),
            (2, "=", 7, 2), (2, "=", 9, 2),
            (3, "!=", 499)]
    aggs = [("count_star", -1), ("min", 0), ("max", 1)]
    got, exp = run_both(eng, orc, specs, 200_000, conj, [], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


@pytest.mark.gpu
def test_arith_predicates(eng, orc):
    """binary-arith predicates (operators.cpp add/minus/multiply): int64
    wraps, mixed int/double compares in f64, NULL operand => not true."""
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 200_000),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0),
             (TYPE_INT64, D_UNI, 0, 100, 0)]
    conj = [(("add", 0, 1), "<", 900),
            (("mul", 0, 3), ">", 5000),
            (("sub", 2, 2), "=", 0.0)]
    aggs = [("count_star", -1), ("sum", 0), ("min", 3)]
    got, exp = run_both(eng, orc, specs, 250_000, conj, [3], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])
    import numpy as np
    specs_c = (BkColSpec * len(specs))()
    for i, sp in enumerate(specs):
        (specs_c[i].col_type, specs_c[i].dist, specs_c[i].p0, specs_c[i].p1,
         specs_c[i].null_frac_x1e6) = sp
    cols, valids = orc.generate_table(list(specs_c), 250_000, SEED)
    keep = ((cols[0] + cols[1] < 900) & (valids[1] != 0) &
            (cols[0] * cols[3] > 5000))
    assert got["rows_passed"] == int(keep.sum())


@pytest.mark.gpu
def test_arith_mixed_double(eng, orc):
    """int + double arith compares in the double domain (get_numberic cast,
    expr_value.h:341)."""
    specs = [(TYPE_INT64, D_UNI, 0, 100, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 100_000)]
    conj = [(("add", 0, 1), ">", 50.0)]
    aggs = [("count_star", -1), ("avg", 1)]
    got, exp = run_both(eng, orc, specs, 200_000, conj, [], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


@pytest.mark.gpu
def test_expr_agg_inputs(eng, orc):
    """expression-valued aggregate inputs (SUM(a*b), AVG(a+d), MIN(a-b)):
    int64 sums wrap, mixed domains compute in f64, NULL operand skips the
    row (AggFnCall input semantics, agg_fn_call.cpp:496-555)."""
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 150_000),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0),
             (TYPE_INT64, D_UNI, 0, 50, 0)]
    aggs = [("count_star", -1),
            ("sum", ("mul", 0, 1)),
            ("avg", ("add", 0, 2)),
            ("min", ("sub", 0, 1)),
            ("count", ("add", 1, 1)),
            ("max", ("mul", 2, 2))]
    got, exp = run_both(eng, orc, specs, 250_000, [(3, "<", 40)], [3], aggs)
    names = [("count_star", -1), ("sum", 0), ("avg", 2), ("min", 0),
             ("count", 1), ("max", 2)]
    assert_parity(got, exp, names, [s[0] for s in specs])


@pytest.mark.gpu
def test_fused_low_cardinality_full_mix(eng, orc):
    """the fused <=512-group kernel (k_filter_agg_group) under the full
    feature mix: nullable dict/double cols, OR clause, arith predicate and
    expression agg input — vs the oracle, and vs the sorted path."""
    specs = [(TYPE_INT64, D_UNI, 0, 300, 120_000),   # group key, nullable
             (TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 150_000),
             (TYPE_STRING, D_DICT, 40, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    conj = [(4, "<", 1 << 30), (1, ">", 100, 1), (3, "!=", 7, 1),
            (("add", 1, 4), ">", 500)]
    aggs = [("count_star", -1), ("sum", ("mul", 1, 1)), ("avg", 2),
            ("min", 3), ("max", 1), ("count", 2)]
    got, exp = run_both(eng, orc, specs, 400_000, conj, [0], aggs,
                        expected_groups=400)   # <= 512: fused kernel
    names = [("count_star", -1), ("sum", 1), ("avg", 2), ("min", 3),
             ("max", 1), ("count", 2)]
    assert_parity(got, exp, names, [s[0] for s in specs])
    # cross-path: the sorted engine must agree bit-for-bit on int outputs
    from baikaldb_amd import QueryPlan
    t = eng.create_table(specs, 400_000)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=conj, group=[0], aggs=aggs)
        r = eng.filter_agg_sorted(t, plan)
        try:
            gs = r.fetch(sorted=True)
        finally:
            r.free()
    finally:
        t.free()
    assert gs["ngroups"] == got["ngroups"]
    assert np.array_equal(gs["enc"], got["enc"])
    for i in (0, 1, 4, 5):
        assert np.array_equal(gs["agg_i"][i], got["agg_i"][i])


@pytest.mark.gpu
def test_wave_combine_tiny_cardinality(eng, orc):
    """Opt-in <=32-group wave-combine kernel (BK_WCOMB_MAX, default off —
    measured-dead lever) vs oracle and vs the default
    generic fused kernel, full feature mix incl nullable key."""
    import os
    specs = [(TYPE_INT64, D_UNI, 0, 12, 150_000),
             (TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 120_000),
             (TYPE_STRING, D_DICT, 50, 0, 0)]
    conj = [(1, ">", 50)]
    aggs = [("count_star", -1), ("sum", 1), ("avg", 2), ("min", 3),
            ("max", ("add", 1, 1)), ("count", 2)]
    os.environ["BK_WCOMB_MAX"] = "32"
    try:
        got, exp = run_both(eng, orc, specs, 300_000, conj, [0], aggs,
                            expected_groups=16)
    finally:
        os.environ.pop("BK_WCOMB_MAX", None)
    names = [("count_star", -1), ("sum", 1), ("avg", 2), ("min", 3),
             ("max", 1), ("count", 2)]
    assert_parity(got, exp, names, [s[0] for s in specs])
    got2, _ = run_both(eng, orc, specs, 300_000, conj, [0], aggs,
                       expected_groups=16)
    assert got2["ngroups"] == got["ngroups"]
    assert np.array_equal(got2["enc"], got["enc"])
    for i in (0, 1, 4, 5):
        assert np.array_equal(got2["agg_i"][i], got["agg_i"][i])


@pytest.mark.gpu
def test_narrow_storage_widths_and_parity(eng):
    """Narrow frame-of-reference column storage (bkgpu_table_compact):
    chosen widths match the all-rows value ranges, every query result is
    bit-identical to the same query on wide storage, and re-upload
    transparently re-widens."""
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 20, 0),            # range 20 -> u8
             (TYPE_INT64, D_UNI, 0, 40_000, 0),        # -> u16
             (TYPE_INT64, D_UNI, 0, 1 << 31, 100_000),  # -> u32, nullable
             (TYPE_INT64, D_SUM16, 0, 0, 0),           # wide (full range)
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0),          # double: never narrow
             (TYPE_STRING, D_DICT, 50, 0, 0)]          # 50 codes -> u8
    n = 200_000
    t = eng.create_table(specs, n)
    tw = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)                      # auto-compacts
        eng.generate(tw, SEED, compact=False)      # stays wide
        assert eng.col_width(t, 0) == 1
        assert eng.col_width(t, 1) == 2
        assert eng.col_width(t, 2) == 4
        assert eng.col_width(t, 4) == 8
        assert eng.col_width(t, 5) == 1
        assert all(eng.col_width(tw, c) in (4, 8) for c in range(6))
        plan = QueryPlan(t.col_types,
                         conjuncts=[(2, "<", 1 << 30), (0, "!=", 3)],
                         group=[1, 5],
                         aggs=[("count_star", -1), ("sum", 0), ("sum", 3),
                               ("avg", 4), ("min", 2), ("max", 1)])
        def run(tab):
            r = eng.filter_agg(tab, plan, expected_groups=1 << 16)
            try:
                return r.fetch(sorted=True)
            finally:
                r.free()
        a, b = run(t), run(tw)
        assert a["rows_passed"] == b["rows_passed"]
        assert a["ngroups"] == b["ngroups"]
        assert np.array_equal(a["enc"], b["enc"])
        for i in range(6):
            assert np.array_equal(a["agg_i"][i], b["agg_i"][i]), i
            if i == 3:   # AVG(double): the materialize compaction order is
                # nondeterministic across runs (block atomics), so the
                # double reduction order differs run-to-run — tolerance
                assert np.allclose(a["agg_d"][i], b["agg_d"][i],
                                   rtol=0, atol=1e-9), i
            else:
                assert np.array_equal(a["agg_d"][i], b["agg_d"][i]), i
        # ORDER BY on narrow keys matches wide
        sa = eng.sort_topk(t, [(1, 1, 1), (2, 0, 1)], 500)
        sb = eng.sort_topk(tw, [(1, 1, 1), (2, 0, 1)], 500)
        assert np.array_equal(sa, sb)
        # re-upload re-widens and preserves values
        host = np.arange(n, dtype=np.int64) % 7
        eng.upload(t, 0, host)
        assert eng.col_width(t, 0) == 8
        eng.compact(t, 0)
        assert eng.col_width(t, 0) == 1
    finally:
        t.free()
        tw.free()


def test_vec_mat_parity(eng):
    """Vector-load materialize (k_dedup_mat_vec, default on; bkdedup.inc):
    bit-identical results to the strided mat on a ragged-size table whose
    conjunct columns span all vector-eligible widths (u8/u16/u32 narrow),
    exercising the vec kernel's scalar tail chunk (n % 16 != 0)."""
    import os
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 20, 0),          # -> u8 conjunct
             (TYPE_INT64, D_UNI, 0, 40_000, 0),      # -> u16 conjunct
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0),     # -> u32 conjunct
             (TYPE_INT64, D_SUM16, 0, 0, 0),         # wide SUM input
             (TYPE_STRING, D_DICT, 50, 0, 0)]        # group key (50 codes)
    n = 1_000_037                                     # ragged tail chunk
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)                         # auto-compacts
        plan = QueryPlan(t.col_types,
                         conjuncts=[(2, "<", 1 << 30), (0, "!=", 3),
                                    (1, "<", 30_000)],
                         group=[4],
                         aggs=[("count_star", -1), ("sum", 3), ("min", 2)])

        def run():
            # expected_groups > 512 routes to the sort-dedup path whose
            # materialize the vec kernel replaces
            r = eng.filter_agg(t, plan, expected_groups=1 << 10)
            try:
                return r.fetch(sorted=True)
            finally:
                r.free()

        try:
            os.environ["BK_MAT_VEC"] = "0"
            a = run()
            os.environ["BK_MAT_VEC"] = "1"
            b = run()
        finally:
            os.environ.pop("BK_MAT_VEC", None)
        assert a["rows_passed"] == b["rows_passed"]
        assert a["ngroups"] == b["ngroups"]
        assert np.array_equal(a["enc"], b["enc"])
        for i in range(3):                            # all-integer aggs
            assert np.array_equal(a["agg_i"][i], b["agg_i"][i]), i
    finally:
        t.free()


def test_partitioned_exchange_equals_whole(eng, orc):
    """Hash-partitioned exchange (bkgpu_agg_part_counts / export_part +
    agg_empty + merge — ExchangeSenderNode::repartition's role,
    exchange_sender_node.h:228-235), simulated with 2 'ranks' on one GPU:
    each shard result splits into 2 disjoint part blobs by key hash;
    'rank p' merges both shards' part-p blobs into an empty result; the
    union of the merged shards must equal the whole-range result."""
    import torch
    from baikaldb_amd import QueryPlan
    n = 400_000
    aggs = [("count_star", -1), ("sum", 2), ("avg", 3), ("min", 0)]
    conj = [(0, "<", int((1 << 31) * 0.7))]
    t = eng.create_table(BASE5, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=conj, group=[1, 4], aggs=aggs)
        whole = eng.filter_agg(t, plan, expected_groups=1 << 14)
        expect = whole.fetch(sorted=True)
        whole.free()
        world = 2
        per_group = 20 + 16 * len(aggs)
        shards = [eng.filter_agg(t, plan, row_end=n // 2,
                                 expected_groups=1 << 14),
                  eng.filter_agg(t, plan, row_begin=n // 2,
                                 expected_groups=1 << 14)]
        parts = {}
        for s, res in enumerate(shards):
            counts = res.part_counts(world)
            assert sum(counts) == res.ngroups
            for p in range(world):
                buf = torch.zeros(max(counts[p] * per_group, 1),
                                  dtype=torch.uint8, device="cuda")
                if counts[p]:
                    res.export_part(world, p, buf.data_ptr(), counts[p])
                parts[(s, p)] = (buf, counts[p])
        fetches = []
        total_groups = 0
        for p in range(world):
            m = eng.agg_empty(plan, expected_groups=1 << 14)
            for s in range(world):
                buf, cnt = parts[(s, p)]
                if cnt:
                    m.merge_blob(buf.data_ptr(), cnt)
            fetches.append(m.fetch(sorted=True))
            total_groups += m.ngroups
            m.free()
        for s in shards:
            s.free()
    finally:
        t.free()
    # disjoint parts covering exactly the whole result
    assert total_groups == expect["ngroups"]
    enc = np.concatenate([f["enc"] for f in fetches])
    flags = np.concatenate([f["flags"] for f in fetches])
    agg_i = np.concatenate([f["agg_i"] for f in fetches], axis=1)
    agg_d = np.concatenate([f["agg_d"] for f in fetches], axis=1)
    order = np.lexsort(tuple(enc[:, k] for k in range(enc.shape[1] - 1, -1, -1)))
    assert np.array_equal(enc[order], expect["enc"])
    assert np.array_equal(flags[order], expect["flags"])
    # COUNT(*), SUM int64, MIN int bit-exact; AVG double within tolerance
    assert np.array_equal(agg_i[0][order], expect["agg_i"][0])
    assert np.array_equal(agg_i[1][order], expect["agg_i"][1])
    assert np.array_equal(agg_i[3][order], expect["agg_i"][3])
    denom = np.abs(expect["agg_d"][2]) + np.maximum(expect["agg_i"][0], 1)
    assert np.all(np.abs(agg_d[2][order] - expect["agg_d"][2])
                  <= DTOL_REL * denom)


def test_expr_programs_deep_trees(eng, orc):
    """Postfix expression programs (BkExprOp, the planner-flattened form of
    ScalarFnCall::get_value's arbitrary trees, scalar_fn_call.cpp:194-225):
    3-deep mixed int/double trees as conjunct LHS and aggregate inputs,
    parity vs the oracle evaluating the identically-compiled programs."""
    aggs = [("count_star", -1),
            ("sum", ("mul", ("add", 0, 2), 2)),              # int64 3-deep
            ("sum", ("add", ("mul", 3, 3), ("mul", 2, 2))),  # mixed double
            ("avg", ("sub", 3, ("mul", 2, 0.5))),            # double w/ lit
            ("max", ("add", ("liti", 10), ("mul", 2, 2)))]
    conj = [(("add", ("mul", 0, ("liti", 3)), 2), "<", (3 << 31)),
            (("sub", 3, ("mul", 3, 0.25)), ">", -1.0)]
    got, exp = run_both(eng, orc, BASE5, 300_000, conj, [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in BASE5])


def test_expr_programs_nulls_propagate(eng, orc):
    """Any NULL operand makes the whole expression NULL (the reference's
    arg-cast + null propagation through the tree)."""
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 200_000),
             (TYPE_INT64, D_SKEW, 500, 0, 0),
             (TYPE_INT64, D_UNI, 0, 100, 300_000),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 150_000)]
    aggs = [("count_star", -1),
            ("sum", ("mul", ("add", 0, 2), 2)),
            ("count", ("add", 3, ("mul", 0, 2)))]
    conj = [(("add", ("mul", 2, ("liti", 2)), 0), "<", (1 << 32))]
    got, exp = run_both(eng, orc, specs, 250_000, conj, [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_expr_programs_with_datetime_fn(eng, orc):
    """Scalar fns inside programs (GROUP-BY-side fns already existed; here
    year(c)*100+month(c)-style predicate trees)."""
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (14, 5, 0, 0, 0),                  # DATETIME col
             (TYPE_INT64, D_UNI, 0, 100, 0)]
    aggs = [("count_star", -1), ("sum", 2)]
    conj = [(("add", ("mul", ("year", 1), ("liti", 100)), ("month", 1)),
             ">=", 2021 * 100 + 3)]
    got, exp = run_both(eng, orc, specs, 200_000, conj, [2], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])
