# Non-dictionary VARCHAR (VERDICT round-1 missing #5): arbitrary strings
# enter through bkgpu_table_upload_strings, which builds the
# order-preserving dictionary host-side (code order == byte order — the
# parquet/cstore ingest policy), so string GROUP BY / MIN / MAX / ORDER BY
# and RANGE predicates execute as integer code ops with the reference's
# string semantics (ExprValue STRING compare, expr_value.h; string keys
# mut_table_key.h:196-208). Checked against independent numpy/python
# brute-force recomputes.
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13
SEED = 424242


@pytest.fixture(scope="module")
def eng():
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()


def make_table(eng, n, rng, null_frac=0.0):
    from baikaldb_amd.engine import GpuTable
    words_pool = ["", "a", "apple", "applesauce", "banana", "zebra",
                  "Zebra", "middle", "mid", "mule",
                  "x" * 40, "yard", "yarn", "m", "mz"]
    strings = [words_pool[i] for i in rng.integers(0, len(words_pool), n)]
    v = rng.integers(0, 1000, n).astype(np.int64)
    valid = None
    if null_frac:
        valid = (rng.random(n) > null_frac).astype(np.uint8)
    t = eng.create_table([(TYPE_STRING, 2, 4, 0, 1 if null_frac else 0),
                          (TYPE_INT64, 0, 0, 1000, 0)], n)
    eng.upload_strings(t, 0, strings, valid)
    eng.upload(t, 1, v)
    return t, strings, v, valid


def test_group_by_arbitrary_strings(eng):
    from baikaldb_amd import QueryPlan
    rng = np.random.default_rng(1)
    n = 120_000
    t, strings, v, _ = make_table(eng, n, rng)
    try:
        plan = QueryPlan(t.col_types, conjuncts=[(1, "<", 800)], group=[0],
                         aggs=[("count_star", -1), ("sum", 1)])
        res = eng.filter_agg(t, plan, expected_groups=64)
        got = res.fetch(sorted=True)
        res.free()
        # decode group keys back to words
        words = [eng.dict_word(t, 0, int(got["enc"][g][0]))
                 for g in range(got["ngroups"])]
        assert all(w is not None for w in words)
    finally:
        t.free()
    # brute force
    exp = {}
    for s, val in zip(strings, v):
        if val < 800:
            c, t2 = exp.get(s, (0, 0))
            exp[s] = (c + 1, t2 + int(val))
    assert sorted(words) == sorted(exp.keys())
    # dict codes are order-preserving: fetch(sorted) ordering == byte order
    assert words == sorted(exp.keys())
    for g, w in enumerate(words):
        assert got["agg_i"][0][g] == exp[w][0], w
        assert got["agg_i"][1][g] == exp[w][1], w


def test_string_range_predicates(eng):
    """s < 'middle' etc. via dict_code lower_bound — integer code compare
    reproduces the reference's byte-wise string order exactly."""
    from baikaldb_amd import QueryPlan
    rng = np.random.default_rng(2)
    n = 80_000
    t, strings, v, _ = make_table(eng, n, rng)
    try:
        for lit, op, pyop in [("middle", "<", lambda s: s < "middle"),
                              ("middle", ">=", lambda s: s >= "middle"),
                              ("banana", "<=", lambda s: s <= "banana"),
                              ("mz", ">", lambda s: s > "mz")]:
            lb = eng.dict_code(t, 0, lit, mode=1)
            exact = eng.dict_code(t, 0, lit, mode=0)
            # < L  -> code <  lb ; >= L -> code >= lb
            # <= L -> code <= (exact if present else lb-1)
            # > L  -> code >  (exact if present else lb-1)
            if op in ("<", ">="):
                code_lit = lb
            else:
                code_lit = exact if exact >= 0 else lb - 1
            plan = QueryPlan(t.col_types, conjuncts=[(0, op, code_lit)],
                             aggs=[("count_star", -1), ("sum", 1)])
            res = eng.filter_agg(t, plan, expected_groups=4)
            got = res.fetch()
            res.free()
            mask = np.array([pyop(s) for s in strings])
            assert got["agg_i"][0][0] == mask.sum(), (lit, op)
            assert got["agg_i"][1][0] == int(v[mask].sum()), (lit, op)
    finally:
        t.free()


def test_string_minmax_and_order_by(eng):
    from baikaldb_amd import QueryPlan
    rng = np.random.default_rng(3)
    n = 50_000
    t, strings, v, _ = make_table(eng, n, rng)
    try:
        plan = QueryPlan(t.col_types, group=[],
                         aggs=[("min", 0), ("max", 0)])
        res = eng.filter_agg(t, plan, expected_groups=1)
        got = res.fetch()
        res.free()
        assert eng.dict_word(t, 0, int(got["agg_i"][0][0])) == min(strings)
        assert eng.dict_word(t, 0, int(got["agg_i"][1][0])) == max(strings)
        # ORDER BY s LIMIT 10: rowids of the byte-wise smallest strings
        rowids = eng.sort_topk(t, [(0, 1, 1)], 10)
        expect = sorted(range(n), key=lambda i: (strings[i], i))[:10]
        assert list(rowids) == expect
    finally:
        t.free()


def test_string_nulls_and_like(eng):
    from baikaldb_amd import QueryPlan
    from baikaldb_amd.like import like_accept_codes
    rng = np.random.default_rng(4)
    n = 60_000
    t, strings, v, valid = make_table(eng, n, rng, null_frac=0.25)
    try:
        # LIKE 'm%' via the real pattern compiler against the built dict
        words = []
        while True:
            w = eng.dict_word(t, 0, len(words))
            if w is None:
                break
            words.append(w)
        nw = len(words)
        codes = like_accept_codes(words, b"m%")
        accept = bytearray((nw + 7) // 8)
        for c in codes:
            accept[c >> 3] |= 1 << (c & 7)
        bm = eng.upload_bytes(bytes(accept))
        try:
            plan = QueryPlan(t.col_types,
                             conjuncts=[(0, "in_bitmap", (bm, nw))],
                             aggs=[("count_star", -1)])
            res = eng.filter_agg(t, plan, expected_groups=4)
            got = res.fetch()
            res.free()
        finally:
            eng.free_ptr(bm)
        expect = sum(1 for i, s in enumerate(strings)
                     if valid[i] and s.startswith("m"))
        assert got["agg_i"][0][0] == expect
    finally:
        t.free()

