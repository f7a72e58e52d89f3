# Dictionary order-preservation invariant: bk_dict_word produces words whose
# byte order (ExprValue STRING compare = byte compare, expr_value.h:895-945)
# equals dict-code order, for ANY seed — the property that makes integer
# MIN/MAX over dict codes a correct string MIN/MAX (DESIGN.md "string agg").
import ctypes as C
import random

import pytest


@pytest.fixture(scope="module")
def eng():
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()



def test_dict_word_order_preserving(oracle):
    lib = oracle.lib
    rng = random.Random(42)
    buf_a = C.create_string_buffer(64)
    buf_b = C.create_string_buffer(64)
    for _ in range(2000):
        seed = rng.getrandbits(64)
        a = rng.randrange(0, 1 << 22)
        b = rng.randrange(0, 1 << 22)
        lib.orc_dict_word(C.c_uint64(seed), C.c_int64(a), buf_a, 64)
        lib.orc_dict_word(C.c_uint64(seed), C.c_int64(b), buf_b, 64)
        wa, wb = buf_a.value, buf_b.value
        if a < b:
            assert wa < wb, (seed, a, b, wa, wb)
        elif a > b:
            assert wa > wb, (seed, a, b, wa, wb)
        else:
            assert wa == wb


# ---------------- derived string-fn columns (dict remap) ----------------

@pytest.mark.gpu
def test_group_by_string_fn_derived_column(eng, oracle):
    """GROUP BY substr(s,1,3): the engine compiles the unary string fn to a
    dict remap (bkgpu_table_derive_remap) — colliding transformed words MERGE
    groups, and the derived codes stay order-preserving. Validated against a
    numpy brute force over the oracle's word materialization."""
    import numpy as np
    from baikaldb_amd import QueryPlan
    lib = oracle.lib
    SEED2 = 77_001
    NCODES, N = 300, 120_000
    T_I, T_S = 6, 13
    specs = [(T_S, 2, NCODES, 0, 0), (T_I, 0, 0, 1000, 0)]
    t = eng.create_table(specs, N)
    try:
        eng.generate(t, SEED2)
        # the generator's dict words (oracle == engine by construction)
        buf = C.create_string_buffer(64)
        words = []
        for code in range(NCODES):
            lib.orc_dict_word(C.c_uint64(SEED2), C.c_int64(code), buf, 64)
            words.append(buf.value.decode())
        nc, new_words = eng.derive_string_fn(t, 0, ("substr", 1, 3), words)
        assert len(new_words) < NCODES  # collisions actually happened
        plan = QueryPlan(t.col_types, group=[nc],
                         aggs=[("count_star", -1), ("sum", 1)])
        r = eng.filter_agg(t, plan, expected_groups=1024)
        try:
            got = r.fetch(sorted=True)
        finally:
            r.free()
    finally:
        t.free()
    # brute force: regenerate rows on host via the oracle's generator
    from oracle.bindings import BkColSpec
    arr = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1,
         arr[i].null_frac_x1e6) = s
    cols, valids = oracle.generate_table(list(arr), N, SEED2)
    tw = np.array([new_words.index(words[c][:3]) for c in range(NCODES)])
    g = tw[cols[0]]
    order = np.argsort(np.unique(g))
    uniq = np.unique(g)
    assert got["ngroups"] == len(uniq)
    # group keys arrive as the derived codes in canonical order
    assert np.array_equal(got["enc"].reshape(-1, 4)[:, 0].astype(np.int64),
                          uniq[order])
    for r_i, gv in enumerate(uniq[order]):
        sel = cols[1][g == gv]
        assert got["agg_i"][0][r_i] == len(sel)
        assert got["agg_i"][1][r_i] == sel.sum()


@pytest.mark.gpu
def test_order_by_string_fn_derived_column(eng, oracle):
    """ORDER BY substr(s,1,3): derived dict-remap codes are order-preserving
    over the transformed words, so top-N on the derived column equals a
    numpy sort of the transformed strings."""
    import numpy as np
    from baikaldb_amd import QueryPlan
    lib = oracle.lib
    SEED3 = 88_002
    NCODES, N = 250, 80_000
    T_I, T_S = 6, 13
    specs = [(T_S, 2, NCODES, 0, 0), (T_I, 0, 0, 1 << 31, 0)]
    t = eng.create_table(specs, N)
    try:
        eng.generate(t, SEED3)
        buf = C.create_string_buffer(64)
        words = []
        for code in range(NCODES):
            lib.orc_dict_word(C.c_uint64(SEED3), C.c_int64(code), buf, 64)
            words.append(buf.value.decode())
        nc, new_words = eng.derive_string_fn(t, 0, ("substr", 1, 3), words)
        limit = 500
        rowids = eng.sort_topk(t, [(nc, 1, 1)], limit)
    finally:
        t.free()
    from oracle.bindings import BkColSpec
    arr = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1,
         arr[i].null_frac_x1e6) = s
    cols, valids = oracle.generate_table(list(arr), N, SEED3)
    tw = np.array([new_words.index(words[c][:3]) for c in range(NCODES)])
    derived = tw[cols[0]]
    order = np.lexsort((np.arange(N), derived))[:limit]
    assert np.array_equal(np.sort(rowids), np.sort(order))
    # arrival-order ties within equal keys (TopNSorter semantics)
    assert np.array_equal(derived[rowids], derived[order])
