# Validates the CPU oracle against an INDEPENDENT numpy restatement of the
# reference row-engine semantics (filter_node.cpp:726-734 NULL logic,
# agg_fn_call.cpp:496-555 aggregate updates, exec_node.cpp:555-571 keys) on
# seeded inputs, plus determinism and multithread==singlethread merge parity
# (the MERGE_AGG path, agg_node.cpp:539-543).
import numpy as np
import pytest

from oracle import (BkColSpec, TYPE_INT64, TYPE_DOUBLE, TYPE_STRING,
                    DIST_UNIFORM, DIST_CUBESKEW, DIST_DICT, DIST_SUMU16,
                    OP_LT, OP_EQ, OP_NE, OP_GT,
                    AGG_COUNT_STAR, AGG_COUNT, AGG_SUM, AGG_AVG, AGG_MIN, AGG_MAX)

OP_IN, OP_NOT_IN = 6, 7
from oracle.bindings import make_query

SEED = 0x5EED


def small_table(oracle, n=20000, null_frac=0):
    specs = (BkColSpec * 5)()
    # c0: int64 uniform, c1: int64 uniform small, c2: skew group, c3: double, c4: dict
    vals = [(TYPE_INT64, DIST_UNIFORM, 0, 2**31, null_frac),
            (TYPE_INT64, DIST_UNIFORM, 0, 100, null_frac),
            (TYPE_INT64, DIST_CUBESKEW, 1000, 0, null_frac),
            (TYPE_DOUBLE, DIST_SUMU16, 0, 0, null_frac),
            (TYPE_STRING, DIST_DICT, 50, 0, null_frac)]
    for i, (t, d, p0, p1, nf) in enumerate(vals):
        specs[i].col_type, specs[i].dist, specs[i].p0, specs[i].p1 = t, d, p0, p1
        specs[i].null_frac_x1e6 = nf
    cols, valids = oracle.generate_table(list(specs), n, SEED)
    types = [TYPE_INT64, TYPE_INT64, TYPE_INT64, TYPE_DOUBLE, TYPE_STRING]
    return cols, valids, types


def numpy_reference(cols, valids, types, conjuncts, group, aggs):
    """Independent numpy restatement of the row-engine semantics."""
    n = len(cols[0])
    mask = np.ones(n, dtype=bool)
    for col, op, cmp_type, lit in conjuncts:
        v = cols[col]
        ok = np.ones(n, dtype=bool) if valids[col] is None else valids[col].astype(bool)
        if cmp_type == TYPE_DOUBLE:
            x = v.astype(np.float64)
            litv = float(lit)
        else:
            x = v.astype(np.int64)
            litv = lit if isinstance(lit, (list, tuple)) else int(lit)
        if op in (OP_IN, OP_NOT_IN):
            member = np.isin(x, np.array(list(lit), dtype=np.int64))
            res = member if op == OP_IN else ~member
        else:
            res = {OP_LT: x < litv, OP_EQ: x == litv, OP_NE: x != litv,
                   OP_GT: x > litv}[op]
        mask &= ok & res
    idx = np.nonzero(mask)[0]
    # group keys: tuple of (is_null, value)
    out = {}
    for r in idx:
        key = tuple((False, None) if (valids[c] is not None and not valids[c][r])
                    else (True, cols[c][r].item()) for c in group)
        st = out.setdefault(key, [dict(i=0, d=0.0, cnt=0, has=False) for _ in aggs])
        for ai, (at, col) in enumerate(aggs):
            s = st[ai]
            if at == AGG_COUNT_STAR:
                s["i"] += 1
                s["has"] = True
                continue
            iv = valids[col] is None or valids[col][r]
            if at == AGG_COUNT:
                if iv:
                    s["i"] += 1
                s["has"] = True
                continue
            if not iv:
                continue
            val = cols[col][r].item()
            if at == AGG_SUM:
                if types[col] == TYPE_DOUBLE:
                    s["d"] = (s["d"] + val) if s["has"] else val
                else:
                    s["i"] = int(np.int64(np.uint64(np.uint64(s["i"] & (2**64 - 1))
                                                    + np.uint64(val & (2**64 - 1)))))
                    if not s["has"]:
                        s["i"] = val
                s["has"] = True
            elif at == AGG_AVG:
                s["d"] += float(val)
                s["cnt"] += 1
                s["has"] = True
            elif at == AGG_MIN:
                cur = s["d"] if types[col] == TYPE_DOUBLE else s["i"]
                if not s["has"] or val < cur:
                    if types[col] == TYPE_DOUBLE:
                        s["d"] = val
                    else:
                        s["i"] = val
                s["has"] = True
            elif at == AGG_MAX:
                cur = s["d"] if types[col] == TYPE_DOUBLE else s["i"]
                if not s["has"] or val > cur:
                    if types[col] == TYPE_DOUBLE:
                        s["d"] = val
                    else:
                        s["i"] = val
                s["has"] = True
    return int(mask.sum()), out


def check_against_numpy(oracle, cols, valids, types, conjuncts, group, aggs,
                        nthreads=1):
    q = make_query(conjuncts, group, aggs, types)
    got = oracle.filter_agg(cols, valids, types, q, nthreads=nthreads, dict_seed=SEED)
    npassed, expect = numpy_reference(cols, valids, types, conjuncts, group, aggs)
    assert got["rows_passed"] == npassed
    assert got["ngroups"] == len(expect) or (len(expect) == 0 and len(group) == 0)
    # reconstruct oracle groups from raw (flag, enc)
    for g in range(got["ngroups"]):
        flag = got["flags"][g]
        key = []
        for k, c in enumerate(group):
            if (flag >> (7 - k)) & 1:
                key.append((False, None))
            else:
                e = int(got["enc"][g][k])
                if types[c] == TYPE_INT64:
                    key.append((True, int(np.int64(np.uint64(e ^ (1 << 63))))))
                elif types[c] == TYPE_DOUBLE:
                    key.append((True, oracle.lib.orc_decode_f64(e)))
                else:
                    key.append((True, int(np.int32(np.uint32(e)))))
        key = tuple(key)
        assert key in expect, key
        st = expect.pop(key)
        for ai, (at, col) in enumerate(aggs):
            s = st[ai]
            has = bool(got["agg_has"][ai][g])
            if at in (AGG_COUNT_STAR, AGG_COUNT):
                assert got["agg_i"][ai][g] == s["i"]
            elif at == AGG_AVG:
                if s["has"] and s["cnt"]:
                    assert has
                    assert got["agg_d"][ai][g] == pytest.approx(s["d"] / s["cnt"], rel=1e-12)
                else:
                    assert not has
            elif types[col] == TYPE_DOUBLE:
                if s["has"]:
                    assert has
                    assert got["agg_d"][ai][g] == pytest.approx(s["d"], rel=1e-12)
                else:
                    assert not has
            else:
                if s["has"]:
                    assert has and got["agg_i"][ai][g] == s["i"]
                else:
                    assert not has
    assert not expect or (len(group) == 0 and got["ngroups"] == 1)


def test_count_star_filter(oracle):
    cols, valids, types = small_table(oracle)
    check_against_numpy(oracle, cols, valids, types,
                        [(0, OP_LT, TYPE_INT64, 2**30)], [], [(AGG_COUNT_STAR, -1)])


def test_group_by_aggs(oracle):
    cols, valids, types = small_table(oracle)
    conj = [(0, OP_LT, TYPE_INT64, int(2**31 * 0.8)), (1, OP_GT, TYPE_INT64, 10)]
    aggs = [(AGG_COUNT_STAR, -1), (AGG_SUM, 1), (AGG_SUM, 3), (AGG_AVG, 3),
            (AGG_MIN, 0), (AGG_MAX, 3)]
    check_against_numpy(oracle, cols, valids, types, conj, [2], aggs)


def test_group_by_two_keys_dict(oracle):
    cols, valids, types = small_table(oracle)
    conj = [(0, OP_LT, TYPE_INT64, int(2**31 * 0.5)), (4, OP_NE, TYPE_STRING, 7)]
    aggs = [(AGG_COUNT_STAR, -1), (AGG_SUM, 1), (AGG_SUM, 3), (AGG_AVG, 3)]
    check_against_numpy(oracle, cols, valids, types, conj, [2, 4], aggs)


def test_nulls_everywhere(oracle):
    cols, valids, types = small_table(oracle, n=5000, null_frac=200000)  # 20% NULLs
    conj = [(0, OP_LT, TYPE_INT64, int(2**31 * 0.9))]
    aggs = [(AGG_COUNT_STAR, -1), (AGG_COUNT, 1), (AGG_SUM, 1), (AGG_AVG, 3),
            (AGG_MIN, 3), (AGG_MAX, 0)]
    check_against_numpy(oracle, cols, valids, types, conj, [2], aggs)


def test_empty_selection_no_group_emits_zero_row(oracle):
    # agg_node.cpp:490-505: no GROUP BY + no rows => one initialized row
    cols, valids, types = small_table(oracle, n=1000)
    q = make_query([(0, OP_LT, TYPE_INT64, -5)], [], [(AGG_COUNT_STAR, -1), (AGG_SUM, 1)],
                   types)
    got = oracle.filter_agg(cols, valids, types, q)
    assert got["ngroups"] == 1
    assert got["rows_passed"] == 0
    assert got["agg_i"][0][0] == 0          # COUNT(*) = 0
    assert got["agg_has"][1][0] == 0        # SUM = NULL


def test_empty_selection_with_group_emits_nothing(oracle):
    cols, valids, types = small_table(oracle, n=1000)
    q = make_query([(0, OP_LT, TYPE_INT64, -5)], [2], [(AGG_COUNT_STAR, -1)], types)
    got = oracle.filter_agg(cols, valids, types, q)
    assert got["ngroups"] == 0


def test_multithread_merge_matches_single(oracle):
    cols, valids, types = small_table(oracle, n=50000, null_frac=50000)
    conj = [(0, OP_LT, TYPE_INT64, int(2**31 * 0.7))]
    aggs = [(AGG_COUNT_STAR, -1), (AGG_SUM, 1), (AGG_SUM, 3), (AGG_AVG, 3),
            (AGG_MIN, 0), (AGG_MAX, 3)]
    q = make_query(conj, [2, 4], aggs, types)
    r1 = oracle.filter_agg(cols, valids, types, q, nthreads=1, dict_seed=SEED)
    r8 = oracle.filter_agg(cols, valids, types, q, nthreads=8, dict_seed=SEED)
    assert r1["ngroups"] == r8["ngroups"]
    assert r1["rows_passed"] == r8["rows_passed"]
    assert r1["keys"] == r8["keys"]
    assert np.array_equal(r1["agg_i"], r8["agg_i"])
    assert np.array_equal(r1["agg_has"], r8["agg_has"])
    # double sums merge in a different order: tolerance
    np.testing.assert_allclose(r1["agg_d"], r8["agg_d"], rtol=1e-12, atol=1e-9)


def test_datagen_deterministic(oracle):
    cols1, _ = oracle.generate_table(_specs(), 1000, 42)
    cols2, _ = oracle.generate_table(_specs(), 1000, 42)
    for a, b in zip(cols1, cols2):
        assert np.array_equal(a, b)
    # sharded generation matches whole-range generation
    whole, _ = oracle.generate_table(_specs(), 1000, 42)
    part, _ = oracle.generate_table(_specs(), 500, 42, row_begin=500)
    for w, p in zip(whole, part):
        assert np.array_equal(w[500:], p)


def _specs():
    specs = (BkColSpec * 3)()
    for i, (t, d, p0, p1) in enumerate([(TYPE_INT64, DIST_UNIFORM, 0, 2**31),
                                        (TYPE_DOUBLE, DIST_SUMU16, 0, 0),
                                        (TYPE_STRING, DIST_DICT, 100, 0)]):
        specs[i].col_type, specs[i].dist, specs[i].p0, specs[i].p1 = t, d, p0, p1
    return list(specs)


def test_sort_topk_matches_numpy(oracle):
    cols, valids, types = small_table(oracle, n=30000)
    order = [(1, 1, 1), (0, 1, 1)]  # ORDER BY c1 ASC, c0 ASC
    got = oracle.sort_topk(cols, valids, types, order, limit=500)
    key = np.lexsort((np.arange(len(cols[0])), cols[0], cols[1]))
    expect = key[:500]
    assert np.array_equal(got, expect)


def test_sort_topk_desc_and_ties(oracle):
    cols, valids, types = small_table(oracle, n=10000)
    order = [(1, 0, 0)]  # ORDER BY c1 DESC (c1 has many ties in [0,100))
    got = oracle.sort_topk(cols, valids, types, order, limit=300)
    # expected: stable by arrival among equal keys, descending by c1
    idx = np.arange(len(cols[0]))
    expect = idx[np.lexsort((idx, -cols[1]))][:300]
    assert np.array_equal(got, expect)


def test_dict_words_unique(oracle):
    words = {oracle.dict_word(SEED, c) for c in range(5000)}
    assert len(words) == 5000


def test_in_predicates(oracle):
    cols, valids, types = small_table(oracle, n=30000, null_frac=100000)
    conj = [(1, OP_IN, TYPE_INT64, [3, 7, 11, 42]),
            (4, OP_NOT_IN, TYPE_STRING, [0, 1, 2])]
    aggs = [(AGG_COUNT_STAR, -1), (AGG_SUM, 1)]
    check_against_numpy(oracle, cols, valids, types, conj, [2], aggs, nthreads=4)
