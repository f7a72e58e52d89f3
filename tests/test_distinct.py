# COUNT/SUM(DISTINCT) — the reference's multi-distinct planner rewrite
# (agg_node.cpp:247-258) executed as level-1 GROUP BY (keys + d) plus the
# bkgpu_agg_rollup / orc_filter_agg_distinct fold.
#
# The oracle itself is validated against a brute-force numpy recompute
# (CPU test); the GPU path is validated against the oracle (gpu tests).
import ctypes as C

import numpy as np
import pytest

from tests.test_gpu_agg import assert_parity, SEED

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13
D_UNI, D_SKEW, D_DICT, D_SUM16, D_ZIPF = 0, 1, 2, 3, 4
AGGMAP = {"count_star": 0, "count": 1, "sum": 2, "avg": 3, "min": 4, "max": 5,
          "count_distinct": 6, "sum_distinct": 7, "avg_distinct": 8}
OPS = {"=": 0, "!=": 1, ">": 2, ">=": 3, "<": 4, "<=": 5}


def oracle_distinct(orc, specs, n, conjuncts, group, aggs, seed=SEED,
                    nthreads=4, group_bits=(), group_base=(),
                    distinct_bits=0, distinct_base=0):
    """Run the oracle's two-level distinct path; returns the fetch dict.
    (The oracle's maps hold raw per-key encodings — the packing bits only
    matter to the ENGINE's key words, so they default off here.)"""
    from oracle.bindings import make_query, BkColSpec
    from baikaldb_amd.plan import QueryPlan

    arr = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1,
         arr[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(arr), n, seed)
    col_types = [s[0] for s in specs]

    plan = QueryPlan(col_types, conjuncts=conjuncts, group=group, aggs=aggs,
                     group_bits=group_bits, group_base=group_base,
                     distinct_bits=distinct_bits, distinct_base=distinct_base)
    l1_plan, _, src_idx = plan.split_distinct()
    oconj = []
    for col, op, lit, *og in conjuncts:
        ct = TYPE_DOUBLE if isinstance(lit, float) else TYPE_INT64
        oconj.append((col, OPS[op], ct, lit, 0, og[0] if og else 0))
    q1 = make_query(oconj, l1_plan.group,
                    [(AGGMAP[a], c) for a, c in l1_plan.aggs], col_types)
    q2 = make_query((), group, [(AGGMAP[a], c) for a, c in aggs], col_types)
    exp = orc.filter_agg_distinct(cols, valids, col_types, q1, q2, src_idx,
                                  nthreads=nthreads, dict_seed=seed)
    return exp, (cols, valids, col_types)


def brute_distinct(cols, valids, col_types, conjuncts, group, aggs):
    """Numpy recompute (independent of both engine and oracle code paths)."""
    n = len(cols[0])
    mask = np.ones(n, dtype=bool)
    for col, op, lit, *og in conjuncts:
        v = cols[col]
        ok = {"=": v == lit, "!=": v != lit, "<": v < lit, "<=": v <= lit,
              ">": v > lit, ">=": v >= lit}[op]
        if valids[col] is not None:
            ok = ok & (valids[col] != 0)
        mask &= ok
    idx = np.nonzero(mask)[0]
    gcol = group[0] if group else None
    if gcol is not None:
        gvals = cols[gcol][idx]
        gnull = (valids[gcol][idx] == 0) if valids[gcol] is not None \
            else np.zeros(len(idx), bool)
        keys = [(bool(gn), None if gn else gv.item())
                for gn, gv in zip(gnull, gvals)]
    else:
        keys = [(False, None)] * len(idx)
    out = {}
    for i, k in zip(idx, keys):
        out.setdefault(k, []).append(i)
    result = {}
    for k, rows in out.items():
        rows = np.array(rows)
        vals = {}
        for a, (name, col) in enumerate(aggs):
            if name == "count_star":
                vals[a] = len(rows)
                continue
            v = cols[col][rows]
            nn = (valids[col][rows] != 0) if valids[col] is not None \
                else np.ones(len(rows), bool)
            v = v[nn]
            if name == "count_distinct":
                vals[a] = len(np.unique(v))
            elif name == "sum_distinct":
                vals[a] = np.unique(v).sum()
            elif name == "avg_distinct":
                u = np.unique(v)
                vals[a] = float(u.mean()) if len(u) else None
            elif name == "sum":
                vals[a] = v.sum() if len(v) else None
            elif name == "count":
                vals[a] = len(v)
        result[k] = vals
    return result


@pytest.fixture(scope="module")
def orc():
    from oracle import Oracle
    return Oracle()


def test_oracle_distinct_vs_brute(orc):
    specs = [(TYPE_INT64, D_UNI, 0, 40, 0),          # group key
             (TYPE_INT64, D_UNI, 0, 25, 200_000),    # distinct col, 20% NULL
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    conj = [(3, "<", 1 << 30)]
    aggs = [("count_star", -1), ("count_distinct", 1), ("sum_distinct", 1),
            ("sum", 2), ("count", 1), ("avg_distinct", 1)]
    exp, (cols, valids, col_types) = oracle_distinct(
        orc, specs, 30_000, conj, [0], aggs)
    brute = brute_distinct(cols, valids, col_types, conj, [0], aggs)
    assert exp["ngroups"] == len(brute)
    # oracle sorts canonically: nulls-first=flag order then encoded key
    enc = exp["enc"][:, 0]
    flags = exp["flags"]
    for g in range(exp["ngroups"]):
        if flags[g] & 0x80:
            k = (True, None)
        else:
            k = (False, int(orc.lib.orc_decode_i64(C.c_uint64(int(enc[g])))))
        b = brute[k]
        assert exp["agg_i"][0][g] == b[0]                      # count_star
        assert exp["agg_i"][1][g] == b[1], f"count_distinct group {k}"
        assert exp["agg_i"][2][g] == b[2], f"sum_distinct group {k}"
        if b[3] is None:
            assert exp["agg_has"][3][g] == 0
        else:
            assert abs(exp["agg_d"][3][g] - b[3]) < 1e-9 * (abs(b[3]) + 1)
        assert exp["agg_i"][4][g] == b[4]                      # count
        if b[5] is None:
            assert exp["agg_has"][5][g] == 0
        else:
            assert abs(exp["agg_d"][5][g] - b[5]) < 1e-9 * (abs(b[5]) + 1)


def test_oracle_distinct_no_group(orc):
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 100_000),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    aggs = [("count_star", -1), ("count_distinct", 0)]
    exp, (cols, valids, col_types) = oracle_distinct(
        orc, specs, 20_000, [(1, "<", 1 << 29)], [], aggs)
    brute = brute_distinct(cols, valids, col_types,
                           [(1, "<", 1 << 29)], [], aggs)
    assert exp["ngroups"] == 1
    b = brute[(False, None)]
    assert exp["agg_i"][0][0] == b[0]
    assert exp["agg_i"][1][0] == b[1]


def test_oracle_distinct_zero_rows(orc):
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 0)]
    aggs = [("count_star", -1), ("count_distinct", 0)]
    exp, _ = oracle_distinct(orc, specs, 1000, [(0, "<", -5)], [], aggs)
    assert exp["ngroups"] == 1          # agg_node.cpp:490-505 single row
    assert exp["agg_i"][0][0] == 0
    assert exp["agg_i"][1][0] == 0


# ---------------- GPU parity ----------------

@pytest.fixture(scope="module")
def eng():
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()


def run_both_distinct(eng, orc, specs, n, conjuncts, group, aggs,
                      seed=SEED, expected_groups=1 << 12):
    from baikaldb_amd import QueryPlan
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, seed)
        plan = QueryPlan(t.col_types, conjuncts=conjuncts, group=group,
                         aggs=aggs)
        res = eng.filter_agg_distinct(t, plan,
                                      expected_l1_groups=1 << 16,
                                      expected_groups=expected_groups)
        try:
            got = res.fetch(sorted=True)
        finally:
            res.free()
    finally:
        t.free()
    exp, _ = oracle_distinct(orc, specs, n, conjuncts, group, aggs, seed=seed)
    return got, exp


def _parity_names(aggs):
    """map distinct names onto their state-shaped plain kin for assert_parity"""
    m = {"count_distinct": "count", "sum_distinct": "sum",
         "avg_distinct": "avg"}
    return [(m.get(n, n), c) for n, c in aggs]


@pytest.mark.gpu
def test_gpu_count_distinct_grouped(eng, orc):
    specs = [(TYPE_INT64, D_UNI, 0, 64, 0),
             (TYPE_INT64, D_UNI, 0, 300, 150_000),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    aggs = [("count_star", -1), ("count_distinct", 1), ("sum", 3),
            ("avg", 2), ("sum_distinct", 1), ("avg_distinct", 1)]
    got, exp = run_both_distinct(eng, orc, specs, 200_000,
                                 [(3, "<", 1 << 30)], [0], aggs)
    assert_parity(got, exp, _parity_names(aggs), [s[0] for s in specs])


@pytest.mark.gpu
def test_gpu_count_distinct_dict(eng, orc):
    # distinct over a dict-encoded VARCHAR; group key nullable
    specs = [(TYPE_INT64, D_UNI, 0, 40, 120_000),
             (TYPE_STRING, D_DICT, 700, 0, 80_000)]
    aggs = [("count_star", -1), ("count_distinct", 1)]
    got, exp = run_both_distinct(eng, orc, specs, 150_000, [], [0], aggs)
    assert_parity(got, exp, _parity_names(aggs), [s[0] for s in specs])


@pytest.mark.gpu
def test_gpu_distinct_double_no_group(eng, orc):
    specs = [(TYPE_DOUBLE, D_SUM16, 0, 0, 100_000),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    aggs = [("count_star", -1), ("count_distinct", 0), ("sum_distinct", 0)]
    got, exp = run_both_distinct(eng, orc, specs, 120_000,
                                 [(1, ">", 1 << 28)], [], aggs)
    assert got["ngroups"] == exp["ngroups"] == 1
    assert got["agg_i"][0][0] == exp["agg_i"][0][0]
    assert got["agg_i"][1][0] == exp["agg_i"][1][0]
    # sum over identical distinct sets: tolerance for reduction order
    d = abs(got["agg_d"][2][0] - exp["agg_d"][2][0])
    assert d <= 1e-9 * (abs(exp["agg_d"][2][0]) + 1)


@pytest.mark.gpu
def test_gpu_distinct_zero_rows(eng, orc):
    specs = [(TYPE_INT64, D_UNI, 0, 1000, 0)]
    aggs = [("count_star", -1), ("count_distinct", 0)]
    got, exp = run_both_distinct(eng, orc, specs, 50_000,
                                 [(0, "<", -1)], [], aggs)
    assert got["ngroups"] == exp["ngroups"] == 1
    assert got["agg_i"][0][0] == 0 and got["agg_i"][1][0] == 0


@pytest.mark.gpu
def test_gpu_distinct_high_cardinality(eng, orc):
    # distinct cardinality near row count: level-1 table must regrow
    specs = [(TYPE_INT64, D_UNI, 0, 16, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 40, 0)]
    aggs = [("count_star", -1), ("count_distinct", 1)]
    got, exp = run_both_distinct(eng, orc, specs, 400_000, [], [0], aggs,
                                 expected_groups=64)
    assert_parity(got, exp, _parity_names(aggs), [s[0] for s in specs])


def _random_distinct_case(rng):
    ncols = rng.randint(2, 5)
    specs = []
    for _ in range(ncols):
        t = rng.choice([TYPE_INT64, TYPE_INT64, TYPE_DOUBLE, TYPE_STRING])
        nf = rng.choice([0, 0, 250_000])
        if t == TYPE_INT64:
            specs.append((t, D_UNI, 0, rng.choice([5, 40, 1000, 1 << 20]), nf))
        elif t == TYPE_DOUBLE:
            specs.append((t, D_SUM16, 0, 0, nf))
        else:
            specs.append((t, D_DICT, rng.choice([4, 90]), 0, nf))
    dcol = rng.randrange(ncols)
    group = rng.sample([c for c in range(ncols)], rng.randint(0, 1))
    aggs = [("count_star", -1), ("count_distinct", dcol)]
    if specs[dcol][0] != TYPE_STRING and rng.random() < 0.7:
        aggs.append(("sum_distinct", dcol))
        aggs.append(("avg_distinct", dcol))
    for _ in range(rng.randint(0, 2)):
        c = rng.randrange(ncols)
        aggs.append((rng.choice(["sum", "count", "min", "max"]), c))
    conj = []
    if rng.random() < 0.6:
        c = rng.randrange(ncols)
        if specs[c][0] == TYPE_DOUBLE:
            conj.append((c, "<", rng.uniform(-1, 1)))
        else:
            conj.append((c, rng.choice(["<", ">", "!="]), rng.randint(0, 500)))
    n = rng.choice([1000, 15_000, 60_000])
    return specs, conj, group, aggs, n


@pytest.mark.parametrize("cs", range(4))
def test_oracle_distinct_fuzz_vs_brute(orc, cs):
    import random
    rng = random.Random(31_000 + cs)
    for sub in range(5):
        specs, conj, group, aggs, n = _random_distinct_case(rng)
        # brute supports: count_star/count/sum/count_distinct/sum_distinct/
        # avg_distinct; restrict to those for the brute comparison
        aggs = [a for a in aggs if a[0] in
                ("count_star", "count", "sum", "count_distinct",
                 "sum_distinct", "avg_distinct")]
        exp, (cols, valids, col_types) = oracle_distinct(
            orc, specs, n, conj, group, aggs, seed=rng.randrange(1 << 40))
        brute = brute_distinct(cols, valids, col_types, conj, group, aggs)
        assert exp["ngroups"] == len(brute), (cs, sub, specs, group, aggs)
        # aggregate-sum check (order-independent totals per agg column)
        for a, (name, col) in enumerate(aggs):
            if name in ("count_star", "count", "count_distinct"):
                tot_exp = int(exp["agg_i"][a].sum())
                tot_brute = sum(v[a] for v in brute.values())
                assert tot_exp == tot_brute, (cs, sub, name)


@pytest.mark.gpu
@pytest.mark.parametrize("cs", range(3))
def test_gpu_distinct_fuzz(eng, orc, cs):
    import random
    rng = random.Random(33_000 + cs)
    for sub in range(4):
        specs, conj, group, aggs, n = _random_distinct_case(rng)
        seed = rng.randrange(1 << 40)
        got, exp = run_both_distinct(eng, orc, specs, n, conj, group, aggs,
                                     seed=seed, expected_groups=1 << 12)
        try:
            assert_parity(got, exp, _parity_names(aggs),
                          [s[0] for s in specs])
        except AssertionError as e:
            raise AssertionError(
                f"distinct fuzz {cs}/{sub}: specs={specs} conj={conj} "
                f"group={group} aggs={aggs}: {e}")


# ---------------- sort-based dedup level 1 (bkgpu_filter_agg_sorted) -------

@pytest.mark.gpu
def test_gpu_sorted_dedup_vs_hash_plain_group(eng):
    """filter_agg_sorted == filter_agg on the same plain GROUP BY."""
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 4000, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 50_000)]
    t = eng.create_table(specs, 500_000)
    try:
        eng.generate(t, SEED + 7)
        plan = QueryPlan(t.col_types, conjuncts=[(1, "<", 1 << 30)],
                         group=[0], aggs=[("count_star", -1), ("sum", 1),
                                          ("avg", 2), ("min", 1)],
                         group_bits=[13], group_base=[0])
        a = eng.filter_agg_sorted(t, plan)
        b = eng.filter_agg(t, plan, expected_groups=8192)
        try:
            ga, gb = a.fetch(sorted=True), b.fetch(sorted=True)
        finally:
            a.free()
            b.free()
    finally:
        t.free()
    assert ga["ngroups"] == gb["ngroups"]
    assert ga["rows_passed"] == gb["rows_passed"]
    assert np.array_equal(ga["enc"], gb["enc"])
    assert np.array_equal(ga["flags"], gb["flags"])
    for i in range(2):
        assert np.array_equal(ga["agg_i"][i], gb["agg_i"][i])
    assert np.allclose(ga["agg_d"][2], gb["agg_d"][2], rtol=0, atol=1e-9)
    assert np.array_equal(ga["agg_i"][3], gb["agg_i"][3])


@pytest.mark.gpu
def test_gpu_sorted_dedup_distinct_parity(eng, orc, monkeypatch):
    """COUNT/SUM(DISTINCT) through the forced sort-dedup level 1 matches the
    oracle (declared distinct_bits make the level-1 keys pack)."""
    from baikaldb_amd import QueryPlan
    monkeypatch.setenv("BK_DEDUP_SORT", "1")
    specs = [(TYPE_INT64, D_UNI, 0, 16, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 40, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 0)]
    aggs = [("count_star", -1), ("count_distinct", 1), ("sum", 2),
            ("sum_distinct", 1)]
    conjuncts = [(2, "<", 900)]
    n = 400_000
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=conjuncts, group=[0],
                         aggs=aggs, group_bits=[6], group_base=[0],
                         distinct_bits=41, distinct_base=0)
        res = eng.filter_agg_distinct(t, plan, expected_l1_groups=1 << 16)
        try:
            got = res.fetch(sorted=True)
        finally:
            res.free()
    finally:
        t.free()
    exp, _ = oracle_distinct(orc, specs, n, conjuncts, [0], aggs)
    assert_parity(got, exp, _parity_names(aggs), [s[0] for s in specs])


@pytest.mark.gpu
def test_gpu_sorted_dedup_nullable_key(eng):
    """nullable group key: the null-flag byte rides above the declared bits
    and must survive the sort round trip."""
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 500, 200_000),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    t = eng.create_table(specs, 300_000)
    try:
        eng.generate(t, SEED + 11)
        plan = QueryPlan(t.col_types, group=[0],
                         aggs=[("count_star", -1), ("max", 1)],
                         group_bits=[10], group_base=[0])
        a = eng.filter_agg_sorted(t, plan)
        b = eng.filter_agg(t, plan, expected_groups=1024)
        try:
            ga, gb = a.fetch(sorted=True), b.fetch(sorted=True)
        finally:
            a.free()
            b.free()
    finally:
        t.free()
    assert ga["ngroups"] == gb["ngroups"]
    assert np.array_equal(ga["enc"], gb["enc"])
    assert np.array_equal(ga["flags"], gb["flags"])
    assert np.array_equal(ga["agg_i"][0], gb["agg_i"][0])
    assert np.array_equal(ga["agg_i"][1], gb["agg_i"][1])


def test_sorted_dedup_rejects_wide_keys():
    """CPU: the C entry refuses keys that don't pack (falls back upstream)."""
    from baikaldb_amd import GpuEngine, QueryPlan
    eng = GpuEngine.__new__(GpuEngine)  # bind lib without device init
    from baikaldb_amd.engine import _load
    eng.lib = _load()
    plan = QueryPlan([TYPE_INT64], group=[0], aggs=[("count_star", -1)])
    q = plan.to_spec()
    h = eng.lib.bkgpu_filter_agg_sorted(None, C.byref(q), 0, 100)
    assert not h
    err = eng.lib.bkgpu_last_error().decode()
    assert "sorted" in err or "GROUP BY" in err


@pytest.mark.gpu
def test_gpu_sorted_rec_mode_vs_partitioned(eng):
    """REC-mode sorted aggregation (>= 2 agg inputs -> record-carrying sort)
    must equal the partitioned hash path on the same query."""
    import os
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_INT64, D_UNI, 0, 4000, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 120_000),
             (TYPE_STRING, D_DICT, 300, 0, 0)]
    t = eng.create_table(specs, 800_000)
    try:
        eng.generate(t, SEED + 21)
        plan = QueryPlan(t.col_types, conjuncts=[(0, "<", 1 << 30)],
                         group=[1, 4],
                         aggs=[("count_star", -1), ("sum", 2), ("avg", 3),
                               ("min", 2)])
        a = eng.filter_agg_sorted(t, plan)            # auto-pack + REC
        os.environ["BK_SORTED_RANGE"] = "0"
        try:
            b = eng.filter_agg(t, plan, expected_groups=1 << 20)  # partitioned
        finally:
            os.environ.pop("BK_SORTED_RANGE", None)
        try:
            ga, gb = a.fetch(sorted=True), b.fetch(sorted=True)
        finally:
            a.free()
            b.free()
    finally:
        t.free()
    assert ga["ngroups"] == gb["ngroups"]
    assert np.array_equal(ga["enc"], gb["enc"])
    assert np.array_equal(ga["flags"], gb["flags"])
    for i in (0, 1, 3):
        assert np.array_equal(ga["agg_i"][i], gb["agg_i"][i])
    assert np.allclose(ga["agg_d"][2], gb["agg_d"][2], rtol=0, atol=1e-9)


@pytest.mark.gpu
def test_gpu_distinct_two_group_keys(eng, orc):
    """DISTINCT aggregates under TWO user group keys (the round-1 cap was
    one): level 1 groups by (k0, k1, d) packed via declared group_bits into
    the two 64-bit key words; k_rollup unpacks all three and repacks the
    user keys per the level-2 spec."""
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 40, 0),       # k0
             (TYPE_INT64, D_UNI, 0, 25, 0),       # k1
             (TYPE_INT64, D_UNI, 0, 500, 100_000),  # d (nullable)
             (TYPE_INT64, D_UNI, 0, 1000, 0)]     # plain agg input
    aggs = [("count_star", -1), ("count_distinct", 2), ("sum", 3),
            ("sum_distinct", 2)]
    conj = [(3, "<", 900)]
    group = [0, 1]
    n = 150_000
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=conj, group=group,
                         aggs=aggs, group_bits=[8, 8], group_base=[0, 0],
                         distinct_bits=10, distinct_base=0)
        res = eng.filter_agg_distinct(t, plan, expected_l1_groups=1 << 16,
                                      expected_groups=1 << 12)
        try:
            got = res.fetch(sorted=True)
        finally:
            res.free()
    finally:
        t.free()
    exp, _ = oracle_distinct(orc, specs, n, conj, group, aggs, seed=SEED,
                             group_bits=[8, 8], group_base=[0, 0],
                             distinct_bits=10, distinct_base=0)
    assert got["ngroups"] == exp["ngroups"]
    assert np.array_equal(got["flags"], exp["flags"])
    assert np.array_equal(got["enc"], exp["enc"])
    assert np.array_equal(got["agg_i"], exp["agg_i"])
    assert np.array_equal(got["agg_has"], exp["agg_has"])


@pytest.mark.gpu
def test_gpu_distinct_two_keys_with_nullable_key(eng, orc):
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, D_UNI, 0, 30, 200_000),   # k0 nullable
             (TYPE_STRING, D_DICT, 16, 0, 0),       # k1 dict
             (TYPE_INT64, D_UNI, 0, 200, 0)]        # d
    aggs = [("count_star", -1), ("count_distinct", 2)]
    group = [0, 1]
    n = 100_000
    t = eng.create_table(specs, n)
    try:
        eng.generate(t, SEED + 1)
        plan = QueryPlan(t.col_types, group=group, aggs=aggs,
                         group_bits=[8, 6], group_base=[0, 0],
                         distinct_bits=9, distinct_base=0)
        res = eng.filter_agg_distinct(t, plan, expected_l1_groups=1 << 14,
                                      expected_groups=1 << 10)
        try:
            got = res.fetch(sorted=True)
        finally:
            res.free()
    finally:
        t.free()
    exp, _ = oracle_distinct(orc, specs, n, [], group, aggs, seed=SEED + 1,
                             group_bits=[8, 6], group_base=[0, 0],
                             distinct_bits=9, distinct_base=0)
    assert got["ngroups"] == exp["ngroups"]
    assert np.array_equal(got["flags"], exp["flags"])
    assert np.array_equal(got["enc"], exp["enc"])
    assert np.array_equal(got["agg_i"], exp["agg_i"])
