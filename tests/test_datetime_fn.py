# Datetime scalar-fn predicates (reference internal_functions.cpp
# hour/minute/second/month/year/dayofmonth, fn_manager.cpp:219-230, packed
# layout datetime.h:35-45 + datetime.cpp:410-419): year(c)=K, hour(c)<K etc.
# pushed down onto a packed-DATETIME column.
import numpy as np
import pytest

from tests.test_gpu_agg import run_both, assert_parity, SEED

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING, TYPE_DATETIME = 6, 12, 13, 14
D_UNI, D_DT = 0, 5
FNS = {"year": 1, "month": 2, "day": 3, "hour": 4, "minute": 5, "second": 6}


def np_extract(fn, v):
    dt = v.astype(np.uint64)
    if fn == "year":
        return ((dt >> 46) & 0x1FFFF) // 13
    if fn == "month":
        return ((dt >> 46) & 0x1FFFF) % 13
    if fn == "day":
        return (dt >> 41) & 0x1F
    if fn == "hour":
        return (dt >> 36) & 0x1F
    if fn == "minute":
        return (dt >> 30) & 0x3F
    if fn == "second":
        return (dt >> 24) & 0x3F


def test_oracle_datetime_fn_vs_numpy(oracle):
    """Oracle datetime predicate counts match a numpy recompute."""
    import ctypes as C
    from oracle.bindings import make_query, BkColSpec
    specs = [(TYPE_DATETIME, D_DT, 0, 0, 0), (TYPE_INT64, D_UNI, 0, 100, 0)]
    arr = (BkColSpec * 2)()
    for i, s in enumerate(specs):
        (arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1,
         arr[i].null_frac_x1e6) = s
    cols, valids = oracle.generate_table(list(arr), 50_000, SEED)
    types = [s[0] for s in specs]
    for fname, (op, opn, lit) in [("year", (0, "==", 2021)),
                                  ("month", (4, "<", 4)),
                                  ("hour", (2, ">", 12)),
                                  ("day", (3, ">=", 15)),
                                  ("minute", (5, "<=", 29)),
                                  ("second", (1, "!=", 0))]:
        q = make_query([(0, op, TYPE_INT64, lit, FNS[fname])], [],
                       [(0, -1)], types)
        exp = oracle.filter_agg(cols, valids, types, q)
        ext = np_extract(fname, cols[0])
        want = {"==": ext == lit, "<": ext < lit, ">": ext > lit,
                ">=": ext >= lit, "<=": ext <= lit, "!=": ext != lit}[opn].sum()
        assert exp["agg_i"][0][0] == want, f"{fname} {opn} {lit}"


@pytest.mark.gpu
def test_gpu_datetime_fn_parity(eng, orc):
    specs = [(TYPE_DATETIME, D_DT, 0, 0, 150_000),   # nullable datetime
             (TYPE_INT64, D_UNI, 0, 500, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 31, 0)]
    aggs = [("count_star", -1), ("sum", 2)]
    for conj in ([(("year", 0), "=", 2022)],
                 [(("hour", 0), "<", 6), (2, "<", 1 << 30)],
                 [(("month", 0), "in", [1, 2, 11, 12])],
                 [(("day", 0), ">=", 20), (("minute", 0), "<=", 30),
                  (("second", 0), ">", 10)],
                 # >4 conjuncts exercises the lazy tail with fns
                 [(2, ">", 0), (2, "<", 1 << 31), (1, ">=", 0), (1, "<", 500),
                  (("hour", 0), "=", 13)]):
        got, exp = run_both(eng, orc, specs, 150_000, conj, [1], aggs)
        assert_parity(got, exp, aggs, [s[0] for s in specs])


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


@pytest.mark.gpu
def test_gpu_group_by_datetime_fn(eng, orc):
    """GROUP BY year(c), month(c) — fn exprs as group keys
    (encode_exprs_key evaluates fn calls, exec_node.cpp:555-571)."""
    from tests.test_gpu_agg import run_both, assert_parity
    specs = [(TYPE_DATETIME, D_DT, 0, 0, 120_000),
             (TYPE_INT64, TYPE_INT64 and 0, 0, 1000, 0),
             (TYPE_DOUBLE, 3, 0, 0, 0)]
    aggs = [("count_star", -1), ("sum", 1), ("avg", 2)]
    got, exp = run_both(eng, orc, specs, 150_000,
                        [(1, "<", 900)], [("year", 0), ("month", 0)], aggs,
                        expected_groups=1 << 10)
    assert_parity(got, exp, aggs, [s[0] for s in specs])
    assert 80 <= got["ngroups"] <= 86   # 7 years x 12 months (+null key rows)
    # decoded keys are the extraction outputs
    years = np.array([orc.lib.orc_decode_i64(__import__("ctypes").c_uint64(int(e)))
                      for e in got["enc"][:, 0]])
    nn = (got["flags"] & 0x80) == 0
    assert set(years[nn]) <= set(range(2019, 2026))
