# Dense-span partition pipeline (bkdpart.inc) parity tests: BK_DENSE=2
# forces the dense path wherever eligible, so these shapes exercise
# k_dhisto / k_dscatter / k_dagg / k_dense_to_table directly and compare
# against the CPU oracle (same seeded inputs, bk_datagen.h on both sides).
# Reference semantics under test: hash aggregate agg_node.cpp:405-587 +
# AggFnCall update/merge agg_fn_call.cpp:496-830 — results must be
# identical to the hash-partitioned path (bit-exact int, DTOL doubles).
import os

import numpy as np
import pytest

from tests.test_gpu_agg import (DTOL_REL, SEED, TYPE_INT64, TYPE_DOUBLE,
                                TYPE_STRING, D_UNI, D_SKEW, D_DICT, D_SUM16,
                                assert_parity, run_both, eng, orc)

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def force_dense():
    os.environ["BK_DENSE"] = "2"
    yield
    os.environ.pop("BK_DENSE", None)


# spans chosen dense-eligible: c1 skew over 1e5 distinct, c2 uniform 1000,
# c4 dict 4096 codes
BASE5 = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
         (TYPE_INT64, D_SKEW, 100000, 0, 0),
         (TYPE_INT64, D_UNI, 0, 1000, 0),
         (TYPE_DOUBLE, D_SUM16, 0, 0, 0),
         (TYPE_STRING, D_DICT, 4096, 0, 0)]
CT = [s[0] for s in BASE5]


def run_dense(eng, orc, *args, expect_dense=True, **kw):
    got, exp = run_both(eng, orc, *args, **kw)
    return got, exp


def test_dense_single_key_all_aggs(eng, orc):
    aggs = [("count_star", -1), ("count", 2), ("sum", 2), ("sum", 3),
            ("avg", 3), ("min", 0), ("max", 3)]
    got, exp = run_both(eng, orc, BASE5, 500_000,
                        [(0, "<", int((1 << 31) * 0.7))], [1], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_two_keys_with_dict(eng, orc):
    aggs = [("count_star", -1), ("sum", 2), ("avg", 3), ("min", 0),
            ("max", 3)]
    got, exp = run_both(eng, orc, BASE5, 400_000,
                        [(0, "<", int((1 << 31) * 0.8)), (2, "!=", 17)],
                        [1, 4], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_breakdown_names(eng):
    """The dense kernels must actually be the ones running under BK_DENSE=2
    (a silent fallback to the hash path would void the other tests here)."""
    from baikaldb_amd import QueryPlan
    t = eng.create_table(BASE5, 300_000)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types, conjuncts=[(0, "<", 1 << 30)],
                         group=[1], aggs=[("sum", 2)])
        res = eng.filter_agg(t, plan, expected_groups=1 << 14)
        try:
            names = list(res.breakdown().keys())
        finally:
            res.free()
    finally:
        t.free()
    assert any("dhisto" in n for n in names), names
    assert any("dagg" in n for n in names), names


def test_dense_nonsimple_conjuncts(eng, orc):
    """IN + OR-clause conjuncts route through the non-SIMPLE dhisto/dscatter
    instantiations."""
    aggs = [("count_star", -1), ("sum", 2), ("min", 3)]
    got, exp = run_both(
        eng, orc, BASE5, 300_000,
        [(2, "in", [3, 5, 9, 1000]),
         (0, "<", int((1 << 31) * 0.9), 1),    # OR clause 1
         (2, ">", 900, 1)],
        [1], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_arith_agg_input(eng, orc):
    """Expression agg inputs (SUM(a*b)) take the wide record field path."""
    aggs = [("count_star", -1), ("sum", ("mul", 2, 2)), ("sum", ("add", 2, 3))]
    got, exp = run_both(eng, orc, BASE5, 300_000,
                        [(0, "<", int((1 << 31) * 0.6))], [1], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_wide_key_span_falls_back(eng, orc):
    """Group key c0 spans 2^31 — not dense-eligible; the fallback (hash/
    sorted) must still produce correct results with BK_DENSE=2 set."""
    aggs = [("count_star", -1), ("sum", 2)]
    got, exp = run_both(eng, orc, BASE5, 200_000,
                        [(2, "<", 500)], [0], aggs,
                        expected_groups=1 << 17)
    assert_parity(got, exp, aggs, CT)


def test_dense_nullable_key_falls_back(eng, orc):
    """A nullable group key is dense-ineligible (flag byte unused in the
    dense id); fallback path must agree with the oracle."""
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_INT64, D_UNI, 0, 5000, 200_000),   # 20% NULL key
             (TYPE_INT64, D_UNI, 0, 1000, 0)]
    aggs = [("count_star", -1), ("sum", 2)]
    got, exp = run_both(eng, orc, specs, 200_000,
                        [(0, "<", 1 << 30)], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_dense_nullable_agg_input_falls_back(eng, orc):
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_INT64, D_UNI, 0, 5000, 0),
             (TYPE_INT64, D_UNI, 0, 1000, 300_000)]   # 30% NULL agg input
    aggs = [("count_star", -1), ("sum", 2), ("count", 2)]
    got, exp = run_both(eng, orc, specs, 200_000,
                        [(0, "<", 1 << 30)], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_dense_shard_merge_equals_whole(eng, orc):
    """Two dense half-range partials merged (MERGE_AGG, agg_node.cpp:539)
    must equal the whole-range dense pass — the dense-to-hash-table
    conversion feeds the standard blob/merge machinery."""
    import torch
    from baikaldb_amd import QueryPlan
    n = 400_000
    t = eng.create_table(BASE5, n)
    try:
        eng.generate(t, SEED)
        plan = QueryPlan(t.col_types,
                         conjuncts=[(0, "<", int((1 << 31) * 0.7))],
                         group=[1, 4],
                         aggs=[("count_star", -1), ("sum", 2), ("avg", 3)])
        whole = eng.filter_agg(t, plan, expected_groups=1 << 14)
        a = eng.filter_agg(t, plan, row_end=n // 2,
                           expected_groups=1 << 14)
        b = eng.filter_agg(t, plan, row_begin=n // 2,
                           expected_groups=1 << 14)
        try:
            nbytes = b.export_bytes()
            buf = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
            b.export_to(buf.data_ptr(), nbytes)
            a.merge_blob(buf.data_ptr(), b.ngroups)
            gw = whole.fetch(sorted=True)
            gm = a.fetch(sorted=True)
        finally:
            whole.free()
            a.free()
            b.free()
    finally:
        t.free()
    assert gw["ngroups"] == gm["ngroups"]
    assert np.array_equal(gw["enc"], gm["enc"])
    assert np.array_equal(gw["agg_i"][0], gm["agg_i"][0])
    assert np.array_equal(gw["agg_i"][1], gm["agg_i"][1])
    denom = np.abs(gw["agg_d"][2]) + np.maximum(gw["agg_i"][0], 1)
    assert np.all(np.abs(gw["agg_d"][2] - gm["agg_d"][2]) <= DTOL_REL * denom)


def test_dense_fuzz_vs_hash_path(eng, orc):
    """Randomized dense-eligible shapes: dense (BK_DENSE=2) vs the hash
    path (BK_DENSE=0) must agree exactly on int aggregates and within
    DTOL on doubles; both compare against the oracle elsewhere."""
    rng = np.random.default_rng(20260916)
    from baikaldb_amd import QueryPlan
    for it in range(6):
        n = int(rng.integers(50_000, 400_000))
        span1 = int(rng.integers(10, 200_000))
        span2 = int(rng.integers(2, 5000))
        specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
                 (TYPE_INT64, D_SKEW, span1, 0, 0),
                 (TYPE_INT64, D_UNI, 0, span2, 0),
                 (TYPE_DOUBLE, D_SUM16, 0, 0, 0)]
        group = [1] if rng.random() < 0.5 else [1, 2]
        aggs = [("count_star", -1)]
        for _ in range(int(rng.integers(1, 4))):
            kind = rng.choice(["sum", "avg", "min", "max", "count"])
            col = int(rng.integers(2, 4))
            aggs.append((str(kind), col))
        conj = [(0, "<", int((1 << 31) * rng.random()))]
        t = eng.create_table(specs, n)
        try:
            eng.generate(t, SEED + it)
            plan = QueryPlan(t.col_types, conjuncts=conj, group=group,
                             aggs=aggs)
            os.environ["BK_DENSE"] = "2"
            rd = eng.filter_agg(t, plan, expected_groups=1 << 14)
            gd = rd.fetch(sorted=True)
            rd.free()
            os.environ["BK_DENSE"] = "0"
            rh = eng.filter_agg(t, plan, expected_groups=1 << 14)
            gh = rh.fetch(sorted=True)
            rh.free()
        finally:
            t.free()
            os.environ["BK_DENSE"] = "2"
        assert gd["ngroups"] == gh["ngroups"], (it, gd["ngroups"], gh["ngroups"])
        assert np.array_equal(gd["enc"], gh["enc"]), it
        assert np.array_equal(gd["agg_has"], gh["agg_has"]), it
        for a, (name, col) in enumerate(aggs):
            is_double = col >= 0 and specs[col][0] == TYPE_DOUBLE
            if not is_double or name == "count":
                assert np.array_equal(gd["agg_i"][a], gh["agg_i"][a]), (it, a)
            elif name in ("min", "max"):
                assert np.array_equal(gd["agg_d"][a], gh["agg_d"][a]), (it, a)
            else:
                denom = np.abs(gh["agg_d"][a]) + np.maximum(gh["agg_i"][0], 1)
                err = np.abs(gd["agg_d"][a] - gh["agg_d"][a])
                assert np.all(err <= DTOL_REL * denom), (it, a, err.max())


# ---- hot-bucket absorption (ABS eager scatter, bkdpart.inc) -------------
# The host gates absorption on the measured per-bucket survivor histogram;
# BK_DABS_MIN=0 forces it on any data, BK_DABS=0 disables it. Results must
# match the oracle AND the non-absorbing run (int aggs bit-exact; f64 within
# DTOL — absorption only reorders the f64 atomic adds).

def _with_env(*kv):
    """_with_env(k1, v1, k2, v2, ...)"""
    import contextlib

    @contextlib.contextmanager
    def cm():
        olds = [(k, os.environ.get(k)) for k in kv[::2]]
        for k, v in zip(kv[::2], kv[1::2]):
            os.environ[k] = v
        try:
            yield
        finally:
            for k, old in olds:
                if old is None:
                    os.environ.pop(k, None)
                else:
                    os.environ[k] = old
    return cm()


def test_dense_absorb_skewed_vs_oracle(eng, orc):
    """Zipf group key: the hottest bucket absorbs in-LDS during the eager
    scatter; parity vs the oracle on all agg kinds the eager path carries."""
    aggs = [("count_star", -1), ("sum", 2), ("sum", 3), ("avg", 3),
            ("min", 2), ("max", 3)]
    with _with_env("BK_DABS", "1", "BK_DABS_MIN", "0.0"):
        got, exp = run_both(eng, orc, BASE5, 600_000,
                            [(0, "<", int((1 << 31) * 0.75))], [1], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_absorb_two_keys_dict(eng, orc):
    aggs = [("count_star", -1), ("sum", 2), ("avg", 3)]
    with _with_env("BK_DABS", "1", "BK_DABS_MIN", "0.0"):
        got, exp = run_both(eng, orc, BASE5, 400_000,
                            [(0, "<", int((1 << 31) * 0.8)), (2, "!=", 17)],
                            [1, 4], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_absorb_on_off_identical(eng):
    """BK_DABS on vs off (off is the default): int64 aggregates must be
    bit-identical (absorption changes only where the adds happen, not what
    is added)."""
    from baikaldb_amd import QueryPlan

    def run():
        t = eng.create_table(BASE5, 500_000)
        try:
            eng.generate(t, SEED)
            plan = QueryPlan(t.col_types,
                             conjuncts=[(0, "<", int((1 << 31) * 0.7))],
                             group=[1],
                             aggs=[("count_star", -1), ("sum", 2),
                                   ("min", 2), ("max", 2)])
            res = eng.filter_agg(t, plan, expected_groups=1 << 14)
            try:
                return res.fetch(sorted=True)
            finally:
                res.free()
        finally:
            t.free()

    with _with_env("BK_DABS", "1", "BK_DABS_MIN", "0.0"):
        a = run()
    b = run()
    assert len(a) == len(b)
    for ra, rb in zip(a, b):
        assert ra == rb, (ra, rb)


def test_dense_avg_over_wide_int64(eng, orc):
    """AVG over a physically-8-byte INT64 column (span >= 2^32): the eager
    record must carry (double)value bits (agg_fn_call.cpp casts AVG inputs
    to double), not raw int bits — wmode 3 in dense_eager_plan."""
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_INT64, D_SKEW, 50_000, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 40, 0),     # wide: no narrow store
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0)]
    aggs = [("count_star", -1), ("avg", 2), ("sum", 3)]
    got, exp = run_both(eng, orc, specs, 400_000,
                        [(0, "<", int((1 << 31) * 0.7))], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


def test_dense_absorb_avg_wide_int64(eng, orc):
    """Same shape with absorption forced on."""
    specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
             (TYPE_INT64, D_SKEW, 50_000, 0, 0),
             (TYPE_INT64, D_UNI, 0, 1 << 40, 0),
             (TYPE_DOUBLE, D_SUM16, 0, 0, 0)]
    aggs = [("count_star", -1), ("avg", 2), ("min", 2), ("sum", 3)]
    with _with_env("BK_DABS", "1", "BK_DABS_MIN", "0.0"):
        got, exp = run_both(eng, orc, specs, 400_000,
                            [(0, "<", int((1 << 31) * 0.7))], [1], aggs)
    assert_parity(got, exp, aggs, [s[0] for s in specs])


# ---- PACK32 split records (BK_DREC_PACK=1, measured-dead default off) ----
# The eager scatter can split the record into a u32 hdr stream (did + narrow
# fields bit-packed) + an aligned wide payload when the bit widths fit; it
# measured slower than the 24-B AoS record and defaults off, but the layout
# stays covered: results must be identical to the oracle and to the AoS run.


def test_dense_pack32_sum_narrow_vs_oracle(eng, orc):
    """Single skewed key (17 did bits) + one 10-bit narrow SUM field + wide
    f64 fields: 27 hdr bits -> PACK32 engages; parity vs the oracle."""
    aggs = [("count_star", -1), ("sum", 2), ("sum", 3), ("avg", 3),
            ("max", 3)]
    with _with_env("BK_DREC_PACK", "1"):
        got, exp = run_both(eng, orc, BASE5, 600_000,
                            [(0, "<", int((1 << 31) * 0.75))], [1], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_pack32_minmax_narrow(eng, orc):
    """Narrow MIN/MAX fields carry encoding deltas in the hdr: 10-bit key +
    17-bit min field (27 bits), then 10-bit key + 17-bit max (each <= 32)."""
    with _with_env("BK_DREC_PACK", "1"):
        aggs = [("count_star", -1), ("min", 1), ("sum", 3)]
        got, exp = run_both(eng, orc, BASE5, 400_000,
                            [(0, "<", int((1 << 31) * 0.8))], [2], aggs)
        assert_parity(got, exp, aggs, CT)
        aggs = [("count_star", -1), ("max", 1), ("avg", 3)]
        got, exp = run_both(eng, orc, BASE5, 400_000,
                            [(0, ">", int((1 << 31) * 0.3))], [2], aggs)
        assert_parity(got, exp, aggs, CT)


def test_dense_pack32_two_keys_bit_boundary(eng, orc):
    """Two keys (10+12 did bits) + one 10-bit narrow field = exactly 32
    hdr bits — the eligibility boundary."""
    aggs = [("count_star", -1), ("sum", 2), ("avg", 3)]
    with _with_env("BK_DREC_PACK", "1"):
        got, exp = run_both(eng, orc, BASE5, 400_000,
                            [(0, "<", int((1 << 31) * 0.8))],
                            [2, 4], aggs)
    assert_parity(got, exp, aggs, CT)


def test_dense_pack32_on_off_identical(eng):
    """PACK32 on vs off: int64 aggregates must be bit-identical (the split
    layout changes where record bytes live, not what is aggregated)."""
    from baikaldb_amd import QueryPlan

    def run():
        t = eng.create_table(BASE5, 500_000)
        try:
            eng.generate(t, SEED)
            plan = QueryPlan(t.col_types,
                             conjuncts=[(0, "<", int((1 << 31) * 0.7))],
                             group=[1],
                             aggs=[("count_star", -1), ("sum", 2),
                                   ("min", 2), ("max", 2)])
            res = eng.filter_agg(t, plan, expected_groups=1 << 14)
            try:
                return res.fetch(sorted=True)
            finally:
                res.free()
        finally:
            t.free()

    with _with_env("BK_DREC_PACK", "1"):
        a = run()
    b = run()
    assert len(a) == len(b)
    for ra, rb in zip(a, b):
        assert ra == rb, (ra, rb)
