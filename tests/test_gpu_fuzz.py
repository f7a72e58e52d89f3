# Randomized query fuzzing: random schemas (types, NULL fractions,
# distributions) x random plans (conjunct ops incl IN-lists, group keys,
# aggregate sets) — GPU pipeline vs CPU oracle on identical seeded inputs.
import random

import numpy as np
import pytest

from tests.test_gpu_agg import run_both, assert_parity  # reuse harness

pytestmark = pytest.mark.gpu

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING, TYPE_DATETIME = 6, 12, 13, 14
D_UNI, D_SKEW, D_DICT, D_SUM16, D_ZIPF, D_DT = 0, 1, 2, 3, 4, 5
_DT_FNS = ["year", "month", "day", "hour", "minute", "second"]


def random_case(rng):
    ncols = rng.randint(2, 6)
    specs = []
    for _ in range(ncols):
        t = rng.choice([TYPE_INT64, TYPE_INT64, TYPE_DOUBLE, TYPE_STRING,
                        TYPE_DATETIME])
        nf = rng.choice([0, 0, 0, 120_000, 400_000])
        if t == TYPE_INT64:
            dist = rng.choice([D_UNI, D_SKEW, D_ZIPF])
            if dist == D_UNI:
                lo = rng.choice([0, -1000, -(1 << 40)])
                hi = rng.choice([10, 1000, 1 << 31, 1 << 41])
                if hi <= lo:
                    hi = lo + 1000
                specs.append((t, dist, lo, hi, nf))
            else:
                specs.append((t, dist, rng.choice([5, 300, 20_000]), 0, nf))
        elif t == TYPE_DOUBLE:
            specs.append((t, D_SUM16, 0, 0, nf))
        elif t == TYPE_DATETIME:
            specs.append((t, D_DT, 0, 0, nf))
        else:
            specs.append((t, D_DICT, rng.choice([3, 64, 3000]), 0, nf))
    # conjuncts
    conjuncts = []
    for _ in range(rng.randint(0, 3)):
        c = rng.randrange(ncols)
        t = specs[c][0]
        if t == TYPE_DOUBLE:
            conjuncts.append((c, rng.choice(["<", ">", ">=", "<="]),
                              rng.uniform(-1.5, 1.5)))
        elif t == TYPE_DATETIME:
            fn = rng.choice(_DT_FNS)
            lim = {"year": (2019, 2026), "month": (1, 13), "day": (1, 29),
                   "hour": (0, 24), "minute": (0, 60), "second": (0, 60)}[fn]
            conjuncts.append(((fn, c), rng.choice(["<", ">", "=", "!="]),
                              rng.randint(*lim)))
        elif rng.random() < 0.3:
            vals = [rng.randint(0, 3000) for _ in range(rng.randint(1, 8))]
            conjuncts.append((c, rng.choice(["in", "not_in"]), vals))
        elif rng.random() < 0.25 and ncols >= 2:
            # binary-arith predicate (add/sub/mul); lit domain follows the
            # operand types (plan casts to f64 when either col is DOUBLE)
            c2 = rng.randrange(ncols)
            while specs[c2][0] == TYPE_DATETIME or specs[c][0] == TYPE_DATETIME:
                c = rng.randrange(ncols); c2 = rng.randrange(ncols)
            arith = rng.choice(["add", "sub", "mul"])
            lit = rng.randint(-2000, 6000)
            conjuncts.append(((arith, c, c2),
                              rng.choice(["<", ">", ">=", "<="]), lit))
        else:
            conjuncts.append((c, rng.choice(["<", ">", "=", "!=", ">=", "<="]),
                              rng.randint(-100, 3000)))
    # OR clauses (BkConjunct.or_group CNF): ~25% of cases wrap 2+ of the
    # conjuncts into one OR clause
    if len(conjuncts) >= 2 and rng.random() < 0.25:
        k = rng.randint(2, len(conjuncts))
        tail = conjuncts[-k:]
        conjuncts = conjuncts[:-k] + [tuple(cj) + (1,) for cj in tail]
    # group keys (DATETIME columns may group through an extraction fn:
    # GROUP BY year(c) etc.)
    group = rng.sample(range(ncols), rng.randint(0, min(2, ncols)))
    group = [(rng.choice(_DT_FNS), c) if specs[c][0] == TYPE_DATETIME
             and rng.random() < 0.7 else c for c in group]
    # ~20%: one aggregate over a two-column arithmetic expression
    arith_agg = None
    num_cols = [c for c in range(ncols)
                if specs[c][0] in (TYPE_INT64, TYPE_DOUBLE)]
    if len(num_cols) >= 2 and rng.random() < 0.2:
        a_c, b_c = rng.sample(num_cols, 2)
        arith_agg = (rng.choice(["sum", "avg", "min", "max", "count"]),
                     (rng.choice(["add", "sub", "mul"]), a_c, b_c))
    # aggs
    aggs = [("count_star", -1)]
    for _ in range(rng.randint(0, 4)):
        c = rng.randrange(ncols)
        name = rng.choice(["count", "sum", "avg", "min", "max"])
        aggs.append((name, c))  # string min/max OK: dict is order-preserving
    if arith_agg is not None:
        aggs.append(arith_agg)
    return specs, conjuncts, group, aggs


@pytest.mark.parametrize("case_seed", range(6))
def test_fuzz_agg(eng, orc, case_seed):
    rng = random.Random(1000 + case_seed)
    for sub in range(6):
        specs, conjuncts, group, aggs = random_case(rng)
        n = rng.choice([1000, 20_000, 60_000])
        got, exp = run_both(eng, orc, specs, n, conjuncts, group, aggs,
                            seed=rng.randrange(1 << 40),
                            expected_groups=1 << 12)
        try:
            assert_parity(got, exp, aggs, [s[0] for s in specs])
        except AssertionError as e:
            raise AssertionError(
                f"fuzz case {case_seed}/{sub}: specs={specs} "
                f"conj={conjuncts} group={group} aggs={aggs}: {e}")


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


@pytest.mark.parametrize("case_seed", range(4))
def test_fuzz_sort(eng, orc, case_seed):
    from tests.test_gpu_sort import run_both as sort_both
    rng = random.Random(7000 + case_seed)
    for sub in range(5):
        ncols = rng.randint(2, 4)
        specs = []
        for _ in range(ncols):
            t = rng.choice([TYPE_INT64, TYPE_INT64, TYPE_DOUBLE])
            nf = rng.choice([0, 0, 300_000])
            if t == TYPE_INT64:
                specs.append((t, D_UNI, rng.choice([0, -(1 << 50)]),
                              rng.choice([50, 1 << 20, 1 << 51]), nf))
            else:
                specs.append((t, D_SUM16, 0, 0, nf))
        norder = rng.randint(1, min(3, ncols))
        order = [(c, rng.randint(0, 1), rng.randint(0, 1))
                 for c in rng.sample(range(ncols), norder)]
        limit = rng.choice([1, 100, 3000])
        n = rng.choice([5000, 40_000])
        got, exp = sort_both(eng, orc, specs, n, order, limit)
        assert np.array_equal(got, exp), \
            f"fuzz sort {case_seed}/{sub}: specs={specs} order={order} limit={limit}"
