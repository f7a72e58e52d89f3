# Concurrent region queries through the C-ABI: the reference store drives
# many exec trees at once from different bthreads (each TREE single-threaded,
# include/exec/exec_node.h threading contract). The engine must return the
# same results under that concurrency (thread-local errors, mutex-guarded
# buffer pool, shared device).
import threading

import numpy as np
import pytest

TYPE_INT64, TYPE_DOUBLE = 6, 12
SEED = 424242


@pytest.fixture(scope="module")
def eng():
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()


@pytest.mark.gpu
def test_concurrent_queries_match_serial(eng):
    from baikaldb_amd import QueryPlan
    specs = [(TYPE_INT64, 0, 0, 1 << 31, 0),
             (TYPE_INT64, 0, 0, 3000, 0),
             (TYPE_INT64, 0, 0, 1000, 0),
             (TYPE_DOUBLE, 3, 0, 0, 0)]
    t = eng.create_table(specs, 400_000)
    try:
        eng.generate(t, SEED)
        eng.sync()
        plans = [
            QueryPlan(t.col_types, conjuncts=[(0, "<", 1 << 30)], group=[1],
                      aggs=[("count_star", -1), ("sum", 2), ("avg", 3)]),
            QueryPlan(t.col_types, conjuncts=[(0, ">", 1 << 29)], group=[1],
                      aggs=[("count_star", -1), ("min", 2), ("max", 2)]),
            QueryPlan(t.col_types, group=[2],
                      aggs=[("count_star", -1), ("sum", 0)]),
            QueryPlan(t.col_types, conjuncts=[(2, "<", 500)], group=[],
                      aggs=[("count_star", -1), ("sum", 2)]),
        ]

        def run_one(plan):
            r = eng.filter_agg(t, plan, expected_groups=1 << 13)
            try:
                return r.fetch(sorted=True)
            finally:
                r.free()

        serial = [run_one(p) for p in plans]

        # 4 threads x 6 rounds of interleaved queries
        results = [[None] * 6 for _ in plans]
        errors = []

        def worker(pi):
            try:
                for rnd in range(6):
                    results[pi][rnd] = run_one(plans[pi])
            except Exception as e:  # noqa: BLE001
                errors.append((pi, repr(e)))

        threads = [threading.Thread(target=worker, args=(pi,))
                   for pi in range(len(plans))]
        for th in threads:
            th.start()
        for th in threads:
            th.join()
        assert not errors, errors
        for pi, exp in enumerate(serial):
            for rnd in range(6):
                got = results[pi][rnd]
                assert got["ngroups"] == exp["ngroups"], (pi, rnd)
                assert np.array_equal(got["enc"], exp["enc"]), (pi, rnd)
                for a in range(len(plans[pi].aggs)):
                    if plans[pi].aggs[a][0] in ("avg",):
                        assert np.allclose(got["agg_d"][a], exp["agg_d"][a],
                                           rtol=0, atol=1e-9), (pi, rnd, a)
                    else:
                        assert np.array_equal(got["agg_i"][a],
                                              exp["agg_i"][a]), (pi, rnd, a)
    finally:
        t.free()
