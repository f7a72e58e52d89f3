# tools/vec_probe.py — k_dedup_mat_vec (BK_MAT_VEC) vs the strided mat on
# the north-star c2b shape: kernel breakdowns + bit-exact parity, plus a
# ragged-size table that forces the vec kernel's scalar tail chunk.
import os
import sys
import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

T_I = 6
eng = GpuEngine()


def run_pair(nrows, reps):
    specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 20, 0),
             (T_I, 4, 100_000, 0, 0), (T_I, 0, 0, 1000, 0)] + \
            [(T_I, 0, 0, 1 << 31, 0)] * 4
    t = eng.create_table(specs, nrows)
    eng.generate(t, 20260915)
    eng.sync()
    plan = QueryPlan(t.col_types,
                     conjuncts=[(0, "<", 1 << 30), (1, "=", 7)],
                     group=[2], aggs=[("sum", 3)])
    res = {}
    for mode in ("0", "1", "2"):
        os.environ["BK_MAT_VEC"] = mode
        best = None
        for rep in range(reps):
            r = eng.filter_agg(t, plan, expected_groups=1 << 18)
            bd = r.breakdown()
            km = r.kernel_ms
            if best is None or km < best[0]:
                best = (km, bd)
            if rep == 0:
                res[mode] = r.fetch(sorted=True)
            r.free()
        print({"0": "strided", "1": "vec    ", "2": "vecukey"}[mode], f"n={nrows}",
              "kernel_ms=%.2f" % best[0],
              " ".join(f"{k}={v:.2f}" for k, v in best[1].items()
                       if v >= 0.05), flush=True)
    a = res["0"]
    for b in (res["1"], res["2"]):
        assert a["rows_passed"] == b["rows_passed"]
        assert a["ngroups"] == b["ngroups"]
        assert np.array_equal(a["enc"], b["enc"])
        for i in range(len(a["agg_i"])):
            assert np.array_equal(a["agg_i"][i], b["agg_i"][i]), i
    print(f"parity OK n={nrows} rows={a['rows_passed']} "
          f"groups={a['ngroups']}", flush=True)
    t.free()


run_pair(10_000_003, 2)      # ragged: exercises the vec kernel's tail path
run_pair(1_000_000_000, 4)   # north-star shape
