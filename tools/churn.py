# allocation-churn stress: tight create/generate/query/free cycles
import sys, os
sys.path.insert(0, "/root/repo")
import numpy as np, torch
torch.cuda.init()
from baikaldb_amd import GpuEngine, QueryPlan
eng = GpuEngine()
T_I, T_D, T_S = 6, 12, 13
it = int(sys.argv[1]) if len(sys.argv) > 1 else 1500
for i in range(it):
    n = [1000, 20_000, 120_000][i % 3]
    t = eng.create_table([(T_I, 0, 0, 500, 0), (T_S, 2, 64, 0, 200_000),
                          (T_D, 3, 0, 0, 0)], n)
    eng.generate(t, i)
    plan = QueryPlan(t.col_types, conjuncts=[(0, "<", 400)], group=[0, 1],
                     aggs=[("count_star", -1), ("sum", 0), ("avg", 2)])
    r = eng.filter_agg(t, plan, expected_groups=1 << 12)
    f = r.fetch(sorted=True, max_groups=3)
    assert f["ngroups"] > 0
    r.free()
    t.free()
    if i % 300 == 0:
        print(f"iter {i} ok", flush=True)
print("CHURN OK", flush=True)
