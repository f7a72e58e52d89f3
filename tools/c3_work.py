# tools/c3_work.py — run a few canonical config3 / config5 queries (default
# engine settings) as the workload under rocprofv3 --pmc passes.
#   python tools/c3_work.py [c3|c5] [reps]
import sys

sys.path.insert(0, "/root/repo")
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

T_I, T_D, T_S = 6, 12, 13
which = sys.argv[1] if len(sys.argv) > 1 else "c3"
reps = int(sys.argv[2]) if len(sys.argv) > 2 else 2
eng = GpuEngine()
if which == "c3":
    specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 1 << 31, 0),
             (T_I, 4, 16384, 0, 0), (T_I, 0, 0, 1000, 0),
             (T_D, 3, 0, 0, 0), (T_D, 3, 0, 0, 0),
             (T_I, 0, 0, 1 << 31, 0), (T_S, 2, 64, 0, 0)]
    t = eng.create_table(specs, 1_000_000_000)
    eng.generate(t, 20260915)
    eng.sync()
    conj = [(0, "<", 1 << 30), (1, "<", int((1 << 31) * 0.9)), (7, "!=", 63)]
    plan = QueryPlan(t.col_types, conjuncts=conj, group=[2, 7],
                     aggs=[("count_star", -1), ("sum", 3), ("sum", 4),
                           ("avg", 5)])
    for _ in range(reps):
        r = eng.filter_agg(t, plan, expected_groups=1 << 21)
        r.free()
elif which == "c2b":
    specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 20, 0),
             (T_I, 4, 100_000, 0, 0), (T_I, 0, 0, 1000, 0)] + \
            [(T_I, 0, 0, 1 << 31, 0)] * 4
    t = eng.create_table(specs, 1_000_000_000)
    eng.generate(t, 20260915)
    eng.sync()
    plan = QueryPlan(t.col_types,
                     conjuncts=[(0, "<", 1 << 30), (1, "=", 7)],
                     group=[2], aggs=[("sum", 3)])
    for _ in range(reps):
        r = eng.filter_agg(t, plan, expected_groups=1 << 18)
        r.free()
elif which == "c5":
    specs = [(T_I, 0, 0, 1 << 31, 0)] * 3
    t = eng.create_table(specs, 1_000_000_000)
    eng.generate(t, 20260915)
    eng.sync()
    for _ in range(reps):
        eng.sort_topk(t, [(0, 1, 1), (1, 1, 1)], 1_000_000)
print("done")
