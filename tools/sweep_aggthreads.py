# part_agg block-size sweep (latency hiding vs LDS-table share per tile).
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan
T_I, T_D, T_S = 6, 12, 13
eng = GpuEngine()
specs = [(T_I,0,0,1<<31,0),(T_I,0,0,1<<31,0),(T_I,4,16384,0,0),(T_I,0,0,1000,0),
         (T_D,3,0,0,0),(T_D,3,0,0,0),(T_I,0,0,1<<31,0),(T_S,2,64,0,0)]
t = eng.create_table(specs, 300_000_000)
eng.generate(t, 20260915); eng.sync()
conj = [(0,"<",1<<30),(1,"<",int((1<<31)*0.9)),(7,"!=",63)]
plan = QueryPlan(t.col_types, conjuncts=conj, group=[2,7],
                 aggs=[("count_star",-1),("sum",3),("sum",4),("avg",5)])
# also a low-cardinality shape (hot buckets -> wave-combine path dominates)
plan_lo = QueryPlan(t.col_types, conjuncts=[], group=[3],
                    aggs=[("count_star",-1),("sum",0)])
for at in ["256", "512", "1024"]:
    os.environ["BK_AGG_THREADS"] = at
    for nm, pl, eg in [("c3~1M", plan, 1<<21), ("lo1000", plan_lo, 1<<12)]:
        best = None
        for rep in range(3):
            r = eng.filter_agg(t, pl, expected_groups=eg)
            bd = r.breakdown(); ng = r.ngroups
            if best is None or bd.get("part_agg", 0) < best[0]:
                best = (bd.get("part_agg", 0), r.kernel_ms, ng)
            r.free()
        print(f"threads={at:4s} {nm}: part_agg={best[0]:6.2f} total={best[1]:7.2f} ng={best[2]}",
              flush=True)
t.free()
