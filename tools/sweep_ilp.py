# part_agg ILP ablation at 1e9 (config3 shape) + parity smoke
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan
T_I, T_D, T_S = 6, 12, 13
eng = GpuEngine()
specs = [(T_I,0,0,1<<31,0),(T_I,0,0,1<<31,0),(T_I,4,16384,0,0),(T_I,0,0,1000,0),
         (T_D,3,0,0,0),(T_D,3,0,0,0),(T_I,0,0,1<<31,0),(T_S,2,64,0,0)]
t = eng.create_table(specs, 1_000_000_000)
eng.generate(t, 20260915); eng.sync()
conj = [(0,"<",1<<30),(1,"<",int((1<<31)*0.9)),(7,"!=",63)]
plan = QueryPlan(t.col_types, conjuncts=conj, group=[2,7],
                 aggs=[("count_star",-1),("sum",3),("sum",4),("avg",5)])
plan_lo = QueryPlan(t.col_types, conjuncts=[], group=[3],
                    aggs=[("count_star",-1),("sum",0)])
base = {}
for ilp in ["1", "2"]:
    os.environ["BK_AGG_ILP"] = ilp
    for nm, pl, eg in [("c3", plan, 1<<21), ("lo", plan_lo, 1<<12)]:
        best = None
        for rep in range(3):
            r = eng.filter_agg(t, pl, expected_groups=eg)
            bd = r.breakdown()
            f = r.fetch(sorted=True, max_groups=5)
            if best is None or r.kernel_ms < best[0]:
                best = (r.kernel_ms, bd.get("part_agg", 0), r.ngroups,
                        f["agg_i"][0][:3].tolist())
            r.free()
        if ilp == "1": base[nm] = best
        ok = "" if ilp == "1" else (
            "PARITY-OK" if best[2] == base[nm][2] and best[3] == base[nm][3]
            else f"PARITY-MISMATCH {best[2]}vs{base[nm][2]} {best[3]}vs{base[nm][3]}")
        print(f"ilp={ilp} {nm}: part_agg={best[1]:6.2f} total={best[0]:7.2f} "
              f"ng={best[2]} {ok}", flush=True)
t.free()
