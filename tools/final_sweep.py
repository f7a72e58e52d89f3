# last P x chunk check at fused records, 1e9
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
for P in ["2048", "4096"]:
    for ch in ["0", "524288"]:
        os.environ["BK_PART_P"] = P
        if ch == "0": os.environ.pop("BK_AGG_CHUNK", None)
        else: os.environ["BK_AGG_CHUNK"] = ch
        best = None
        for rep in range(4):
            r = eng.filter_agg(t, plan, expected_groups=1<<21)
            bd = r.breakdown()
            if best is None or r.kernel_ms < best[0]: best = (r.kernel_ms, bd)
            r.free()
        print(f"P={P} chunk={ch:7s} total={best[0]:6.2f} " +
              " ".join(f"{k}={v:.2f}" for k,v in best[1].items() if v>=0.1), flush=True)
t.free()
