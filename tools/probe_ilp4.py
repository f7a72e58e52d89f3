# is part_agg still latency-bound? compare chunk sizes + ILP at 1e9
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
for env in [{}, {"BK_AGG_LDS_KB": "64"}, {"BK_AGG_LDS_KB": "100"},
             {"BK_AGG_THREADS": "512"}]:
    for k in ["BK_AGG_LDS_KB", "BK_AGG_THREADS"]: os.environ.pop(k, None)
    os.environ.update(env)
    best = None
    for rep in range(3):
        r = eng.filter_agg(t, plan, expected_groups=1<<21)
        bd = r.breakdown()
        if best is None or r.kernel_ms < best[0]: best = (r.kernel_ms, bd)
        r.free()
    print(f"{env}: total={best[0]:6.2f} " +
          " ".join(f"{k}={v:.2f}" for k,v in best[1].items() if v>=0.1), flush=True)
t.free()
