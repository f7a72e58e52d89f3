# part_agg op-count ablation: same partition shape, varying aggregate count.
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
for name, aggs in [("4aggs",[("count_star",-1),("sum",3),("sum",4),("avg",5)]),
                   ("2aggs",[("count_star",-1),("sum",3)]),
                   ("1agg",[("count_star",-1)])]:
    plan = QueryPlan(t.col_types, conjuncts=conj, group=[2,7], aggs=aggs)
    best=None
    for rep in range(3):
        r = eng.filter_agg(t, plan, expected_groups=1<<21)
        bd = r.breakdown(); tot=r.kernel_ms
        if best is None or tot<best[0]: best=(tot,bd)
        r.free()
    print(name, {k: round(v,2) for k,v in best[1].items()}, flush=True)
t.free()
