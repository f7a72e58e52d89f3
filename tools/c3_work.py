import os, sys
sys.path.insert(0, "/root/repo")
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
os.environ["BK_PIPE"] = "0"
for rep in range(2):
    r = eng.filter_agg(t, plan, expected_groups=1<<21)
    r.free()
print("done")
