import os, sys
sys.path.insert(0, "/root/repo")
from baikaldb_amd import GpuEngine, QueryPlan
T_I = 6
eng = GpuEngine()
specs = [(T_I,0,0,1<<31,0),(T_I,0,0,20,0),(T_I,4,100_000,0,0),(T_I,0,0,1000,0)] + [(T_I,0,0,1<<31,0)]*4
t = eng.create_table(specs, 1_000_000_000)
eng.generate(t, 20260915); eng.sync()
os.environ["BK_SORTED_RANGE"] = "1100000000"
plan = QueryPlan(t.col_types, conjuncts=[(0,"<",1<<30),(1,"=",7)], group=[2], aggs=[("sum",3)])
for rep in range(3):
    r = eng.filter_agg(t, plan, expected_groups=1<<18)
    r.free()
print("done")
