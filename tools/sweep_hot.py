# Sweep the k_part_histo hot-table knobs on the config3 shape.
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["BK_DEBUG"] = "1"
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
CASES = [
  ("base(off)",  {}),
  ("min10%",     {"BK_HOT_MIN":"400"}),
  ("min10%+cap384", {"BK_HOT_MIN":"400","BK_HOT_CAP":"384","BK_HOT_PROBE":"6"}),
  ("1024/512",   {"BK_HOT_MIN":"400","BK_HOT_SLOTS":"1024","BK_HOT_LDS_KB":"95"}),
  ("1024/768p8", {"BK_HOT_MIN":"400","BK_HOT_SLOTS":"1024","BK_HOT_LDS_KB":"95",
                  "BK_HOT_CAP":"768","BK_HOT_PROBE":"8"}),
]
KN = ["BK_HOT_MIN","BK_HOT_CAP","BK_HOT_PROBE","BK_HOT_SLOTS","BK_HOT_LDS_KB"]
for name, env in CASES:
    for k in KN: os.environ.pop(k, None)
    os.environ.update(env)
    best = None
    for rep in range(3):
        r = eng.filter_agg(t, plan, expected_groups=1<<21)
        bd = r.breakdown(); tot = r.kernel_ms; ng = r.ngroups
        if best is None or tot < best[0]: best = (tot, bd, ng)
        r.free()
    tot, bd, ng = best
    print(f"{name:14s} total={tot:7.2f} ng={ng} " +
          " ".join(f"{k}={v:.2f}" for k,v in bd.items() if v >= 0.1), flush=True)
t.free()
