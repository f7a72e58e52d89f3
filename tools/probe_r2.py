# tools/probe_r2.py — fresh-session hypotheses on the config3 1e9 shape:
#  A. histo LDS occupancy: with the hot path default-off, the warmup still
#     carves a 45 KB LDS table; does shrinking it (BK_HOT_SLOTS=16) buy
#     histo occupancy/time?
#  B. pipeline depth 3 vs 2 (BK_PIPE).
#  C. part_agg LDS table 64 KB (2 blocks/CU, more generations) vs 135 KB.
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

def run(tag, env):
    saved = {}
    for k, v in env.items():
        saved[k] = os.environ.get(k)
        if v is None: os.environ.pop(k, None)
        else: os.environ[k] = str(v)
    best = None; bbd = None
    for rep in range(4):
        r = eng.filter_agg(t, plan, expected_groups=1<<21)
        bd = r.breakdown()
        if best is None or r.kernel_ms < best:
            best, bbd, ng = r.kernel_ms, bd, r.ngroups
        r.free()
    print(f"{tag:34s} total={best:6.2f} ng={ng} " +
          " ".join(f"{k}={v:.2f}" for k, v in bbd.items() if v >= 0.1),
          flush=True)
    for k, v in saved.items():
        if v is None: os.environ.pop(k, None)
        else: os.environ[k] = v

# baseline (pipelined default)
run("base pipe2", {})
# A: tiny hot table in the pipelined path
run("hotslots16 pipe2", {"BK_HOT_SLOTS": 16})
# A': unpipelined so the per-kernel breakdown is visible
run("base nopipe", {"BK_PIPE": 0})
run("hotslots16 nopipe", {"BK_HOT_SLOTS": 16, "BK_PIPE": 0})
run("hotslots64 nopipe", {"BK_HOT_SLOTS": 64, "BK_PIPE": 0})
# B: pipe depth
run("pipe3", {"BK_PIPE": 3})
run("pipe4", {"BK_PIPE": 4})
run("pipe3 hotslots16", {"BK_PIPE": 3, "BK_HOT_SLOTS": 16})
# C: agg LDS size (2 blocks/CU at <=64 KB)
run("agglds64 nopipe", {"BK_AGG_LDS_KB": 64, "BK_PIPE": 0})
run("agglds64 pipe2", {"BK_AGG_LDS_KB": 64})
run("agglds100 pipe2", {"BK_AGG_LDS_KB": 100})
# combos
run("pipe2 hot16 agg64", {"BK_HOT_SLOTS": 16, "BK_AGG_LDS_KB": 64})
run("pipe3 hot16 agg64", {"BK_PIPE": 3, "BK_HOT_SLOTS": 16, "BK_AGG_LDS_KB": 64})
t.free()
print("probe_r2 done")
