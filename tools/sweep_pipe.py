# pipelined pipeline: chunks x overlap measurement + parity check, 1e9
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan
import numpy as np
T_I, T_D, T_S = 6, 12, 13
eng = GpuEngine()
specs = [(T_I,0,0,1<<31,0),(T_I,0,0,1<<31,0),(T_I,4,16384,0,0),(T_I,0,0,1000,0),
         (T_D,3,0,0,0),(T_D,3,0,0,0),(T_I,0,0,1<<31,0),(T_S,2,64,0,0)]
t = eng.create_table(specs, 1_000_000_000)
eng.generate(t, 20260915); eng.sync()
conj = [(0,"<",1<<30),(1,"<",int((1<<31)*0.9)),(7,"!=",63)]
plan = QueryPlan(t.col_types, conjuncts=conj, group=[2,7],
                 aggs=[("count_star",-1),("sum",3),("sum",4),("avg",5)])
ref = None
CASES = [("0", None), ("2", None), ("2", "192"), ("2", "128"), ("2", "96"),
         ("3", "128"), ("4", "128"), ("4", "96")]
for pipe, gmax in CASES:
    os.environ["BK_PIPE"] = pipe
    if gmax: os.environ["BK_PIPE_AGG_GRID"] = gmax
    else: os.environ.pop("BK_PIPE_AGG_GRID", None)
    best = None
    for rep in range(3):
        r = eng.filter_agg(t, plan, expected_groups=1<<21)
        f = r.fetch(sorted=True, max_groups=10)
        if best is None or r.kernel_ms < best[0]:
            best = (r.kernel_ms, r.ngroups, f["agg_i"][:, :5].copy(),
                    r.rows_passed)
        r.free()
    ms, ng, head, rp = best
    if ref is None:
        ref = (ng, head, rp); ok = "REF"
    else:
        ok = "PARITY-OK" if (ng == ref[0] and np.array_equal(head, ref[1])
                             and rp == ref[2]) else "PARITY-MISMATCH"
    print(f"pipe={pipe:2s} agg_grid={gmax or '-':4s} kernel={ms:7.2f} ms "
          f"ng={ng} {ok}", flush=True)
t.free()
