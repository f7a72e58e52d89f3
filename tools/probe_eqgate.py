# Probe the EQ-selectivity gate (ADVICE round-1 item 2): a LOW-selectivity
# equality (2-valued column) over a >2e8-row range must NOT take the raised
# sorted-path cap — it would materialize ~rows/2 key/rowid records before
# the OOM fallback. With the gate, est = rows/2 > 2e8 keeps the default cap
# and the query runs the partitioned/dense pipeline directly. A selective EQ
# (span 2000) on the same range still qualifies for the sorted path.
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
if torch.cuda.is_available():
    torch.cuda.init()
from baikaldb_amd import GpuEngine, QueryPlan

eng = GpuEngine()
N = 300_000_000
TYPE_INT64 = 6

for tag, span in (("low-sel  (span 2)", 2), ("high-sel (span 2000)", 2000)):
    t = eng.create_table([(TYPE_INT64, 0, 0, span, 0),        # EQ column
                          (TYPE_INT64, 4, 100_000, 0, 0),     # group key
                          (TYPE_INT64, 0, 0, 1000, 0)], N)    # agg input
    try:
        eng.generate(t, 1234)
        plan = QueryPlan(t.col_types, conjuncts=[(0, "=", 1)],
                         group=[1], aggs=[("count_star", -1), ("sum", 2)])
        t0 = time.time()
        res = eng.filter_agg(t, plan, expected_groups=1 << 18)
        got = res.fetch(max_groups=1)
        ng = got["ngroups"]
        res.free()
        dt = time.time() - t0
        print(f"{tag}: {dt*1e3:8.1f} ms  ngroups={ng}", flush=True)
    finally:
        t.free()
print("PROBE DONE", flush=True)
