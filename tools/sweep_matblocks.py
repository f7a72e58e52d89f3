# tools/sweep_matblocks.py — BK_MAT_BLOCKS grid sweep for the vector-load
# mat on the c2b (north-star) shape.
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

T_I = 6
eng = GpuEngine()
specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 20, 0),
         (T_I, 4, 100_000, 0, 0), (T_I, 0, 0, 1000, 0)] + \
        [(T_I, 0, 0, 1 << 31, 0)] * 4
t = eng.create_table(specs, 1_000_000_000)
eng.generate(t, 20260915)
eng.sync()
plan = QueryPlan(t.col_types, conjuncts=[(0, "<", 1 << 30), (1, "=", 7)],
                 group=[2], aggs=[("sum", 3)])
for blocks in ("2048", "4096", "8192", "16384", "32768"):
    os.environ["BK_MAT_BLOCKS"] = blocks
    best = None
    for rep in range(4):
        r = eng.filter_agg(t, plan, expected_groups=1 << 18)
        bd = r.breakdown()
        if best is None or r.kernel_ms < best[0]:
            best = (r.kernel_ms, bd)
        r.free()
    print(f"blocks={blocks:6s} total={best[0]:5.2f} " +
          " ".join(f"{k}={v:.2f}" for k, v in best[1].items() if v >= 0.05),
          flush=True)
t.free()
