# tools/sweep_part.py — sweep partition count P and grid size for the
# partitioned aggregate pipeline on the config-3-shaped workload.
import itertools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13
SEED = 20260915
N = int(os.environ.get("SWEEP_ROWS", 300_000_000))

eng = GpuEngine()
specs = [(TYPE_INT64, 0, 0, 1 << 31, 0),
         (TYPE_INT64, 0, 0, 1 << 31, 0),
         (TYPE_INT64, 4, 16384, 0, 0),
         (TYPE_INT64, 0, 0, 1000, 0),
         (TYPE_DOUBLE, 3, 0, 0, 0),
         (TYPE_DOUBLE, 3, 0, 0, 0),
         (TYPE_STRING, 2, 64, 0, 0)]
t = eng.create_table(specs, N)
eng.generate(t, SEED)
eng.sync()
plan = QueryPlan(t.col_types,
                 conjuncts=[(0, "<", 1 << 30), (1, "<", int((1 << 31) * 0.9)),
                            (6, "!=", 63)],
                 group=[2, 6],
                 aggs=[("count_star", -1), ("sum", 3), ("sum", 4), ("avg", 5)])

for P, blocks in itertools.product([128, 256, 512, 1024], [1024, 2048]):
    os.environ["BK_PART_P"] = str(P)
    os.environ["BK_PART_BLOCKS"] = str(blocks)
    best = None
    for rep in range(3):
        r = eng.filter_agg(t, plan, expected_groups=1 << 21)
        bd = r.breakdown()
        tot = r.kernel_ms
        if best is None or tot < best[0]:
            best = (tot, bd, r.ngroups)
        r.free()
    tot, bd, ng = best
    print(f"P={P:5d} blocks={blocks:5d} total={tot:8.3f}ms groups={ng} "
          f"{ {k: round(v,3) for k,v in bd.items()} }", flush=True)
t.free()
