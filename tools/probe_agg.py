# tools/probe_agg.py — perf ablation probe for the fused filter+agg kernel.
# Runs variants of the config-2 workload on one GPU and prints kernel ms:
#   full    : WHERE c0<K AND c1<K2 GROUP BY c2 SUM(c3)   (the real query)
#   nogroup : same WHERE, COUNT(*) only (scalar register path — no hash table)
#   nofilter: GROUP BY c2 SUM(c3) with no predicates
#   1group  : GROUP BY constant-ish (c2 domain 1) — pure hot-slot contention
# Meant to run under rocprofv3 --kernel-trace --stats as well.
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

TYPE_INT64 = 6
D_UNI, D_SKEW = 0, 1
SEED = 20260915
N = int(os.environ.get("PROBE_ROWS", 100_000_000))

eng = GpuEngine()
specs = [(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
         (TYPE_INT64, D_UNI, 0, 1 << 31, 0),
         (TYPE_INT64, 4, 100_000, 0, 0),     # Zipf-like group key
         (TYPE_INT64, D_UNI, 0, 1000, 0),
         (TYPE_INT64, D_UNI, 0, 2, 0)]
t = eng.create_table(specs, N)
t0 = time.perf_counter()
eng.generate(t, SEED)
eng.sync()
print(f"generate: {time.perf_counter()-t0:.3f}s for {N} rows", flush=True)

conj = [(0, "<", 1 << 30), (1, "<", int((1 << 31) * 0.9))]
variants = {
    "full":     dict(conjuncts=conj, group=[2], aggs=[("sum", 3)]),
    "nogroup":  dict(conjuncts=conj, group=[], aggs=[("count_star", -1)]),
    "nofilter": dict(conjuncts=[], group=[2], aggs=[("sum", 3)]),
    "1group":   dict(conjuncts=conj, group=[4], aggs=[("sum", 3)]),
}
for name, kw in variants.items():
    plan = QueryPlan(t.col_types, **kw)
    times = []
    bd = None
    for rep in range(4):
        r = eng.filter_agg(t, plan, expected_groups=1 << 18)
        times.append(r.kernel_ms)
        if rep == 0:
            print(f"{name}: groups={r.ngroups} rows_passed={r.rows_passed}",
                  flush=True)
        bd = r.breakdown()
        r.free()
    best = min(times[1:])
    gbs = N * 32 / best / 1e6
    print(f"{name}: kernel_ms={best:.3f}  ({N/best/1e6:.2f} Grows/s, "
          f"{gbs:.0f} GB/s if 32B/row)  breakdown={ {k: round(v,3) for k,v in bd.items()} }",
          flush=True)
t.free()
