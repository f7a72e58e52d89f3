# tools/narrow_probe.py — c2b (north-star 1e9x8 INT64) and c3 with narrow
# physical column storage ON vs OFF.  python tools/narrow_probe.py [c2b|c3]
import sys
import time

sys.path.insert(0, "/root/repo")
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

T_I, T_D, T_S = 6, 12, 13
which = sys.argv[1] if len(sys.argv) > 1 else "c2b"
eng = GpuEngine()
if which == "c2b":
    specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 20, 0),
             (T_I, 4, 100_000, 0, 0), (T_I, 0, 0, 1000, 0)] + \
            [(T_I, 0, 0, 1 << 31, 0)] * 4
    conj = [(0, "<", 1 << 30), (1, "=", 7)]
    group, aggs, eg = [2], [("sum", 3)], 1 << 18
else:
    specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 1 << 31, 0),
             (T_I, 4, 16384, 0, 0), (T_I, 0, 0, 1000, 0),
             (T_D, 3, 0, 0, 0), (T_D, 3, 0, 0, 0),
             (T_I, 0, 0, 1 << 31, 0), (T_S, 2, 64, 0, 0)]
    conj = [(0, "<", 1 << 30), (1, "<", int((1 << 31) * 0.9)), (7, "!=", 63)]
    group, aggs, eg = [2, 7], [("count_star", -1), ("sum", 3), ("sum", 4),
                               ("avg", 5)], 1 << 21
if which == "c5":
    specs = [(T_I, 0, 0, 1 << 31, 0)] * 3
    t = eng.create_table(specs, 1_000_000_000)
    for narrow in (False, True):
        eng.generate(t, 20260915, compact=narrow)
        eng.sync()
        print("widths", [eng.col_width(t, c) for c in range(3)])
        for rep in range(3):
            t0 = time.time()
            ids = eng.sort_topk(t, [(0, 1, 1), (1, 1, 1)], 1_000_000)
            dt = (time.time() - t0) * 1e3
            print(("narrow" if narrow else "wide"), rep,
                  "wall_ms=%.2f" % dt, "kernel_ms=%.2f" % eng.topk_kernel_ms())
    t.free()
    sys.exit(0)
t = eng.create_table(specs, 1_000_000_000)
plan = QueryPlan(t.col_types, conjuncts=conj, group=group, aggs=aggs)
base = {}
for narrow in (False, True):
    eng.generate(t, 20260915, compact=narrow)
    eng.sync()
    print("widths", [eng.col_width(t, c) for c in range(len(specs))])
    for rep in range(3):
        t0 = time.time()
        r = eng.filter_agg(t, plan, expected_groups=eg)
        eng.sync()
        dt = (time.time() - t0) * 1e3
        tag = "narrow" if narrow else "wide"
        print(f"{tag} rep{rep} wall_ms={dt:.2f} kernel_ms={r.kernel_ms:.2f} "
              f"ngroups={r.ngroups} rows_passed={r.rows_passed}")
        if rep == 2:
            base[tag] = (r.ngroups, r.rows_passed)
            print(tag, "breakdown", r.breakdown())
        r.free()
assert base["wide"] == base["narrow"], base
print("RESULTS MATCH", base)
t.free()
