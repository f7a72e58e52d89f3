# tools/probe_dabs.py — hot-bucket absorption (ABS) probe on the config3
# shape: per-kernel breakdown (BK_DPIPE=1) with BK_DABS on/off + result
# checksum equality.
import os
import sys

sys.path.insert(0, "/root/repo")
os.environ["BK_DPIPE"] = "1"
os.environ["BK_DABS"] = "1"   # opt-in (measured-dead default)
from baikaldb_amd import GpuEngine, QueryPlan  # noqa: E402

T_I, T_D, T_S = 6, 12, 13
N = int(sys.argv[1]) if len(sys.argv) > 1 else 1_000_000_000
eng = GpuEngine()
specs = [(T_I, 0, 0, 1 << 31, 0), (T_I, 0, 0, 1 << 31, 0),
         (T_I, 4, 16384, 0, 0), (T_I, 0, 0, 1000, 0),
         (T_D, 3, 0, 0, 0), (T_D, 3, 0, 0, 0),
         (T_I, 0, 0, 1 << 31, 0), (T_S, 2, 64, 0, 0)]
t = eng.create_table(specs, N)
eng.generate(t, 20260915)
eng.sync()
conj = [(0, "<", 1 << 30), (1, "<", int((1 << 31) * 0.9)), (7, "!=", 63)]
plan = QueryPlan(t.col_types, conjuncts=conj, group=[2, 7],
                 aggs=[("count_star", -1), ("sum", 3), ("sum", 4),
                       ("avg", 5)])


def run(tag):
    sums = None
    for rep in range(3):
        r = eng.filter_agg(t, plan, expected_groups=1 << 21)
        bd = r.breakdown()
        ng = r.ngroups
        if rep == 2:
            sums = (ng, r.rows_passed)
        r.free()
        tot = sum(bd.values())
        print(f"[{tag}] rep{rep} ngroups={ng} total={tot:.2f}ms "
              + " ".join(f"{k}={v:.2f}" for k, v in bd.items()),
              flush=True)
    return sums


a = run("DABS=on")
os.environ["BK_DABS"] = "0"
b = run("DABS=off")
print("absorb checksum:", a)
print("noabs  checksum:", b)
print("MATCH" if a == b else "MISMATCH", flush=True)
