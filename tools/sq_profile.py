# tools/sq_profile.py — summarize a rocprofv3 --pmc run's SQ wait/active
# counters per kernel: decomposes wave cycles into parked-wait (s_waitcnt /
# barrier), issue-stall, and active-issue (MI355X_MICROARCH.md §PMC slots:
# WAIT_ANY + WAIT_INST_ANY + ACTIVE_INST_ANY ~= WAVE_CYCLES, disjoint).
import glob
import sqlite3
import sys

f = sorted(glob.glob(sys.argv[1]))[-1]
db = sqlite3.connect(f)
cur = db.cursor()
tabs = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
g = [t for t in tabs if t.startswith('rocpd_kernel_dispatch')][0] \
    .replace('rocpd_kernel_dispatch_', '')
pe = [t for t in tabs if t.startswith('rocpd_pmc_event')][0]
# counter-name table (rocpd_info_pmc_<guid>: id, name, ...)
ctab = [t for t in tabs if t.startswith('rocpd_info_pmc')]
name_expr = "c.name"
join_expr = f"JOIN {ctab[0]} c ON p.pmc_id = c.id"
q = f"""SELECT s.display_name, {name_expr}, COUNT(*), SUM(p.value),
               SUM(d.end-d.start)/COUNT(*)/1e6
        FROM {pe} p JOIN rocpd_kernel_dispatch_{g} d ON p.event_id=d.event_id
        JOIN rocpd_info_kernel_symbol_{g} s ON d.kernel_id=s.id
        {join_expr}
        GROUP BY s.display_name, {name_expr}"""
rows = {}
for kn, cn, c, v, ms in cur.execute(q):
    k = kn.split('(')[0][:36]
    rows.setdefault(k, {"n": c, "ms": ms})[str(cn)] = v
print(f"# SQ counters — {f}")
for k, d in sorted(rows.items(), key=lambda kv: -kv[1]["ms"]):
    n, ms = d.pop("n"), d.pop("ms")
    wave = d.get("SQ_WAVE_CYCLES", 0) or 1
    parts = " ".join(f"{cn.replace('SQ_', '')}={v / wave * 100:5.1f}%"
                     for cn, v in sorted(d.items()) if cn != "SQ_WAVE_CYCLES")
    print(f"{k:38s} n={n:3d} avg={ms:7.3f} ms  {parts}")
