# Summarize a rocprofv3 kernel-trace rocpd DB into a per-kernel table.
import glob
import sqlite3
import sys

pat = sys.argv[1]
f = sorted(glob.glob(pat))[-1]
db = sqlite3.connect(f)
cur = db.cursor()
tabs = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
g = [t for t in tabs if t.startswith('rocpd_kernel_dispatch')][0] \
    .replace('rocpd_kernel_dispatch_', '')
rows = list(cur.execute(f"""
    SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
           AVG(d.end-d.start)/1e6
    FROM rocpd_kernel_dispatch_{g} d
    JOIN rocpd_info_kernel_symbol_{g} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC"""))
total = sum(r[2] for r in rows)
print(f"# per-kernel GPU time — {f}")
print(f"{'kernel':44s} {'calls':>5s} {'total_ms':>9s} {'avg_ms':>8s} {'%':>6s}")
for name, calls, tot, avg in rows:
    name = name.split('(')[0][:44]
    print(f"{name:44s} {calls:5d} {tot:9.2f} {avg:8.3f} {100*tot/total:6.2f}")
print(f"{'TOTAL':44s} {'':5s} {total:9.2f}")
