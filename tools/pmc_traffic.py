# tools/pmc_traffic.py — turn rocprofv3 --pmc FETCH_SIZE / WRITE_SIZE rocpd
# databases (collected in SEPARATE passes, per the gfx950 TCC slot limits)
# into per-launch HBM traffic for the dominant kernel of each workload, and
# commit them as profiles/traffic_<config>.json for bench.py's
# roofline.traffic field.
#
# Calibration (MI355X_MICROARCH.md §HBM): on gfx950 FETCH_SIZE reports 1/2 of
# the bytes of a wide coalesced streaming read — verified here against
# k_generate, whose WRITE_SIZE reads exactly the table bytes (60.00 GB for
# config3) while its FETCH is ~0; we double FETCH before summing. WRITE_SIZE
# is used uncalibrated (1:1 on the k_generate known byte count).
import glob
import json
import os
import sqlite3
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
FETCH_CAL = 2.0


def per_kernel(dbpat):
    f = sorted(glob.glob(dbpat))[-1]
    db = sqlite3.connect(f)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    g = [t for t in tabs if t.startswith('rocpd_kernel_dispatch')][0] \
        .replace('rocpd_kernel_dispatch_', '')
    pe = [t for t in tabs if t.startswith('rocpd_pmc_event')][0]
    q = f"""SELECT s.display_name, COUNT(*), SUM(p.value),
                   SUM(d.end-d.start)/COUNT(*)/1e6
            FROM {pe} p JOIN rocpd_kernel_dispatch_{g} d ON p.event_id=d.event_id
            JOIN rocpd_info_kernel_symbol_{g} s ON d.kernel_id=s.id
            GROUP BY s.display_name"""
    out = {}
    for n, c, v, ms in cur.execute(q):
        key = n.split('(')[0]
        out[key] = {"launches": c, "kb_total": v, "avg_ms": ms}
    return out, f


def main(config, dominant):
    fetch, ff = per_kernel(
        f"{REPO}/gpurun_out/pmcZ_{config}_FETCH_SIZE/runc/*_results.db")
    write, wf = per_kernel(
        f"{REPO}/gpurun_out/pmcZ_{config}_WRITE_SIZE/runc/*_results.db")
    lines = [f"# PMC HBM traffic — {config} (separate FETCH_SIZE / WRITE_SIZE passes)",
             f"# sources: {os.path.basename(ff)}, {os.path.basename(wf)}",
             f"# FETCH calibrated x{FETCH_CAL} (wide coalesced reads report 1/2; "
             f"MI355X_MICROARCH.md §HBM), WRITE 1:1 (k_generate writes its exact "
             f"table bytes)",
             f"{'kernel':28s} {'n':>3s} {'read_GB/launch':>15s} "
             f"{'write_GB/launch':>16s} {'total_GB':>9s} {'avg_ms':>7s}"]
    total = {}
    for k in sorted(set(fetch) | set(write)):
        fe = fetch.get(k, {"launches": 1, "kb_total": 0, "avg_ms": 0})
        wr = write.get(k, {"launches": 1, "kb_total": 0, "avg_ms": 0})
        rd_gb = fe["kb_total"] * 1024 * FETCH_CAL / fe["launches"] / 1e9
        wr_gb = wr["kb_total"] * 1024 / wr["launches"] / 1e9
        total[k] = (rd_gb + wr_gb, fe["avg_ms"] or wr["avg_ms"])
        lines.append(f"{k[:28]:28s} {fe['launches']:3d} {rd_gb:15.2f} "
                     f"{wr_gb:16.2f} {rd_gb + wr_gb:9.2f} {total[k][1]:7.2f}")
    txt = "\n".join(lines) + "\n"
    open(f"{REPO}/profiles/r01_pmc_{config}.txt", "w").write(txt)
    print(txt)
    dk = [k for k in total if dominant in k]
    dk.sort(key=lambda k: total[k][0], reverse=True)
    dom_bytes = total[dk[0]][0] * 1e9
    json.dump({"dominant": dominant,
               "bytes_per_launch": dom_bytes,
               "note": "HBM bytes/launch of the dominant kernel "
                       "(FETCH x2 calibrated + WRITE), rocprofv3 --pmc"},
              open(f"{REPO}/profiles/traffic_{config}.json", "w"), indent=1)
    print(f"-> traffic_{config}.json: {dom_bytes/1e9:.2f} GB/launch ({dominant})")


if __name__ == "__main__":
    import sys as _sys
    which = _sys.argv[1] if len(_sys.argv) > 1 else "all"
    # dominant kernels for the round-2 dense-span pipeline (bkdpart.inc);
    # pass a third arg to override (old hash-path names: k_part_scatter /
    # k_dedup_mat)
    dom = _sys.argv[2] if len(_sys.argv) > 2 else None
    if which in ("all", "c3"):
        main("config3_1e9_mixed", dom or "k_dscatter")
    if which in ("all", "c5"):
        main("config5_1e9_sort", dom or "k_topk_scan")
    if which in ("all", "c2b"):
        main("config2_1e9_8int64", dom or "k_dhisto")
