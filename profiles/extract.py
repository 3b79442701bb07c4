#!/usr/bin/env python3
"""Extract per-kernel stats from rocprofv3 result databases (gpurun_out/*)
into committed text summaries under profiles/. Usage:
    python profiles/extract.py gpurun_out/prof4 profiles/r01_q7_kernels.txt
"""
import glob
import sqlite3
import sys


def summarize(db_dir, out_path):
    dbs = glob.glob(f"{db_dir}/*/*.db") + glob.glob(f"{db_dir}/*.db")
    if not dbs:
        raise SystemExit(f"no .db under {db_dir}")
    lines = [f"# rocprofv3 summary extracted from {dbs[0]}"]
    db = sqlite3.connect(dbs[0])
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")]
    if disp:
        sfx = disp[0].split("rocpd_kernel_dispatch_")[1]
        lines.append(f"{'kernel':60s} {'n':>6s} {'total_ms':>10s} {'avg_us':>9s} "
                     f"{'min_us':>8s} {'max_us':>9s}")
        q = (f"SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6, "
             f"AVG(k.end-k.start)/1e3, MIN(k.end-k.start)/1e3, MAX(k.end-k.start)/1e3 "
             f"FROM rocpd_kernel_dispatch_{sfx} k "
             f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
             f"GROUP BY 1 ORDER BY 3 DESC")
        for name, n, tot, avg, mn, mx in db.execute(q):
            lines.append(f"{str(name)[:60]:60s} {n:6d} {tot:10.3f} {avg:9.1f} "
                         f"{mn:8.1f} {mx:9.1f}")
        # PMC events if present
        try:
            q2 = (f"SELECT ks.display_name, p.name, COUNT(*), AVG(e.value), "
                  f"MIN(e.value), MAX(e.value) FROM rocpd_pmc_event_{sfx} e "
                  f"JOIN rocpd_info_pmc_{sfx} p ON e.pmc_id=p.id "
                  f"JOIN rocpd_kernel_dispatch_{sfx} k ON e.event_id=k.event_id "
                  f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
                  f"GROUP BY 1,2 ORDER BY 4 DESC")
            rows = list(db.execute(q2))
            if rows:
                lines.append("")
                lines.append(f"{'kernel':50s} {'counter':>12s} {'n':>5s} "
                             f"{'avg':>14s} {'min':>12s} {'max':>14s}")
                for name, pn, n, avg, mn, mx in rows:
                    lines.append(f"{str(name)[:50]:50s} {str(pn):>12s} {n:5d} "
                                 f"{avg:14,.0f} {mn:12,.0f} {mx:14,.0f}")
        except sqlite3.OperationalError:
            pass
    with open(out_path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print(f"wrote {out_path} ({len(lines)} lines)")


if __name__ == "__main__":
    summarize(sys.argv[1], sys.argv[2])
