"""rocpd results.db -> kernel-stats CSV (Name,Calls,TotalDurationNs,
AverageNs,Percentage). rocprofv3 on this image sometimes emits only the db."""

import csv
import sqlite3
import sys


def main(db_path: str, out_csv: str) -> None:
    con = sqlite3.connect(db_path)
    tables = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0]
    uuid = disp.replace("rocpd_kernel_dispatch_", "")
    rows = con.execute(
        f"""SELECT s.display_name, COUNT(*), SUM(k.end-k.start), AVG(k.end-k.start)
            FROM rocpd_kernel_dispatch_{uuid} k
            JOIN rocpd_info_kernel_symbol_{uuid} s ON k.kernel_id = s.id
            GROUP BY s.display_name ORDER BY SUM(k.end-k.start) DESC"""
    ).fetchall()
    total = sum(r[2] for r in rows) or 1
    with open(out_csv, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["Name", "Calls", "TotalDurationNs", "AverageNs", "Percentage"])
        for name, calls, tot, avg in rows:
            w.writerow([name, calls, tot, round(avg, 1), round(100.0 * tot / total, 2)])
    print(f"wrote {out_csv}: {len(rows)} kernels, total {total/1e9:.3f}s")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
