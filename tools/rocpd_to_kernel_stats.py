"""Convert a rocprofv3 rocpd .db (ROCm 7.2 default output) into the
kernel-stats CSV committed under profiles/.

Usage: python tools/rocpd_to_kernel_stats.py results.db out.csv
"""
import csv
import sqlite3
import sys


def convert(db_path, csv_path):
    c = sqlite3.connect(db_path)
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch_"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol_"))
    rows = c.execute(f"""
        SELECT k.display_name, COUNT(*), SUM(d.end-d.start),
               AVG(d.end-d.start), MIN(d.end-d.start), MAX(d.end-d.start)
        FROM {disp} d JOIN {sym} k ON d.kernel_id = k.id
        GROUP BY k.display_name ORDER BY 3 DESC""").fetchall()
    tot = sum(r[2] for r in rows) or 1
    with open(csv_path, "w", newline="") as f:
        w = csv.writer(f, quoting=csv.QUOTE_ALL)
        w.writerow(["Name", "Calls", "TotalDurationNs", "AverageNs",
                    "Percentage", "MinNs", "MaxNs"])
        for n, cnt, s, a, mn, mx in rows:
            w.writerow([n, cnt, s, round(a, 2), round(100 * s / tot, 2),
                        mn, mx])
    return rows, tot


if __name__ == "__main__":
    rows, tot = convert(sys.argv[1], sys.argv[2])
    for n, cnt, s, a, mn, mx in rows[:12]:
        print(f"{100*s/tot:5.1f}%  {cnt:5d}x  {a/1000:9.1f}us  {n[:70]}")
