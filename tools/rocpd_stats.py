"""Extract per-kernel stats CSV from a rocprofv3 rocpd SQLite database.

Newer rocprofv3 builds emit `<name>_results.db` (rocpd schema) instead of
the older kernel-stats CSV; this reproduces the CSV summary
(Name,Calls,TotalDurationNs,AverageNs,Percentage,MinNs,MaxNs) from the
kernel-dispatch table.

    python tools/rocpd_stats.py gpurun_out/prof/<x>_results.db out.csv
"""

import csv
import sqlite3
import sys


def extract(db_path, out_path):
    c = sqlite3.connect(db_path)
    tables = {r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")}
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    rows = list(c.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
               AVG(d.end - d.start), MIN(d.end - d.start),
               MAX(d.end - d.start)
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"""))
    total = sum(r[2] for r in rows) or 1
    with open(out_path, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["Name", "Calls", "TotalDurationNs", "AverageNs",
                    "Percentage", "MinNs", "MaxNs"])
        for r in rows:
            w.writerow([r[0], r[1], r[2], round(r[3], 1),
                        round(100 * r[2] / total, 2), r[4], r[5]])
    return len(rows), total


if __name__ == "__main__":
    n, total = extract(sys.argv[1], sys.argv[2])
    print(f"{n} kernels, {total / 1e6:.2f} ms total -> {sys.argv[2]}")
