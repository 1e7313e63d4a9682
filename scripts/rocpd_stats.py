"""Aggregate a rocprofv3 rocpd .db into the kernel-stats CSV shape the
r1 rocprofv3 emitted (Name,Calls,TotalDurationNs,AverageNs,Percentage,
MinNs,MaxNs,StdDev) so `profiles/` stays one consistent format.

Usage: python scripts/rocpd_stats.py <results.db> <out.csv>
"""
import csv
import math
import sqlite3
import sys


def main() -> None:
    db_path, out_path = sys.argv[1], sys.argv[2]
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(cur.execute(
        f"SELECT s.display_name, d.end - d.start FROM {kd} d "
        f"JOIN {ks} s ON s.id = d.kernel_id"))
    agg = {}
    for name, dur in rows:
        agg.setdefault(name, []).append(dur)
    grand = sum(sum(v) for v in agg.values())
    out = []
    for name, durs in agg.items():
        n = len(durs)
        tot = sum(durs)
        mean = tot / n
        var = sum((d - mean) ** 2 for d in durs) / n if n > 1 else 0.0
        out.append((name, n, tot, mean, 100.0 * tot / grand,
                    min(durs), max(durs), math.sqrt(var)))
    out.sort(key=lambda r: -r[2])
    with open(out_path, "w", newline="") as f:
        w = csv.writer(f, quoting=csv.QUOTE_ALL)
        w.writerow(["Name", "Calls", "TotalDurationNs", "AverageNs",
                    "Percentage", "MinNs", "MaxNs", "StdDev"])
        for name, n, tot, mean, pct, mn, mx, sd in out:
            w.writerow([name, n, tot, f"{mean:.6f}", f"{pct:.2f}",
                        mn, mx, f"{sd:.6f}"])
    print(f"{out_path}: {len(out)} kernels, {grand/1e6:.1f} ms total")


if __name__ == "__main__":
    main()
