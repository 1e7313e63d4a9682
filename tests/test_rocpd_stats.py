"""scripts/rocpd_stats.py: rocpd db -> r1-format kernel-stats CSV."""
import csv
import sqlite3
import subprocess
import sys
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_converter_aggregates(tmp_path):
    db = sqlite3.connect(tmp_path / "r.db")
    cur = db.cursor()
    sfx = "testguid"
    cur.execute(f"CREATE TABLE rocpd_info_kernel_symbol_{sfx} "
                "(id INTEGER PRIMARY KEY, display_name TEXT)")
    cur.execute(f"CREATE TABLE rocpd_kernel_dispatch_{sfx} "
                "(kernel_id INTEGER, start BIGINT, end BIGINT)")
    cur.execute(f"INSERT INTO rocpd_info_kernel_symbol_{sfx} VALUES "
                "(1, 'fast_kernel'), (2, 'slow_kernel')")
    rows = [(1, 0, 100), (1, 200, 320), (2, 0, 1000)]
    cur.executemany(
        f"INSERT INTO rocpd_kernel_dispatch_{sfx} VALUES (?, ?, ?)", rows)
    db.commit()
    db.close()

    out = tmp_path / "stats.csv"
    r = subprocess.run(
        [sys.executable, "scripts/rocpd_stats.py", str(tmp_path / "r.db"),
         str(out)], cwd=REPO, capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    got = {row["Name"]: row for row in csv.DictReader(open(out))}
    assert got["fast_kernel"]["Calls"] == "2"
    assert got["fast_kernel"]["TotalDurationNs"] == "220"
    assert got["slow_kernel"]["TotalDurationNs"] == "1000"
    # percentages sum to ~100, sorted by total desc
    names = [row["Name"] for row in csv.DictReader(open(out))]
    assert names[0] == "slow_kernel"
