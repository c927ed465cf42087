#!/usr/bin/env python3
"""Aggregate a rocprofv3 results DB into a small kernel-stats table."""
import glob
import re
import sqlite3
import sys

pat = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/prof/*/*.db"
for db_path in glob.glob(pat):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    rows = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    if not rows:
        continue
    sfx = rows[0].replace("rocpd_kernel_dispatch_", "")
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
           AVG(kd.end-kd.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 15
    """
    print(f"== {db_path}")
    print(f"{'total ms':>10} {'calls':>7} {'avg us':>9}  kernel")
    for name, n, tot_ms, avg_us in cur.execute(q):
        nm = re.sub(r"<[^>]*>", "", name).split("(")[0].replace("void ", "")
        print(f"{tot_ms:10.2f} {n:7d} {avg_us:9.1f}  {nm[:64]}")
