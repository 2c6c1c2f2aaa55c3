#!/usr/bin/env python3
"""Summarize per-kernel totals from a rocprofv3 rocpd results.db."""
import glob
import sqlite3
import sys

path = sys.argv[1] if len(sys.argv) > 1 else None
if path is None or not path.endswith(".db"):
    cands = glob.glob((path or ".") + "/**/*_results.db", recursive=True)
    path = sorted(cands)[-1]
db = sqlite3.connect(path)
cur = db.cursor()
u = [r[0] for r in cur.execute("select name from sqlite_master where type='table'")
     if r[0].startswith("rocpd_kernel_dispatch_")][0].removeprefix("rocpd_kernel_dispatch_")
q = f"""
select ks.display_name, count(*), sum(k."end"-k.start)/1e6, avg(k."end"-k.start)/1e3
from rocpd_kernel_dispatch_{u} k
join rocpd_info_kernel_symbol_{u} ks on k.kernel_id = ks.id
group by ks.display_name order by 3 desc limit 25
"""
print(f"{'total_ms':>10} {'count':>7} {'avg_us':>9}  kernel   ({path})")
for name, n, tot_ms, avg_us in cur.execute(q):
    print(f"{tot_ms:10.2f} {n:7d} {avg_us:9.1f}  {name[:80]}")
