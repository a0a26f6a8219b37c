#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database (kernel-trace --stats run)
into a per-kernel CSV: name, calls, total ms, mean/min/max us.
Usage: python tools/rocpd_stats.py <results.db> [out.csv]"""
import sqlite3, sys

db = sys.argv[1]
out = sys.argv[2] if len(sys.argv) > 2 else None
c = sqlite3.connect(db)
uuid = [r[0] for r in c.execute(
    "select name from sqlite_master where name like 'rocpd_kernel_dispatch%'")][0]
uuid = uuid[len("rocpd_kernel_dispatch_"):]
q = f"""
select k.display_name as name, count(*) as calls,
       sum(d.end-d.start)/1e6 as total_ms,
       avg(d.end-d.start)/1e3 as mean_us,
       min(d.end-d.start)/1e3 as min_us,
       max(d.end-d.start)/1e3 as max_us
from rocpd_kernel_dispatch_{uuid} d
join rocpd_info_kernel_symbol_{uuid} k on d.kernel_id = k.id
group by k.display_name order by total_ms desc
"""
rows = c.execute(q).fetchall()
lines = ["name,calls,total_ms,mean_us,min_us,max_us"]
for r in rows:
    lines.append(f'"{r[0]}",{r[1]},{r[2]:.3f},{r[3]:.2f},{r[4]:.2f},{r[5]:.2f}')
text = "\n".join(lines)
print(text)
if out:
    open(out, "w").write(text + "\n")
