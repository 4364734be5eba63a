#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd db: top kernels by total time -> stdout."""
import sqlite3, sys
db = sys.argv[1]
c = sqlite3.connect(db)
sfx = [r[0] for r in c.execute("select name from sqlite_master where type='table'")
       if r[0].startswith('rocpd_kernel_dispatch')][0].replace('rocpd_kernel_dispatch_', '')
q = f"""
select ks.display_name, count(*), sum(kd.end-kd.start)
from rocpd_kernel_dispatch_{sfx} kd
join rocpd_info_kernel_symbol_{sfx} ks on kd.kernel_id = ks.id
group by ks.display_name order by 3 desc limit 15
"""
rows = list(c.execute(q))
tot = sum(r[2] for r in rows)
for name, cnt, dur in rows:
    print(f"{dur/1e6:9.2f} ms ({100*dur/tot:4.1f}%) n={cnt:6d}  {name[:90]}")
