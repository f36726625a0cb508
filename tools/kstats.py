#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db: total time per kernel, hot first."""
import sqlite3, sys, re
path = sys.argv[1]
db = sqlite3.connect(path)
cur = db.cursor()
t = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0]
u = t[len("rocpd_kernel_dispatch_"):]
q = f"""
SELECT ks.display_name, COUNT(*) cnt, SUM(k.end-k.start)/1e6 ms,
       AVG(k.end-k.start)/1e3 us, MAX(k.grid_size_x) gx
FROM rocpd_kernel_dispatch_{u} k
JOIN rocpd_info_kernel_symbol_{u} ks ON k.kernel_id = ks.id
GROUP BY ks.display_name ORDER BY ms DESC LIMIT {int(sys.argv[2]) if len(sys.argv)>2 else 30}
"""
total = cur.execute(f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{u}").fetchone()[0]
print(f"total GPU kernel time: {total:.2f} ms")
for name, cnt, ms, us, gx in cur.execute(q):
    name = re.sub(r"\s+", " ", name)[:110]
    print(f"{ms:9.3f} ms {100*ms/total:5.1f}%  x{cnt:<5d} {us:9.1f} us/call grid={gx:<8d} {name}")
