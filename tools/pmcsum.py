#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc rocpd .db: per-kernel counter totals."""
import re
import sqlite3
import sys

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
tables = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
u = None
for t in tables:
    m = re.match(r"rocpd_kernel_dispatch_(.*)", t)
    if m:
        u = m.group(1)
        break
assert u, tables


def cols(t):
    return [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]


pmc_t = f"rocpd_pmc_event_{u}"
info_t = f"rocpd_info_pmc_{u}"
kd_t = f"rocpd_kernel_dispatch_{u}"
ks_t = f"rocpd_info_kernel_symbol_{u}"
print("pmc_event cols:", cols(pmc_t))
print("info_pmc cols:", cols(info_t))

q = f"""
SELECT ks.display_name, ip.name, SUM(pe.value) total, COUNT(*) n
FROM {pmc_t} pe
JOIN {kd_t} k ON pe.event_id = k.event_id
JOIN {ks_t} ks ON k.kernel_id = ks.id
JOIN {info_t} ip ON pe.pmc_id = ip.id
GROUP BY ks.display_name, ip.name
ORDER BY total DESC
LIMIT 40
"""
try:
    for name, ctr, total, n in cur.execute(q):
        name = re.sub(r"\s+", " ", name)[:80]
        print(f"{ctr:28s} {total:>16.0f} x{n:<5d} {name}")
except Exception as e:
    print("join failed:", e)
    # fallback: dump raw
    for row in cur.execute(f"SELECT * FROM {pmc_t} LIMIT 10"):
        print(row)
