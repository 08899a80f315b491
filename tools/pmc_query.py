#!/usr/bin/env python3
"""Print per-kernel PMC averages from a rocpd DB.
Usage: python tools/pmc_query.py <db-or-glob> [kernel-substring]"""
import glob
import sqlite3
import sys

paths = glob.glob(sys.argv[1])
like = sys.argv[2] if len(sys.argv) > 2 else "net_"
db = sqlite3.connect(paths[0])
cur = db.cursor()
tabs = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
suff = [t for t in tabs if t.startswith("rocpd_pmc_event")][0][
    len("rocpd_pmc_event_"):]
q = (f"SELECT ks.display_name, ip.name, AVG(pe.value), COUNT(*) "
     f"FROM rocpd_pmc_event_{suff} pe "
     f"JOIN rocpd_info_pmc_{suff} ip ON pe.pmc_id=ip.id "
     f"JOIN rocpd_kernel_dispatch_{suff} k ON pe.event_id=k.event_id "
     f"JOIN rocpd_info_kernel_symbol_{suff} ks ON k.kernel_id=ks.id "
     f"WHERE ks.display_name LIKE ? GROUP BY 1,2")
for kname, cname, avg, n in cur.execute(q, (f"%{like}%",)):
    print(f"{kname.split('(')[0][:44]:46s} {cname:18s} {avg:10.1f} (n={n})")
