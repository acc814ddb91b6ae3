#!/usr/bin/env python3
"""Aggregate rocprofv3 PMC counters per kernel from a rocpd DB."""
import re
import sqlite3
import sys

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
try:
    rows = cur.execute(
        "SELECT s.display_name, p.name, SUM(pe.value), COUNT(*) "
        "FROM rocpd_pmc_event pe "
        "JOIN rocpd_kernel_dispatch d ON pe.event_id = d.event_id "
        "JOIN rocpd_info_kernel_symbol s ON d.kernel_id = s.id "
        "JOIN rocpd_info_pmc p ON pe.pmc_id = p.id "
        "GROUP BY s.display_name, p.name ORDER BY 3 DESC"
    ).fetchall()
except sqlite3.OperationalError as e:
    print("schema issue:", e)
    for t in cur.execute("SELECT name FROM sqlite_master WHERE type='table' AND name NOT LIKE '%_0000%'"):
        print(t[0])
    for c in cur.execute("PRAGMA table_info(rocpd_pmc_event)"):
        print("pmc_event col:", c[1])
    sys.exit(0)
agg = {}
for name, counter, val, cnt in rows:
    short = re.sub(r"<[^>]*>", "", name.split("(")[0]).strip()[:70]
    agg.setdefault(short, {})[counter] = (val, cnt)
for name, counters in agg.items():
    print(name)
    for c, (v, n) in sorted(counters.items()):
        print(f"    {c:<24} total={v:.3e}  dispatches={n}")
