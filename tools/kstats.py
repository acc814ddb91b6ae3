#!/usr/bin/env python3
"""Summarise a rocpd kernel-trace DB: per-kernel totals + gap analysis."""
import sqlite3
import sys
import re

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
rows = cur.execute(
    "SELECT d.start, d.end, s.display_name FROM rocpd_kernel_dispatch d "
    "JOIN rocpd_info_kernel_symbol s ON d.kernel_id = s.id ORDER BY d.start"
).fetchall()
print(f"total dispatches: {len(rows)}")
if not rows:
    sys.exit(0)
span = (rows[-1][1] - rows[0][0]) / 1e6
busy = 0
last_end = 0
agg = {}
for st, en, name in rows:
    if st > last_end:
        busy += en - st
        last_end = en
    elif en > last_end:
        busy += en - last_end
        last_end = en
    short = re.sub(r"<[^>]*>", "", name.split("(")[0]).strip()
    a = agg.setdefault(short, [0, 0.0])
    a[0] += 1
    a[1] += (en - st) / 1e6
print(f"wall span: {span:.1f} ms   kernel-busy: {busy/1e6:.1f} ms   gap: {span - busy/1e6:.1f} ms ({(1-busy/1e6/span)*100:.0f}%)")
print(f"{'kernel':<72}{'count':>8}{'total ms':>12}{'avg us':>10}")
for name, (cnt, ms) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:30]:
    print(f"{name[:72]:<72}{cnt:>8}{ms:>12.2f}{ms/cnt*1000:>10.1f}")

# RAWDUMP: top raw kernel names (un-shortened) for identification
raw = {}
for st, en, name in rows:
    a = raw.setdefault(name, [0, 0.0])
    a[0] += 1
    a[1] += (en - st) / 1e6
print("\n-- top raw names --")
for name, (c, tot) in sorted(raw.items(), key=lambda kv: -kv[1][1])[:18]:
    print(f"{tot:9.2f} ms  x{c:<6d} {name[:150]}")
