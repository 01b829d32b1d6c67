#!/usr/bin/env python3
"""Per-kernel PMC counter sums/averages from a rocprofv3 --pmc rocpd DB
(the `pmc_events` view). Usage: pmc_summary.py <results.db> [kernel-substr]"""
import sqlite3
import sys
from collections import defaultdict

con = sqlite3.connect(sys.argv[1])
filt = sys.argv[2] if len(sys.argv) > 2 else ""
agg = defaultdict(lambda: [0, 0])           # (kernel, counter) -> [sum, n]
for kname, cname, val in con.execute(
        "SELECT name, counter_name, counter_value FROM pmc_events"):
    if filt and filt not in kname:
        continue
    a = agg[(kname.split("(")[0][:48], cname)]
    a[0] += val
    a[1] += 1
for (k, c), (s, n) in sorted(agg.items()):
    print(f"{k:50s} {c:24s} sum={s:>16.0f} n={n:>3} avg={s/max(n,1):>14.1f}")
