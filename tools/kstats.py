#!/usr/bin/env python3
"""Summarize per-kernel GPU time from a rocprofv3 rocpd results.db."""
import sqlite3, sys, re, collections

db = sys.argv[1]
c = sqlite3.connect(db)
tabs = [t[0] for t in c.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sym = [t for t in tabs if 'kernel_symbol' in t][0]
disp = [t for t in tabs if 'kernel_dispatch' in t][0]
names = {r[0]: r[1] for r in c.execute(f"SELECT id, display_name FROM {sym}")}
agg = collections.defaultdict(lambda: [0, 0.0])
total = 0.0
first_end, last_start = None, None
for kid, s, e in c.execute(f"SELECT kernel_id, start, end FROM {disp}"):
    nm = names.get(kid, str(kid))
    nm = re.sub(r'<.*', '<...>', nm)[:80]
    agg[nm][0] += 1
    agg[nm][1] += (e - s) / 1e6
    total += (e - s) / 1e6
for nm, (cnt, ms) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
    print(f"{ms:10.3f} ms  {100*ms/total:5.1f}%  n={cnt:5d}  {nm}")
print(f"{total:10.3f} ms  total GPU kernel time")
