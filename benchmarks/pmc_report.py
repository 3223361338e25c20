"""Aggregate rocprofv3 PMC rocpd .db: per-kernel counter sums + durations.

Usage: python benchmarks/pmc_report.py <db> — schema-introspecting so it
survives rocpd layout changes between rocprofv3 versions.
"""
import sqlite3
import sys
from collections import defaultdict

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
tabs = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
def find(sub):
    return [t for t in tabs if sub in t]
pmc_t = find("pmc_event")
if not pmc_t:
    print("tables:", tabs)
    sys.exit("no pmc_event table")
pmc_t = pmc_t[0]
sfx = pmc_t.replace("rocpd_pmc_event_", "")
cols = {t: [c[1] for c in cur.execute(f"PRAGMA table_info({t})")] for t in tabs}
print("# pmc cols:", cols[pmc_t])
kd = f"rocpd_kernel_dispatch_{sfx}"
ks = f"rocpd_info_kernel_symbol_{sfx}"
# counter names table
cn = find("info_pmc")
name_by_id = {}
if cn:
    ccols = cols[cn[0]]
    idc = "id" if "id" in ccols else ccols[0]
    nmc = "name" if "name" in ccols else ("symbol" if "symbol" in ccols else ccols[1])
    for i, n in cur.execute(f"SELECT {idc}, {nmc} FROM {cn[0]}"):
        name_by_id[i] = n
pcols = cols[pmc_t]
kcols = cols[kd]
print("# kd cols:", kcols)
cid = ("pmc_id" if "pmc_id" in pcols else
       "counter_id" if "counter_id" in pcols else
       [c for c in pcols if "counter" in c or "pmc" in c][0])
val = "value" if "value" in pcols else pcols[-1]
if "event_id" in pcols and "event_id" in kcols:
    join = f"p.event_id = d.event_id"
elif "dispatch_id" in pcols:
    join = f"p.dispatch_id = d.id"
else:
    join = f"p.{pcols[0]} = d.id"
rows = cur.execute(f"""
  SELECT sym.display_name, p.{cid}, SUM(p.{val})
  FROM {pmc_t} p JOIN {kd} d ON {join}
  JOIN {ks} sym ON d.kernel_id = sym.id
  GROUP BY 1, 2""").fetchall()
durs = dict(cur.execute(f"""
  SELECT sym.display_name, SUM(d.end-d.start)/1e6 FROM {kd} d
  JOIN {ks} sym ON d.kernel_id = sym.id GROUP BY 1""").fetchall())
agg = defaultdict(dict)
for name, c, v in rows:
    agg[str(name).split("(")[0][:60]][name_by_id.get(c, c)] = v
for k, cc in sorted(agg.items()):
    short = k
    dur = [v for n, v in durs.items() if str(n).split("(")[0][:60] == k]
    print(f"\n== {short}  total_ms={dur[0] if dur else '?':.2f}" if dur else f"\n== {short}")
    for n, v in sorted(cc.items()):
        print(f"   {n}: {v:.3e}")
