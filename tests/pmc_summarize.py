import glob
import sqlite3
import sys

db = sqlite3.connect(glob.glob(sys.argv[1] + "/*.db")[0])
cur = db.cursor()
tables = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
print("tables:", [t for t in tables if "counter" in t or "pmc" in t][:6])
try:
    rows = list(cur.execute("""
        SELECT kernel_name, counter_name, sum(value), count(*)
        FROM counters_collection GROUP BY kernel_name, counter_name"""))
except Exception as e:
    print("query1 failed:", e)
    rows = []
agg = {}
for name, cname, val, cnt in rows:
    short = name.split("(")[0][:60]
    agg.setdefault(short, {})[cname] = (val, cnt)
for kern, cs in sorted(agg.items()):
    if not any(s in kern for s in ("adamw", "swiglu", "rmsnorm", "rope",
                                   "cross_entropy", "colsum", "attn", "fmha",
                                   "Cijk")):
        continue
    parts = []
    for cname, (val, cnt) in sorted(cs.items()):
        parts.append(f"{cname}={val:.3e}(n={cnt})")
    print(f"{kern}: " + " ".join(parts))
