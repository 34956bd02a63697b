"""Summarize a rocprofv3 --pmc run (rocpd SQLite output) into per-kernel
counter totals. Usage: python tests/pmc_summarize.py <dir-with-db>"""
import glob
import sqlite3
import sys

paths = (glob.glob(sys.argv[1] + "/*.db")
         + glob.glob(sys.argv[1] + "/**/*.db", recursive=True))
db = sqlite3.connect(paths[0])
cur = db.cursor()
tables = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
print("tables:", [t for t in tables if "pmc" in t][:6])

sfx = None
for t in tables:
    if t.startswith("rocpd_pmc_event_"):
        sfx = t[len("rocpd_pmc_event_"):]
        break

if sfx:
    cols = {t: [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
            for t in (f"rocpd_pmc_event_{sfx}", f"rocpd_info_pmc_{sfx}",
                      f"rocpd_kernel_dispatch_{sfx}",
                      f"rocpd_info_kernel_symbol_{sfx}")}
    for t, c in cols.items():
        print(t.split("_0000")[0], c)
    q = f"""
    SELECT ks.display_name, ip.name, SUM(pe.value), COUNT(*)
    FROM rocpd_pmc_event_{sfx} pe
    JOIN rocpd_info_pmc_{sfx} ip ON pe.pmc_id = ip.id
    JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id = kd.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name, ip.name
    """
    agg = {}
    try:
        for name, cname, val, cnt in cur.execute(q):
            short = name.split("(")[0][:60]
            agg.setdefault(short, {})[cname] = (val, cnt)
    except Exception as e:
        print("join query failed:", e)
    for name in sorted(agg):
        parts = " ".join(f"{c.split('.')[-1]}={v:.3e}(n={n})"
                         for c, (v, n) in sorted(agg[name].items()))
        print(f"{name}: {parts}")
else:
    print("no rocpd_pmc_event table found")
