import sqlite3, glob, sys
db = glob.glob(sys.argv[1])[0]
con = sqlite3.connect(db)
cur = con.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = [t for t in tables if t.startswith("rocpd_region_")][0].split("rocpd_region_")[1]
# region table = host API spans
q = f"""
SELECT s.string, COUNT(*), SUM(r.end-r.start)/1e6, MAX(r.end-r.start)/1e6
FROM rocpd_region_{sfx} r JOIN rocpd_string_{sfx} s ON r.name_id = s.id
GROUP BY s.string ORDER BY 4 DESC LIMIT 15
"""
print(f"{'api':45s}{'calls':>7s}{'tot_ms':>10s}{'max_ms':>9s}")
for name, cnt, tot, mx in cur.execute(q):
    print(f"{name[:45]:45s}{cnt:7d}{tot:10.2f}{mx:9.2f}")
# top individual slow calls
q2 = f"""
SELECT s.string, (r.end-r.start)/1e6, r.start FROM rocpd_region_{sfx} r
JOIN rocpd_string_{sfx} s ON r.name_id = s.id
ORDER BY (r.end-r.start) DESC LIMIT 12
"""
print("\nslowest individual calls:")
for name, dur, st in cur.execute(q2):
    print(f"  {name[:50]:50s} {dur:9.2f} ms")
