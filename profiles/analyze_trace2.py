import sqlite3, glob, sys
db = glob.glob(sys.argv[1])[0]
con = sqlite3.connect(db)
cur = con.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = [t for t in tables if t.startswith("rocpd_region_")][0].split("rocpd_region_")[1]
# slowest kernel INSTANCES
q = f"""
SELECT ks.display_name, (k.end-k.start)/1e6, k.start
FROM rocpd_kernel_dispatch_{sfx} k
JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
ORDER BY (k.end-k.start) DESC LIMIT 10
"""
print("slowest kernel instances:")
for name, dur, st in cur.execute(q):
    print(f"  {name[:60]:60s} {dur:9.3f} ms @ {st}")
# GPU idle gaps
rows = list(cur.execute(f"SELECT start, end FROM rocpd_kernel_dispatch_{sfx} ORDER BY start"))
gaps = []
cur_end = rows[0][1]
for s, e in rows[1:]:
    if s > cur_end:
        gaps.append((s - cur_end, cur_end, s))
    cur_end = max(cur_end, e)
gaps.sort(reverse=True)
print("\nlargest GPU idle gaps:")
for g, a, b in gaps[:10]:
    print(f"  {g/1e6:9.2f} ms idle  [{a} .. {b}]")
# what host API spans overlap the biggest gap
if gaps:
    g, a, b = gaps[0]
    q3 = f"""
    SELECT s.string, r.start, r.end FROM rocpd_region_{sfx} r
    JOIN rocpd_string_{sfx} s ON r.name_id = s.id
    WHERE r.end > {a} AND r.start < {b} AND (r.end-r.start) > 1000000
    ORDER BY (r.end-r.start) DESC LIMIT 8
    """
    print("\nhost spans overlapping biggest gap:")
    for name, st, en in cur.execute(q3):
        print(f"  {name[:50]:50s} {(en-st)/1e6:9.2f} ms")
