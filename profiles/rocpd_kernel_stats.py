"""Extract a per-kernel time table from a rocprofv3 rocpd SQLite database
(ROCm 7.2 default output: `rocprofv3 --kernel-trace --stats` writes
<name>_results.db). Usage:

    python profiles/rocpd_kernel_stats.py gpurun_out/profadult/adultprof_results.db > profiles/adult_kernel_stats_r02.csv
"""
import sqlite3
import sys


def main(path: str) -> None:
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'"
    )]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(d.end - d.start)/1000.0,
           AVG(d.end - d.start)/1000.0
    FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """
    rows = list(cur.execute(q))
    tot = sum(r[2] for r in rows)
    print('"Name","Calls","TotalDurationUs","AverageUs","Percentage"')
    for name, calls, tus, aus in rows:
        clean = name.replace('"', "'")
        print(f'"{clean}",{calls},{tus:.1f},{aus:.2f},{tus / tot * 100:.2f}')


if __name__ == "__main__":
    main(sys.argv[1])
