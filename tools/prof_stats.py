"""Summarize a rocprofv3 results.db: per-kernel total time, calls, mean.

Usage: python tools/prof_stats.py <results.db> [top_n]
"""
import sqlite3
import sys


def summarize(path, top=25):
    db = sqlite3.connect(path)
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0]
    sfx = t[len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start), AVG(d.end - d.start),
               MAX(s.arch_vgpr_count), MAX(s.accum_vgpr_count), MAX(d.grid_size_x*d.grid_size_y*d.grid_size_z)
        FROM rocpd_kernel_dispatch_{sfx} d
        JOIN rocpd_info_kernel_symbol_{sfx} s ON s.id = d.kernel_id
        GROUP BY s.display_name ORDER BY 3 DESC""").fetchall()
    total = sum(r[2] for r in rows)
    print(f"{'kernel':64s} {'calls':>6s} {'total ms':>9s} {'%':>5s} {'mean µs':>8s} {'vgpr':>5s} {'grid':>9s}")
    for name, calls, tot, avg, vgpr, agpr, grid in rows[:top]:
        short = name.split("(")[0][:64]
        print(f"{short:64s} {calls:6d} {tot/1e6:9.3f} {100*tot/total:5.1f} "
              f"{avg/1e3:8.1f} {vgpr or 0:5d} {grid or 0:9d}")
    print(f"{'TOTAL':64s} {sum(r[1] for r in rows):6d} {total/1e6:9.3f}")
    return rows, total


if __name__ == "__main__":
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 25)
