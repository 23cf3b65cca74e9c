#!/usr/bin/env python3
"""Extract per-kernel stats (count, total/avg/min/max duration) and PMC
counter sums from rocprofv3 SQLite result databases (gpurun_out/prof_*/runc/
*_results.db) into the committed profiles/ summaries.

Usage: python3 profiles/extract_stats.py <results.db> [--pmc]
"""
import sqlite3
import sys
from collections import defaultdict


def open_db(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    uuid = next(t.split("rocpd_metadata_")[1] for t in tabs
                if t.startswith("rocpd_metadata_"))
    return db, uuid


def kernel_stats(db, u):
    cur = db.cursor()
    rows = cur.execute(f"""
        SELECT s.display_name, d.start, d.end,
               d.grid_size_x * d.grid_size_y * d.grid_size_z,
               s.arch_vgpr_count, s.sgpr_count
        FROM rocpd_kernel_dispatch_{u} d
        JOIN rocpd_info_kernel_symbol_{u} s ON d.kernel_id = s.id
    """).fetchall()
    agg = defaultdict(lambda: [0, 0.0, float("inf"), 0.0, 0, 0, 0])
    for name, start, end, grid, vgpr, sgpr in rows:
        dur = (end - start) / 1e6  # ns -> ms
        a = agg[name]
        a[0] += 1
        a[1] += dur
        a[2] = min(a[2], dur)
        a[3] = max(a[3], dur)
        a[4] = grid
        a[5] = vgpr
        a[6] = sgpr
    total = sum(a[1] for a in agg.values())
    out = []
    out.append(f"{'kernel':<44} {'calls':>5} {'total_ms':>10} {'avg_ms':>9} "
               f"{'min_ms':>8} {'max_ms':>9} {'%':>5} {'grid':>10} {'vgpr':>4}")
    for name, a in sorted(agg.items(), key=lambda kv: -kv[1][1]):
        short = name.split("(")[0].replace("em::", "")[:44]
        out.append(f"{short:<44} {a[0]:>5} {a[1]:>10.3f} {a[1]/a[0]:>9.3f} "
                   f"{a[2]:>8.3f} {a[3]:>9.3f} {100*a[1]/total:>5.1f} "
                   f"{a[4]:>10} {a[5]:>4}")
    out.append(f"total kernel time: {total:.3f} ms over {len(rows)} dispatches")
    return "\n".join(out)


def pmc_stats(db, u):
    cur = db.cursor()
    try:
        rows = cur.execute(f"""
            SELECT s.display_name, pi.name, SUM(pe.value), COUNT(*)
            FROM rocpd_pmc_event_{u} pe
            JOIN rocpd_info_pmc_{u} pi ON pe.pmc_id = pi.id
            JOIN rocpd_event_{u} ev ON pe.event_id = ev.id
            JOIN rocpd_kernel_dispatch_{u} d ON d.event_id = ev.id
            JOIN rocpd_info_kernel_symbol_{u} s ON d.kernel_id = s.id
            GROUP BY s.display_name, pi.name
        """).fetchall()
    except sqlite3.OperationalError as e:
        return f"(pmc join failed: {e}; falling back to raw pmc table)\n" + \
            "\n".join(str(r) for r in cur.execute(
                f"SELECT * FROM rocpd_pmc_event_{u} LIMIT 20"))
    agg = defaultdict(dict)
    for kname, cname, val, cnt in rows:
        short = kname.split("(")[0].replace("em::", "")[:44]
        agg[short][cname] = (val, cnt)
    out = []
    for k in sorted(agg):
        out.append(k)
        for c, (v, n) in sorted(agg[k].items()):
            out.append(f"    {c:<28} sum={v:>18.0f}  (dispatches={n})")
    return "\n".join(out)


if __name__ == "__main__":
    path = sys.argv[1]
    db, u = open_db(path)
    print(f"== {path} ==")
    print(kernel_stats(db, u))
    if "--pmc" in sys.argv:
        print("\n-- PMC --")
        print(pmc_stats(db, u))
