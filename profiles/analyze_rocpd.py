#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database (kernel-trace --stats output)
into the committed per-kernel text summaries under profiles/.

Usage: python profiles/analyze_rocpd.py gpurun_out/prof/xxx_results.db > profiles/rNN_xxx.txt
"""
import sqlite3
import sys


def main(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    print(f"# rocprofv3 kernel summary of {path}")
    print(f"# {'total_ms':>10} {'calls':>7} {'avg_us':>10} {'pct':>6}  name")
    rows = list(cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
               AVG(d.end-d.start)/1e3
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY 3 DESC"""))
    total = sum(r[2] for r in rows)
    for name, cnt, ms, avg in rows:
        print(f"{ms:12.3f} {cnt:7d} {avg:10.1f} {100*ms/total:5.1f}%  {name[:90]}")
    span = cur.execute(
        f"SELECT SUM(end-start)/1e6, (MAX(end)-MIN(start))/1e6 FROM {disp}"
    ).fetchone()
    print(f"\n# kernel-busy {span[0]:.1f} ms over wall span {span[1]:.1f} ms "
          f"(includes untimed generation + warmup + all steps)")

    # PMC counters (from --pmc passes): aggregate per kernel per counter
    pmc = next((t for t in tables if t.startswith("rocpd_pmc_event")), None)
    info = next((t for t in tables if t.startswith("rocpd_info_pmc")), None)
    if pmc and cur.execute(f"SELECT COUNT(*) FROM {pmc}").fetchone()[0]:
        print("\n# PMC counters per kernel: counter sum over dispatches "
              "(and per-dispatch average)")
        rows = list(cur.execute(f"""
            SELECT s.display_name, i.name, SUM(p.value), COUNT(*)
            FROM {pmc} p
            JOIN {disp} d ON p.event_id = d.event_id
            JOIN {sym} s ON d.kernel_id = s.id
            JOIN {info} i ON p.pmc_id = i.id
            GROUP BY s.display_name, i.name ORDER BY 3 DESC"""))
        for name, counter, total, cnt in rows[:60]:
            print(f"{counter:>22} {total:18.0f} over {cnt:5d} dispatches "
                  f"(avg {total/cnt:14.1f})  {name[:60]}")


if __name__ == "__main__":
    main(sys.argv[1])
