#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite database into a per-kernel stats table
(like `rocprofv3 --stats`, but robust to the profiled process dying before the
CSV pass).  Usage: python tools/rocpd_stats.py <results.db> [top_n]"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    top_n = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch_"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol_"))
    rows = cur.execute(
        f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start), AVG(d.end - d.start)
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
        """
    ).fetchall()
    total = sum(r[2] for r in rows) or 1
    print(f"{'%':>6} {'total_ms':>10} {'calls':>8} {'avg_us':>9}  kernel")
    for name, calls, tot, avg in rows[:top_n]:
        label = name if len(name) < 120 else name[:117] + "..."
        print(
            f"{100.0 * tot / total:6.2f} {tot / 1e6:10.3f} {calls:8d} "
            f"{avg / 1e3:9.2f}  {label}"
        )
    print(f"total GPU kernel time: {total / 1e6:.3f} ms over {sum(r[1] for r in rows)} dispatches")
    # wall-clock span of all dispatches
    lo, hi = cur.execute(f"SELECT MIN(start), MAX(end) FROM {disp}").fetchone()
    if lo is not None:
        print(f"dispatch span: {(hi - lo) / 1e6:.3f} ms  (GPU busy {100.0 * total / (hi - lo):.1f}%)")


if __name__ == "__main__":
    main()
