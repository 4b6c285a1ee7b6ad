#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database: per-kernel totals.

Usage: python scripts/rocpd_stats.py <results.db> [top_n]
Prints per-kernel call count / total / mean µs sorted by total time, plus
the whole-GPU busy time — the evidence file committed under profiles/.
"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    st = next(t for t in tables if t.startswith("rocpd_string"))

    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    sym_cols = [r[1] for r in cur.execute(f"PRAGMA table_info({ks})")]
    name_col = ("display_name" if "display_name" in sym_cols
                else "kernel_name" if "kernel_name" in sym_cols else None)
    if name_col:
        q = (f"SELECT s.{name_col}, COUNT(*), SUM(d.end-d.start), "
             f"AVG(d.end-d.start) FROM {kd} d JOIN {ks} s "
             f"ON d.kernel_id = s.id GROUP BY s.{name_col} "
             f"ORDER BY SUM(d.end-d.start) DESC")
        rows = list(cur.execute(q))
        if rows and isinstance(rows[0][0], int):
            # name is a string-table id
            smap = dict(cur.execute(f"SELECT id, string FROM {st}"))
            rows = [(smap.get(r[0], r[0]),) + r[1:] for r in rows]
    else:
        raise SystemExit(f"no kernel name column in {sym_cols}")

    total = sum(r[2] for r in rows)
    n_disp = sum(r[1] for r in rows)
    span = cur.execute(f"SELECT MAX(end)-MIN(start) FROM {kd}").fetchone()[0]
    print(f"{'kernel':<72} {'calls':>7} {'total_us':>12} {'mean_us':>9} "
          f"{'%':>6}")
    for name, calls, tot, mean in rows[:top]:
        nm = str(name)
        nm = nm if len(nm) <= 70 else nm[:67] + "..."
        print(f"{nm:<72} {calls:>7} {tot / 1e3:>12.1f} {mean / 1e3:>9.2f} "
              f"{100.0 * tot / total:>6.2f}")
    print(f"\ntotal kernel time: {total / 1e6:.3f} ms over {n_disp} "
          f"dispatches; wall span {span / 1e6:.3f} ms; "
          f"gpu busy {100.0 * total / span:.1f}%")


if __name__ == "__main__":
    main()
