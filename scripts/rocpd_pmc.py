#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc rocpd database: per-kernel counter totals.

Usage: python scripts/rocpd_pmc.py <results.db> [top_n]
Joins rocpd_pmc_event (event_id, pmc_id, value) -> kernel dispatches via
dispatch.event_id, prints per-kernel duration + counter sums and derived
MFMA utilization (SQ_VALU_MFMA_BUSY_CYCLES / GRBM_GUI_ACTIVE; GRBM counts
one unit, SQ counters aggregate all CUs — MfmaUtil = mfma_cycles /
(grbm_active * 256 CUs), capped for the gfx94x-formula caveat).
"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    pe = next(t for t in tabs if t.startswith("rocpd_pmc_event"))
    pi = next(t for t in tabs if t.startswith("rocpd_info_pmc"))

    q = f"""
      SELECT s.display_name, p.name, SUM(v.value)
      FROM {pe} v
      JOIN {kd} d ON v.event_id = d.event_id
      JOIN {ks} s ON d.kernel_id = s.id
      JOIN {pi} p ON v.pmc_id = p.id
      GROUP BY s.display_name, p.name
    """
    agg = {}
    for kname, counter, val in cur.execute(q):
        k = kname.split("(")[0][:52]
        agg.setdefault(k, {})[counter] = val
    dq = f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start)
             FROM {kd} d JOIN {ks} s ON d.kernel_id=s.id
             GROUP BY s.display_name"""
    dur = {}
    for name, calls, tot in cur.execute(dq):
        dur[name.split("(")[0][:52]] = (calls, tot)
    rows = sorted(agg.items(), key=lambda kv: -dur.get(kv[0], (0, 0))[1])
    print(f"{'kernel':<52} {'calls':>6} {'tot_us':>9} "
          f"{'MfmaUtil%':>9} {'waves_cyc':>10} {'grbm_act':>10}")
    for k, c in rows[:top]:
        calls, tot = dur.get(k, (0, 0))
        mfma = c.get("SQ_VALU_MFMA_BUSY_CYCLES")
        grbm = c.get("GRBM_GUI_ACTIVE")
        util = (100.0 * mfma / (grbm * 256)
                if mfma is not None and grbm else None)
        print(f"{k:<52} {calls:>6} {tot/1e3:>9.1f} "
              f"{(f'{util:8.2f}' if util is not None else '      --'):>9} "
              f"{c.get('SQ_WAVE_CYCLES', 0):>10.3g} {grbm or 0:>10.3g}")


if __name__ == "__main__":
    main()
