#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results .db: per-kernel counter sums and
per-dispatch averages (rocpd schema: pmc_event(event_id,pmc_id,value) ->
kernel_dispatch.event_id / info_pmc.id).

Usage: python tools/pmc_summary.py <results.db> [kernel-substr ...]
"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    pats = sys.argv[2:] or ["rank_grp", "copy_chunks", "copy_rows", "gather_meta"]
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    pe = next(t for t in tabs if t.startswith("rocpd_pmc_event"))
    pi = next(t for t in tabs if t.startswith("rocpd_info_pmc"))
    where = " OR ".join(f"ks.display_name LIKE '%{p}%'" for p in pats)
    q = f"""
    SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(DISTINCT kd.id)
    FROM {pe} pe
    JOIN {kd} kd ON pe.event_id = kd.event_id
    JOIN {ks} ks ON kd.kernel_id = ks.id
    JOIN {pi} pi ON pe.pmc_id = pi.id
    WHERE {where}
    GROUP BY 1, 2 ORDER BY 1, 2
    """
    rows = list(cur.execute(q))
    per = {}
    for name, ctr, tot, ndisp in rows:
        per.setdefault(name, {})[ctr] = (tot, ndisp)
    for name, ctrs in per.items():
        print(name[:60])
        wave = ctrs.get("SQ_WAVE_CYCLES", (0, 1))[0]
        for ctr, (tot, ndisp) in sorted(ctrs.items()):
            pct = f"  {100.0 * tot / wave:5.1f}% of WAVE_CYCLES" if (
                wave and ctr != "SQ_WAVE_CYCLES" and ctr.startswith("SQ_")) else ""
            print(f"    {ctr:<22} {tot:.4g} total  {tot / ndisp:.4g}/dispatch"
                  f" ({ndisp} disp){pct}")


if __name__ == "__main__":
    main()
