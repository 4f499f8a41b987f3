"""Summarize a rocprofv3 --pmc rocpd database: per-kernel counter totals and
derived duty ratios (ACTIVE / WAIT vs wave-cycles). Handles the GUID-suffixed
rocpd table names. Usage:

    python scripts/pmc_summary.py <results.db> [kernel-name-filter]
"""

from __future__ import annotations

import sqlite3
import sys


def main() -> None:
    path = sys.argv[1]
    filt = sys.argv[2] if len(sys.argv) > 2 else ""
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [n for (n,) in cur.execute(
        "select name from sqlite_master where type='table'")]

    def tbl(prefix: str) -> str:
        return next(t for t in tables if t.startswith(prefix))

    ev = tbl("rocpd_pmc_event")
    dis = tbl("rocpd_kernel_dispatch")
    sym = tbl("rocpd_info_kernel_symbol")
    pmc = tbl("rocpd_info_pmc")
    rows = cur.execute(
        f"""
        select k.display_name, p.name, sum(e.value)
        from {ev} e
        join {pmc} p on e.pmc_id = p.id
        join {dis} d on e.event_id = d.event_id
        join {sym} k on d.kernel_id = k.id
        group by k.display_name, p.name
        """
    ).fetchall()
    agg: dict[str, dict[str, float]] = {}
    for kname, cname, total in rows:
        if filt and filt not in kname:
            continue
        agg.setdefault(kname, {})[cname] = total
    for kname, c in agg.items():
        print(f"== {kname[:100]}")
        for cname, total in sorted(c.items()):
            print(f"   {cname:<24} {total:,.0f}")
        wc = c.get("SQ_WAVE_CYCLES")
        if wc:
            for k in ("SQ_ACTIVE_INST_ANY", "SQ_WAIT_ANY"):
                if k in c:
                    print(f"   {k}/WAVE_CYCLES = {c[k] / wc:.3f}")
            for k in ("SQ_INSTS_MFMA", "SQ_INSTS_VALU", "SQ_INSTS_LDS"):
                if k in c:
                    print(f"   {k}/WAVE_CYCLES = {c[k] / wc:.4f}")


if __name__ == "__main__":
    main()
