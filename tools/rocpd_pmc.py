#!/usr/bin/env python3
"""Digest a rocprofv3 --pmc rocpd sqlite DB: per (kernel, grid) mean counter
values per dispatch (PMC rows are per-SE instances; we SUM instances within a
dispatch, then average over dispatches).  Also derives an MFMA-utilization
ratio when the needed SQ counters are present.

Usage: python tools/rocpd_pmc.py <results.db>
"""
import sqlite3
import sys
from collections import defaultdict


def main():
    db = sqlite3.connect(sys.argv[1])
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def t(prefix):
        return next(x for x in tabs if x.startswith(prefix))

    pe, pi = t("rocpd_pmc_event_"), t("rocpd_info_pmc_")
    kd, ks = t("rocpd_kernel_dispatch_"), t("rocpd_info_kernel_symbol_")
    names = dict(cur.execute(f"SELECT id, name FROM {pi}"))
    # per-dispatch counter sums
    rows = cur.execute(
        f"""
        SELECT d.event_id, s.display_name, d.grid_size_x, e.pmc_id,
               SUM(e.value)
        FROM {pe} e
        JOIN {kd} d ON e.event_id = d.event_id
        JOIN {ks} s ON d.kernel_id = s.id
        GROUP BY d.event_id, e.pmc_id
        """
    ).fetchall()
    agg = defaultdict(lambda: defaultdict(list))  # (kernel,grid) -> ctr -> []
    for _eid, kname, gx, pmc_id, val in rows:
        kname = kname.replace("(anonymous namespace)::", "").split("(")[0]
        agg[(kname, gx)][names[pmc_id]].append(val)
    for (kname, gx), ctrs in sorted(agg.items()):
        n = max(len(v) for v in ctrs.values())
        print(f"\n{kname}  grid_x={gx}  dispatches={n}")
        means = {c: sum(v) / len(v) for c, v in ctrs.items()}
        for c in sorted(means):
            print(f"  {c:<28} {means[c]:16,.0f}")
        wave = means.get("SQ_WAVE_CYCLES")
        mfma = means.get("SQ_VALU_MFMA_BUSY_CYCLES")
        if wave and mfma:
            # WAVE_CYCLES counts quad-cycles, MFMA_BUSY plain cycles
            print(f"  -> MFMA busy / wave cycles   {mfma / (4 * wave):.1%}"
                  f"  (per-wave-slot; xcycle units normalized)")
        if wave and means.get("SQ_WAIT_ANY") is not None:
            print(f"  -> wait(parked)/wave         "
                  f"{means['SQ_WAIT_ANY'] / wave:.1%}   "
                  f"issue-stall/wave {means.get('SQ_WAIT_INST_ANY', 0) / wave:.1%}   "
                  f"active-issue/wave {means.get('SQ_ACTIVE_INST_ANY', 0) / wave:.1%}")
        if means.get("SQ_INSTS_LDS") and means.get("SQ_LDS_BANK_CONFLICT") is not None:
            print(f"  -> LDS conflict-cycles/inst  "
                  f"{means['SQ_LDS_BANK_CONFLICT'] / means['SQ_INSTS_LDS']:.2f}")


if __name__ == "__main__":
    main()
