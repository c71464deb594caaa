#!/usr/bin/env python3
"""Summarize a rocprofv3 SQLite results db into a small per-kernel CSV.

Usage: python scripts/pmc_summary.py <results.db> <out.csv>
Aggregates dispatch durations and (if present) PMC counter totals per kernel.
"""
import sqlite3
import sys


def main(db_path, out_path):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]

    def t(prefix):
        for name in tables:
            if name.startswith(prefix):
                return name
        return None

    disp = t("rocpd_kernel_dispatch")
    sym = t("rocpd_info_kernel_symbol")
    pmc_ev = t("rocpd_pmc_event")
    pmc_info = t("rocpd_info_pmc")

    lines = ["kernel,ndispatch,total_ms,avg_us,vgpr,lds_bytes"]
    q = f"""SELECT k.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
            AVG(d.end-d.start)/1e3, MAX(k.arch_vgpr_count), MAX(d.group_segment_size)
            FROM {disp} d JOIN {sym} k ON d.kernel_id=k.id
            GROUP BY k.display_name ORDER BY SUM(d.end-d.start) DESC LIMIT 30"""
    for name, n, tot, avg, vgpr, lds in cur.execute(q):
        lines.append(f'"{name}",{n},{tot:.3f},{avg:.1f},{vgpr},{lds}')

    if pmc_ev and pmc_info:
        try:
            cols_ev = [r[1] for r in cur.execute(f"PRAGMA table_info({pmc_ev})")]
            cols_info = [r[1] for r in cur.execute(f"PRAGMA table_info({pmc_info})")]
            lines.append("")
            lines.append(f"# pmc_event cols: {cols_ev}")
            lines.append(f"# pmc_info cols: {cols_info}")
            # typical schema: pmc_event(pmc_id, event_id, value);
            # dispatch rows have event_id
            q2 = f"""SELECT k.display_name, i.name, SUM(e.value)
                     FROM {pmc_ev} e
                     JOIN {pmc_info} i ON e.pmc_id = i.id
                     JOIN {disp} d ON e.event_id = d.event_id
                     JOIN {sym} k ON d.kernel_id = k.id
                     GROUP BY k.display_name, i.name"""
            lines.append("kernel,counter,total")
            for name, cname, val in cur.execute(q2):
                lines.append(f'"{name}",{cname},{val}')
        except Exception as e:  # noqa: BLE001
            lines.append(f"# pmc join failed: {e}")
            for tb in (pmc_ev, pmc_info):
                rows = list(cur.execute(f"SELECT * FROM {tb} LIMIT 3"))
                lines.append(f"# {tb} sample: {rows}")
    with open(out_path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print(f"wrote {out_path}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
