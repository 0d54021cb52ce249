#!/usr/bin/env python3
"""Aggregate rocprofv3 .db output (rocpd schema): per (kernel, grid) counter
sums. Usage: python scripts/pmc_db_extract.py <results.db>

(The CSV flow is scripts/pmc_extract.py; rocprofv3 writes .db by default
when --output-format csv is not given.)"""
import sqlite3
import sys
from collections import defaultdict


def main():
    db = sys.argv[1]
    con = sqlite3.connect(db)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "select name from sqlite_master where type='table'")]
    pmc_tab = next(t for t in tabs if t.startswith("rocpd_pmc_event_"))
    sfx = pmc_tab[len("rocpd_pmc_event_"):]
    q = f"""
    select ks.display_name, kd.grid_size_x, p.name, sum(pe.value), count(*)
    from rocpd_pmc_event_{sfx} pe
    join rocpd_kernel_dispatch_{sfx} kd on kd.event_id = pe.event_id
    join rocpd_info_kernel_symbol_{sfx} ks on ks.id = kd.kernel_id
    join rocpd_info_pmc_{sfx} p on p.id = pe.pmc_id
    group by ks.display_name, kd.grid_size_x, p.name
    """
    agg = defaultdict(dict)
    for name, grid, ctr, tot, n in cur.execute(q):
        agg[(name.split("(")[0], grid)][ctr] = (tot, n)
    rows = sorted(agg.items(),
                  key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", (0, 0))[0])
    for (name, grid), ctrs in rows:
        for ctr, (tot, n) in sorted(ctrs.items()):
            print(f"{name:46.46s} grid={grid:>12d} {ctr:26s} n={n:5d} "
                  f"sum={tot:.6e} mean={tot / n:.6e}")


if __name__ == "__main__":
    main()
