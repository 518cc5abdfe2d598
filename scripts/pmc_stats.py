"""Aggregate rocprofv3 --pmc rocpd output: per-counter totals per kernel."""
from __future__ import annotations

import glob
import sqlite3
import sys


def main(pattern: str) -> None:
    db = sqlite3.connect(sorted(glob.glob(pattern))[-1])
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tbl(prefix):
        m = [t for t in tables if t.startswith('rocpd_' + prefix)]
        return m[0] if m else None

    pmc_event = tbl('pmc_event')
    info_pmc = tbl('info_pmc')
    kd = tbl('kernel_dispatch')
    ksym = tbl('info_kernel_symbol')
    for t in (info_pmc,):
        cols = [r[1] for r in db.execute('PRAGMA table_info(%s)' % t)]
        print(t, cols)
    # counter totals joined to kernel names via dispatch event
    q = """
    SELECT ks.display_name, p.name, SUM(e.value), COUNT(DISTINCT k.id)
    FROM {e} e
    JOIN {p} p ON e.pmc_id = p.id
    JOIN {k} k ON e.event_id = k.event_id
    JOIN {s} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name, p.name ORDER BY 1, 2
    """.format(e=pmc_event, p=info_pmc, k=kd, s=ksym)
    try:
        for name, counter, total, disp in db.execute(q):
            print('%-60s %-28s %14.0f  (%d dispatches)' % (name[:60], counter, total, disp))
    except Exception as ex:
        print('join failed:', ex)
        cols = [r[1] for r in db.execute('PRAGMA table_info(%s)' % pmc_event)]
        print(pmc_event, cols)


if __name__ == '__main__':
    main(sys.argv[1])
