"""Aggregate rocprofv3 rocpd SQLite output into per-kernel totals."""

from __future__ import annotations

import glob
import sqlite3
import sys


def main(pattern: str, out_path: str | None = None) -> None:
    paths = sorted(glob.glob(pattern))
    if not paths:
        print(f'no db matching {pattern}')
        return
    db = sqlite3.connect(paths[-1])
    # rocpd schema: kernel dispatches with start/end timestamps.
    tables = [
        r[0]
        for r in db.execute(
            "SELECT name FROM sqlite_master WHERE type='table'",
        )
    ]
    lines = []
    cand = [t for t in tables if 'kernel_dispatch' in t]
    if not cand:
        lines.append(f'tables: {tables}')
    else:
        t = cand[0]

        def tbl(prefix: str) -> str:
            match = [x for x in tables if x.startswith(prefix)]
            return match[0] if match else prefix

        ksym = tbl('rocpd_info_kernel_symbol')
        sstr = tbl('rocpd_string')
        try:
            kcols = [r[1] for r in db.execute(f'PRAGMA table_info({ksym})')]
            name_col = (
                'display_name' if 'display_name' in kcols else 'kernel_name'
            )
            q = f"""
            SELECT s.string AS name,
                   COUNT(*) AS calls,
                   SUM(k.end - k.start) / 1e6 AS total_ms,
                   AVG(k.end - k.start) / 1e3 AS avg_us
            FROM {t} k
            JOIN {ksym} ks ON k.kernel_id = ks.id
            JOIN {sstr} s ON ks.{name_col} = s.id
            GROUP BY s.string ORDER BY total_ms DESC LIMIT 30
            """
            for row in db.execute(q):
                name = row[0][:80]
                lines.append(
                    f'{row[2]:10.2f} ms  {row[1]:6d} calls  {row[3]:9.1f} us/call  {name}',
                )
        except Exception as e:
            lines.append(f'join failed: {e}')
            # dump schema of related tables
            for tt in tables:
                cols = [r[1] for r in db.execute(f'PRAGMA table_info({tt})')]
                lines.append(f'{tt}: {cols}')
    text = '\n'.join(lines)
    print(text)
    if out_path:
        with open(out_path, 'w') as f:
            f.write(text + '\n')


if __name__ == '__main__':
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
