"""Aggregate rocprofv3 rocpd SQLite output into per-kernel totals."""

from __future__ import annotations

import glob
import sqlite3
import sys


def main(pattern: str, out_path: str | None = None) -> None:
    paths = sorted(glob.glob(pattern))
    lines: list[str] = []
    if not paths:
        lines.append(f'no db matching {pattern}')
        _emit(lines, out_path)
        return
    db = sqlite3.connect(paths[-1])
    tables = [
        r[0]
        for r in db.execute(
            "SELECT name FROM sqlite_master WHERE type='table'",
        )
    ]

    def tbl(prefix: str) -> str | None:
        match = [x for x in tables if x.startswith(prefix)]
        return match[0] if match else None

    kd = tbl('rocpd_kernel_dispatch')
    ksym = tbl('rocpd_info_kernel_symbol')
    sstr = tbl('rocpd_string')
    if kd is None:
        lines.append(f'tables: {tables}')
        _emit(lines, out_path)
        return
    n = db.execute(f'SELECT COUNT(*) FROM {kd}').fetchone()[0]
    lines.append(f'dispatches: {n}')
    if ksym is not None:
        kcols = [r[1] for r in db.execute(f'PRAGMA table_info({ksym})')]
        lines.append(f'{ksym} cols: {kcols}')
        sample = db.execute(f'SELECT * FROM {ksym} LIMIT 2').fetchall()
        lines.append(f'sample: {str(sample)[:300]}')
    done = False
    if ksym is not None:
        try:
            q = f"""
            SELECT ks.display_name, COUNT(*), SUM(k.end - k.start)/1e6,
                   AVG(k.end - k.start)/1e3
            FROM {kd} k
            JOIN {ksym} ks ON k.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30
            """
            rows = db.execute(q).fetchall()
            if rows:
                for r in rows:
                    lines.append(
                        f'{r[2]:10.2f} ms {r[1]:6d} calls '
                        f'{r[3]:9.1f} us/call  {str(r[0])[:100]}',
                    )
                done = True
        except Exception as e:
            lines.append(f'join failed: {e}')
    if not done:
        # fall back: group by kernel_id only
        q = f"""
        SELECT k.kernel_id, COUNT(*), SUM(k.end - k.start)/1e6
        FROM {kd} k GROUP BY k.kernel_id ORDER BY 3 DESC LIMIT 20
        """
        for r in db.execute(q):
            lines.append(f'{r[2]:10.2f} ms {r[1]:6d} calls  kernel_id={r[0]}')
    _emit(lines, out_path)


def _emit(lines: list[str], out_path: str | None) -> None:
    text = '\n'.join(lines)
    print(text)
    if out_path:
        with open(out_path, 'w') as f:
            f.write(text + '\n')


if __name__ == '__main__':
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
