#!/usr/bin/env python3
"""Summarize rocprofv3 kernel stats: top kernels by total time.

Accepts either a *kernel_stats.csv (older rocprofv3 output) or the
prof_results.db SQLite (rocpd) written by the ROCm 7.2 rocprofv3."""
import csv
import sys


def from_csv(path, top):
    rows = list(csv.DictReader(open(path)))
    rows.sort(key=lambda r: -float(r['TotalDurationNs']))
    tot = sum(float(r['TotalDurationNs']) for r in rows)
    calls = sum(int(r['Calls']) for r in rows)
    out = [(float(r['TotalDurationNs']), int(r['Calls']), r['Name'])
           for r in rows[:top]]
    return tot, calls, out


def from_db(path, top):
    import sqlite3
    con = sqlite3.connect(path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    # rocpd schema: kernel dispatches reference a kernel-info row which
    # references a string row.  Discover column names defensively.
    kd = next(t for t in tables if 'kernel_dispatch' in t)
    cols = [c[1] for c in cur.execute(f'PRAGMA table_info({kd})')]
    tcol = 'end' if 'end' in cols else 'end_timestamp'
    scol = 'start' if 'start' in cols else 'start_timestamp'
    # find a joinable kernel name: try common layouts
    q = None
    for ki in [t for t in tables if 'kernel' in t and 'dispatch' not in t]:
        kcols = [c[1] for c in cur.execute(f'PRAGMA table_info({ki})')]
        if 'display_name' in kcols or 'kernel_name' in kcols or 'name' in kcols:
            namecol = ('display_name' if 'display_name' in kcols else
                       'kernel_name' if 'kernel_name' in kcols else 'name')
            for join in ('kernel_id', 'id'):
                if join in kcols and 'kernel_id' in cols:
                    q = (f'SELECT k.{namecol}, COUNT(*), '
                         f'SUM(d.{tcol}-d.{scol}) FROM {kd} d '
                         f'JOIN {ki} k ON d.kernel_id = k.{join} '
                         f'GROUP BY k.{namecol}')
                    break
        if q:
            break
    if q is None:
        raise RuntimeError(f'unknown rocpd schema; tables={tables}')
    rows = cur.execute(q).fetchall()
    # name may be a string id -> resolve through rocpd_string
    if rows and isinstance(rows[0][0], int) and any('string' in t for t in tables):
        st = next(t for t in tables if 'string' in t)
        scols = [c[1] for c in cur.execute(f'PRAGMA table_info({st})')]
        sval = 'string' if 'string' in scols else 'value'
        sid = 'id'
        smap = dict(cur.execute(f'SELECT {sid}, {sval} FROM {st}'))
        rows = [(smap.get(n, str(n)), c, t) for n, c, t in rows]
    rows.sort(key=lambda r: -(r[2] or 0))
    tot = sum(r[2] or 0 for r in rows)
    calls = sum(r[1] for r in rows)
    out = [(r[2] or 0, r[1], str(r[0])) for r in rows[:top]]
    return tot, calls, out


def main():
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    if path.endswith('.db'):
        tot, calls, rows = from_db(path, top)
    else:
        tot, calls, rows = from_csv(path, top)
    print('total kernel time: %.2fs over %d launches' % (tot / 1e9, calls))
    for t, c, name in rows:
        print('%9.1fms %7d x %7.1fus  %s' % (t / 1e6, c, t / c / 1e3,
                                             name[:110]))


if __name__ == '__main__':
    main()
