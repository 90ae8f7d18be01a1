"""Summarize rocprofv3 --pmc output (rocpd .db): per-kernel mean counter
values.  Usage: python scripts/pmcsum.py <results.db> [top]"""
import sys
import sqlite3


def main():
    con = sqlite3.connect(sys.argv[1])
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 12
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    ct = [t for t in tables if 'counter' in t.lower()]
    print('tables:', ', '.join(tables))
    if not ct:
        print('no counter tables found')
        return
    for t in ct:
        cols = [c[1] for c in cur.execute(f'PRAGMA table_info({t})')]
        print(f'\n== {t} cols: {cols}')
        n = cur.execute(f'SELECT COUNT(*) FROM {t}').fetchone()[0]
        print('rows:', n)
        # common rocpd shape: (dispatch) id, counter name/id, value
        name_col = next((c for c in cols if 'name' in c.lower()), None)
        val_col = next((c for c in cols
                        if c.lower() in ('value', 'counter_value')), None)
        kern_col = next((c for c in cols if 'kernel' in c.lower()), None)
        if name_col and val_col:
            q = (f'SELECT {name_col}, COUNT(*), AVG({val_col}), '
                 f'SUM({val_col}) FROM {t} GROUP BY {name_col}')
            for row in cur.execute(q):
                print('  %-28s n=%-8d avg=%.3e sum=%.3e' % row)
        elif val_col and kern_col:
            for row in cur.execute(
                    f'SELECT {kern_col}, COUNT(*), AVG({val_col}) FROM {t} '
                    f'GROUP BY {kern_col} LIMIT {top}'):
                print(' ', row)
    # try the joined per-kernel view if present
    for v in tables:
        if 'counters_collection' in v or 'counter_value' in v:
            cols = [c[1] for c in cur.execute(f'PRAGMA table_info({v})')]
            print(f'\n== joined {v}: {cols}')


if __name__ == '__main__':
    main()
