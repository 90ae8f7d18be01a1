"""Summarize rocprofv3 --pmc output (rocpd .db): per-(kernel, counter)
mean values for the hand kernels.  Usage:
    python scripts/pmcsum.py <results.db> [name_filter]"""
import sys
import sqlite3


def main():
    con = sqlite3.connect(sys.argv[1])
    filt = sys.argv[2] if len(sys.argv) > 2 else ''
    cur = con.cursor()
    q = ("SELECT kernel_name, counter_name, COUNT(*), AVG(value), "
         "AVG(grid_size), AVG(duration) FROM counters_collection "
         "GROUP BY kernel_name, counter_name")
    rows = [r for r in cur.execute(q)]
    seen = {}
    for kname, cname, n, avg, grid, dur in rows:
        short = kname.split('<')[0].split('(')[0]
        if filt and filt not in short:
            continue
        seen.setdefault(short, []).append((cname, n, avg, grid, dur))
    for short, entries in sorted(seen.items()):
        n = entries[0][1]
        dur = entries[0][4] or 0
        print(f'{short}  (n={n}, avg_dur={dur / 1e3:.1f}us, '
              f'grid={entries[0][3]:.0f})')
        for cname, _, avg, _, _ in sorted(entries):
            print(f'    {cname:<26} avg={avg:.4g}')


if __name__ == '__main__':
    main()
