#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel_stats.csv: top kernels by total time."""
import csv
import sys

path = sys.argv[1]
top = int(sys.argv[2]) if len(sys.argv) > 2 else 20
rows = list(csv.DictReader(open(path)))
rows.sort(key=lambda r: -float(r['TotalDurationNs']))
tot = sum(float(r['TotalDurationNs']) for r in rows)
calls = sum(int(r['Calls']) for r in rows)
print('total kernel time: %.2fs over %d launches' % (tot / 1e9, calls))
for r in rows[:top]:
    print('%9.1fms %7s x %7.1fus  %s' % (
        float(r['TotalDurationNs']) / 1e6, r['Calls'],
        float(r['AverageNs']) / 1e3, r['Name'][:100]))
