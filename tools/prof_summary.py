"""Summarizes a rocprofv3 kernel_stats.csv (sorted by total time)."""
import csv
import sys

path = sys.argv[1] if len(sys.argv) > 1 else \
    '/root/repo/gpurun_out/prof/conf_kernel_stats.csv'
rows = list(csv.DictReader(open(path)))
tot = sum(float(r['TotalDurationNs']) for r in rows)
print('total GPU ms (all steps):', round(tot / 1e6, 1))
for r in rows[:24]:
  name = r['Name'][:78]
  ms = float(r['TotalDurationNs']) / 1e6
  print('%8.2f ms %6sx %5s%% %s' % (ms, r['Calls'], r['Percentage'][:5],
                                    name))
