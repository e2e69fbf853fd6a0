"""Print a compact summary of a rocprofv3 kernel_stats csv."""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
tot = sum(float(r['TotalDurationNs']) for r in rows)
print(f'total GPU time: {tot/1e9:.2f}s over {sum(int(r["Calls"]) for r in rows)} calls')
for r in rows[:20]:
    name = r['Name'][:72]
    print('%5.2f%% %9.1fms %6dx  %s' % (float(r['Percentage']),
                                        float(r['TotalDurationNs']) / 1e6,
                                        int(r['Calls']), name))
