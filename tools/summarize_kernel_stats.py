"""Print top kernels from a rocprofv3 *kernel_stats.csv: python tools/summarize_kernel_stats.py <csv> [n]"""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
n = int(sys.argv[2]) if len(sys.argv) > 2 else 14


def dur(r):
    for k in ("TotalDurationNs", "DurationNs", "TOTAL_DURATION_NS"):
        if k in r:
            return float(r[k])
    return 0.0


rows.sort(key=lambda r: -dur(r))
total = sum(dur(r) for r in rows)
print(f"total kernel time: {total/1e6:.1f} ms")
for r in rows[:n]:
    name = (r.get("Name") or r.get("KernelName") or "?")[:110]
    calls = r.get("Calls") or r.get("TotalCalls") or "?"
    print(f"{dur(r)/1e6:9.1f}ms {str(calls):>7} {100*dur(r)/max(total,1):5.1f}% {name}")
