"""Summarize a rocprofv3 kernel_stats.csv (top-N kernels by total time)."""
import csv
import sys


def main(path, topn=8):
    rows = sorted(csv.DictReader(open(path)),
                  key=lambda r: -float(r.get("TotalDurationNs", 0)))
    tot = sum(float(r.get("TotalDurationNs", 0)) for r in rows)
    for r in rows[:topn]:
        t = float(r.get("TotalDurationNs", 0))
        name = r["Name"].split("(")[0][:42]
        n = r.get("Calls", "?")
        avg = float(r.get("AverageNs", 0)) / 1e3
        print(f"  {name:44s} n={n:>6s} tot={t/1e6:8.1f}ms "
              f"avg={avg:7.1f}us share={100*t/max(tot,1):5.1f}%")


if __name__ == "__main__":
    main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 8)
