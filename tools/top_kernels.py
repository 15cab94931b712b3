"""Print top kernels from a rocprofv3 kernel_stats.csv (or kernel trace).

    python tools/top_kernels.py <dir-or-csv> [N]
"""

import csv
import glob
import sys


def main():
    path = sys.argv[1]
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    files = ([path] if path.endswith(".csv")
             else sorted(glob.glob(f"{path}/**/*kernel_stats.csv",
                                   recursive=True)))
    if not files:
        files = sorted(glob.glob(f"{path}/**/*stats*.csv", recursive=True))
    print(files)
    rows = list(csv.DictReader(open(files[-1])))
    key = ("TotalDurationNs" if "TotalDurationNs" in rows[0]
           else "DurationNs")
    calls_key = "Calls" if "Calls" in rows[0] else "calls"
    rows.sort(key=lambda r: -float(r[key]))
    tot = sum(float(r[key]) for r in rows)
    print(f"total kernel time: {tot / 1e6:.2f} ms over {len(rows)} kernels")
    for r in rows[:n]:
        name = r.get("Name") or r.get("Kernel_Name") or "?"
        print(f"{float(r[key]) / 1e6:9.2f} ms {int(float(r[calls_key])):5d}x"
              f"  {name[:100]}")


if __name__ == "__main__":
    main()
