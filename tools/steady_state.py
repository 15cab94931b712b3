"""Aggregate a rocprofv3 kernel trace over its steady-state tail.

    python tools/steady_state.py <trace-dir> <window_ms> [N]

Finds *kernel_trace.csv under <trace-dir>, keeps dispatches whose end
falls in the last <window_ms> of the trace, and prints a per-kernel
total table (the r01 steady-state methodology: warmup / MIOpen-Find
excluded by the window).
"""

import csv
import glob
import sys
from collections import defaultdict


def main():
    d = sys.argv[1]
    window_ms = float(sys.argv[2]) if len(sys.argv) > 2 else 180.0
    n = int(sys.argv[3]) if len(sys.argv) > 3 else 40
    files = sorted(glob.glob(f"{d}/**/*kernel_trace.csv", recursive=True))
    if not files:
        # tolerate other rocprofv3 naming: any csv with kernel columns
        for c in sorted(glob.glob(f"{d}/**/*.csv", recursive=True)):
            with open(c) as f:
                head = f.readline()
            if "Kernel_Name" in head and ("Start_Timestamp" in head
                                          or "BeginNs" in head):
                files.append(c)
    assert files, f"no kernel trace under {d}: " + str(
        glob.glob(f"{d}/**/*", recursive=True)[:20])
    rows = []
    with open(files[-1]) as f:
        rd = csv.DictReader(f)
        cols = rd.fieldnames
        kname = ("Kernel_Name" if "Kernel_Name" in cols else "Name")
        t0c = "Start_Timestamp" if "Start_Timestamp" in cols else "BeginNs"
        t1c = "End_Timestamp" if "End_Timestamp" in cols else "EndNs"
        for r in rd:
            rows.append((r[kname], int(r[t0c]), int(r[t1c])))
    t_end = max(r[2] for r in rows)
    t_lo = t_end - window_ms * 1e6
    agg = defaultdict(lambda: [0.0, 0])
    busy = 0.0
    for name, t0, t1 in rows:
        if t1 < t_lo:
            continue
        a = agg[name]
        a[0] += (t1 - t0) / 1e6
        a[1] += 1
        busy += (t1 - t0) / 1e6
    print(f"window {window_ms:.0f} ms, GPU busy {busy:.1f} ms "
          f"({100 * busy / window_ms:.0f}%), {len(agg)} distinct kernels")
    items = sorted(agg.items(), key=lambda kv: -kv[1][0])
    for name, (ms, calls) in items[:n]:
        print(f"{ms:8.2f} ms {calls:5d}x  {name[:100]}")


if __name__ == "__main__":
    main()
