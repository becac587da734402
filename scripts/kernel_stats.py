#!/usr/bin/env python3
"""Print the top kernels by total device time from a rocprofv3 stats run.

Usage: python scripts/kernel_stats.py <dir-or-csv> [N]
Accepts either a *kernel_stats.csv file or a directory to search
recursively (rocprofv3 -d output tree)."""

import csv
import glob
import sys


def main() -> None:
    target = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out"
    topn = int(sys.argv[2]) if len(sys.argv) > 2 else 12
    if target.endswith(".csv"):
        files = [target]
    else:
        files = glob.glob(f"{target}/**/*kernel_stats.csv", recursive=True)
    if not files:
        print(f"no kernel_stats.csv under {target}")
        return
    rows = list(csv.DictReader(open(files[0])))

    def dur(r):
        for k in ("TotalDurationNs", "TOTAL_DURATION_NS", "DurationNs"):
            if k in r:
                return float(r[k])
        return 0.0

    rows.sort(key=lambda r: -dur(r))
    total = sum(dur(r) for r in rows)
    print(f"{files[0]}: {total/1e6:.1f} ms total device time")
    for r in rows[:topn]:
        name = (r.get("Name") or r.get("KERNEL_NAME") or "")[:90]
        calls = r.get("Calls") or r.get("CALLS") or ""
        print(f"{dur(r)/1e6:9.1f} ms {100*dur(r)/max(total,1):5.1f}% "
              f"x{calls:>5} {name}")


if __name__ == "__main__":
    main()
