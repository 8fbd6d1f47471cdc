"""Print the top kernels from a rocprofv3 kernel-stats CSV directory.

Usage: python tools/kstats.py <dir> [n]
Finds *kernel_stats*.csv (or falls back to the kernel trace) under <dir>.
"""

import csv
import glob
import sys
from collections import defaultdict


def main() -> None:
    d = sys.argv[1]
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    files = glob.glob(f"{d}/**/*kernel_stats*.csv", recursive=True)
    if files:
        rows = []
        for f in files:
            rows.extend(csv.DictReader(open(f)))
        key_dur = next(k for k in rows[0] if "Duration" in k or "DURATION" in k)
        key_name = next(k for k in rows[0] if k.lower() == "name")
        key_calls = next((k for k in rows[0] if "Call" in k or "CALLS" in k), None)
        rows.sort(key=lambda r: -float(r[key_dur]))
        for r in rows[:n]:
            calls = r.get(key_calls, "?") if key_calls else "?"
            print(f"{float(r[key_dur]) / 1e6:10.3f} ms  n={calls:>6}  {r[key_name][:80]}")
        return
    # fallback: aggregate the kernel trace
    files = glob.glob(f"{d}/**/*kernel_trace*.csv", recursive=True)
    agg = defaultdict(lambda: [0.0, 0])
    for f in files:
        for r in csv.DictReader(open(f)):
            name = r.get("Kernel_Name") or r.get("Name")
            dur = float(r.get("End_Timestamp", 0)) - float(r.get("Start_Timestamp", 0))
            agg[name][0] += dur
            agg[name][1] += 1
    for name, (dur, calls) in sorted(agg.items(), key=lambda kv: -kv[1][0])[:n]:
        print(f"{dur / 1e6:10.3f} ms  n={calls:>6}  {name[:80]}")


if __name__ == "__main__":
    main()
