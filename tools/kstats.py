# Convert rocprofv3 --kernel-trace --stats CSV output into the compact
# per-kernel JSON committed under profiles/ (name/calls/total_ns/avg_ns,
# sorted by total time).
#   python tools/kstats.py <rocprof_out_dir> <out_json> [top_n]
from __future__ import annotations

import csv
import json
import pathlib
import sys


def main():
    d = pathlib.Path(sys.argv[1])
    out = sys.argv[2]
    top = int(sys.argv[3]) if len(sys.argv) > 3 else 25
    rows = []
    for f in sorted(d.rglob("*kernel_stats.csv")):
        with open(f) as fh:
            for r in csv.DictReader(fh):
                name = r.get("Name") or r.get("NAME") or r.get("Kernel_Name")
                calls = r.get("Calls") or r.get("CALLS")
                tot = (r.get("TotalDurationNs") or r.get("DurationNs")
                       or r.get("TOTAL_DURATION_NS"))
                avg = r.get("AverageNs") or r.get("AVERAGE_NS")
                if not name or not calls:
                    continue
                total_ns = float(tot) if tot else 0.0
                rows.append({
                    "name": name.split("(")[0][:80],
                    "calls": int(calls),
                    "total_ns": total_ns,
                    "avg_ns": float(avg) if avg else
                    (total_ns / max(1, int(calls))),
                })
    rows.sort(key=lambda r: -r["total_ns"])
    pathlib.Path(out).write_text(json.dumps(rows[:top], indent=1))
    print(f"{len(rows)} kernels -> {out} (top {top})")


if __name__ == "__main__":
    main()
