# Parse rocprofv3 --pmc counter_collection CSVs from two separate passes
# (FETCH_SIZE and WRITE_SIZE cannot share a pass: TCC slot budget,
# MI355X_MICROARCH.md "rocprofv3 PMC slots") into the per-launch HBM
# traffic of the dominant GEMM kernel, applying the gfx950 FETCH x2
# correction for wide coalesced reads (guide §HBM).
#
#   python tools/pmc_parse.py <fetch_dir> <write_dir> <out_json> \
#       [--kernel-substr SUBSTR] [--algorithmic-bytes N]
from __future__ import annotations

import argparse
import csv
import json
import pathlib
from collections import defaultdict


def load_counter(d: pathlib.Path, counter: str,
                 kernel_substr: str | None):
    """-> {kernel_name: (total_kb, n_dispatches)} for `counter`."""
    acc: dict[str, list[float]] = defaultdict(list)
    for f in sorted(d.rglob("*counter_collection.csv")):
        with open(f) as fh:
            for row in csv.DictReader(fh):
                if row.get("Counter_Name") != counter:
                    continue
                name = row.get("Kernel_Name", "")
                if kernel_substr and kernel_substr not in name:
                    continue
                acc[name].append(float(row["Counter_Value"]))
    return {k: (sum(v), len(v)) for k, v in acc.items()}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("fetch_dir")
    ap.add_argument("write_dir")
    ap.add_argument("out_json")
    ap.add_argument("--kernel-substr", default=None)
    ap.add_argument("--algorithmic-bytes", type=float, default=None)
    args = ap.parse_args()

    fetch = load_counter(pathlib.Path(args.fetch_dir), "FETCH_SIZE",
                         args.kernel_substr)
    write = load_counter(pathlib.Path(args.write_dir), "WRITE_SIZE",
                         args.kernel_substr)
    if not fetch:
        raise SystemExit("no FETCH_SIZE rows matched")
    # dominant kernel = most dispatches in the fetch pass
    kname = max(fetch, key=lambda k: fetch[k][1])
    fkb, fn = fetch[kname]
    wkb, wn = write.get(kname, (0.0, 1))
    read_b = fkb / fn * 1024 * 2  # gfx950: FETCH_SIZE reports 1/2 of wide reads
    write_b = wkb / max(wn, 1) * 1024
    out = {
        "kernel": kname[:200],
        "fetch_size_kb_avg": fkb / fn,
        "write_size_kb_avg": wkb / max(wn, 1),
        "dispatches_fetch_pass": fn,
        "fetch_correction": "x2 (gfx950, MI355X_MICROARCH.md §HBM)",
        "hbm_read_bytes_per_launch": read_b,
        "hbm_write_bytes_per_launch": write_b,
        "fc_fwd_hbm_bytes_per_launch": read_b + write_b,
        "note": ("Infinity Cache (256 MiB) hits are counted by the "
                 "memory-side counters; per guide §L3 small working sets "
                 "under-report true HBM traffic"),
    }
    if args.algorithmic_bytes:
        out["algorithmic_bytes_per_launch"] = args.algorithmic_bytes
    pathlib.Path(args.out_json).write_text(json.dumps(out, indent=1))
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
