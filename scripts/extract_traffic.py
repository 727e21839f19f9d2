"""Summarize rocprofv3 PMC counter CSVs into per-launch HBM traffic for the
fd stencil kernels, applying the documented gfx950 correction
(MI355X_MICROARCH.md §HBM: FETCH_SIZE reports exactly half the bytes of a
wide coalesced streaming read — double it; WRITE_SIZE taken as reported and
cross-checked against the kernel's algorithmic write bytes).

Usage: python scripts/extract_traffic.py <counter_csv>... [--out traffic.json --dims 2048x2048x128 --n-gpus 1]
"""
import argparse
import csv
import json
import re
from collections import defaultdict


def parse(files):
    # rocprofv3 csv: one row per (dispatch, counter) with Counter_Name,
    # Counter_Value, Kernel_Name (column names vary slightly by version)
    per_kernel = defaultdict(lambda: defaultdict(list))
    for f in files:
        with open(f) as fh:
            rd = csv.DictReader(fh)
            cols = rd.fieldnames or []
            kcol = next((c for c in cols if "Kernel_Name" in c or c == "Name"), None)
            ccol = next((c for c in cols if "Counter_Name" in c), None)
            vcol = next((c for c in cols if "Counter_Value" in c), None)
            dcol = next((c for c in cols if "Dispatch" in c and "Id" in c), None)
            if not (kcol and ccol and vcol and dcol):
                continue  # agent_info / other sidecar CSVs
            rows = defaultdict(dict)
            for row in rd:
                rows[row[dcol]][row[ccol]] = float(row[vcol])
                rows[row[dcol]]["__k"] = row[kcol]
            for disp, vals in rows.items():
                k = vals.pop("__k")
                for c, v in vals.items():
                    per_kernel[k][c].append(v)
    return per_kernel


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("csvs", nargs="+")
    ap.add_argument("--out", default=None)
    ap.add_argument("--dims", default=None)
    ap.add_argument("--n-gpus", type=int, default=1)
    args = ap.parse_args()
    per_kernel = parse(args.csvs)
    summary = {}
    for k, counters in sorted(per_kernel.items()):
        short = re.sub(r"\s+", " ", k)[:90]
        entry = {}
        for c, vs in counters.items():
            entry[c] = {"mean": sum(vs) / len(vs), "n": len(vs)}
        # FETCH_SIZE/WRITE_SIZE are reported in KiB by rocprofv3
        fetch = entry.get("FETCH_SIZE", {}).get("mean")
        write = entry.get("WRITE_SIZE", {}).get("mean")
        if fetch is not None:
            entry["read_bytes_corrected"] = fetch * 1024 * 2  # gfx950 x2
        if write is not None:
            entry["write_bytes"] = write * 1024
        summary[short] = entry
        print(short)
        for c, v in entry.items():
            print(f"   {c}: {v}")
    if args.out and args.dims:
        # per-launch traffic of the matvec stencil (fd_kernel ...ILi4E...)
        tot = None
        for k, e in summary.items():
            if "fd_kernel" in k and ("ILi4ELi2E" in k or "double, 4, 2" in k):
                tot = e.get("read_bytes_corrected", 0) + e.get("write_bytes", 0)
        if tot:
            try:
                cur = json.load(open(args.out))
            except Exception:
                cur = {}
            cur[args.dims] = {"bytes_per_launch": tot, "n_gpus": args.n_gpus}
            json.dump(cur, open(args.out, "w"), indent=1)
            print(f"wrote {args.out}: {args.dims} -> {tot:.3e} B/launch")


if __name__ == "__main__":
    main()
