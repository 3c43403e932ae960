#!/usr/bin/env python3
"""Build profiles/traffic_calibration.json from a rocprofv3 --pmc run.

Usage: python scripts/make_traffic_cal.py <pmc_counter_csv> <workload> \
           <kernel_regex> <nprobe> <lut> <source_note>

Reads the per-dispatch FETCH_SIZE / WRITE_SIZE rows of the scan kernel,
applies the gfx950 wide-read correction (FETCH_SIZE reports 1/2 of wide
coalesced reads — MI355X_MICROARCH.md §HBM; calibrated in
profiles/r01/README.md), and records mean counter bytes per launch. The
bench reports this as roofline.traffic when its operating point matches.
"""
import csv
import json
import os
import re
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    pmc_csv, workload, kregex, nprobe, lut, source = sys.argv[1:7]
    rows = []
    with open(pmc_csv) as f:
        for row in csv.DictReader(f):
            name = row.get("Kernel_Name", row.get("kernel", ""))
            if re.search(kregex, name):
                rows.append(row)
    # keep only the FULL-BATCH dispatches (largest grid): the bench also
    # launches small recall-sweep searches that would skew the mean
    gmax = max(int(r["Grid_Size"]) for r in rows)
    fetch, write = [], []
    for row in rows:
        if int(row["Grid_Size"]) != gmax:
            continue
        cname = row.get("Counter_Name", row.get("counter", ""))
        val = float(row.get("Counter_Value", row.get("value", 0)))
        if cname == "FETCH_SIZE":
            fetch.append(val)
        elif cname == "WRITE_SIZE":
            write.append(val)
    if not fetch:
        sys.exit(f"no FETCH_SIZE rows matched {kregex!r} in {pmc_csv}")
    # counters are reported in KB per dispatch; gfx950 FETCH_SIZE
    # undercounts wide coalesced reads by 2x (calibrated r01)
    fetch_b = 2.0 * 1024.0 * sum(fetch) / len(fetch)
    write_b = 1024.0 * (sum(write) / len(write) if write else 0.0)
    out_path = os.path.join(REPO, "profiles", "traffic_calibration.json")
    cal = {}
    if os.path.exists(out_path):
        cal = json.load(open(out_path))
    cal[workload] = {
        "kernel_regex": kregex,
        "nprobe": int(nprobe),
        "lut": lut,
        "bytes_per_launch": fetch_b + write_b,
        "fetch_bytes_x2_corrected": fetch_b,
        "write_bytes": write_b,
        "dispatches": len(fetch),
        "correction": "FETCH_SIZE x2 (gfx950 wide-read undercount, "
                      "MI355X_MICROARCH.md §HBM; profiles/r01/README.md)",
        "source": source,
    }
    json.dump(cal, open(out_path, "w"), indent=1)
    print(f"{workload}: {len(fetch)} dispatches, "
          f"fetch(x2) {fetch_b/1e6:.1f} MB + write {write_b/1e6:.1f} MB "
          f"per launch -> {out_path}")


if __name__ == "__main__":
    main()
