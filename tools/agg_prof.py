#!/usr/bin/env python3
"""Aggregate rocprofv3 output CSVs (kernel traces + counter collections) under
a directory tree into one JSON summary: per (pass, kernel) dispatch counts,
total/avg durations, and summed PMC counter values."""
import csv
import glob
import json
import os
import sys


def short(name):
    return name.split("(")[0].replace(".kd", "").strip()


def main(root):
    passes = {}
    for path in glob.glob(os.path.join(root, "**", "*.csv"), recursive=True):
        rel = os.path.relpath(path, root)
        pass_name = rel.split(os.sep)[0]
        p = passes.setdefault(pass_name, {"kernels": {}, "files": []})
        p["files"].append(rel)
        with open(path, newline="") as f:
            try:
                rows = list(csv.DictReader(f))
            except Exception:
                continue
        if not rows:
            continue
        cols = {c.lower().replace('"', ""): c for c in rows[0].keys()}

        def col(*names):
            for n in names:
                if n in cols:
                    return cols[n]
            return None

        kname = col("kernel_name", "name", "kernelname")
        cname = col("counter_name", "counter")
        cval = col("counter_value", "value")
        t0 = col("start_timestamp", "begin_ns", "start")
        t1 = col("end_timestamp", "end_ns", "end")
        for r in rows:
            if not kname or not r.get(kname):
                continue
            k = p["kernels"].setdefault(short(r[kname]),
                                        {"dispatches": 0, "total_ms": 0.0,
                                         "counters": {}})
            if cname and cval and r.get(cname):
                c = k["counters"].setdefault(r[cname],
                                             {"sum": 0.0, "dispatches": 0})
                try:
                    c["sum"] += float(r[cval])
                    c["dispatches"] += 1
                except ValueError:
                    pass
            elif t0 and t1:
                try:
                    k["total_ms"] += (int(r[t1]) - int(r[t0])) / 1e6
                    k["dispatches"] += 1
                except ValueError:
                    pass
    # derive averages
    for p in passes.values():
        for k in p["kernels"].values():
            if k["dispatches"]:
                k["avg_ms"] = round(k["total_ms"] / k["dispatches"], 4)
            k["total_ms"] = round(k["total_ms"], 3)
            for c in k["counters"].values():
                if c["dispatches"]:
                    c["per_dispatch"] = c["sum"] / c["dispatches"]
        p["files"] = sorted(p["files"])[:8]
    print(json.dumps(passes, indent=1, sort_keys=True))


if __name__ == "__main__":
    main(sys.argv[1])
