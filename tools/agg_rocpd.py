#!/usr/bin/env python3
"""Aggregate rocprofv3 rocpd SQLite result databases (ROCm 7.2 default output)
under a directory tree into one JSON summary: per (pass, kernel) dispatch
counts, total/avg durations, VGPR/scratch sizes, and summed PMC counters."""
import glob
import json
import os
import sqlite3
import sys


def short(name):
    return name.split("(")[0].replace(".kd", "").strip()


def read_db(path):
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    meta = [t for t in tabs if t.startswith("rocpd_metadata_")]
    if not meta:
        return {}
    uuid = meta[0][len("rocpd_metadata_"):]

    def T(t):
        return f"{t}_{uuid}"

    kernels = {}
    syms = {r[0]: (short(r[1]), r[2], r[3]) for r in db.execute(
        f"SELECT id, display_name, arch_vgpr_count, private_segment_size "
        f"FROM {T('rocpd_info_kernel_symbol')}")}
    disp = {}
    for did, kid, start, end, ev in db.execute(
            f"SELECT id, kernel_id, start, end, event_id "
            f"FROM {T('rocpd_kernel_dispatch')}"):
        name, vgpr, scratch = syms.get(kid, ("?", None, None))
        k = kernels.setdefault(name, {"dispatches": 0, "total_ms": 0.0,
                                      "vgpr": vgpr, "scratch_per_lane": scratch,
                                      "counters": {}})
        k["dispatches"] += 1
        k["total_ms"] += (end - start) / 1e6
        disp[ev] = name
    pmc_names = {r[0]: r[1] for r in db.execute(
        f"SELECT id, name FROM {T('rocpd_info_pmc')}")}
    for ev, pmc_id, value in db.execute(
            f"SELECT event_id, pmc_id, value FROM {T('rocpd_pmc_event')}"):
        name = disp.get(ev)
        if name is None:
            continue
        c = kernels[name]["counters"].setdefault(
            pmc_names.get(pmc_id, str(pmc_id)), {"sum": 0.0, "n": 0})
        c["sum"] += value
        c["n"] += 1
    for k in kernels.values():
        if k["dispatches"]:
            k["avg_ms"] = round(k["total_ms"] / k["dispatches"], 4)
        k["total_ms"] = round(k["total_ms"], 3)
        for c in k["counters"].values():
            c["per_dispatch"] = c["sum"] / max(1, k["dispatches"])
    return kernels


def main(root):
    out = {}
    for path in glob.glob(os.path.join(root, "**", "*_results.db"),
                          recursive=True):
        pass_name = os.path.relpath(path, root).split(os.sep)[0]
        kernels = read_db(path)
        if pass_name in out:  # multiple procs in one pass: merge
            for name, k in kernels.items():
                out[pass_name][name] = k
        else:
            out[pass_name] = kernels
    print(json.dumps(out, indent=1, sort_keys=True))


if __name__ == "__main__":
    main(sys.argv[1])
