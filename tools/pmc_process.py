#!/usr/bin/env python3
"""Post-process rocprofv3 CSV evidence from gpurun_out into profiles/.

1. PMC traffic (r02_cfg3_fetch.csv / r02_cfg3_write.csv -> traffic json):
   per-kernel per-launch HBM bytes. FETCH_SIZE/WRITE_SIZE report
   kilobytes; gfx950 FETCH_SIZE reads half the bytes of wide coalesced
   loads -> x2 correction calibrated against k_copy's known read volume
   (MI355X_MICROARCH.md, r01 calibration).
2. Ingest overlap (ing_memory_copy_trace.csv + ing_kernel_trace.csv):
   how much k_prepare time lies inside H2D copy intervals.
"""
import csv
import json
import re
import sys


def kshort(name: str) -> str:
    m = re.match(r"(?:void )?(k_\w+|__amd\w+)", name)
    if m:
        base = m.group(1)
        if base.startswith("k_"):
            return re.sub(r"prepare|corank|rankreduce", lambda x: x.group(0),
                          base)
        return base
    if "DeviceScan" in name or "device_scan" in name.lower() or \
            "lookback" in name.lower():
        return "rocprim_scan_u64" if "unsigned long" in name else \
            "rocprim_scan_u32"
    return name[:40]


def load_counters(path):
    per = {}
    with open(path) as f:
        for row in csv.DictReader(f):
            name = kshort(row["Kernel_Name"])
            if name.startswith("__amd"):
                continue
            per.setdefault(name, []).append(float(row["Counter_Value"]))
    return per


def traffic(fetch_csv, write_csv, out_json, n_steps=4):
    f = load_counters(fetch_csv)
    w = load_counters(write_csv)
    kernels = {}
    for name in sorted(set(f) | set(w)):
        fv = f.get(name, [])
        wv = w.get(name, [])
        fetch_raw = int(sum(fv) * 1024)
        write_b = int(sum(wv) * 1024)
        # per-step totals: the bench ran n_steps pipeline passes (warmup
        # included); k_maxks runs ONCE at job create, not per step
        div = 1 if name == "k_maxks" else n_steps
        kernels[name] = {
            "launches_sampled": max(len(fv), len(wv)),
            "fetch_bytes_raw": fetch_raw // div,
            "fetch_bytes_corrected": fetch_raw * 2 // div,
            "write_bytes": write_b // div,
            "traffic_bytes": (fetch_raw * 2 + write_b) // div,
        }
    doc = {
        "_doc": ("Per-STEP HBM traffic per kernel on cfg3 (8x1GiB runs, "
                 "seed 0xDBEE1), rocprofv3 --pmc FETCH_SIZE / WRITE_SIZE in "
                 "separate passes (TCC slot limit); totals over the run "
                 "divided by its pipeline passes, so multi-launch stages "
                 "(rocPRIM scans) are fully counted. k_maxks is a "
                 "once-per-job-create cost reported per CREATE. "
                 "fetch_bytes_corrected = 2x raw per the gfx950 "
                 "calibration (MI355X_MICROARCH.md; calibrated against "
                 "k_copy's algorithmic read volume in r01)."),
        "workload": "cfg3_8run_x_1GiB",
        "steps": n_steps,
        "kernels": kernels,
        "pipeline_traffic_bytes": sum(
            k["traffic_bytes"] for n, k in kernels.items()
            if n != "k_maxks" and n.startswith(("k_", "rocprim"))),
    }
    json.dump(doc, open(out_json, "w"), indent=1)
    print(f"{out_json}: pipeline traffic/step "
          f"{doc['pipeline_traffic_bytes']/1e9:.2f} GB")


def overlap(copy_csv, kernel_csv, out_json):
    copies = []
    with open(copy_csv) as f:
        for row in csv.DictReader(f):
            if row["Direction"] == "MEMORY_COPY_HOST_TO_DEVICE":
                copies.append((int(row["Start_Timestamp"]),
                               int(row["End_Timestamp"])))
    preps = []
    with open(kernel_csv) as f:
        for row in csv.DictReader(f):
            if "k_prepare" in row["Kernel_Name"]:
                preps.append((int(row["Start_Timestamp"]),
                              int(row["End_Timestamp"])))
    copies.sort()
    # merge copy intervals
    merged = []
    for s, e in copies:
        if merged and s <= merged[-1][1]:
            merged[-1][1] = max(merged[-1][1], e)
        else:
            merged.append([s, e])
    total_prep = sum(e - s for s, e in preps)
    inside = 0
    for s, e in preps:
        for ms, me in merged:
            lo, hi = max(s, ms), min(e, me)
            if hi > lo:
                inside += hi - lo
    total_copy = sum(e - s for s, e in merged)
    doc = {
        "_doc": ("Streamed-ingest overlap evidence (tools/ingest_probe.py "
                 "under rocprofv3 --sys-trace): fraction of k_prepare "
                 "execution time lying INSIDE H2D copy intervals. ~1.0 "
                 "means the prepare stage is fully hidden behind the "
                 "PCIe transfer."),
        "h2d_copies": len(copies),
        "h2d_busy_ms": round(total_copy / 1e6, 3),
        "k_prepare_launches": len(preps),
        "k_prepare_busy_ms": round(total_prep / 1e6, 3),
        "k_prepare_overlapped_ms": round(inside / 1e6, 3),
        "overlap_fraction": round(inside / total_prep, 4) if total_prep
        else None,
    }
    json.dump(doc, open(out_json, "w"), indent=1)
    print(f"{out_json}: overlap {doc['overlap_fraction']}")


def stalls(counter_csv, out_json, workload):
    """Wave-state breakdown per kernel from one SQ-counter pass:
    SQ_WAVE_CYCLES = parked (SQ_WAIT_ANY: waitcnt/barrier) + issue-stall
    (SQ_WAIT_INST_ANY) + issuing (SQ_ACTIVE_INST_ANY), disjoint
    (MI355X_MICROARCH.md SQ table). parked_frac near 1 means waves sit on
    memory; issue/active-dominated means the kernel is issue-bound."""
    per = {}
    with open(counter_csv) as f:
        for row in csv.DictReader(f):
            name = kshort(row["Kernel_Name"])
            if name.startswith("__amd"):
                continue
            per.setdefault(name, {}).setdefault(
                row["Counter_Name"], []).append(float(row["Counter_Value"]))
    doc = {"_doc": ("Per-kernel wave-state breakdown (one rocprofv3 --pmc "
                    "pass: SQ_WAVE_CYCLES / SQ_WAIT_ANY / SQ_WAIT_INST_ANY "
                    "/ SQ_ACTIVE_INST_ANY, quad-cycle units, averaged over "
                    "launches). parked = waiting on waitcnt/barrier "
                    "(memory); issue_stall = instruction-issue stalls; "
                    "active = issuing."),
           "workload": workload, "kernels": {}}
    for name, c in sorted(per.items()):
        def avg(k):
            v = c.get(k, [])
            return sum(v) / len(v) if v else 0.0
        wave = avg("SQ_WAVE_CYCLES")
        if not wave:
            continue
        doc["kernels"][name] = {
            "wave_cycles": int(wave),
            "parked_frac": round(avg("SQ_WAIT_ANY") / wave, 4),
            "issue_stall_frac": round(avg("SQ_WAIT_INST_ANY") / wave, 4),
            "active_frac": round(avg("SQ_ACTIVE_INST_ANY") / wave, 4),
        }
    json.dump(doc, open(out_json, "w"), indent=1)
    kc = doc["kernels"].get("k_copy", {})
    print(f"{out_json}: k_copy parked={kc.get('parked_frac')} "
          f"issue={kc.get('issue_stall_frac')} "
          f"active={kc.get('active_frac')}")


if __name__ == "__main__":
    base = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out"
    import os

    if os.path.exists(f"{base}/r02b_cfg3_fetch.csv"):
        traffic(f"{base}/r02b_cfg3_fetch.csv", f"{base}/r02b_cfg3_write.csv",
                "profiles/r02_traffic_cfg3.json")
    if os.path.exists(f"{base}/ing_memory_copy_trace.csv"):
        overlap(f"{base}/ing_memory_copy_trace.csv",
                f"{base}/ing_kernel_trace.csv",
                "profiles/r02_ingest_overlap.json")
    for w in ("cfg3", "cfg5"):
        p = f"{base}/r02_{w}_stalls.csv"
        if os.path.exists(p):
            stalls(p, f"profiles/r02_{w}_stalls.json", w)
