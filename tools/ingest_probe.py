#!/usr/bin/env python3
"""Streamed-ingest overlap probe (run under rocprofv3 --sys-trace to
capture the copy-engine / compute concurrency evidence for profiles/).

Creates a cfg3-shaped job at reduced scale, pins the host buffers, then
ingests + runs a few times printing the ingest stats. The sys-trace's
memory-copy rows and kernel rows overlapping in time are the evidence
that k_prepare executes while chunks stream."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import dbeel_amd  # noqa: E402
from dbeel_amd.engine import Job, pin_host, unpin_host  # noqa: E402
from dbeel_amd.genruns import CONFIGS, make_runs  # noqa: E402


def main():
    scale = float(sys.argv[1]) if len(sys.argv) > 1 else 0.25
    cfg = dict(CONFIGS["cfg3"])
    cfg["entries_per_run"] = int(cfg["entries_per_run"] * scale)
    runs = make_runs(seed=0xDBEE1, **cfg)
    bufs = [a for d, i in runs for a in (d, i)]
    for a in bufs:
        pin_host(a)
    try:
        with Job(runs, device=0) as job:
            for step in range(3):
                st = job.ingest(runs)
                out_bytes, out_entries, t = job.run(False)
                print(json.dumps({
                    "step": step,
                    "ingest": {k: round(v, 3) if isinstance(v, float) else v
                               for k, v in st.items()},
                    "run_prep_ms": round(t["prep_ms"], 3),
                    "run_kernel_ms": round(t["kernel_ms"], 3),
                    "out_bytes": out_bytes,
                }))
    finally:
        for a in bufs:
            unpin_host(a)


if __name__ == "__main__":
    main()
