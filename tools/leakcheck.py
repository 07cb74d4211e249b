#!/usr/bin/env python3
"""Device-memory stability check: churn job create/run/fetch/destroy,
streamed ingests (each creates ~per-chunk HIP events), batched jobs,
scans and sliced compactions, reporting hipMemGetInfo drift. A leak in
any path shows as monotonically shrinking free memory.

    python tools/leakcheck.py --iters 40
"""
import argparse
import ctypes
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import dbeel_amd  # noqa: E402
from dbeel_amd.engine import (  # noqa: E402
    BatchJob,
    Job,
    compact_sliced,
    load,
    pin_host,
    scan,
    unpin_host,
)
from dbeel_amd.genruns import make_runs  # noqa: E402


def free_mem():
    lib = ctypes.CDLL("libamdhip64.so")
    free = ctypes.c_size_t()
    total = ctypes.c_size_t()
    lib.hipMemGetInfo(ctypes.byref(free), ctypes.byref(total))
    return int(free.value)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=40)
    args = ap.parse_args()
    load()

    runs = make_runs(4, 20_000, 16, 256, overlap_frac=0.4,
                     tombstone_frac=0.1, seed=77)
    runs2 = make_runs(4, 20_000, 16, 256, overlap_frac=0.4,
                      tombstone_frac=0.1, seed=78)
    # equal shapes for ingest (tombstone draws differ -> regenerate
    # without tombstones for the ingest pair)
    ri_a = make_runs(3, 15_000, 16, 128, overlap_frac=0.3, seed=80)
    ri_b = make_runs(3, 15_000, 16, 128, overlap_frac=0.3, seed=81)
    for d, i in ri_b:
        pin_host(d)
        pin_host(i)

    os.environ["DBEEL_STREAM_CHUNK_MB"] = "1"
    base = None
    total_in = sum(d.nbytes + i.nbytes for d, i in runs)
    try:
        for it in range(args.iters):
            # plain compact (job create/run/fetch/destroy)
            dbeel_amd.compact(runs, keep_tombstones=False, device=0)
            # resident job + ingest churn
            with Job(ri_a, device=0) as job:
                job.ingest(ri_b)
                job.run(False)
                job.fetch()
            # batched
            with BatchJob([runs, runs2], device=0) as bj:
                bj.run(True)
                bj.fetch_job(1)
            # scan + sliced
            scan(runs, hash_ranges=[(0, 1 << 31)], device=0)
            compact_sliced(runs, keep_tombstones=True, device=0,
                           max_resident_bytes=total_in // 3)
            f = free_mem()
            if base is None:
                base = f
            drift = base - f
            if it % 5 == 4 or it == args.iters - 1:
                print(f"iter {it + 1}/{args.iters}: free={f / 1e9:.3f} GB "
                      f"drift={drift / 1e6:.1f} MB")
            if drift > 512 * 1024 * 1024:
                print("LEAKCHECK FAILED: device memory drift "
                      f"{drift / 1e6:.1f} MB after {it + 1} iters")
                sys.exit(1)
    finally:
        for d, i in ri_b:
            unpin_host(d)
            unpin_host(i)
        os.environ.pop("DBEEL_STREAM_CHUNK_MB", None)
    print(f"LEAKCHECK OK: {args.iters} iterations across every API path, "
          f"drift {(base - free_mem()) / 1e6:.1f} MB")


if __name__ == "__main__":
    main()
