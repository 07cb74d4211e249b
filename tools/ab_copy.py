#!/usr/bin/env python3
"""Within-probe interleaved A/B of copy-kernel geometry variants
(cdna_hip_programming.md §5.4 rules 13/24: sub-5% deltas need interleaved
rounds in ONE process; report the distribution).

    python tools/ab_copy.py --rounds 12
"""
import argparse
import os
import statistics
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import dbeel_amd  # noqa: E402
from dbeel_amd.genruns import make_config  # noqa: E402

VARIANTS = {
    0: "256thr x 16KiB (default)",
    1: "256thr x 8KiB",
    2: "512thr x 16KiB",
    3: "512thr x 32KiB",
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=12)
    ap.add_argument("--scale", type=float, default=1.0)
    args = ap.parse_args()

    runs = make_config("cfg3", scale=args.scale)
    job = dbeel_amd.Job(runs, device=0)
    # warmup
    for v in VARIANTS:
        os.environ["DBEEL_COPY_VARIANT"] = str(v)
        job.run(False)

    times = {v: [] for v in VARIANTS}
    for r in range(args.rounds):
        for v in VARIANTS:
            os.environ["DBEEL_COPY_VARIANT"] = str(v)
            _, _, t = job.run(False)
            times[v].append(t["copy_ms"])
    for v, name in VARIANTS.items():
        ts = sorted(times[v])
        print(f"variant {v} ({name}): median={statistics.median(ts):.3f} ms "
              f"min={ts[0]:.3f} max={ts[-1]:.3f}")
    job.close()


if __name__ == "__main__":
    main()
