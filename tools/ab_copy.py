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
    4: "512thr x 64KiB",
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=12)
    ap.add_argument("--scale", type=float, default=1.0)
    ap.add_argument("--workload", default="cfg3",
                    choices=["cfg2", "cfg3", "cfg5"])
    ap.add_argument("--grids", default="",
                    help="comma list of DBEEL_COPY_GRID caps to sweep "
                         "(per variant); empty = default 4096")
    args = ap.parse_args()

    if args.workload == "cfg2":
        runs = make_config("cfg2", scale=args.scale)
    elif args.workload == "cfg5":
        from dbeel_amd.genruns import CONFIGS, make_runs_varkey

        cfg = dict(CONFIGS["cfg5"])
        if args.scale != 1.0:
            cfg["entries_per_run"] = int(cfg["entries_per_run"] * args.scale)
        runs = make_runs_varkey(seed=0xDBEE1, **cfg)
    else:
        runs = make_config("cfg3", scale=args.scale)
    job = dbeel_amd.Job(runs, device=0)
    # warmup
    for v in VARIANTS:
        os.environ["DBEEL_COPY_VARIANT"] = str(v)
        job.run(False)

    grids = [int(g) for g in args.grids.split(",") if g] or [None]
    combos = [(v, g) for v in VARIANTS for g in grids]
    times = {c: [] for c in combos}
    for r in range(args.rounds):
        for v, g in combos:
            os.environ["DBEEL_COPY_VARIANT"] = str(v)
            if g is None:
                os.environ.pop("DBEEL_COPY_GRID", None)
            else:
                os.environ["DBEEL_COPY_GRID"] = str(g)
            _, _, t = job.run(False)
            times[(v, g)].append(t["copy_ms"])
    for (v, g), ts in times.items():
        ts = sorted(ts)
        gs = f" grid={g}" if g else ""
        print(f"variant {v} ({VARIANTS[v]}){gs}: "
              f"median={statistics.median(ts):.3f} ms "
              f"min={ts[0]:.3f} max={ts[-1]:.3f}")
    job.close()


if __name__ == "__main__":
    main()
