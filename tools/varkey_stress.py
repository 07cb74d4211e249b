#!/usr/bin/env python3
"""Randomized variable-length-key stress: compact, scan and sliced
compaction on cfg5-style msgpack-key runs, each trial checked bit-exact
against the oracle / AsyncIter model.

    python tools/varkey_stress.py --trials 8 --seed 17
"""
import argparse
import os
import sys

import numpy as np

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)
sys.path.insert(0, os.path.join(_ROOT, "tests"))

import dbeel_amd  # noqa: E402
import oracle  # noqa: E402
from dbeel_amd.engine import compact_sliced, scan  # noqa: E402
from dbeel_amd.genruns import make_runs_varkey  # noqa: E402
from pymm3 import scan_model  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=8)
    ap.add_argument("--seed", type=int, default=17)
    args = ap.parse_args()
    rng = np.random.default_rng(args.seed)
    for t in range(args.trials):
        runs = make_runs_varkey(
            int(rng.integers(2, 8)), int(rng.integers(1000, 6000)),
            value_size=int(rng.integers(64, 1024)),
            overlap_frac=float(rng.random() * 0.8),
            tombstone_frac=float(rng.random() * 0.3),
            seed=int(rng.integers(1, 1 << 30)),
        )
        keep = bool(rng.integers(0, 2))
        od, oi, on = oracle.compact(runs, keep_tombstones=keep)
        gd, gi, gn = dbeel_amd.compact(runs, keep_tombstones=keep, device=0)
        assert (gn, gi, gd) == (on, oi, od), f"varkey compact trial {t}"
        kw = dict(hash_ranges=[(int(rng.integers(0, 2**32)),
                                int(rng.integers(0, 2**32)))])
        sd, si, sn = scan(runs, device=0, **kw)
        md, mi, mn = scan_model(runs, **kw)
        assert (sn, si, sd) == (mn, mi, md), f"varkey scan trial {t}"
        total = sum(d.nbytes + i.nbytes for d, i in runs)
        xd, xi, xn = compact_sliced(runs, keep_tombstones=keep, device=0,
                                    max_resident_bytes=max(total // 3,
                                                           4096))
        assert (xn, xi, xd) == (on, oi, od), f"varkey sliced trial {t}"
        print(f"varkey trial {t} OK")
    print(f"VARKEY STRESS OK: {args.trials} trials "
          "(compact + scan + sliced, bit-exact)")


if __name__ == "__main__":
    main()
